"""Per-shape kernel variant table (measured, not guessed).

``tools/autotune_conv.py`` sweeps the gather-GEMM / wgrad variants (see
the table in ``ops/hip/conv.hip``) over the zoo's real layer shapes on
an MI355X and writes the winners to ``conv_tuned.json`` next to this
file.  The layers look their shape up here on first use; a shape with
no entry falls back to variant 0 (the launcher's built-in heuristic,
round-1 behavior).

Key format: ``"<op>:<M>x<BN>x<Kdf>g<G>"`` where op is fwd/dgrad/wgrad,
M = GEMM rows (N*P*Q), BN = output channels per group (fwd/dgrad) or
Kgp (wgrad), Kdf = flat reduction length.  Values: int variant, or for
wgrad ``[variant, split_target]``.

Env: ``THEANOMPI_TUNED_JSON=<path>`` overlays/replaces the shipped
table (used by the autotuner itself); ``THEANOMPI_TUNED_OFF=1``
disables lookups entirely (A/B baseline).
"""

import json
import os

_TABLE = None


def _load():
    global _TABLE
    if _TABLE is None:
        t = {}
        if os.environ.get("THEANOMPI_TUNED_OFF", "0") in ("0", ""):
            p = os.path.join(os.path.dirname(__file__), "conv_tuned.json")
            try:
                with open(p) as f:
                    t = json.load(f)
            except (OSError, ValueError):
                t = {}
            env = os.environ.get("THEANOMPI_TUNED_JSON")
            if env:
                try:
                    with open(env) as f:
                        t.update(json.load(f))
                except (OSError, ValueError):
                    pass
        _TABLE = t
    return _TABLE


def key(op, m, bn, kdf, g):
    return f"{op}:{m}x{bn}x{kdf}g{g}"


def variant_for(op, m, bn, kdf, g):
    """Returns the tuned variant int (0 = launcher heuristic)."""
    v = _load().get(key(op, m, bn, kdf, g), 0)
    if isinstance(v, (list, tuple)):
        v = v[0]
    return int(v)


def wgrad_for(m, kgp, kdf, g):
    """Returns (variant, split_target) for a wgrad shape."""
    v = _load().get(key("wgrad", m, kgp, kdf, g), 0)
    if isinstance(v, (list, tuple)):
        return int(v[0]), int(v[1]) if len(v) > 1 else 0
    return int(v), 0


def reset():
    """Drop the cached table (tests / autotuner reload)."""
    global _TABLE
    _TABLE = None
