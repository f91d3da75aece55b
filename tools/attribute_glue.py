"""Attribute residual torch-glue ops (fills, adds, muls, copies) in a
training step to their Python call sites via TorchDispatchMode.

The rocprof kernel stats show WHAT at::native glue runs
(profiles/*_final_r2_kernel_stats.txt); this shows WHERE it comes
from, so the round-3 glue work targets real sources.  GPU or CPU.

Usage: python tools/attribute_glue.py [googlenet] [steps]
"""
import collections
import importlib
import sys
import traceback

import torch
from torch.utils._python_dispatch import TorchDispatchMode

sys.path.insert(0, ".")
from bench import MODELS  # noqa: E402

GLUE = ("aten.zero_", "aten.fill_", "aten.zeros", "aten.add.Tensor",
        "aten.add_.Tensor", "aten.mul.Tensor", "aten.mul_.Tensor",
        "aten.copy_", "aten.cat")


class _Attr(TorchDispatchMode):
    def __init__(self):
        super().__init__()
        self.hits = collections.Counter()
        self.bytes = collections.Counter()
        self.anon = []

    def __torch_dispatch__(self, func, types, args=(), kwargs=None):
        name = str(func)
        if any(name.startswith(g) for g in GLUE):
            # innermost repo frame = the responsible call site
            site = "?"
            for fr in reversed(traceback.extract_stack()):
                if ("theanompi_amd" in fr.filename
                        or fr.filename.endswith("bench.py")):
                    site = f"{fr.filename.split('/')[-1]}:{fr.lineno}"
                    break
            key = (name.split(".Tensor")[0], site)
            self.hits[key] += 1
            t = args[0] if args and isinstance(args[0], torch.Tensor) \
                else None
            if t is not None:
                self.bytes[key] += t.numel() * t.element_size()
            if site == "?" and t is not None:
                src = args[1] if (len(args) > 1
                                  and isinstance(args[1], torch.Tensor)) \
                    else None
                self.anon.append(
                    (name.split(".Tensor")[0], tuple(t.shape),
                     tuple(t.stride()), str(t.dtype).replace(
                         "torch.", ""),
                     tuple(src.stride()) if src is not None else None))
        return func(*args, **(kwargs or {}))


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "googlenet"
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 3
    modname, clsname, batch = MODELS[name]
    cls = getattr(importlib.import_module(modname), clsname)
    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    model = cls({"rank": 0, "size": 1, "device": dev,
                 "verbose": False, "batch_size": batch, "n_iters": 16})
    model.compile_iter_fns("grad")
    for i in range(3):                      # warm caches/buffers
        model.train_iter(i % model.data.n_batch_train)
        model.apply_update()
    mode = _Attr()
    with mode:
        for i in range(steps):
            model.train_iter((3 + i) % model.data.n_batch_train)
            model.apply_update()
    if dev != "cpu":
        torch.cuda.synchronize()
    print(f"== glue ops over {steps} steps of {name} (per-step counts)")
    for (op, site), n in sorted(mode.hits.items(),
                                key=lambda kv: -kv[1])[:30]:
        mb = mode.bytes[(op, site)] / steps / 1e6
        print(f"{n / steps:7.1f}/step  {mb:9.2f} MB/step  {op:14s} {site}")
    if mode.anon:
        print("== engine-internal ops (no repo frame), dst shape/stride:")
        an = collections.Counter(mode.anon)
        for sig, n in sorted(an.items(), key=lambda kv: -kv[1])[:20]:
            print(f"{n / steps:7.1f}/step  {sig}")


if __name__ == "__main__":
    main()
