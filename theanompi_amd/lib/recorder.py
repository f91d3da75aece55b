"""Recorder — wall-clock calc/comm/wait segmentation + training history.

Rebuild of the reference's metrics system (Theano-MPI
``theanompi/lib/recorder.py`` [pub]): explicit
``start()`` / ``end(tag)`` calls segment every iteration into calc / comm /
wait buckets; train cost/error and val top-1/top-5 accumulate; the whole
history pickles to ``inforec.pkl`` and is half of the resume state
(SURVEY.md §5.4).  Additions over the reference: a JSONL emitter for the
benchmark harness and optional roctx-style stream markers are left to the
worker; the Recorder itself stays a plain host-side timer so its output
matches the reference's scaling tables.
"""

import json
import pickle
import time


class Recorder:

    TAGS = ("calc", "comm", "wait", "load")

    def __init__(self, config=None):
        config = config or {}
        self.rank = int(config.get("rank", 0))
        self.size = int(config.get("size", 1))
        self.verbose = bool(config.get("verbose", self.rank == 0))
        self.print_freq = int(config.get("print_freq", 40))
        self._t0 = None
        self._sync = config.get("sync_fn")  # e.g. torch.cuda.synchronize
        # per-iteration buckets (reset after each print)
        self._acc = {t: 0.0 for t in self.TAGS}
        # history
        self.epoch = 0
        self.uidx = 0
        self.train_info = []       # (epoch, uidx, cost, error, calc, comm, wait)
        self.val_info = []         # (epoch, uidx, cost, top1, top5)
        self.epoch_time = []
        self._epoch_t0 = None
        self._costs = []
        self._errors = []

    # -- timing ---------------------------------------------------------------

    def start(self):
        if self._sync:
            self._sync()
        self._t0 = time.perf_counter()

    def end(self, tag):
        assert tag in self.TAGS, tag
        if self._sync:
            self._sync()
        now = time.perf_counter()
        if self._t0 is not None:
            self._acc[tag] += now - self._t0
        self._t0 = now

    def start_epoch(self):
        self._epoch_t0 = time.perf_counter()

    def end_epoch(self):
        if self._epoch_t0 is not None:
            self.epoch_time.append(time.perf_counter() - self._epoch_t0)
            self.epoch += 1

    # -- accumulation ---------------------------------------------------------

    def train_error(self, uidx, cost, error):
        # cost/error may be device tensors; they are only converted to
        # floats at print time so the hot loop never syncs the GPU.
        self.uidx = uidx
        self._costs.append(cost)
        self._errors.append(error)

    def val_error(self, uidx, cost, top1, top5):
        self.val_info.append((self.epoch, uidx, float(cost),
                              float(top1), float(top5)))
        if self.verbose:
            print(f"[rank {self.rank}] epoch {self.epoch} val: "
                  f"cost {cost:.4f} top1 {top1:.4f} top5 {top5:.4f}",
                  flush=True)

    # -- reporting ------------------------------------------------------------

    def print_train_info(self, uidx, force=False):
        if uidx % self.print_freq != 0 and not force:
            return
        n = max(len(self._costs), 1)
        cost = float(sum(float(c) for c in self._costs)) / n
        err = float(sum(float(e) for e in self._errors)) / n
        c, m, w = (self._acc["calc"], self._acc["comm"], self._acc["wait"])
        self.train_info.append((self.epoch, uidx, cost, err, c, m, w))
        if self.verbose:
            print(f"[rank {self.rank}] e{self.epoch} it{uidx} "
                  f"cost {cost:.4f} err {err:.4f} | "
                  f"calc {c:.3f}s comm {m:.3f}s wait {w:.3f}s",
                  flush=True)
        self._costs, self._errors = [], []
        for t in self.TAGS:
            self._acc[t] = 0.0

    def get_latest_durations(self):
        return dict(self._acc)

    # -- persistence (reference: pickled inforec.pkl) ---------

    def save(self, path="inforec.pkl"):
        state = {
            "epoch": self.epoch,
            "uidx": self.uidx,
            "train_info": self.train_info,
            "val_info": self.val_info,
            "epoch_time": self.epoch_time,
        }
        with open(path, "wb") as f:
            pickle.dump(state, f)

    def load(self, path="inforec.pkl"):
        with open(path, "rb") as f:
            state = pickle.load(f)
        self.epoch = state["epoch"]
        self.uidx = state["uidx"]
        self.train_info = state["train_info"]
        self.val_info = state["val_info"]
        self.epoch_time = state["epoch_time"]
        return self

    def dump_jsonl(self, path):
        """Machine-readable addition for the scaling-curve harness."""
        with open(path, "a") as f:
            for row in self.train_info:
                f.write(json.dumps({
                    "epoch": row[0], "iter": row[1], "cost": row[2],
                    "error": row[3], "calc_s": row[4], "comm_s": row[5],
                    "wait_s": row[6], "rank": self.rank}) + "\n")
