"""Profile bench-style training steps with python stacks attached to
each GPU kernel, to attribute residual at::native glue (fills /
elementwise) in the hot path.  GPU-only; run under gpurun.

Usage: python tools/torch_prof_step.py [alexnet] [steps]
"""
import sys

import torch

sys.path.insert(0, ".")
from bench import MODELS  # noqa: E402
import importlib  # noqa: E402


def main():
    name = sys.argv[1] if len(sys.argv) > 1 else "alexnet"
    steps = int(sys.argv[2]) if len(sys.argv) > 2 else 6
    modname, clsname, batch = MODELS[name]
    cls = getattr(importlib.import_module(modname), clsname)
    model = cls({"rank": 0, "size": 1, "device": "cuda:0",
                 "verbose": False, "batch_size": batch, "n_iters": 16})
    model.compile_iter_fns("grad")
    for i in range(4):
        model.train_iter(i % model.data.n_batch_train)
        model.apply_update()
    torch.cuda.synchronize()
    with torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CUDA,
                        torch.profiler.ProfilerActivity.CPU],
            with_stack=True) as prof:
        for i in range(steps):
            model.train_iter((4 + i) % model.data.n_batch_train)
            model.apply_update()
        torch.cuda.synchronize()
    print(prof.key_averages(group_by_stack_n=6).table(
        sort_by="cuda_time_total", row_limit=40, max_src_column_width=120))


if __name__ == "__main__":
    main()
