"""BSP worker — rebuild of ``theanompi/bsp_worker.py`` [pub]
(SURVEY.md §2.2, §3.2).  One process per GPU; per iteration:
train_iter (fwd/bwd + cdd phase 1) -> exchanger.exchange (RCCL over
xGMI) -> apply_update (cdd phase 2) -> periodic recorder print; per
epoch: sharded validation with all-reduced stats, adjust_hyperp,
checkpoint from rank 0.
"""

import os

import torch
import torch.distributed as dist

from theanompi_amd.lib.heartbeat import maybe_start
from theanompi_amd.lib.base import GPUProcess, _env_opt, maybe_resume
from theanompi_amd.lib.exchanger import (BSP_Exchanger,
                                         OverlappedGradExchanger)
from theanompi_amd.lib.recorder import Recorder
from theanompi_amd.models.base_model import _rtx


def default_strategy(sync_type, on_gpu):
    """bf16 wire is safe for deltas (cdd velocities / grads) but not for
    'avg', which ships the fp32 master weights themselves: quantizing
    those to bf16 every iteration erases updates below the weights' bf16
    ulp.  Weight averaging therefore defaults to an fp32 wire."""
    if not on_gpu:
        return "ar"
    return "rccl32" if sync_type == "avg" else "rccl_bf16"


def run():
    proc = GPUProcess()
    device = proc.init_device()
    proc.init_comm()
    hb = maybe_start(proc)
    model = proc.build_model()

    sync_type = _env_opt("sync_type", "cdd")
    strategy = _env_opt("exchange_strategy",
                        default_strategy(sync_type, proc.on_gpu))
    n_epochs = int(_env_opt("epochs", model.n_epochs))
    save_dir = _env_opt("save_dir", "")
    save_freq = int(_env_opt("save_freq", 0))

    model.compile_iter_fns(sync_type)
    overlap = (sync_type == "grad"
               and _env_opt("overlap", "1") not in ("0", "false"))
    config = {"sync_type": sync_type, "exchange_strategy": strategy,
              "size": proc.size, "rank": proc.rank,
              "bucket_mb": float(_env_opt("bucket_mb", 25))}
    if overlap:
        exchanger = OverlappedGradExchanger(
            model.recs, proc.size, bucket_mb=config["bucket_mb"])
    else:
        exchanger = BSP_Exchanger(config, model.params, model.vels)

    proc.broadcast_params(model)
    rec = Recorder({"rank": proc.rank, "size": proc.size,
                    "verbose": proc.verbose,
                    "print_freq": int(_env_opt("print_freq", 40)),
                    "sync_fn": (torch.cuda.synchronize if proc.on_gpu
                                else None)})
    start_epoch = maybe_resume(model, rec)
    proc.barrier()

    # test hooks: simulate a failing rank.  die_mode "exit" dies
    # abruptly (transport-detectable); "wedge" keeps the process and
    # its sockets alive but silences the heartbeat — the analog of a
    # rank wedged in a hung collective, which only the watchdog catches
    die_at = int(_env_opt("die_at", -1))
    die_rank = int(_env_opt("die_rank", 0))
    die_mode = _env_opt("die_mode", "exit")
    for epoch in range(start_epoch, n_epochs):
        rec.start_epoch()
        if epoch > start_epoch:
            model.reset_iter("train")   # per-epoch shard reshuffle
        n_batches = model.data.n_batch_train
        for i in range(n_batches):
            if die_at >= 0 and rec.uidx >= die_at \
                    and proc.rank == die_rank:
                if die_mode == "wedge":
                    if hb:
                        hb.stop()
                    import time as _t
                    _t.sleep(3600)
                os._exit(1)
            rec.start()
            model.train_iter(i, rec)
            with _rtx("exchange"):
                if overlap:
                    exchanger.finish(rec)
                elif sync_type == "grad":
                    exchanger.set_grad_buffers(model.opt.grads())
                    exchanger.exchange(rec)
                elif sync_type == "cdd" and hasattr(model, "vel_arena"):
                    exchanger.exchange_flat_inplace([model.vel_arena],
                                                    rec)
                else:
                    exchanger.exchange(rec)
            rec.start()
            model.apply_update()
            rec.end("calc")
            rec.print_train_info(i)
        rec.print_train_info(n_batches - 1, force=True)

        # sharded validation, stats all-reduced over the control group
        stats = torch.zeros(4)
        for j in range(proc.rank, model.data.n_batch_val, proc.size):
            cost, top1, top5 = model.val_iter(j, rec)
            stats += torch.tensor([cost, top1, top5, 1.0])
        if dist.is_initialized():
            dist.all_reduce(stats, group=proc.ctrl)
        if stats[3] > 0:
            rec.val_error(rec.uidx, *(stats[:3] / stats[3]).tolist())

        model.adjust_hyperp(epoch + 1)
        rec.end_epoch()
        if save_dir and save_freq and proc.rank == 0 \
                and (epoch + 1) % save_freq == 0:
            model.save(save_dir, epoch + 1)
            rec.save(os.path.join(save_dir, "inforec.pkl"))
        proc.barrier()

    dump_dir = _env_opt("dump_dir", "")
    if dump_dir:    # test hook: dump final params per rank
        import numpy as np
        arrs = {f"p{i}": p.detach().float().cpu().numpy()
                for i, p in enumerate(model.params)}
        np.savez(os.path.join(dump_dir, f"params_rank{proc.rank}.npz"),
                 **arrs)
    if hb:
        hb.stop()
    proc.finalize()
    return model, rec


if __name__ == "__main__":
    run()
