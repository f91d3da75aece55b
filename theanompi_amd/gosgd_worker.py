"""GOSGD worker — rebuild of ``theanompi/gosgd_worker.py`` [pub]
(Blot et al. GoSGD): local SGD; each iteration with
probability p push (w_i, alpha_i/2) to a uniformly drawn peer; merge any
received pushes  w <- (a*w + a_j*w_j)/(a + a_j),  a += a_j.

Transport: pushes are announced on the TCP control store (non-blocking
``check`` — gloo cannot poll a pending irecv in this torch build) and the
payload travels as a matched gloo isend/recv pair from pinned host
staging, so a push never blocks the sender's GPU stream (reference used
MPI Isend/Irecv + probe, C5).
"""

import os
import random
import time

import torch
import torch.distributed as dist

from theanompi_amd.lib.heartbeat import maybe_start
from theanompi_amd.lib.base import GPUProcess, _env_opt, maybe_resume
from theanompi_amd.lib.exchanger import GOSGD_Exchanger
from theanompi_amd.lib.recorder import Recorder


def run():
    proc = GPUProcess()
    proc.init_device()
    proc.init_comm()
    hb = maybe_start(proc)
    model = proc.build_model()
    model.compile_iter_fns("avg")

    p_push = float(_env_opt("p_push", 0.05))
    n_epochs = int(_env_opt("epochs", model.n_epochs))
    rng = random.Random(1234 + proc.rank)

    proc.broadcast_params(model)
    # GPU: one directed RCCL communicator per ordered rank pair so a
    # blocked A->B send can never head-of-line block the B->A direction
    # (new_group is collective: every rank iterates the same order)
    pair_groups = None
    if proc.on_gpu and proc.size > 1:
        pair_groups = {}
        for s in range(proc.size):
            for d in range(proc.size):
                if s != d:
                    pair_groups[(s, d)] = dist.new_group([s, d])
    exch = GOSGD_Exchanger({}, model.params, group=proc.ctrl,
                           pair_groups=pair_groups, rank=proc.rank)
    rec = Recorder({"rank": proc.rank, "size": proc.size,
                    "verbose": proc.verbose,
                    "print_freq": int(_env_opt("print_freq", 40)),
                    "sync_fn": (torch.cuda.synchronize if proc.on_gpu
                                else None)})
    store = proc.control_store()
    peers = [r for r in range(proc.size) if r != proc.rank]
    send_seq = {dst: 0 for dst in peers}    # pushes I sent to dst
    recv_seq = {src: 0 for src in peers}    # pushes I consumed from src
    me = proc.rank
    start_epoch = maybe_resume(model, rec)
    proc.barrier()

    def drain():
        """Receive + merge every announced, not-yet-consumed push."""
        merged = False
        for src in peers:
            while True:
                key = f"go/push/{me}/{src}/{recv_seq[src]}"
                if not store.check([key]):
                    break
                exch.recv_merge(src, store.get(key), tag=400 + src)
                recv_seq[src] += 1
                merged = True
        if merged:
            model.refresh_all()

    uidx = 0
    for epoch in range(start_epoch, n_epochs):
        rec.start_epoch()
        for i in range(model.data.n_batch_train):
            rec.start()
            model.train_iter(i, rec)
            model.apply_update()
            rec.end("calc")
            rec.start()
            drain()
            if peers and rng.random() < p_push:
                dst = rng.choice(peers)
                a = exch.push(dst, tag=400 + me)   # isend (non-blocking)
                # alpha rides the announcement (GPU wire = params only)
                store.set(f"go/push/{dst}/{me}/{send_seq[dst]}", repr(a))
                send_seq[dst] += 1
            rec.end("comm")
            uidx += 1
            rec.print_train_info(uidx)
        model.adjust_hyperp(epoch + 1)
        rec.end_epoch()

    # shutdown: finish my last send (receivers keep draining until every
    # rank has announced done), then drain stragglers until all done.
    if exch._pending_send is not None:
        exch._pending_send.wait()
    store.add("go/done", 1)
    t0 = time.time()
    while int(store.add("go/done", 0)) < proc.size:
        drain()
        time.sleep(0.001)
        if time.time() - t0 > 300:
            raise RuntimeError("gosgd shutdown drain timed out "
                               "(peer likely died)")
    drain()                                   # announced-but-unreceived
    if proc.on_gpu:
        # pair-group comm streams are not covered by the nccl barrier;
        # make sure every p2p kernel retired before teardown
        torch.cuda.synchronize()
    proc.barrier()

    stats = torch.zeros(3)
    n = 0
    for j in range(proc.rank, model.data.n_batch_val, proc.size):
        c, t1, t5 = model.val_iter(j, rec)
        stats += torch.tensor([c, t1, t5])
        n += 1
    if proc.verbose and n:
        rec.val_error(uidx, *(stats / n).tolist())
    save_dir = _env_opt("save_dir", "")
    if save_dir and proc.rank == 0:
        model.save(save_dir, model.epoch)
        rec.save(os.path.join(save_dir, "inforec.pkl"))
    if hb:
        hb.stop()
    proc.finalize()
    return model, rec


if __name__ == "__main__":
    run()
