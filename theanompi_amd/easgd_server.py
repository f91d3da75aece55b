"""EASGD parameter server — rebuild of ``theanompi/easgd_server.py`` [pub]
(SURVEY.md §2.2, §3.3).  Rank 0 owns the center variables w~ on its own
GPU.  Event loop: poll per-worker request keys on the TCP control store
(the reference used MPI Iprobe; gloo cannot poll a pending irecv, so
requests signal through the store); per request run the elastic update
w~ += alpha*(w_i - w~) with the parameter swap riding RCCL
point-to-point over the direct xGMI link; periodically validate and
checkpoint; count total worker iterations to decide stop.
"""

import os
import time

import torch

from theanompi_amd.lib.heartbeat import maybe_start
from theanompi_amd.lib.base import GPUProcess, _env_opt, maybe_resume
from theanompi_amd.lib.exchanger import EASGD_Exchanger
from theanompi_amd.lib.recorder import Recorder


def run():
    proc = GPUProcess()
    proc.init_device()
    proc.init_comm()
    hb = maybe_start(proc)
    model = proc.build_model()
    model.compile_iter_fns("avg")        # server never trains; needs params

    alpha = float(_env_opt("alpha", 0.5))
    n_total = int(_env_opt("n_total_iters",
                           model.data.n_batch_train * model.n_epochs
                           * max(proc.size - 1, 1)))
    val_freq = int(_env_opt("val_freq", model.data.n_batch_train))
    save_dir = _env_opt("save_dir", "")

    # resume BEFORE the broadcast so workers start from the restored
    # center variables
    start_epoch = maybe_resume(model)
    proc.broadcast_params(model)
    exch = EASGD_Exchanger({"alpha": alpha}, model.params, role="server")
    rec = Recorder({"rank": proc.rank, "size": proc.size,
                    "verbose": proc.verbose})
    if start_epoch:
        ip = os.path.join(_env_opt("resume_dir", ""), "inforec.pkl")
        if os.path.exists(ip):
            rec.load(ip)
    store = proc.control_store()
    proc.barrier()

    workers = set(range(1, proc.size))
    seq = {w: 0 for w in workers}

    uidx = 0
    last_val = 0
    epoch = start_epoch
    t0 = time.time()
    rr = 0          # rotating scan start: fairness under 7 workers
    while workers:
        progressed = False
        # rotate the poll order each sweep so a low rank with
        # back-to-back requests cannot starve the high ranks (the
        # sorted scan always served rank 1 first)
        order = sorted(workers)
        rr = (rr + 1) % max(len(order), 1)
        for w in order[rr:] + order[:rr]:
            key = f"ea/req/{w}/{seq[w]}"
            if not store.check([key]):
                continue
            progressed = True
            tau = int(store.get(key))
            flag = 1 if uidx + tau < n_total else 0
            store.set(f"ea/resp/{w}/{seq[w]}", str(flag))
            exch.server_exchange(w)          # recv params / send center
            uidx += tau
            seq[w] += 1
            if flag == 0:
                workers.discard(w)
        if not progressed:
            time.sleep(0.0005)
            continue
        if uidx - last_val >= val_freq:
            last_val = uidx
            epoch += 1
            model.refresh_all()      # bf16 working copies, for val only
            stats = torch.zeros(3)
            n = 0
            for j in range(model.data.n_batch_val):
                c, t1, t5 = model.val_iter(j, rec)
                stats += torch.tensor([c, t1, t5])
                n += 1
            rec.epoch = epoch
            rec.val_error(uidx, *(stats / max(n, 1)).tolist())
            model.adjust_hyperp(epoch)
            if save_dir:
                model.save(save_dir, epoch)
                rec.save(os.path.join(save_dir, "inforec.pkl"))

    if proc.verbose:
        print(f"[easgd server] done: {uidx} worker iters in "
              f"{time.time() - t0:.1f}s", flush=True)
    proc.barrier()
    if hb:
        hb.stop()
    proc.finalize()
    return model, rec


if __name__ == "__main__":
    run()
