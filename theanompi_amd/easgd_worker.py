"""EASGD worker — rebuild of ``theanompi/easgd_worker.py`` [pub]
(SURVEY.md §2.2, §3.3): train tau local iterations, then request an
elastic exchange from the server (request/response over the TCP control
store, parameter swap over RCCL p2p): send params, receive center,
apply  w_i -= alpha*(w_i - w~).
"""

import torch

from theanompi_amd.lib.heartbeat import maybe_start
from theanompi_amd.lib.base import GPUProcess, _env_opt
from theanompi_amd.lib.exchanger import EASGD_Exchanger
from theanompi_amd.lib.recorder import Recorder


def run():
    proc = GPUProcess()
    proc.init_device()
    proc.init_comm()
    hb = maybe_start(proc)
    model = proc.build_model()
    model.compile_iter_fns("avg")        # full local SGD steps

    alpha = float(_env_opt("alpha", 0.5))
    tau = int(_env_opt("tau", 4))
    proc.broadcast_params(model)
    exch = EASGD_Exchanger({"alpha": alpha}, model.params, role="worker")
    rec = Recorder({"rank": proc.rank, "size": proc.size,
                    "verbose": proc.rank == 1,
                    "print_freq": int(_env_opt("print_freq", 40)),
                    "sync_fn": (torch.cuda.synchronize if proc.on_gpu
                                else None)})
    store = proc.control_store()
    proc.barrier()

    uidx = 0
    seq = 0
    running = True
    while running:
        for _ in range(tau):
            rec.start()
            model.train_iter(uidx, rec)
            model.apply_update()
            rec.end("calc")
            uidx += 1
            rec.print_train_info(uidx)
        # request exchange: announce on the store, then matched p2p swap
        rec.start()
        store.set(f"ea/req/{proc.rank}/{seq}", str(tau))
        store.wait([f"ea/resp/{proc.rank}/{seq}"])
        flag = int(store.get(f"ea/resp/{proc.rank}/{seq}"))
        exch.worker_exchange(server_rank=0)  # send params / recv center
        model.refresh_all()
        rec.end("comm")
        seq += 1
        if flag == 0:
            running = False

    proc.barrier()
    if hb:
        hb.stop()
    proc.finalize()
    return model, rec


if __name__ == "__main__":
    run()
