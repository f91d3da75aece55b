"""Round-2 GPU session evidence: full rule->worker->RCCL(1-rank)->
checkpoint workflow + end-to-end RESUME on hardware."""
import glob
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

from theanompi_amd import BSP  # noqa: E402

ck = "gpurun_out/sess_ck"
rule = BSP()
rule.init(devices=["cuda0"], modelfile="theanompi_amd.models.cifar10",
          modelclass="Cifar10_model", epochs=2, n_iters=16, seed=7,
          save_dir=ck, save_freq=2)
assert rule.wait() == [0]
rule2 = BSP()
rule2.init(devices=["cuda0"], modelfile="theanompi_amd.models.cifar10",
           modelclass="Cifar10_model", epochs=3, n_iters=16, seed=7,
           resume_dir=ck, resume_epoch=2, save_dir=ck, save_freq=1)
assert rule2.wait() == [0]
ws = sorted(os.path.basename(p) for p in glob.glob(ck + "/W_0_*.npy"))
assert "W_0_2.npy" in ws and "W_0_3.npy" in ws, ws
print("SESSION+RESUME OK:", ws)
