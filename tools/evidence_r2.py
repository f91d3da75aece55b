"""Round-2 convergence evidence: CIFAR-10 quick-start + AlexNet short
run, loss must decrease; ImageNet shard pipeline end-to-end on GPU."""
import sys, os
sys.path.insert(0, ".")
import torch
from theanompi_amd.lib.recorder import Recorder

def train(modelcls, cfg, iters, tag):
    m = modelcls(cfg)
    m.compile_iter_fns("grad")
    losses = []
    for i in range(iters):
        loss, err = m.train_iter(i % m.data.n_batch_train)
        m.apply_update()
        if i % 50 == 0 or i == iters - 1:
            losses.append((i, float(loss), float(err)))
    print(f"[{tag}]")
    for i, l, e in losses:
        print(f"  iter {i:4d} loss {l:.4f} err {e:.4f}")
    assert losses[-1][1] < losses[0][1] * 0.7, "loss did not decrease"
    return losses

from theanompi_amd.models.cifar10 import Cifar10_model
train(Cifar10_model, {"device": "cuda:0", "batch_size": 128,
                      "n_iters": 32}, 600, "cifar10-600it")

from theanompi_amd.models.alex_net import AlexNet
train(AlexNet, {"device": "cuda:0", "batch_size": 128, "n_iters": 16},
      300, "alexnet-300it")

# ImageNet shard pipeline (sub-batch semantics) on hardware
from theanompi_amd.models.data.imagenet import (ImageNetData,
                                                write_synthetic_shards)
paths = write_synthetic_shards("/tmp/shards_r2", n_shards=6,
                               images_per_shard=256, hw=256)
d = ImageNetData(paths, paths[:1], batch_size=128, crop=227,
                 n_classes=1000, device="cuda:0")
try:
    assert d.sub_batches == 2 and d.n_batch_train == 12
    seen = 0
    for i in range(d.n_batch_train):
        x, y = d.get_train_batch(i)
        assert x.shape == (128, 227, 227, 4) and x.is_cuda
        seen += x.shape[0]
    torch.cuda.synchronize()
    print(f"[imagenet-shards] consumed {seen} images over "
          f"{d.n_batch_train} sub-batches (2 per 256-image shard) OK")
finally:
    d.close()
print("ALL CONVERGENCE/PIPELINE CHECKS OK")
