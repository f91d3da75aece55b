"""Round-2 GPU evidence: learnable-task convergence (mirrors
tests/test_models_cpu.py::test_convergence_gpu) + the ImageNet shard
pipeline with the round-2 sub-batch semantics, end-to-end on hardware."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))

import torch as T  # noqa: E402


def main():

    from theanompi_amd.models.cifar10 import Cifar10_model  # noqa: E402

    T.manual_seed(0)
    m = Cifar10_model({"device": "cuda:0", "batch_size": 128,
                       "n_iters": 32})
    m.compile_iter_fns("grad")
    d = m.data
    templates = T.randint(0, 256, (10, 32, 32, 4), dtype=T.int16)
    templates[..., 3:] = 0
    with T.no_grad():
        y_all = d.y.cpu()
        noise = T.randint(-60, 60, d.x.shape, dtype=T.int16)
        newx = (templates[y_all] + noise).clamp(0, 255).to(T.uint8)
        newx[..., 3:] = 0
        d.x.copy_(newx.to(d.x.device))
    print("[cifar10 learnable-template task, 600 iters]")
    for i in range(600):
        loss, err = m.train_iter(i % d.n_batch_train)
        m.apply_update()
        if i % 100 == 0 or i == 599:
            print(f"  iter {i:4d} loss {float(loss):.4f} "
                  f"err {float(err):.4f}")
    assert float(err) < 0.02, f"did not learn: {float(err)}"

    # ImageNet shard pipeline (sub-batch semantics) on hardware
    from theanompi_amd.models.data.imagenet import (  # noqa: E402
        ImageNetData, write_synthetic_shards)

    paths = write_synthetic_shards("/tmp/shards_r2", n_shards=6,
                                   images_per_shard=256, hw=256)
    d2 = ImageNetData(paths, paths[:1], batch_size=128, crop=227,
                      n_classes=1000, device="cuda:0")
    try:
        assert d2.sub_batches == 2 and d2.n_batch_train == 12
        seen = 0
        for i in range(d2.n_batch_train):
            x, y = d2.get_train_batch(i)
            assert x.shape == (128, 227, 227, 4) and x.is_cuda
            seen += x.shape[0]
        T.cuda.synchronize()
        print(f"[imagenet-shards] consumed {seen} images over "
              f"{d2.n_batch_train} sub-batches (2 per 256-image shard) OK")
    finally:
        d2.close()
    print("ALL EVIDENCE CHECKS OK")


if __name__ == "__main__":   # REQUIRED: the shard loader
    main()                   # child spawns re-import this module
