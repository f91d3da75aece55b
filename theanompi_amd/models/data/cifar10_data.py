"""CIFAR-10 data object — rebuild of ``theanompi/models/data/cifar10.py``
[pub]: the whole dataset lives in host memory, no child
process.  Accepts the standard python-pickle batch files
(``data_batch_1..5`` + ``test_batch``) or any (X uint8 [N,32,32,3],
y int) arrays; falls back to synthetic arrays when no path is given
(no dataset downloads in this environment).
"""

import os
import pickle

import numpy as np
import torch


class Cifar10Data:

    def __init__(self, batch_size, device, path=None, n_classes=10,
                 c_phys=4, rank=0, size=1, seed=0,
                 synthetic_train=8192, synthetic_val=1024):
        self.batch_size = batch_size
        self.device = torch.device(device)
        self.gpu = self.device.type == "cuda"
        self.c_phys = c_phys
        self.rank, self.size = rank, size

        if path and os.path.isdir(path):
            xs, ys = [], []
            for i in range(1, 6):
                with open(os.path.join(path, f"data_batch_{i}"),
                          "rb") as f:
                    d = pickle.load(f, encoding="bytes")
                xs.append(d[b"data"])
                ys.append(d[b"labels"])
            x = np.concatenate(xs).reshape(-1, 3, 32, 32)
            x = x.transpose(0, 2, 3, 1)          # NHWC
            y = np.concatenate(ys)
            with open(os.path.join(path, "test_batch"), "rb") as f:
                d = pickle.load(f, encoding="bytes")
            xv = d[b"data"].reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1)
            yv = np.asarray(d[b"labels"])
        else:
            rng = np.random.RandomState(seed)
            x = rng.randint(0, 256, (synthetic_train, 32, 32, 3),
                            dtype=np.uint8)
            y = rng.randint(0, n_classes, (synthetic_train,))
            xv = rng.randint(0, 256, (synthetic_val, 32, 32, 3),
                             dtype=np.uint8)
            yv = rng.randint(0, n_classes, (synthetic_val,))

        def prep(a):
            out = np.zeros((*a.shape[:3], c_phys), dtype=np.uint8)
            out[..., :3] = a
            return torch.from_numpy(out)

        # shard by rank, as the reference partitioned shards per worker
        self.x = prep(x[rank::size])
        self.y = torch.from_numpy(y[rank::size].astype(np.int64))
        self.xv = prep(xv)
        self.yv = torch.from_numpy(yv.astype(np.int64))
        if self.gpu:
            self.x = self.x.to(self.device)
            self.y = self.y.to(self.device)
            self.xv = self.xv.to(self.device)
            self.yv = self.yv.to(self.device)
        self.n_batch_train = len(self.x) // batch_size
        self.n_batch_val = max(len(self.xv) // batch_size, 1)

    def _finish(self, u8, y):
        if self.gpu:
            from theanompi_amd import ops as _ops
            return _ops.ext().u8_finish(u8.contiguous(), 3), y
        x = u8.float() * (1 / 128.0) - 1.0
        return x[..., :3], y

    def get_train_batch(self, i):
        b = self.batch_size
        j = (i % self.n_batch_train) * b
        return self._finish(self.x[j:j + b], self.y[j:j + b])

    def get_val_batch(self, i):
        b = self.batch_size
        j = (i % self.n_batch_val) * b
        return self._finish(self.xv[j:j + b], self.yv[j:j + b])

    def reset(self, mode):
        pass
