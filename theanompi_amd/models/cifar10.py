"""CIFAR-10 quick-start model — rebuild of ``theanompi/models/cifar10.py``
``Cifar10_model`` [pub]: the README's 2-GPU BSP example
and the de-facto integration test.  Small conv net on 32x32x3 images.
Data is synthetic CIFAR-shaped (no dataset downloads in this
environment); plug a real array store into ``build_data`` to train for
accuracy.
"""

import torch.nn as nn

from theanompi_amd.models import layers as L
from theanompi_amd.models.base_model import BaseModel


class Cifar10_model(BaseModel):

    name = "cifar10"
    batch_size = 128
    n_epochs = 50
    lr = 0.01
    momentum = 0.9
    weight_decay = 1e-4
    n_classes = 10
    image_size = 32

    def lr_steps(self):
        return (25, 40)

    def build_module(self):
        dev = str(self.device)
        B = self.batch_size
        m = nn.Sequential()
        m.append(L.Conv2d((32, 32, 4 if self.gpu else 3), 32, 5, stride=1,
                          pad=2, relu=True, init=("normal", 0.05),
                          device=dev, batch=B, needs_dgrad=False))
        m.append(L.MaxPool2d(3, 2, 1))                   # 16x16x32
        m.append(L.Conv2d((16, 16, 32), 32, 5, stride=1, pad=2, relu=True,
                          init=("normal", 0.05), device=dev, batch=B))
        m.append(L.AvgPool2d(3, 2, 1))                   # 8x8x32
        m.append(L.Conv2d((8, 8, 32), 64, 5, stride=1, pad=2, relu=True,
                          init=("normal", 0.05), device=dev, batch=B))
        m.append(L.AvgPool2d(3, 2, 1))                   # 4x4x64
        m.append(L.Flatten())
        m.append(L.Linear(4 * 4 * 64, 64, relu=True, device=dev, batch=B))
        m.append(L.Linear(64, self.n_classes, device=dev, batch=B))
        return m

    def build_data(self):
        from theanompi_amd.models.data.cifar10_data import Cifar10Data
        n_iters = int(self.config.get("n_iters", 32))
        return Cifar10Data(
            batch_size=self.batch_size, device=self.device,
            path=self.config.get("data_path"),
            n_classes=self.n_classes, c_phys=4,
            rank=self.rank, size=self.size,
            synthetic_train=self.batch_size * n_iters,
            synthetic_val=self.batch_size
            * int(self.config.get("val_iters", 4)))
