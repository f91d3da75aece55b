"""AlexNet-128b — rebuild of ``theanompi/models/alex_net.py`` [pub]
(SURVEY.md §2.4): classic 1-crop AlexNet, batch 128/GPU, LRN, grouped
conv2/4/5, dropout on FC6/7; SGD + momentum, step LR schedule.

This is the BASELINE headline model (images/sec, BSP, 1/2/4/8 GPUs).
"""

import torch.nn as nn

from theanompi_amd.models import layers as L
from theanompi_amd.models.base_model import BaseModel
from theanompi_amd.models.data.synthetic import SyntheticData


class AlexNet(BaseModel):

    name = "alexnet"
    batch_size = 128
    n_epochs = 70
    lr = 0.01
    momentum = 0.9
    weight_decay = 5e-4
    n_classes = 1000
    image_size = 227

    def lr_steps(self):
        return (20, 40, 60)

    def build_module(self):
        dev = str(self.device)
        B = self.batch_size
        use_lrn = True
        m = nn.Sequential()
        # conv1: 227x227x3(pad 4) -> 55x55x96
        m.append(L.Conv2d((227, 227, 4 if self.gpu else 3), 96, 11,
                          stride=4, pad=0, relu=True,
                          init=("normal", 0.01), bias_init=0.0,
                          device=dev, batch=B, needs_dgrad=False))
        if use_lrn:
            m.append(L.LRN(5, 1e-4, 0.75, 1.0))
        m.append(L.MaxPool2d(3, 2))                      # -> 27x27x96
        # conv2: 5x5 pad 2, groups 2 -> 27x27x256
        m.append(L.Conv2d((27, 27, 96), 256, 5, stride=1, pad=2, groups=2,
                          relu=True, init=("normal", 0.01), bias_init=0.1,
                          device=dev, batch=B))
        if use_lrn:
            m.append(L.LRN(5, 1e-4, 0.75, 1.0))
        m.append(L.MaxPool2d(3, 2))                      # -> 13x13x256
        m.append(L.Conv2d((13, 13, 256), 384, 3, stride=1, pad=1,
                          relu=True, init=("normal", 0.01), bias_init=0.0,
                          device=dev, batch=B))
        m.append(L.Conv2d((13, 13, 384), 384, 3, stride=1, pad=1, groups=2,
                          relu=True, init=("normal", 0.01), bias_init=0.1,
                          device=dev, batch=B))
        m.append(L.Conv2d((13, 13, 384), 256, 3, stride=1, pad=1, groups=2,
                          relu=True, init=("normal", 0.01), bias_init=0.1,
                          device=dev, batch=B))
        m.append(L.MaxPool2d(3, 2))                      # -> 6x6x256
        m.append(L.Flatten())
        m.append(L.Dropout(0.5))
        m.append(L.Linear(9216, 4096, relu=True, init=("normal", 0.005),
                          bias_init=0.1, device=dev, batch=B))
        m.append(L.Dropout(0.5))
        m.append(L.Linear(4096, 4096, relu=True, init=("normal", 0.005),
                          bias_init=0.1, device=dev, batch=B))
        m.append(L.Linear(4096, self.n_classes, init=("normal", 0.01),
                          bias_init=0.0, device=dev, batch=B))
        return m

    def build_data(self):
        return self.imagenet_or_synthetic(
            self.image_size, lambda: self._synthetic_data())

    def _synthetic_data(self):
        return SyntheticData(
            batch_size=self.batch_size,
            image_hw=self.image_size,
            n_classes=self.n_classes,
            device=self.device,
            c_phys=4 if self.gpu else 3,
            n_train_batches=int(self.config.get("n_iters", 64)),
            n_val_batches=int(self.config.get("val_iters", 8)),
            rank=self.rank, size=self.size)
