"""VGG-16 — rebuild of ``theanompi/models/lasagne_model_zoo/vgg.py``
[pub]: the reference's EASGD example model (BASELINE
config #4: 1-server / 7-worker elastic averaging over xGMI).
"""

import torch.nn as nn

from theanompi_amd.models import layers as L
from theanompi_amd.models.base_model import BaseModel
from theanompi_amd.models.data.synthetic import SyntheticData


class VGG16(BaseModel):

    name = "vgg16"
    batch_size = 64
    n_epochs = 74
    lr = 0.01
    momentum = 0.9
    weight_decay = 5e-4
    n_classes = 1000
    image_size = 224

    def lr_steps(self):
        return (25, 50)

    def build_module(self):
        dev = str(self.device)
        B = self.batch_size
        cfg = [(64, 2), (128, 2), (256, 3), (512, 3), (512, 3)]
        m = nn.Sequential()
        hw, c = 224, 4 if self.gpu else 3
        first = True
        for k, n in cfg:
            for _ in range(n):
                conv = L.Conv2d((hw, hw, c), k, 3, stride=1, pad=1,
                                relu=True, device=dev, batch=B,
                                needs_dgrad=not first, init=("he",))
                first = False
                m.append(conv)
                c = conv.out_shape[2]
            m.append(L.MaxPool2d(2, 2))
            hw //= 2
        m.append(L.Flatten())
        m.append(L.Linear(hw * hw * c, 4096, relu=True,
                          init=("normal", 0.005), bias_init=0.1,
                          device=dev, batch=B))
        m.append(L.Dropout(0.5))
        m.append(L.Linear(4096, 4096, relu=True, init=("normal", 0.005),
                          bias_init=0.1, device=dev, batch=B))
        m.append(L.Dropout(0.5))
        m.append(L.Linear(4096, self.n_classes, init=("normal", 0.01),
                          device=dev, batch=B))
        return m

    def build_data(self):
        return self.imagenet_or_synthetic(
            self.image_size, lambda: self._synthetic_data())

    def _synthetic_data(self):
        return SyntheticData(
            batch_size=self.batch_size, image_hw=self.image_size,
            n_classes=self.n_classes, device=self.device,
            c_phys=4 if self.gpu else 3,
            n_train_batches=int(self.config.get("n_iters", 32)),
            n_val_batches=int(self.config.get("val_iters", 4)),
            rank=self.rank, size=self.size)
