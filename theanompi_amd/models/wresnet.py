"""Wide-ResNet (CIFAR-shaped) — rebuild of
``theanompi/models/keras_model_zoo/wresnet.py`` [pub].
The reference wrapped a Keras model to prove the model contract is
framework-agnostic; here the same WRN-16-4 architecture runs natively
on the layer library, and any object satisfying the contract (see
models/__init__.py) still trains under every rule.
"""

import torch.nn as nn

from theanompi_amd.models import layers as L
from theanompi_amd.models.base_model import BaseModel
from theanompi_amd.models.data.synthetic import SyntheticData
from theanompi_amd.ops import functional as OF


class _WRNBlock(nn.Module):
    """BN-ReLU-Conv x2 residual block (pre-activation)."""

    def __init__(self, hw, cin, cout, stride, dev, B):
        super().__init__()
        self.bn1 = L.BatchNorm2d(cin, relu=True, device=dev)
        self.c1 = L.Conv2d((hw, hw, cin), cout, 3, stride=stride, pad=1,
                           device=dev, batch=B, init=("he",))
        ohw = self.c1.out_shape[0]
        self.bn2 = L.BatchNorm2d(self.c1.out_shape[2], relu=True,
                                 device=dev)
        self.c2 = L.Conv2d((ohw, ohw, self.c1.out_shape[2]), cout, 3,
                           stride=1, pad=1, device=dev, batch=B,
                           init=("he",))
        if stride != 1 or cin != self.c2.out_shape[2]:
            self.down = L.Conv2d((hw, hw, cin), cout, 1, stride=stride,
                                 device=dev, batch=B, init=("he",))
        else:
            self.down = None
        self.out_hw = ohw
        self.out_c = self.c2.out_shape[2]

    def forward(self, x):
        y = self.bn1(x)
        s = self.down(y) if self.down is not None else x
        y = self.c2(self.bn2(self.c1(y)))
        return OF.add(y, s, relu=False)


class _WRNNet(nn.Module):
    def __init__(self, dev, B, n_classes, gpu, depth=16, widen=4):
        super().__init__()
        n = (depth - 4) // 6
        widths = [16, 16 * widen, 32 * widen, 64 * widen]
        cin = 4 if gpu else 3
        self.stem = L.Conv2d((32, 32, cin), widths[0], 3, pad=1,
                             device=dev, batch=B, needs_dgrad=False,
                             init=("he",))
        blocks = []
        hw, c = 32, self.stem.out_shape[2]
        for stage, w in enumerate(widths[1:]):
            for i in range(n):
                stride = 2 if (stage > 0 and i == 0) else 1
                b = _WRNBlock(hw, c, w, stride, dev, B)
                blocks.append(b)
                hw, c = b.out_hw, b.out_c
        self.blocks = nn.Sequential(*blocks)
        self.bn = L.BatchNorm2d(c, relu=True, device=dev)
        self.gap = L.AvgPool2d(hw, 1, 0)
        self.fc = L.Linear(c, n_classes, device=dev, batch=B)

    def forward(self, x):
        y = self.blocks(self.stem(x))
        y = self.gap(self.bn(y)).reshape(y.shape[0], -1)
        return self.fc(y)


class WResNet(BaseModel):

    name = "wresnet"
    batch_size = 128
    n_epochs = 200
    lr = 0.1
    momentum = 0.9
    weight_decay = 5e-4
    n_classes = 10
    image_size = 32

    def lr_steps(self):
        return (60, 120, 160)

    def build_module(self):
        return _WRNNet(str(self.device), self.batch_size, self.n_classes,
                       self.gpu)

    def build_data(self):
        return SyntheticData(
            batch_size=self.batch_size, image_hw=self.image_size,
            n_classes=self.n_classes, device=self.device,
            c_phys=4 if self.gpu else 3,
            n_train_batches=int(self.config.get("n_iters", 32)),
            n_val_batches=int(self.config.get("val_iters", 4)),
            rank=self.rank, size=self.size, use_loader=False)
