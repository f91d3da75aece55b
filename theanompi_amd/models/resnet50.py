"""ResNet-50 BSP bf16 — rebuild of ``theanompi/models/resnet50.py`` /
``models/lasagne_model_zoo/resnet50.py`` [pub]:
bottleneck residual blocks with BatchNorm (N7 kernels), batch 256/GPU
per BASELINE config #5 (288 GB HBM sizing).
"""

import os

import torch.nn as nn

from theanompi_amd.models import layers as L
from theanompi_amd.models.base_model import BaseModel
from theanompi_amd.models.data.synthetic import SyntheticData
from theanompi_amd.ops import functional as OF


# fused conv-epilogue BN statistics (the conv's LDS-staged epilogue
# also emits per-channel sum/sum-sq, replacing the separate bn_sums
# pass).  Measured: ~1% SLOWER end-to-end once the standalone bn_sums
# kernel had been tuned (adaptive width + parallel epilogue), so OFF
# by default; kept gated for re-evaluation on future shapes.
_BN_FUSE = os.environ.get("THEANOMPI_BN_FUSE", "0") not in ("0", "")


class _ConvBN(nn.Module):
    def __init__(self, hw, cin, k, r, stride=1, pad=0, relu=True,
                 dev="cpu", B=1, needs_dgrad=True):
        super().__init__()
        gpu = str(dev).startswith("cuda")
        self.conv = L.Conv2d((hw, hw, cin), k, r, stride=stride, pad=pad,
                             relu=False, device=dev, batch=B,
                             needs_dgrad=needs_dgrad, init=("he",),
                             has_bias=False,    # BN supplies the shift
                             collect_stats=_BN_FUSE and gpu)
        self.bn = L.BatchNorm2d(self.conv.out_shape[2], relu=relu,
                                device=dev)
        self.out_hw = self.conv.out_shape[0]
        self.out_c = self.conv.out_shape[2]

    def forward(self, x, residual=None, relu=None):
        y = self.conv(x)
        pre = self.conv.pop_stats() if self.conv.gpu else None
        return self.bn(y, prestats=pre, residual=residual, relu=relu)


class _Bottleneck(nn.Module):
    """1x1 -> 3x3 -> 1x1 (x4) with identity/projection shortcut."""

    def __init__(self, hw, cin, planes, stride, dev, B):
        super().__init__()
        cout = planes * 4
        self.c1 = _ConvBN(hw, cin, planes, 1, dev=dev, B=B)
        self.c2 = _ConvBN(hw, self.c1.out_c, planes, 3, stride=stride,
                          pad=1, dev=dev, B=B)
        self.c3 = _ConvBN(self.c2.out_hw, self.c2.out_c, cout, 1,
                          relu=False, dev=dev, B=B)
        if stride != 1 or cin != self.c3.out_c:
            self.down = _ConvBN(hw, cin, cout, 1, stride=stride,
                                relu=False, dev=dev, B=B)
        else:
            self.down = None
        self.out_hw = self.c3.out_hw
        self.out_c = self.c3.out_c

    def forward(self, x):
        s = self.down(x) if self.down is not None else x
        # residual join fused into the exit BN's apply pass:
        # y = relu(bn(conv(..)) + s) in one kernel (the separate
        # add_relu passes were ~8% of the step)
        return self.c3(self.c2(self.c1(x)), residual=s, relu=True)


class _ResNet50Net(nn.Module):
    def __init__(self, dev, B, n_classes, gpu):
        super().__init__()
        cin = 4 if gpu else 3
        self.stem = _ConvBN(224, cin, 64, 7, stride=2, pad=3, dev=dev,
                            B=B, needs_dgrad=False)        # -> 112
        self.pool = L.MaxPool2d(3, 2, 1)                   # -> 56
        blocks = []
        hw, c = 56, self.stem.out_c
        for planes, n, stride in ((64, 3, 1), (128, 4, 2),
                                  (256, 6, 2), (512, 3, 2)):
            for i in range(n):
                b = _Bottleneck(hw, c, planes, stride if i == 0 else 1,
                                dev, B)
                blocks.append(b)
                hw, c = b.out_hw, b.out_c
        self.blocks = nn.Sequential(*blocks)
        self.gap = L.AvgPool2d(hw, 1, 0)
        self.fc = L.Linear(c, n_classes, device=dev, batch=B)

    def forward(self, x):
        y = self.pool(self.stem(x))
        y = self.blocks(y)
        y = self.gap(y).reshape(y.shape[0], -1)
        return self.fc(y)


class ResNet50(BaseModel):

    name = "resnet50"
    batch_size = 256
    n_epochs = 90
    lr = 0.1
    momentum = 0.9
    weight_decay = 1e-4
    n_classes = 1000
    image_size = 224

    def lr_steps(self):
        return (30, 60, 80)

    def build_module(self):
        return _ResNet50Net(str(self.device), self.batch_size,
                            self.n_classes, self.gpu)

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
