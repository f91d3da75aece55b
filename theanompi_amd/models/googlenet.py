"""GoogLeNet-32b (Inception v1) — rebuild of
``theanompi/models/googlenet.py`` [pub]: 9 inception
modules, 2 auxiliary classifiers (0.3-weighted losses), batch 32/GPU.
The BASELINE headline model alongside AlexNet; its ~120 small tensors
stress exchanger bucketing, and the 1x1/3x3/5x5 mix stresses the
implicit-GEMM tiler.
"""


import torch
import torch.nn as nn

from theanompi_amd.models import layers as L
from theanompi_amd.models.base_model import BaseModel
from theanompi_amd.models.data.synthetic import SyntheticData
from theanompi_amd.ops import functional as OF


class _Inception(nn.Module):
    """branches: 1x1 | 1x1->3x3 | 1x1->5x5 | pool->1x1, concat on C.

    GPU: the three same-input 1x1s (b1, 3x3-reduce, 5x5-reduce) run as
    ONE fused GEMM (L.MultiConv1x1) — GoogLeNet-32b is launch-latency
    bound, and this removes ~8 launches per module per step.  The block
    owns its checkpoint order (owns_logical_params) so the entries stay
    in the reference's unmerged per-branch sequence."""

    owns_logical_params = True

    def __init__(self, hw, cin, c1, c3r, c3, c5r, c5, cp, dev, B):
        super().__init__()
        kw = dict(device=dev, batch=B, init=("he",), relu=True)
        self.merged = str(dev).startswith("cuda")
        if self.merged:
            self.b135 = L.MultiConv1x1((hw, hw, cin), [c1, c3r, c5r],
                                       **kw)
            kp = self.b135.branch_kp
            w1, w3r, w5r = kp
        else:
            self.b1 = L.Conv2d((hw, hw, cin), c1, 1, **kw)
            self.b3a = L.Conv2d((hw, hw, cin), c3r, 1, **kw)
            self.b5a = L.Conv2d((hw, hw, cin), c5r, 1, **kw)
            w1 = self.b1.out_shape[2]
            w3r = self.b3a.out_shape[2]
            w5r = self.b5a.out_shape[2]
        self.b3b = L.Conv2d((hw, hw, w3r), c3, 3, pad=1, **kw)
        self.b5b = L.Conv2d((hw, hw, w5r), c5, 5, pad=2, **kw)
        self.pool = L.MaxPool2d(3, 1, 1)
        self.bp = L.Conv2d((hw, hw, cin), cp, 1, **kw)
        self._w1 = w1
        self.out_c = (w1 + self.b3b.out_shape[2]
                      + self.b5b.out_shape[2] + self.bp.out_shape[2])

    def logical_params(self):
        """Reference per-branch order: b1, b3a, b3b, b5a, b5b, bp."""
        if self.merged:
            br = self.b135.branch_logical_params()
            return (br[0] + br[1] + self.b3b.logical_params()
                    + br[2] + self.b5b.logical_params()
                    + self.bp.logical_params())
        return (self.b1.logical_params() + self.b3a.logical_params()
                + self.b3b.logical_params() + self.b5a.logical_params()
                + self.b5b.logical_params() + self.bp.logical_params())

    def forward(self, x):
        if self.merged and x.is_cuda:
            # one fused GEMM for the three 1x1s, then concat-writing
            # branch convs: every final output lands in the concat
            # buffer in place
            N = x.shape[0]
            P = self.b3b.out_shape[0]
            m = self.b135(x)
            # one-cat backward instead of per-slice scatter+add glue
            s1, s3, s5 = L.split_channels(m, self.b135.branch_kp)
            buf = torch.empty((N, P, P, self.out_c),
                              dtype=torch.bfloat16, device=x.device)
            ws = [self._w1, self.b3b.out_shape[2],
                  self.b5b.out_shape[2], self.bp.out_shape[2]]
            o = [0, ws[0], ws[0] + ws[1], ws[0] + ws[1] + ws[2]]
            y1 = L.copy_slice(buf[..., o[0]:o[1]], s1)
            y3 = self.b3b(s3, out=buf[..., o[1]:o[2]])
            y5 = self.b5b(s5, out=buf[..., o[2]:o[3]])
            yp = self.bp(self.pool(x), out=buf[..., o[3]:])
            return L.join_views(buf, y1, y3, y5, yp)
        y1 = self.b1(x)
        y3 = self.b3b(self.b3a(x))
        y5 = self.b5b(self.b5a(x))
        yp = self.bp(self.pool(x))
        return torch.cat([y1, y3, y5, yp], dim=-1)


class _AuxHead(nn.Module):
    """avgpool 5x5/3 -> 1x1 conv 128 -> FC 1024 -> dropout .7 -> FC 1000."""

    def __init__(self, hw, cin, n_classes, dev, B):
        super().__init__()
        self.pool = L.AvgPool2d(5, 3, 0)
        phw = (hw - 5) // 3 + 1
        self.conv = L.Conv2d((phw, phw, cin), 128, 1, relu=True,
                             device=dev, batch=B)
        cphys = self.conv.out_shape[2]
        self.fc1 = L.Linear(phw * phw * cphys, 1024, relu=True,
                            device=dev, batch=B)
        self.drop = L.Dropout(0.7)
        self.fc2 = L.Linear(1024, n_classes, device=dev, batch=B)

    def forward(self, x):
        y = self.conv(self.pool(x))
        y = y.reshape(y.shape[0], -1)
        return self.fc2(self.drop(self.fc1(y)))


class _GoogLeNetNet(nn.Module):
    def __init__(self, dev, B, n_classes, gpu):
        super().__init__()
        cin = 4 if gpu else 3
        kw = dict(device=dev, batch=B, relu=True)
        self.conv1 = L.Conv2d((224, 224, cin), 64, 7, stride=2, pad=3,
                              needs_dgrad=False, **kw)    # -> 112
        self.pool1 = L.MaxPool2d(3, 2, 1)                 # -> 56
        self.lrn1 = L.LRN(5, 1e-4, 0.75, 1.0)
        self.conv2 = L.Conv2d((56, 56, self.conv1.out_shape[2]), 64, 1,
                              **kw)
        self.conv3 = L.Conv2d((56, 56, self.conv2.out_shape[2]), 192, 3,
                              pad=1, **kw)
        self.lrn2 = L.LRN(5, 1e-4, 0.75, 1.0)
        self.pool2 = L.MaxPool2d(3, 2, 1)                 # -> 28
        c = self.conv3.out_shape[2]
        self.i3a = _Inception(28, c, 64, 96, 128, 16, 32, 32, dev, B)
        self.i3b = _Inception(28, self.i3a.out_c, 128, 128, 192, 32, 96,
                              64, dev, B)
        self.pool3 = L.MaxPool2d(3, 2, 1)                 # -> 14
        self.i4a = _Inception(14, self.i3b.out_c, 192, 96, 208, 16, 48,
                              64, dev, B)
        self.i4b = _Inception(14, self.i4a.out_c, 160, 112, 224, 24, 64,
                              64, dev, B)
        self.i4c = _Inception(14, self.i4b.out_c, 128, 128, 256, 24, 64,
                              64, dev, B)
        self.i4d = _Inception(14, self.i4c.out_c, 112, 144, 288, 32, 64,
                              64, dev, B)
        self.i4e = _Inception(14, self.i4d.out_c, 256, 160, 320, 32, 128,
                              128, dev, B)
        self.pool4 = L.MaxPool2d(3, 2, 1)                 # -> 7
        self.i5a = _Inception(7, self.i4e.out_c, 256, 160, 320, 32, 128,
                              128, dev, B)
        self.i5b = _Inception(7, self.i5a.out_c, 384, 192, 384, 48, 128,
                              128, dev, B)
        self.gap = L.AvgPool2d(7, 1, 0)
        self.drop = L.Dropout(0.4)
        self.fc = L.Linear(self.i5b.out_c, n_classes, device=dev,
                           batch=B)
        self.aux1 = _AuxHead(14, self.i4a.out_c, n_classes, dev, B)
        self.aux2 = _AuxHead(14, self.i4d.out_c, n_classes, dev, B)

    def forward(self, x):
        y = self.pool2(self.lrn2(self.conv3(self.conv2(
            self.lrn1(self.pool1(self.conv1(x)))))))
        y = self.pool3(self.i3b(self.i3a(y)))
        y4a = self.i4a(y)
        y4d = self.i4d(self.i4c(self.i4b(y4a)))
        y = self.pool4(self.i4e(y4d))
        y = self.i5b(self.i5a(y))
        y = self.gap(y).reshape(y.shape[0], -1)
        logits = self.fc(self.drop(y))
        if self.training:
            return logits, self.aux1(y4a), self.aux2(y4d)
        return logits


class GoogLeNet(BaseModel):

    name = "googlenet"
    batch_size = 32
    n_epochs = 70
    lr = 0.01
    momentum = 0.9
    weight_decay = 2e-4
    n_classes = 1000
    image_size = 224

    def lr_steps(self):
        return (24, 48)

    def build_module(self):
        return _GoogLeNetNet(str(self.device), self.batch_size,
                             self.n_classes, self.gpu)

    def forward(self, x):
        return self.module(x)

    def _fwd_bwd(self, x, y):
        self.opt.zero_grad()
        out = self.module(x)
        if isinstance(out, tuple):
            logits, a1, a2 = out
            loss0, _, err = OF.softmax_xent_err(logits, y, self.n_classes)
            loss1, _ = OF.softmax_xent(a1, y, self.n_classes)
            loss2, _ = OF.softmax_xent(a2, y, self.n_classes)
            loss = loss0 + 0.3 * (loss1 + loss2)
        else:
            logits = out
            loss, _, err = OF.softmax_xent_err(logits, y, self.n_classes)
        loss.backward()
        return loss.detach(), err

    def build_data(self):
        return self.imagenet_or_synthetic(
            self.image_size, lambda: self._synthetic_data())

    def _synthetic_data(self):
        return SyntheticData(
            batch_size=self.batch_size, image_hw=self.image_size,
            n_classes=self.n_classes, device=self.device,
            c_phys=4 if self.gpu else 3,
            n_train_batches=int(self.config.get("n_iters", 64)),
            n_val_batches=int(self.config.get("val_iters", 8)),
            rank=self.rank, size=self.size)
