"""Tiny synthetic model — rebuild of the reference's exchanger smoke-test
model (``theanompi/models/test_model.py`` [MED]):
a minimal CNN + random data that satisfies the model contract, used to
exercise rules/exchangers without real data or a big graph.  Runs on CPU
in seconds — the workhorse of the CPU test suite.
"""

import torch.nn as nn

from theanompi_amd.models import layers as L
from theanompi_amd.models.base_model import BaseModel
from theanompi_amd.models.data.synthetic import SyntheticData


class TestModel(BaseModel):

    __test__ = False            # not a pytest class
    name = "test_model"
    batch_size = 8
    n_epochs = 2
    lr = 0.05
    momentum = 0.9
    weight_decay = 1e-4
    n_classes = 10
    image_size = 16

    def lr_steps(self):
        return (1,)

    def build_module(self):
        dev = str(self.device)
        B = self.batch_size
        m = nn.Sequential()
        m.append(L.Conv2d((16, 16, 4 if self.gpu else 3), 16, 3, stride=1,
                          pad=1, relu=True, device=dev, batch=B,
                          needs_dgrad=False))
        m.append(L.MaxPool2d(2, 2))                      # 8x8x16
        m.append(L.Conv2d((8, 8, 16), 32, 3, stride=1, pad=1, relu=True,
                          device=dev, batch=B))
        m.append(L.MaxPool2d(2, 2))                      # 4x4x32
        m.append(L.Flatten())
        m.append(L.Linear(4 * 4 * 32, self.n_classes, device=dev, batch=B))
        return m

    def build_data(self):
        return SyntheticData(
            batch_size=self.batch_size, image_hw=self.image_size,
            n_classes=self.n_classes, device=self.device,
            c_phys=4 if self.gpu else 3,
            n_train_batches=int(self.config.get("n_iters", 8)),
            n_val_batches=int(self.config.get("val_iters", 2)),
            rank=self.rank, size=self.size, use_loader=False)
