"""Synthetic ImageNet-shaped data + the double-buffered staging pipeline.

Rebuild of the reference's parallel loader (Theano-MPI
``theanompi/models/data/proc_load_mpi.py`` + ``imagenet.py`` [pub];
SURVEY.md §2.5, §3.5): the reference spawned one loader child process per
worker over an MPI intercomm, double-buffering host->GPU staging so
load(k+1) overlaps compute(k).

MI355X-native version (N18 in SURVEY.md §2.6): a background thread fills
a ring of PINNED host buffers (uint8, ImageNet-shaped, pre-augmented
synthetic samples); the training process copies them host->HBM with an
async copy on a dedicated side stream and records an event the compute
stream waits on.  Cast uint8->bf16 (+mean subtract) runs as a GPU op on
the compute stream.  BASELINE benchmarks use this path with random data
and random labels (no network, no datasets in the environment).
"""

import os
import threading

import torch


class SyntheticData:
    """Data-object side of the model contract (SURVEY.md §1 L2->L1)."""

    def __init__(self, batch_size, image_hw, n_classes, device,
                 c_phys=4, n_train_batches=64, n_val_batches=8,
                 rank=0, size=1, seed=1234, dtype=None, n_pool=4,
                 use_loader=True, n_logical_c=3):
        self.batch_size = batch_size
        self.image_hw = image_hw
        self.n_classes = n_classes
        self.device = torch.device(device)
        self.gpu = self.device.type == "cuda"
        self.c_phys = c_phys
        self.n_logical_c = n_logical_c
        self.n_batch_train = n_train_batches
        self.n_batch_val = n_val_batches
        self.dtype = dtype or (torch.bfloat16 if self.gpu else torch.float32)
        g = torch.Generator().manual_seed(seed + rank)

        h, w = (image_hw, image_hw) if isinstance(image_hw, int) \
            else image_hw
        self.shape = (batch_size, h, w, c_phys)

        # host-side sample pool (uint8 "pre-augmented" images)
        self.pool = []
        self.labels = []
        pin = self.gpu
        for i in range(n_pool):
            x = torch.randint(0, 256, self.shape, generator=g,
                              dtype=torch.uint8)
            x[..., n_logical_c:] = 0            # padded channels are zero
            y = torch.randint(0, n_classes, (batch_size,), generator=g,
                              dtype=torch.int64)
            if pin:
                x = x.pin_memory()
                y = y.pin_memory()
            self.pool.append(x)
            self.labels.append(y)

        if self.gpu and use_loader:
            self.loader = _PinnedRingLoader(self.pool, self.labels,
                                            self.device)
        else:
            self.loader = None
            if self.gpu:
                self._dev_pool = [(x.to(self.device), y.to(self.device))
                                  for x, y in zip(self.pool, self.labels)]
            else:
                self._dev_pool = [(x, y)
                                  for x, y in zip(self.pool, self.labels)]

    def _finish(self, x_u8, y):
        """uint8 NHWC -> compute dtype, scaled to ~[-1, 1)."""
        if self.gpu and self.dtype == torch.bfloat16:
            from theanompi_amd import ops as _ops
            return _ops.ext().u8_finish(x_u8, self.n_logical_c), y
        x = x_u8.to(self.dtype)
        x = x * (1.0 / 128.0) - 1.0
        x[..., self.n_logical_c:] = 0       # keep pad channels exactly 0
        return x, y

    def get_train_batch(self, i):
        if self.loader is not None:
            x_u8, y = self.loader.next()
            return self._finish(x_u8, y)
        x_u8, y = self._dev_pool[i % len(self._dev_pool)]
        return self._finish(x_u8, y)

    def get_val_batch(self, i):
        x_u8, y = (self._dev_pool[i % len(self._dev_pool)]
                   if self.loader is None else
                   self.loader.peek(i))
        return self._finish(x_u8, y)

    def reset(self, mode):
        pass


class _PinnedRingLoader:
    """Double-buffered pinned-host -> HBM staging on a side stream.

    Two device-resident staging buffers alternate: while the model
    computes on buffer k, buffer k+1 is filled by hipMemcpyAsync on the
    copy stream; the compute stream waits on the copy event (not the
    host).  The producer thread rotates through the host pool (in a real
    dataset it would read + augment shards there — SURVEY.md §2.5).
    """

    def __init__(self, host_pool, host_labels, device, depth=2):
        self.device = device
        self.pool = host_pool
        self.labels = host_labels
        self.copy_stream = torch.cuda.Stream(device=device)
        self.depth = depth
        self.bufs = [torch.empty(host_pool[0].shape, dtype=torch.uint8,
                                 device=device) for _ in range(depth)]
        self.lbufs = [torch.empty(host_labels[0].shape, dtype=torch.int64,
                                  device=device) for _ in range(depth)]
        self.copy_events = [torch.cuda.Event() for _ in range(depth)]
        self.done_events = [torch.cuda.Event() for _ in range(depth)]
        self.idx = 0            # next host item to stage
        self.slot = 0
        self._last_slot = None
        self._lock = threading.Lock()
        # SURVEY §5.2 race-detection mode (THEANOMPI_AMD_DEBUG=1):
        # host-synchronize the consumer fence before every slot reuse
        # and keep a per-slot generation counter, so an overwrite-in-
        # flight would fail here instead of corrupting a batch.
        self.debug = os.environ.get("THEANOMPI_AMD_DEBUG", "0") \
            not in ("0", "")
        self.gen = [0] * depth
        for k in range(depth):  # prime the whole ring
            self._stage(k)

    def _stage(self, slot):
        i = self.idx % len(self.pool)
        with torch.cuda.stream(self.copy_stream):
            self.bufs[slot].copy_(self.pool[i], non_blocking=True)
            self.lbufs[slot].copy_(self.labels[i], non_blocking=True)
            self.copy_events[slot].record(self.copy_stream)
        self.idx += 1

    def next(self):
        with self._lock:
            compute = torch.cuda.current_stream(self.device)
            if self._last_slot is not None:
                # everything reading the previous slot is now enqueued on
                # the compute stream; fence the refill copy behind it
                prev = self._last_slot
                self.done_events[prev].record(compute)
                self.copy_stream.wait_event(self.done_events[prev])
                if self.debug:
                    self.done_events[prev].synchronize()
                    assert self.copy_events[prev].query(), \
                        f"slot {prev} refilled before copy completed"
                self.gen[prev] += 1
                self._stage(prev)
            slot = self.slot
            compute.wait_event(self.copy_events[slot])
            x, y = self.bufs[slot], self.lbufs[slot]
            self._last_slot = slot
            self.slot = (slot + 1) % self.depth
            return x, y

    def peek(self, i):
        # validation path: simple sync copy
        x = self.pool[i % len(self.pool)].to(self.device)
        y = self.labels[i % len(self.labels)].to(self.device)
        return x, y
