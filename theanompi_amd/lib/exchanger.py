"""Parameter exchangers — rebuild of ``theanompi/lib/exchanger.py`` [pub]
(SURVEY.md §2.3, §3.2, §3.3).

BSP_Exchanger
    Called once per iteration.  Two sync modes, as in the reference:
      'cdd'  exchange per-iteration update deltas (velocities): worker
             computes vels locally (opt.py phase 1), the exchanger
             all-reduces the flat vel buffer, and phase 2 applies the
             averaged delta to the weights.
      'avg'  worker takes a full local SGD step, the exchanger
             all-reduces the *weights* and multiplies by 1/N.
    A third mode 'grad' (not in the reference; the modern formulation)
    all-reduces gradients before a single fused optimizer step — it is
    numerically equivalent to 'cdd' for plain SGD+momentum and is the
    fast path used by bench.py.

EASGD_Exchanger
    Role 'server' or 'worker'.  Worker<->server parameter swaps ride RCCL
    point-to-point send/recv over the single xGMI hop that directly
    connects any two GPUs (reference used CUDA-aware MPI Send/Recv, C4).
    The elastic update w += a*(w_other - w) is a fused step on the flat
    buffer on each side.

GOSGD_Exchanger
    Probabilistic peer pushes (Blot et al. GoSGD).  Pushes travel over
    the gloo control group from pinned host staging so a push never
    blocks the sender's training stream (reference used Isend/Irecv, C5).
"""

import torch
import torch.distributed as dist

from theanompi_amd.lib.exchanger_strategy import make_strategy, flatten_params


class BSP_Exchanger:

    def __init__(self, config, param_list, vel_list=None, group=None):
        self.sync_type = config.get("sync_type", "cdd")
        self.size = int(config.get("size", dist.get_world_size(group)
                                   if dist.is_initialized() else 1))
        strategy = config.get("exchange_strategy", "rccl32")
        per_tensor = bool(config.get("per_tensor", False))
        bucket_mb = float(config.get("bucket_mb", 64))
        self.strategy = make_strategy(strategy, group=group)
        self.params = list(param_list)
        self.vels = list(vel_list) if vel_list is not None else None

        if self.sync_type == "avg":
            src = self.params
        elif self.sync_type == "cdd":
            assert self.vels is not None, "cdd mode needs velocity buffers"
            src = self.vels
        elif self.sync_type == "grad":
            src = None   # bound later to grad tensors via set_grad_buffers
        else:
            raise ValueError(f"unknown sync_type {self.sync_type!r}")

        self._flat = None
        self._views = None
        self._src = src
        self._per_tensor = per_tensor
        self._bucket_bytes = int(bucket_mb * (1 << 20))
        if src is not None:
            self._bind(src)

    def _bind(self, tensors):
        if self._per_tensor:
            self._flat = None
        else:
            self._flat, self._views = flatten_params(
                tensors, self._bucket_bytes)
        self._src = tensors

    def set_grad_buffers(self, grads):
        """'grad' mode: bind the gradient tensors produced by the model's
        backward.  Re-binding the same-shaped list (the per-iteration
        case: autograd replaces .grad tensors) reuses the flat buffers."""
        grads = list(grads)
        if (self._src is not None and len(grads) == len(self._src)
                and all(a.numel() == b.numel()
                        for a, b in zip(grads, self._src))):
            self._src = grads
            return
        self._bind(grads)

    # -- the per-iteration entry point ------------------------

    @torch.no_grad()
    def exchange(self, recorder=None):
        if self.size == 1:
            return
        if recorder:
            recorder.start()
        scale = 1.0 / self.size
        if self._per_tensor:
            self.strategy.exchange([t.data for t in self._src], scale=scale)
        else:
            for v, t in zip(self._views, self._src):
                v.copy_(t.data)
            self.strategy.exchange(self._flat, scale=scale)
            for v, t in zip(self._views, self._src):
                t.data.copy_(v)
        if recorder:
            recorder.end("comm")

    def exchange_flat_inplace(self, flat_buffers, recorder=None, scale="avg"):
        """Fast path: the caller's grads/vels already live in flat buffers
        (models allocate grads as bucket views); no copies needed."""
        if self.size == 1:
            return
        if recorder:
            recorder.start()
        s = 1.0 / self.size if scale == "avg" else scale
        self.strategy.exchange(flat_buffers, scale=s)
        if recorder:
            recorder.end("comm")


class OverlappedGradExchanger:
    """BSP 'grad'-mode exchanger that overlaps the all-reduce with the
    backward pass — headroom the reference never had (SURVEY.md §7
    phase 4: "exchange buckets as soon as ... finishes").

    Parameters are bucketed in reverse registration order (approximate
    backward order).  A post-accumulate-grad hook fires per parameter;
    when a bucket is complete its grads are packed to the bf16 wire
    buffer and an async all-reduce launches while backward continues.
    ``finish()`` waits, scales by 1/N and unpacks.  Numerically this is
    the reference's cdd rule (averaged deltas) for SGD+momentum.
    """

    def __init__(self, recs, size, group=None, bucket_mb=25,
                 wire_dtype=torch.bfloat16):
        self.size = size
        self.group = group
        self.enabled = size > 1
        dev = recs[0].master.device
        wire_dtype = wire_dtype if dev.type == "cuda" else torch.float32
        self.buckets = []
        cur, cur_elems = [], 0
        max_elems = int(bucket_mb * (1 << 20)) // 2
        for r in reversed(list(recs)):          # ~backward order
            if cur and cur_elems + r.master.numel() > max_elems:
                self.buckets.append(cur)
                cur, cur_elems = [], 0
            cur.append(r)
            cur_elems += r.master.numel()
        if cur:
            self.buckets.append(cur)
        self._state = []
        self._by_param = {}
        for bi, group_recs in enumerate(self.buckets):
            n = sum(r.master.numel() for r in group_recs)
            wire = torch.empty(n, dtype=wire_dtype, device=dev)
            offs = {}
            off = 0
            for r in group_recs:
                offs[id(r.master)] = off
                off += r.master.numel()
            st = {"recs": group_recs, "wire": wire, "offs": offs,
                  "ready": 0, "work": None}
            self._state.append(st)
            for r in group_recs:
                self._by_param[id(r.master)] = st
        if self.enabled:
            for st in self._state:
                for r in st["recs"]:
                    r.master.register_post_accumulate_grad_hook(
                        self._hook)

    @torch.no_grad()
    def _hook(self, param):
        st = self._by_param.get(id(param))
        if st is None:
            return
        st["ready"] += 1
        if st["ready"] == len(st["recs"]):
            self._launch(st)

    @torch.no_grad()
    def _launch(self, st):
        wire = st["wire"]
        for r in st["recs"]:
            off = st["offs"][id(r.master)]
            n = r.master.numel()
            g = r.master.grad
            if g is None:       # param got no grad this step (unused head)
                wire[off:off + n].zero_()
            else:
                wire[off:off + n].copy_(g.view(-1))
        st["work"] = dist.all_reduce(wire, group=self.group,
                                     async_op=True)

    @torch.no_grad()
    def finish(self, recorder=None):
        """Wait for in-flight reductions, unpack averaged grads."""
        if not self.enabled:
            return
        if recorder:
            recorder.start()
        inv = 1.0 / self.size
        for st in self._state:
            if st["ready"] == 0:
                continue
            if st["work"] is None:      # bucket partially filled: flush
                self._launch(st)
            st["work"].wait()
            st["wire"].mul_(inv)
            for r in st["recs"]:
                off = st["offs"][id(r.master)]
                n = r.master.numel()
                if r.master.grad is None:
                    # other ranks may still have contributed a gradient
                    r.master.grad = torch.zeros_like(r.master)
                r.master.grad.view(-1).copy_(st["wire"][off:off + n])
            st["ready"] = 0
            st["work"] = None
        if recorder:
            recorder.end("comm")


class EASGD_Exchanger:
    """Elastic-averaging swaps (reference EASGD_Exchanger [pub], C4)."""

    def __init__(self, config, param_list, role, group=None):
        self.role = role                      # 'server' | 'worker'
        self.alpha = float(config.get("alpha", 0.5))
        self.group = group
        self.params = list(param_list)
        self._flat, self._views = flatten_params(self.params)
        assert len(self._flat) >= 1
        # a single staging buffer the size of the largest bucket
        self._stage = [torch.empty_like(f) for f in self._flat]

    @torch.no_grad()
    def _pack(self):
        for v, p in zip(self._views, self.params):
            v.copy_(p.data)

    @torch.no_grad()
    def _unpack(self):
        for v, p in zip(self._views, self.params):
            p.data.copy_(v)

    @torch.no_grad()
    def server_exchange(self, worker_rank):
        """Server side of one elastic cycle: recv worker params, do
        center += alpha * (w_i - center), send center back."""
        self._pack()
        for f, s in zip(self._flat, self._stage):
            dist.recv(s, src=worker_rank, group=self.group)
        for f, s in zip(self._flat, self._stage):
            # center (f) += alpha * (worker (s) - center)
            f.add_(s.sub_(f), alpha=self.alpha)
        for f in self._flat:
            dist.send(f, dst=worker_rank, group=self.group)
        self._unpack()

    @torch.no_grad()
    def worker_exchange(self, server_rank=0):
        """Worker side: send own params, recv center, apply
        w -= alpha * (w - center)."""
        self._pack()
        for f in self._flat:
            dist.send(f, dst=server_rank, group=self.group)
        for s in self._stage:
            dist.recv(s, src=server_rank, group=self.group)
        for f, s in zip(self._flat, self._stage):
            # w (f) -= alpha * (w - center (s))  ==  f += alpha*(s - f)
            f.add_(s.sub_(f), alpha=self.alpha)
        self._unpack()


class GOSGD_Exchanger:
    """Gossip pushes (reference gosgd_worker.py [pub], C5).

    GPU path (reference C5 semantics: device-buffer Isend): the payload
    is one flat fp32 device tensor riding RCCL p2p over the direct xGMI
    link; alpha travels in the store announcement, so the merge
    w <- (a*w + a_j*w_j)/(a+a_j) runs entirely on-device (a fused lerp
    per parameter) with no host staging and no device<->host sync.
    Each ordered rank pair gets its OWN communicator (directed pair
    group): an RCCL send blocks its communicator's stream until the
    matching recv is posted, so A->B and B->A pushes sharing one
    communicator could head-of-line deadlock; per-direction comms make
    the matching order independent.

    CPU path: one flat fp32 tensor [alpha, w0, w1, ...] over gloo from
    pinned host staging.  Sender halves its own alpha and sends; the
    receiver merges  w <- (a*w + a_j*w_j) / (a + a_j),  a <- a + a_j.
    """

    def __init__(self, config, param_list, group=None, pair_groups=None,
                 rank=0):
        self.group = group
        self.params = list(param_list)
        self._flat, self._views = flatten_params(self.params)
        self.n = sum(f.numel() for f in self._flat)
        self.alpha = 1.0
        self.rank = rank
        dev = self.params[0].device
        self.on_gpu = dev.type == "cuda" and pair_groups is not None
        self.pair_groups = pair_groups or {}
        self._pending_send = None
        if self.on_gpu:
            self._dev_out = torch.empty(self.n, dtype=torch.float32,
                                        device=dev)
            self._dev_in = torch.empty_like(self._dev_out)
            self._out_views, self._in_views = [], []
            off = 0
            for p in self.params:
                n = p.numel()
                self._out_views.append(
                    self._dev_out[off:off + n].view(p.shape))
                self._in_views.append(
                    self._dev_in[off:off + n].view(p.shape))
                off += n
            return
        self._host_out = torch.empty(self.n + 1, dtype=torch.float32,
                                     pin_memory=dev.type == "cuda")
        self._host_in = torch.empty_like(self._host_out)
        self._rbufs = {}

    def _gather_flat(self, out):
        off = 1
        for p in self.params:
            n = p.numel()
            out[off:off + n].copy_(p.detach().reshape(-1).to(torch.float32))
            off += n

    def _scatter_flat(self, src):
        off = 1
        for p in self.params:
            n = p.numel()
            with torch.no_grad():
                p.copy_(src[off:off + n].view(p.shape).to(p.dtype))
            off += n

    def push(self, dst, tag=0):
        """Send (alpha_i/2, w_i) to dst; halve own alpha.  Returns the
        alpha that rides the announcement (GPU wire carries params
        only)."""
        if self._pending_send is not None:
            # stream-level on RCCL: also orders the repack below after
            # the previous send's kernel on the comm stream
            self._pending_send.wait()
        self.alpha *= 0.5
        if self.on_gpu:
            with torch.no_grad():
                for v, p in zip(self._out_views, self.params):
                    v.copy_(p.detach().to(torch.float32))
            self._pending_send = dist.isend(
                self._dev_out, dst=dst,
                group=self.pair_groups[(self.rank, dst)])
            return self.alpha
        self._host_out[0] = self.alpha
        self._gather_flat(self._host_out)
        self._pending_send = dist.isend(self._host_out, dst=dst,
                                        group=self.group, tag=tag)
        return self.alpha

    def recv_merge(self, src, announced, tag=0):
        """Receive one announced push from src and merge it.
        ``announced`` is the store value the sender published (its
        alpha_j on GPU; unused on CPU where alpha rides the wire)."""
        if self.on_gpu:
            a_j = float(announced)
            dist.recv(self._dev_in, src=src,
                      group=self.pair_groups[(src, self.rank)])
            t = a_j / (self.alpha + a_j)
            with torch.no_grad():
                for v, p in zip(self._in_views, self.params):
                    # w <- (1-t)*w + t*w_j  ==  (a*w + a_j*w_j)/(a+a_j)
                    p.data.lerp_(v.to(p.dtype), t)
            self.alpha += a_j
            return
        buf = self._rbufs.setdefault(
            src, torch.empty_like(self._host_in))
        dist.recv(buf, src=src, group=self.group, tag=tag)
        self.try_merge_buf(buf)

    def try_merge_buf(self, buf):
        """Merge a fully-received push buffer (CPU/gloo wire format)."""
        a_j = float(buf[0])
        a = self.alpha
        denom = a + a_j
        cur = torch.empty_like(buf)
        cur[0] = 0.0
        self._gather_flat(cur)
        merged = (cur * a + buf * a_j) / denom
        self._scatter_flat(merged)
        self.alpha = denom
