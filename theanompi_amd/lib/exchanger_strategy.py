"""Exchange strategies — the BSP data plane.

Rebuild of the reference's ``theanompi/lib/exchanger_strategy.py`` [pub]
(SURVEY.md §2.3, §2.8 C1-C3, §5.8).  The reference offered:

  ar      host-staged MPI Allreduce fp32
  nccl32  NCCL allReduce per tensor, fp32
  nccl16  NCCL allReduce with fp32->fp16 pack/unpack GPU kernels
  asa32   3-phase GPU-direct alltoall + segment-sum kernel + allgather
  asa16   asa with fp16 wire format

MI355X mapping: the data plane is RCCL over xGMI via
torch.distributed.  Each MI355X has 7 point-to-point xGMI links, so a ring
all-reduce is per-link bound (~153 GB/s); RCCL's fully-connected
reduce-scatter/all-gather path uses all links at once, which is exactly
the reference's "asa" decomposition.  Strategies here therefore are:

  ar        host-staged gloo allreduce (CPU plumbing mode + fallback)
  rccl32    ncclAllReduce on flat fp32 buckets      (alias: nccl32)
  rccl_bf16 bf16 wire format (pack/allreduce/unpack) (alias: nccl16)
  asa       reduce_scatter_tensor + all_gather_tensor on flat buckets —
            the 7-link path                          (alias: asa32)
  asa_bf16  asa with bf16 wire                       (alias: asa16)

All strategies operate on pre-flattened bucket tensors (the reference
exchanged tensor-by-tensor — a known weakness for GoogLeNet's ~120 small
tensors; ``per_tensor=True`` reproduces it for parity).  Buckets default
to sizes that keep all 7 xGMI links busy (~64 MB).
"""

import torch
import torch.distributed as dist

__all__ = ["make_strategy", "flatten_params"]


def flatten_params(tensors, bucket_bytes=64 << 20):
    """Group tensors into flat buckets; returns (buckets, views) where
    views[i] is a view into its bucket shaped like tensors[i].

    The flat buffers are allocated once; callers copy grads/params in and
    out (or better, arrange for grads to accumulate directly into views).
    """
    buckets, views = [], []
    cur, cur_elems = [], 0
    dtype = tensors[0].dtype
    device = tensors[0].device
    elem = tensors[0].element_size()
    max_elems = max(bucket_bytes // elem, 1)
    for t in tensors:
        assert t.dtype == dtype and t.device == device
        if cur and cur_elems + t.numel() > max_elems:
            buckets.append((cur, cur_elems))
            cur, cur_elems = [], 0
        cur.append(t)
        cur_elems += t.numel()
    if cur:
        buckets.append((cur, cur_elems))

    flat_list = []
    for group, n in buckets:
        flat = torch.empty(n, dtype=dtype, device=device)
        off = 0
        for t in group:
            v = flat[off:off + t.numel()].view(t.shape)
            views.append(v)
            off += t.numel()
        flat_list.append(flat)
    return flat_list, views


class _Strategy:
    name = None

    def __init__(self, process=None, group=None):
        self.process = process
        self.group = group

    def exchange(self, flat_buffers, scale=None):
        """All-reduce (sum) every flat buffer in place across ranks,
        multiplying by ``scale`` afterwards if given."""
        raise NotImplementedError


class ExchAllreduceHost(_Strategy):
    """'ar': stage to host, reduce over gloo, stage back.  On CPU tensors
    this is a straight gloo allreduce (the CPU plumbing mode)."""
    name = "ar"

    def exchange(self, flat_buffers, scale=None):
        for buf in flat_buffers:
            if buf.is_cuda:
                host = buf.detach().cpu()
                dist.all_reduce(host, group=self.group)
                buf.copy_(host)
            else:
                dist.all_reduce(buf, group=self.group)
            if scale is not None:
                buf.mul_(scale)


class ExchRccl32(_Strategy):
    """'rccl32': device-direct allReduce on each flat bucket (RCCL ring /
    fully-connected, RCCL's choice)."""
    name = "rccl32"

    def exchange(self, flat_buffers, scale=None):
        for buf in flat_buffers:
            dist.all_reduce(buf, group=self.group)
            if scale is not None:
                buf.mul_(scale)


class ExchRcclBf16(_Strategy):
    """'rccl_bf16': bf16 on the wire.  fp32 buckets are packed to bf16
    (replacing the reference's fp32->fp16 GPU pack kernels, N13), reduced,
    and unpacked; native-bf16 buckets go straight through."""
    name = "rccl_bf16"

    def __init__(self, process=None, group=None):
        super().__init__(process, group)
        self._wire = {}

    def exchange(self, flat_buffers, scale=None):
        for buf in flat_buffers:
            if buf.dtype == torch.bfloat16:
                dist.all_reduce(buf, group=self.group)
                if scale is not None:
                    buf.mul_(scale)
            else:
                key = id(buf)
                wire = self._wire.get(key)
                if wire is None or wire.numel() != buf.numel():
                    wire = torch.empty_like(buf, dtype=torch.bfloat16)
                    self._wire[key] = wire
                wire.copy_(buf)                      # pack fp32 -> bf16
                dist.all_reduce(wire, group=self.group)
                buf.copy_(wire)                      # unpack
                if scale is not None:
                    buf.mul_(scale)


class ExchAsa(_Strategy):
    """'asa': reduce-scatter + all-gather on each flat bucket.  This is the
    reference paper's 3-phase exchange (alltoall, local segment-sum,
    allgather — its segment-sum CUDA kernel N14 is subsumed by RCCL's
    fused ncclReduceScatter), and on an 8-GPU xGMI node it drives all 7
    links simultaneously instead of a ring's one."""
    name = "asa"
    wire_dtype = None  # same as buffer

    def __init__(self, process=None, group=None):
        super().__init__(process, group)
        self._pad = {}
        self._wire = {}

    def exchange(self, flat_buffers, scale=None):
        world = dist.get_world_size(self.group)
        for buf in flat_buffers:
            src = buf
            if self.wire_dtype is not None and buf.dtype != self.wire_dtype:
                key = id(buf)
                wire = self._wire.get(key)
                if wire is None:
                    wire = torch.empty_like(buf, dtype=self.wire_dtype)
                    self._wire[key] = wire
                wire.copy_(buf)
                src = wire
            n = src.numel()
            if n % world:                      # pad to a multiple of world
                key = ("pad", id(buf))
                padded = self._pad.get(key)
                pn = ((n + world - 1) // world) * world
                if padded is None or padded.numel() != pn:
                    padded = torch.zeros(pn, dtype=src.dtype, device=src.device)
                    self._pad[key] = padded
                padded[:n].copy_(src)
                work = padded
            else:
                work = src
            shard = work.view(world, -1)[dist.get_rank(self.group)].contiguous()
            dist.reduce_scatter_tensor(shard, work, group=self.group)
            dist.all_gather_into_tensor(work, shard, group=self.group)
            if work is not src:
                src.copy_(work[:n])
            if src is not buf:
                buf.copy_(src)
            if scale is not None:
                buf.mul_(scale)


class ExchAsaBf16(ExchAsa):
    name = "asa_bf16"
    wire_dtype = torch.bfloat16


class ExchAsaXgmi(_Strategy):
    """'asa_xgmi': the SURVEY §5.8 design, fully explicit — the
    reference paper's 3-phase ASA exchange with OUR reduction kernel:

      1. all-to-all of 1/world bucket shards.  RCCL implements this as
         grouped ncclSend/ncclRecv — on an 8-GPU xGMI node every rank
         sends a distinct 1/8 shard to each of its 7 peers over that
         peer's dedicated point-to-point link simultaneously (~7x a
         ring's per-link bound).
      2. local segment-sum across the world received copies — the
         hand-written HIP kernel N14 (fp32 accumulation even on a bf16
         wire, unlike ncclReduceScatter's wire-dtype math).
      3. all-gather of the reduced shards.

    Compared to 'asa' (fused ncclReduceScatter+AllGather) this trades
    one extra HBM round-trip for explicit control of the reduction
    dtype and per-link traffic; A/B both on the 8-GPU node.
    """
    name = "asa_xgmi"
    wire_dtype = None

    def __init__(self, process=None, group=None):
        super().__init__(process, group)
        self._scratch = {}

    def _buffers(self, key, pn, dtype, device):
        s = self._scratch.get(key)
        if s is None or s[0].numel() != pn or s[0].dtype != dtype:
            work = torch.zeros(pn, dtype=dtype, device=device)
            recv = torch.empty(pn, dtype=dtype, device=device)
            shard = torch.empty(pn // dist.get_world_size(self.group),
                                dtype=dtype, device=device)
            s = (work, recv, shard)
            self._scratch[key] = s
        return s

    def exchange(self, flat_buffers, scale=None):
        world = dist.get_world_size(self.group)
        rank = dist.get_rank(self.group)
        for buf in flat_buffers:
            src = buf
            if self.wire_dtype is not None and buf.dtype != self.wire_dtype:
                w = self._scratch.get(("wire", id(buf)))
                if w is None or w.numel() != buf.numel():
                    w = torch.empty_like(buf, dtype=self.wire_dtype)
                    self._scratch[("wire", id(buf))] = w
                w.copy_(buf)
                src = w
            n = src.numel()
            # pad to a multiple of 4*world: equal shards, 4-elem
            # vectorized segsum
            q = 4 * world
            pn = ((n + q - 1) // q) * q
            work, recv, shard = self._buffers(id(buf), pn, src.dtype,
                                              src.device)
            work[:n].copy_(src.view(-1))
            if pn > n:
                work[n:].zero_()
            # phase 1: all-to-all (grouped RCCL send/recv, all links)
            dist.all_to_all_single(recv, work, group=self.group)
            # phase 2: local segment-sum (N14)
            if work.is_cuda:
                from theanompi_amd import ops as _ops
                _ops.ext().segsum(recv, shard, world)
            else:
                torch.sum(recv.view(world, -1), dim=0, out=shard)
            # phase 3: all-gather the reduced shards
            dist.all_gather_into_tensor(work, shard, group=self.group)
            buf.view(-1).copy_(work[:n])
            if scale is not None:
                buf.mul_(scale)


class ExchAsaXgmiBf16(ExchAsaXgmi):
    name = "asa_xgmi_bf16"
    wire_dtype = torch.bfloat16


_STRATEGIES = {
    "ar": ExchAllreduceHost,
    "rccl32": ExchRccl32,
    "nccl32": ExchRccl32,
    "rccl_bf16": ExchRcclBf16,
    "nccl16": ExchRcclBf16,
    "bf16": ExchRcclBf16,
    "asa": ExchAsa,
    "asa32": ExchAsa,
    "asa_bf16": ExchAsaBf16,
    "asa16": ExchAsaBf16,
    "asa_xgmi": ExchAsaXgmi,
    "asa_xgmi_bf16": ExchAsaXgmiBf16,
}


def make_strategy(name, process=None, group=None):
    try:
        cls = _STRATEGIES[name.lower()]
    except KeyError:
        raise ValueError(
            f"unknown exchange strategy {name!r}; "
            f"choose from {sorted(set(_STRATEGIES))}") from None
    return cls(process, group)
