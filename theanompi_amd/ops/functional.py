"""Functional ops: CPU reference numerics + GPU autograd wrappers.

Conventions (both paths):
  * images are NHWC, channels padded to CPAD (16; stems 4) — see
    ``cpad_for``;
  * GPU compute dtype is bf16 with fp32 master weights / statistics;
    the CPU path runs fp32 and defines the semantics the HIP kernels in
    ``ops/hip/`` are tested against (tests/test_kernels_gpu.py).

The GPU conv/linear Functions live in models/layers.py next to their
layout state; this module holds the stateless ops (pool, LRN, BN,
dropout, relu, add, softmax-xent) and the CPU references for everything,
including conv (reference: the ops of Theano-MPI's layer library
``theanompi/models/layers2.py`` [pub]).
"""

import torch
import torch.nn.functional as F

from theanompi_amd import ops as _ops

# ---------------------------------------------------------------------------
# channel padding helpers
# ---------------------------------------------------------------------------

CPAD = 16


def cpad_for(c):
    """Physical channel count for a logical channel count."""
    if c <= 4:
        return 4
    return (c + CPAD - 1) // CPAD * CPAD


def pad_channels(x_nhwc, c_phys):
    """Zero-pad the last dim of an NHWC tensor to c_phys."""
    c = x_nhwc.shape[-1]
    if c == c_phys:
        return x_nhwc
    return F.pad(x_nhwc, (0, c_phys - c))


# ---------------------------------------------------------------------------
# CPU reference implementations (fp32, NHWC in/out)
# ---------------------------------------------------------------------------

def conv2d_ref(x, w_kcrs, bias, stride=(1, 1), pad=(0, 0), groups=1,
               relu=False):
    """x: [N,H,W,C] ; w: [K,C/G,R,S] ; returns [N,P,Q,K]."""
    xc = x.permute(0, 3, 1, 2).contiguous()
    y = F.conv2d(xc, w_kcrs, bias, stride=stride, padding=pad, groups=groups)
    y = y.permute(0, 2, 3, 1).contiguous()
    return F.relu(y) if relu else y


def linear_ref(x, w_in_out, bias, relu=False):
    """x: [B, IN]; w: [IN, OUT] (Theano layout)."""
    y = x @ w_in_out
    if bias is not None:
        y = y + bias
    return F.relu(y) if relu else y


def maxpool2d_ref(x, kernel, stride, pad=(0, 0)):
    xc = x.permute(0, 3, 1, 2)
    y = F.max_pool2d(xc, kernel, stride, pad)
    return y.permute(0, 2, 3, 1).contiguous()


def avgpool2d_ref(x, kernel, stride, pad=(0, 0)):
    xc = x.permute(0, 3, 1, 2)
    y = F.avg_pool2d(xc, kernel, stride, pad, count_include_pad=True)
    return y.permute(0, 2, 3, 1).contiguous()


def lrn_ref(x, size=5, alpha=1e-4, beta=0.75, k=1.0):
    """Cross-map LRN, AlexNet-style:  y = x * (k + alpha/size * sum x^2)^-beta
    over a window of `size` channels centred on c (clamped at the edges).
    Matches theano dnn LRN / F.local_response_norm semantics."""
    xc = x.permute(0, 3, 1, 2)
    y = F.local_response_norm(xc, size=size, alpha=alpha, beta=beta, k=k)
    return y.permute(0, 2, 3, 1).contiguous()


def batchnorm_ref(x, gamma, beta, running_mean, running_var, training,
                  momentum=0.1, eps=1e-5, relu=False):
    xc = x.permute(0, 3, 1, 2)
    y = F.batch_norm(xc, running_mean, running_var, gamma, beta,
                     training=training, momentum=momentum, eps=eps)
    y = y.permute(0, 2, 3, 1).contiguous()
    return F.relu(y) if relu else y


def dropout_ref(x, p, training):
    return F.dropout(x, p=p, training=training)


def softmax_xent_ref(logits, labels, n_classes=None):
    """logits [B, Cp] (maybe channel-padded), labels int64 [B].
    Returns (mean loss, probs [B, Cp]).  Padded columns are masked."""
    if n_classes is not None and n_classes < logits.shape[1]:
        mask = torch.full_like(logits, float("-inf"))
        mask[:, :n_classes] = 0.0
        logits = logits + mask
    probs = F.softmax(logits.float(), dim=1)
    loss = F.cross_entropy(logits.float(), labels)
    return loss, probs


# ---------------------------------------------------------------------------
# GPU autograd wrappers (stateless ops). Each calls theanompi_amd._C.
# ---------------------------------------------------------------------------

class _MaxPool2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, pad):
        C = _ops.ext()
        kh, kw = kernel
        sh, sw = stride
        ph, pw = pad
        xp = C.halo_pad(x, ph, pw, float("-inf")) if (ph or pw) else x
        n, hp, wp, c = xp.shape
        p = (hp - kh) // sh + 1
        q = (wp - kw) // sw + 1
        y = torch.empty((n, p, q, c), dtype=x.dtype, device=x.device)
        argmax = torch.empty((n, p, q, c), dtype=torch.uint8, device=x.device)
        C.maxpool_fwd(xp, y, argmax, kh, kw, sh, sw)
        ctx.save_for_backward(argmax)
        ctx.meta = (x.shape, kernel, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (argmax,) = ctx.saved_tensors
        xshape, kernel, stride, pad = ctx.meta
        C = _ops.ext()
        # maxpool_bwd_k writes every dx element - no pre-zero needed
        dx = torch.empty(xshape, dtype=dy.dtype, device=dy.device)
        C.maxpool_bwd(dy.contiguous(), argmax, dx,
                      kernel[0], kernel[1], stride[0], stride[1],
                      pad[0], pad[1])
        return dx, None, None, None


class _AvgPool2dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, kernel, stride, pad):
        C = _ops.ext()
        kh, kw = kernel
        sh, sw = stride
        ph, pw = pad
        xp = C.halo_pad(x, ph, pw, 0.0) if (ph or pw) else x
        n, hp, wp, c = xp.shape
        p = (hp - kh) // sh + 1
        q = (wp - kw) // sw + 1
        y = torch.empty((n, p, q, c), dtype=x.dtype, device=x.device)
        C.avgpool_fwd(xp, y, kh, kw, sh, sw)
        ctx.meta = (x.shape, kernel, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        xshape, kernel, stride, pad = ctx.meta
        C = _ops.ext()
        dx = torch.empty(xshape, dtype=dy.dtype, device=dy.device)
        C.avgpool_bwd(dy.contiguous(), dx, kernel[0], kernel[1],
                      stride[0], stride[1], pad[0], pad[1])
        return dx, None, None, None


class _LRNFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, size, alpha, beta, k):
        C = _ops.ext()
        y = torch.empty_like(x)
        # 1/denom in bf16: halves the save traffic, and backward
        # reconstructs x from y so only (y, scale) need saving at all
        scale = torch.empty(x.shape, dtype=torch.bfloat16,
                            device=x.device)
        C.lrn_fwd(x, y, scale, size, alpha, beta, k)   # scale = 1/denom
        ctx.save_for_backward(y, scale)
        ctx.meta = (size, alpha, beta, k)
        return y

    @staticmethod
    def backward(ctx, dy):
        y, scale = ctx.saved_tensors
        size, alpha, beta, k = ctx.meta
        C = _ops.ext()
        dx = torch.empty_like(y)
        C.lrn_bwd(dy.contiguous(), y, scale, dx, size, alpha, beta, k)
        return dx, None, None, None, None


class _BatchNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var,
                training, momentum, eps, relu, prestats=None,
                residual=None):
        # residual: fused residual join y = [relu](bn(x) + residual) —
        # one pass instead of bn_apply + add_relu (ResNet bottleneck
        # exits; the separate adds were ~8% of its step)
        C = _ops.ext()
        y = torch.empty_like(x)
        c = x.shape[-1]
        save_mean = torch.empty(c, dtype=torch.float32, device=x.device)
        save_invstd = torch.empty(c, dtype=torch.float32, device=x.device)
        C.bn_fwd(x, gamma, beta, y, save_mean, save_invstd,
                 running_mean, running_var, float(momentum), float(eps),
                 bool(training), bool(relu), prestats,
                 residual.contiguous() if residual is not None else None)
        ctx.save_for_backward(x, gamma, beta, save_mean, save_invstd, y)
        ctx.relu = relu
        ctx.training = training
        ctx.has_res = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, beta, save_mean, save_invstd, y = ctx.saved_tensors
        C = _ops.ext()
        dx = torch.empty_like(x)
        dgamma = torch.empty_like(gamma)
        dbeta = torch.empty_like(gamma)
        dy = dy.contiguous()
        dres = None
        if ctx.has_res and ctx.relu:
            # masked dy comes out of the same kernel pass
            dres = torch.empty_like(dy)
        C.bn_bwd(dy, x, gamma, beta, save_mean,
                 save_invstd, y, dx, dgamma, dbeta, bool(ctx.relu),
                 dres, bool(ctx.has_res))
        if ctx.has_res and not ctx.relu:
            dres = dy              # pass-through, no kernel needed
        return (dx, dgamma, dbeta, None, None, None, None, None, None,
                None, dres)


class _ReluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        C = _ops.ext()
        y = torch.empty_like(x)
        C.relu_fwd(x, y)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        C = _ops.ext()
        dx = torch.empty_like(dy)
        C.relu_bwd(dy.contiguous(), y, dx)
        return dx


class _AddFn(torch.autograd.Function):
    """Residual add (+ optional relu) for ResNet."""
    @staticmethod
    def forward(ctx, a, b, relu):
        C = _ops.ext()
        y = torch.empty_like(a)
        C.add_relu(a, b, y, bool(relu))
        ctx.relu = relu
        if relu:
            ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        if ctx.relu:
            (y,) = ctx.saved_tensors
            C = _ops.ext()
            dx = torch.empty_like(dy)
            C.relu_bwd(dy.contiguous(), y, dx)
            return dx, dx, None
        return dy, dy, None


class _DropoutFn(torch.autograd.Function):
    _seed_buf = {}

    @staticmethod
    def _seed(device):
        t = _DropoutFn._seed_buf.get(device)
        if t is None:
            t = torch.tensor([12345], dtype=torch.int64, device=device)
            _DropoutFn._seed_buf[device] = t
        return t

    @staticmethod
    def forward(ctx, x, p):
        C = _ops.ext()
        y = torch.empty_like(x)
        mask = torch.empty(x.shape, dtype=torch.uint8, device=x.device)
        # device-resident seed (bumped in-kernel) keeps masks fresh under
        # hipGraph replay
        C.dropout_fwd(x, y, mask, float(p), _DropoutFn._seed(x.device))
        ctx.save_for_backward(mask)
        ctx.p = p
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        C = _ops.ext()
        dx = torch.empty_like(dy)
        C.dropout_bwd(dy.contiguous(), mask, dx, float(ctx.p))
        return dx, None


class _SoftmaxXentFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, n_classes):
        C = _ops.ext()
        b = logits.shape[0]
        probs = torch.empty_like(logits)
        loss = torch.empty(b, dtype=torch.float32, device=logits.device)
        # fused batch reductions: stats = [sum(loss), count(top1 err)],
        # accumulated by the kernel's per-row atomics — no torch
        # mean/argmax glue in the train step
        stats = torch.zeros(2, dtype=torch.float32, device=logits.device)
        C.softmax_xent_fwd(logits, labels, probs, loss, int(n_classes),
                           stats)
        ctx.save_for_backward(probs, labels)
        ctx.n_classes = n_classes
        sm = stats * (1.0 / b)
        # probs/err are inference-only outputs: backward below only
        # accounts for dloss, so autograd must not route gradients
        # through them silently
        ctx.mark_non_differentiable(probs)
        err = sm[1]
        ctx.mark_non_differentiable(err)
        return sm[0], probs, err

    @staticmethod
    def backward(ctx, dloss, dprobs, derr):
        probs, labels = ctx.saved_tensors
        C = _ops.ext()
        dlogits = torch.empty_like(probs)
        scale = 1.0 / probs.shape[0]
        C.softmax_xent_bwd(probs, labels, dlogits, scale, int(ctx.n_classes))
        if dloss.numel() == 1:
            dlogits = dlogits * dloss.to(dlogits.dtype)  # no host sync
        return dlogits, None, None


# ---------------------------------------------------------------------------
# public dispatchers
# ---------------------------------------------------------------------------

def _pair(v):
    return (v, v) if isinstance(v, int) else tuple(v)


def maxpool2d(x, kernel, stride, pad=0):
    kernel, stride, pad = _pair(kernel), _pair(stride), _pair(pad)
    if x.is_cuda:
        return _MaxPool2dFn.apply(x, kernel, stride, pad)
    return maxpool2d_ref(x, kernel, stride, pad)


def avgpool2d(x, kernel, stride, pad=0):
    kernel, stride, pad = _pair(kernel), _pair(stride), _pair(pad)
    if x.is_cuda:
        return _AvgPool2dFn.apply(x, kernel, stride, pad)
    return avgpool2d_ref(x, kernel, stride, pad)


def lrn(x, size=5, alpha=1e-4, beta=0.75, k=1.0):
    if x.is_cuda:
        return _LRNFn.apply(x, size, alpha, beta, k)
    return lrn_ref(x, size, alpha, beta, k)


def batchnorm(x, gamma, beta, running_mean, running_var, training,
              momentum=0.1, eps=1e-5, relu=False, prestats=None,
              residual=None):
    if x.is_cuda:
        return _BatchNormFn.apply(x, gamma, beta, running_mean,
                                  running_var, training, momentum, eps,
                                  relu, prestats, residual)
    y = batchnorm_ref(x, gamma.float(), beta.float(), running_mean,
                      running_var, training, momentum, eps, relu=False)
    if residual is not None:
        y = y + residual
    return F.relu(y) if relu else y


def relu(x):
    if x.is_cuda:
        return _ReluFn.apply(x)
    return F.relu(x)


def add(a, b, relu=False):
    if a.is_cuda:
        return _AddFn.apply(a, b, relu)
    y = a + b
    return F.relu(y) if relu else y


def dropout(x, p, training):
    if not training or p == 0.0:
        return x
    if x.is_cuda:
        return _DropoutFn.apply(x, p)
    return F.dropout(x, p=p, training=True)


def softmax_xent(logits, labels, n_classes=None):
    """Fused softmax + cross-entropy.  Returns (mean_loss, probs)."""
    n_classes = n_classes or logits.shape[1]
    if logits.is_cuda:
        loss, probs, _err = _SoftmaxXentFn.apply(logits, labels, n_classes)
        return loss, probs
    return softmax_xent_ref(logits, labels, n_classes)


def softmax_xent_err(logits, labels, n_classes=None):
    """softmax_xent plus the fused top-1 training error: on GPU the
    loss mean AND the err count ride the xent kernel's atomics, so the
    train step needs no torch argmax/mean glue.  Returns
    (mean_loss, probs, err)."""
    n_classes = n_classes or logits.shape[1]
    if logits.is_cuda:
        return _SoftmaxXentFn.apply(logits, labels, n_classes)
    loss, probs = softmax_xent_ref(logits, labels, n_classes)
    with torch.no_grad():
        pred = logits[:, :n_classes].argmax(dim=1)
        err = (pred != labels).float().mean()
    return loss, probs, err
