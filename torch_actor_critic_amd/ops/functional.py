"""Dispatching functional ops: fused HIP kernels on GPU, eager PyTorch on CPU.

Each op has an eager composite that defines the numerics contract (and is
the fp32 reference the GPU parity tests compare against) and a native path
backed by hand-written gfx950 kernels via custom autograd Functions.

Numerics parity targets in the reference implementation:
  * linear+relu trunk          — reference networks/linear.py:32-35
  * tanh-Gaussian head         — reference networks/linear.py:37-53
  * twin-Q Bellman backup/MSE  — reference sac/algorithm.py:46-74
  * policy loss                — reference sac/algorithm.py:30-43
  * polyak target update       — reference sac/algorithm.py:77-81
"""

import math

import torch
import torch.nn.functional as F

from . import use_native, require_extension

_LOG_2PI = math.log(2.0 * math.pi)
_2LOG2 = 2.0 * math.log(2.0)

_compute_dtype = "fp32"


def set_compute_dtype(dtype: str):
    """'bf16' switches GEMM kernels to bf16 MFMA inputs (fp32 accumulate,
    fp32 master weights); 'fp32' uses the exact f32 MFMA path."""
    global _compute_dtype
    assert dtype in ("fp32", "bf16")
    _compute_dtype = dtype
    from . import extension
    ext = extension()
    if ext is not None:
        ext.set_compute_bf16(dtype == "bf16")


def get_compute_dtype() -> str:
    return _compute_dtype


# per-device Philox noise state (graph-replay-safe rsample noise)
_philox_ctr = {}
_philox_seed = 0


def set_philox_seed(seed: int):
    global _philox_seed
    _philox_seed = int(seed)
    _philox_ctr.clear()


def randn_like_philox(t: torch.Tensor) -> torch.Tensor:
    """Standard-normal noise; on GPU drawn by the Philox kernel with a
    device-side counter so a captured hipGraph replays fresh noise."""
    if use_native(t):
        dev = t.device
        if dev not in _philox_ctr:
            _philox_ctr[dev] = torch.zeros(1, dtype=torch.int64, device=dev)
        out = torch.empty_like(t)
        require_extension().philox_randn_(out, _philox_ctr[dev],
                                          _philox_seed)
        return out
    return torch.randn_like(t)


# ---------------------------------------------------------------------------
# Graph-update optimizations (activated by GraphedSACUpdate only):
#  * cached transposed weights — backward dgrads read a cache entry that
#    the update refreshes ONCE per phase after its Adam step, instead of
#    transposing per backward call;
#  * direct wgrad — weight/bias gradients are written straight into the
#    .grad views of the flat-gradient buffer (each layer receives exactly
#    one contribution per captured update, so overwrite == accumulate
#    from zero), eliminating autograd's per-parameter accumulate-add
#    kernels (~46/update on the visual critic).
# Both are scoped: only active between set_graph_opt(cache, True) and
# set_graph_opt(None, False), i.e. during graph warmup + capture.
# ---------------------------------------------------------------------------

_graph_wt_cache = None   # data_ptr(w) -> (w, wt, block)
_graph_direct_wgrad = False


def set_graph_opt(wt_cache, direct_wgrad: bool):
    global _graph_wt_cache, _graph_direct_wgrad
    _graph_wt_cache = wt_cache
    _graph_direct_wgrad = direct_wgrad


def _dense_wt(ext, w):
    """Transposed dense weight [K,N] for dgrad, cached when active."""
    c = _graph_wt_cache
    if c is not None:
        ent = c.get(w.data_ptr())
        if ent is not None:
            return ent[1]
    wt = torch.empty(w.shape[1], w.shape[0], device=w.device,
                     dtype=w.dtype)
    ext.transpose_multi([w], [wt])
    if c is not None:
        c[w.data_ptr()] = (w, wt, 1)
    return wt


def _conv_wt(ext, w):
    """Conv dgrad weight layout [IC, OC*kh*kw], cached when active."""
    oc, ic, kh, kw = w.shape
    c = _graph_wt_cache
    if c is not None:
        ent = c.get(w.data_ptr())
        if ent is not None:
            return ent[1]
        wt = torch.empty(ic, oc * kh * kw, device=w.device, dtype=w.dtype)
        ext.transpose_multi([w], [wt], [kh * kw])
        c[w.data_ptr()] = (w, wt, kh * kw)
        return wt
    return w.permute(1, 0, 2, 3).reshape(ic, oc * kh * kw).contiguous()


def refresh_wt_cache(cache, weights):
    """Refresh the cached transposed layouts of `weights` (those already
    registered) in batched transpose_multi launches — called right after
    the owning optimizer's Adam step, inside the captured graph."""
    ext = require_extension()
    ws, wts, bss = [], [], []
    for w in weights:
        ent = cache.get(w.data_ptr())
        if ent is not None:
            ws.append(ent[0])
            wts.append(ent[1])
            bss.append(ent[2])
    for i in range(0, len(ws), 12):
        ext.transpose_multi(ws[i:i + 12], wts[i:i + 12], bss[i:i + 12])


def _direct_outs(pairs):
    """(w.grad, b.grad) buffers for in-place wgrad writes, or None when
    the direct path is off / a .grad view is missing."""
    if not _graph_direct_wgrad:
        return None
    outs = []
    for w, b in pairs:
        # frozen weights (e.g. the critic during the policy phase) must
        # NOT have their phase-1 gradients clobbered — fall back to the
        # allocating path whose outputs autograd discards
        if not w.requires_grad or w.grad is None or \
                (b is not None and (not b.requires_grad or b.grad is None)):
            return None
        outs.append((w.grad, b.grad if b is not None else None))
    return outs


# ---------------------------------------------------------------------------
# Linear (+ optional ReLU)
# ---------------------------------------------------------------------------

class _NativeLinear(torch.autograd.Function):
    """y = x @ w^T + b, optional fused ReLU — MFMA GEMM on gfx950."""

    @staticmethod
    def forward(ctx, x, w, b, relu):
        ext = require_extension()
        x = x.contiguous()
        M, K = x.shape
        N = w.shape[0]
        y = torch.empty(M, N, device=x.device, dtype=x.dtype)
        # pipelined multi-problem GEMM kernel (fused.hip), single problem
        ext.mgemm([x], [w], [b], [y], [None], M, N, K, K, N, relu,
                  [], [], [], 0, 0, 0, [])
        ctx.save_for_backward(x, w, y, b)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y, b = ctx.saved_tensors
        ext = require_extension()
        dy = dy.contiguous()
        M, K = x.shape
        N = w.shape[0]
        ymask = y if ctx.relu else None
        dw = db = None
        if ctx.needs_input_grad[1]:
            outs = _direct_outs([(w, b)])
            if outs is None:
                dw = torch.empty_like(w)
                db = torch.empty(N, device=w.device, dtype=w.dtype)
            else:
                dw, db = outs[0]
                if db is None:
                    db = torch.empty(0, device=w.device, dtype=w.dtype)
            # pipelined coalesced wgrad (split-M at large batch) +
            # fused db
            ext.mwgrad([dy], [ymask], [x], [dw], [db], M, N, K, N, K, 0)
            if outs is not None:
                dw = db = None
        dx = None
        if ctx.needs_input_grad[0]:
            wt = _dense_wt(ext, w)
            dx = torch.empty_like(x)
            ext.mgemm([dy], [wt], [None], [dx], [ymask], M, K, N, N, K,
                      False, [], [], [], 0, 0, 0, [])
        return dx, dw, db, None


def linear_relu(x, w, b, relu: bool = True):
    if use_native(x, w):
        return _NativeLinear.apply(x, w, b, relu)
    y = F.linear(x, w, b)
    return F.relu(y) if relu else y


def mlp_forward(x, layers, relu_last: bool = True):
    """Run x through a list of nn.Linear, ReLU after every layer (or every
    layer but the last when relu_last=False — the critic trunk shape,
    reference networks/linear.py:61-67)."""
    n = len(layers)
    for i, layer in enumerate(layers):
        relu = relu_last or (i + 1 < n)
        x = linear_relu(x, layer.weight, layer.bias, relu=relu)
    return x


# ---------------------------------------------------------------------------
# Conv2d (implicit-GEMM MFMA; valid padding, square stride — the visual
# trunk's shape family, reference networks/convolutional.py:30-51)
# ---------------------------------------------------------------------------


def _conv_mask_in_kernel() -> bool:
    """TAC_AMD_CONV_MASK=1: apply the relu mask inside the conv
    dgrad/wgrad gathers instead of a threshold_backward launch per dy
    (re-A/B'd after the stride-class dgrad)."""
    import os
    return os.environ.get("TAC_AMD_CONV_MASK") == "1"


class _NativeConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, stride, relu):
        ext = require_extension()
        y = ext.conv2d_fwd(x.contiguous(), w.contiguous(),
                           b.contiguous() if b is not None else None,
                           stride, relu)
        if relu:
            ctx.save_for_backward(x, w, y, b)
        else:
            ctx.save_for_backward(x, w, b)
        ctx.stride = stride
        ctx.relu = relu
        ctx.has_bias = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_extension()
        dy = dy.contiguous()
        m = None
        if ctx.relu:
            x, w, y, b = ctx.saved_tensors
            if _conv_mask_in_kernel():
                m = y
            else:
                # one elementwise relu-backward instead of masked
                # gathers inside the conv kernels (A/B, default)
                dy = torch.ops.aten.threshold_backward(dy, y, 0)
        else:
            x, w, b = ctx.saved_tensors
        dx = None
        if ctx.needs_input_grad[0]:
            wt = _conv_wt(ext, w)
            dx = ext.conv2d_dgrad(dy, m, wt, x, w, ctx.stride)
        dw = db = None
        if ctx.needs_input_grad[1]:
            outs = _direct_outs([(w, b)]) if ctx.has_bias else None
            if outs is not None:
                ext.conv2d_wgrad_multi([dy], [m], [x], w, ctx.stride,
                                       out=[outs[0][0], outs[0][1]])
            else:
                dw, db = ext.conv2d_wgrad(dy, m, x, w, ctx.stride)
                if not ctx.has_bias:
                    db = None
        return dx, dw, db, None, None


def conv2d(x, w, b, stride: int, relu: bool = False):
    """Valid-padding conv with square stride (the reference CNN family),
    optionally with the trailing ReLU fused into the kernel."""
    if use_native(x, w):
        return _NativeConv2d.apply(x, w, b, int(stride), relu)
    y = F.conv2d(x, w, b, stride=stride)
    return F.relu(y) if relu else y


# ---------------------------------------------------------------------------
# Paired twin-critic ops: both critics' identically-shaped layers run in
# ONE launch (blockIdx.z), forward and backward (the visual DoubleCritic
#runs lockstep through these)
# ---------------------------------------------------------------------------

class _PairedLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x1, x2, w1, b1, w2, b2, relu):
        ext = require_extension()
        x1 = x1.contiguous()
        x2 = x2.contiguous()
        M, K = x1.shape
        N = w1.shape[0]
        y1 = torch.empty(M, N, device=x1.device, dtype=x1.dtype)
        y2 = torch.empty_like(y1)
        ext.mgemm([x1, x2], [w1, w2], [b1, b2], [y1, y2], [None, None],
                  M, N, K, K, N, relu, [], [], [], 0, 0, 0, [])
        ctx.save_for_backward(x1, x2, w1, w2, y1, y2, b1, b2)
        ctx.relu = relu
        ctx.shared_x = x1 is x2
        return y1, y2

    @staticmethod
    def backward(ctx, dy1, dy2):
        x1, x2, w1, w2, y1, y2, b1, b2 = ctx.saved_tensors
        ext = require_extension()
        dy1 = dy1.contiguous()
        dy2 = dy2.contiguous()
        M, K = x1.shape
        N = w1.shape[0]
        m1 = y1 if ctx.relu else None
        m2 = y2 if ctx.relu else None
        dw1 = dw2 = db1 = db2 = None
        if ctx.needs_input_grad[2]:
            outs = _direct_outs([(w1, b1), (w2, b2)])
            if outs is None:
                dw1 = torch.empty_like(w1)
                dw2 = torch.empty_like(w2)
                db1 = torch.empty(N, device=w1.device, dtype=w1.dtype)
                db2 = torch.empty_like(db1)
                gws = (dw1, db1, dw2, db2)
            else:
                (gw1, gb1), (gw2, gb2) = outs
                if gb1 is None:
                    gb1 = torch.empty(0, device=w1.device, dtype=w1.dtype)
                    gb2 = torch.empty_like(gb1)
                gws = (gw1, gb1, gw2, gb2)
            ext.mwgrad([dy1, dy2], [m1, m2], [x1, x2],
                       [gws[0], gws[2]], [gws[1], gws[3]],
                       M, N, K, N, K, 0)
        dx1 = dx2 = None
        if ctx.needs_input_grad[0]:
            wt1 = _dense_wt(ext, w1)
            wt2 = _dense_wt(ext, w2)
            dx1 = torch.empty_like(x1)
            if ctx.shared_x:
                # both critics share the input: dx = dy1@wt1 + dy2@wt2
                # in ONE sum2 GEMM launch; autograd sees dx2=None so no
                # accumulate-add kernel runs
                ext.mgemm([dy1], [wt1], [None], [dx1], [m1], M, K, N, N,
                          K, False, [dy2], [wt2], [m2], N, 0, 0, [])
            else:
                dx2 = torch.empty_like(x2)
                ext.mgemm([dy1, dy2], [wt1, wt2], [None, None],
                          [dx1, dx2], [m1, m2], M, K, N, N, K, False,
                          [], [], [], 0, 0, 0, [])
        return dx1, dx2, dw1, db1, dw2, db2, None


def linear_pair(x1, x2, w1, b1, w2, b2, relu: bool):
    if use_native(x1, w1):
        return _PairedLinear.apply(x1, x2, w1, b1, w2, b2, relu)
    y1 = F.linear(x1, w1, b1)
    y2 = F.linear(x2, w2, b2)
    if relu:
        y1, y2 = F.relu(y1), F.relu(y2)
    return y1, y2


class _PairedConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x1, x2, w1, b1, w2, b2, stride, relu):
        ext = require_extension()
        y1, y2 = ext.conv2d_fwd_multi(
            [x1.contiguous(), x2.contiguous()],
            [w1.contiguous(), w2.contiguous()],
            [b1.contiguous() if b1 is not None else None,
             b2.contiguous() if b2 is not None else None], stride, relu)
        if relu:
            ctx.save_for_backward(x1, x2, w1, w2, y1, y2, b1, b2)
        else:
            ctx.save_for_backward(x1, x2, w1, w2, b1, b2)
        ctx.stride = stride
        ctx.relu = relu
        ctx.has_bias = b1 is not None
        return y1, y2

    @staticmethod
    def backward(ctx, dy1, dy2):
        ext = require_extension()
        dy1 = dy1.contiguous()
        dy2 = dy2.contiguous()
        m1 = m2 = None
        if ctx.relu:
            x1, x2, w1, w2, y1, y2, b1, b2 = ctx.saved_tensors
            if _conv_mask_in_kernel():
                m1, m2 = y1, y2
            else:
                dy1 = torch.ops.aten.threshold_backward(dy1, y1, 0)
                dy2 = torch.ops.aten.threshold_backward(dy2, y2, 0)
        else:
            x1, x2, w1, w2, b1, b2 = ctx.saved_tensors
        dx1 = dx2 = None
        if ctx.needs_input_grad[0]:
            wt1 = _conv_wt(ext, w1)
            wt2 = _conv_wt(ext, w2)
            dx1, dx2 = ext.conv2d_dgrad_multi([dy1, dy2], [m1, m2],
                                              [wt1, wt2], x1, w1,
                                              ctx.stride)
        dw1 = db1 = dw2 = db2 = None
        if ctx.needs_input_grad[2]:
            outs = (_direct_outs([(w1, b1), (w2, b2)]) if ctx.has_bias
                    else None)
            if outs is not None:
                ext.conv2d_wgrad_multi(
                    [dy1, dy2], [m1, m2], [x1, x2], w1, ctx.stride,
                    out=[outs[0][0], outs[0][1], outs[1][0], outs[1][1]])
            else:
                dw1, db1, dw2, db2 = ext.conv2d_wgrad_multi(
                    [dy1, dy2], [m1, m2], [x1, x2], w1, ctx.stride)
                if not ctx.has_bias:
                    db1 = db2 = None
        return dx1, dx2, dw1, db1, dw2, db2, None, None


def conv2d_pair(x1, x2, w1, b1, w2, b2, stride: int, relu: bool):
    if use_native(x1, w1):
        return _PairedConv2d.apply(x1, x2, w1, b1, w2, b2, int(stride),
                                   relu)
    y1 = F.conv2d(x1, w1, b1, stride=stride)
    y2 = F.conv2d(x2, w2, b2, stride=stride)
    if relu:
        y1, y2 = F.relu(y1), F.relu(y2)
    return y1, y2


# ---------------------------------------------------------------------------
# Quad ops: target-twin AND live-twin critic layers in ONE 4-problem
# launch (the critic phase runs both stacks on identical shapes — same
# trick as the MLP engine's 4-problem GEMMs, algo/engine.py).  Problems
# [0,1] are the target critics: non-differentiable, no backward work.
# ---------------------------------------------------------------------------

class _QuadLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xt1, xt2, x1, x2, wt1, bt1, wt2, bt2,
                w1, b1, w2, b2, relu):
        ctx.set_materialize_grads(False)  # target outputs get dy=None
        ext = require_extension()
        xt1, xt2 = xt1.contiguous(), xt2.contiguous()
        x1c, x2c = x1.contiguous(), x2.contiguous()
        M, K = x1c.shape
        N = w1.shape[0]
        ys = [torch.empty(M, N, device=x1c.device, dtype=x1c.dtype)
              for _ in range(4)]
        ext.mgemm([xt1, xt2, x1c, x2c], [wt1, wt2, w1, w2],
                  [bt1, bt2, b1, b2], ys, [None] * 4,
                  M, N, K, K, N, relu, [], [], [], 0, 0, 0, [])
        ctx.save_for_backward(x1c, x2c, w1, w2, ys[2], ys[3], b1, b2)
        ctx.relu = relu
        ctx.shared_x = x1c is x2c
        ctx.mark_non_differentiable(ys[0], ys[1])
        return ys[0], ys[1], ys[2], ys[3]

    @staticmethod
    def backward(ctx, dyt1, dyt2, dy1, dy2):
        x1, x2, w1, w2, y1, y2, b1, b2 = ctx.saved_tensors
        ext = require_extension()
        dy1 = dy1.contiguous()
        dy2 = dy2.contiguous()
        M, K = x1.shape
        N = w1.shape[0]
        m1 = y1 if ctx.relu else None
        m2 = y2 if ctx.relu else None
        dw1 = dw2 = db1 = db2 = None
        if ctx.needs_input_grad[8]:
            outs = _direct_outs([(w1, b1), (w2, b2)])
            if outs is None:
                dw1 = torch.empty_like(w1)
                dw2 = torch.empty_like(w2)
                db1 = torch.empty(N, device=w1.device, dtype=w1.dtype)
                db2 = torch.empty_like(db1)
                gws = (dw1, db1, dw2, db2)
            else:
                (gw1, gb1), (gw2, gb2) = outs
                if gb1 is None:
                    gb1 = torch.empty(0, device=w1.device, dtype=w1.dtype)
                    gb2 = torch.empty_like(gb1)
                gws = (gw1, gb1, gw2, gb2)
            ext.mwgrad([dy1, dy2], [m1, m2], [x1, x2],
                       [gws[0], gws[2]], [gws[1], gws[3]],
                       M, N, K, N, K, 0)
        dx1 = dx2 = None
        if ctx.needs_input_grad[2]:
            wt1 = _dense_wt(ext, w1)
            wt2 = _dense_wt(ext, w2)
            dx1 = torch.empty_like(x1)
            if ctx.shared_x:
                ext.mgemm([dy1], [wt1], [None], [dx1], [m1], M, K, N, N,
                          K, False, [dy2], [wt2], [m2], N, 0, 0, [])
            else:
                dx2 = torch.empty_like(x2)
                ext.mgemm([dy1, dy2], [wt1, wt2], [None, None],
                          [dx1, dx2], [m1, m2], M, K, N, N, K, False,
                          [], [], [], 0, 0, 0, [])
        return (None, None, dx1, dx2, None, None, None, None,
                dw1, db1, dw2, db2, None)


def linear_quad(xt1, xt2, x1, x2, t1, t2, l1, l2, relu: bool):
    """(target twin, live twin) linear layers, one 4-problem launch;
    t*/l* are nn.Linear modules."""
    return _QuadLinear.apply(xt1, xt2, x1, x2,
                             t1.weight, t1.bias, t2.weight, t2.bias,
                             l1.weight, l1.bias, l2.weight, l2.bias,
                             relu)


class _QuadConv2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xt1, xt2, x1, x2, wt1, bt1, wt2, bt2,
                w1, b1, w2, b2, stride, relu):
        ctx.set_materialize_grads(False)  # target outputs get dy=None
        ext = require_extension()
        x1c, x2c = x1.contiguous(), x2.contiguous()
        ys = ext.conv2d_fwd_multi(
            [xt1.contiguous(), xt2.contiguous(), x1c, x2c],
            [wt1, wt2, w1, w2], [bt1, bt2, b1, b2], stride, relu)
        if relu:
            ctx.save_for_backward(x1c, x2c, w1, w2, ys[2], ys[3], b1, b2)
        else:
            ctx.save_for_backward(x1c, x2c, w1, w2, b1, b2)
        ctx.stride = stride
        ctx.relu = relu
        ctx.mark_non_differentiable(ys[0], ys[1])
        return ys[0], ys[1], ys[2], ys[3]

    @staticmethod
    def backward(ctx, dyt1, dyt2, dy1, dy2):
        ext = require_extension()
        dy1 = dy1.contiguous()
        dy2 = dy2.contiguous()
        m1 = m2 = None
        if ctx.relu:
            x1, x2, w1, w2, y1, y2, b1, b2 = ctx.saved_tensors
            if _conv_mask_in_kernel():
                m1, m2 = y1, y2
            else:
                dy1 = torch.ops.aten.threshold_backward(dy1, y1, 0)
                dy2 = torch.ops.aten.threshold_backward(dy2, y2, 0)
        else:
            x1, x2, w1, w2, b1, b2 = ctx.saved_tensors
        dx1 = dx2 = None
        if ctx.needs_input_grad[2]:
            wt1 = _conv_wt(ext, w1)
            wt2 = _conv_wt(ext, w2)
            dx1, dx2 = ext.conv2d_dgrad_multi([dy1, dy2], [m1, m2],
                                              [wt1, wt2], x1, w1,
                                              ctx.stride)
        dw1 = db1 = dw2 = db2 = None
        if ctx.needs_input_grad[8]:
            outs = _direct_outs([(w1, b1), (w2, b2)])
            if outs is not None:
                ext.conv2d_wgrad_multi(
                    [dy1, dy2], [m1, m2], [x1, x2], w1, ctx.stride,
                    out=[outs[0][0], outs[0][1], outs[1][0], outs[1][1]])
            else:
                dw1, db1, dw2, db2 = ext.conv2d_wgrad_multi(
                    [dy1, dy2], [m1, m2], [x1, x2], w1, ctx.stride)
        return (None, None, dx1, dx2, None, None, None, None,
                dw1, db1, dw2, db2, None, None)


def conv2d_quad(xt1, xt2, x1, x2, t1, t2, c1, c2, stride: int,
                relu: bool):
    """(target twin, live twin) conv layers, one 4-problem launch;
    t*/c* are conv modules."""
    return _QuadConv2d.apply(xt1, xt2, x1, x2,
                             t1.weight, t1.bias, t2.weight, t2.bias,
                             c1.weight, c1.bias, c2.weight, c2.bias,
                             int(stride), relu)


# ---------------------------------------------------------------------------
# Fused tanh-Gaussian head (sample + squash + log-prob)
# ---------------------------------------------------------------------------

def _eager_tanh_gauss(mu, log_std, eps, act_limit, log_min_std, log_max_std,
                      deterministic, with_logprob):
    log_std = torch.clip(log_std, log_min_std, log_max_std)
    std = torch.exp(log_std)
    prob = mu if deterministic else mu + std * eps
    pi_action = torch.tanh(prob) * act_limit
    logprob = None
    if with_logprob:
        # N(mu, std).log_prob(prob) with prob = mu + std*eps
        gauss = -0.5 * ((prob - mu) / std) ** 2 - log_std - 0.5 * _LOG_2PI
        logprob = gauss.sum(dim=-1)
        # numerically-stable tanh Jacobian correction
        logprob = logprob - (_2LOG2 - prob - F.softplus(-2.0 * prob)).sum(dim=-1)
    return pi_action, logprob


class _NativeTanhGaussHead(torch.autograd.Function):
    @staticmethod
    def forward(ctx, mu, log_std, eps, act_limit, log_min_std, log_max_std,
                deterministic, with_logprob):
        ext = require_extension()
        pi, logp, prob, log_std_c = ext.tanh_gauss_fwd(
            mu, log_std, eps, act_limit, log_min_std, log_max_std,
            deterministic, with_logprob)
        ctx.save_for_backward(mu, log_std, eps, prob, log_std_c)
        ctx.meta = (act_limit, log_min_std, log_max_std, deterministic,
                    with_logprob)
        return pi, (logp if with_logprob else None)

    @staticmethod
    def backward(ctx, dpi, dlogp):
        mu, log_std, eps, prob, log_std_c = ctx.saved_tensors
        act_limit, log_min_std, log_max_std, deterministic, with_logprob = ctx.meta
        ext = require_extension()
        if dlogp is None:
            dlogp = torch.zeros(mu.shape[0], device=mu.device, dtype=mu.dtype)
        dmu, dlog_std = ext.tanh_gauss_bwd(
            dpi.contiguous(), dlogp.contiguous(), mu, log_std, eps, prob,
            log_std_c, act_limit, log_min_std, log_max_std, deterministic,
            with_logprob)
        return dmu, dlog_std, None, None, None, None, None, None


def tanh_gauss_head(mu, log_std, eps, act_limit, log_min_std, log_max_std,
                    deterministic: bool = False, with_logprob: bool = True):
    """Squashed-Gaussian policy head. Returns (pi_action, logprob|None)."""
    if use_native(mu, log_std):
        return _NativeTanhGaussHead.apply(
            mu, log_std, eps, act_limit, log_min_std, log_max_std,
            deterministic, with_logprob)
    return _eager_tanh_gauss(mu, log_std, eps, act_limit, log_min_std,
                             log_max_std, deterministic, with_logprob)


# ---------------------------------------------------------------------------
# Fused SAC losses
# ---------------------------------------------------------------------------

def _eager_q_loss(q1, q2, q1t, q2t, logp_next, rewards, done, alpha, gamma,
                  reward_scale):
    with torch.no_grad():
        q_target = torch.min(q1t, q2t)
        backup = reward_scale * rewards + gamma * (1.0 - done) * (
            q_target - alpha * logp_next)
    return ((q1 - backup) ** 2).mean() + ((q2 - backup) ** 2).mean()


class _NativeQLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q1, q2, q1t, q2t, logp_next, rewards, done, alpha,
                gamma, reward_scale):
        ext = require_extension()
        loss, dq1, dq2 = ext.sac_q_loss_fwd(q1, q2, q1t, q2t, logp_next,
                                            rewards, done, alpha, gamma,
                                            reward_scale)
        ctx.save_for_backward(dq1, dq2)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        dq1, dq2 = ctx.saved_tensors
        return dq1 * dloss, dq2 * dloss, None, None, None, None, None, \
            None, None, None


def sac_q_loss(q1, q2, q1t, q2t, logp_next, rewards, done, alpha, gamma,
               reward_scale):
    """Twin-Q Bellman MSE: loss = mse(q1, backup) + mse(q2, backup) with
    backup = scale*r + gamma*(1-d)*(min(q1t,q2t) - alpha*logp_next)."""
    if use_native(q1, q2):
        return _NativeQLoss.apply(q1, q2, q1t, q2t, logp_next, rewards,
                                  done, alpha, gamma, reward_scale)
    return _eager_q_loss(q1, q2, q1t, q2t, logp_next, rewards, done, alpha,
                         gamma, reward_scale)


def _eager_pi_loss(q1, q2, logp, alpha):
    return (alpha * logp - torch.min(q1, q2)).mean()


class _NativePiLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q1, q2, logp, alpha):
        ext = require_extension()
        loss, dq1, dq2, dlogp = ext.sac_pi_loss_fwd(q1, q2, logp, alpha)
        ctx.save_for_backward(dq1, dq2, dlogp)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        dq1, dq2, dlogp = ctx.saved_tensors
        return dq1 * dloss, dq2 * dloss, dlogp * dloss, None


def sac_pi_loss(q1, q2, logp, alpha):
    """Policy loss: (alpha*logp - min(q1,q2)).mean()."""
    if use_native(q1, q2):
        return _NativePiLoss.apply(q1, q2, logp, alpha)
    return _eager_pi_loss(q1, q2, logp, alpha)


# ---------------------------------------------------------------------------
# Flat-buffer maintenance ops (no autograd)
# ---------------------------------------------------------------------------

def polyak_(flat_target: torch.Tensor, flat_src: torch.Tensor, polyak: float):
    """In-place: target = polyak*target + (1-polyak)*src — ONE kernel over
    the module's whole flattened parameter buffer (the reference loops
    per-parameter, sac/algorithm.py:77-81)."""
    if use_native(flat_target, flat_src):
        require_extension().polyak_(flat_target, flat_src, polyak)
    else:
        flat_target.mul_(polyak).add_(flat_src, alpha=1.0 - polyak)


def adam_step_(p, g, m, v, step_t, lr, beta1, beta2, eps, weight_decay=0.0):
    """Fused Adam over flat buffers. step_t is a device int64 scalar that
    the kernel increments — hipGraph-replay-safe (no host-side state)."""
    if use_native(p, g):
        require_extension().adam_step_(p, g, m, v, step_t, lr, beta1, beta2,
                                       eps, weight_decay)
    else:
        step_t += 1
        step = int(step_t.item())
        if weight_decay != 0.0:
            g = g.add(p, alpha=weight_decay)
        m.mul_(beta1).add_(g, alpha=1.0 - beta1)
        v.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
        bc1 = 1.0 - beta1 ** step
        bc2 = 1.0 - beta2 ** step
        denom = (v / bc2).sqrt_().add_(eps)
        p.addcdiv_(m, denom, value=-lr / bc1)
