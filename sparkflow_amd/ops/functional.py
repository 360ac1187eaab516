"""Autograd wrappers over the CDNA4 HIP kernels (fp32).

Each op is a ``torch.autograd.Function`` whose GPU path calls the gfx950
extension and whose CPU path implements identical fp32 math with plain
torch ops (used by the no-GPU tests; numerics tests compare the HIP path
against this path).

Replaces the reference's implicit TF ops: dense
GEMM+bias+activation, conv2d, maxpool, softmax-cross-entropy, MSE, argmax.
"""
from __future__ import annotations

import contextlib
import threading

import torch

from sparkflow_amd import ops

ACT_NONE, ACT_RELU, ACT_SIGMOID = 0, 1, 2
ACT_CODES = {"none": ACT_NONE, "linear": ACT_NONE, "relu": ACT_RELU,
             "sigmoid": ACT_SIGMOID}

_TLS = threading.local()


def _want_loss_value() -> bool:
    import os

    if os.environ.get("SPARKFLOW_GRAD_ONLY", "1") == "0":
        return True  # debug toggle: always compute the scalar
    return not getattr(_TLS, "grad_only", False)


@contextlib.contextmanager
def grad_only():
    """Inside this context the loss ops may skip computing the scalar loss
    VALUE (it is undefined) while producing exact gradients — used by the
    training hot loop, where the value is never read.  Saves the
    zero-fill / atomic-reduce / mean kernels per step (GPU path only; the
    CPU path always computes the value)."""
    prev = getattr(_TLS, "grad_only", False)
    _TLS.grad_only = True
    try:
        yield
    finally:
        _TLS.grad_only = prev


@contextlib.contextmanager
def defer_wgrad(records: list):
    """Inside this context LinearAct.backward appends
    ``(weight_data_ptr, workspace)`` to ``records`` instead of reducing
    the wgrad split-K workspace to per-tensor grads (the weight/bias
    grads it returns are then None).  The captured step feeds the raw
    workspaces to ``reduce_apply_adam`` — one launch replaces every
    wgrad reduce, pack_segments, and the Adam apply.  GPU biased-Linear
    layers only; other ops ignore the context (the caller checks
    coverage and falls back)."""
    prev = getattr(_TLS, "defer_wgrad", None)
    _TLS.defer_wgrad = records
    try:
        yield
    finally:
        _TLS.defer_wgrad = prev


def _defer_records():
    return getattr(_TLS, "defer_wgrad", None)


def _dev_scale(dloss: torch.Tensor) -> torch.Tensor:
    """Upstream loss grad as a device fp32 scalar for the HIP bwd kernels.

    Kept on-device (the kernel dereferences it) so the backward pass never
    d2h-syncs — a host `float(dloss)` would break hipGraph capture of the
    full training step."""
    return dloss.detach().to(dtype=torch.float32).contiguous()


def _act_fwd(pre: torch.Tensor, act: int) -> torch.Tensor:
    if act == ACT_RELU:
        return torch.relu(pre)
    if act == ACT_SIGMOID:
        return torch.sigmoid(pre)
    return pre


def _act_bwd_from_out(y: torch.Tensor, dy: torch.Tensor, act: int) -> torch.Tensor:
    # activation grads recomputed from the *output* (saves the preactivation)
    if act == ACT_RELU:
        return dy * (y > 0).to(dy.dtype)
    if act == ACT_SIGMOID:
        return dy * y * (1.0 - y)
    return dy


class LinearAct(torch.autograd.Function):
    """y = act(x @ W + b) with x:[M,K], W:[K,N] (TF dense layout), b:[N]."""

    @staticmethod
    def forward(ctx, x, w, b, act: int):
        if ops.use_hip(x):
            y = ops.hip.gemm_act(x, w, b if b is not None else torch.empty(0, device=x.device), act)
        else:
            pre = x @ w
            if b is not None:
                pre = pre + b
            y = _act_fwd(pre, act)
        ctx.save_for_backward(x, w, y)
        ctx.act = act
        ctx.has_bias = b is not None
        # Captured at FORWARD time (defer_wgrad is entered on the thread
        # running the step; autograd executes backward on its own engine
        # thread where the thread-local would be invisible).
        ctx.defer_sink = _defer_records()
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        act = ctx.act
        need_dx = ctx.needs_input_grad[0]
        dy = dy.contiguous()
        if ops.use_hip(x):
            import os

            recs = ctx.defer_sink
            if recs is not None and ctx.has_bias:
                # deferred-reduce mode: hand the raw split-K workspace to
                # the fused reduce+apply; weight/bias grads stay None
                if act != ACT_NONE and \
                        os.environ.get("SPARKFLOW_FUSED_ACTBWD", "1") != "0":
                    dx = ops.hip.gemm_dgrad_act(dy, y, w, act) \
                        if need_dx else None
                    ws = ops.hip.gemm_wgrad_bias_act_ws(x, dy, y, act)
                else:
                    dpre = ops.hip.act_bwd(y, dy, act) \
                        if act != ACT_NONE else dy
                    dx = ops.hip.gemm_dgrad(dpre, w) if need_dx else None
                    ws = ops.hip.gemm_wgrad_bias_ws(x, dpre)
                recs.append((w.data_ptr(), ws))
                return dx, None, None, None
            if act != ACT_NONE and ctx.has_bias and \
                    os.environ.get("SPARKFLOW_FUSED_ACTBWD", "1") != "0":
                # activation backward fused into BOTH backward GEMMs'
                # operand loads — no standalone act_bwd pass or dpre buf
                dx = ops.hip.gemm_dgrad_act(dy, y, w, act) \
                    if need_dx else None
                dwb = ops.hip.gemm_wgrad_bias_act(x, dy, y, act)
                k = x.shape[1]
                dw, db = dwb[:k], dwb[k]
                return dx, dw, db, None
            dpre = ops.hip.act_bwd(y, dy, act) if act != ACT_NONE else dy
            dx = ops.hip.gemm_dgrad(dpre, w) if need_dx else None
            if ctx.has_bias:
                # bias grad rides the wgrad GEMM as a virtual ones-row
                dwb = ops.hip.gemm_wgrad_bias(x, dpre)
                k = x.shape[1]
                dw, db = dwb[:k], dwb[k]
            else:
                dw, db = ops.hip.gemm_wgrad(x, dpre), None
        else:
            dpre = _act_bwd_from_out(y, dy, act)
            dx = dpre @ w.t() if need_dx else None
            dw = x.t() @ dpre
            db = dpre.sum(dim=0) if ctx.has_bias else None
        return dx, dw, db, None


def linear_act(x, w, b=None, act="none"):
    return LinearAct.apply(x, w, b, ACT_CODES[act] if isinstance(act, str) else act)


class SoftmaxXent(torch.autograd.Function):
    """Mean softmax-cross-entropy over rows; labels are one-hot [M,C]."""

    @staticmethod
    def forward(ctx, logits, onehot):
        if ops.use_hip(logits):
            loss, probs = ops.hip.softmax_xent_fwd(logits, onehot,
                                                   _want_loss_value())
        else:
            lse = torch.logsumexp(logits, dim=1, keepdim=True)
            logp = logits - lse
            loss = -(onehot * logp).sum(dim=1).mean()
            probs = torch.exp(logp)
        ctx.save_for_backward(probs, onehot)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        probs, onehot = ctx.saved_tensors
        m = probs.shape[0]
        if ops.use_hip(probs):
            dlogits = ops.hip.softmax_xent_bwd(probs, onehot,
                                               _dev_scale(dloss), 1.0 / m)
        else:
            dlogits = (probs - onehot) * (dloss / m)
        return dlogits, None


class SoftmaxXentIdx(torch.autograd.Function):
    """Class-index label path — no one-hot materialization (the fused
    kernel reads int64 OR integral-float labels directly).

    Under ``grad_only()`` the forward kernel writes the GRADIENT
    ``(probs - onehot)/m`` instead of the probs and backward returns it
    as-is — the whole loss fwd+bwd is ONE kernel.  That contract assumes
    a unit upstream gradient (which ``grad_only`` callers guarantee by
    backpropagating a ones root)."""

    @staticmethod
    def forward(ctx, logits, labels):
        grad_fused = False
        if ops.use_hip(logits):
            grad_fused = not _want_loss_value()
            loss, probs = ops.hip.softmax_xent_idx_fwd(
                logits, labels, not grad_fused, grad_fused)
        else:
            lab = labels.long()
            lse = torch.logsumexp(logits, dim=1, keepdim=True)
            logp = logits - lse
            loss = -logp.gather(1, lab.view(-1, 1)).mean()
            probs = torch.exp(logp)
        ctx.save_for_backward(probs, labels)
        ctx.grad_fused = grad_fused
        return loss

    @staticmethod
    def backward(ctx, dloss):
        probs, labels = ctx.saved_tensors
        m = probs.shape[0]
        if ctx.grad_fused:
            return probs, None  # probs holds (softmax - onehot)/m already
        if ops.use_hip(probs):
            dlogits = ops.hip.softmax_xent_idx_bwd(probs, labels,
                                                   _dev_scale(dloss), 1.0 / m)
        else:
            dlogits = probs.clone()
            dlogits[torch.arange(m, device=probs.device),
                    labels.long()] -= 1.0
            dlogits.mul_(dloss / m)
        return dlogits, None


def softmax_xent(logits, labels):
    """labels: one-hot [M,C], class-index [M], or class-index [M,1]
    (the engine's ``handle_features`` wraps scalar labels to [y])."""
    if labels.dim() == 2 and labels.size(1) == 1 and logits.size(1) != 1:
        labels = labels.reshape(-1)
    if labels.dim() == 1:
        if ops.use_hip(logits) and labels.dtype in (torch.float32,
                                                    torch.int64):
            # HIP kernels read float labels directly — no cast kernel
            return SoftmaxXentIdx.apply(logits, labels.contiguous())
        return SoftmaxXentIdx.apply(logits, labels.long().contiguous())
    return SoftmaxXent.apply(logits, labels.to(logits.dtype).contiguous())


class MseLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, pred, target):
        ctx.save_for_backward(pred, target)
        if ops.use_hip(pred):
            return ops.hip.mse_fwd(pred, target, _want_loss_value())
        return ((pred - target) ** 2).mean()

    @staticmethod
    def backward(ctx, dloss):
        pred, target = ctx.saved_tensors
        n = pred.numel()
        if ops.use_hip(pred):
            dpred = ops.hip.mse_bwd(pred, target, _dev_scale(dloss), 2.0 / n)
        else:
            dpred = (pred - target) * (2.0 * dloss / n)
        return dpred, None


def mse_loss(pred, target):
    return MseLoss.apply(pred, target)


class Conv2dAct(torch.autograd.Function):
    """NCHW conv2d + bias + activation. w: [Cout, Cin, Kh, Kw]."""

    @staticmethod
    def forward(ctx, x, w, b, stride: int, pad: int, act: int):
        if ops.use_hip(x):
            y = ops.hip.conv_gemm_fwd(
                x, w, b if b is not None else torch.empty(0, device=x.device),
                stride, pad, act)
        else:
            pre = torch.nn.functional.conv2d(x, w, b, stride=stride, padding=pad)
            y = _act_fwd(pre, act)
        ctx.save_for_backward(x, w, y)
        ctx.conf = (stride, pad, act, b is not None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        stride, pad, act, has_bias = ctx.conf
        need_dx = ctx.needs_input_grad[0]
        dy = dy.contiguous()
        if ops.use_hip(x):
            dpre = ops.hip.act_bwd(y, dy, act) if act != ACT_NONE else dy
            if not need_dx:
                dx = None
            else:
                # dgrad IS a conv: dx = corr(dy, W swapped+rotated) with
                # padding K-1-p on the implicit-GEMM MFMA path; stride>1
                # zero-stuffs dy first (the trailing remainder rows/cols
                # recover the exact input size) — ~10x over the direct
                # strided dgrad kernel on ResNet shapes.
                wt = w.transpose(0, 1).flip(2, 3).contiguous()
                kk = w.shape[2]
                if stride == 1:
                    src = dpre
                else:
                    rh = (x.shape[2] + 2 * pad - kk) % stride
                    rw = (x.shape[3] + 2 * pad - kk) % stride
                    src = ops.hip.conv_dilate(dpre, stride, rh, rw)
                dx = ops.hip.conv_gemm_fwd(
                    src, wt, torch.empty(0, device=x.device), 1,
                    kk - 1 - pad, ACT_NONE)
            dw = ops.hip.conv_gemm_wgrad(x, dpre, w.shape[2], w.shape[3],
                                         stride, pad)
            db = ops.hip.conv_bias_grad(dpre) if has_bias else None
        else:
            dpre = _act_bwd_from_out(y, dy, act)
            dx = torch.nn.grad.conv2d_input(x.shape, w, dpre, stride=stride,
                                            padding=pad) if need_dx else None
            dw = torch.nn.grad.conv2d_weight(x, w.shape, dpre, stride=stride,
                                             padding=pad)
            db = dpre.sum(dim=(0, 2, 3)) if has_bias else None
        return dx, dw, db, None, None, None


def conv2d_act(x, w, b=None, stride=1, pad=0, act="none"):
    return Conv2dAct.apply(x, w, b, stride, pad,
                           ACT_CODES[act] if isinstance(act, str) else act)


class Conv2dActNHWC(torch.autograd.Function):
    """Channel-last conv: x [N,H,W,Cin], w [Cout,Kh,Kw,Cin] — im2col
    gathers and output stores are channel-contiguous float4s (the NCHW
    kernels were gather-latency-bound on ResNet shapes).  CPU fallback
    permutes through torch's NCHW conv."""

    @staticmethod
    def forward(ctx, x, w, b, stride: int, pad: int, act: int):
        # MEASURED NEGATIVE (round 2): zero-padding Cin 1->4 / 3->4 to
        # reach the float4 gather path multiplies the GEMM K work by
        # 4/Cin, which costs more than the scalar gathers it removes
        # (MNIST CNN -10%, ResNet-18 -1.3%).  Keep the scalar path.
        cpad = 0
        if ops.use_hip(x):
            y = ops.hip.conv_nhwc_fwd(
                x, w, b if b is not None else torch.empty(0, device=x.device),
                stride, pad, act)
        else:
            xc = x.permute(0, 3, 1, 2)
            wc = w.permute(0, 3, 1, 2)
            pre = torch.nn.functional.conv2d(xc, wc, b, stride=stride,
                                             padding=pad)
            y = _act_fwd(pre, act).permute(0, 2, 3, 1).contiguous()
        ctx.save_for_backward(x, w, y)
        ctx.conf = (stride, pad, act, b is not None, cpad)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, y = ctx.saved_tensors
        stride, pad, act, has_bias, cpad = ctx.conf
        need_dx = ctx.needs_input_grad[0]
        dy = dy.contiguous()
        kk = w.shape[1]
        if ops.use_hip(x):
            dpre = ops.hip.act_bwd(y, dy, act) if act != ACT_NONE else dy
            if not need_dx:
                dx = None
            else:
                # dgrad IS a stride-1 conv of dpre with spatially-
                # flipped, in/out-swapped weights.  stride > 1 uses the
                # direct s^2-sub-convolution kernel (no zero-stuffed
                # dilation: the dilated form ran s^2 x the useful MFMA
                # work); the dilate path stays as the Cout%4!=0 fallback.
                wt = w.flip(1, 2).permute(3, 1, 2, 0).contiguous()
                if stride == 1:
                    dx = ops.hip.conv_nhwc_fwd(
                        dpre, wt, torch.empty(0, device=x.device), 1,
                        kk - 1 - pad, ACT_NONE)
                elif w.shape[0] % 4 == 0:
                    dx = ops.hip.conv_nhwc_dgrad_strided(
                        dpre, wt, x.shape[1], x.shape[2], stride, pad)
                else:
                    rh = (x.shape[1] + 2 * pad - kk) % stride
                    rw = (x.shape[2] + 2 * pad - kk) % stride
                    src = ops.hip.conv_nhwc_dilate(dpre, stride, rh, rw)
                    dx = ops.hip.conv_nhwc_fwd(
                        src, wt, torch.empty(0, device=x.device), 1,
                        kk - 1 - pad, ACT_NONE)
            dw = ops.hip.conv_nhwc_wgrad(x, dpre, kk, w.shape[2], stride,
                                         pad)
            db = ops.hip.conv_nhwc_bias_grad(dpre) if has_bias else None
        else:
            dpre = _act_bwd_from_out(y, dy, act)
            xc = x.permute(0, 3, 1, 2)
            wc = w.permute(0, 3, 1, 2)
            dpc = dpre.permute(0, 3, 1, 2)
            dx = None
            if need_dx:
                dx = torch.nn.grad.conv2d_input(
                    xc.shape, wc, dpc, stride=stride, padding=pad) \
                    .permute(0, 2, 3, 1).contiguous()
            dw = torch.nn.grad.conv2d_weight(
                xc, wc.shape, dpc, stride=stride, padding=pad) \
                .permute(0, 2, 3, 1).contiguous()
            db = dpre.sum(dim=(0, 1, 2)) if has_bias else None
        return dx, dw, db, None, None, None


def conv2d_act_nhwc(x, w, b=None, stride=1, pad=0, act="none"):
    return Conv2dActNHWC.apply(x, w, b, stride, pad,
                               ACT_CODES[act] if isinstance(act, str)
                               else act)


class BatchNorm2dActNHWC(torch.autograd.Function):
    """NHWC training BN with optional fused trailing ReLU AND optional
    fused ResNet-block residual add: out = relu(bn(x) + res).  Fusing
    the add+relu into the norm kernel (and the relu-masked d_res into
    the bwd-dx kernel) removes two full activation-tensor passes per
    block boundary."""

    @staticmethod
    def forward(ctx, x, gamma, beta, run_mean, run_var, momentum, eps,
                relu, res=None):
        empty = torch.empty(0, device=x.device, dtype=x.dtype)
        y, mean, invstd = ops.hip.bn2d_nhwc_fwd(
            x, gamma, beta, run_mean, run_var, momentum, eps, relu,
            res.contiguous() if res is not None else empty)
        ctx.save_for_backward(x, y, gamma, mean, invstd)
        ctx.relu = relu
        ctx.has_res = res is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, gamma, mean, invstd = ctx.saved_tensors
        dx, dgamma, dbeta, dres = ops.hip.bn2d_nhwc_bwd(
            x, dy.contiguous(), y, gamma, mean, invstd, ctx.relu,
            ctx.has_res)
        return (dx, dgamma, dbeta, None, None, None, None, None,
                dres if ctx.has_res else None)


def batchnorm2d_act_nhwc(x, gamma, beta, run_mean, run_var, momentum=0.1,
                         eps=1e-5, relu=False, training=True,
                         residual=None):
    """x: [N, H, W, C] channel-last.  ``residual`` fuses a ResNet block
    boundary: out = relu(bn(x) + residual)."""
    if training and ops.use_hip(x):
        return BatchNorm2dActNHWC.apply(x.contiguous(), gamma, beta,
                                        run_mean, run_var, momentum, eps,
                                        relu, residual)
    xc = x.permute(0, 3, 1, 2)
    y = torch.nn.functional.batch_norm(xc, run_mean, run_var, gamma, beta,
                                       training, momentum, eps)
    y = y.permute(0, 2, 3, 1)
    if residual is not None:
        y = y + residual
    if relu:
        y = torch.relu(y)
    return y.contiguous()


class MaxPool2d(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, k: int, stride: int, pad: int):
        if ops.use_hip(x):
            y, idx = ops.hip.maxpool2d_fwd(x, k, stride, pad)
        else:
            y, idx = torch.nn.functional.max_pool2d(x, k, stride, padding=pad,
                                                    return_indices=True)
        ctx.save_for_backward(idx)
        ctx.conf = (x.shape, k, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        shape, k, stride, pad = ctx.conf
        dy = dy.contiguous()
        if ops.use_hip(dy):
            dx = ops.hip.maxpool2d_bwd(dy, idx, shape[2], shape[3])
        else:
            # scatter_add, NOT max_unpool2d: with overlapping windows
            # (stride < k) one input can be the max of several windows
            # and unpool would overwrite instead of accumulating
            n, c = shape[0], shape[1]
            dx = torch.zeros(n, c, shape[2] * shape[3], dtype=dy.dtype)
            dx.scatter_add_(2, idx.reshape(n, c, -1), dy.reshape(n, c, -1))
            dx = dx.reshape(shape)
        return dx, None, None, None


def maxpool2d(x, k, stride=None, pad=0):
    return MaxPool2d.apply(x, k, stride or k, pad)


class MaxPool2dNHWC(torch.autograd.Function):
    """Channel-last maxpool (GPU only; the module's CPU path permutes
    through the torch NCHW pool)."""

    @staticmethod
    def forward(ctx, x, k: int, stride: int, pad: int):
        y, idx = ops.hip.maxpool_nhwc_fwd(x, k, stride, pad)
        ctx.save_for_backward(idx)
        ctx.conf = (x.shape[1], x.shape[2], k, stride, pad)
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        H, W, k, stride, pad = ctx.conf
        return ops.hip.maxpool_nhwc_bwd(dy.contiguous(), idx, H, W, k,
                                        stride, pad), \
            None, None, None


def maxpool2d_nhwc(x, k, stride=None, pad=0):
    return MaxPool2dNHWC.apply(x, k, stride or k, pad)


class BatchNorm2dAct(torch.autograd.Function):
    """Training-mode BatchNorm2d with optional fused ReLU on the custom
    CDNA4 kernels (MIOpen's spatial BN was 27% of the ResNet-shape step).
    Updates running stats in-place as a forward side effect."""

    @staticmethod
    def forward(ctx, x, gamma, beta, run_mean, run_var, momentum, eps,
                relu):
        y, mean, invstd = ops.hip.bn2d_fwd(x, gamma, beta, run_mean,
                                           run_var, momentum, eps, relu)
        ctx.save_for_backward(x, y, gamma, mean, invstd)
        ctx.relu = relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, gamma, mean, invstd = ctx.saved_tensors
        dx, dgamma, dbeta = ops.hip.bn2d_bwd(x, dy.contiguous(), y, gamma,
                                             mean, invstd, ctx.relu)
        return dx, dgamma, dbeta, None, None, None, None, None


def batchnorm2d_act(x, gamma, beta, run_mean, run_var, momentum=0.1,
                    eps=1e-5, relu=False, training=True):
    if training and ops.use_hip(x):
        return BatchNorm2dAct.apply(x.contiguous(), gamma, beta, run_mean,
                                    run_var, momentum, eps, relu)
    y = torch.nn.functional.batch_norm(x, run_mean, run_var, gamma, beta,
                                       training, momentum, eps)
    return torch.relu(y) if relu else y


def argmax_rows(y: torch.Tensor) -> torch.Tensor:
    """Rowwise argmax prediction head (reference tf.argmax(out, 1))."""
    if ops.use_hip(y):
        return ops.hip.argmax_rows(y)
    return torch.argmax(y, dim=1).to(torch.float32)
