"""cilfw functional ops: autograd Functions with device dispatch.

Conventions (cilfw-native, chosen for CDNA4 — not the reference's):
- Activations are NHWC contiguous: ``(N, H, W, C)``. Channels innermost gives
  coalesced per-channel access for BN stats and a dense GEMM K-dim (C*R*S) for the
  implicit-GEMM MFMA convolutions.
- Conv weights are ``(R, S, C, K)`` fp32 masters; forward casts to the compute dtype
  (bf16) once per use. Weight grads come back fp32.
- CPU tensors run the torch reference implementations (fp32 math) — these double as
  the numerics oracles for the HIP kernels (tests/test_ops_gpu.py).
- CUDA tensors run the hand-written HIP kernels via ``cilfw._hip_ops`` and raise if
  the extension is missing (no silent ATen fallback).

Reference behavior being reimplemented: the implicit cuDNN /
cuBLAS kernels behind reference ``resnet.py`` and ``template.py:99-101, 259-263``.
"""

import torch
import torch.nn.functional as F

from ._backend import use_hip, ext


_wgrad_stream = None


def _wstream():
    """Side stream for conv weight-grad kernels: dW is off the backward's
    critical path (only dX feeds the next layer), so bwd-weight runs
    concurrently with bwd-data — sum becomes max of the two."""
    global _wgrad_stream
    if _wgrad_stream is None:
        _wgrad_stream = torch.cuda.Stream()
    return _wgrad_stream


def _to_nchw(x):
    return x.permute(0, 3, 1, 2).contiguous()


def _to_nhwc(x):
    return x.permute(0, 2, 3, 1).contiguous()


# --------------------------------------------------------------------------- conv2d


def _sink_on():
    import os
    return os.environ.get("CILFW_SINK_MODE", "join") != "off"


def _bnbwd_fuse_on():
    # default off: measured slower than the standalone vectorized sums pass
    # on both rn18 and rn50 (see conv.hip bnbwd_fuse_enabled) — the kernel
    # path stays covered by tests either way and flips on via env
    import os
    return os.environ.get("CILFW_BNBWD_FUSE", "0") == "1"


def _conv_bwd_data_with_bn(ctx, E, dy, wc, stride, padding, x):
    """bwd-data; when this conv's input is a single-consumer training-mode
    BN(+ReLU) output, the bwd-data kernel epilogue also emits that BN's
    backward (dgamma, dbeta) partials — attached to the returned dx, which
    IS the BN's incoming grad, so its backward skips the sums pass. The
    save_mean tensor rides along as an identity tag: the BN only accepts
    partials stamped with its own per-call statistics."""
    meta = getattr(ctx, "bnbwd_src", None)
    if meta is None or not _bnbwd_fuse_on():
        return E.conv2d_bwd_data(dy, wc, stride, padding,
                                 x.shape[1], x.shape[2])
    bn_x, bn_mean, bn_invstd, bn_relu = meta
    dx, parts = E.conv2d_bwd_data(
        dy, wc, stride, padding, x.shape[1], x.shape[2],
        bn_meta=(x, bn_x, bn_mean, bn_invstd, bn_relu))
    if parts is not None:
        dx._cilfw_bnbwd = (parts, bn_mean)
    return dx


class Conv2dNHWC(torch.autograd.Function):
    """2D convolution, NHWC activations, (R,S,C,K) weight, symmetric padding.

    ``w_param`` (non-tensor arg) is set when the weight is engine-registered
    for grad-sink delivery: ``w`` is then the DETACHED weight, so autograd
    creates no AccumulateGrad edge for it (returning None would otherwise
    still materialize a zero grad, run an accumulation kernel and fire the
    post-accumulate hook — double-counting bucket arrivals), and backward
    writes dW straight into the engine's flat slot."""

    @staticmethod
    def forward(ctx, x, w, stride, padding, w_param=None):
        ctx.stride, ctx.padding = stride, padding
        if w_param is not None:
            w_param = w_param[0]  # holder list: keeps the param OUT of the graph
        wp = w_param if w_param is not None else w
        # engine-maintained bf16 mirror (updated inside the fused SGD kernel)
        # avoids a per-step cast of the fp32 master
        wc = getattr(wp, "_cilfw_bf16", None)
        if wc is None or wc.dtype != x.dtype:
            wc = w.to(x.dtype)
        ctx.save_for_backward(x, wc)
        ctx.w_dtype = w.dtype
        ctx.w_ref = wp  # grad-sink lookup (engine flat-slot delivery)
        ctx.sinked = w_param is not None
        # input produced by a single-consumer training BN(+ReLU): backward
        # can emit that BN's (dgamma, dbeta) partials from the bwd-data
        # epilogue (tensors held as plain refs — the BN's ctx saves them too)
        ctx.bnbwd_src = getattr(x, "_cilfw_bnbwd_src", None)
        if use_hip(x):
            y, parts = ext().conv2d_fwd(x, wc, stride, padding,
                                        want_bn_parts=ctx.sinked)
            if parts is not None:
                # a following training-mode BN consumes these instead of
                # running its own bn_sums pass over y
                y._cilfw_bn_parts = (parts, (y.numel() // y.shape[-1]
                                             + 127) // 128)
            return y
        # CPU reference: fp32 NCHW conv
        xf = _to_nchw(x).float()
        wf = w.permute(3, 2, 0, 1).contiguous().float()  # (K,C,R,S)
        y = F.conv2d(xf, wf, stride=stride, padding=padding)
        return _to_nhwc(y).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, wc = ctx.saved_tensors
        stride, padding = ctx.stride, ctx.padding
        dy = dy.contiguous()
        need_dx = ctx.needs_input_grad[0]  # stems skip the whole bwd-data pass
        sink = getattr(ctx.w_ref, "_cilfw_sink", None) if ctx.sinked else None
        if use_hip(dy):
            R, S = wc.shape[0], wc.shape[1]
            cur = torch.cuda.current_stream()
            ws = _wstream()
            ws.wait_stream(cur)
            if sink is not None:
                # write dW straight into the engine's flat-grad slot on the
                # side stream (no fresh tensor, no AccumulateGrad add)
                eng, pi = sink
                slot, accum = eng.sink_acquire(pi)
                out = slot.view(R, S, wc.shape[2], wc.shape[3])
                with torch.cuda.stream(ws):
                    ext().conv2d_bwd_weight(dy, x, stride, padding, R, S,
                                            out=out, accum=accum)
                dx = _conv_bwd_data_with_bn(ctx, ext(), dy, wc, stride, padding, x) \
                    if need_dx else None
                cur.wait_stream(ws)
                eng.sink_delivered(pi)
                return dx, None, None, None, None
            with torch.cuda.stream(ws):  # dW concurrent with dX (fork/join)
                dw = ext().conv2d_bwd_weight(dy, x, stride, padding, R, S)
            dx = _conv_bwd_data_with_bn(ctx, ext(), dy, wc, stride, padding, x) \
                if need_dx else None
            cur.wait_stream(ws)
            dw.record_stream(cur)
            return dx, dw.to(ctx.w_dtype), None, None, None
        xf = _to_nchw(x).float()
        wf = wc.permute(3, 2, 0, 1).contiguous().float()
        dyf = _to_nchw(dy).float()
        dx = None
        if need_dx:
            dxf = torch.nn.grad.conv2d_input(xf.shape, wf, dyf, stride=stride,
                                             padding=padding)
            dx = _to_nhwc(dxf).to(x.dtype)
        dwf = torch.nn.grad.conv2d_weight(xf, wf.shape, dyf, stride=stride,
                                          padding=padding)
        dw = dwf.permute(2, 3, 1, 0).contiguous().to(ctx.w_dtype)  # (R,S,C,K)
        if sink is not None:
            eng, pi = sink
            slot, accum = eng.sink_acquire(pi)
            out = slot.view_as(dw)
            (out.add_(dw) if accum else out.copy_(dw))
            eng.sink_delivered(pi)
            return dx, None, None, None, None
        return dx, dw, None, None, None


def conv2d(x, w, stride=1, padding=1):
    # engine-registered weights take the grad-sink path: DETACHED weight in
    # the graph + the param as a non-tensor ref (x must carry requires_grad
    # or the graph would die at input-adjacent convs, e.g. the stem)
    if (_sink_on() and x.requires_grad
            and getattr(w, "_cilfw_sink", None) is not None):
        return Conv2dNHWC.apply(x, w.detach(), stride, padding, [w])
    return Conv2dNHWC.apply(x, w, stride, padding)


# ----------------------------------------------------------------- batchnorm (+ReLU)


def _gb_sinkable(x, gamma, beta):
    """True when both BN params are engine-registered for grad-sink delivery
    (and x carries the graph, so detaching them cannot kill it)."""
    if not (_sink_on() and x.requires_grad):
        return False
    sg = getattr(gamma, "_cilfw_sink", None)
    sb = getattr(beta, "_cilfw_sink", None)
    return sg is not None and sb is not None and sg[0] is sb[0]


def _gb_deliver_hip(eng, gi, bi, bn_bwd_args, want_dres, ext_parts=None):
    """BN backward with dgamma/dbeta written into the flat slots. During
    grad accumulation the reduced grads go to scratch first (the dx formula
    needs THIS batch's dgamma/dbeta, not the accumulated slot) and are added."""
    gs, gacc = eng.sink_acquire(gi)
    bs, bacc = eng.sink_acquire(bi)
    if gacc or bacc:
        dx, dgamma, dbeta, dres = ext().bn_bwd(*bn_bwd_args,
                                               want_dres=want_dres,
                                               ext_parts=ext_parts)
        gs.add_(dgamma)
        bs.add_(dbeta)
    else:
        dx, _, _, dres = ext().bn_bwd(*bn_bwd_args, want_dres=want_dres,
                                      out_gamma=gs, out_beta=bs,
                                      ext_parts=ext_parts)
    eng.sink_delivered(gi)
    eng.sink_delivered(bi)
    return dx, dres


def _gb_deliver_cpu(eng, gi, bi, dgamma, dbeta):
    gs, gacc = eng.sink_acquire(gi)
    bs, bacc = eng.sink_acquire(bi)
    (gs.add_(dgamma) if gacc else gs.copy_(dgamma))
    (bs.add_(dbeta) if bacc else bs.copy_(dbeta))
    eng.sink_delivered(gi)
    eng.sink_delivered(bi)


class BatchNormAct(torch.autograd.Function):
    """Training-mode BN over NHWC (per-channel stats on N*H*W), optional fused ReLU.

    Matches torch BN semantics: biased variance for normalization, unbiased for the
    running-stat update; per-rank statistics (the reference runs plain BN under DDP —
    no SyncBN).
    """

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, momentum, eps,
                training, relu, g_param=None, b_param=None, fuse_bwd=True):
        ctx.relu = relu
        ctx.eps = eps
        # grad-sink: g_param/b_param set => gamma/beta are DETACHED (see
        # Conv2dNHWC docstring for why returning None is not enough); the
        # params arrive in holder lists so apply() does not graph-track them
        if g_param is not None:
            g_param, b_param = g_param[0], b_param[0]
        ctx.g_ref = g_param if g_param is not None else gamma
        ctx.b_ref = b_param if b_param is not None else beta
        ctx.sinked = g_param is not None
        if use_hip(x):
            pg = getattr(x, "_cilfw_bn_parts", None)
            y, save_mean, save_invstd = ext().bn_fwd(
                x, gamma, beta, running_mean, running_var,
                momentum, eps, training, relu,
                ext_parts=pg[0] if pg else None,
                ext_gy=pg[1] if pg else 0)
            ctx.training = training
            ctx.save_for_backward(x, gamma, save_mean, save_invstd, y)
            if training and fuse_bwd and _bnbwd_fuse_on():
                # a single downstream conv may emit this BN's backward
                # partials from its bwd-data epilogue (models opt OUT for
                # multi-consumer BN outputs, e.g. stems feeding identity
                # skips — correctness is preserved either way, the partials
                # would just be dead work)
                y._cilfw_bnbwd_src = (x, save_mean, save_invstd, relu)
            return y
        N, H, W, C = x.shape
        xf = x.float().reshape(-1, C)
        if training:
            mean = xf.mean(dim=0)
            var = xf.var(dim=0, unbiased=False)
            m = xf.shape[0]
            with torch.no_grad():
                running_mean.mul_(1 - momentum).add_(momentum * mean)
                unbiased = var * (m / max(m - 1, 1))
                running_var.mul_(1 - momentum).add_(momentum * unbiased)
        else:
            mean, var = running_mean.float(), running_var.float()
        invstd = torch.rsqrt(var + eps)
        yf = (xf - mean) * invstd * gamma.float() + beta.float()
        if relu:
            yf = yf.clamp_min(0)
        y = yf.reshape(N, H, W, C).to(x.dtype)
        ctx.training = training
        ctx.save_for_backward(x, gamma, mean, invstd, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, invstd, y = ctx.saved_tensors
        relu = ctx.relu
        dy = dy.contiguous()
        sink = (getattr(ctx.g_ref, "_cilfw_sink", None)
                if ctx.sinked else None)
        if use_hip(dy):
            pgb = getattr(dy, "_cilfw_bnbwd", None)
            # accept only partials stamped with THIS call's statistics (the
            # save_mean tensor identity ties producer and consumer)
            ext_parts = (pgb[0] if pgb is not None and pgb[1] is mean
                         else None)
            if sink is not None:  # reduce writes straight into the slots
                eng, gi = sink
                bi = ctx.b_ref._cilfw_sink[1]
                dx, _ = _gb_deliver_hip(
                    eng, gi, bi,
                    (dy, x, gamma, mean, invstd, y, relu, ctx.training),
                    want_dres=False, ext_parts=ext_parts)
                return (dx, None, None) + (None,) * 9
            dx, dgamma, dbeta, _ = ext().bn_bwd(dy, x, gamma, mean, invstd,
                                                y, relu, ctx.training,
                                                ext_parts=ext_parts)
            return (dx, dgamma, dbeta) + (None,) * 9
        C = x.shape[-1]
        dyf = dy.float().reshape(-1, C)
        if relu:
            dyf = dyf * (y.float().reshape(-1, C) > 0)
        xf = x.float().reshape(-1, C)
        xhat = (xf - mean) * invstd
        dgamma = (dyf * xhat).sum(dim=0)
        dbeta = dyf.sum(dim=0)
        if ctx.training:
            m = xf.shape[0]
            dxf = (gamma.float() * invstd / m) * (
                m * dyf - dbeta - xhat * dgamma)
        else:
            dxf = dyf * gamma.float() * invstd
        dx = dxf.reshape_as(x).to(x.dtype)
        if sink is not None:
            eng, gi = sink
            bi = ctx.b_ref._cilfw_sink[1]
            _gb_deliver_cpu(eng, gi, bi, dgamma, dbeta)
            return (dx, None, None) + (None,) * 9
        return (dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype)) + (None,) * 9


def batchnorm_act(x, gamma, beta, running_mean, running_var, momentum=0.1, eps=1e-5,
                  training=True, relu=False, fuse_bwd=True):
    if _gb_sinkable(x, gamma, beta):
        return BatchNormAct.apply(x, gamma.detach(), beta.detach(),
                                  running_mean, running_var, momentum, eps,
                                  training, relu, [gamma], [beta], fuse_bwd)
    return BatchNormAct.apply(x, gamma, beta, running_mean, running_var, momentum,
                              eps, training, relu, None, None, fuse_bwd)


class BatchNormAddReLU(torch.autograd.Function):
    """y = relu(bn(x) + residual) — the residual-block tail fused into the BN
    apply/backward kernels (saves the add_relu round-trips; dres == the
    relu-masked dy the BN backward already computes)."""

    @staticmethod
    def forward(ctx, x, residual, gamma, beta, running_mean, running_var,
                momentum, eps, training, g_param=None, b_param=None):
        ctx.eps = eps
        if g_param is not None:  # holder lists (see BatchNormAct)
            g_param, b_param = g_param[0], b_param[0]
        ctx.g_ref = g_param if g_param is not None else gamma
        ctx.b_ref = b_param if b_param is not None else beta
        ctx.sinked = g_param is not None
        if use_hip(x):
            pg = getattr(x, "_cilfw_bn_parts", None)
            y, save_mean, save_invstd = ext().bn_fwd(
                x, gamma, beta, running_mean, running_var, momentum, eps,
                training, True, residual=residual,
                ext_parts=pg[0] if pg else None,
                ext_gy=pg[1] if pg else 0)
            ctx.training = training
            ctx.save_for_backward(x, gamma, save_mean, save_invstd, y)
            return y
        N, H, W, C = x.shape
        xf = x.float().reshape(-1, C)
        if training:
            mean = xf.mean(dim=0)
            var = xf.var(dim=0, unbiased=False)
            m = xf.shape[0]
            with torch.no_grad():
                running_mean.mul_(1 - momentum).add_(momentum * mean)
                unbiased = var * (m / max(m - 1, 1))
                running_var.mul_(1 - momentum).add_(momentum * unbiased)
        else:
            mean, var = running_mean.float(), running_var.float()
        invstd = torch.rsqrt(var + eps)
        yf = (xf - mean) * invstd * gamma.float() + beta.float()
        yf = (yf + residual.float().reshape(-1, C)).clamp_min(0)
        y = yf.reshape(N, H, W, C).to(x.dtype)
        ctx.training = training
        ctx.save_for_backward(x, gamma, mean, invstd, y)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, gamma, mean, invstd, y = ctx.saved_tensors
        dy = dy.contiguous()
        sink = (getattr(ctx.g_ref, "_cilfw_sink", None)
                if ctx.sinked else None)
        if use_hip(dy):
            if sink is not None:  # reduce writes straight into the slots
                eng, gi = sink
                bi = ctx.b_ref._cilfw_sink[1]
                dx, dres = _gb_deliver_hip(
                    eng, gi, bi,
                    (dy, x, gamma, mean, invstd, y, True, ctx.training),
                    want_dres=True)
                return (dx, dres, None, None) + (None,) * 7
            dx, dgamma, dbeta, dres = ext().bn_bwd(
                dy, x, gamma, mean, invstd, y, True, ctx.training,
                want_dres=True)
            return (dx, dres, dgamma, dbeta) + (None,) * 7
        C = x.shape[-1]
        dyf = dy.float().reshape(-1, C) * (y.float().reshape(-1, C) > 0)
        xf = x.float().reshape(-1, C)
        xhat = (xf - mean) * invstd
        dgamma = (dyf * xhat).sum(dim=0)
        dbeta = dyf.sum(dim=0)
        if ctx.training:
            m = xf.shape[0]
            dxf = (gamma.float() * invstd / m) * (
                m * dyf - dbeta - xhat * dgamma)
        else:
            dxf = dyf * gamma.float() * invstd
        dres = dyf.reshape_as(x).to(dy.dtype)
        if sink is not None:
            eng, gi = sink
            bi = ctx.b_ref._cilfw_sink[1]
            _gb_deliver_cpu(eng, gi, bi, dgamma, dbeta)
            return (dxf.reshape_as(x).to(x.dtype), dres, None,
                    None) + (None,) * 7
        return (dxf.reshape_as(x).to(x.dtype), dres, dgamma.to(gamma.dtype),
                dbeta.to(gamma.dtype)) + (None,) * 7


def batchnorm_add_relu(x, residual, gamma, beta, running_mean, running_var,
                       momentum=0.1, eps=1e-5, training=True):
    if _gb_sinkable(x, gamma, beta):
        return BatchNormAddReLU.apply(x, residual, gamma.detach(),
                                      beta.detach(), running_mean,
                                      running_var, momentum, eps, training,
                                      [gamma], [beta])
    return BatchNormAddReLU.apply(x, residual, gamma, beta, running_mean,
                                  running_var, momentum, eps, training)


# ------------------------------------------------------------------------- add+ReLU


class AddReLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        if use_hip(a):
            y = ext().add_relu_fwd(a, b)
        else:
            y = (a.float() + b.float()).clamp_min(0).to(a.dtype)
        ctx.save_for_backward(y)
        return y

    @staticmethod
    def backward(ctx, dy):
        (y,) = ctx.saved_tensors
        dy = dy.contiguous()
        if use_hip(dy):
            da = ext().add_relu_bwd(dy, y)
        else:
            da = (dy * (y > 0)).to(dy.dtype)
        return da, da


def add_relu(a, b):
    return AddReLU.apply(a, b)


# ------------------------------------------------------- downsample-A (reference C9)


class DownsampleA(torch.autograd.Function):
    """Stride-2 1x1 avg-pool + zero-channel pad (reference resnet.py:9-17):
    (N,H,W,C) -> (N,H/2,W/2,2C) where the second C channels are zeros."""

    @staticmethod
    def forward(ctx, x):
        ctx.in_shape = x.shape
        if use_hip(x):
            return ext().downsample_a_fwd(x)
        N, H, W, C = x.shape
        y = x.new_zeros(N, H // 2, W // 2, 2 * C)
        y[:, :, :, :C] = x[:, ::2, ::2, :]
        return y

    @staticmethod
    def backward(ctx, dy):
        N, H, W, C = ctx.in_shape
        dy = dy.contiguous()
        if use_hip(dy):
            return ext().downsample_a_bwd(dy, H, W)
        dx = dy.new_zeros(N, H, W, C)
        dx[:, ::2, ::2, :] = dy[:, :, :, :C]
        return dx


def downsample_a(x):
    return DownsampleA.apply(x)


# --------------------------------------------------------------- global average pool


class GlobalAvgPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ctx.in_shape = x.shape
        if use_hip(x):
            return ext().gap_fwd(x)
        N, H, W, C = x.shape
        return x.float().mean(dim=(1, 2)).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        N, H, W, C = ctx.in_shape
        dy = dy.contiguous()
        if use_hip(dy):
            return ext().gap_bwd(dy, H, W)
        scale = 1.0 / (H * W)
        return (dy * scale).reshape(N, 1, 1, C).expand(N, H, W, C).to(dy.dtype)


def global_avg_pool(x):
    return GlobalAvgPool.apply(x)


# --------------------------------------------------------------------------- linear


class LinearFn(torch.autograd.Function):
    """y = x @ w^T + b with fp32 master w/b; grads fp32."""

    @staticmethod
    def forward(ctx, x, w, b):
        wc = getattr(w, "_cilfw_bf16", None)
        if wc is None or wc.dtype != x.dtype:
            wc = w.to(x.dtype)
        ctx.save_for_backward(x, wc)
        ctx.w_dtype = w.dtype
        ctx.has_bias = b is not None
        if use_hip(x):
            return ext().linear_fwd(x, wc, b.float() if b is not None else None)
        y = x.float() @ w.float().t()
        if b is not None:
            y = y + b.float()
        return y.to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        x, wc = ctx.saved_tensors
        dy = dy.contiguous()
        if use_hip(dy):
            dx, dw, db = ext().linear_bwd(dy, x, wc, ctx.has_bias)
            return dx, dw.to(ctx.w_dtype), (db.to(ctx.w_dtype) if ctx.has_bias
                                            else None)
        dyf = dy.float()
        dx = (dyf @ wc.float()).to(x.dtype)
        dw = (dyf.t() @ x.float()).to(ctx.w_dtype)
        db = dyf.sum(dim=0).to(ctx.w_dtype) if ctx.has_bias else None
        return dx, dw, db


def linear(x, w, b=None):
    return LinearFn.apply(x, w, b)


# ---------------------------------------------------- fused cross-entropy (+ smooth)


class CrossEntropyLS(torch.autograd.Function):
    """Fused log-softmax + NLL with label smoothing; mean over batch.

    Reference: torch.nn.CrossEntropyLoss(label_smoothing=args.smooth) at
    template.py:219,259.
    """

    @staticmethod
    def forward(ctx, logits, targets, smoothing):
        ctx.smoothing = smoothing
        if use_hip(logits):
            loss, probs = ext().ce_fwd(logits, targets, smoothing)
            ctx.save_for_backward(probs, targets)
            return loss
        lf = logits.float()
        logp = F.log_softmax(lf, dim=1)
        n, c = lf.shape
        nll = -logp.gather(1, targets.view(-1, 1)).squeeze(1)
        if smoothing > 0:
            loss = ((1 - smoothing) * nll - smoothing * logp.mean(dim=1)).mean()
        else:
            loss = nll.mean()
        ctx.save_for_backward(logp.exp(), targets)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        probs, targets = ctx.saved_tensors
        s = ctx.smoothing
        if use_hip(probs):
            dlogits = ext().ce_bwd(probs, targets, s, dloss)
            return dlogits, None, None
        n, c = probs.shape
        g = probs.clone()
        g.scatter_add_(1, targets.view(-1, 1), torch.full_like(targets, -1,
                       dtype=g.dtype).view(-1, 1) * (1 - s))
        if s > 0:
            g -= s / c
        dlogits = (g * (dloss / n)).to(probs.dtype)
        return dlogits, None, None


def cross_entropy(logits, targets, smoothing=0.0):
    return CrossEntropyLS.apply(logits, targets, smoothing)


# -------------------------------------------------- fused KD loss (SoftTarget, T=2)


class SoftTargetKD(torch.autograd.Function):
    """Hinton KD: KLDiv(log_softmax(s/T), softmax(t/T), batchmean) * T^2.

    Reference utils.py:121-132 (SoftTarget). Gradient flows to student only.
    """

    @staticmethod
    def forward(ctx, s_logits, t_logits, T):
        ctx.T = T
        if use_hip(s_logits):
            loss, ps, pt = ext().kd_fwd(s_logits, t_logits, T)
            ctx.save_for_backward(ps, pt)
            return loss
        sf, tf = s_logits.float() / T, t_logits.float() / T
        ps = F.softmax(sf, dim=1)
        pt = F.softmax(tf, dim=1)
        logps = F.log_softmax(sf, dim=1)
        logpt = F.log_softmax(tf, dim=1)
        n = sf.shape[0]
        loss = (pt * (logpt - logps)).sum() / n * (T * T)
        ctx.save_for_backward(ps, pt)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        ps, pt = ctx.saved_tensors
        T = ctx.T
        if use_hip(ps):
            ds = ext().kd_bwd(ps, pt, T, dloss)
            return ds, None, None
        n = ps.shape[0]
        ds = ((ps - pt) * (T / n) * dloss)
        return ds, None, None


def kd_loss(s_logits, t_logits, T=2.0):
    return SoftTargetKD.apply(s_logits, t_logits, T)


# ------------------------------------------------------------------------ max pool


class MaxPoolNHWC(torch.autograd.Function):
    """3x3/s2/p1-style max pool over NHWC (ImageNet-stem ResNets)."""

    @staticmethod
    def forward(ctx, x, kernel, stride, padding):
        ctx.params = (kernel, stride, padding)
        ctx.in_shape = x.shape
        if use_hip(x):
            y, idx = ext().maxpool_fwd(x, kernel, stride, padding)
            ctx.save_for_backward(idx)
            return y
        xf = _to_nchw(x).float()
        y, idx = F.max_pool2d(xf, kernel, stride, padding, return_indices=True)
        ctx.save_for_backward(idx)
        return _to_nhwc(y).to(x.dtype)

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        kernel, stride, padding = ctx.params
        N, H, W, C = ctx.in_shape
        dy = dy.contiguous()
        if use_hip(dy):
            return (ext().maxpool_bwd(dy, idx, H, W, kernel, stride, padding),
                    None, None, None)
        dyf = _to_nchw(dy).float()
        # scatter-ADD (max_unpool overwrites on duplicate winners)
        Nn, Cc = dyf.shape[0], dyf.shape[1]
        dxf = dyf.new_zeros(Nn, Cc, H * W)
        dxf.scatter_add_(2, idx.reshape(Nn, Cc, -1), dyf.reshape(Nn, Cc, -1))
        dxf = dxf.reshape(Nn, Cc, H, W)
        return _to_nhwc(dxf).to(dy.dtype), None, None, None


def max_pool(x, kernel=3, stride=2, padding=1):
    return MaxPoolNHWC.apply(x, kernel, stride, padding)


# ----------------------------------------------------- fused WA loss (CE + KD)


class WALoss(torch.autograd.Function):
    """The whole WA training objective in one fused kernel pair:
    loss = CE(logits, targets; smoothing) + lam * SoftTarget(logits[:, :Ck],
    t_logits; T). Reads bf16 logits directly and emits bf16 dlogits (no fp32
    cast round-trips). Only the TOTAL loss output participates in autograd;
    loss_ce / loss_kd are detached (for logging)."""

    @staticmethod
    def forward(ctx, logits, t_logits, targets, smooth, T, lam):
        ctx.params = (smooth, T, lam,
                      0 if t_logits is None else t_logits.shape[1])
        if use_hip(logits):
            total, ce, kd, probs, ps, pt = ext().wa_loss_fwd(
                logits, t_logits, targets, smooth, T, lam)
            ctx.save_for_backward(probs, ps, pt, targets)
            return total, ce, kd
        lf = logits.float()
        logp = F.log_softmax(lf, dim=1)
        nll = -logp.gather(1, targets.view(-1, 1)).squeeze(1)
        ce = ((1 - smooth) * nll - smooth * logp.mean(dim=1)).mean() \
            if smooth > 0 else nll.mean()
        probs = logp.exp()
        if t_logits is not None:
            Ck = t_logits.shape[1]
            sf = lf[:, :Ck] / T
            tf = t_logits.float() / T
            ps, pt = F.softmax(sf, 1), F.softmax(tf, 1)
            kd = (pt * (F.log_softmax(tf, 1) - F.log_softmax(sf, 1))
                  ).sum() / lf.shape[0] * (T * T)
        else:
            ps = pt = lf.new_zeros(lf.shape[0], 1)
            kd = lf.new_zeros(())
        ctx.save_for_backward(probs, ps, pt, targets)
        return ce + lam * kd, ce, kd

    @staticmethod
    def backward(ctx, dtotal, _dce, _dkd):
        # only the TOTAL output participates in training backward; the ce/kd
        # outputs are logging-only (their grads, if any, are ignored)
        probs, ps, pt, targets = ctx.saved_tensors
        smooth, T, lam, Ck = ctx.params
        if use_hip(probs):
            dlogits = ext().wa_loss_bwd(probs, ps, pt, targets, dtotal,
                                        smooth, T, lam, Ck)
            return dlogits, None, None, None, None, None
        M, C = probs.shape
        g = probs - smooth / C
        g.scatter_add_(1, targets.view(-1, 1),
                       torch.full((M, 1), -(1 - smooth), dtype=g.dtype))
        if Ck > 0:
            g[:, :Ck] += lam * T * (ps - pt)
        return (g * (dtotal / M), None, None, None, None, None)


def wa_loss(logits, t_logits, targets, smooth=0.0, T=2.0, lam=0.5):
    return WALoss.apply(logits, t_logits, targets, smooth, T, lam)


# ------------------------------------------------------------------ top-k accuracy


def accuracy(logits, targets, topk=(1,)):
    """Top-k accuracy in percent (reference: timm.utils.accuracy, used
    template.py:179-180, 267-268)."""
    maxk = max(topk)
    if logits.is_cuda and use_hip(logits):
        counts = ext().topk_correct(logits, targets, maxk)  # (maxk,) cumulative
        n = targets.shape[0]
        return [counts[k - 1].item() * 100.0 / n for k in topk]
    _, pred = logits.float().topk(maxk, dim=1)
    correct = pred.eq(targets.view(-1, 1))
    n = targets.shape[0]
    return [correct[:, :k].any(dim=1).float().sum().item() * 100.0 / n for k in topk]
