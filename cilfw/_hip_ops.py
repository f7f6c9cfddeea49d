"""ctypes bridge to the in-tree HIP kernel library (cilfw/_hip_lib.so).

Raises ImportError at import time when the library isn't built — ``ops._backend``
then refuses to run GPU compute (no silent ATen fallback).

Every function allocates outputs with torch (same allocator/stream semantics as
the rest of the program) and launches the hand-written gfx950 kernels on the
CURRENT torch HIP stream, so ordering with surrounding torch ops is implicit and
hipGraph capture of a training step captures these launches too.
"""

import ctypes
import os

import torch

_LIB_PATH = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                         "_hip_lib.so")
if not os.path.exists(_LIB_PATH):
    raise ImportError(f"cilfw HIP library not built: {_LIB_PATH}")
_lib = ctypes.CDLL(_LIB_PATH)

_DEBUG = os.environ.get("CILFW_SYNC_DEBUG") == "1"

c_vp = ctypes.c_void_p
c_i = ctypes.c_int
c_l = ctypes.c_long
c_f = ctypes.c_float

_lib.cilfw_sync.restype = c_i
_lib.cilfw_error_string.restype = ctypes.c_char_p
_lib.cilfw_error_string.argtypes = [c_i]

# Explicit prototypes: ctypes cannot catch arity/type mismatches on its own and
# a wrong call corrupts device memory (see gap_bwd incident, round 1).
_PROTOS = {
    "cilfw_conv2d_fwd": [c_vp] * 4 + [c_i] * 12 + [c_vp, c_vp],
    "cilfw_conv2d_bwd_data": [c_vp] * 4 + [c_i] * 12 + [c_vp] * 5 + [c_i, c_vp],
    "cilfw_conv2d_bwd_weight": [c_vp] * 5 + [c_i] * 13 + [c_vp],
    "cilfw_fill_mtable": [c_vp] + [c_i] * 4 + [c_vp],
    "cilfw_im2col_smallc": [c_vp] * 3 + [c_i] * 11 + [c_vp],
    "cilfw_bn_apply_only": [c_vp] * 7 + [c_l, c_i, c_i, c_vp],
    "cilfw_bn_fwd": [c_vp] * 11 + [c_i, c_l, c_i, c_f, c_f, c_i, c_i, c_vp],
    "cilfw_bn_bwd": [c_vp] * 11 + [c_vp, c_i] + [c_l, c_i, c_i, c_i, c_vp],
    "cilfw_add_relu_fwd": [c_vp] * 3 + [c_l, c_vp],
    "cilfw_add_relu_bwd": [c_vp] * 3 + [c_l, c_vp],
    "cilfw_downsample_a_fwd": [c_vp] * 2 + [c_i] * 4 + [c_vp],
    "cilfw_downsample_a_bwd": [c_vp] * 2 + [c_i] * 4 + [c_vp],
    "cilfw_gap_fwd": [c_vp] * 2 + [c_i] * 3 + [c_vp],
    "cilfw_gap_bwd": [c_vp] * 2 + [c_i] * 3 + [c_vp],
    "cilfw_conv2d_bwd_data_sub": [c_vp] * 4 + [c_i] * 12 + [c_vp],
    "cilfw_parity_scatter": [c_vp] * 2 + [c_i] * 8 + [c_vp],
    "cilfw_stride_gather": [c_vp] * 2 + [c_i] * 7 + [c_vp],
    "cilfw_stride_scatter": [c_vp] * 2 + [c_i] * 7 + [c_vp],
    "cilfw_maxpool_fwd": [c_vp] * 3 + [c_i] * 9 + [c_vp],
    "cilfw_maxpool_bwd": [c_vp] * 3 + [c_i] * 9 + [c_vp],
    "cilfw_linear_fwd": [c_vp] * 5 + [c_i] * 4 + [c_vp],
    "cilfw_linear_dx": [c_vp] * 4 + [c_i] * 4 + [c_vp],
    "cilfw_linear_dw": [c_vp] * 5 + [c_i] * 4 + [c_vp],
    "cilfw_ce_fwd": [c_vp] * 5 + [c_i, c_i, c_f, c_vp],
    "cilfw_ce_bwd": [c_vp] * 4 + [c_i, c_i, c_f, c_vp],
    "cilfw_kd_fwd": [c_vp] * 6 + [c_i, c_i, c_f, c_vp],
    "cilfw_kd_bwd": [c_vp] * 4 + [c_i, c_i, c_f, c_vp],
    "cilfw_wa_loss_fwd": [c_vp] * 8 + [c_i] * 3 + [c_f] * 3 + [c_vp],
    "cilfw_wa_loss_bwd": [c_vp] * 6 + [c_i] * 3 + [c_f] * 3 + [c_vp],
    "cilfw_sgd_step": [c_vp] * 4 + [c_l, c_vp, c_f, c_f, c_vp],
    "cilfw_topk_correct": [c_vp] * 3 + [c_i] * 3 + [c_vp],
    "cilfw_herding_select": [c_vp] * 3 + [c_i] * 3 + [c_vp],
    "cilfw_herding_select_batch": [c_vp] * 6 + [c_i] * 3 + [c_vp],
}
for _name, _args in _PROTOS.items():
    _fn = getattr(_lib, _name)
    _fn.argtypes = _args
    _fn.restype = None

_lib.cilfw_conv2d_bwd_data.restype = c_i  # 1 = BN partials were emitted

for _name, _n in [("cilfw_linear_ksplit", 3),
                  ("cilfw_conv2d_fwd_ksplit", 7),
                  ("cilfw_conv2d_bwd_data_ksplit", 7),
                  ("cilfw_conv2d_bwd_data_can_fuse_bn", 8),
                  ("cilfw_conv2d_bwd_data_bn_gy", 3),
                  ("cilfw_conv2d_bwd_weight_nslices", 7)]:
    _fn = getattr(_lib, _name)
    _fn.argtypes = [c_i] * _n
    _fn.restype = c_i


def _stream():
    return c_vp(torch.cuda.current_stream().cuda_stream)


def _ptr(t):
    return c_vp(0 if t is None else t.data_ptr())


def _check(name):
    if _DEBUG:
        e = _lib.cilfw_sync()
        if e != 0:
            raise RuntimeError(
                f"HIP error after {name}: "
                f"{_lib.cilfw_error_string(e).decode()}")


def _bf16(t, name):
    assert t.dtype == torch.bfloat16, \
        f"{name}: GPU compute path is bf16 (got {t.dtype}); run with " \
        f"--dtype bf16 or set CILFW_FORCE_TORCH=1 for debugging"
    assert t.is_contiguous(), f"{name}: needs contiguous tensor"
    return t


# ------------------------------------------------------------------------ conv

def _stem_cols(x, stride, pad, R, S, CRSpad):
    """Padded im2col for tiny-C stems -> (M, CRSpad) bf16 (pre-zeroed)."""
    N, H, W_, C = x.shape
    Ho = (H + 2 * pad - R) // stride + 1
    Wo = (W_ + 2 * pad - S) // stride + 1
    M = N * Ho * Wo
    mt = _mtable(N, Ho, Wo, stride, x.device)
    col = torch.zeros(M, CRSpad, dtype=torch.bfloat16, device=x.device)
    _lib.cilfw_im2col_smallc(_ptr(x), _ptr(mt), _ptr(col), c_i(N), c_i(H),
                             c_i(W_), c_i(C), c_i(R), c_i(S), c_i(stride),
                             c_i(pad), c_i(Ho), c_i(Wo), c_i(CRSpad),
                             _stream())
    _check("im2col_smallc")
    return col, Ho, Wo


def conv2d_fwd(x, w, stride, pad, want_bn_parts=False):
    """want_bn_parts: also produce per-M-block per-channel (sum, sumsq)
    partials of y in the conv epilogue — a following training-mode BN skips
    its own bn_sums pass (and a full re-read of y). Returns (y, parts|None);
    parts is None when the shape routed off the v2 kernel or used split-K."""
    _bf16(x, "conv2d_fwd.x")
    _bf16(w, "conv2d_fwd.w")
    N, H, W_, C = x.shape
    R, S, Cw, K = w.shape
    assert Cw == C
    assert K % 8 == 0, "conv kernels vector-stage over output channels (K%8==0)"
    Ho = (H + 2 * pad - R) // stride + 1
    Wo = (W_ + 2 * pad - S) // stride + 1
    if C < 16 and R * S > 1 and K % 16 == 0 \
            and os.environ.get("CILFW_STEM_IM2COL") == "1":
        # measured: the generic gather already fills the chip at stem sizes —
        # this alternate path is kept for experiments (CILFW_STEM_IM2COL=1)
        CRS = C * R * S
        CRSpad = (CRS + 15) // 16 * 16
        col, Ho, Wo = _stem_cols(x, stride, pad, R, S, CRSpad)
        M = N * Ho * Wo
        wpad = torch.zeros(CRSpad, K, dtype=torch.bfloat16, device=x.device)
        wpad[:CRS] = w.reshape(CRS, K)
        y = torch.empty(M, K, dtype=torch.bfloat16, device=x.device)
        ks = _lib.cilfw_conv2d_fwd_ksplit(M, CRSpad, K, 1, 1, 1, 1)
        ws = (torch.empty(ks * M * K, dtype=torch.float32, device=x.device)
              if ks > 1 else None)
        _lib.cilfw_conv2d_fwd(_ptr(col), _ptr(wpad), _ptr(y), _ptr(ws),
                              c_i(M), c_i(1), c_i(1), c_i(CRSpad), c_i(K),
                              c_i(1), c_i(1), c_i(1), c_i(0), c_i(1), c_i(1),
                              c_i(ks), _ptr(None), _stream())
        _check("conv2d_fwd_stem")
        return y.view(N, Ho, Wo, K), None
    y = torch.empty(N, Ho, Wo, K, dtype=torch.bfloat16, device=x.device)
    ks = _lib.cilfw_conv2d_fwd_ksplit(N, C, K, R, S, Ho, Wo)
    ws = (torch.empty(ks * N * Ho * Wo * K, dtype=torch.float32,
                      device=x.device) if ks > 1 else None)
    parts = None
    if want_bn_parts and ks == 1 and C % 8 == 0             and os.environ.get("CILFW_CONV_V2", "1") != "0"             and os.environ.get("CILFW_CONV_BN_FUSE", "1") != "0":
        gy = (N * Ho * Wo + 127) // 128
        parts = torch.empty(gy * 2 * K, dtype=torch.float32,
                            device=x.device)
    _lib.cilfw_conv2d_fwd(_ptr(x), _ptr(w), _ptr(y), _ptr(ws), c_i(N),
                          c_i(H), c_i(W_), c_i(C), c_i(K), c_i(R), c_i(S),
                          c_i(stride), c_i(pad), c_i(Ho), c_i(Wo), c_i(ks),
                          _ptr(parts), _stream())
    _check("conv2d_fwd")
    return y, parts


def conv2d_bwd_data(dy, w, stride, pad, H, W_, bn_meta=None):
    """bn_meta = (bn_y, bn_x, bn_mean, bn_invstd, bn_relu): this conv's input
    is a training BN(+ReLU) output with a single consumer — the kernel
    epilogue then also emits that BN's backward (dgamma, dbeta) partials so
    the BN can skip its sums pass. Returns (dx, parts-or-None) when bn_meta
    is given, plain dx otherwise."""
    _bf16(dy, "conv2d_bwd_data.dy")
    N, Ho, Wo, K = dy.shape
    R, S, C, Kw = w.shape
    assert Kw == K
    if R == 1 and S == 1 and stride > 1:
        # strided 1x1 projection: 3/4 of positions are structurally zero —
        # solve on the subsampled (stride-1) grid, then scatter
        dxs = conv2d_bwd_data(dy, w, 1, 0, Ho, Wo)
        dx = torch.empty(N, H, W_, C, dtype=torch.bfloat16, device=dy.device)
        _lib.cilfw_stride_scatter(_ptr(dxs), _ptr(dx), c_i(N), c_i(H),
                                  c_i(W_), c_i(C), c_i(stride), c_i(Ho),
                                  c_i(Wo), _stream())
        _check("stride_scatter")
        # keep the return shape contract: callers that passed bn_meta unpack
        # a (dx, parts) pair (no fusion on the strided-1x1 scatter path)
        return (dx, None) if bn_meta is not None else dx
    # NOTE: a 4-way parity decomposition of stride-2 bwd-data was measured
    # SLOWER than the single zero-structured kernel (12 small launches vs one
    # big one) — kernels kept (cilfw_conv2d_bwd_data_sub/parity_scatter) for a
    # future fused single-launch variant.
    dx = torch.empty(N, H, W_, C, dtype=torch.bfloat16, device=dy.device)
    ks = _lib.cilfw_conv2d_bwd_data_ksplit(N, H, W_, C, K, R, S)
    ws = (torch.empty(ks * N * H * W_ * C, dtype=torch.float32,
                      device=dy.device) if ks > 1 else None)
    parts = None
    if (bn_meta is not None
            and _lib.cilfw_conv2d_bwd_data_can_fuse_bn(
                N, H, W_, C, K, R, S, stride)):
        gy = _lib.cilfw_conv2d_bwd_data_bn_gy(N, H, W_)
        parts = torch.empty(gy, 2, C, dtype=torch.float32, device=dy.device)
    if parts is not None:
        bn_y, bn_x, bn_mean, bn_invstd, bn_relu = bn_meta
        fused = _lib.cilfw_conv2d_bwd_data(
            _ptr(dy), _ptr(w), _ptr(dx), _ptr(ws), c_i(N),
            c_i(H), c_i(W_), c_i(C), c_i(K), c_i(R),
            c_i(S), c_i(stride), c_i(pad), c_i(Ho),
            c_i(Wo), c_i(ks), _ptr(bn_y), _ptr(bn_x), _ptr(bn_mean),
            _ptr(bn_invstd), _ptr(parts), c_i(1 if bn_relu else 0),
            _stream())
        if not fused:
            parts = None
    else:
        _lib.cilfw_conv2d_bwd_data(
            _ptr(dy), _ptr(w), _ptr(dx), _ptr(ws), c_i(N),
            c_i(H), c_i(W_), c_i(C), c_i(K), c_i(R),
            c_i(S), c_i(stride), c_i(pad), c_i(Ho),
            c_i(Wo), c_i(ks), c_vp(0), c_vp(0), c_vp(0), c_vp(0), c_vp(0),
            c_i(0), _stream())
    _check("conv2d_bwd_data")
    if bn_meta is not None:
        return dx, parts
    return dx


_mtable_cache = {}


def _mtable(N, Ho, Wo, stride, device):
    """Packed im2col pixel table mt[m] = n<<20 | ho*stride<<10 | wo*stride,
    built once per conv geometry and cached for the process lifetime."""
    key = (N, Ho, Wo, stride, device)
    mt = _mtable_cache.get(key)
    if mt is None:
        assert N < 4096 and Ho * stride < 1024 and Wo * stride < 1024, \
            "mtable packing limits exceeded"
        M = N * Ho * Wo
        mt = torch.empty(M, dtype=torch.int32, device=device)
        _lib.cilfw_fill_mtable(_ptr(mt), c_i(M), c_i(Ho * Wo), c_i(Wo),
                               c_i(stride), _stream())
        _mtable_cache[key] = mt
    return mt


def conv2d_bwd_weight(dy, x, stride, pad, R, S, out=None, accum=False):
    """dW (R,S,C,K) fp32. With ``out`` the reduce writes straight into the
    given buffer (the DataParallelEngine's flat-grad slot — no AccumulateGrad
    add); ``accum=True`` adds instead of overwriting (grad accumulation)."""
    _bf16(dy, "conv2d_bwd_weight.dy")
    _bf16(x, "conv2d_bwd_weight.x")
    N, H, W_, C = x.shape
    _, Ho, Wo, K = dy.shape
    # 7x7 stems (ImageNet): the per-element gather path measured 542 us on
    # the 224^2 stem dW; padded-im2col + flat fast-path GEMM is far cheaper
    # there (3x3 CIFAR stems measured better on the direct gather)
    use_im2col = (os.environ.get("CILFW_STEM_IM2COL") == "1"
                  or (R * S >= 25
                      and os.environ.get("CILFW_STEM_IM2COL") != "0"))
    if C < 16 and R * S > 1 and K % 16 == 0 and use_im2col:
        CRS = C * R * S
        CRSpad = (CRS + 15) // 16 * 16
        col, _, _ = _stem_cols(x, stride, pad, R, S, CRSpad)
        M = N * Ho * Wo
        dyf = dy.reshape(M, K)
        dwp = torch.empty(CRSpad, K, dtype=torch.float32, device=x.device)
        ns = _lib.cilfw_conv2d_bwd_weight_nslices(M, CRSpad, K, 1, 1, 1, 1)
        ws = torch.empty(ns * CRSpad * K, dtype=torch.float32,
                         device=x.device)
        _lib.cilfw_conv2d_bwd_weight(_ptr(dyf), _ptr(col), _ptr(None),
                                     _ptr(dwp), _ptr(ws), c_i(M), c_i(1),
                                     c_i(1), c_i(CRSpad), c_i(K), c_i(1),
                                     c_i(1), c_i(1), c_i(0), c_i(1), c_i(1),
                                     c_i(ns), c_i(0), _stream())
        _check("conv2d_bwd_weight_stem")
        dw = dwp[:CRS].reshape(R, S, C, K)
        if out is None:
            return dw
        (out.add_(dw) if accum else out.copy_(dw))
        return out
    if R == 1 and S == 1 and stride > 1:
        xg = torch.empty(N, Ho, Wo, C, dtype=torch.bfloat16, device=x.device)
        _lib.cilfw_stride_gather(_ptr(x), _ptr(xg), c_i(N), c_i(H), c_i(W_),
                                 c_i(C), c_i(stride), c_i(Ho), c_i(Wo),
                                 _stream())
        _check("stride_gather")
        return conv2d_bwd_weight(dy, xg, 1, 0, R, S, out=out, accum=accum)
    mt = _mtable(N, Ho, Wo, stride, dy.device)
    if out is None:
        dw = torch.empty(R, S, C, K, dtype=torch.float32, device=dy.device)
        accum = False
    else:
        assert out.is_contiguous() and out.dtype == torch.float32 \
            and out.shape == (R, S, C, K)
        dw = out
    ns = _lib.cilfw_conv2d_bwd_weight_nslices(N, C, K, R, S, Ho, Wo)
    ws = torch.empty(ns * R * S * C * K, dtype=torch.float32,
                     device=dy.device)
    _lib.cilfw_conv2d_bwd_weight(_ptr(dy), _ptr(x), _ptr(mt), _ptr(dw),
                                 _ptr(ws), c_i(N), c_i(H), c_i(W_), c_i(C),
                                 c_i(K), c_i(R), c_i(S), c_i(stride),
                                 c_i(pad), c_i(Ho), c_i(Wo), c_i(ns),
                                 c_i(1 if accum else 0), _stream())
    _check("conv2d_bwd_weight")
    return dw


# -------------------------------------------------------------------------- bn

def bn_fwd(x, gamma, beta, running_mean, running_var, momentum, eps, training,
           relu, residual=None, ext_parts=None, ext_gy=0):
    """ext_parts/ext_gy: per-M-block (sum, sumsq) partials produced by the
    conv that made x (conv2d_fwd want_bn_parts) — skips the bn_sums pass."""
    _bf16(x, "bn_fwd.x")
    if residual is not None:
        _bf16(residual, "bn_fwd.residual")
    frozen = (None if training
              else getattr(running_mean, "_cilfw_frozen", None))
    if frozen is not None:
        mean, invstd = frozen
        C = x.shape[-1]
        M = x.numel() // C
        y = torch.empty_like(x)
        gf = gamma.float().contiguous()
        bf = beta.float().contiguous()
        _lib.cilfw_bn_apply_only(_ptr(x), _ptr(y), _ptr(residual), _ptr(gf),
                                 _ptr(bf), _ptr(mean), _ptr(invstd),
                                 c_l(M * C), c_i(C),
                                 c_i(1 if relu else 0), _stream())
        _check("bn_apply_only")
        return y, mean, invstd
    C = x.shape[-1]
    assert C % 8 == 0, "bn kernels vectorize over channels (C%8==0)"
    M = x.numel() // C
    y = torch.empty_like(x)
    mean = torch.empty(C, dtype=torch.float32, device=x.device)
    invstd = torch.empty(C, dtype=torch.float32, device=x.device)
    gy = (M + 255) // 256  # keep in sync with rows_per_blk=256 in norm.hip
    scratch = torch.empty((gy + 1) * 2 * C, dtype=torch.float32,
                          device=x.device)
    gf = gamma.float().contiguous()
    bf = beta.float().contiguous()
    if not training:
        ext_parts = None
    _lib.cilfw_bn_fwd(_ptr(x), _ptr(y), _ptr(residual), _ptr(gf), _ptr(bf),
                      _ptr(running_mean), _ptr(running_var), _ptr(mean),
                      _ptr(invstd), _ptr(scratch), _ptr(ext_parts),
                      c_i(ext_gy), c_l(M), c_i(C),
                      c_f(momentum), c_f(eps), c_i(1 if training else 0),
                      c_i(1 if relu else 0), _stream())
    _check("bn_fwd")
    return y, mean, invstd


def bn_bwd(dy, x, gamma, mean, invstd, y, relu, training, want_dres=False,
           out_gamma=None, out_beta=None, ext_parts=None):
    """With out_gamma/out_beta the reduced per-channel grads are written
    straight into the given fp32 buffers (flat-grad slot delivery) and the
    returned dgamma/dbeta are those buffers. ext_parts: [gy][2][C]
    (dgamma, dbeta) partials already emitted by the producing conv's
    bwd-data epilogue — the sums pass over (dy, x, y) is skipped."""
    _bf16(dy, "bn_bwd.dy")
    C = x.shape[-1]
    M = x.numel() // C
    dx = torch.empty_like(x)
    dres = torch.empty_like(x) if want_dres else None
    if ext_parts is not None:
        ext_gy = ext_parts.shape[0]
        dgb = torch.empty(2 * C, dtype=torch.float32, device=x.device)
        base = 0
    else:
        ext_gy = 0
        gy = (M + 255) // 256  # keep in sync with rows_per_blk in norm.hip
        dgb = torch.empty((gy + 1) * 2 * C, dtype=torch.float32,
                          device=x.device)
        base = gy * 2 * C
    gf = gamma.float().contiguous()
    _lib.cilfw_bn_bwd(_ptr(dy), _ptr(x), _ptr(y), _ptr(dx), _ptr(dres),
                      _ptr(gf), _ptr(mean), _ptr(invstd), _ptr(dgb),
                      _ptr(out_gamma), _ptr(out_beta),
                      _ptr(ext_parts), c_i(ext_gy),
                      c_l(M), c_i(C), c_i(1 if relu else 0),
                      c_i(1 if training else 0), _stream())
    _check("bn_bwd")
    dgamma = out_gamma if out_gamma is not None else dgb[base:base + C]
    dbeta = out_beta if out_beta is not None else dgb[base + C:base + 2 * C]
    return dx, dgamma, dbeta, dres


# ------------------------------------------------------------------ elementwise

def add_relu_fwd(a, b):
    _bf16(a, "add_relu.a")
    y = torch.empty_like(a)
    _lib.cilfw_add_relu_fwd(_ptr(a), _ptr(b), _ptr(y), c_l(a.numel()),
                            _stream())
    _check("add_relu_fwd")
    return y


def add_relu_bwd(dy, y):
    da = torch.empty_like(dy)
    _lib.cilfw_add_relu_bwd(_ptr(dy), _ptr(y), _ptr(da), c_l(dy.numel()),
                            _stream())
    _check("add_relu_bwd")
    return da


def downsample_a_fwd(x):
    _bf16(x, "downsample_a.x")
    N, H, W_, C = x.shape
    y = torch.empty(N, H // 2, W_ // 2, 2 * C, dtype=x.dtype, device=x.device)
    _lib.cilfw_downsample_a_fwd(_ptr(x), _ptr(y), c_i(N), c_i(H), c_i(W_),
                                c_i(C), _stream())
    _check("downsample_a_fwd")
    return y


def downsample_a_bwd(dy, H, W_):
    N = dy.shape[0]
    C = dy.shape[3] // 2
    dx = torch.empty(N, H, W_, C, dtype=dy.dtype, device=dy.device)
    _lib.cilfw_downsample_a_bwd(_ptr(dy), _ptr(dx), c_i(N), c_i(H), c_i(W_),
                                c_i(C), _stream())
    _check("downsample_a_bwd")
    return dx


def gap_fwd(x):
    _bf16(x, "gap.x")
    N, H, W_, C = x.shape
    y = torch.empty(N, C, dtype=x.dtype, device=x.device)
    _lib.cilfw_gap_fwd(_ptr(x), _ptr(y), c_i(N), c_i(H * W_), c_i(C),
                       _stream())
    _check("gap_fwd")
    return y


def gap_bwd(dy, H, W_):
    N, C = dy.shape
    dx = torch.empty(N, H, W_, C, dtype=dy.dtype, device=dy.device)
    _lib.cilfw_gap_bwd(_ptr(dy), _ptr(dx), c_i(N), c_i(H * W_), c_i(C),
                       _stream())
    _check("gap_bwd")
    return dx


def maxpool_fwd(x, kernel, stride, pad):
    _bf16(x, "maxpool.x")
    N, H, W_, C = x.shape
    Ho = (H + 2 * pad - kernel) // stride + 1
    Wo = (W_ + 2 * pad - kernel) // stride + 1
    y = torch.empty(N, Ho, Wo, C, dtype=x.dtype, device=x.device)
    idx = torch.empty(N, Ho, Wo, C, dtype=torch.int32, device=x.device)
    _lib.cilfw_maxpool_fwd(_ptr(x), _ptr(y), _ptr(idx), c_i(N), c_i(H),
                           c_i(W_), c_i(C), c_i(kernel), c_i(stride),
                           c_i(pad), c_i(Ho), c_i(Wo), _stream())
    _check("maxpool_fwd")
    return y, idx


def maxpool_bwd(dy, idx, H, W_, kernel, stride, pad):
    N, Ho, Wo, C = dy.shape
    dx = torch.empty(N, H, W_, C, dtype=dy.dtype, device=dy.device)
    _lib.cilfw_maxpool_bwd(_ptr(dy), _ptr(idx), _ptr(dx), c_i(N), c_i(H),
                           c_i(W_), c_i(C), c_i(kernel), c_i(stride),
                           c_i(pad), c_i(Ho), c_i(Wo), _stream())
    _check("maxpool_bwd")
    return dx


# ---------------------------------------------------------------------- linear

def linear_fwd(x, w, bias):
    _bf16(x, "linear.x")
    _bf16(w, "linear.w")
    M, K = x.shape
    N, _ = w.shape
    y = torch.empty(M, N, dtype=torch.bfloat16, device=x.device)
    bf = bias.float().contiguous() if bias is not None else None
    ks = _lib.cilfw_linear_ksplit(M, N, K)
    ws = (torch.empty(ks * M * N, dtype=torch.float32, device=x.device)
          if ks > 1 else None)
    _lib.cilfw_linear_fwd(_ptr(x), _ptr(w), _ptr(bf), _ptr(y), _ptr(ws),
                          c_i(M), c_i(N), c_i(K), c_i(ks), _stream())
    _check("linear_fwd")
    return y


def linear_bwd(dy, x, w, has_bias):
    _bf16(dy, "linear_bwd.dy")
    M, K = x.shape
    N, _ = w.shape
    dx = torch.empty(M, K, dtype=torch.bfloat16, device=x.device)
    dw = torch.empty(N, K, dtype=torch.float32, device=x.device)
    db = torch.empty(N, dtype=torch.float32, device=x.device) if has_bias \
        else None
    ks_dx = _lib.cilfw_linear_ksplit(M, K, N)
    ws_dx = (torch.empty(ks_dx * M * K, dtype=torch.float32, device=x.device)
             if ks_dx > 1 else None)
    _lib.cilfw_linear_dx(_ptr(dy), _ptr(w), _ptr(dx), _ptr(ws_dx), c_i(M),
                         c_i(N), c_i(K), c_i(ks_dx), _stream())
    ks_dw = _lib.cilfw_linear_ksplit(N, K, M)
    ws_dw = (torch.empty(ks_dw * N * K, dtype=torch.float32, device=x.device)
             if ks_dw > 1 else None)
    _lib.cilfw_linear_dw(_ptr(dy), _ptr(x), _ptr(dw), _ptr(db), _ptr(ws_dw),
                         c_i(M), c_i(N), c_i(K), c_i(ks_dw), _stream())
    _check("linear_bwd")
    return dx, dw, db


# ---------------------------------------------------------------------- losses

def ce_fwd(logits, targets, smooth):
    logits = logits.float().contiguous()
    M, C = logits.shape
    probs = torch.empty_like(logits)
    loss = torch.empty((), dtype=torch.float32, device=logits.device)
    rowloss = torch.empty(M, dtype=torch.float32, device=logits.device)
    _lib.cilfw_ce_fwd(_ptr(logits), _ptr(targets.contiguous()), _ptr(probs),
                      _ptr(loss), _ptr(rowloss), c_i(M), c_i(C), c_f(smooth),
                      _stream())
    _check("ce_fwd")
    return loss, probs


def ce_bwd(probs, targets, smooth, dloss):
    M, C = probs.shape
    dlogits = torch.empty_like(probs)
    dl = dloss.float().reshape(1).contiguous()
    _lib.cilfw_ce_bwd(_ptr(probs), _ptr(targets.contiguous()), _ptr(dl),
                      _ptr(dlogits), c_i(M), c_i(C), c_f(smooth), _stream())
    _check("ce_bwd")
    return dlogits


def kd_fwd(s_logits, t_logits, T):
    s = s_logits.float().contiguous()
    t = t_logits.float().contiguous()
    M, C = s.shape
    ps = torch.empty_like(s)
    pt = torch.empty_like(s)
    loss = torch.empty((), dtype=torch.float32, device=s.device)
    rowloss = torch.empty(M, dtype=torch.float32, device=s.device)
    _lib.cilfw_kd_fwd(_ptr(s), _ptr(t), _ptr(ps), _ptr(pt), _ptr(loss),
                      _ptr(rowloss), c_i(M), c_i(C), c_f(T), _stream())
    _check("kd_fwd")
    return loss, ps, pt


def kd_bwd(ps, pt, T, dloss):
    M, C = ps.shape
    ds = torch.empty_like(ps)
    dl = dloss.float().reshape(1).contiguous()
    _lib.cilfw_kd_bwd(_ptr(ps), _ptr(pt), _ptr(dl), _ptr(ds), c_i(M), c_i(C),
                      c_f(T), _stream())
    _check("kd_bwd")
    return ds


def wa_loss_fwd(s_logits, t_logits, targets, smooth, T, lam):
    """Fused CE(+smoothing) + lambda*KD over bf16 logits. t_logits may be
    None (plain CE). Returns (loss_total, loss_ce, loss_kd, probs, ps, pt)."""
    _bf16(s_logits, "wa_loss.s")
    M, C = s_logits.shape
    Ck = 0
    if t_logits is not None:
        _bf16(t_logits, "wa_loss.t")
        Ck = t_logits.shape[1]
    dev = s_logits.device
    probs = torch.empty(M, C, dtype=torch.float32, device=dev)
    ps = torch.empty(M, max(Ck, 1), dtype=torch.float32, device=dev)
    pt = torch.empty(M, max(Ck, 1), dtype=torch.float32, device=dev)
    rl2 = torch.empty(2 * M, dtype=torch.float32, device=dev)
    out3 = torch.empty(3, dtype=torch.float32, device=dev)
    _lib.cilfw_wa_loss_fwd(_ptr(s_logits), _ptr(t_logits),
                           _ptr(targets.contiguous()), _ptr(probs), _ptr(ps),
                           _ptr(pt), _ptr(rl2), _ptr(out3), c_i(M), c_i(C),
                           c_i(Ck), c_f(smooth), c_f(T), c_f(lam), _stream())
    _check("wa_loss_fwd")
    return out3[2], out3[0], out3[1], probs, ps, pt


def wa_loss_bwd(probs, ps, pt, targets, dtotal, smooth, T, lam, Ck):
    M, C = probs.shape
    dlogits = torch.empty(M, C, dtype=torch.bfloat16, device=probs.device)
    dl = dtotal.float().reshape(1).contiguous()
    _lib.cilfw_wa_loss_bwd(_ptr(probs), _ptr(ps), _ptr(pt),
                           _ptr(targets.contiguous()), _ptr(dl),
                           _ptr(dlogits), c_i(M), c_i(C), c_i(Ck),
                           c_f(smooth), c_f(T), c_f(lam), _stream())
    _check("wa_loss_bwd")
    return dlogits


# ------------------------------------------------------------- optimizer / misc

def sgd_step(p, g, m, lr_dev, momentum, wd, p_bf16=None):
    """lr_dev: 1-elem fp32 DEVICE tensor (graph-replayable lr input);
    a plain float is boxed into one for convenience."""
    if not torch.is_tensor(lr_dev):
        lr_dev = torch.full((1,), float(lr_dev), dtype=torch.float32,
                            device=p.device)
    _lib.cilfw_sgd_step(_ptr(p), _ptr(g), _ptr(m), _ptr(p_bf16),
                        c_l(p.numel()), _ptr(lr_dev), c_f(momentum), c_f(wd),
                        _stream())
    _check("sgd_step")


def topk_correct(logits, targets, maxk):
    lf = logits.float().contiguous()
    M, C = lf.shape
    counts = torch.empty(maxk, dtype=torch.int64, device=lf.device)
    _lib.cilfw_topk_correct(_ptr(lf), _ptr(targets.contiguous()),
                            _ptr(counts), c_i(M), c_i(C), c_i(maxk),
                            _stream())
    _check("topk_correct")
    return counts


def herding_select(f, mu, m):
    n, D = f.shape
    assert (n + D) * 4 <= 60_000, \
        "herding_select kernel supports n+D <= 15000 (per-class sample sets)"
    order = torch.empty(m, dtype=torch.int64, device=f.device)
    _lib.cilfw_herding_select(_ptr(f), _ptr(mu), _ptr(order), c_i(n), c_i(D),
                              c_i(m), _stream())
    _check("herding_select")
    return order


def herding_select_batch(feats_list, m_list):
    """All classes' greedy herding in ONE launch (block per class).
    feats_list: list of (n_c, D) fp32 GPU tensors; returns list of int64
    ranked-index tensors (local indices, length m_c)."""
    device = feats_list[0].device
    D = feats_list[0].shape[1]
    ns = [int(f.shape[0]) for f in feats_list]
    assert all((n + D) * 4 <= 60_000 for n in ns), "herding LDS limit"
    fall = torch.cat([f.float().contiguous() for f in feats_list])
    mus = torch.stack([f.float().mean(0) for f in feats_list]).contiguous()
    foff = torch.tensor([0] + list(torch.tensor(ns).cumsum(0)),
                        dtype=torch.int32, device=device)
    ms = [min(m, n) for m, n in zip(m_list, ns)]
    ooff = torch.tensor([0] + list(torch.tensor(ms).cumsum(0)),
                        dtype=torch.int32, device=device)
    mvec = torch.tensor(ms, dtype=torch.int32, device=device)
    order = torch.empty(sum(ms), dtype=torch.int64, device=device)
    _lib.cilfw_herding_select_batch(_ptr(fall), _ptr(foff), _ptr(mus),
                                    _ptr(ooff), _ptr(mvec), _ptr(order),
                                    c_i(len(ns)), c_i(D), c_i(max(ns)),
                                    _stream())
    _check("herding_select_batch")
    outs, o = [], 0
    for m in ms:
        outs.append(order[o:o + m])
        o += m
    return outs
