"""Device-side augmentation — the full train recipe as batched torch ops.

The reference's timm stack (RandAugment ``rand-m9-mstd0.5-inc1`` + color
jitter + random erasing, utils.py:217-229) runs per-sample on CPU workers;
that pipeline cannot feed the ~30k imgs/s MI355X step. This module applies
the same ops BATCHED on the device tensors the GpuTaskLoader already holds:
each op processes the sub-batch that drew it, every sample still gets its
own magnitude/sign, and the semantics mirror cilfw's host pipeline
(cilfw/data/transforms.py — itself the from-scratch timm-contract
implementation; tests/test_device_augment.py checks the device ops against
the same numpy/scipy oracles the host ops use).

Geometric ops use affine_grid/grid_sample over (x - 128) so the zero
padding becomes the 128 constant fill the host ops use. All ops take and
return float tensors in the 0..255 range, NHWC.
"""

import math

import torch
import torch.nn.functional as F


def _nchw(x):
    return x.permute(0, 3, 1, 2)


def _nhwc(x):
    return x.permute(0, 2, 3, 1)


def _affine_sample(imgs, mats, offs):
    """scipy.ndimage.affine_transform semantics, batched: output pixel (i,j)
    samples input at (row, col) = M @ (i, j) + off, bilinear, 128-fill.
    imgs (N,H,W,C) float 0..255; mats (N,2,2), offs (N,2) in pixel units."""
    N, H, W, C = imgs.shape
    dev = imgs.device
    ii = torch.arange(H, device=dev, dtype=torch.float32)
    jj = torch.arange(W, device=dev, dtype=torch.float32)
    gi, gj = torch.meshgrid(ii, jj, indexing="ij")
    # in_r/in_c per batch: (N,H,W)
    in_r = (mats[:, 0, 0, None, None] * gi + mats[:, 0, 1, None, None] * gj
            + offs[:, 0, None, None])
    in_c = (mats[:, 1, 0, None, None] * gi + mats[:, 1, 1, None, None] * gj
            + offs[:, 1, None, None])
    # align_corners=False: pixel p center at normalized (2p+1)/S - 1
    gx = (2.0 * in_c + 1.0) / W - 1.0
    gy = (2.0 * in_r + 1.0) / H - 1.0
    grid = torch.stack([gx, gy], dim=-1)
    shifted = _nchw(imgs - 128.0)
    out = F.grid_sample(shifted, grid, mode="bilinear", padding_mode="zeros",
                        align_corners=False)
    res = (_nhwc(out) + 128.0).clamp_(0, 255).floor_()
    # scipy mode='constant' gives EXACT cval for any coordinate outside
    # [0, n-1] (no edge blending, unlike grid_sample's zero padding)
    outside = ((in_r < 0) | (in_r > H - 1) | (in_c < 0) | (in_c > W - 1))
    return torch.where(outside.unsqueeze(-1), torch.full_like(res, 128.0),
                       res)


def _blend(a, b, factor):
    """factor (N,1,1,1): b + f*(a-b), clamped then truncated — the host ops
    cast to uint8 after every op (numpy astype truncates toward zero)."""
    return (b + factor * (a - b)).clamp_(0, 255).floor_()


def autocontrast(imgs, _mag, _sign):
    lo = imgs.amin(dim=(1, 2), keepdim=True)
    hi = imgs.amax(dim=(1, 2), keepdim=True)
    scale = 255.0 / (hi - lo).clamp_min(1e-6)
    out = (imgs - lo) * scale
    return torch.where(hi > lo, out, imgs).clamp_(0, 255).floor_()


def equalize(imgs, _mag, _sign):
    """PIL-style per-channel equalization (matches transforms._equalize)."""
    N, H, W, C = imgs.shape
    dev = imgs.device
    v = imgs.round().to(torch.int64).clamp_(0, 255)
    # histogram per (image, channel): flatten to one bincount
    chan = torch.arange(C, device=dev).view(1, 1, 1, C)
    img_i = torch.arange(N, device=dev).view(N, 1, 1, 1)
    flat = ((img_i * C + chan) * 256 + v).reshape(-1)
    hist = torch.bincount(flat, minlength=N * C * 256).view(N, C, 256)
    total = hist.sum(dim=2)
    # last nonzero bin's count
    has = hist > 0
    last_idx = 255 - has.flip(dims=[2]).float().argmax(dim=2)
    last_cnt = hist.gather(2, last_idx.unsqueeze(2).long()).squeeze(2)
    nz_count = has.sum(dim=2)
    step = torch.div(total - last_cnt, 255, rounding_mode="floor")
    lut = torch.div(hist.cumsum(dim=2) - hist, step.clamp_min(1).unsqueeze(2),
                    rounding_mode="floor").clamp_(0, 255)
    skip = (step == 0) | (nz_count <= 1)  # identity channels
    ramp = torch.arange(256, device=dev).view(1, 1, 256).expand_as(lut)
    lut = torch.where(skip.unsqueeze(2), ramp, lut)
    out = lut.gather(2, v.permute(0, 3, 1, 2).reshape(N, C, H * W))
    return out.view(N, C, H, W).permute(0, 2, 3, 1).float()


def invert(imgs, _mag, _sign):
    return 255.0 - imgs


def rotate(imgs, mag, sign):
    deg = mag / 10.0 * 30.0 * sign
    th = deg * math.pi / 180.0
    N, H, W, _ = imgs.shape
    cos, sin = torch.cos(th), torch.sin(th)
    # scipy ndi.rotate(reshape=False): rotation about the center in (r,c)
    mats = torch.stack([torch.stack([cos, sin], -1),
                        torch.stack([-sin, cos], -1)], -2)
    cr, cc = (H - 1) / 2.0, (W - 1) / 2.0
    ctr = torch.tensor([cr, cc], device=imgs.device)
    offs = ctr - torch.einsum("nij,j->ni", mats, ctr)
    return _affine_sample(imgs, mats, offs)


def posterize(imgs, mag, _sign):
    bits = (8 - (mag / 10.0 * 4).round()).clamp_min(1)
    q = torch.pow(2.0, 8 - bits).view(-1, 1, 1, 1)
    return torch.div(imgs.round(), q, rounding_mode="floor") * q


def solarize(imgs, mag, _sign):
    thr = 256.0 - (mag / 10.0 * 256).round()
    thr = thr.view(-1, 1, 1, 1)
    r = imgs.round()
    return torch.where(r >= thr, 255.0 - r, r)


def solarize_add(imgs, mag, _sign):
    add = (mag / 10.0 * 110).round().view(-1, 1, 1, 1)
    r = imgs.round()
    return torch.where(r < 128, (r + add).clamp(0, 255), r)


def _enh_factor(mag, sign):
    return (1.0 + sign * mag / 10.0 * 0.9).view(-1, 1, 1, 1)


def color(imgs, mag, sign):
    gray = imgs.round().mean(dim=3, keepdim=True)
    return _blend(imgs.round(), gray, _enh_factor(mag, sign))


def contrast(imgs, mag, sign):
    mean = imgs.round().mean(dim=(1, 2, 3), keepdim=True).floor()
    return _blend(imgs.round(), mean, _enh_factor(mag, sign))


def brightness(imgs, mag, sign):
    return _blend(imgs.round(), torch.zeros_like(imgs), _enh_factor(mag, sign))


def sharpness(imgs, mag, sign):
    # 3x3 [[1,1,1],[1,5,1],[1,1,1]]/13 smoothing by shift-sum — NOT
    # F.conv2d: MIOpen re-tunes the depthwise conv for every new sub-batch
    # shape (~100 ms per fresh shape on MI355X, measured)
    x = _nchw(imgs.round())
    xp = F.pad(x, (1, 1, 1, 1), mode="replicate")
    sm = (xp[..., :-2, :-2] + xp[..., :-2, 1:-1] + xp[..., :-2, 2:]
          + xp[..., 1:-1, :-2] + 5.0 * xp[..., 1:-1, 1:-1]
          + xp[..., 1:-1, 2:]
          + xp[..., 2:, :-2] + xp[..., 2:, 1:-1] + xp[..., 2:, 2:]) / 13.0
    # host blends against the uint8-cast smooth image
    return _blend(_nhwc(x), _nhwc(sm).clamp(0, 255).to(torch.uint8).float(),
                  _enh_factor(mag, sign))


def shear_x(imgs, mag, sign):
    s = (mag / 10.0 * 0.3 * sign)
    N, H, W, _ = imgs.shape
    one = torch.ones_like(s)
    zero = torch.zeros_like(s)
    mats = torch.stack([torch.stack([one, s], -1),
                        torch.stack([zero, one], -1)], -2)
    offs = torch.stack([-s * W / 2.0, zero], -1)
    return _affine_sample(imgs, mats, offs)


def shear_y(imgs, mag, sign):
    s = (mag / 10.0 * 0.3 * sign)
    N, H, W, _ = imgs.shape
    one = torch.ones_like(s)
    zero = torch.zeros_like(s)
    mats = torch.stack([torch.stack([one, zero], -1),
                        torch.stack([s, one], -1)], -2)
    offs = torch.stack([zero, -s * H / 2.0], -1)
    return _affine_sample(imgs, mats, offs)


def translate_x(imgs, mag, sign):
    N, H, W, _ = imgs.shape
    t = mag / 10.0 * 0.45 * W * sign
    eye = torch.eye(2, device=imgs.device).expand(N, 2, 2)
    offs = torch.stack([torch.zeros_like(t), t], -1)
    return _affine_sample(imgs, eye, offs)


def translate_y(imgs, mag, sign):
    N, H, W, _ = imgs.shape
    t = mag / 10.0 * 0.45 * H * sign
    eye = torch.eye(2, device=imgs.device).expand(N, 2, 2)
    offs = torch.stack([t, torch.zeros_like(t)], -1)
    return _affine_sample(imgs, eye, offs)


DEVICE_OPS = [autocontrast, equalize, invert, rotate, posterize, solarize,
              solarize_add, color, contrast, brightness, sharpness, shear_x,
              shear_y, translate_x, translate_y]


class DeviceAugment:
    """Batched RandAugment + color jitter (pre-normalize, uint8-range floats)
    and random erasing (post-normalize), driven by a device generator."""

    def __init__(self, aa_policy="rand-m9-mstd0.5-inc1", num_ops=2,
                 magnitude=9.0, mstd=0.5, color_jitter=0.4, reprob=0.0,
                 remode="pixel", recount=1):
        if aa_policy:
            from .transforms import RandAugment as HostRA
            host = HostRA.from_policy(aa_policy)
            num_ops, magnitude, mstd = host.num_ops, host.magnitude, host.mstd
            self.enabled = True
        else:
            self.enabled = False
        self.num_ops, self.magnitude, self.mstd = num_ops, magnitude, mstd
        self.color_jitter = color_jitter or 0.0
        self.reprob, self.remode, self.recount = reprob or 0.0, remode, recount

    def __call__(self, imgs_u8, g):
        """imgs_u8: (N,H,W,C) uint8 or float tensor; returns float 0..255."""
        imgs = imgs_u8.float()
        N = imgs.shape[0]
        dev = imgs.device
        if self.enabled:
            for _ in range(self.num_ops):
                ops = torch.randint(0, len(DEVICE_OPS), (N,), device=dev,
                                    generator=g)
                mag = torch.full((N,), float(self.magnitude), device=dev)
                if self.mstd > 0:
                    mag = (mag + torch.randn(N, device=dev, generator=g)
                           * self.mstd).clamp(0, 10)
                sign = torch.where(
                    torch.rand(N, device=dev, generator=g) < 0.5, 1.0, -1.0)
                for oi, op in enumerate(DEVICE_OPS):
                    mask = ops == oi
                    if not torch.any(mask):
                        continue
                    idx = mask.nonzero(as_tuple=True)[0]
                    imgs[idx] = op(imgs[idx], mag[idx], sign[idx])
        if self.color_jitter > 0:
            s = self.color_jitter
            r = imgs.round()
            for kind in ("brightness", "contrast", "color"):
                f = (1.0 + (torch.rand(N, device=dev, generator=g) * 2 - 1)
                     * s).view(N, 1, 1, 1)
                if kind == "brightness":
                    base = torch.zeros_like(r)
                elif kind == "contrast":
                    base = r.mean(dim=(1, 2, 3), keepdim=True).floor()
                else:
                    base = r.mean(dim=3, keepdim=True)
                r = _blend(r, base, f)
            imgs = r
        return imgs

    def erase(self, t, g):
        """Random erasing on the NORMALIZED tensor (timm order). t: (N,H,W,C)
        float/bf16 normalized; in-place fill, returns t."""
        if self.reprob <= 0:
            return t
        N, H, W, C = t.shape
        dev = t.device
        doit = torch.rand(N, device=dev, generator=g) < self.reprob
        for _ in range(self.recount):
            area = (H * W) * (torch.rand(N, device=dev, generator=g)
                              * (1 / 3 - 0.02) + 0.02)
            logr = (torch.rand(N, device=dev, generator=g)
                    * (math.log(10 / 3) - math.log(0.3)) + math.log(0.3))
            ar = torch.exp(logr)
            h = torch.sqrt(area * ar).round().long().clamp(1, H - 1)
            w = torch.sqrt(area / ar).round().long().clamp(1, W - 1)
            y = (torch.rand(N, device=dev, generator=g)
                 * (H - h).float()).long()
            x = (torch.rand(N, device=dev, generator=g)
                 * (W - w).float()).long()
            ii = torch.arange(H, device=dev).view(1, H, 1)
            jj = torch.arange(W, device=dev).view(1, 1, W)
            box = ((ii >= y.view(N, 1, 1)) & (ii < (y + h).view(N, 1, 1))
                   & (jj >= x.view(N, 1, 1)) & (jj < (x + w).view(N, 1, 1)))
            box = box & doit.view(N, 1, 1)
            if self.remode == "pixel":
                fill = torch.randn(N, H, W, C, device=dev, generator=g,
                                   dtype=torch.float32).to(t.dtype)
            else:
                fill = torch.zeros(N, H, W, C, device=dev, dtype=t.dtype)
            t = torch.where(box.unsqueeze(-1), fill, t)
        return t
