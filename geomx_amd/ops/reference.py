"""Pure-torch fp32 reference implementations of the kvstore hot-path ops.

These define the SEMANTICS (matching the GeoMX reference algorithms) and
serve three purposes:
  1. golden references for the HIP kernels (tests compare HIP vs these),
  2. the CPU execution path (gloo backend, no GPU),
  3. documentation of each algorithm.

Reference semantics sources (behavioral, re-implemented — not copied):
  - 2bit quantization with error feedback:
      /root/reference/src/kvstore/gradient_compression-inl.h:40-139
  - Bi-Sparse compress / pull-compress / decompress:
      /root/reference/src/kvstore/gradient_compression.cc:191-336
  - DGT 4-bit linear quantization with residual:
      /root/reference/3rdparty/ps-lite/src/van.cc:750-824
  - DCASGD optimizer:
      /root/reference/python/mxnet/optimizer/optimizer.py:872-926
  - fused SGD/Adam updates:
      /root/reference/src/operator/optimizer_op.cc:43-651
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

BSC_PLACEHOLDER = -65530.0
BSC_MOMENTUM = 0.9


# ---------------------------------------------------------------------------
# 2bit: threshold-sign quantization, 16 fp32 -> 1 uint32, error feedback
# ---------------------------------------------------------------------------
# Code per element (2 bits, little-endian slot order inside each uint32):
#   0b11 -> +threshold ; 0b10 -> -threshold ; 0b00 -> 0
# residual accumulates the quantization error (error feedback):
#   residual += grad; emit sign(residual) * threshold where |residual| >= threshold
#   and subtract the emitted value from residual.

def quantized_words(n: int) -> int:
    return (n + 15) // 16


def quantize_2bit(grad: torch.Tensor, residual: torch.Tensor,
                  threshold: float) -> torch.Tensor:
    """Returns packed int32 tensor of ceil(N/16) words. Mutates residual."""
    g = grad.reshape(-1).float()
    r = residual.reshape(-1)
    n = g.numel()
    r += g
    pos = r >= threshold
    neg = r <= -threshold
    # 11 for pos, 10 for neg, 00 otherwise
    codes = torch.zeros(n, dtype=torch.int64, device=g.device)
    codes[pos] = 3
    codes[neg] = 2
    r[pos] -= threshold
    r[neg] += threshold
    nw = quantized_words(n)
    padded = torch.zeros(nw * 16, dtype=torch.int64, device=g.device)
    padded[:n] = codes
    slots = padded.reshape(nw, 16)
    shifts = torch.arange(16, device=g.device, dtype=torch.int64) * 2
    words = (slots << shifts).sum(dim=1) & 0xFFFFFFFF
    # wrap to int32 range (bit pattern preserved)
    words = torch.where(words >= (1 << 31), words - (1 << 32), words)
    return words.to(torch.int32)


def dequantize_2bit(packed: torch.Tensor, n: int, threshold: float,
                    out: Optional[torch.Tensor] = None) -> torch.Tensor:
    words = packed.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    shifts = torch.arange(16, device=packed.device, dtype=torch.int64) * 2
    codes = (words.unsqueeze(1) >> shifts) & 3
    vals = torch.zeros_like(codes, dtype=torch.float32)
    vals[codes == 3] = threshold
    vals[codes == 2] = -threshold
    flat = vals.reshape(-1)[:n]
    if out is not None:
        out.reshape(-1).copy_(flat)
        return out
    return flat


# ---------------------------------------------------------------------------
# Bi-Sparse (BSC): momentum-corrected top-k sparsification
# ---------------------------------------------------------------------------

def bsc_capacity(n: int, ratio: float, multiplier: int = 1) -> int:
    return int(float(n) * ratio) * multiplier


def bsc_sample_size(n: int, ratio: float) -> int:
    """Sample ~0.5% of elements (scaled check vs 10 as in the reference)."""
    if n * 0.005 * ratio >= 10:
        return int(n * 0.005)
    return int(10 / ratio)


def bsc_sample_indices(n: int, sample_size: int, seed: int = 42
                       ) -> torch.Tensor:
    """Seeded random sample positions, shared by the CPU and GPU
    boundary estimators so a mixed CPU/GPU party computes the SAME
    threshold (the paths used to diverge — strided vs
    randperm — which a structured gradient could bias). Sampling is
    with replacement (torch.randint): for threshold ESTIMATION over
    0.5% of elements it is statistically indistinguishable from the
    reference's srand/rand draw (gradient_compression.cc:213-231,
    itself with replacement)."""
    gen = torch.Generator(device="cpu").manual_seed(seed)
    return torch.randint(0, n, (sample_size,), generator=gen)


def bsc_boundary(v: torch.Tensor, ratio: float, seed: int = 42) -> float:
    """Sampled top-k threshold estimation over |v|."""
    n = v.numel()
    sample_size = min(bsc_sample_size(n, ratio), n)
    top_k = max(1, int(sample_size * ratio))
    idx = bsc_sample_indices(n, sample_size, seed).to(v.device)
    sample = v.reshape(-1)[idx].abs()
    return torch.topk(sample, min(top_k, sample.numel())).values[-1].item()


def bsc_compress(grad: torch.Tensor, u: torch.Tensor, v: torch.Tensor,
                 ratio: float, momentum: float = BSC_MOMENTUM,
                 seed: int = 42,
                 boundary: Optional[float] = None
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Momentum-corrected sampled-top-k select+pack.

    Mutates u (momentum) and v (error accumulation); zeroes u,v at the
    positions actually sent. Returns (values fp32[k], indices int32[k])
    with k = bsc_capacity(n, ratio); unused slots hold
    (BSC_PLACEHOLDER, -1).
    """
    g = grad.reshape(-1).float()
    n = g.numel()
    k = bsc_capacity(n, ratio)
    u.mul_(momentum).add_(g)
    v.add_(u)
    if boundary is None:
        boundary = bsc_boundary(v, ratio, seed)
    mask = v.abs() >= boundary
    sel = mask.nonzero(as_tuple=False).reshape(-1)[:k]   # index order, capacity-bounded
    vals = torch.full((k,), BSC_PLACEHOLDER, dtype=torch.float32, device=g.device)
    idx = torch.full((k,), -1, dtype=torch.int32, device=g.device)
    m = sel.numel()
    vals[:m] = v[sel]
    idx[:m] = sel.to(torch.int32)
    v[sel] = 0.0
    u[sel] = 0.0
    return vals, idx


def bsc_pull_compress(x: torch.Tensor, capacity: int
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Pack nonzeros of the aggregated tensor (the pull-side half)."""
    f = x.reshape(-1).float()
    sel = (f != 0).nonzero(as_tuple=False).reshape(-1)[:capacity]
    vals = torch.full((capacity,), BSC_PLACEHOLDER, dtype=torch.float32,
                      device=f.device)
    idx = torch.full((capacity,), -1, dtype=torch.int32, device=f.device)
    m = sel.numel()
    vals[:m] = f[sel]
    idx[:m] = sel.to(torch.int32)
    return vals, idx


def bsc_decompress(vals: torch.Tensor, idx: torch.Tensor, n: int,
                   out: Optional[torch.Tensor] = None,
                   accumulate: bool = False) -> torch.Tensor:
    """Scatter values by index; idx<0 entries are placeholders."""
    if out is None:
        out = torch.zeros(n, dtype=torch.float32, device=vals.device)
        accumulate_into = out
    else:
        accumulate_into = out.reshape(-1)
        if not accumulate:
            accumulate_into.zero_()
    valid = idx >= 0
    ii = idx[valid].long()
    vv = vals[valid]
    if accumulate:
        accumulate_into.index_add_(0, ii, vv)
    else:
        accumulate_into[ii] = vv
    return out


# ---------------------------------------------------------------------------
# DGT: per-chunk contribution + 4-bit linear quantization with residual
# ---------------------------------------------------------------------------

def dgt_contribution(grad: torch.Tensor, chunk_elems: int) -> torch.Tensor:
    """mean(|grad|) per chunk (kv_app.h:853-876)."""
    g = grad.reshape(-1).abs().float()
    n = g.numel()
    nchunks = (n + chunk_elems - 1) // chunk_elems
    padded = torch.zeros(nchunks * chunk_elems, device=g.device)
    padded[:n] = g
    counts = torch.full((nchunks,), chunk_elems, device=g.device,
                        dtype=torch.float32)
    counts[-1] = n - (nchunks - 1) * chunk_elems
    return padded.reshape(nchunks, chunk_elems).sum(dim=1) / counts


def quantize_4bit(x: torch.Tensor, residual: Optional[torch.Tensor] = None
                  ) -> Tuple[torch.Tensor, float, float]:
    """Linear min/max 4-bit quantization with residual feedback.

    Returns (packed uint8[ceil(n/2)], min, max). Codes 0..15 map to the
    16 midpoints of equal bins on [min, max] (van.cc:750-824 semantics).
    If residual is given, x+residual is quantized and residual is
    updated to the quantization error.
    """
    f = x.reshape(-1).float()
    if residual is not None:
        f = f + residual.reshape(-1)
    lo = f.min().item()
    hi = f.max().item()
    span = max(hi - lo, 1e-30)
    step = span / 16.0
    codes = ((f - lo) / step).floor().clamp_(0, 15).to(torch.uint8)
    deq = lo + (codes.float() + 0.5) * step
    if residual is not None:
        residual.reshape(-1).copy_(f - deq)
    n = f.numel()
    if n % 2:
        codes = torch.cat([codes, torch.zeros(1, dtype=torch.uint8,
                                              device=codes.device)])
    packed = (codes[0::2] | (codes[1::2] << 4))
    return packed, lo, hi


def dequantize_4bit(packed: torch.Tensor, n: int, lo: float, hi: float
                    ) -> torch.Tensor:
    step = max(hi - lo, 1e-30) / 16.0
    low = (packed & 0x0F).float()
    high = ((packed >> 4) & 0x0F).float()
    codes = torch.stack([low, high], dim=1).reshape(-1)[:n]
    return lo + (codes + 0.5) * step


# ---------------------------------------------------------------------------
# FP16 transmission helpers (fp32 master copy semantics)
# ---------------------------------------------------------------------------

def cast_fp16(x: torch.Tensor) -> torch.Tensor:
    return x.to(torch.float16)


def cast_fp32(x: torch.Tensor) -> torch.Tensor:
    return x.to(torch.float32)


# ---------------------------------------------------------------------------
# Fused optimizer update references (server-side ApplyUpdates)
# ---------------------------------------------------------------------------

def sgd_update(w: torch.Tensor, g: torch.Tensor, lr: float, wd: float = 0.0,
               rescale: float = 1.0):
    g = g * rescale + wd * w
    w.add_(g, alpha=-lr)


def sgd_mom_update(w: torch.Tensor, g: torch.Tensor, mom: torch.Tensor,
                   lr: float, momentum: float = 0.9, wd: float = 0.0,
                   rescale: float = 1.0):
    g = g * rescale + wd * w
    mom.mul_(momentum).add_(g, alpha=-lr)
    w.add_(mom)


def adam_update(w: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
                v: torch.Tensor, t: int, lr: float, beta1: float = 0.9,
                beta2: float = 0.999, eps: float = 1e-8, wd: float = 0.0,
                rescale: float = 1.0):
    """Adam with bias correction (matches mxnet adam_update semantics:
    lr_t = lr * sqrt(1-b2^t)/(1-b1^t))."""
    g = g * rescale + wd * w
    m.mul_(beta1).add_(g, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    lr_t = lr * math.sqrt(1 - beta2 ** t) / (1 - beta1 ** t)
    w.addcdiv_(m, v.sqrt().add_(eps), value=-lr_t)


def rmsprop_update(w: torch.Tensor, g: torch.Tensor, n: torch.Tensor,
                   lr: float, rho: float = 0.9, eps: float = 1e-8,
                   wd: float = 0.0, rescale: float = 1.0):
    """RMSProp (optimizer_op.cc rmsprop_update semantics)."""
    g = g * rescale + wd * w
    n.mul_(rho).addcmul_(g, g, value=1 - rho)
    w.addcdiv_(g, n.sqrt().add_(eps), value=-lr)


def adagrad_update(w: torch.Tensor, g: torch.Tensor, h: torch.Tensor,
                   lr: float, eps: float = 1e-7, wd: float = 0.0,
                   rescale: float = 1.0):
    g = g * rescale + wd * w
    h.addcmul_(g, g, value=1.0)
    w.addcdiv_(g, h.sqrt().add_(eps), value=-lr)


def signsgd_update(w: torch.Tensor, g: torch.Tensor, lr: float,
                   wd: float = 0.0, rescale: float = 1.0):
    """SignSGD (optimizer_op.cc signsgd_update)."""
    g = g * rescale + wd * w
    w.add_(torch.sign(g), alpha=-lr)


def signum_update(w: torch.Tensor, g: torch.Tensor, mom: torch.Tensor,
                  lr: float, momentum: float = 0.9, wd: float = 0.0,
                  rescale: float = 1.0):
    """Signum: momentum then sign (optimizer_op.cc signum_update)."""
    g = g * rescale + wd * w
    mom.mul_(momentum).add_(g, alpha=1 - momentum)
    w.add_(torch.sign(mom), alpha=-lr)


def dcasgd_update(w: torch.Tensor, g: torch.Tensor, prev_w: torch.Tensor,
                  mom: Optional[torch.Tensor], lr: float, lamda: float = 0.04,
                  momentum: float = 0.0, wd: float = 0.0, rescale: float = 1.0):
    """Delay-compensated ASGD (optimizer.py:872-926):
    w += -lr*(g + wd*w + lamda*g*g*(w - prev_w)); prev_w = w."""
    g = g * rescale
    upd = -lr * (g + wd * w + lamda * g * g * (w - prev_w))
    if mom is not None and momentum != 0.0:
        mom.mul_(momentum).add_(upd)
        upd = mom
    prev_w.copy_(w)
    w.add_(upd)
