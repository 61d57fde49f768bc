"""Op dispatch: hand-written gfx950 HIP kernels on GPU, torch reference on CPU.

Policy (matches the project contract):
  - CUDA (ROCm) tensors MUST go through the native `_geops` extension;
    if the extension is missing on a GPU machine the op RAISES — there
    is no silent eager fallback on the GPU path.
  - CPU tensors use the pure-torch reference implementations
    (`geomx_amd.ops.reference`), which also serve as the golden model
    for the HIP kernels in tests.
"""

from __future__ import annotations


from typing import Optional, Tuple

import torch

from . import reference as ref

_geops = None
_geops_err: Optional[str] = None
try:
    from geomx_amd import _geops as _geops_mod  # built in-tree by setup.py
    _geops = _geops_mod
except Exception as e:  # pragma: no cover - exercised only when ext missing
    _geops_err = repr(e)


def native_available() -> bool:
    return _geops is not None


def native_error() -> Optional[str]:
    return _geops_err


def _require_native():
    if _geops is None:
        raise RuntimeError(
            "geomx_amd native extension (_geops) is required for GPU tensors "
            "but failed to import: %s. Build it with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
            % _geops_err)
    return _geops


def _on_gpu(*tensors) -> bool:
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


# ---------------------------------------------------------------------------
# 2bit
# ---------------------------------------------------------------------------

def quantize_2bit(grad: torch.Tensor, residual: torch.Tensor, threshold: float,
                  out: Optional[torch.Tensor] = None) -> torch.Tensor:
    n = grad.numel()
    nw = ref.quantized_words(n)
    if _on_gpu(grad):
        g = _require_native()
        if out is None:
            out = torch.empty(nw, dtype=torch.int32, device=grad.device)
        g.quantize_2bit(grad.reshape(-1), residual.reshape(-1), out, threshold)
        return out
    packed = ref.quantize_2bit(grad, residual, threshold)
    if out is not None:
        out.copy_(packed)
        return out
    return packed


def dequantize_2bit(packed: torch.Tensor, n: int, threshold: float,
                    out: Optional[torch.Tensor] = None) -> torch.Tensor:
    if _on_gpu(packed):
        g = _require_native()
        if out is None:
            out = torch.empty(n, dtype=torch.float32, device=packed.device)
        g.dequantize_2bit(packed, out.reshape(-1), threshold)
        return out
    return ref.dequantize_2bit(packed, n, threshold, out=out)


# ---------------------------------------------------------------------------
# Bi-Sparse
# ---------------------------------------------------------------------------

def bsc_compress(grad, u, v, ratio, momentum=ref.BSC_MOMENTUM, seed=42
                 ) -> Tuple[torch.Tensor, torch.Tensor]:
    if _on_gpu(grad):
        g = _require_native()
        n = grad.numel()
        k = ref.bsc_capacity(n, ratio)
        vals = torch.empty(k, dtype=torch.float32, device=grad.device)
        idx = torch.empty(k, dtype=torch.int32, device=grad.device)
        gflat = grad.reshape(-1)
        # boundary PREDICTED from the seeded sample of post-momentum v
        # (v + mu*u + g at the sample positions) as a DEVICE scalar, so
        # the fused kernel path never synchronizes with the host
        bt = _bsc_boundary_pred_gpu(gflat, u, v, ratio, momentum, seed)
        if g.bsc_compress_fused(gflat, u, v, vals, idx, bt, momentum,
                                ref.BSC_PLACEHOLDER):
            return vals, idx
        # huge-n fallback: momentum + count + scan + pack chain
        g.bsc_momentum(gflat, u, v, momentum)
        boundary = _bsc_boundary_gpu(v, ratio, seed)
        g.bsc_pack(v, u, vals, idx, boundary, ref.BSC_PLACEHOLDER)
        return vals, idx
    return ref.bsc_compress(grad, u, v, ratio, momentum, seed)


_bsc_idx_cache: dict = {}


def _bsc_sample_idx(n: int, sample_size: int, seed: int, device):
    ck = (n, sample_size, seed, device)
    idx = _bsc_idx_cache.get(ck)
    if idx is None:
        if len(_bsc_idx_cache) > 64:
            _bsc_idx_cache.clear()
        idx = ref.bsc_sample_indices(n, sample_size, seed).to(device)
        _bsc_idx_cache[ck] = idx
    return idx


def _bsc_boundary_pred_gpu(g: torch.Tensor, u: torch.Tensor,
                           v: torch.Tensor, ratio: float, momentum: float,
                           seed: int) -> torch.Tensor:
    """Sampled top-k boundary over the POST-momentum |v|, computed from
    (g,u,v) BEFORE the fused kernel mutates them: same value the CPU
    golden model derives after its momentum pass (same seeded sample,
    same op order mu*u + g then + v). Returns a device fp32 scalar."""
    n = v.numel()
    sample_size = min(ref.bsc_sample_size(n, ratio), n)
    top_k = max(1, int(sample_size * ratio))
    si = _bsc_sample_idx(n, sample_size, seed, v.device)
    sample = (v[si] + (u[si] * momentum + g[si])).abs()
    k = min(top_k, sample.numel())
    # clone: a slice view is 4B-aligned at best; the kernel's
    # uniform scalar load wants its own 16B-aligned storage
    return torch.topk(sample, k).values[k - 1:k].clone()


def _bsc_boundary_gpu(v: torch.Tensor, ratio: float, seed: int) -> float:
    """Same seeded-sample estimator as the CPU golden model
    (ref.bsc_boundary): gather |v| at precomputed random positions,
    small topk. Indices are generated once per (n, sample_size, seed)
    on the CPU generator and cached on-device, so the per-call cost is
    one tiny gather + topk."""
    n = v.numel()
    sample_size = min(ref.bsc_sample_size(n, ratio), n)
    top_k = max(1, int(sample_size * ratio))
    ck = (n, sample_size, seed, v.device)
    idx = _bsc_idx_cache.get(ck)
    if idx is None:
        if len(_bsc_idx_cache) > 64:
            _bsc_idx_cache.clear()
        idx = ref.bsc_sample_indices(n, sample_size, seed).to(v.device)
        _bsc_idx_cache[ck] = idx
    sample = v[idx].abs()
    k = min(top_k, sample.numel())
    return torch.topk(sample, k).values[-1].item()


def bsc_pull_compress(x: torch.Tensor, capacity: int
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    if _on_gpu(x):
        g = _require_native()
        vals = torch.empty(capacity, dtype=torch.float32, device=x.device)
        idx = torch.empty(capacity, dtype=torch.int32, device=x.device)
        g.bsc_pull_pack(x.reshape(-1), vals, idx, ref.BSC_PLACEHOLDER)
        return vals, idx
    return ref.bsc_pull_compress(x, capacity)


def bsc_decompress(vals, idx, n, out=None, accumulate=False) -> torch.Tensor:
    if _on_gpu(vals):
        g = _require_native()
        if out is None:
            out = torch.zeros(n, dtype=torch.float32, device=vals.device)
            g.bsc_unpack(vals, idx, out.reshape(-1), True)
            return out
        g.bsc_unpack(vals, idx, out.reshape(-1), accumulate)
        return out
    return ref.bsc_decompress(vals, idx, n, out=out, accumulate=accumulate)


# ---------------------------------------------------------------------------
# DGT
# ---------------------------------------------------------------------------

def dgt_contribution(grad: torch.Tensor, chunk_elems: int) -> torch.Tensor:
    if _on_gpu(grad):
        g = _require_native()
        n = grad.numel()
        nchunks = (n + chunk_elems - 1) // chunk_elems
        out = torch.empty(nchunks, dtype=torch.float32, device=grad.device)
        g.dgt_contribution(grad.reshape(-1), out, chunk_elems)
        return out
    return ref.dgt_contribution(grad, chunk_elems)


def quantize_4bit_chunked(x: torch.Tensor, chunk_elems: int,
                          residual: Optional[torch.Tensor] = None):
    """Per-chunk min/max 4-bit quantization. Returns (packed u8, minmax f32[nchunks,2])."""
    n = x.numel()
    nchunks = (n + chunk_elems - 1) // chunk_elems
    if _on_gpu(x):
        g = _require_native()
        packed = torch.empty((n + 1) // 2, dtype=torch.uint8, device=x.device)
        minmax = torch.empty((nchunks, 2), dtype=torch.float32, device=x.device)
        res = residual.reshape(-1) if residual is not None else torch.Tensor()
        g.quantize_4bit(x.reshape(-1), res, packed, minmax, chunk_elems)
        return packed, minmax
    packs, mins, maxs = [], [], []
    f = x.reshape(-1)
    for c in range(nchunks):
        sl = f[c * chunk_elems:(c + 1) * chunk_elems]
        rsl = residual.reshape(-1)[c * chunk_elems:(c + 1) * chunk_elems] \
            if residual is not None else None
        p, lo, hi = ref.quantize_4bit(sl, rsl)
        packs.append(p)
        mins.append(lo)
        maxs.append(hi)
    minmax = torch.tensor([[lo, hi] for lo, hi in zip(mins, maxs)],
                          dtype=torch.float32)
    return torch.cat(packs), minmax


def dequantize_4bit_chunked(packed: torch.Tensor, minmax: torch.Tensor,
                            n: int, chunk_elems: int,
                            out: Optional[torch.Tensor] = None) -> torch.Tensor:
    nchunks = minmax.shape[0]
    if _on_gpu(packed):
        g = _require_native()
        if out is None:
            out = torch.empty(n, dtype=torch.float32, device=packed.device)
        g.dequantize_4bit(packed, minmax, out.reshape(-1), chunk_elems)
        return out
    res = []
    # CPU reference packs each chunk independently (chunk_elems even except last)
    half = (chunk_elems + 1) // 2
    for c in range(nchunks):
        lo, hi = minmax[c, 0].item(), minmax[c, 1].item()
        m = min(chunk_elems, n - c * chunk_elems)
        res.append(ref.dequantize_4bit(packed[c * half:c * half + (m + 1) // 2],
                                       m, lo, hi))
    full = torch.cat(res)
    if out is not None:
        out.reshape(-1).copy_(full)
        return out
    return full


# ---------------------------------------------------------------------------
# Fused optimizer updates
# ---------------------------------------------------------------------------

def sgd_update(w, g, lr, wd=0.0, rescale=1.0):
    if _on_gpu(w):
        _require_native().sgd_update(w.reshape(-1), g.reshape(-1), lr, wd, rescale)
        return
    ref.sgd_update(w, g, lr, wd, rescale)


def sgd_mom_update(w, g, mom, lr, momentum=0.9, wd=0.0, rescale=1.0):
    if _on_gpu(w):
        _require_native().sgd_mom_update(w.reshape(-1), g.reshape(-1),
                                         mom.reshape(-1), lr, momentum, wd, rescale)
        return
    ref.sgd_mom_update(w, g, mom, lr, momentum, wd, rescale)


def adam_update(w, g, m, v, t, lr, beta1=0.9, beta2=0.999, eps=1e-8, wd=0.0,
                rescale=1.0):
    if _on_gpu(w):
        _require_native().adam_update(w.reshape(-1), g.reshape(-1),
                                      m.reshape(-1), v.reshape(-1),
                                      t, lr, beta1, beta2, eps, wd, rescale)
        return
    ref.adam_update(w, g, m, v, t, lr, beta1, beta2, eps, wd, rescale)


def rmsprop_update(w, g, n, lr, rho=0.9, eps=1e-8, wd=0.0, rescale=1.0):
    if _on_gpu(w):
        _require_native().rmsprop_update(w.reshape(-1), g.reshape(-1),
                                         n.reshape(-1), lr, rho, eps, wd,
                                         rescale)
        return
    ref.rmsprop_update(w, g, n, lr, rho, eps, wd, rescale)


def adagrad_update(w, g, h, lr, eps=1e-7, wd=0.0, rescale=1.0):
    if _on_gpu(w):
        _require_native().adagrad_update(w.reshape(-1), g.reshape(-1),
                                         h.reshape(-1), lr, eps, wd, rescale)
        return
    ref.adagrad_update(w, g, h, lr, eps, wd, rescale)


def signsgd_update(w, g, lr, wd=0.0, rescale=1.0):
    if _on_gpu(w):
        _require_native().signsgd_update(w.reshape(-1), g.reshape(-1), lr,
                                         wd, rescale)
        return
    ref.signsgd_update(w, g, lr, wd, rescale)


def signum_update(w, g, mom, lr, momentum=0.9, wd=0.0, rescale=1.0):
    if _on_gpu(w):
        _require_native().signum_update(w.reshape(-1), g.reshape(-1),
                                        mom.reshape(-1), lr, momentum, wd,
                                        rescale)
        return
    ref.signum_update(w, g, mom, lr, momentum, wd, rescale)


def dcasgd_update(w, g, prev_w, mom, lr, lamda=0.04, momentum=0.0, wd=0.0,
                  rescale=1.0):
    if _on_gpu(w):
        _require_native().dcasgd_update(
            w.reshape(-1), g.reshape(-1), prev_w.reshape(-1),
            mom.reshape(-1) if mom is not None else torch.Tensor(),
            lr, lamda, momentum, wd, rescale)
        return
    ref.dcasgd_update(w, g, prev_w, mom, lr, lamda, momentum, wd, rescale)
