"""GeoConv5 / GeoConv5Pool: 5x5 stride-1 convolution stages on the
hand-written gfx950 MFMA kernels (csrc/conv.hip, convwrw2.hip).

r02 state — every op of both flagship conv stages is custom:
  * forward: fused conv+bias+ReLU+maxpool kernels (k_conv5_pool*):
    block-shared LDS row staging, pooling in-register, only the pooled
    output + argmax-quadrant mask reach HBM.
  * data gradient: LDS-staged direct conv (k_conv5_lds_nhwc) with
    in-kernel virtual padding and an XOR bank swizzle (conv2 class);
    interior direct conv (k_conv5_nhwc) for conv1-class shapes.
  * weight+bias gradients: ds_read_b64_tr_b16 fragment kernels
    (convwrw2.hip) with the bias fused as an all-ones tap tile.
Anything outside the supported geometry falls back to ATen/F.conv2d
(same math, library kernels).
"""

from __future__ import annotations



import torch
import torch.nn.functional as F

from . import native_available

_SUPPORTED_FWD = {(4, 16), (16, 32), (16, 16), (32, 32)}
_SUPPORTED_DGRAD = {(32, 16), (16, 16)}

# shapes where the custom kernel MEASURES faster than MIOpen on MI355X
# (conv1 3->16 @224px: 0.73 vs 1.63 ms). conv2-class shapes currently tie
# or trail MIOpen's igemm (0.63 vs 0.54 fwd) and stay on the ATen path;
# flip them on here as the kernel improves.
DEFAULT_ENABLED = {(4, 16)}


def _frag_index(CO: int, CI: int, numel: int, entry) -> torch.Tensor:
    """Build the gather index mapping weight.flatten() -> w_frags layout
    [CO/16][nK][64][8]; index `numel` means 'zero' (the caller appends a
    zero element to the flat weight). `entry(o, k)` returns the flat
    weight index for output channel o, im2col position k, or None."""
    S = 5 * CI
    Sp = (S + 7) & ~7
    K = 5 * Sp
    nK = (K + 31) // 32
    cot = CO // 16
    idx = torch.full((cot, nK, 64, 8), numel, dtype=torch.int64)
    for ct in range(cot):
        for km in range(nK):
            for lane in range(64):
                q, n = lane >> 4, lane & 15
                o = ct * 16 + n
                for j in range(8):
                    k = km * 32 + q * 8 + j
                    if k >= K:
                        continue
                    kh, jj = divmod(k, Sp)
                    if jj >= S or kh >= 5:
                        continue
                    kw, ci = divmod(jj, CI)
                    e = entry(o, kh, kw, ci)
                    if e is not None:
                        idx[ct, km, lane, j] = e
    return idx.reshape(-1)


def build_fwd_index(weight_shape) -> torch.Tensor:
    """weight [CO][CIr][5][5] (torch OIHW), CI zero-padded to mult of 4."""
    CO, CIr, KH, KW = weight_shape
    CI = (CIr + 3) & ~3
    numel = CO * CIr * KH * KW

    def entry(o, kh, kw, ci):
        if ci >= CIr:
            return None
        return ((o * CIr + ci) * KH + kh) * KW + kw

    return _frag_index(CO, CI, numel, entry)


def build_dgrad_index(weight_shape) -> torch.Tensor:
    """Data-grad pass: roles swap (CI'=CO, CO'=CIr padded to 16) and the
    kernel is spatially flipped: W'[o'=ci][kh][kw][ci'=o] =
    W[o][ci][4-kh][4-kw]."""
    CO, CIr, KH, KW = weight_shape
    COp = (CIr + 15) & ~15
    numel = CO * CIr * KH * KW

    def entry(op, kh, kw, cip):
        # op: output channel of the dgrad conv = original input channel
        if op >= CIr:
            return None
        o, ci = cip, op
        return ((o * CIr + ci) * KH + (4 - kh)) * KW + (4 - kw)

    return _frag_index(COp, CO, numel, entry)


_WRW_SUPPORTED = {(4, 16), (16, 32), (16, 16)}
_WRW_NWG = 768  # partial slabs (3 workgroups per CU)
# The custom wrw kernel is numerically verified (tests) but currently
# measures SLOWER than MIOpen's igemm_wrw (conv2: 2.07 vs 0.67 ms;
# conv1: 2.41 vs 1.64 ms) — LDS gather-bound. Off by default until the
# A-fragment gather is restructured.
# opt-in via env for A/B runs (GEOPS_WRW=1); module default stays off
# because v1 loses net in-step (see docs/kernels.md wrw v2 notes)
import os as _os
# r02: default ON — conv1's wrw runs the CI=4 kh-interleaved v2 kernel
# (MIOpen's igemm took 1.61 ms/step for it); GEOPS_WRW=0 reverts to ATen
WRW_ENABLED = _os.environ.get("GEOPS_WRW", "1") == "1"


def build_wrw_unpack_index(weight_shape) -> torch.Tensor:
    """Gather index from the wrw kernel slab [T16][CO] back to OIHW.

    CI=16 (v2): tap = (kh*5+kw)*16 + ci.
    CI=4  (v2-4, kh-interleaved): slot = (kw*2 + kh//4)*16 + (kh%4)*4+ci
    (38% of the 160 slots are dead kh>4 padding, dropped here).
    CI=8.. other: v1 layout tap = (kh*5+kw)*CI + ci.
    """
    CO, CIr, KH, KW = weight_shape
    CI = (CIr + 3) & ~3
    idx = torch.empty(CO, CIr, KH, KW, dtype=torch.int64)
    if CI == 4 and CO == 16:
        T16 = 176  # 10 tap tiles + the fused-bias tile
        for o in range(CO):
            for ci in range(CIr):
                for kh in range(KH):
                    for kw in range(KW):
                        slot = (kw * 2 + kh // 4) * 16 + (kh % 4) * 4 + ci
                        idx[o, ci, kh, kw] = slot * CO + o
        return idx.reshape(-1), T16
    NT = (25 * CI + 15) // 16
    T16 = NT * 16 + (16 if CI == 16 else 0)  # v2: +fused-bias tile
    for o in range(CO):
        for ci in range(CIr):
            for kh in range(KH):
                for kw in range(KW):
                    t = (kh * 5 + kw) * CI + ci
                    idx[o, ci, kh, kw] = t * CO + o
    return idx.reshape(-1), T16


def wrw_via_kernel(xb_padded: torch.Tensor, go: torch.Tensor,
                   unpack_idx: torch.Tensor, T16: int, weight_shape,
                   want_bias: bool = False):
    """Run the custom wrw kernel; returns dW in OIHW fp32 (and, with
    want_bias, the bias gradient the kernel fused via its all-ones tap
    tile — slab row T16-16)."""
    from geomx_amd import _geops
    N, CI, Hi, Wi = xb_padded.shape
    CO = go.shape[1]
    Ho, Wo = go.shape[2], go.shape[3]
    n_blocks = N * ((Ho + 3) // 4)
    n_wg = min(_WRW_NWG, n_blocks)
    part = torch.zeros(n_wg, T16, CO, dtype=torch.float32,
                       device=go.device)
    _geops.conv5_wrw_nhwc(xb_padded, go, part, N, Hi, Wi, Ho, Wo, CI, CO,
                          n_wg)
    full = part.sum(dim=0)
    dw = full.reshape(-1)[unpack_idx].reshape(weight_shape)
    if want_bias:
        return dw, full[T16 - 16]
    return dw


class _Conv5Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, fwd_idx, dgrad_idx, wrw_idx, wrw_t16):
        from geomx_amd import _geops
        N, CIr, Hi, Wi = x.shape
        CO = weight.shape[0]
        CI = (CIr + 3) & ~3
        Ho, Wo = Hi - 4, Wi - 4
        if CI != CIr:  # zero-pad channels (conv1: 3 -> 4), fused kernel
            xc = x.contiguous(memory_format=torch.channels_last)
            xb = torch.empty(N, CI, Hi, Wi, dtype=torch.bfloat16,
                             device=x.device,
                             memory_format=torch.channels_last)
            _geops.pad_ch3to4_nhwc(xc, xb, N * Hi * Wi)
            xs = xb[:, :CIr]  # saved for the weight-grad pass
        else:
            xb = x.to(torch.bfloat16) \
                .contiguous(memory_format=torch.channels_last)
            xs = xb
        wb = weight.detach().to(torch.bfloat16).reshape(-1)
        wz = torch.cat([wb, wb.new_zeros(1)])
        w_frags = wz[fwd_idx].contiguous()
        out = torch.empty(N, CO, Ho, Wo, dtype=torch.bfloat16,
                          device=x.device,
                          memory_format=torch.channels_last)
        b = bias.detach().float() if bias is not None else torch.Tensor()
        _geops.conv5_nhwc(xb, w_frags, b, out, N, Hi, Wi, Ho, Wo, CI, CO, 0)
        use_wrw = (WRW_ENABLED and wrw_idx is not None and
                   (CI, CO) in _WRW_SUPPORTED)
        # save the PADDED input when the custom wrw runs (it wants CI%4==0);
        # the unpadded view is recovered as xb[:, :CIr]
        ctx.save_for_backward(xb if use_wrw else xs, weight, dgrad_idx)
        ctx.wrw_pack = (wrw_idx, wrw_t16) if use_wrw else None
        ctx.has_bias = bias is not None
        ctx.dims = (N, CIr, CI, CO, Hi, Wi, Ho, Wo)
        # outside autocast (fp32 graphs) the downstream layers expect
        # the input dtype back; under autocast bf16 IS the compute dtype
        if x.dtype == torch.bfloat16 or torch.is_autocast_enabled():
            return out
        return out.to(x.dtype)

    @staticmethod
    def backward(ctx, grad_out):
        from geomx_amd import _geops
        x, weight, dgrad_idx = ctx.saved_tensors
        N, CIr, CI, CO, Hi, Wi, Ho, Wo = ctx.dims
        go = grad_out.to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        grad_x = None
        if ctx.needs_input_grad[0]:
            COp = (CIr + 15) & ~15
            wb = weight.detach().to(torch.bfloat16).reshape(-1)
            wz = torch.cat([wb, wb.new_zeros(1)])
            w_frags = wz[dgrad_idx].contiguous()
            # physical zero-pad of grad_out: lets the dgrad run on the
            # interior-only kernel (the in-kernel masked path cost 256
            # VGPRs -> 1 wave/SIMD)
            gop = F.pad(go, (4, 4, 4, 4)) \
                .contiguous(memory_format=torch.channels_last)
            gx = torch.empty(N, COp, Hi, Wi, dtype=torch.bfloat16,
                             device=x.device,
                             memory_format=torch.channels_last)
            _geops.conv5_nhwc(gop, w_frags, torch.Tensor(), gx, N, Ho + 8,
                              Wo + 8, Hi, Wi, CO, COp, 0)
            grad_x = gx[:, :CIr] if COp != CIr else gx
        grad_w = grad_b = None
        if ctx.needs_input_grad[1] or (ctx.has_bias and
                                       ctx.needs_input_grad[2]):
            if ctx.wrw_pack is not None:
                xb_pad = x  # saved padded (see forward)
                unpack_idx, T16 = ctx.wrw_pack
                grad_w, gb = wrw_via_kernel(xb_pad, go, unpack_idx, T16,
                                            weight.shape, want_bias=True)
                grad_w = grad_w.to(weight.dtype)
                if ctx.has_bias:
                    grad_b = gb.to(weight.dtype)  # fused in the kernel
            else:
                gi, gw, gb = torch.ops.aten.convolution_backward(
                    go, x, weight.to(torch.bfloat16),
                    [CO] if ctx.has_bias else None,
                    [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
                    [False, True, ctx.has_bias])
                grad_w = gw.to(weight.dtype)
                grad_b = gb.to(weight.dtype) if ctx.has_bias else None
        return grad_x, grad_w, grad_b, None, None, None, None


_wrw_idx_cache = {}


def _wrw_cached_index(weight_shape, device):
    key = (tuple(weight_shape), str(device))
    ent = _wrw_idx_cache.get(key)
    if ent is None:
        idx, t16 = build_wrw_unpack_index(weight_shape)
        ent = (idx.to(device), t16)
        _wrw_idx_cache[key] = ent
    return ent


class _AtenSplitConvFn(torch.autograd.Function):
    """ATen forward + SPLIT backward: dgrad via ATen, weight grad via
    the custom gfx950 wrw kernel where supported (1.10 vs the 1.62 ms
    igemm MIOpen picks in the training context), bias via a channel
    sum. Falls back to a separate ATen wrw call otherwise."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        xb = x.to(torch.bfloat16)
        wb = weight.detach().to(torch.bfloat16)
        bb = bias.detach().to(torch.bfloat16) if bias is not None else None
        out = torch.nn.functional.conv2d(xb, wb, bb)
        ctx.save_for_backward(xb, wb)
        ctx.has_bias = bias is not None
        if x.dtype == torch.bfloat16 or torch.is_autocast_enabled():
            return out
        return out.to(x.dtype)

    @staticmethod
    def backward(ctx, grad_out):
        x, w = ctx.saved_tensors
        go = grad_out.to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        CO = w.shape[0]
        grad_x = grad_w = grad_b = None
        if ctx.needs_input_grad[0]:
            gx, _, _ = torch.ops.aten.convolution_backward(
                go, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
                [True, False, False])
            grad_x = gx
        if ctx.needs_input_grad[1]:
            CIr = w.shape[1]
            if native_available() and (CIr, CO) in _WRW_SUPPORTED \
                    and CIr % 4 == 0:
                idx, t16 = _wrw_cached_index(w.shape, go.device)
                grad_w, gb = wrw_via_kernel(
                    x.contiguous(memory_format=torch.channels_last), go,
                    idx, t16, w.shape, want_bias=True)
                grad_b = gb if ctx.has_bias else None
            else:
                _, gw, gb = torch.ops.aten.convolution_backward(
                    go, x, w, [CO] if ctx.has_bias else None, [1, 1],
                    [0, 0], [1, 1], False, [0, 0], 1,
                    [False, True, ctx.has_bias])
                grad_w = gw.float()
                grad_b = gb.float() if ctx.has_bias else None
        return grad_x, grad_w, grad_b


class GeoConv5(torch.nn.Conv2d):
    """nn.Conv2d drop-in (kernel 5, stride 1, pad 0) running on the
    gfx950 MFMA direct-conv kernel when eligible."""

    def __init__(self, in_channels, out_channels, enabled_shapes=None, **kw):
        super().__init__(in_channels, out_channels, kernel_size=5, **kw)
        self._fwd_idx = None
        self._dgrad_idx = None
        self.enabled_shapes = (DEFAULT_ENABLED if enabled_shapes is None
                               else enabled_shapes)

    def _eligible(self, x) -> bool:
        CI = (self.in_channels + 3) & ~3
        dgrad_ok = (not x.requires_grad) or \
            ((self.out_channels, (self.in_channels + 15) & ~15)
             in _SUPPORTED_DGRAD)
        return (x.is_cuda and native_available()
                and (CI, self.out_channels) in _SUPPORTED_FWD
                and (CI, self.out_channels) in self.enabled_shapes
                and dgrad_ok
                and self.stride == (1, 1) and self.padding == (0, 0)
                and self.kernel_size == (5, 5) and self.groups == 1)

    # split-backward ATen path on non-custom shapes: dgrad by ATen,
    # weight grad by the custom wrw v2 kernel (0.208 ms vs MIOpen's
    # 1.61 ms igemm on conv2, r02 steady profile). DEFAULT ON since
    # wrw v2; GEOPS_SPLIT_BWD=0 reverts to the fused ATen backward
    SPLIT_BACKWARD = __import__("os").environ.get(
        "GEOPS_SPLIT_BWD", "1") == "1"

    def forward(self, x):
        if not self._eligible(x):
            if self.SPLIT_BACKWARD and x.is_cuda:
                return _AtenSplitConvFn.apply(x, self.weight, self.bias)
            return super().forward(x)
        if self._fwd_idx is None or self._fwd_idx.device != x.device:
            self._fwd_idx = build_fwd_index(self.weight.shape).to(x.device)
            self._dgrad_idx = build_dgrad_index(self.weight.shape).to(x.device)
            wi, t16 = build_wrw_unpack_index(self.weight.shape)
            self._wrw_idx = wi.to(x.device)
            self._wrw_t16 = t16
        return _Conv5Fn.apply(x, self.weight, self.bias, self._fwd_idx,
                              self._dgrad_idx, self._wrw_idx, self._wrw_t16)


class _Conv5PoolFn(torch.autograd.Function):
    """Fused conv1 stage: conv5 + bias + ReLU + 2x2 maxpool in ONE
    kernel (k_conv5_pool_nhwc) — the full-resolution conv output
    (1.55 GB at the bench shape) never exists in memory. Backward:
    the recorded argmax-quadrant mask drives the existing
    relu_maxpool2_bwd scatter, then the custom dgrad conv and the
    wrw kernel (fused bias tile) as in _Conv5Fn."""

    @staticmethod
    def forward(ctx, x, weight, bias, fwd_idx, dgrad_idx, wrw_idx, wrw_t16):
        from geomx_amd import _geops
        N, CIr, Hi, Wi = x.shape
        CO = weight.shape[0]
        CI = (CIr + 3) & ~3
        Ho, Wo = Hi - 4, Wi - 4
        if CI != CIr:
            xc = x.contiguous(memory_format=torch.channels_last)
            xb = torch.empty(N, CI, Hi, Wi, dtype=torch.bfloat16,
                             device=x.device,
                             memory_format=torch.channels_last)
            _geops.pad_ch3to4_nhwc(xc, xb, N * Hi * Wi)
        else:
            xb = x.to(torch.bfloat16) \
                .contiguous(memory_format=torch.channels_last)
        wb = weight.detach().to(torch.bfloat16).reshape(-1)
        wz = torch.cat([wb, wb.new_zeros(1)])
        w_frags = wz[fwd_idx].contiguous()
        Hop, Wop = Ho // 2, Wo // 2
        out = torch.empty(N, CO, Hop, Wop, dtype=torch.bfloat16,
                          device=x.device,
                          memory_format=torch.channels_last)
        mask = torch.empty(N * Hop * Wop * CO, dtype=torch.uint8,
                           device=x.device)
        b = bias.detach().float() if bias is not None else torch.Tensor()
        _geops.conv5_pool_nhwc(xb, w_frags, b, out, mask, N, Hi, Wi, Ho,
                               Wo, CI, CO)
        ctx.save_for_backward(xb, weight, dgrad_idx, mask)
        ctx.wrw_pack = (wrw_idx, wrw_t16)
        ctx.has_bias = bias is not None
        ctx.dims = (N, CIr, CI, CO, Hi, Wi, Ho, Wo)
        if x.dtype == torch.bfloat16 or torch.is_autocast_enabled():
            return out
        return out.to(x.dtype)

    @staticmethod
    def backward(ctx, grad_pooled):
        from geomx_amd import _geops
        import torch.nn.functional as F
        xb, weight, dgrad_idx, mask = ctx.saved_tensors
        N, CIr, CI, CO, Hi, Wi, Ho, Wo = ctx.dims
        gp = grad_pooled.to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        go = torch.empty(N, CO, Ho, Wo, dtype=torch.bfloat16,
                         device=gp.device,
                         memory_format=torch.channels_last)
        _geops.relu_maxpool2_bwd(gp, mask, go, N, CO, Ho, Wo)
        grad_x = None
        if ctx.needs_input_grad[0]:
            if CI == 4:
                COp = (CIr + 15) & ~15
                wb = weight.detach().to(torch.bfloat16).reshape(-1)
                wz = torch.cat([wb, wb.new_zeros(1)])
                w_frags = wz[dgrad_idx].contiguous()
                gop = F.pad(go, (4, 4, 4, 4)) \
                    .contiguous(memory_format=torch.channels_last)
                gx = torch.empty(N, COp, Hi, Wi, dtype=torch.bfloat16,
                                 device=xb.device,
                                 memory_format=torch.channels_last)
                _geops.conv5_nhwc(gop, w_frags, torch.Tensor(), gx, N,
                                  Ho + 8, Wo + 8, Hi, Wi, CO, COp, 0)
                grad_x = gx[:, :CIr] if COp != CIr else gx
            else:
                # conv2-class dgrad on the LDS-staged custom kernel with
                # IN-KERNEL virtual padding (pad=4): 0.56 ms vs MIOpen's
                # in-step igemm 0.64, and no F.pad round trip
                COp = (CIr + 15) & ~15
                wb = weight.detach().to(torch.bfloat16).reshape(-1)
                wz = torch.cat([wb, wb.new_zeros(1)])
                w_frags = wz[dgrad_idx].contiguous()
                gx = torch.empty(N, COp, Hi, Wi, dtype=torch.bfloat16,
                                 device=xb.device,
                                 memory_format=torch.channels_last)
                _geops.conv5_nhwc(go, w_frags, torch.Tensor(), gx, N,
                                  Ho + 8, Wo + 8, Hi, Wi, CO, COp, 4)
                grad_x = gx[:, :CIr] if COp != CIr else gx
        grad_w = grad_b = None
        if ctx.needs_input_grad[1] or (ctx.has_bias and
                                       ctx.needs_input_grad[2]):
            unpack_idx, T16 = ctx.wrw_pack
            grad_w, gb = wrw_via_kernel(xb, go, unpack_idx, T16,
                                        weight.shape, want_bias=True)
            grad_w = grad_w.to(weight.dtype)
            grad_b = gb.to(weight.dtype) if ctx.has_bias else None
        return grad_x, grad_w, grad_b, None, None, None, None


class GeoConv5Pool(GeoConv5):
    """Drop-in for GeoConv5 -> ReLU -> MaxPool2d(2,2): one fused kernel
    on the GPU path, eager fallback elsewhere. Parameters live on this
    module exactly as on GeoConv5 (state-dict compatible)."""

    _POOL_SHAPES = {(4, 16), (16, 32), (16, 16)}

    def _pool_eligible(self, x) -> bool:
        CI = (self.in_channels + 3) & ~3
        Ho, Wo = x.shape[2] - 4, x.shape[3] - 4
        return (x.is_cuda and native_available()
                and (CI, self.out_channels) in self._POOL_SHAPES
                and self.stride == (1, 1) and self.padding == (0, 0)
                and self.kernel_size == (5, 5) and self.groups == 1
                and Ho % 2 == 0 and Wo % 2 == 0 and x.shape[3] % 2 == 0)

    def forward(self, x):
        if not self._pool_eligible(x):
            y = super().forward(x)
            return torch.nn.functional.max_pool2d(
                torch.nn.functional.relu(y), 2, 2)
        if self._fwd_idx is None or self._fwd_idx.device != x.device:
            self._fwd_idx = build_fwd_index(self.weight.shape).to(x.device)
            self._dgrad_idx = build_dgrad_index(self.weight.shape).to(x.device)
            wi, t16 = build_wrw_unpack_index(self.weight.shape)
            self._wrw_idx = wi.to(x.device)
            self._wrw_t16 = t16
        return _Conv5PoolFn.apply(x, self.weight, self.bias, self._fwd_idx,
                                  self._dgrad_idx, self._wrw_idx,
                                  self._wrw_t16)
