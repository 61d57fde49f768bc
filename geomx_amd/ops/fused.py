"""Fused NN-op modules backed by the gfx950 kernels.

FusedReLUPool2: relu -> maxpool(2,2) as ONE kernel pair (NHWC bf16).
Replaces ATen's separate relu elementwise + max_pool_{fwd,bwd}_nhwc
(together ~40% of the CNN benchmark step on MI355X). Uses the identity
relu(maxpool(x)) == maxpool(relu(x)) for max pooling, and records a
per-channel argmax quadrant (255 = relu-clamped) for an index-exact
backward with no atomics.

Falls back to the ATen ops when the shape/layout/device doesn't fit
(CPU, odd H/W, C % 8 != 0, non-bf16).
"""

from __future__ import annotations

import torch

from . import native_available


def _nhwc(t: torch.Tensor) -> torch.Tensor:
    return t.contiguous(memory_format=torch.channels_last)


class _ReLUPool2Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        from geomx_amd import _geops
        N, C, H, W = x.shape
        x = _nhwc(x)
        out = torch.empty((N, C, H // 2, W // 2), dtype=x.dtype,
                          device=x.device,
                          memory_format=torch.channels_last)
        idx = torch.empty(N * (H // 2) * (W // 2) * C, dtype=torch.uint8,
                          device=x.device)
        _geops.relu_maxpool2_fwd(x, out, idx, N, C, H, W)
        ctx.save_for_backward(idx)
        ctx.shape = (N, C, H, W)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        from geomx_amd import _geops
        (idx,) = ctx.saved_tensors
        N, C, H, W = ctx.shape
        grad_out = _nhwc(grad_out.to(torch.bfloat16))
        grad_in = torch.empty((N, C, H, W), dtype=torch.bfloat16,
                              device=grad_out.device,
                              memory_format=torch.channels_last)
        _geops.relu_maxpool2_bwd(grad_out, idx, grad_in, N, C, H, W)
        return grad_in


class FusedReLUPool2(torch.nn.Module):
    """Drop-in for nn.Sequential(nn.ReLU(), nn.MaxPool2d(2, 2))."""

    def _eligible(self, x: torch.Tensor) -> bool:
        return (x.is_cuda and native_available()
                and x.dtype == torch.bfloat16
                and x.dim() == 4 and x.shape[1] % 8 == 0
                and x.shape[2] % 2 == 0 and x.shape[3] % 2 == 0)

    def forward(self, x):
        if self._eligible(x):
            return _ReLUPool2Fn.apply(x)
        return torch.nn.functional.max_pool2d(
            torch.nn.functional.relu(x), 2, 2)
