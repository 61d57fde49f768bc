#!/usr/bin/env python3
"""A/B: conv2 dgrad via MIOpen (ATen) vs the custom k_conv5_nhwc<32,1>."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from geomx_amd.ops import conv as C
from geomx_amd import _geops

DEV = "cuda:0"
torch.manual_seed(3)
N, CI, CO, H = 512, 16, 32, 110
Ho = H - 4
x = torch.randn(N, CI, H, H, device=DEV, dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
w = torch.randn(CO, CI, 5, 5, device=DEV, dtype=torch.bfloat16) * 0.05
go = torch.randn(N, CO, Ho, Ho, device=DEV, dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)

def aten():
    gx, _, _ = torch.ops.aten.convolution_backward(
        go, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
        [True, False, False])
    return gx

didx = C.build_dgrad_index(w.shape).to(DEV)
COp = (CI + 15) & ~15
wb = w.detach().reshape(-1)
wz = torch.cat([wb, wb.new_zeros(1)])
w_frags = wz[didx].contiguous()

def custom():
    gop = F.pad(go, (4, 4, 4, 4)).contiguous(
        memory_format=torch.channels_last)
    gx = torch.empty(N, COp, H, H, dtype=torch.bfloat16, device=DEV,
                     memory_format=torch.channels_last)
    _geops.conv5_nhwc(gop, w_frags, torch.Tensor(), gx, N, Ho + 8, Ho + 8,
                      H, H, CO, COp, 0)
    return gx

# correctness
ga = aten().float()
gc = custom().float()
err = (ga - gc).abs().max().item()
print(f"maxerr {err:.4f} scale {ga.abs().max().item():.2f}")

def bench(f, iters=20):
    for _ in range(5): f()
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(iters): f()
    torch.cuda.synchronize()
    return (time.perf_counter() - t) / iters * 1e3

print(f"aten dgrad   {bench(aten):8.3f} ms")
print(f"custom dgrad {bench(custom):8.3f} ms  (incl. F.pad)")
