#!/usr/bin/env python3
"""Error-structure dump for the wrw v2 kernel vs ATen."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from geomx_amd.ops import conv as C

DEV = "cuda:0"
torch.manual_seed(1)
for (ci, co, hw) in [(16, 32, 40), (16, 16, 40)]:
    N = 2
    x = torch.randn(N, ci, hw, hw, device=DEV, dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    go = torch.randn(N, co, hw - 4, hw - 4, device=DEV,
                     dtype=torch.bfloat16) \
        .to(memory_format=torch.channels_last)
    idx, t16 = C.build_wrw_unpack_index((co, ci, 5, 5))
    dw = C.wrw_via_kernel(x, go, idx.to(DEV), t16, (co, ci, 5, 5))
    w = torch.zeros(co, ci, 5, 5, device=DEV, dtype=torch.bfloat16)
    _, dwr, _ = torch.ops.aten.convolution_backward(
        go, x, w, None, [1, 1], [0, 0], [1, 1], False, [0, 0], 1,
        [False, True, False])
    dwr = dwr.float()
    err = (dw - dwr).abs()
    print(f"=== ci={ci} co={co} hw={hw}  scale={dwr.abs().max():.1f} "
          f"maxerr={err.max():.3f} relerr={(err.max()/dwr.abs().max()):.4f}")
    # error by (kh, kw)
    e_khkw = err.amax(dim=(0, 1))
    print("err by (kh,kw):")
    for kh in range(5):
        print("  " + " ".join(f"{e_khkw[kh, kw]:8.3f}" for kw in range(5)))
    # error by o (rows=o)
    e_o = err.amax(dim=(1, 2, 3))
    print("err by o tile0:", [f"{v:.2f}" for v in e_o[:16:4]],
          "tile1:", [f"{v:.2f}" for v in e_o[16::4]] if co > 16 else "-")
    e_ci = err.amax(dim=(0, 2, 3))
    print("err by ci:", [f"{v:.2f}" for v in e_ci[::4]])
    # ratio structure: is dw a permutation/shift of ref?
    c0 = torch.corrcoef(torch.stack([dw.reshape(-1), dwr.reshape(-1)]))[0, 1]
    print("corr(dw, ref) =", float(c0))

# ---- staging dump check ----------------------------------------------
from geomx_amd import _geops
ci, co, hw = 16, 32, 40
Wi = Hi = hw; Wo = Ho = hw - 4
x = torch.randn(1, ci, Hi, Wi, device=DEV, dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
go = torch.randn(1, co, Ho, Wo, device=DEV, dtype=torch.bfloat16) \
    .to(memory_format=torch.channels_last)
RPB = ((136 * 16 * 2 + 1023) // 1024) * 1024  # 5120 bytes
dump = torch.zeros((6 + 4) * RPB // 2, dtype=torch.bfloat16, device=DEV)
sz = _geops.wrw2_dump(x, go, dump, 1, Hi, Wi, Ho, Wo, co)
d = dump.cpu().float()
xl = x.permute(0, 2, 3, 1).cpu().float()   # [n][h][w][ci]
gl = go.permute(0, 2, 3, 1).cpu().float()  # [n][h][w][o]
RP = RPB // 2  # elements per region
ok_a = True
for ir in range(6):
    img = d[ir * RP:(ir + 1) * RP]
    for pix in range(0, Wi, 7):
        got = img[pix * 16:(pix + 1) * 16]
        exp = xl[0, ir, pix]
        if not torch.allclose(got, exp):
            print("A-image mismatch ir", ir, "pix", pix, got[:4], exp[:4])
            ok_a = False
            break
    pad = img[Wi * 16: (RPB // 32) * 16]
    if pad.abs().max() > 0:
        print("A pad nonzero at ir", ir, float(pad.abs().max()))
        ok_a = False
print("A image ok:", ok_a)
ok_b = True
for rr in range(2):
    for ot in range(2):
        img = d[(6 + rr * 2 + ot) * RP:(6 + rr * 2 + ot + 1) * RP]
        for pix in range(0, Wo, 5):
            got = img[pix * 16:(pix + 1) * 16]
            exp = gl[0, rr, pix, ot * 16:(ot + 1) * 16]
            if not torch.allclose(got, exp):
                print("B mismatch rr", rr, "ot", ot, "pix", pix,
                      got[:4], exp[:4])
                ok_b = False
                break
print("B image ok:", ok_b)
