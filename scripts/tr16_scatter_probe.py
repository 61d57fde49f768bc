#!/usr/bin/env python3
"""Does ds_read_b64_tr_b16 gather per-lane 8B quarters?
Model: out[g][j] = mem[addr[4*j + (g>>2)] + (g&3) elements]."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from geomx_amd import _geops

DEV = "cuda:0"
torch.manual_seed(5)
content = torch.arange(512, dtype=torch.float32).to(torch.bfloat16)
# scattered (8B-aligned = multiples of 4 elements), unique-ish offsets
g = torch.Generator().manual_seed(9)
addr = (torch.randperm(128, generator=g)[:16] * 4).to(torch.int32)
# replicate per 16-lane group
addr64 = addr.repeat(4).contiguous()
out = torch.empty(256, dtype=torch.bfloat16, device=DEV)
_geops.tr16_probe2(content.to(DEV), addr64.to(DEV), out)
got = out.cpu().float().reshape(64, 4)
c = content.float()
ok = True
for g_ in range(16):
    for j in range(4):
        exp = c[int(addr[4 * j + (g_ >> 2)]) + (g_ & 3)]
        if got[g_, j] != exp:
            ok = False
            if j == 0 and g_ < 4:
                print("mismatch", g_, j, float(got[g_, j]), float(exp))
print("PER_LANE_GATHER:", "CONFIRMED" if ok else "REFUTED")
if not ok:
    # alternative: contiguous at lane-0 base?
    base = int(addr[0])
    ok2 = all(got[g_, j] == c[base + j * 16 + g_] for g_ in range(16)
              for j in range(4))
    print("contiguous-at-lane0-base:", ok2)
