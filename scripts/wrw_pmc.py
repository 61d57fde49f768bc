import sys
sys.path.insert(0, "/root/repo")
import torch
from geomx_amd.ops import conv as C
x2 = torch.randn(512,16,110,110,device="cuda",dtype=torch.bfloat16).to(memory_format=torch.channels_last)
g2 = torch.randn(512,32,106,106,device="cuda",dtype=torch.bfloat16).to(memory_format=torch.channels_last)
idx2, t16 = C.build_wrw_unpack_index((32,16,5,5)); idx2=idx2.cuda()
for _ in range(6):
    C.wrw_via_kernel(x2, g2, idx2, t16, (32,16,5,5))
torch.cuda.synchronize()
