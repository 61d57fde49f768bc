import sys, os
sys.path.insert(0, "/root/repo")
import torch, time, sys
from geomx_amd.ops.conv import GeoConv5
which = sys.argv[1]
if which == "conv1":
    m = GeoConv5(3,16).cuda(); x = torch.randn(512,3,224,224,device="cuda",dtype=torch.bfloat16).to(memory_format=torch.channels_last)
else:
    m = GeoConv5(16,32).cuda(); x = torch.randn(512,16,110,110,device="cuda",dtype=torch.bfloat16).to(memory_format=torch.channels_last)
for _ in range(8):
    y = m(x)
torch.cuda.synchronize()
