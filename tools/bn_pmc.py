"""Minimal BN fwd/bwd dispatches for PMC collection (BN single-pass
lever: does the apply phase's re-read hit L2?).  Run under:
    rocprofv3 --pmc FETCH_SIZE -f csv -d out -- python tools/bn_pmc.py
"""
import os, sys, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from adaptdl_amd.torch.layers import FusedBatchNormAct2d

dev = torch.device("cuda")
for (n, c, hw) in [(1024, 64, 32), (1024, 128, 16), (1024, 256, 8)]:
    bn = FusedBatchNormAct2d(c, relu=True).to(dev)
    x = (torch.randn(n, c, hw, hw, device=dev).to(torch.bfloat16)
         .contiguous(memory_format=torch.channels_last).requires_grad_(True))
    for _ in range(3):
        y = bn(x)
        y.backward(torch.ones_like(y))
        x.grad = None
    torch.cuda.synchronize()
    print("done", n, c, hw,
          "bytes_in =", n * c * hw * hw * 2)
