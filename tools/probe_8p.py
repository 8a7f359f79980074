#!/usr/bin/env python3
"""A/B probe for the 8-phase vs 2-buffer GEMM under rocprofv3 --pmc.
Runs each kernel a few times on the CLIP qkv (fb384) and 8192^3 shapes.
Select with VFA_8P=1 / unset."""
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), '..'))
import torch
from video_features_amd import ops

assert torch.cuda.is_available()
dev = 'cuda:0'
torch.manual_seed(0)
for m, n, k in [(19200, 2304, 768), (8192, 8192, 8192)]:
    x = (torch.randn(m, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    w = (torch.randn(n, k, device=dev) / (k ** 0.25)).to(torch.bfloat16)
    b = torch.randn(n, device=dev).to(torch.bfloat16)
    for _ in range(6):
        ops.linear_act(x, w, b, 'none')
    torch.cuda.synchronize()
print('done')
