#!/usr/bin/env python
"""Run one conv shape in a loop for PMC profiling."""
import sys
import torch
import os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from heterofl_amd.ops import require_native

ext = require_native()
shape = sys.argv[1] if len(sys.argv) > 1 else 'L1'
iters = int(sys.argv[2]) if len(sys.argv) > 2 else 200
S = {'L1': (5, 10, 64, 32, 64, 3, 1, 1),
     'L4': (5, 10, 512, 4, 512, 3, 1, 1)}[shape]
G, N, Cin, H, Cout, k, s, p = S
x = torch.randn(N, G * Cin, H, H, device='cuda', dtype=torch.bfloat16)
w = torch.randn(G * Cout, Cin, k, k, device='cuda') * 0.1
for _ in range(iters):
    ext.conv_fwd(x, w, torch.Tensor(), torch.Tensor(), G, s, p, 0)
torch.cuda.synchronize()
print('done')
