"""Micro-bench of the one convbench row slower than MIOpen (L3 bwdW:
G=5 N=10 Cin=256 H=8 Cout=256 k=3 s=1 — VERDICT r1 item 3).
    python scripts/bwdw_l3.py          # time native vs MIOpen
Env: HETEROFL_CONV_SPLIT_TARGET sweeps the split-P heuristic."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from heterofl_amd.ops import require_native

ext = require_native()
dev = 'cuda'
dt = torch.bfloat16
G, N, Cin, H, Cout, k, s, p = 5, 10, 256, 8, 256, 3, 1, 1
torch.manual_seed(0)
x = torch.randn(N, G * Cin, H, H, device=dev, dtype=dt)
w = torch.randn(G * Cout, Cin, k, k, device=dev) * 0.1
OH = (H + 2 * p - k) // s + 1
dy = torch.randn(N, G * Cout, OH, OH, device=dev, dtype=dt)


def timeit(fn, iters=100):
    st = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    st.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return st.elapsed_time(e) / iters * 1000


tn = timeit(lambda: ext.conv_bwd_weight(dy, x, G, s, p, k, 0))
x32 = x.float().requires_grad_(True)
w32 = w.to(dt)


def miopen():
    return torch.ops.aten.convolution_backward(
        dy, x, w32, None, [s, s], [p, p], [1, 1], False, [0, 0], G,
        [False, True, False])[1]


tm = timeit(miopen)
tag = os.environ.get('HETEROFL_CONV_SPLIT_TARGET', 'default')
nostg = 'nostaged' if os.environ.get('HETEROFL_CONV_NO_STAGED') else 'staged'
print(f'L3 bwdW [{nostg}, split_target={tag}]: native {tn:.1f}us '
      f'miopen {tm:.1f}us ratio {tn / tm:.2f}')
err = (ext.conv_bwd_weight(dy, x, G, s, p, k, 0)
       - miopen().float()).abs().max().item()
print(f'  max err vs miopen: {err:.3e}')
