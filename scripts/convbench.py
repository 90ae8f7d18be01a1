#!/usr/bin/env python
"""Per-shape A/B of the MFMA grouped conv vs MIOpen (F.conv2d): fwd,
bwd-data, bwd-weight timed separately with hip events.  Run on a GPU box:
    python scripts/convbench.py [iters]
"""
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, '.')
from heterofl_amd.ops import require_native  # noqa: E402

SHAPES = [
    # (name, G, N, Cin, H, Cout, k, stride, pad)
    ('stem', 5, 10, 3, 32, 64, 3, 1, 1),
    ('L1', 5, 10, 64, 32, 64, 3, 1, 1),
    ('L2d', 5, 10, 64, 32, 128, 3, 2, 1),
    ('L2', 5, 10, 128, 16, 128, 3, 1, 1),
    ('L3', 5, 10, 256, 8, 256, 3, 1, 1),
    ('L4d', 5, 10, 256, 8, 512, 3, 2, 1),
    ('L4', 5, 10, 512, 4, 512, 3, 1, 1),
    ('sc2', 5, 10, 64, 32, 128, 1, 2, 0),
    ('sc4', 5, 10, 256, 8, 512, 1, 2, 0),
]


def timeit(fn, iters):
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


def main():
    iters = int(sys.argv[1]) if len(sys.argv) > 1 else 50
    ext = require_native()
    dt = torch.bfloat16
    print(f'{"shape":6} {"dir":5} {"native_us":>10} {"miopen_us":>10} '
          f'{"ratio":>6}  gflop')
    tot_n = tot_m = 0.0
    for name, G, N, Cin, H, Cout, k, s, p in SHAPES:
        x = torch.randn(N, G * Cin, H, H, device='cuda', dtype=dt)
        w = torch.randn(G * Cout, Cin, k, k, device='cuda') * 0.1
        wb = w.to(dt)
        OH = (H + 2 * p - k) // s + 1
        dy = torch.randn(N, G * Cout, OH, OH, device='cuda', dtype=dt)
        gflop = 2.0 * N * G * Cout * Cin * k * k * OH * OH / 1e9
        rows = [
            ('fwd',
             lambda: ext.conv_fwd(x, w, torch.Tensor(), torch.Tensor(),
                                  G, s, p, 0),
             lambda: F.conv2d(x, wb, None, s, p, 1, G)),
            ('bwdD',
             lambda: ext.conv_bwd_data(dy, w, G, s, p, H, H, 0),
             lambda: torch.nn.grad.conv2d_input(
                 (N, G * Cin, H, H), wb, dy, (s, s), (p, p), (1, 1), G)),
            ('bwdW',
             lambda: ext.conv_bwd_weight(dy, x, G, s, p, k, 0),
             lambda: torch.nn.grad.conv2d_weight(
                 x, (G * Cout, Cin, k, k), dy, (s, s), (p, p), (1, 1), G)),
        ]
        for dname, fn_n, fn_m in rows:
            tn = timeit(fn_n, iters)
            tm = timeit(fn_m, iters)
            tot_n += tn
            tot_m += tm
            print(f'{name:6} {dname:5} {tn:10.1f} {tm:10.1f} '
                  f'{tn / tm:6.2f}  {gflop:.2f}')
    print(f'TOTAL native {tot_n:.0f}us  miopen {tot_m:.0f}us  '
          f'ratio {tot_n / tot_m:.2f}')


if __name__ == '__main__':
    main()
