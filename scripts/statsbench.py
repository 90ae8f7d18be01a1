"""A/B the sBN statistics pass: native batched engine vs eager MIOpen
test-model, plus per-shape conv timings at the stats-pass geometry
(G=1, R=1, N=500).  Run on a GPU box."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from heterofl_amd.ops import require_native

ext = require_native()
dev = 'cuda:0'
dt = torch.float32   # the stats pass runs un-autocast fp32

SHAPES = [
    ('stem', 1, 500, 3, 32, 64, 3, 1, 1),
    ('L1', 1, 500, 64, 32, 64, 3, 1, 1),
    ('L2d', 1, 500, 64, 32, 128, 3, 2, 1),
    ('L2', 1, 500, 128, 16, 128, 3, 1, 1),
    ('L3d', 1, 500, 128, 16, 256, 3, 2, 1),
    ('L3', 1, 500, 256, 8, 256, 3, 1, 1),
    ('L4d', 1, 500, 256, 8, 512, 3, 2, 1),
    ('L4', 1, 500, 512, 4, 512, 3, 1, 1),
    ('sc2', 1, 500, 64, 32, 128, 1, 2, 0),
    ('sc4', 1, 500, 256, 8, 512, 1, 2, 0),
]


def timeit(fn, iters=20):
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


for dt in (torch.float32, torch.bfloat16):
    print(f'-- dtype {dt} --')
    print(f'{"shape":6} {"native_us":>10} {"miopen_us":>10} {"ratio":>6}')
    tot_n = tot_m = 0.0
    for name, G, N, Cin, H, Cout, k, s, p in SHAPES:
        x = torch.randn(N, G * Cin, H, H, device=dev, dtype=dt)
        w = torch.randn(G * Cout, Cin, k, k, device=dev) * 0.1
        tn = timeit(lambda: ext.conv_fwd(x, w, torch.Tensor(), torch.Tensor(),
                                         G, s, p, 0))
        tm = timeit(lambda: F.conv2d(x, w.to(dt), None, s, p, 1, G))
        tot_n += tn
        tot_m += tm
        print(f'{name:6} {tn:10.1f} {tm:10.1f} {tn / tm:6.2f}')
    print(f'TOTAL  {tot_n:10.1f} {tot_m:10.1f} {tot_n / tot_m:6.2f}')

# whole stats pass A/B
from heterofl_amd.config import default_config
from heterofl_amd.control import process_control, CONTROL_FIELDS
from heterofl_amd.data import fetch_dataset, split_dataset
from heterofl_amd.fed import FedRunner
from heterofl_amd.models import make_model
from heterofl_amd.utils import process_dataset, make_optimizer

cfg = default_config()
control = '1_100_0.1_iid_fix_a1-e1_bn_1_1'
cfg['control'] = dict(zip(CONTROL_FIELDS, control.split('_')))
cfg['control_name'] = control
cfg['data_name'] = 'CIFAR10'
cfg['model_name'] = 'resnet18'
cfg['device'] = dev
cfg['engine'] = 'batched'
cfg['compute_dtype'] = 'bfloat16'
cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                      'test': {'Global': ['Global-Loss']}}
process_control(cfg)
spu = int(os.environ.get('HETEROFL_BENCH_SPU', '500'))
ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=spu * 100)
process_dataset(ds, cfg)
data_split, label_split = split_dataset(ds, 100, 'iid', 10)
model = make_model(cfg).to(dev)
runner = FedRunner(cfg, ds, data_split, label_split, model,
                   make_optimizer(model, cfg['lr'], cfg))
runner.train_round(1)
torch.cuda.synchronize()

for mode, env in [('native', '1'), ('eager', '0')]:
    os.environ['HETEROFL_NATIVE_STATS'] = env
    runner.stats()  # warm caches
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(3):
        runner.stats()
    torch.cuda.synchronize()
    print(f'stats pass [{mode}]: {(time.perf_counter() - t0) / 3 * 1000:.0f} ms')
print('DONE')
