#!/usr/bin/env python
"""Long-run stability soak: many federated rounds of the bench config on a
learnable synthetic problem; prints loss trajectory and peak memory."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from heterofl_amd.config import default_config
from heterofl_amd.control import process_control, CONTROL_FIELDS
from heterofl_amd.data import fetch_dataset, split_dataset
from heterofl_amd.fed import FedRunner
from heterofl_amd.models import make_model
from heterofl_amd.utils import process_dataset, make_optimizer

rounds = int(sys.argv[1]) if len(sys.argv) > 1 else 30
cfg = default_config()
control = '1_100_0.1_iid_fix_a1-e1_bn_1_1'
cfg['control'] = dict(zip(CONTROL_FIELDS, control.split('_')))
cfg['data_name'] = 'CIFAR10'
cfg['model_name'] = 'resnet18'
cfg['device'] = 'cuda:0'
cfg['engine'] = 'batched'
cfg['compute_dtype'] = os.environ.get('HETEROFL_SOAK_DTYPE', 'bfloat16')
cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                      'test': {'Global': ['Global-Loss']}}
process_control(cfg)
torch.manual_seed(0)
ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=20000)
img = ds['train'].img.float()
ds['train'].target = (img.reshape(len(ds['train']), -1).mean(1) * 10 / 256
                      ).long().clamp(0, 9).tolist()
process_dataset(ds, cfg)
data_split, label_split = split_dataset(ds, 100, 'iid', 10)
model = make_model(cfg).to('cuda:0')
opt = make_optimizer(model, cfg['lr'], cfg)
runner = FedRunner(cfg, ds, data_split, label_split, model, opt)

probe = {'img': torch.stack([ds['train'][i]['img'] for i in range(512)]
                            ).to('cuda:0'),
         'label': torch.tensor(ds['train'].target[:512], device='cuda:0')}


def global_loss():
    model.load_state_dict(runner.federation.global_parameters)
    model.train(True)
    with torch.no_grad():
        return model(probe)['loss'].item()


t0 = time.perf_counter()
for ep in range(1, rounds + 1):
    runner.train_round(ep)
    if ep % 5 == 0 or ep == 1:
        torch.cuda.synchronize()
        print(f'round {ep:3d}  loss {global_loss():.4f}  '
              f'mem {torch.cuda.max_memory_allocated()/2**30:.2f} GiB  '
              f'{(time.perf_counter()-t0)/ep*1000:.0f} ms/round', flush=True)
print('soak done')
