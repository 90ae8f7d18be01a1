#!/usr/bin/env python
"""Transformer (masked-LM) federated-training benchmark — BASELINE config 4:
WikiText2 transformer, 100 clients, 0.1 active, fix a1-e1.  Measures
local-train tokens/sec/node on synthetic token streams (no network).  Same
torchrun contract as bench.py for N>1."""
import argparse
import json
import os
import time

import torch

from heterofl_amd.config import default_config
from heterofl_amd.control import process_control, CONTROL_FIELDS
from heterofl_amd.data import fetch_dataset, split_dataset
from heterofl_amd.fed import FedRunner
from heterofl_amd.models import make_model
from heterofl_amd.utils import process_dataset, make_optimizer


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=3)
    p.add_argument('--warmup', type=int, default=1)
    p.add_argument('--dtype', type=str, default='bfloat16')
    args = p.parse_args()

    world_size = int(os.environ.get('WORLD_SIZE', '1'))
    n = max(args.gpus, world_size)
    on_gpu = torch.cuda.is_available()
    dist_ctx = None
    if world_size > 1:
        from heterofl_amd.parallel import init_distributed
        dist_ctx = init_distributed()
        device = str(dist_ctx.device)
        rank = dist_ctx.rank
    else:
        device = 'cuda:0' if on_gpu else 'cpu'
        rank = 0

    cfg = default_config()
    control = f'1_{100 * n}_0.1_iid_fix_a1-e1_bn_1_1'
    cfg['control'] = dict(zip(CONTROL_FIELDS, control.split('_')))
    cfg['control_name'] = control
    cfg['data_name'] = 'WikiText2'
    cfg['model_name'] = 'transformer'
    cfg['device'] = device
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = args.dtype if on_gpu else 'float32'
    cfg['world_size'] = world_size
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                          'test': {'Global': ['Global-Loss',
                                              'Global-Perplexity']}}
    process_control(cfg)

    torch.manual_seed(0)
    # real WikiText2 scale: ~2M train tokens; batchified to 100*n rows
    toks = int(os.environ.get('HETEROFL_BENCH_TOKENS', str(2_000_000 * n)))
    ds = fetch_dataset('WikiText2', synthetic=True, synthetic_size=toks)
    cfg['batch_size'] = {'train': 100 * n, 'test': 10}
    process_dataset(ds, cfg)
    st = torch.random.get_rng_state()
    torch.manual_seed(1234)
    data_split, label_split = split_dataset(ds, cfg['num_users'], 'iid')
    torch.random.set_rng_state(st)

    model = make_model(cfg).to(device)
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt,
                       dist_ctx=dist_ctx)

    def sync():
        if dist_ctx is not None:
            torch.distributed.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for ep in range(1, args.warmup + 1):
        runner.train_round(ep)
    sync()
    t0 = time.perf_counter()
    for ep in range(args.warmup + 1, args.warmup + args.steps + 1):
        runner.train_round(ep)
    sync()
    elapsed = time.perf_counter() - t0
    if dist_ctx is not None:
        t = torch.tensor([elapsed], device=device if on_gpu else 'cpu')
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    active = int(torch.tensor(cfg['frac'] * cfg['num_users']).ceil())
    rows_per_user = cfg['batch_size']['train'] // cfg['num_users']
    L = ds['train'].token.size(1)
    tokens_per_round = active * cfg['num_epochs']['local'] * rows_per_user * L
    value = tokens_per_round * args.steps / elapsed
    if rank == 0:
        print(json.dumps({
            'metric': 'local-train tokens/sec/node', 'value': value,
            'unit': 'tokens/s', 'n_gpus': n, 'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': elapsed / args.steps * 1000.0,
            'higher_is_better': True, 'scaling': 'weak',
            'vs_baseline': None, 'dtype': cfg['compute_dtype'],
            'data': 'synthetic',
            'config': {'model': 'transformer', 'dataset': 'WikiText2',
                       'control_name': cfg['control_name'],
                       'bptt': cfg['bptt'], 'num_users': cfg['num_users'],
                       'active_clients': active,
                       'local_epochs': cfg['num_epochs']['local'],
                       'parallelism': f'client-dp{n}'},
        }))


if __name__ == '__main__':
    main()
