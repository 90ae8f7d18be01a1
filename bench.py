#!/usr/bin/env python
"""Flagship benchmark: HeteroFL federated training throughput on MI355X.

Measures the BASELINE.json headline metric — local-train samples/sec/node on
CIFAR10 ResNet18, 100-client, active fraction 0.1, fix a1-e1 (half the users
at full width, half at 1/16 width), sBN + Scaler + masked-CE — on synthetic
data with random-init weights (no network for datasets).

One step = one communication round: distribute slices -> local training of
the 10 active clients (5 local epochs, batch 10) -> padded combine.  The
per-round sBN statistics pass and evaluation are excluded from the timed
region (they are evaluation machinery, not local training; the reference
publishes no throughput number at all — BASELINE.md).

Scaling is WEAK: with N GPUs the federation holds 100*N users (10*N active
clients per round, sharded one rank per GPU over RCCL); per-GPU work is
fixed.  Run under torchrun for N>1:
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
"""
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


def build_cfg(n_gpus, device, dtype):
    cfg = default_config()
    control = f'1_{100 * n_gpus}_0.1_iid_fix_a1-e1_bn_1_1'
    cfg['control'] = dict(zip(CONTROL_FIELDS, control.split('_')))
    cfg['control_name'] = control
    cfg['data_name'] = 'CIFAR10'
    cfg['model_name'] = 'resnet18'
    cfg['device'] = device
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = dtype
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss', 'Local-Accuracy']},
                          'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                                   'Global': ['Global-Loss', 'Global-Accuracy']}}
    process_control(cfg)
    return cfg


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--gpus', type=int, default=1)
    p.add_argument('--steps', type=int, default=5)
    p.add_argument('--warmup', type=int, default=3)
    p.add_argument('--dtype', type=str, default='bfloat16',
                   choices=['float32', 'bfloat16', 'fp8'])
    p.add_argument('--engine', type=str, default='batched',
                   choices=['batched', 'sequential'])
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
    dtype = args.dtype if on_gpu else 'float32'

    cfg = build_cfg(n, device, dtype)
    cfg['engine'] = args.engine
    cfg['world_size'] = world_size

    torch.manual_seed(0)
    # real CIFAR10 shape: 500 samples per user (50k/100); HETEROFL_BENCH_SPU
    # shrinks it for plumbing tests only
    spu = int(os.environ.get('HETEROFL_BENCH_SPU', '500'))
    ds = fetch_dataset('CIFAR10', synthetic=True,
                       synthetic_size=spu * cfg['num_users'])
    process_dataset(ds, cfg)
    # deterministic split across ranks
    g = torch.Generator().manual_seed(1234)
    st = torch.random.get_rng_state()
    torch.manual_seed(1234)
    data_split, label_split = split_dataset(ds, cfg['num_users'], 'iid',
                                            cfg['classes_size'])
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

    # secondary whole-round metric: train round + the full-train-set sBN
    # statistics pass the reference pays every round
    # (src/train_classifier_fed.py:127-138); runs on the native batched
    # engine (runner._native_stats).  Kept outside the headline timed
    # region, reported so the headline cannot be read as end-to-end round
    # wall clock.
    whole_steps = min(2, args.steps)
    sync()
    t1 = time.perf_counter()
    for ep in range(args.warmup + args.steps + 1,
                    args.warmup + args.steps + whole_steps + 1):
        runner.train_round(ep)
        runner.stats()
    sync()
    whole = time.perf_counter() - t1
    if dist_ctx is not None:
        t = torch.tensor([whole], device=device if on_gpu else 'cpu')
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        whole = t.item()
    round_ms_incl_stats = whole / whole_steps * 1000.0

    if os.environ.get('HETEROFL_TIMING') == '1' and rank == 0:
        from heterofl_amd.fed.runner import _phase_timer
        import sys
        print('[timing]', json.dumps(_phase_timer.report()), file=sys.stderr)

    active = int(torch.tensor(cfg['frac'] * cfg['num_users']).ceil())
    samples_per_user = len(ds['train']) // cfg['num_users']
    samples_per_round = active * cfg['num_epochs']['local'] * samples_per_user
    total = samples_per_round * args.steps
    value = total / elapsed
    if rank == 0:
        print(json.dumps({
            'metric': 'local-train samples/sec/node',
            'value': value,
            'unit': 'samples/s',
            'n_gpus': n,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': elapsed / args.steps * 1000.0,
            'round_ms_incl_stats': round_ms_incl_stats,
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': None,
            'dtype': dtype,
            'data': 'synthetic',
            'config': {
                'model': 'resnet18', 'dataset': 'CIFAR10',
                'control_name': cfg['control_name'],
                'num_users': cfg['num_users'], 'active_clients': active,
                'local_epochs': cfg['num_epochs']['local'],
                'global_batch': cfg['batch_size']['train'],
                'samples_per_user': samples_per_user,
                'parallelism': f'client-dp{n}',
                'engine': cfg['engine'],
                'timed_region': 'distribute+local_train+combine per round',
            },
        }))


if __name__ == '__main__':
    main()
