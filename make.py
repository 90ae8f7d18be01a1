#!/usr/bin/env python
"""Experiment sweep generator (reference: src/make.py): emits bash scripts
that launch training runs round-robin over the node's GPUs via
HIP_VISIBLE_DEVICES (the ROCm equivalent of the reference's
CUDA_VISIBLE_DEVICES fan-out, src/make.py:88-101)."""
import argparse
import itertools

import numpy as np


def fixed_combinations():
    levels = ['a', 'b', 'c', 'd', 'e']
    modes = [l + '1' for l in levels]
    dynamic = []
    for k in range(2, 6):
        for combo in itertools.combinations(modes, k):
            dynamic.append('-'.join(combo))
    interp = []
    # all 10 level pairs x 9 ratios = 90 (reference: src/make.py:62-66)
    for n in range(1, 10):
        for hi, lo in itertools.combinations(levels, 2):
            interp.append('{}{}-{}{}'.format(hi, n, lo, 10 - n))
    return modes, dynamic, interp


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--run', default='train')
    p.add_argument('--num_gpus', type=int, default=8)
    p.add_argument('--world_size', type=int, default=1)
    p.add_argument('--round', type=int, default=4)
    p.add_argument('--experiment_step', type=int, default=1)
    p.add_argument('--num_experiments', type=int, default=1)
    p.add_argument('--resume_mode', type=int, default=0)
    p.add_argument('--file', default='classifier')
    p.add_argument('--data', default='CIFAR10')
    p.add_argument('--model', default='resnet18')
    args = p.parse_args()
    script = '{}_{}_fed.py'.format(args.run, args.file)
    modes, dynamic, interp = fixed_combinations()
    controls = []
    for mm in modes + dynamic + interp:
        for split in ['iid', 'non-iid-2']:
            smode = 'fix' if '-' not in mm or any(c.isdigit() and c != '1'
                                                  for c in mm) else 'dynamic'
            controls.append('1_100_0.1_{}_{}_{}_bn_1_1'.format(split, smode, mm))
    lines = []
    k = 0
    for seed in range(args.num_experiments):
        for c in controls:
            gpu = k % args.num_gpus
            lines.append(
                'HIP_VISIBLE_DEVICES={} python {} --data_name {} '
                '--model_name {} --init_seed {} --control_name {} &'.format(
                    gpu, script, args.data, args.model, seed, c))
            k += 1
            if k % (args.round * args.num_gpus) == 0:
                lines.append('wait')
    lines.append('wait')
    out = '{}_{}_{}.sh'.format(args.run, args.file, args.data)
    with open(out, 'w') as f:
        f.write('#!/bin/bash\n' + '\n'.join(lines) + '\n')
    print('wrote {} with {} runs'.format(out, k))


if __name__ == '__main__':
    main()
