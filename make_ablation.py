#!/usr/bin/env python
"""Ablation sweep generator (reference: src/make_ablation.py): norm
({bn,none} vs {in,ln,gn}), Scaler on/off, mask on/off across the fixed and
dynamic model modes."""
import argparse


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--run', default='train')
    p.add_argument('--num_gpus', type=int, default=8)
    p.add_argument('--round', type=int, default=4)
    p.add_argument('--num_experiments', type=int, default=1)
    p.add_argument('--file', default='classifier')
    p.add_argument('--data', default='CIFAR10')
    p.add_argument('--model', default='resnet18')
    args = p.parse_args()
    script = '{}_{}_fed.py'.format(args.run, args.file)
    modes = ['a1', 'a1-e1', 'a1-b1-c1-d1-e1']
    ablations = []
    for mm in modes:
        for split in ['iid', 'non-iid-2']:
            for norm in ['bn', 'in', 'ln', 'gn', 'none']:
                for scale in ['1', '0']:
                    for mask in ['1', '0']:
                        smode = 'fix'
                        ablations.append('1_100_0.1_{}_{}_{}_{}_{}_{}'.format(
                            split, smode, mm, norm, scale, mask))
    lines = []
    k = 0
    for seed in range(args.num_experiments):
        for c in ablations:
            gpu = k % args.num_gpus
            lines.append(
                'HIP_VISIBLE_DEVICES={} python {} --data_name {} '
                '--model_name {} --init_seed {} --control_name {} &'.format(
                    gpu, script, args.data, args.model, seed, c))
            k += 1
            if k % (args.round * args.num_gpus) == 0:
                lines.append('wait')
    lines.append('wait')
    out = '{}_{}_{}_ablation.sh'.format(args.run, args.file, args.data)
    with open(out, 'w') as f:
        f.write('#!/bin/bash\n' + '\n'.join(lines) + '\n')
    print('wrote {} with {} runs'.format(out, k))


if __name__ == '__main__':
    main()
