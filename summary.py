#!/usr/bin/env python
"""Model-cost summary CLI (reference: src/summary.py): writes
./output/result/{data}_{model}_{level}.pt with {num_params, num_flops, space}
for every width level of the current control."""
import torch

from heterofl_amd.entry import parse_args
from heterofl_amd.control import process_control
from heterofl_amd.data import fetch_dataset
from heterofl_amd.profiler import summarize_level
from heterofl_amd.utils import process_dataset, save


def main():
    cfg = parse_args()
    process_control(cfg)
    cfg['device'] = 'cpu'
    dataset = fetch_dataset(cfg['data_name'], cfg['subset'],
                            synthetic=cfg.get('synthetic', False))
    process_dataset(dataset, cfg)
    for level, rate in cfg['model_split_rate'].items():
        s = summarize_level(cfg, rate)
        out = {'num_params': s['num_params'], 'num_flops': s['num_flops'],
               'space': s['space']}
        path = './output/result/{}_{}_{}.pt'.format(cfg['data_name'],
                                                    cfg['model_name'], level)
        save(out, path)
        print(level, out)


if __name__ == '__main__':
    main()
