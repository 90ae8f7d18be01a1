#!/usr/bin/env python
"""Federated masked-LM training: same CLI as the reference
train_transformer_fed.py (pivot = Global-Perplexity, minimized).
(reference: src/train_transformer_fed.py)
"""
from heterofl_amd.entry import parse_args, run_fed_experiment


def main():
    cfg = parse_args()
    metric_name = {'train': {'Local': ['Local-Loss', 'Local-Perplexity']},
                   'test': {'Global': ['Global-Loss', 'Global-Perplexity']}}
    run_fed_experiment(cfg, pivot_metric='Global-Perplexity', pivot_sign=-1,
                       metric_name=metric_name)


if __name__ == '__main__':
    main()
