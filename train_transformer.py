#!/usr/bin/env python
"""Centralized masked-LM baseline (reference: src/train_transformer.py)."""
from heterofl_amd.entry import parse_args, run_centralized_experiment


def main():
    cfg = parse_args()
    metric_name = {'train': ['Loss', 'Perplexity'],
                   'test': ['Loss', 'Perplexity']}
    run_centralized_experiment(cfg, 'Perplexity', -1, metric_name)


if __name__ == '__main__':
    main()
