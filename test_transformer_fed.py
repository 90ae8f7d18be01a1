#!/usr/bin/env python
"""Federated masked-LM evaluation (reference: src/test_transformer_fed.py)."""
from heterofl_amd.entry import parse_args, run_fed_eval


def main():
    cfg = parse_args()
    metric_name = {'train': {'Local': ['Local-Loss', 'Local-Perplexity']},
                   'test': {'Global': ['Global-Loss', 'Global-Perplexity']}}
    run_fed_eval(cfg, metric_name)


if __name__ == '__main__':
    main()
