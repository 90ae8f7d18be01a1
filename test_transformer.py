#!/usr/bin/env python
"""Centralized masked-LM evaluation (reference: src/test_transformer.py):
load {tag}_best.pt, evaluate the test stream, save
./output/result/{tag}.pt."""
from heterofl_amd.entry import parse_args, run_centralized_eval


def main():
    cfg = parse_args()
    metric_name = {'train': ['Loss', 'Perplexity'],
                   'test': ['Loss', 'Perplexity']}
    run_centralized_eval(cfg, metric_name)


if __name__ == '__main__':
    main()
