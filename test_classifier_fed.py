#!/usr/bin/env python
"""Federated classifier evaluation (reference: src/test_classifier_fed.py):
load {tag}_best.pt, re-run sBN stats, evaluate Local+Global metrics, save
./output/result/{tag}.pt."""
from heterofl_amd.entry import parse_args, run_fed_eval


def main():
    cfg = parse_args()
    metric_name = {'train': {'Local': ['Local-Loss', 'Local-Accuracy']},
                   'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                            'Global': ['Global-Loss', 'Global-Accuracy']}}
    run_fed_eval(cfg, metric_name)


if __name__ == '__main__':
    main()
