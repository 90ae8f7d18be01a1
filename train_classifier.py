#!/usr/bin/env python
"""Centralized classifier baseline (reference: src/train_classifier.py)."""
from heterofl_amd.entry import parse_args, run_centralized_experiment


def main():
    cfg = parse_args()
    metric_name = {'train': ['Loss', 'Accuracy'], 'test': ['Loss', 'Accuracy']}
    run_centralized_experiment(cfg, 'Accuracy', +1, metric_name)


if __name__ == '__main__':
    main()
