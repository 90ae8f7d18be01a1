#!/usr/bin/env python
"""Federated classifier training (vision): same CLI as the reference
train_classifier_fed.py — every cfg key is a flag, plus --control_name, e.g.

  python train_classifier_fed.py --data_name CIFAR10 --model_name resnet18 \
      --control_name 1_100_0.1_iid_fix_a1-e1_bn_1_1

(reference: src/train_classifier_fed.py)
"""
from heterofl_amd.entry import parse_args, run_fed_experiment


def main():
    cfg = parse_args()
    metric_name = {'train': {'Local': ['Local-Loss', 'Local-Accuracy']},
                   'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                            'Global': ['Global-Loss', 'Global-Accuracy']}}
    run_fed_experiment(cfg, pivot_metric='Global-Accuracy', pivot_sign=+1,
                       metric_name=metric_name)


if __name__ == '__main__':
    main()
