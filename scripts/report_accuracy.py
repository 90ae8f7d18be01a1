"""Summarize a fed-training checkpoint's accuracy history:
    python scripts/report_accuracy.py output/model/<tag>_checkpoint.pt
Prints the per-eval-round Global-Accuracy series, the best value and the
final value (the BASELINE quality metric is IID global accuracy)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main(path):
    ck = torch.load(path, map_location='cpu', weights_only=False)
    logger = ck.get('logger')
    hist = getattr(logger, 'history', {}) or {}
    for key in ('test/Global-Accuracy', 'test/Global-Perplexity'):
        series = hist.get(key)
        if not series:
            continue
        print(f'{key}: {len(series)} eval points')
        for i, v in enumerate(series):
            print(f'  eval {i + 1}: {v:.2f}')
        best = max(series) if 'Accuracy' in key else min(series)
        print(f'  best: {best:.2f}   final: {series[-1]:.2f}')
    print('epoch (next to run):', ck.get('epoch'))
    cfg = ck.get('cfg', {})
    print('tag:', cfg.get('model_tag'))


if __name__ == '__main__':
    main(sys.argv[1])
