"""Metric registry (reference: src/metrics/metrics.py:7-44):
Loss/Accuracy/Perplexity each in plain, Local-, Global- flavors.
"""
import torch


def accuracy(output, target, topk=1):
    """Top-k accuracy in percent (reference: src/metrics/metrics.py:7-13)."""
    with torch.no_grad():
        if output.dim() == 3:
            # LM scores (N, vocab, S): flatten positions
            output = output.permute(0, 2, 1).reshape(-1, output.size(1))
            target = target.reshape(-1)
        batch_size = target.size(0)
        pred_k = output.topk(topk, 1, True, True)[1]
        correct_k = pred_k.eq(target.unsqueeze(1).expand_as(pred_k)).float().sum()
        return (correct_k * (100.0 / batch_size)).item()


def perplexity(output, target):
    """exp(CE) recomputed from the score (reference: src/metrics/metrics.py:16-25)."""
    with torch.no_grad():
        ce = torch.nn.functional.cross_entropy(output, target)
        return torch.exp(ce).item()


class Metric:
    def __init__(self):
        base = {
            'Loss': lambda inp, out: out['loss'].item(),
            'Accuracy': lambda inp, out: accuracy(out['score'], inp['label']),
            'Perplexity': lambda inp, out: perplexity(out['score'], inp['label']),
        }
        self.metric = {}
        for name, fn in base.items():
            self.metric[name] = fn
            self.metric['Local-' + name] = fn
            self.metric['Global-' + name] = fn

    def evaluate(self, metric_names, input, output):
        return {m: self.metric[m](input, output) for m in metric_names}
