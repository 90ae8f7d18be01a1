"""Analytic model-cost profiler: per-module parameter counts, forward FLOPs
and checkpoint space, via forward hooks (reference: src/summary.py:44-47
output contract {num_params, num_flops, space}; FLOP formulas
src/summary.py:200-276 — re-derived here, not copied).

Used by the summary CLI to produce the Ratio/Params/FLOPs/Space columns of
the HeteroFL poster table (combined across levels by process.make_stats,
reference: src/process.py:345-374).
"""
import numpy as np
import torch
import torch.nn as nn


def _module_flops(m, inp, out):
    """Multiply-accumulate-based forward FLOPs of one module call."""
    if isinstance(m, nn.Conv2d):
        # out elems x (2 * Cin/groups * kh * kw) (+1 per out elem for bias)
        oe = out.numel()
        kh, kw = m.kernel_size
        f = oe * 2 * (m.in_channels // m.groups) * kh * kw
        if m.bias is not None:
            f += oe
        return f
    if isinstance(m, nn.Linear):
        f = out.numel() * 2 * m.in_features
        if m.bias is not None:
            f += out.numel()
        return f
    if isinstance(m, (nn.BatchNorm2d, nn.GroupNorm, nn.InstanceNorm2d,
                      nn.LayerNorm)):
        # mean/var pass + normalize+affine ~ 4 ops per element
        return 4 * inp.numel()
    if isinstance(m, (nn.ReLU, nn.GELU)):
        return inp.numel()
    if isinstance(m, nn.MaxPool2d):
        k = m.kernel_size if isinstance(m.kernel_size, int) else m.kernel_size[0]
        return out.numel() * k * k
    if isinstance(m, (nn.AdaptiveAvgPool2d, nn.AvgPool2d)):
        return inp.numel()
    if isinstance(m, nn.Embedding):
        return 0
    return 0


def summarize(model, input):
    """Run one forward with hooks; returns
    {'num_params', 'num_flops', 'space' (MB), 'per_module'}."""
    records = []
    handles = []

    def hook(m, i, o):
        x = i[0] if isinstance(i, tuple) and len(i) else None
        y = o[0] if isinstance(o, tuple) else o
        if not isinstance(y, torch.Tensor) or x is None or \
                not isinstance(x, torch.Tensor):
            return
        p = sum(q.numel() for q in m.parameters(recurse=False))
        records.append({'name': type(m).__name__, 'params': p,
                        'flops': _module_flops(m, x, y)})

    for m in model.modules():
        if len(list(m.children())) == 0:
            handles.append(m.register_forward_hook(hook))
    model.train(False)
    with torch.no_grad():
        model(input)
    for h in handles:
        h.remove()
    num_params = sum(p.numel() for p in model.parameters())
    num_flops = sum(r['flops'] for r in records)
    space = num_params * 4 / (1024 ** 2)  # fp32 checkpoint MB
    return {'num_params': num_params, 'num_flops': num_flops, 'space': space,
            'per_module': records}


def summarize_level(cfg, level_rate, batch_size=1):
    """Cost of one width level's local model on a synthetic input of the
    dataset's shape (reference: src/summary.py main loop)."""
    from .models import make_model
    model = make_model(cfg, model_rate=level_rate)
    if cfg['model_name'] == 'transformer':
        input = {'label': torch.randint(0, cfg['num_tokens'],
                                        (batch_size, cfg['bptt']))}
    else:
        input = {'img': torch.randn(batch_size, *cfg['data_shape']),
                 'label': torch.zeros(batch_size, dtype=torch.long)}
    return summarize(model, input)
