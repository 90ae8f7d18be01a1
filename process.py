#!/usr/bin/env python
"""Result aggregator (reference: src/process.py): crawl
./output/result/*.pt, nest by control fields, mean/std across seeds, export
a spreadsheet, and combine per-level summary outputs into weighted
Params/FLOPs/Space/Ratio stats (reference: src/process.py:345-374)."""
import argparse
import os
from collections import defaultdict

import numpy as np

from heterofl_amd.utils import load, save


def crawl_results(result_dir='./output/result'):
    """-> {control_key: {seed: logger_history}}"""
    out = defaultdict(dict)
    if not os.path.isdir(result_dir):
        return out
    for fn in sorted(os.listdir(result_dir)):
        if not fn.endswith('.pt'):
            continue
        parts = fn[:-3].split('_')
        if len(parts) < 5:
            continue  # summary files handled by make_stats
        seed = parts[0]
        key = '_'.join(parts[1:])
        try:
            out[key][seed] = load(os.path.join(result_dir, fn))
        except Exception as e:
            print('skip {}: {}'.format(fn, e))
    return out


def final_metrics(result):
    """Pull the last-epoch mean of every test metric from a result file."""
    vals = {}
    logger = result.get('logger')
    test_logger = logger.get('test') if isinstance(logger, dict) else logger
    mean = getattr(test_logger, 'mean', None)
    if mean:
        for k, v in mean.items():
            if k.startswith('test/'):
                vals[k[5:]] = v
    return vals


def aggregate(results):
    """mean/std across seeds per control key."""
    table = {}
    for key, by_seed in results.items():
        metrics = defaultdict(list)
        for seed, res in by_seed.items():
            for m, v in final_metrics(res).items():
                metrics[m].append(v)
        table[key] = {m: (float(np.mean(v)), float(np.std(v)))
                      for m, v in metrics.items()}
    return table


def make_stats(data_name, model_name, model_mode, model_split_rate,
               result_dir='./output/result'):
    """Combine per-level summary outputs into proportion-weighted
    Params/FLOPs/Space and the Ratio column (reference:
    src/process.py:345-374): each mode term '<level><weight>' contributes
    weight/total of its level's cost; Ratio = weighted params / full params.
    """
    levels = [t[0] for t in model_mode.split('-')]
    weights = [int(t[1:]) for t in model_mode.split('-')]
    total = sum(weights)
    per_level = {}
    for lv in set(levels) | {'a'}:
        path = os.path.join(result_dir, '{}_{}_{}.pt'.format(
            data_name, model_name, lv))
        per_level[lv] = load(path)
    params = sum(w / total * per_level[l]['num_params']
                 for l, w in zip(levels, weights))
    flops = sum(w / total * per_level[l]['num_flops']
                for l, w in zip(levels, weights))
    space = sum(w / total * per_level[l]['space']
                for l, w in zip(levels, weights))
    ratio = params / per_level['a']['num_params']
    return {'Ratio': ratio, 'Params': params, 'FLOPs': flops, 'Space': space}


def make_learning_curves(results, out_dir, metric='Global-Accuracy',
                         save_format='png'):
    """Per-control learning curves: metric vs global round, one line per
    seed (reference: src/process.py:233-342, 'vis' learning-curve half).
    Uses the per-round history snapshots pickled inside each result's
    Logger.  Returns the list of written figure paths."""
    try:
        import matplotlib
        matplotlib.use('Agg')
        import matplotlib.pyplot as plt
    except ImportError:
        print('matplotlib not available; skipping learning curves')
        return []
    os.makedirs(out_dir, exist_ok=True)
    written = []
    for key, by_seed in results.items():
        fig, ax = plt.subplots(figsize=(6, 4))
        any_line = False
        for seed, res in sorted(by_seed.items()):
            logger = res.get('logger')
            test_logger = logger.get('test') if isinstance(logger, dict) \
                else logger
            hist = getattr(test_logger, 'history', None)
            if not hist:
                continue
            series = hist.get('test/{}'.format(metric))
            if not series:
                continue
            ax.plot(range(1, len(series) + 1), series, label='seed {}'.format(seed))
            any_line = True
        if not any_line:
            plt.close(fig)
            continue
        ax.set_xlabel('global round')
        ax.set_ylabel(metric)
        ax.set_title(key, fontsize=8)
        ax.legend(fontsize=7)
        path = os.path.join(out_dir, 'curve_{}.{}'.format(key, save_format))
        fig.savefig(path, bbox_inches='tight')
        plt.close(fig)
        written.append(path)
    return written


def make_interpolation_plot(table, out_dir, metric='Global-Accuracy',
                            save_format='png'):
    """Interpolation plot (reference: src/process.py 'interp' half, driven
    by src/make.py:62-66's xN-y(10-N) sweeps): for each pair of levels
    (x, y), plot the final metric against the fraction of full-width
    clients N/10.  Returns the written figure paths."""
    try:
        import matplotlib
        matplotlib.use('Agg')
        import matplotlib.pyplot as plt
    except ImportError:
        print('matplotlib not available; skipping interpolation plot')
        return []
    import re
    os.makedirs(out_dir, exist_ok=True)
    # group control keys by the (dataset, model, level-pair) they interpolate
    groups = defaultdict(list)
    pat = re.compile(r'^([a-e])(\d+)-([a-e])(\d+)$')
    for key, metrics in table.items():
        parts = key.split('_')
        mode = next((p for p in parts if pat.match(p)), None)
        if mode is None or metric not in metrics:
            continue
        m = pat.match(mode)
        lx, wx, ly, wy = m.group(1), int(m.group(2)), m.group(3), int(m.group(4))
        frac = wx / (wx + wy)
        gkey = key.replace(mode, '{}-{}'.format(lx, ly))
        groups[gkey].append((frac, metrics[metric][0], metrics[metric][1]))
    written = []
    for gkey, pts in groups.items():
        if len(pts) < 2:
            continue
        pts.sort()
        xs, ys, es = zip(*pts)
        fig, ax = plt.subplots(figsize=(6, 4))
        ax.errorbar(xs, ys, yerr=es, marker='o')
        ax.set_xlabel('fraction of wider-level clients')
        ax.set_ylabel(metric)
        ax.set_title(gkey, fontsize=8)
        path = os.path.join(out_dir, 'interp_{}.{}'.format(gkey, save_format))
        fig.savefig(path, bbox_inches='tight')
        plt.close(fig)
        written.append(path)
    return written


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--result_dir', default='./output/result')
    p.add_argument('--out', default='./output/processed')
    p.add_argument('--save_format', default='png')
    p.add_argument('--metric', default='Global-Accuracy')
    args = p.parse_args()
    results = crawl_results(args.result_dir)
    table = aggregate(results)
    os.makedirs(args.out, exist_ok=True)
    save(table, os.path.join(args.out, 'aggregate.pt'))
    make_learning_curves(results, os.path.join(args.out, 'vis'),
                         metric=args.metric, save_format=args.save_format)
    make_interpolation_plot(table, os.path.join(args.out, 'vis'),
                            metric=args.metric, save_format=args.save_format)
    try:
        import pandas as pd
        rows = []
        for key, ms in table.items():
            row = {'control': key}
            for m, (mu, sd) in ms.items():
                row[m] = mu
                row[m + '_std'] = sd
            rows.append(row)
        df = pd.DataFrame(rows)
        df.to_csv(os.path.join(args.out, 'aggregate.csv'), index=False)
        print(df.to_string())
    except ImportError:
        for key, ms in table.items():
            print(key, ms)


if __name__ == '__main__':
    main()
