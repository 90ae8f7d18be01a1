"""Metric logger: per-key weighted series with round snapshots and console
line assembly.

Plays the role of the reference's Logger (src/logger.py:8-87) and keeps its
*external* contract — `append/write/safe/reset/flush`, a `mean` mapping, a
`history` of per-round snapshots, and picklability into checkpoints
(src/train_classifier_fed.py:85-88) — but the internals are a different
design: each key accumulates a weighted sum and a total weight, and the mean
is derived on demand instead of being maintained by an incremental running
update.  TensorBoard is optional; without it scalars go to a JSONL event
stream so runs stay inspectable in no-network environments.
"""
import json
import os

from .utils.core import makedir_exist_ok

try:
    from torch.utils.tensorboard import SummaryWriter  # needs tensorboard pkg
    _HAS_TB = True
except Exception:
    SummaryWriter = None
    _HAS_TB = False


class _MeanView(dict):
    """Dict of derived means; missing keys read as 0 (callers probe pivot
    metrics that may not have been logged yet)."""

    def __missing__(self, key):
        return 0


def _is_scalar(x):
    return isinstance(x, (int, float)) and not isinstance(x, bool)


class Logger:
    def __init__(self, log_path):
        # log_path=None makes a writer-less logger (non-main ranks in
        # multi-GPU runs accumulate series for sync() but never write files)
        self.log_path = log_path
        self.writer = None
        self._wsum = {}    # key -> sum of weight*value, this rank, unsynced
        self._wtot = {}    # key -> sum of weights, this rank, unsynced
        self._ssum = {}    # key -> already rank-merged sum (see sync())
        self._stot = {}    # key -> already rank-merged weight
        self.tracker = {}  # key -> most recent raw appended value (any type)
        self.history = {}  # key -> [snapshot mean per round]
        self._step = {}    # key -> scalar event counter for the writer

    # -------------------------------------------------------------- derived
    @property
    def mean(self):
        """Weighted mean per key, derived from the accumulated series
        (local pending contributions plus any rank-merged ones)."""
        view = _MeanView()
        for k in set(self._wsum) | set(self._ssum):
            s = self._wsum.get(k, 0.0) + self._ssum.get(k, 0.0)
            w = self._wtot.get(k, 0) + self._stot.get(k, 0)
            if w:
                view[k] = s / w
        return view

    def sync(self):
        """Merge this logger's pending series across ranks (C3 metric
        reduction — the reference accumulates metrics in one process,
        src/logger.py:35-55; sharded evaluation needs the shards' weighted
        sums summed so every rank sees identical means).  Collective: all
        ranks must call it at the same point.  No-op when not distributed."""
        import torch.distributed as dist
        if not dist.is_initialized() or dist.get_world_size() <= 1:
            return
        world = dist.get_world_size()
        gathered = [None] * world
        dist.all_gather_object(gathered, (self._wsum, self._wtot))
        for wsum, wtot in gathered:
            for k, s in wsum.items():
                self._ssum[k] = self._ssum.get(k, 0.0) + s
                self._stot[k] = self._stot.get(k, 0) + wtot[k]
        self._wsum = {}
        self._wtot = {}

    # -------------------------------------------------------------- rounds
    def safe(self, write):
        """Bracket a round: True opens the event writer; False closes it and
        snapshots every key's current mean into `history`."""
        if write:
            if self.log_path is not None:
                if _HAS_TB:
                    self.writer = SummaryWriter(self.log_path)
                else:
                    self.writer = _Jsonl(self.log_path)
            return
        if self.writer is not None:
            self.writer.close()
            self.writer = None
        for k, m in self.mean.items():
            self.history.setdefault(k, []).append(m)

    def reset(self):
        self._wsum = {}
        self._wtot = {}
        self._ssum = {}
        self._stot = {}
        self.tracker = {}

    # -------------------------------------------------------------- logging
    def append(self, result, tag, n=1, mean=True):
        """Record a batch of named values under `tag/`; scalars with
        mean=True join the weighted series with weight n (= batch size)."""
        for name, value in result.items():
            key = '{}/{}'.format(tag, name)
            self.tracker[key] = value
            if mean and _is_scalar(value):
                self._wsum[key] = self._wsum.get(key, 0.0) + n * value
                self._wtot[key] = self._wtot.get(key, 0) + n

    def write(self, tag, metric_names):
        """Emit one console line (info prefix + current means) and push each
        numeric mean to the event writer."""
        parts = []
        info = self.tracker.get('{}/info'.format(tag))
        if isinstance(info, dict):
            info = info.get('info')
        if isinstance(info, list):
            parts.extend(str(x) for x in info[:3])
        means = self.mean
        for name in metric_names:
            key = '{}/{}'.format(tag, name)
            raw = self.tracker.get(key)
            if _is_scalar(raw):
                m = means[key]
                parts.append('{}: {:.4f}'.format(name, m))
                if self.writer is not None:
                    step = self._step.get(key, 0) + 1
                    self._step[key] = step
                    self.writer.add_scalar(key, m, step)
            elif isinstance(raw, list) and raw:
                parts.append('{}: {}'.format(name, raw[0]))
        print(' | '.join(parts))

    def flush(self):
        if self.writer is not None:
            self.writer.flush()

    # ------------------------------------------------------------- pickling
    def __getstate__(self):
        # writer handles never survive the checkpoint
        state = dict(self.__dict__)
        state['writer'] = None
        return state

    def __setstate__(self, state):
        # migrate checkpoints pickled by the earlier running-mean layout
        # (mean/counter/iterator attributes) into the series form
        state.setdefault('_ssum', {})
        state.setdefault('_stot', {})
        if '_wsum' not in state:
            counter = state.pop('counter', {}) or {}
            old_mean = state.pop('mean', {}) or {}
            state['_wsum'] = {k: old_mean.get(k, 0) * w
                              for k, w in counter.items()}
            state['_wtot'] = dict(counter)
            state['_step'] = dict(state.pop('iterator', {}) or {})
            state['_ssum'] = {}
            state['_stot'] = {}
            state.setdefault('tracker', {})
            state['history'] = dict(state.get('history', {}) or {})
        self.__dict__.update(state)


class _Jsonl:
    """Minimal SummaryWriter stand-in writing JSON lines."""

    def __init__(self, path):
        makedir_exist_ok(path)
        self.f = open(os.path.join(path, 'events.jsonl'), 'a')

    def add_scalar(self, name, value, step):
        self.f.write(json.dumps({'name': name, 'value': value,
                                 'step': step}) + '\n')

    def flush(self):
        self.f.flush()

    def close(self):
        self.f.close()
