"""Running-mean tracker with per-key counters, epoch history snapshots and
console line assembly (reference: src/logger.py:8-87).

The reference writes TensorBoard summaries; tensorboard is optional here —
when it is absent the Logger writes a JSONL event stream instead, so the
object stays picklable into checkpoints exactly like the reference
(reference: src/train_classifier_fed.py:85-88).
"""
import json
import os
from collections import defaultdict

from .utils.core import makedir_exist_ok

try:
    from torch.utils.tensorboard import SummaryWriter  # needs tensorboard pkg
    _HAS_TB = True
except Exception:
    SummaryWriter = None
    _HAS_TB = False


class Logger:
    def __init__(self, log_path):
        self.log_path = log_path
        self.writer = None
        self.tracker = defaultdict(int)
        self.counter = defaultdict(int)
        self.mean = defaultdict(int)
        self.history = defaultdict(list)
        self.iterator = defaultdict(int)

    def safe(self, write):
        """Bracket a round: open the writer on True, snapshot means into
        history and close on False (reference: src/logger.py:18-27)."""
        if write:
            self.writer = SummaryWriter(self.log_path) if _HAS_TB else _Jsonl(self.log_path)
        else:
            if self.writer is not None:
                self.writer.close()
                self.writer = None
            for name in self.mean:
                self.history[name].append(self.mean[name])
        return

    def reset(self):
        self.tracker = defaultdict(int)
        self.counter = defaultdict(int)
        self.mean = defaultdict(int)
        return

    def append(self, result, tag, n=1, mean=True):
        for k in result:
            name = '{}/{}'.format(tag, k)
            self.tracker[name] = result[k]
            if mean and isinstance(result[k], (int, float)):
                self.counter[name] += n
                self.mean[name] = ((self.counter[name] - n) * self.mean[name]
                                   + n * result[k]) / self.counter[name]
        return

    def write(self, tag, metric_names):
        names = ['{}/{}'.format(tag, k) for k in metric_names]
        evaluation_info = []
        for name in names:
            tag_, k = name.split('/')
            if isinstance(self.tracker[name], (int, float)):
                s = self.mean[name]
                evaluation_info.append('{}: {:.4f}'.format(k, s))
                if self.writer is not None:
                    self.iterator[name] += 1
                    self.writer.add_scalar(name, s, self.iterator[name])
            elif isinstance(self.tracker[name], list) and self.tracker[name]:
                evaluation_info.append('{}: {}'.format(k, self.tracker[name][0]))
        info_name = '{}/info'.format(tag)
        info = self.tracker[info_name]
        if isinstance(info, dict) and 'info' in info:
            info = info['info']
        if isinstance(info, list):
            print(' | '.join(info[:3] + evaluation_info))
        else:
            print(' | '.join(evaluation_info))
        return

    def flush(self):
        if self.writer is not None:
            self.writer.flush()

    # keep the pickled checkpoint logger usable (writer handles are dropped)
    def __getstate__(self):
        state = self.__dict__.copy()
        state['writer'] = None
        return state


class _Jsonl:
    """Minimal SummaryWriter stand-in writing JSON lines."""

    def __init__(self, path):
        makedir_exist_ok(path)
        self.f = open(os.path.join(path, 'events.jsonl'), 'a')

    def add_scalar(self, name, value, step):
        self.f.write(json.dumps({'name': name, 'value': value, 'step': step}) + '\n')

    def flush(self):
        self.f.flush()

    def close(self):
        self.f.close()
