"""Unit tests for the utility layer: Stats, batchify, optimizer/scheduler
factories, logger running means, metrics registry."""
import math

import pytest
import torch

from heterofl_amd.logger import Logger
from heterofl_amd.metrics import Metric
from heterofl_amd.utils import (Stats, batchify, make_optimizer,
                                make_scheduler, collate, to_device, recur)


def test_stats_running_mean_std():
    """Chunked updates equal whole-tensor statistics
    (reference: src/utils.py:231-257)."""
    torch.manual_seed(0)
    data = torch.randn(100, 8)
    s = Stats(dim=1)
    for chunk in data.split(25, dim=0):
        s.update(chunk)
    assert torch.allclose(s.mean, data.mean(0), atol=1e-5)
    assert torch.allclose(s.std, data.std(0, unbiased=False), atol=1e-5)


def test_batchify_folds_stream():
    """LM stream -> (batch_size, -1) (reference: src/utils.py:353-357)."""
    from heterofl_amd.data import LanguageModeling, Vocab
    ds = LanguageModeling('WikiText2', torch.arange(103), Vocab(['a']))
    batchify(ds, 10)
    assert ds.token.shape == (10, 10)
    assert ds.token[0, 0] == 0 and ds.token[-1, -1] == 99


def test_optimizer_factory_kinds():
    model = torch.nn.Linear(4, 2)
    cfg = {'momentum': 0.9, 'weight_decay': 5e-4}
    for name, cls in [('SGD', torch.optim.SGD),
                      ('RMSprop', torch.optim.RMSprop),
                      ('Adam', torch.optim.Adam),
                      ('Adamax', torch.optim.Adamax)]:
        opt = make_optimizer(model, 0.1, dict(cfg, optimizer_name=name))
        assert isinstance(opt, cls)
    with pytest.raises(ValueError):
        make_optimizer(model, 0.1, dict(cfg, optimizer_name='nope'))


def test_scheduler_factory_kinds():
    model = torch.nn.Linear(4, 2)
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    base = {'step_size': 1, 'milestones': [2, 4], 'factor': 0.5,
            'patience': 2, 'threshold': 1e-3, 'min_lr': 1e-4, 'lr': 0.1,
            'num_epochs': {'global': 10}}
    for name in ['None', 'StepLR', 'MultiStepLR', 'ExponentialLR',
                 'CosineAnnealingLR', 'ReduceLROnPlateau', 'CyclicLR']:
        sch = make_scheduler(opt, dict(base, scheduler_name=name))
        assert sch is not None
    # MultiStepLR decays at the milestones (reference hyperparameters use
    # milestones 150/250 with factor 0.1)
    opt2 = torch.optim.SGD(model.parameters(), lr=1.0)
    sch2 = make_scheduler(opt2, dict(base, scheduler_name='MultiStepLR',
                                     factor=0.1))
    lrs = []
    for _ in range(5):
        lrs.append(opt2.param_groups[0]['lr'])
        sch2.step()
    assert lrs[0] == 1.0 and abs(lrs[2] - 0.1) < 1e-9 \
        and abs(lrs[4] - 0.01) < 1e-9


def test_logger_running_means_and_history(tmp_path):
    lg = Logger(str(tmp_path / 'run'))
    lg.safe(True)
    lg.append({'train/Loss': 2.0}, 'train', n=10)
    lg.append({'train/Loss': 4.0}, 'train', n=30)
    assert abs(lg.mean['train/train/Loss'] - 3.5) < 1e-9
    lg.safe(False)
    assert lg.history['train/train/Loss']
    lg.reset()
    assert lg.mean == {} or all(v == 0 for v in lg.tracker.values())


def test_metric_registry_flavors():
    m = Metric()
    score = torch.tensor([[2.0, 1.0], [0.0, 3.0]])
    inp = {'label': torch.tensor([0, 1])}
    out = {'score': score, 'loss': torch.tensor(0.5)}
    ev = m.evaluate(['Loss', 'Local-Accuracy', 'Global-Accuracy'], inp, out)
    assert ev['Loss'] == 0.5
    assert ev['Local-Accuracy'] == 100.0 == ev['Global-Accuracy']
    pp = m.evaluate(['Perplexity'], inp, out)['Perplexity']
    ce = torch.nn.functional.cross_entropy(score, inp['label'])
    assert abs(pp - math.exp(ce.item())) < 1e-4


def test_recur_and_collate_and_to_device():
    x = {'a': [torch.ones(2), torch.zeros(2)], 'b': 'keep'}
    y = recur(lambda t: t + 1, x)
    assert y['b'] == 'keep' and torch.equal(y['a'][0], torch.full((2,), 2.0))
    c = collate({'img': [torch.ones(3), torch.zeros(3)]})
    assert c['img'].shape == (2, 3)
    d = to_device({'t': torch.ones(1)}, 'cpu')
    assert d['t'].device.type == 'cpu'


def test_no_undefined_names_static():
    """Static undefined-name sweep over the package (scripts/namecheck.py):
    GPU-only code paths never execute in CPU CI, so a typo there would
    otherwise only surface on the GPU box."""
    import importlib.util
    import os
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    spec = importlib.util.spec_from_file_location(
        'namecheck', os.path.join(root, 'scripts', 'namecheck.py'))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    problems = mod.check_tree(os.path.join(root, 'heterofl_amd'))
    assert not problems, problems
