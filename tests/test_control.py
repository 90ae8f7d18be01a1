"""control_name grammar tests against hand-computed expansions
(reference semantics: src/utils.py:113-215)."""
import os

import numpy as np
import pytest

from heterofl_amd.config import default_config
from heterofl_amd.control import (process_control, parse_control_name,
                                  parse_model_mode, MODEL_SPLIT_RATE,
                                  CONTROL_FIELDS)


def _cfg(control_name, data_name='CIFAR10'):
    cfg = default_config()
    cfg['control'] = parse_control_name(control_name)
    cfg['data_name'] = data_name
    process_control(cfg)
    return cfg


def test_rates():
    assert MODEL_SPLIT_RATE == {'a': 1, 'b': 0.5, 'c': 0.25, 'd': 0.125, 'e': 0.0625}


def test_parse_fields():
    c = parse_control_name('1_100_0.1_iid_fix_a1-e1_bn_1_1')
    assert c['num_users'] == '100' and c['frac'] == '0.1'
    assert c['model_mode'] == 'a1-e1' and c['norm'] == 'bn'
    with pytest.raises(ValueError):
        parse_control_name('1_100_0.1_iid')


def test_fix_partition_a1_e1():
    cfg = _cfg('1_100_0.1_iid_fix_a1-e1_bn_1_1')
    # 100 users // (1+1) = 50 per term: first 50 at rate 1, next 50 at 1/16
    assert cfg['model_rate'][:50] == [1] * 50
    assert cfg['model_rate'][50:] == [0.0625] * 50
    assert cfg['global_model_rate'] == 1


def test_fix_partition_remainder():
    cfg = _cfg('1_10_0.5_iid_fix_a1-b2_bn_1_1')
    # 10 // 3 = 3 per unit: 3 at a, 6 at b, 1 leftover gets last rate (b)
    assert cfg['model_rate'] == [1] * 3 + [0.5] * 7


def test_dynamic_proportion():
    cfg = _cfg('1_100_0.1_iid_dynamic_a1-b1-c2_bn_1_1')
    assert cfg['model_rate'] == [1, 0.5, 0.25]
    assert np.allclose(cfg['proportion'], [0.25, 0.25, 0.5])


def test_hyperparameters_cifar_iid():
    cfg = _cfg('1_100_0.1_iid_fix_a1_bn_1_1', 'CIFAR10')
    assert cfg['lr'] == 0.1 and cfg['optimizer_name'] == 'SGD'
    assert cfg['num_epochs'] == {'global': 400, 'local': 5}
    assert cfg['batch_size'] == {'train': 10, 'test': 50}
    assert cfg['milestones'] == [150, 250]
    assert cfg['data_shape'] == [3, 32, 32]


def test_hyperparameters_cifar_noniid():
    cfg = _cfg('1_100_0.1_non-iid-2_fix_a1_gn_1_1', 'CIFAR10')
    assert cfg['num_epochs'] == {'global': 800, 'local': 5}
    assert cfg['milestones'] == [300, 500]
    assert cfg['norm'] == 'gn'


def test_hyperparameters_mnist():
    cfg = _cfg('1_100_0.1_iid_fix_a1_bn_1_1', 'MNIST')
    assert cfg['lr'] == 1e-2
    assert cfg['num_epochs'] == {'global': 200, 'local': 5}


def test_hyperparameters_wikitext2():
    cfg = _cfg('1_100_0.1_iid_fix_a1-e1_bn_1_1', 'WikiText2')
    assert cfg['bptt'] == 64 and cfg['mask_rate'] == 0.15
    assert cfg['num_epochs'] == {'global': 200, 'local': 1}
    assert cfg['batch_size'] == {'train': 100, 'test': 10}
    assert cfg['transformer'] == {'embedding_size': 256, 'num_heads': 8,
                                  'hidden_size': 512, 'num_layers': 4,
                                  'dropout': 0.2}


def test_hyperparameters_centralized_none():
    """'none' data_split_mode branches used by the centralized entries
    (reference: src/utils.py:166-170, 188-192, 207-211)."""
    cfg = _cfg('0_1_1_none_fix_a1_bn_1_1', 'MNIST')
    assert cfg['num_epochs'] == 200
    assert cfg['batch_size'] == {'train': 100, 'test': 500}
    assert cfg['milestones'] == [100]
    cfg = _cfg('0_1_1_none_fix_a1_bn_1_1', 'CIFAR10')
    assert cfg['num_epochs'] == 400
    assert cfg['batch_size'] == {'train': 100, 'test': 500}
    assert cfg['milestones'] == [150, 250]
    cfg = _cfg('0_1_1_none_fix_a1_bn_1_1', 'WikiText2')
    assert cfg['num_epochs'] == 100
    assert cfg['batch_size'] == {'train': 100, 'test': 100}
    assert cfg['milestones'] == [25, 50]
    with pytest.raises(ValueError):
        _cfg('0_1_1_bogus_fix_a1_bn_1_1', 'CIFAR10')


def test_hyperparameters_mnist_noniid():
    cfg = _cfg('1_100_0.1_non-iid-2_fix_a1_bn_1_1', 'MNIST')
    assert cfg['num_epochs'] == {'global': 400, 'local': 5}
    assert cfg['milestones'] == [200]
    assert cfg['batch_size'] == {'train': 10, 'test': 50}


def test_scale_mask_flags():
    cfg = _cfg('1_10_0.1_iid_fix_a1_bn_0_0')
    assert cfg['scale'] is False and cfg['mask'] is False


def test_model_mode_parse():
    rates, props = parse_model_mode('a5-b10-e1')
    assert rates == [1, 0.5, 0.0625] and props == [5, 10, 1]


def test_profiler_summary():
    """Analytic cost profiler: full-rate resnet18 on CIFAR shape is ~11.2M
    params (reference poster quotes 9.6M trainable-equivalent scale)."""
    from tests.conftest import make_cfg
    from heterofl_amd.config import default_config
    from heterofl_amd.profiler import summarize_level
    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg = make_cfg(cfg, '1_10_0.1_iid_fix_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    s_full = summarize_level(cfg, 1.0)
    s_e = summarize_level(cfg, 0.0625)
    assert 9e6 < s_full['num_params'] < 13e6
    assert s_e['num_params'] < s_full['num_params'] / 100
    assert s_full['num_flops'] > 10 * s_e['num_flops']


def test_process_make_stats(tmp_path):
    from process import make_stats
    from heterofl_amd.utils import save
    for lv, p in [('a', 100), ('e', 2)]:
        save({'num_params': p, 'num_flops': 10 * p, 'space': p / 4},
             str(tmp_path / 'CIFAR10_resnet18_{}.pt'.format(lv)))
    st = make_stats('CIFAR10', 'resnet18', 'a1-e1', {'a': 1, 'e': 0.0625},
                    result_dir=str(tmp_path))
    assert abs(st['Params'] - 51) < 1e-6
    assert abs(st['Ratio'] - 0.51) < 1e-6


def test_process_plots(tmp_path):
    """Learning-curve + interpolation plots (reference: src/process.py
    visualization half)."""
    pytest.importorskip('matplotlib')
    from process import (crawl_results, aggregate, make_learning_curves,
                         make_interpolation_plot)
    from heterofl_amd.logger import Logger
    from heterofl_amd.utils import save
    rdir = tmp_path / 'result'
    rdir.mkdir()
    # two interpolation points x2 seeds, with per-round history
    for mode, acc in [('a3-e7', 60.0), ('a7-e3', 70.0)]:
        for seed in (0, 1):
            lg = Logger(str(tmp_path / 'runs' / f'{seed}_{mode}'))
            for ep in range(3):
                lg.safe(True)
                lg.append({'Global-Accuracy': acc + ep + seed}, 'test', n=10)
                lg.safe(False)
                if ep < 2:   # keep the final epoch's running mean, as the
                    lg.reset()   # entries do before pickling the logger
            tag = f'{seed}_CIFAR10_label_resnet18_1_100_0.1_iid_fix_{mode}_bn_1_1'
            save({'cfg': {}, 'epoch': 3,
                  'logger': {'train': None, 'test': lg}},
                 str(rdir / f'{tag}.pt'))
    results = crawl_results(str(rdir))
    assert len(results) == 2 and all(len(v) == 2 for v in results.values())
    curves = make_learning_curves(results, str(tmp_path / 'vis'))
    assert len(curves) == 2 and all(os.path.exists(p) for p in curves)
    table = aggregate(results)
    interp = make_interpolation_plot(table, str(tmp_path / 'vis'))
    assert len(interp) == 1 and os.path.exists(interp[0])


def test_make_sweep_generator(tmp_path, monkeypatch):
    """make.py emits a bash sweep (reference: src/make.py:88-101)."""
    import subprocess, sys, os
    out = subprocess.run(
        [sys.executable, os.path.join(os.path.dirname(__file__), '..',
                                      'make.py'),
         '--num_gpus', '8', '--num_experiments', '1'],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    sh = [f for f in os.listdir(tmp_path) if f.endswith('.sh')]
    assert sh, out.stdout
    body = open(str(tmp_path / sh[0])).read()
    assert 'HIP_VISIBLE_DEVICES' in body and 'control_name' in body


def test_make_sweep_composition():
    """Sweep sizes pin the reference's enumeration: 5 fixed levels, 26
    multi-level dynamic combinations, 90 interpolation ratios
    (reference: src/make.py:55-66)."""
    from make import fixed_combinations
    modes, dynamic, interp = fixed_combinations()
    assert len(modes) == 5
    assert len(dynamic) == 26      # C(5,2)+C(5,3)+C(5,4)+C(5,5)
    assert len(interp) == 90       # 10 level pairs x 9 ratios
    assert 'a1-b1-c1-d1-e1' in dynamic
    assert 'a5-e5' in interp and 'd9-e1' in interp


def test_summary_cli(tmp_path):
    """summary.py writes one cost file per width level
    (reference: src/summary.py:44-47)."""
    import subprocess, sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, 'summary.py'),
         '--data_name', 'MNIST', '--model_name', 'conv',
         '--control_name', '1_4_0.5_iid_fix_a1-e1_bn_1_1',
         '--synthetic', '1'],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=300,
        env=dict(os.environ, PYTHONPATH=root))
    assert out.returncode == 0, out.stderr
    from heterofl_amd.utils import load
    for lv in 'abcde':
        p = tmp_path / 'output' / 'result' / f'MNIST_conv_{lv}.pt'
        assert p.exists(), (lv, out.stdout)
    a = load(str(tmp_path / 'output' / 'result' / 'MNIST_conv_a.pt'))
    e = load(str(tmp_path / 'output' / 'result' / 'MNIST_conv_e.pt'))
    assert a['num_params'] > 50 * e['num_params']


def test_make_ablation_generator(tmp_path):
    """make_ablation.py emits the norm/scaler/mask ablation sweep
    (reference: src/make_ablation.py:69-88)."""
    import subprocess, sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, 'make_ablation.py'),
         '--num_gpus', '4', '--num_experiments', '1'],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    sh = [f for f in os.listdir(tmp_path) if f.endswith('.sh')]
    assert sh, out.stdout
    body = open(str(tmp_path / sh[0])).read()
    # ablation axes present: a non-bn norm, scaler off, mask off
    assert '_gn_' in body or '_in_' in body or '_ln_' in body
    assert '_0_1' in body or '_1_0' in body or '_0_0' in body


def test_mnist_idx_parser_roundtrip(tmp_path):
    """Real-file MNIST idx parser (reference: src/datasets/mnist.py)."""
    import gzip
    import struct
    import numpy as np
    from heterofl_amd.data import fetch_dataset
    raw = tmp_path / 'raw'
    raw.mkdir()
    rng = np.random.RandomState(0)
    for split, prefix, n in [('train', 'train', 12), ('test', 't10k', 6)]:
        imgs = rng.randint(0, 256, (n, 28, 28), dtype=np.uint8)
        labs = rng.randint(0, 10, (n,), dtype=np.uint8)
        with open(raw / f'{prefix}-images-idx3-ubyte', 'wb') as f:
            f.write(struct.pack('>I', 0x00000803))
            f.write(struct.pack('>III', n, 28, 28))
            f.write(imgs.tobytes())
        with gzip.open(raw / f'{prefix}-labels-idx1-ubyte.gz', 'wb') as f:
            f.write(struct.pack('>I', 0x00000801))
            f.write(struct.pack('>I', n))
            f.write(labs.tobytes())
    ds = fetch_dataset('MNIST', root=str(tmp_path))
    assert len(ds['train']) == 12 and len(ds['test']) == 6
    item = ds['train'][0]
    assert item['img'].shape == (1, 28, 28)
    assert 0 <= int(item['label']) < 10


def test_cifar_pickle_parser_roundtrip(tmp_path):
    """Real-file CIFAR10 pickle parser (reference: src/datasets/cifar.py)."""
    import pickle
    import numpy as np
    from heterofl_amd.data import fetch_dataset
    base = tmp_path / 'cifar-10-batches-py'
    base.mkdir()
    rng = np.random.RandomState(0)
    for fn, n in [('data_batch_%d' % i, 4) for i in range(1, 6)] + \
                 [('test_batch', 8)]:
        d = {b'data': rng.randint(0, 256, (n, 3072), dtype=np.uint8),
             b'labels': rng.randint(0, 10, (n,)).tolist()}
        with open(base / fn, 'wb') as f:
            pickle.dump(d, f)
    ds = fetch_dataset('CIFAR10', root=str(tmp_path))
    assert len(ds['train']) == 20 and len(ds['test']) == 8
    assert ds['train'][0]['img'].shape == (3, 32, 32)


def test_imagefolder_parser_roundtrip(tmp_path):
    """Class-per-subdir image tree (reference: src/datasets/folder.py)."""
    import numpy as np
    from PIL import Image
    from heterofl_amd.data import fetch_dataset
    rng = np.random.RandomState(0)
    for split, n in [('train', 3), ('test', 2)]:
        for cls in ['cat', 'dog']:
            d = tmp_path / split / cls
            d.mkdir(parents=True)
            for i in range(n):
                arr = rng.randint(0, 256, (16, 16, 3), dtype=np.uint8)
                Image.fromarray(arr).save(d / f'{i}.png')
    ds = fetch_dataset('ImageFolder', root=str(tmp_path))
    assert len(ds['train']) == 6 and len(ds['test']) == 4
    item = ds['train'][0]
    assert item['img'].shape == (3, 16, 16)
    assert ds['train'].classes_size == 2
    # sorted dir names define labels: cat=0, dog=1
    assert int(ds['train'][0]['label']) == 0
    assert int(ds['train'][5]['label']) == 1


def test_wikitext_raw_parser(tmp_path):
    from heterofl_amd.data import fetch_dataset
    (tmp_path / 'wiki.train.tokens').write_text('a b c a b c d e')
    (tmp_path / 'wiki.test.tokens').write_text('a b x')
    ds = fetch_dataset('WikiText2', root=str(tmp_path))
    assert len(ds['train'].vocab) >= 6   # <unk> + a b c d e
    assert ds['train'].token.numel() == 8
    # vocab is train-built (reference: src/datasets/lm.py:157-164);
    # unseen-at-train 'x' maps to <unk>
    assert ds['test'].token.numel() == 3
    unk = ds['train'].vocab['<unk>']
    assert int(ds['test'].token[2]) == unk


def test_noniid_split_properties():
    """non-iid-2: every user's train indices carry only their 2 assigned
    classes, and the train indices partition the kept samples
    (reference: src/data.py:79-110)."""
    import torch
    from heterofl_amd.data import fetch_dataset, split_dataset
    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=400)
    data_split, label_split = split_dataset(ds, 10, 'non-iid-2',
                                            classes_size=10)
    all_idx = []
    for u in range(10):
        labels = {int(ds['train'].target[i]) for i in data_split['train'][u]}
        assert labels <= set(label_split[u]), (u, labels, label_split[u])
        assert len(label_split[u]) <= 2
        all_idx.extend(data_split['train'][u])
    assert len(all_idx) == len(set(all_idx))   # disjoint shards
    # 2 shards x 10 users / 10 classes = 2 shards per class -> full coverage
    assert len(all_idx) == len(ds['train'])
    # test split follows the SAME label assignment
    for u in range(10):
        tl = {int(ds['test'].target[i]) for i in data_split['test'][u]}
        assert tl <= set(label_split[u]), u


def test_iid_split_properties():
    """iid: equal disjoint shards; label_split records observed labels
    (reference: src/data.py:61-76)."""
    import torch
    from heterofl_amd.data import fetch_dataset, split_dataset
    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=100)
    data_split, label_split = split_dataset(ds, 4, 'iid', classes_size=10)
    sizes = [len(data_split['train'][u]) for u in range(4)]
    assert sizes == [25, 25, 25, 25]
    seen = sum((data_split['train'][u] for u in range(4)), [])
    assert len(seen) == len(set(seen))
    for u in range(4):
        labels = {int(ds['train'].target[i]) for i in data_split['train'][u]}
        assert labels == set(label_split[u])
