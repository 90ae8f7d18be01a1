"""End-to-end federated training on CPU with synthetic data
(BASELINE config 1: MNIST conv, 10 clients IID, fix a1 = homogeneous FedAvg).
"""
import torch
import pytest

from heterofl_amd.data import fetch_dataset, split_dataset
from heterofl_amd.fed import FedRunner
from heterofl_amd.models import make_model
from heterofl_amd.utils import process_dataset, make_optimizer
from tests.conftest import make_cfg


def _run(cfg, rounds=2, n_data=200):
    torch.manual_seed(0)
    ds = fetch_dataset(cfg['data_name'], synthetic=True, synthetic_size=n_data)
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, cfg['num_users'],
                                            cfg['data_split_mode'],
                                            cfg.get('classes_size'))
    model = make_model(cfg)
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)
    losses = []
    for ep in range(1, rounds + 1):
        runner.train_round(ep)
    return runner


def test_mnist_conv_fedavg(base_cfg):
    cfg = make_cfg(base_cfg, '1_10_0.3_iid_fix_a1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    cfg['num_epochs'] = {'global': 2, 'local': 1}
    runner = _run(cfg)
    tm = runner.stats()
    # sBN stats pass produced running buffers
    sd = tm.state_dict()
    assert any('running_mean' in k for k in sd)
    runner.test(tm, 2)


def test_cifar_resnet_heterogeneous(base_cfg):
    cfg = make_cfg(base_cfg, '1_4_0.5_iid_fix_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    runner = _run(cfg, rounds=1, n_data=80)


def test_cifar_noniid_dynamic(base_cfg):
    cfg = make_cfg(base_cfg, '1_10_0.2_non-iid-2_dynamic_a1-e1_gn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    runner = _run(cfg, rounds=1, n_data=400)


def test_transformer_fed(base_cfg):
    cfg = make_cfg(base_cfg, '1_4_0.5_iid_fix_a1-e1_bn_1_1',
                   data_name='WikiText2', model_name='transformer')
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                          'test': {'Global': ['Global-Loss', 'Global-Perplexity']}}
    torch.manual_seed(0)
    ds = fetch_dataset('WikiText2', synthetic=True, synthetic_size=120_000)
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, cfg['num_users'], 'iid')
    model = make_model(cfg)
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)
    runner.train_round(1)
    tm = runner.stats()
    runner.test(tm, 1)


def test_loss_decreases_homogeneous(base_cfg):
    """Training sanity: FedAvg on a tiny learnable problem reduces loss."""
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    cfg['num_epochs'] = {'global': 3, 'local': 2}
    cfg['lr'] = 0.05
    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=64)
    # make it learnable: label = pixel-sum parity buckets
    img = ds['train'].img.float()
    ds['train'].target = (img.view(64, -1).mean(1) > 127.5).long().tolist()
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 2, 'iid', cfg['classes_size'])
    model = make_model(cfg)
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)

    def global_loss():
        model.load_state_dict(runner.federation.global_parameters)
        model.train(True)
        with torch.no_grad():
            batch = {'img': torch.stack([ds['train'][i]['img'] for i in range(64)]),
                     'label': torch.tensor(ds['train'].target)}
            return model(batch)['loss'].item()

    l0 = global_loss()
    for ep in range(1, 4):
        runner.train_round(ep)
    l1 = global_loss()
    assert l1 < l0, (l0, l1)


def test_checkpoint_resume_roundtrip(base_cfg, tmp_path, monkeypatch):
    """Full entry-point run writes the reference checkpoint layout
    (./output/model/{tag}_checkpoint.pt + _best.pt) and resume_mode=1
    continues from the saved epoch (reference: src/utils.py:300-344)."""
    import os
    monkeypatch.chdir(tmp_path)
    from heterofl_amd.entry import run_fed_experiment
    from heterofl_amd.config import default_config
    cfg = default_config()
    cfg.update({'data_name': 'MNIST', 'model_name': 'conv', 'device': 'cpu',
                'engine': 'sequential', 'synthetic': True,
                'num_experiments': 1, 'init_seed': 0, 'resume_mode': 0})
    cfg['control'] = {'fed': '1', 'num_users': '4', 'frac': '0.5',
                      'data_split_mode': 'iid', 'model_split_mode': 'fix',
                      'model_mode': 'a1', 'norm': 'bn', 'scale': '1',
                      'mask': '1'}
    cfg['control_name'] = '1_4_0.5_iid_fix_a1_bn_1_1'
    import heterofl_amd.data as data_mod
    import heterofl_amd.entry as entry_mod
    orig = data_mod.fetch_dataset
    monkeypatch.setattr(
        'heterofl_amd.entry.fetch_dataset',
        lambda name, subset=None, synthetic=False, **kw: orig(
            name, subset, synthetic=True, synthetic_size=40))
    orig_pc = entry_mod.process_control

    def small_pc(c):
        orig_pc(c)
        c['num_epochs'] = {'global': 2, 'local': 1}

    monkeypatch.setattr(entry_mod, 'process_control', small_pc)
    metric_name = {'train': {'Local': ['Local-Loss', 'Local-Accuracy']},
                   'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                            'Global': ['Global-Loss', 'Global-Accuracy']}}
    run_fed_experiment(dict(cfg), 'Global-Accuracy', +1, metric_name)
    tag = '0_MNIST_label_conv_1_4_0.5_iid_fix_a1_bn_1_1'
    ck = './output/model/{}_checkpoint.pt'.format(tag)
    assert os.path.exists(ck), os.listdir('./output/model')
    assert os.path.exists('./output/model/{}_best.pt'.format(tag))
    from heterofl_amd.utils import load
    saved = load(ck)
    for key in ('cfg', 'epoch', 'data_split', 'label_split', 'model_dict',
                'optimizer_dict', 'scheduler_dict', 'logger'):
        assert key in saved, key
    assert saved['epoch'] == 3  # next epoch to run after 2 rounds
    # resume_mode=1: restarts from the saved epoch (past the end here, so
    # the run completes immediately but must load the checkpoint cleanly)
    run_fed_experiment(dict(cfg, resume_mode=1), 'Global-Accuracy', +1,
                       metric_name)
    # resume_mode=2: reload weights + splits only, train from epoch 1 with a
    # fresh logger (reference: src/train_classifier_fed.py:57-69)
    run_fed_experiment(dict(cfg, resume_mode=2), 'Global-Accuracy', +1,
                       metric_name)
    saved2 = load(ck)
    assert saved2['epoch'] == 3
    # the data/label splits must survive both resumes unchanged
    assert saved2['data_split']['train'].keys() == \
        saved['data_split']['train'].keys()


def test_resnet50_bottleneck_engines(base_cfg, monkeypatch):
    """Bottleneck resnets default to the batched BBottleneck engine;
    HETEROFL_BATCHED_BOTTLENECK=0 forces the sequential oracle."""
    from heterofl_amd.fed.batched import BatchedClientTrainer
    from heterofl_amd.fed.sequential import SequentialClientTrainer
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet50')
    cfg['engine'] = 'batched'
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    runner = _run(cfg, rounds=1, n_data=20)
    assert isinstance(runner.trainer, BatchedClientTrainer)
    monkeypatch.setenv('HETEROFL_BATCHED_BOTTLENECK', '0')
    runner = _run(cfg, rounds=1, n_data=20)
    assert isinstance(runner.trainer, SequentialClientTrainer)


@pytest.mark.parametrize('norm,scale,mask', [
    ('none', '1', '1'), ('in', '0', '1'), ('ln', '1', '0'),
    ('gn', '0', '0')])
def test_ablation_combos(base_cfg, norm, scale, mask):
    """Ablation grid (reference: src/make_ablation.py:69-88): every
    norm/scaler/mask combination trains a round end-to-end."""
    cfg = make_cfg(base_cfg,
                   f'1_4_0.5_iid_fix_a1-e1_{norm}_{scale}_{mask}',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    _run(cfg, rounds=1, n_data=40)


def test_centralized_entry(base_cfg, tmp_path, monkeypatch):
    """Non-fed baseline entry trains, evaluates and checkpoints
    (reference: src/train_classifier.py)."""
    import os
    monkeypatch.chdir(tmp_path)
    import heterofl_amd.entry as entry
    from heterofl_amd.config import default_config
    orig_pc = entry.process_control

    def small_pc(cfg):
        orig_pc(cfg)
        cfg['num_epochs'] = 2

    monkeypatch.setattr(entry, 'process_control', small_pc)
    import heterofl_amd.data as data_mod
    orig_fetch = data_mod.fetch_dataset
    monkeypatch.setattr(
        entry, 'fetch_dataset',
        lambda name, subset=None, synthetic=False, **kw: orig_fetch(
            name, subset, synthetic=True, synthetic_size=40))
    cfg = default_config()
    cfg.update({'data_name': 'MNIST', 'model_name': 'conv', 'device': 'cpu',
                'num_experiments': 1, 'init_seed': 0, 'resume_mode': 0,
                'batch_size': {'train': 10, 'test': 10}})
    cfg['control_name'] = '1_4_0.5_iid_fix_a1_bn_1_1'
    entry.run_centralized_experiment(
        cfg, 'Accuracy', +1,
        {'train': ['Loss', 'Accuracy'], 'test': ['Loss', 'Accuracy']})
    tag = cfg['model_tag']
    assert os.path.exists('./output/model/{}_checkpoint.pt'.format(tag))
    assert os.path.exists('./output/model/{}_best.pt'.format(tag))
    # centralized eval entry (reference: src/test_classifier.py): loads the
    # best checkpoint, re-runs sBN stats, saves ./output/result/{tag}.pt
    entry.run_centralized_eval(
        cfg, {'train': ['Loss', 'Accuracy'], 'test': ['Loss', 'Accuracy']})
    assert os.path.exists('./output/result/{}.pt'.format(tag))
    from heterofl_amd.utils import load
    result = load('./output/result/{}.pt'.format(tag))
    assert 'test/Accuracy' in result['logger']['test'].mean


def test_centralized_lm_entry(base_cfg, tmp_path, monkeypatch):
    """Centralized masked-LM train + eval entries
    (reference: src/train_transformer.py, src/test_transformer.py)."""
    import os
    monkeypatch.chdir(tmp_path)
    import heterofl_amd.entry as entry
    from heterofl_amd.config import default_config
    orig_pc = entry.process_control

    def small_pc(cfg):
        orig_pc(cfg)
        cfg['num_epochs'] = 1
        cfg['transformer'] = {'embedding_size': 16, 'num_heads': 2,
                              'hidden_size': 16, 'num_layers': 1,
                              'dropout': 0.0}
        cfg['bptt'] = 8

    monkeypatch.setattr(entry, 'process_control', small_pc)
    import heterofl_amd.data as data_mod
    orig_fetch = data_mod.fetch_dataset
    monkeypatch.setattr(
        entry, 'fetch_dataset',
        lambda name, subset=None, synthetic=False, **kw: orig_fetch(
            name, subset, synthetic=True, synthetic_size=400))
    cfg = default_config()
    cfg.update({'data_name': 'WikiText2', 'subset': 'label',
                'model_name': 'transformer', 'device': 'cpu',
                'num_experiments': 1, 'init_seed': 0, 'resume_mode': 0,
                'batch_size': {'train': 4, 'test': 4}})
    cfg['control'] = dict(cfg['control'], data_split_mode='none')
    cfg['control_name'] = '0_1_1_none_fix_a1_bn_1_1'
    entry.run_centralized_experiment(
        cfg, 'Perplexity', -1,
        {'train': ['Loss', 'Perplexity'], 'test': ['Loss', 'Perplexity']})
    tag = cfg['model_tag']
    assert os.path.exists('./output/model/{}_best.pt'.format(tag))
    entry.run_centralized_eval(
        cfg, {'train': ['Loss', 'Perplexity'],
              'test': ['Loss', 'Perplexity']})
    from heterofl_amd.utils import load
    result = load('./output/result/{}.pt'.format(tag))
    assert 'test/Perplexity' in result['logger']['test'].mean


@pytest.mark.parametrize('model_name', ['resnet34', 'resnet101', 'resnet152'])
def test_deep_resnet_factories(base_cfg, model_name):
    """Deep resnet factories build and run forward/backward at full and
    fractional width (reference: src/models/resnet.py:161-208)."""
    from tests.conftest import make_cfg
    from heterofl_amd.models import make_model
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1_bn_1_1',
                   data_name='CIFAR10', model_name=model_name)
    for rate in (1.0, 0.25):
        model = make_model(cfg, model_rate=rate)
        x = torch.randn(2, 3, 32, 32)
        out = model({'img': x, 'label': torch.tensor([0, 1]),
                     'label_split': torch.arange(10)})
        assert out['score'].shape == (2, 10)
        out['loss'].backward()
        assert all(p.grad is not None for p in model.parameters()
                   if p.requires_grad)


def test_golden_path_trajectory(base_cfg):
    """Golden-path regression: the BASELINE config-1 loss
    trajectory under fixed seeds is pinned so any refactor that changes the
    engine's numerics is caught immediately.  Values captured on this
    torch build; tolerance 1e-4 absorbs BLAS nondeterminism, not semantic
    changes."""
    cfg = make_cfg(base_cfg, '1_10_0.1_iid_fix_a1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    cfg['engine'] = 'sequential'
    cfg['num_epochs'] = {'global': 3, 'local': 1}
    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=100)
    process_dataset(ds, cfg)
    torch.manual_seed(7)
    data_split, label_split = split_dataset(ds, 10, 'iid', cfg['classes_size'])
    torch.manual_seed(1)
    model = make_model(cfg)
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)

    def global_loss():
        model.load_state_dict(runner.federation.global_parameters)
        model.train(True)
        with torch.no_grad():
            batch = {'img': torch.stack([ds['train'][i]['img']
                                         for i in range(100)]),
                     'label': torch.tensor(ds['train'].target)}
            return model(batch)['loss'].item()

    traj = []
    for ep in range(1, 4):
        torch.manual_seed(100 + ep)   # pin per-round batch shuffling
        runner.train_round(ep)
        traj.append(global_loss())
    golden = [2.297021, 2.291516, 2.288667]
    for got, want in zip(traj, golden):
        assert abs(got - want) < 1e-4, (traj, golden)


def test_cli_train_classifier_fed_subprocess(tmp_path):
    """The real train_classifier_fed.py CLI (argv -> checkpoint) runs
    end-to-end; HETEROFL_MAX_ROUNDS caps the round count for CI."""
    import json
    import os
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, HETEROFL_MAX_ROUNDS='2', PYTHONPATH=root,
               HETEROFL_SYNTHETIC_SIZE='40')
    out = subprocess.run(
        [sys.executable, os.path.join(root, 'train_classifier_fed.py'),
         '--data_name', 'MNIST', '--model_name', 'conv',
         '--control_name', '1_4_0.5_iid_fix_a1_bn_1_1',
         '--device', 'cpu', '--synthetic', '1'],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=900,
        env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    tag = '0_MNIST_label_conv_1_4_0.5_iid_fix_a1_bn_1_1'
    assert (tmp_path / 'output' / 'model' / f'{tag}_checkpoint.pt').exists(), \
        out.stdout[-2000:]
    assert (tmp_path / 'output' / 'model' / f'{tag}_best.pt').exists()


def test_run_to_run_reproducibility(base_cfg):
    """Two identically-seeded runs produce identical global parameters —
    the determinism discipline behind graph-replay bit-stability holds at
    the round-engine level too."""
    def run_once():
        cfg = make_cfg(base_cfg, '1_4_0.5_iid_fix_a1-e1_bn_1_1',
                       data_name='MNIST', model_name='conv')
        cfg['engine'] = 'sequential'
        cfg['num_epochs'] = {'global': 2, 'local': 1}
        torch.manual_seed(0)
        ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=40)
        process_dataset(ds, cfg)
        torch.manual_seed(7)
        data_split, label_split = split_dataset(ds, 4, 'iid',
                                                cfg['classes_size'])
        torch.manual_seed(1)
        model = make_model(cfg)
        opt = make_optimizer(model, cfg['lr'], cfg)
        runner = FedRunner(cfg, ds, data_split, label_split, model, opt)
        for ep in (1, 2):
            torch.manual_seed(50 + ep)
            runner.train_round(ep)
        return {k: v.clone() for k, v in
                runner.federation.global_parameters.items()}

    a, b = run_once(), run_once()
    for k in a:
        assert torch.equal(a[k], b[k]), k


def test_native_stats_pass_matches_eager(base_cfg, monkeypatch):
    """runner.stats() routed through the batched engine's stats sink
    (fused-BN mean/var harvest) reproduces the eager per-batch cumulative-BN
    pass exactly (MNIST: no train-time augmentation, so both passes are
    deterministic)."""
    cfg = make_cfg(base_cfg, '1_4_0.5_iid_fix_a1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    cfg['engine'] = 'batched'
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=45)
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 4, 'iid', cfg['classes_size'])
    model = make_model(cfg)
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)
    runner.train_round(1)
    assert runner._native_stats_ok()
    tm_native = runner.stats()
    monkeypatch.setenv('HETEROFL_NATIVE_STATS', '0')
    assert not runner._native_stats_ok()
    tm_eager = runner.stats()
    sd_n, sd_e = tm_native.state_dict(), tm_eager.state_dict()
    for k in sd_e:
        if 'running' in k or 'num_batches' in k:
            diff = (sd_n[k].float() - sd_e[k].float()).abs().max().item()
            assert diff < 1e-4, (k, diff)


def test_eval_every_knob(base_cfg, tmp_path, monkeypatch):
    """HETEROFL_EVAL_EVERY=k evaluates every k-th round plus the final
    round; skipped rounds never update the best-checkpoint pivot."""
    import os
    monkeypatch.chdir(tmp_path)
    monkeypatch.setenv('HETEROFL_EVAL_EVERY', '2')
    monkeypatch.setenv('HETEROFL_MAX_ROUNDS', '3')
    monkeypatch.setenv('HETEROFL_SYNTHETIC_SIZE', '40')
    from heterofl_amd.entry import run_fed_experiment
    from heterofl_amd.config import default_config
    from heterofl_amd.utils import load
    cfg = default_config()
    cfg.update({'data_name': 'MNIST', 'model_name': 'conv', 'device': 'cpu',
                'engine': 'sequential', 'synthetic': True,
                'num_experiments': 1, 'init_seed': 0, 'resume_mode': 0})
    cfg['control'] = {'fed': '1', 'num_users': '4', 'frac': '0.5',
                      'data_split_mode': 'iid', 'model_split_mode': 'fix',
                      'model_mode': 'a1', 'norm': 'bn', 'scale': '1',
                      'mask': '1'}
    cfg['control_name'] = '1_4_0.5_iid_fix_a1_bn_1_1'
    metric_name = {'train': {'Local': ['Local-Loss', 'Local-Accuracy']},
                   'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                            'Global': ['Global-Loss', 'Global-Accuracy']}}
    run_fed_experiment(dict(cfg), 'Global-Accuracy', +1, metric_name)
    tag = '0_MNIST_label_conv_1_4_0.5_iid_fix_a1_bn_1_1'
    ck = load('./output/model/{}_checkpoint.pt'.format(tag))
    hist = ck['logger'].history
    # rounds 2 and 3 (final) evaluated; round 1 skipped
    assert len(hist.get('test/Global-Accuracy', [])) == 2
    assert os.path.exists('./output/model/{}_best.pt'.format(tag))


def test_learnable_synthetic_mode(monkeypatch):
    """HETEROFL_SYNTHETIC_MODE=learnable: deterministic class-template data,
    train/test share the template bank but draw disjoint samples."""
    import torch
    from heterofl_amd.data import fetch_dataset
    monkeypatch.setenv('HETEROFL_SYNTHETIC_MODE', 'learnable')
    a = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=300)
    b = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=300)
    assert torch.equal(a['train'].img, b['train'].img)  # deterministic
    assert a['train'].img.shape == (300, 32, 32, 3)
    assert not torch.equal(a['train'].img[:60], a['test'].img)  # disjoint draw
    # labels roughly balanced over 10 classes
    counts = torch.bincount(torch.tensor(a['train'].target), minlength=10)
    assert int(counts.min()) > 10
