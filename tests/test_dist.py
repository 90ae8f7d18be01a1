"""Multi-process CPU tests (gloo, world_size 2): the padded all-reduce
combine must equal the sequential combine on the union of both ranks'
clients (SURVEY §7 step 6 verification contract)."""
import os

import pytest
import torch
import torch.multiprocessing as mp

from tests.conftest import make_cfg


def _worker(rank, world, port, tmpdir):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    os.environ['WORLD_SIZE'] = str(world)
    os.environ['RANK'] = str(rank)
    os.environ['LOCAL_RANK'] = str(rank)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    from heterofl_amd.config import default_config
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.models import make_model
    from heterofl_amd.parallel import init_distributed
    from heterofl_amd.utils import process_dataset, make_optimizer

    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg['engine'] = 'sequential'
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                          'test': {'Global': ['Global-Loss']}}
    cfg = make_cfg(cfg, '1_6_1_iid_fix_a1-e1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    cfg['world_size'] = world

    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=60)
    process_dataset(ds, cfg)
    torch.manual_seed(7)
    data_split, label_split = split_dataset(ds, 6, 'iid', cfg['classes_size'])
    torch.manual_seed(1)
    model = make_model(cfg)
    opt = make_optimizer(model, cfg['lr'], cfg)
    ctx = init_distributed(backend='gloo')
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt,
                       dist_ctx=ctx)
    runner.train_round(1)
    gp = {k: v.clone() for k, v in runner.federation.global_parameters.items()}

    # sequential oracle: same round on one rank, all clients
    cfg2 = dict(cfg)
    cfg2['world_size'] = 1
    torch.manual_seed(1)
    model2 = make_model(cfg2)
    opt2 = make_optimizer(model2, cfg2['lr'], cfg2)
    runner2 = FedRunner(cfg2, ds, data_split, label_split, model2, opt2)
    # force the same seeded sampling path the distributed round used
    runner2.cfg['world_size'] = 2
    runner2.dist_ctx = None
    g = runner2._round_generator(1)
    from heterofl_amd.fed.runner import sample_active_users
    user_idx = sample_active_users(cfg2, 1, generator=g)
    runner2.federation.make_model_rate(generator=g)
    local_parameters, param_idx = runner2.federation.distribute(
        user_idx, resample=False)
    trained = dict(runner2.trainer.train_clients(
        list(range(len(user_idx))), user_idx, local_parameters,
        runner2.federation.model_rate, runner2._make_loader,
        label_split, cfg2['lr']))
    ordered = [trained[m] for m in range(len(user_idx))]
    runner2.federation.combine(ordered, param_idx, user_idx)

    for k, v in runner2.federation.global_parameters.items():
        if v.is_floating_point():
            diff = (gp[k] - v).abs().max().item()
            assert diff < 1e-5, (rank, k, diff)
    torch.distributed.destroy_process_group()


def test_distributed_combine_matches_sequential(tmp_path):
    port = 29541
    mp.spawn(_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)


def _stats_worker(rank, world, port, tmpdir):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    from heterofl_amd.config import default_config
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.models import make_model
    from heterofl_amd.parallel import init_distributed
    from heterofl_amd.utils import process_dataset, make_optimizer

    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg['engine'] = 'sequential'
    cfg = make_cfg_local(cfg)
    torch.manual_seed(0)
    # 85 samples with batch 10 -> a partial tail batch, exercising the
    # whole-batch sharding that hands the tail to exactly one rank
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=85)
    process_dataset(ds, cfg)
    torch.manual_seed(7)
    data_split, label_split = split_dataset(ds, 4, 'iid', cfg['classes_size'])
    torch.manual_seed(1)
    model = make_model(cfg)
    ctx = init_distributed(backend='gloo')
    runner = FedRunner(cfg, ds, data_split, label_split, model,
                       make_optimizer(model, cfg['lr'], cfg), dist_ctx=ctx)
    tm = runner.stats()
    # sequential oracle on one rank
    runner1 = FedRunner(dict(cfg, world_size=1), ds, data_split, label_split,
                        make_model(cfg),
                        make_optimizer(model, cfg['lr'], cfg))
    runner1.federation.global_parameters = runner.federation.global_parameters
    runner1.global_model = runner.global_model
    tm1 = runner1.stats()
    for (k, v), (k1, v1) in zip(tm.state_dict().items(),
                                tm1.state_dict().items()):
        if 'running' in k:
            diff = (v.float() - v1.float()).abs().max().item()
            assert diff < 1e-4, (rank, k, diff)
    torch.distributed.destroy_process_group()


def make_cfg_local(cfg):
    cfg = dict(cfg)
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                          'test': {'Global': ['Global-Loss']}}
    c = make_cfg(cfg, '1_4_0.5_iid_fix_a1_bn_1_1',
                 data_name='MNIST', model_name='conv')
    c['num_epochs'] = {'global': 1, 'local': 1}
    c['world_size'] = 2
    c['batch_size'] = {'train': 10, 'test': 50}
    return c


def test_distributed_sbn_stats_matches_sequential(tmp_path):
    mp.spawn(_stats_worker, args=(2, 29551, str(tmp_path)), nprocs=2,
             join=True)


def test_round_sampling_rank_invariant():
    """Every rank draws identical active users and dynamic rates from the
    round-seeded generator — the no-communication reproducibility contract
    (fed/runner.py:98-132)."""
    from heterofl_amd.config import default_config
    from heterofl_amd.control import process_control
    from heterofl_amd.fed.runner import sample_active_users
    from heterofl_amd.fed.federation import Federation
    from heterofl_amd.models import make_model

    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg['data_name'] = 'CIFAR10'
    cfg['model_name'] = 'resnet18'
    cfg['control'] = {'fed': '1', 'num_users': '100', 'frac': '0.1',
                      'data_split_mode': 'iid', 'model_split_mode': 'dynamic',
                      'model_mode': 'a1-e1', 'norm': 'bn', 'scale': '1',
                      'mask': '1'}
    process_control(cfg)
    cfg['classes_size'] = 10

    def round_gen(epoch, seed=0):
        g = torch.Generator()
        g.manual_seed((seed * 1000003 + epoch) % (2 ** 63 - 1))
        return g

    model = make_model(cfg, model_rate=1.0)
    label_split = {i: [0, 1] for i in range(100)}
    seen = []
    for epoch in (1, 2):
        per_rank = []
        for rank in range(2):   # both "ranks" replay the same drawing
            g = round_gen(epoch)
            users = sample_active_users(cfg, epoch, generator=g)
            fed = Federation(model.state_dict(), cfg['model_rate'],
                             label_split, cfg)
            fed.make_model_rate(generator=g)
            per_rank.append((users, list(fed.model_rate)))
        assert per_rank[0] == per_rank[1], epoch
        assert len(per_rank[0][0]) == 10          # ceil(0.1 * 100)
        assert set(per_rank[0][1]) <= {1, 0.0625}  # dynamic a1-e1 levels
        seen.append(per_rank[0])
    assert seen[0] != seen[1]   # different rounds draw differently


def test_dynamic_rate_distribution():
    """Dynamic-mode multinomial follows the mode proportions
    (reference: src/fed.py:15-24 semantics)."""
    from heterofl_amd.config import default_config
    from heterofl_amd.control import process_control
    from heterofl_amd.fed.federation import Federation
    from heterofl_amd.models import make_model

    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg['data_name'] = 'CIFAR10'
    cfg['model_name'] = 'resnet18'
    cfg['control'] = {'fed': '1', 'num_users': '1000', 'frac': '0.1',
                      'data_split_mode': 'iid', 'model_split_mode': 'dynamic',
                      'model_mode': 'a1-e3', 'norm': 'bn', 'scale': '1',
                      'mask': '1'}
    process_control(cfg)
    cfg['classes_size'] = 10
    model = make_model(cfg, model_rate=1.0)
    fed = Federation(model.state_dict(), cfg['model_rate'],
                     {i: [0] for i in range(1000)}, cfg)
    g = torch.Generator().manual_seed(7)
    fed.make_model_rate(generator=g)
    frac_full = sum(1 for r in fed.model_rate if r == 1) / 1000
    assert 0.18 < frac_full < 0.32   # expected 0.25


def test_bench_torchrun_contract(tmp_path):
    """bench.py under torchrun (the driver's exact N>1 launch shape) prints
    ONE JSON line from rank 0 with the BASELINE metric/config."""
    import json
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, HETEROFL_BENCH_SPU='10', OMP_NUM_THREADS='2')
    out = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29531', os.path.join(root, 'bench.py'),
         '--gpus', '2', '--steps', '1', '--warmup', '0'],
        cwd=root, capture_output=True, text=True, timeout=600, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith('{')]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])
    assert rec['metric'] == 'local-train samples/sec/node'
    assert rec['n_gpus'] == 2 and rec['scaling'] == 'weak'
    assert rec['config']['num_users'] == 200   # weak scaling: 100 users/GPU
    assert rec['config']['active_clients'] == 20
    assert rec['value'] > 0


def _collective_worker(rank, world, port, tmpdir):
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    from heterofl_amd.parallel import init_distributed
    from heterofl_amd.parallel.dist import (broadcast_state_dict,
                                            allreduce_bn_stats)
    ctx = init_distributed(backend='gloo')
    # broadcast: rank 0's values win
    sd = {'w': torch.full((4, 3), float(rank)),
          'b': torch.full((5,), float(rank) + 10),
          'n': torch.tensor(rank, dtype=torch.long)}  # int: untouched
    broadcast_state_dict(sd, ctx, src=0)
    assert torch.equal(sd['w'], torch.zeros(4, 3)), rank
    assert torch.equal(sd['b'], torch.full((5,), 10.0)), rank
    assert sd['n'].item() == rank   # non-float buffers are not broadcast
    # allreduce: sums partials across ranks
    parts = [torch.full((3,), float(rank + 1)), torch.tensor([2.0 * rank])]
    out = allreduce_bn_stats(parts, ctx)
    assert torch.equal(out[0], torch.full((3,), 3.0)), rank   # 1 + 2
    assert torch.equal(out[1], torch.tensor([2.0])), rank     # 0 + 2
    torch.distributed.destroy_process_group()


def test_collective_helpers():
    """broadcast_state_dict / allreduce_bn_stats utilities (C1/C2 helpers;
    the round engine itself needs no W_g broadcast — every rank finalizes
    identical global params from the deterministic padded combine)."""
    mp.spawn(_collective_worker, args=(2, 29561, ''), nprocs=2, join=True)


def test_fed_cli_torchrun_ws2(tmp_path):
    """The REAL fed CLI entry (train_classifier_fed.py) under torchrun
    world_size=2 on gloo: clients shard across ranks, evaluation is sharded
    + merged, and only rank 0 writes the checkpoint."""
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, HETEROFL_MAX_ROUNDS='2',
               HETEROFL_SYNTHETIC_SIZE='40', OMP_NUM_THREADS='2')
    out = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29571',
         os.path.join(root, 'train_classifier_fed.py'),
         '--data_name', 'MNIST', '--model_name', 'conv',
         '--device', 'cpu', '--synthetic', '1', '--num_experiments', '1',
         '--control_name', '1_4_0.5_iid_fix_a1-e1_bn_1_1'],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=600,
        env=env)
    assert out.returncode == 0, out.stderr[-3000:]
    tag = '0_MNIST_label_conv_1_4_0.5_iid_fix_a1-e1_bn_1_1'
    ck = tmp_path / 'output' / 'model' / f'{tag}_checkpoint.pt'
    assert ck.exists(), out.stdout[-2000:]
    # exactly one rank printed the per-round console lines
    assert out.stdout.count('Experiment: ') == 1, out.stdout[-2000:]


def _missing_client_worker(rank, world, port, tmpdir):
    """One rank's client never reports: the scalar completeness check must
    route the combine onto the count-reducing fallback and still match the
    sequential combine over the REPORTING clients."""
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    from heterofl_amd.config import default_config
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.models import make_model
    from heterofl_amd.parallel import init_distributed
    from heterofl_amd.parallel.dist import distributed_combine
    from heterofl_amd.utils import process_dataset, make_optimizer
    from heterofl_amd.fed.runner import sample_active_users

    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg['engine'] = 'sequential'
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                          'test': {'Global': ['Global-Loss']}}
    cfg = make_cfg(cfg, '1_6_1_iid_fix_a1-e1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    cfg['world_size'] = world
    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=60)
    process_dataset(ds, cfg)
    torch.manual_seed(7)
    data_split, label_split = split_dataset(ds, 6, 'iid', cfg['classes_size'])
    torch.manual_seed(1)
    model = make_model(cfg)
    ctx = init_distributed(backend='gloo')
    runner = FedRunner(cfg, ds, data_split, label_split, model,
                       make_optimizer(model, cfg['lr'], cfg), dist_ctx=ctx)
    g = runner._round_generator(1)
    user_idx = sample_active_users(cfg, 1, generator=g)
    runner.federation.make_model_rate(generator=g)
    my_clients = list(range(rank, len(user_idx), world))
    local_parameters, param_idx = runner.federation.distribute(
        user_idx, resample=False, slots=my_clients)
    trained = dict(runner.trainer.train_clients(
        my_clients, user_idx, local_parameters,
        runner.federation.model_rate, runner._make_loader,
        label_split, cfg['lr']))
    # drop slot 3 (it lives on rank 3 % world)
    trained.pop(3, None)
    distributed_combine(runner.federation, trained, param_idx, user_idx, ctx)
    gp = {k: v.clone() for k, v in runner.federation.global_parameters.items()}

    # sequential oracle over the reporting clients only
    torch.manual_seed(1)
    model2 = make_model(dict(cfg, world_size=1))
    runner2 = FedRunner(dict(cfg, world_size=1), ds, data_split, label_split,
                        model2, make_optimizer(model2, cfg['lr'], cfg))
    local2, pidx2 = runner2.federation.distribute(user_idx, resample=False)
    runner2.federation.model_rate = list(runner.federation.model_rate)
    local2, pidx2 = runner2.federation.distribute(user_idx, resample=False)
    trained2 = dict(runner2.trainer.train_clients(
        list(range(len(user_idx))), user_idx, local2,
        runner2.federation.model_rate, runner2._make_loader,
        label_split, cfg['lr']))
    trained2.pop(3, None)
    slots2 = sorted(trained2.keys())
    tmp_d, cnt_d = runner2.federation.accumulate(trained2, pidx2, user_idx,
                                                 slots=slots2)
    runner2.federation.finalize(tmp_d, cnt_d)
    for k, v in runner2.federation.global_parameters.items():
        if v.is_floating_point():
            diff = (gp[k] - v).abs().max().item()
            assert diff < 1e-5, (rank, k, diff)
    torch.distributed.destroy_process_group()


def test_distributed_combine_missing_client_fallback():
    mp.spawn(_missing_client_worker, args=(2, 29581, ''), nprocs=2, join=True)


def _eval_sync_worker(rank, world, port, tmpdir):
    """Sharded evaluation (C3): per-user Local loop and Global set sharded
    across ranks; after logger.sync() every rank's means equal the
    single-rank oracle's."""
    os.environ['MASTER_ADDR'] = '127.0.0.1'
    os.environ['MASTER_PORT'] = str(port)
    torch.distributed.init_process_group('gloo', rank=rank, world_size=world)
    from heterofl_amd.config import default_config
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.logger import Logger
    from heterofl_amd.models import make_model
    from heterofl_amd.parallel import init_distributed
    from heterofl_amd.utils import process_dataset, make_optimizer

    cfg = default_config()
    cfg['device'] = 'cpu'
    cfg['engine'] = 'sequential'
    cfg['metric_name'] = {'train': {'Local': ['Local-Loss']},
                          'test': {'Local': ['Local-Loss', 'Local-Accuracy'],
                                   'Global': ['Global-Loss',
                                              'Global-Accuracy']}}
    cfg = make_cfg(cfg, '1_5_0.4_iid_fix_a1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    cfg['world_size'] = world
    torch.manual_seed(0)
    ds = fetch_dataset('MNIST', synthetic=True, synthetic_size=60)
    process_dataset(ds, cfg)
    torch.manual_seed(7)
    data_split, label_split = split_dataset(ds, 5, 'iid', cfg['classes_size'])
    torch.manual_seed(1)
    model = make_model(cfg)
    ctx = init_distributed(backend='gloo')
    logger = Logger(None)
    runner = FedRunner(cfg, ds, data_split, label_split, model,
                       make_optimizer(model, cfg['lr'], cfg),
                       logger=logger, dist_ctx=ctx)
    tm = runner.stats()
    runner.test(tm, 1)
    means = dict(logger.mean)

    # single-rank oracle with the same test model
    logger1 = Logger(None)
    runner1 = FedRunner(dict(cfg, world_size=1), ds, data_split, label_split,
                        make_model(cfg),
                        make_optimizer(model, cfg['lr'], cfg), logger=logger1)
    runner1.test(tm, 1)
    for k, v in logger1.mean.items():
        assert k in means, (rank, k)
        assert abs(means[k] - v) < 1e-6, (rank, k, means[k], v)
    torch.distributed.destroy_process_group()


def test_sharded_eval_sync_matches_sequential():
    mp.spawn(_eval_sync_worker, args=(2, 29591, ''), nprocs=2, join=True)
