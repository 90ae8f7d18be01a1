"""Shared entry-point machinery: CLI flags mirroring every cfg key, the
experiment loop, checkpointing with the reference's exact layout
(./output/model/{tag}_checkpoint.pt + {tag}_best.pt, reference:
src/train_classifier_fed.py:37-96).
"""
import argparse
import ast
import os
import shutil

import torch

from .config import cfg as global_cfg
from .control import process_control, CONTROL_FIELDS
from .data import fetch_dataset, split_dataset
from .fed import FedRunner
from .logger import Logger
from .metrics import Metric
from .models import make_model
from .utils import (save, process_dataset, make_optimizer, make_scheduler,
                    resume, model_tag_of)


def build_parser(cfg):
    parser = argparse.ArgumentParser(description='cfg')
    for k, v in cfg.items():
        if isinstance(v, (dict, list, bool)):
            parser.add_argument(f'--{k}', default=v, type=_literal)
        else:
            parser.add_argument(f'--{k}', default=v, type=type(v) if v is not None else str)
    parser.add_argument('--control_name', default=None, type=str)
    parser.add_argument('--synthetic', default=0, type=int,
                        help='use deterministic synthetic data (no-network envs)')
    return parser


def _literal(x):
    if isinstance(x, str):
        try:
            return ast.literal_eval(x)
        except (ValueError, SyntaxError):
            return x
    return x


def parse_args(argv=None, cfg=None):
    cfg = dict(global_cfg if cfg is None else cfg)
    parser = build_parser(cfg)
    args = vars(parser.parse_args(argv))
    for k in cfg:
        cfg[k] = args[k]
    if args.get('control_name') and args['control_name'] != 'None':
        cfg['control'] = dict(zip(CONTROL_FIELDS, args['control_name'].split('_')))
    cfg['control_name'] = '_'.join([str(cfg['control'][k]) for k in cfg['control']])
    cfg['synthetic'] = bool(args.get('synthetic', 0))
    return cfg


def run_fed_experiment(cfg, pivot_metric, pivot_sign, metric_name):
    """One full federated experiment per seed (reference:
    src/train_classifier_fed.py:48-96).  pivot_sign=+1 maximizes (accuracy),
    -1 minimizes (perplexity)."""
    cfg['pivot_metric'] = pivot_metric
    cfg['metric_name'] = metric_name
    process_control(cfg)
    _apply_round_cap(cfg)
    seeds = list(range(cfg['init_seed'], cfg['init_seed'] + cfg['num_experiments']))
    for seed in seeds:
        cfg['model_tag'] = model_tag_of(seed, cfg)
        cfg['pivot'] = -float('inf') * pivot_sign if pivot_sign > 0 else float('inf')
        if os.environ.get('RANK', '0') == '0':  # pre-init: env rank gates
            print('Experiment: {}'.format(cfg['model_tag']))
        _run_one(cfg, seed, pivot_sign)


def _synthetic_size(default=None):
    """HETEROFL_SYNTHETIC_SIZE shrinks synthetic datasets for smoke/CI runs
    of the CLI entries (None = the real dataset's size)."""
    v = os.environ.get('HETEROFL_SYNTHETIC_SIZE')
    return int(v) if v else default


def _apply_round_cap(cfg):
    """HETEROFL_MAX_ROUNDS caps the global-round count (smoke/CI runs of the
    real CLI scripts; the reference has no such knob — round counts come
    from the dataset tables, src/utils.py:150-212)."""
    cap = os.environ.get('HETEROFL_MAX_ROUNDS')
    if not cap:
        return
    cap = int(cap)
    if isinstance(cfg.get('num_epochs'), dict):
        cfg['num_epochs'] = dict(cfg['num_epochs'],
                                 **{'global': min(cfg['num_epochs']['global'], cap)})
    else:
        cfg['num_epochs'] = min(int(cfg['num_epochs']), cap)


def _run_one(cfg, seed, pivot_sign):
    # multi-GPU: under torchrun every rank runs this same function; the
    # round engine shards active clients / sBN batches / evaluation across
    # ranks and rank 0 owns all file writes (the reference is single-process,
    # src/train_classifier_fed.py:48-96 — client sharding over the node's
    # GPUs is the MI355X-native replacement for its sequential client loop)
    from .parallel import init_distributed
    ctx = init_distributed()
    is_main = ctx is None or ctx.is_main
    if ctx is not None:
        cfg['world_size'] = ctx.world_size
        if ctx.device.type == 'cuda':
            cfg['device'] = str(ctx.device)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed(seed)
    if cfg['device'].startswith('cuda') and not torch.cuda.is_available():
        print('[heterofl_amd] no GPU visible; falling back to cpu')
        cfg['device'] = 'cpu'
    dataset = fetch_dataset(cfg['data_name'], cfg['subset'],
                            synthetic=cfg.get('synthetic', False),
                            synthetic_size=_synthetic_size())
    process_dataset(dataset, cfg)
    model = make_model(cfg, model_rate=cfg['global_model_rate']).to(cfg['device'])
    optimizer = make_optimizer(model, cfg['lr'], cfg)
    scheduler = make_scheduler(optimizer, cfg)
    if cfg['resume_mode'] == 1:
        last_epoch, data_split, label_split, model, optimizer, scheduler, logger = resume(
            model, cfg['model_tag'], optimizer, scheduler)
        if not is_main:
            logger.log_path = None
    elif cfg['resume_mode'] == 2:
        last_epoch = 1
        _, data_split, label_split, model, _, _, _ = resume(model, cfg['model_tag'])
        logger = Logger(os.path.join('output', 'runs', cfg['model_tag'])
                        if is_main else None)
    else:
        last_epoch = 1
        data_split, label_split = None, None
        logger = Logger(os.path.join('output', 'runs',
                                     'train_{}'.format(cfg['model_tag']))
                        if is_main else None)
    if data_split is None:
        # identical on every rank: all ranks seeded the same above
        data_split, label_split = split_dataset(
            dataset, cfg['num_users'], cfg['data_split_mode'],
            classes_size=cfg.get('classes_size'))
    runner = FedRunner(cfg, dataset, data_split, label_split, model, optimizer,
                       logger=logger, dist_ctx=ctx)
    # HETEROFL_EVAL_EVERY=k evaluates every k-th round (+ the final round)
    # instead of the reference's every-round evaluation — a cost knob for
    # long runs; k=1 (default) is reference behavior
    eval_every = int(os.environ.get('HETEROFL_EVAL_EVERY', '1'))
    total_epochs = cfg['num_epochs']['global']
    for epoch in range(last_epoch, total_epochs + 1):
        logger.safe(True)
        runner.train_round(epoch)
        do_eval = (epoch % eval_every == 0) or epoch == total_epochs
        if do_eval:
            test_model = runner.stats()
            runner.test(test_model, epoch)
        if cfg['scheduler_name'] == 'ReduceLROnPlateau':
            # post-sync means are identical on every rank
            scheduler.step(metrics=logger.mean['train/{}'.format(cfg['pivot_metric'])])
        else:
            # the server-side optimizer never steps by design (it only
            # carries the lr the scheduler decays; local optimizers are
            # fresh per client per round — reference
            # src/train_classifier_fed.py:79-82,195) so torch's
            # "step() before lr_scheduler.step()" warning does not apply
            import warnings
            with warnings.catch_warnings():
                warnings.filterwarnings(
                    'ignore', message='.*lr_scheduler.step.*')
                scheduler.step()
        logger.safe(False)
        cur = logger.mean['test/{}'.format(cfg['pivot_metric'])]
        better = do_eval and (cur > cfg['pivot'] if pivot_sign > 0
                              else cur < cfg['pivot'])
        if better:
            cfg['pivot'] = cur
        if is_main:
            save_result = {
                'cfg': cfg, 'epoch': epoch + 1, 'data_split': data_split,
                'label_split': label_split, 'model_dict': model.state_dict(),
                'optimizer_dict': optimizer.state_dict(),
                'scheduler_dict': scheduler.state_dict(), 'logger': logger}
            save(save_result, './output/model/{}_checkpoint.pt'.format(cfg['model_tag']))
            if better:
                shutil.copy('./output/model/{}_checkpoint.pt'.format(cfg['model_tag']),
                            './output/model/{}_best.pt'.format(cfg['model_tag']))
        logger.reset()
    logger.safe(False)


def _sync_bn_running_stats(model):
    """Batch-count-weighted all-reduce of cumulative BN running stats after a
    rank-sharded stats pass (C2; momentum=None BN keeps running stats as the
    mean over batches, and num_batches_tracked counts them)."""
    import torch.distributed as dist
    parts, weights = [], []
    for mod in model.modules():
        if getattr(mod, 'running_mean', None) is not None:
            w = float(mod.num_batches_tracked.item())
            parts.append(mod.running_mean.reshape(-1).float() * w)
            parts.append(mod.running_var.reshape(-1).float() * w)
            weights.append(w)
    if not parts:
        return
    dev = parts[0].device
    flat = torch.cat(parts)
    wt = torch.tensor(weights, dtype=torch.float32, device=dev)
    dist.all_reduce(flat)
    dist.all_reduce(wt)
    off = i = 0
    for mod in model.modules():
        if getattr(mod, 'running_mean', None) is not None:
            for buf in (mod.running_mean, mod.running_var):
                n = buf.numel()
                buf.copy_((flat[off:off + n] / wt[i].clamp(min=1)).view_as(buf))
                off += n
            i += 1


def run_centralized_experiment(cfg, pivot_metric, pivot_sign, metric_name):
    """Non-federated baseline: train ONE model at the global rate on the full
    dataset (reference: src/train_classifier.py:48-133).  Where the reference
    uses torch.nn.DataParallel for multi-GPU (src/train_classifier.py:65-66),
    the MI355X-native path is one process per GPU with DDP over RCCL (run
    under torchrun); single-process runs need no wrapper."""
    from .data import make_data_loader
    from .metrics import Metric
    from .utils import collate, to_device
    cfg['pivot_metric'] = pivot_metric
    cfg['metric_name'] = metric_name
    process_control(cfg)
    _apply_round_cap(cfg)
    from .parallel import init_distributed
    ctx = init_distributed()
    is_main = ctx is None or ctx.is_main
    rank, world = (0, 1) if ctx is None else (ctx.rank, ctx.world_size)
    seeds = list(range(cfg['init_seed'], cfg['init_seed'] + cfg['num_experiments']))
    for seed in seeds:
        cfg['model_tag'] = model_tag_of(seed, cfg)
        cfg['pivot'] = -float('inf') * pivot_sign if pivot_sign > 0 else float('inf')
        if is_main:
            print('Experiment: {}'.format(cfg['model_tag']))
        torch.manual_seed(seed)
        if ctx is not None and ctx.device.type == 'cuda':
            cfg['device'] = str(ctx.device)
        if cfg['device'].startswith('cuda') and not torch.cuda.is_available():
            cfg['device'] = 'cpu'
        dataset = fetch_dataset(cfg['data_name'], cfg['subset'],
                                synthetic=cfg.get('synthetic', False),
                                synthetic_size=_synthetic_size())
        process_dataset(dataset, cfg)
        model = make_model(cfg, model_rate=cfg['global_model_rate']).to(cfg['device'])
        optimizer = make_optimizer(model, cfg['lr'], cfg)
        scheduler = make_scheduler(optimizer, cfg)
        ddp_model = model
        if ctx is not None and world > 1:
            ddp_model = torch.nn.parallel.DistributedDataParallel(
                model, device_ids=[ctx.local_rank] if ctx.device.type == 'cuda'
                else None)
        if cfg['resume_mode'] == 1:
            last_epoch, _, _, model, optimizer, scheduler, logger = resume(
                model, cfg['model_tag'], optimizer, scheduler)
            if not is_main:
                logger.log_path = None
        else:
            last_epoch = 1
            logger = Logger(os.path.join('output', 'runs',
                                         'train_{}'.format(cfg['model_tag']))
                            if is_main else None)
        metric = Metric()
        is_lm = cfg['model_name'] == 'transformer'
        num_epochs = cfg['num_epochs']
        if isinstance(num_epochs, dict):
            num_epochs = num_epochs.get('global', 200)
        # per-rank train shard (rank-strided): DDP all-reduces gradients, so
        # each rank must see a DISJOINT slice of the data or world_size
        # ranks do world_size-times duplicated work
        from .data import SplitDataset
        train_ds = dataset['train']
        if world > 1 and not is_lm:
            train_ds = SplitDataset(
                dataset['train'], list(range(rank, len(dataset['train']), world)))
        for epoch in range(last_epoch, num_epochs + 1):
            logger.safe(True)
            ddp_model.train(True)
            if is_lm:
                from .data import BatchDataset
                train_iter = BatchDataset(dataset['train'], cfg['bptt'])
                batches = (train_iter[i]
                           for i in range(rank, len(train_iter), world))
            else:
                loader = make_data_loader({'train': train_ds}, cfg)['train']
                batches = (collate(b) for b in loader)
            for input in batches:
                input = to_device(input, cfg['device'])
                optimizer.zero_grad()
                output = ddp_model(input)
                output['loss'].backward()
                torch.nn.utils.clip_grad_norm_(model.parameters(), 1)
                optimizer.step()
                n = input['label'].size(0)
                ev = metric.evaluate(cfg['metric_name']['train'], input, output)
                logger.append(ev, 'train', n=n)
            # sBN stats pass + evaluation (vision; reference
            # src/train_classifier.py:123-133), sharded across ranks like
            # the fed path and merged by logger.sync()
            with torch.no_grad():
                if not is_lm:
                    test_model = make_model(cfg, model_rate=cfg['global_model_rate'],
                                            track=True).to(cfg['device'])
                    test_model.load_state_dict(model.state_dict(), strict=False)
                    test_model.train(True)
                    stats_ds = dataset['train'] if world == 1 else SplitDataset(
                        dataset['train'],
                        list(range(rank, len(dataset['train']), world)))
                    loader = make_data_loader({'train': stats_ds}, cfg)['train']
                    for input in loader:
                        input = collate(input)
                        test_model(to_device(input, cfg['device']))
                    if ctx is not None and world > 1:
                        _sync_bn_running_stats(test_model)
                else:
                    test_model = model
                test_model.train(False)
                if is_lm:
                    from .data import BatchDataset
                    ds = BatchDataset(dataset['test'], cfg['bptt'])
                    test_batches = (ds[i] for i in range(rank, len(ds), world))
                else:
                    test_ds = dataset['test'] if world == 1 else SplitDataset(
                        dataset['test'],
                        list(range(rank, len(dataset['test']), world)))
                    loader = make_data_loader({'test': test_ds}, cfg)['test']
                    test_batches = (collate(b) for b in loader)
                for input in test_batches:
                    input = to_device(input, cfg['device'])
                    output = test_model(input)
                    ev = metric.evaluate(cfg['metric_name']['test'], input, output)
                    logger.append(ev, 'test', n=input['label'].size(0))
            if ctx is not None:
                logger.sync()
            if is_main:
                logger.write('test', cfg['metric_name']['test'])
            scheduler.step()
            logger.safe(False)
            cur = logger.mean['test/{}'.format(cfg['pivot_metric'])]
            better = cur > cfg['pivot'] if pivot_sign > 0 else cur < cfg['pivot']
            if better:
                cfg['pivot'] = cur
            if is_main:
                save_result = {
                    'cfg': cfg, 'epoch': epoch + 1, 'data_split': None,
                    'label_split': None, 'model_dict': model.state_dict(),
                    'optimizer_dict': optimizer.state_dict(),
                    'scheduler_dict': scheduler.state_dict(), 'logger': logger}
                save(save_result,
                     './output/model/{}_checkpoint.pt'.format(cfg['model_tag']))
                if better:
                    shutil.copy('./output/model/{}_checkpoint.pt'.format(cfg['model_tag']),
                                './output/model/{}_best.pt'.format(cfg['model_tag']))
            logger.reset()
        logger.safe(False)


def run_centralized_eval(cfg, metric_name):
    """Centralized (non-fed) evaluation entry (reference:
    src/test_classifier.py:41-91 / src/test_transformer.py): load
    {tag}_best.pt, re-run the sBN stats pass over the train set (vision
    only), evaluate the test set, save ./output/result/{tag}.pt."""
    from .data import make_data_loader, BatchDataset
    from .metrics import Metric
    from .utils import collate, to_device
    cfg['metric_name'] = metric_name
    process_control(cfg)
    seeds = list(range(cfg['init_seed'], cfg['init_seed'] + cfg['num_experiments']))
    for seed in seeds:
        cfg['model_tag'] = model_tag_of(seed, cfg)
        print('Eval: {}'.format(cfg['model_tag']))
        torch.manual_seed(seed)
        if cfg['device'].startswith('cuda') and not torch.cuda.is_available():
            cfg['device'] = 'cpu'
        dataset = fetch_dataset(cfg['data_name'], cfg['subset'],
                                synthetic=cfg.get('synthetic', False),
                                synthetic_size=_synthetic_size())
        process_dataset(dataset, cfg)
        model = make_model(cfg, model_rate=cfg['global_model_rate']).to(cfg['device'])
        last_epoch, _, _, model, _, _, train_logger = resume(
            model, cfg['model_tag'], load_tag='best', strict=False)
        logger = Logger(os.path.join('output', 'runs',
                                     'test_{}'.format(cfg['model_tag'])))
        metric = Metric()
        is_lm = cfg['model_name'] == 'transformer'
        logger.safe(True)
        with torch.no_grad():
            if not is_lm:
                test_model = make_model(cfg, model_rate=cfg['global_model_rate'],
                                        track=True).to(cfg['device'])
                test_model.load_state_dict(model.state_dict(), strict=False)
                test_model.train(True)
                loader = make_data_loader({'train': dataset['train']}, cfg)['train']
                for input in loader:
                    test_model(to_device(collate(input), cfg['device']))
            else:
                test_model = model
            test_model.train(False)
            if is_lm:
                ds = BatchDataset(dataset['test'], cfg['bptt'])
                test_batches = (ds[i] for i in range(len(ds)))
            else:
                loader = make_data_loader({'test': dataset['test']}, cfg)['test']
                test_batches = (collate(b) for b in loader)
            for input in test_batches:
                input = to_device(input, cfg['device'])
                output = test_model(input)
                ev = metric.evaluate(cfg['metric_name']['test'], input, output)
                logger.append(ev, 'test', n=input['label'].size(0))
        logger.write('test', cfg['metric_name']['test'])
        logger.safe(False)
        result = {'cfg': cfg, 'epoch': last_epoch,
                  'logger': {'train': train_logger, 'test': logger}}
        save(result, './output/result/{}.pt'.format(cfg['model_tag']))


def run_fed_eval(cfg, metric_name, result_key='test'):
    """Evaluation entry (reference: src/test_classifier_fed.py:41-60): load
    {tag}_best.pt, re-run sBN stats (vision), evaluate, save
    ./output/result/{tag}.pt.  Under torchrun the stats pass and evaluation
    shard across ranks (FedRunner dist_ctx) and rank 0 writes the result."""
    from .parallel import init_distributed
    ctx = init_distributed()
    is_main = ctx is None or ctx.is_main
    cfg['metric_name'] = metric_name
    process_control(cfg)
    seeds = list(range(cfg['init_seed'], cfg['init_seed'] + cfg['num_experiments']))
    for seed in seeds:
        cfg['model_tag'] = model_tag_of(seed, cfg)
        if is_main:
            print('Eval: {}'.format(cfg['model_tag']))
        torch.manual_seed(seed)
        if ctx is not None:
            cfg['world_size'] = ctx.world_size
            if ctx.device.type == 'cuda':
                cfg['device'] = str(ctx.device)
        dataset = fetch_dataset(cfg['data_name'], cfg['subset'],
                                synthetic=cfg.get('synthetic', False),
                                synthetic_size=_synthetic_size())
        process_dataset(dataset, cfg)
        if cfg['device'].startswith('cuda') and not torch.cuda.is_available():
            cfg['device'] = 'cpu'
        model = make_model(cfg, model_rate=cfg['global_model_rate']).to(cfg['device'])
        last_epoch, data_split, label_split, model, _, _, train_logger = resume(
            model, cfg['model_tag'], load_tag='best', strict=False)
        if data_split is None:
            data_split, label_split = split_dataset(
                dataset, cfg['num_users'], cfg['data_split_mode'],
                classes_size=cfg.get('classes_size'))
        logger = Logger(os.path.join('output', 'runs',
                                     'test_{}'.format(cfg['model_tag']))
                        if is_main else None)
        runner = FedRunner(cfg, dataset, data_split, label_split, model,
                           make_optimizer(model, cfg['lr'], cfg), logger=logger,
                           dist_ctx=ctx)
        logger.safe(True)
        test_model = runner.stats()
        runner.test(test_model, last_epoch - 1)
        logger.safe(False)
        if is_main:
            result = {'cfg': cfg, 'epoch': last_epoch,
                      'logger': {'train': train_logger, 'test': logger}}
            save(result, './output/result/{}.pt'.format(cfg['model_tag']))
