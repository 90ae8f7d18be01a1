"""Federated round orchestration.

Drives the reference round structure (src/train_classifier_fed.py:74-94):
sample active clients -> distribute slices -> local training -> combine ->
sBN statistics pass -> evaluation -> checkpoint.  The local-training engine
is pluggable: 'sequential' (reference-faithful oracle) or 'batched'
(MI355X fast path, clients of one rate trained in one grouped model).

Multi-rank operation (one process per GPU over RCCL): active clients are
sharded across ranks; every rank samples identical user_idx and dynamic
rates from a round-seeded generator, trains its shard, and combine runs as
a padded all-reduce (see parallel/dist.py).
"""
import math
import os
import time

import numpy as np
import torch

_TIMING = os.environ.get('HETEROFL_TIMING') == '1'


class _phase_timer:
    acc = {}

    def __init__(self, name):
        self.name = name

    def __enter__(self):
        if _TIMING:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            self.t0 = time.perf_counter()
        return self

    def __exit__(self, *a):
        if _TIMING:
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            dt = time.perf_counter() - self.t0
            _phase_timer.acc[self.name] = _phase_timer.acc.get(self.name, 0.0) + dt

    @classmethod
    def report(cls):
        return {k: round(v, 4) for k, v in sorted(cls.acc.items())}

from ..data import SplitDataset, BatchDataset, make_data_loader
from ..metrics import Metric
from ..models import make_model
from ..utils import collate, to_device
from .federation import Federation
from .sequential import SequentialClientTrainer


def sample_active_users(cfg, epoch=None, generator=None):
    """ceil(frac*num_users) clients by randperm
    (reference: src/train_classifier_fed.py:172-181).  With a generator the
    sampling is reproducible across ranks and GPU counts."""
    num_active = int(np.ceil(cfg['frac'] * cfg['num_users']))
    perm = torch.randperm(cfg['num_users'], generator=generator)
    return perm[:num_active].tolist()


class FedRunner:
    def __init__(self, cfg, dataset, data_split, label_split, model, optimizer,
                 logger=None, dist_ctx=None):
        self.cfg = cfg
        self.dataset = dataset
        self.data_split = data_split
        self.label_split = label_split
        self.global_model = model
        self.optimizer = optimizer
        self.logger = logger
        self.dist_ctx = dist_ctx  # parallel.DistContext or None
        self.federation = Federation(model.state_dict(), cfg['model_rate'],
                                     label_split, cfg)
        self.is_lm = cfg['model_name'] == 'transformer'
        use_batched = cfg.get('engine', 'sequential') == 'batched'
        # The grouped BBottleneck engine passed its GPU numerics suite
        # (GPUTEST_r01) so Bottleneck resnets now use the batched engine by
        # default; HETEROFL_BATCHED_BOTTLENECK=0 forces the sequential oracle
        if use_batched and cfg['model_name'] in ('resnet50', 'resnet101',
                                                 'resnet152') and \
                os.environ.get('HETEROFL_BATCHED_BOTTLENECK', '1') == '0':
            use_batched = False
        if use_batched and self.is_lm:
            from .batched_lm_trainer import BatchedLMClientTrainer
            self.trainer = BatchedLMClientTrainer(cfg)
            self.trainer.set_data(dataset, data_split)
        elif use_batched:
            from .batched import BatchedClientTrainer
            self.trainer = BatchedClientTrainer(cfg)
            self.trainer.set_data(dataset, data_split)
        else:
            self.trainer = SequentialClientTrainer(cfg)
        self._round_gen = None

    # ------------------------------------------------------------------ round
    def _round_generator(self, epoch):
        """Seeded per round so all ranks agree on sampling without
        communication (SURVEY §7 reproducibility contract)."""
        seed = int(self.cfg.get('round_seed_base', self.cfg.get('init_seed', 0)))
        g = torch.Generator()
        g.manual_seed((seed * 1000003 + epoch) % (2 ** 63 - 1))
        return g

    def _make_loader(self, user):
        cfg = self.cfg
        if self.is_lm:
            # LM clients own rows of the batchified token matrix
            # (reference: src/train_transformer_fed.py:158-162)
            return BatchDataset(SplitDataset(self.dataset['train'],
                                             self.data_split['train'][user]),
                                cfg['bptt'])
        loader = make_data_loader(
            {'train': SplitDataset(self.dataset['train'],
                                   self.data_split['train'][user])}, cfg)['train']
        return loader

    def train_round(self, epoch):
        cfg = self.cfg
        self.global_model.load_state_dict(self.federation.global_parameters)
        self.global_model.train(True)
        lr = self.optimizer.param_groups[0]['lr']

        g = self._round_generator(epoch)
        if cfg['world_size'] > 1 or self.dist_ctx is not None:
            user_idx = sample_active_users(cfg, epoch, generator=g)
            self.federation.make_model_rate(generator=g)
            resample = False
        else:
            user_idx = sample_active_users(cfg, epoch)
            resample = True
        # shard clients across ranks (identity shard when single rank)
        rank, world = (0, 1) if self.dist_ctx is None else \
            (self.dist_ctx.rank, self.dist_ctx.world_size)
        my_clients = list(range(rank, len(user_idx), world))
        with _phase_timer('1.distribute'):
            local_parameters, param_idx = self.federation.distribute(
                user_idx, resample=resample,
                slots=my_clients if world > 1 else None)

        with _phase_timer('2.local_train'):
            trained = dict(self.trainer.train_clients(
                my_clients, user_idx, local_parameters,
                self.federation.model_rate, self._make_loader,
                self.label_split, lr, self.logger))

        with _phase_timer('3.combine'):
            if self.dist_ctx is None:
                # combine averages whoever reported — a client that failed
                # mid-round simply contributes nothing (reference tolerance
                # semantics, src/fed.py:180-298)
                tmp_d, cnt_d = self.federation.accumulate(
                    trained, param_idx, user_idx,
                    slots=sorted(trained.keys()))
                self.federation.finalize(tmp_d, cnt_d)
            else:
                from ..parallel.dist import distributed_combine
                distributed_combine(self.federation, trained, param_idx,
                                    user_idx, self.dist_ctx)
        self.global_model.load_state_dict(self.federation.global_parameters)
        if self.logger is not None:
            # per-round console line (the engine-level equivalent of the
            # reference's per-batch Train Epoch/ETA prints,
            # src/train_classifier_fed.py:108-119 — the graphed inner loop
            # has no per-batch host involvement to log from)
            if self.dist_ctx is not None:
                self.logger.sync()  # merge each rank's clients' train metrics
            total = cfg['num_epochs']['global'] if \
                isinstance(cfg['num_epochs'], dict) else cfg['num_epochs']
            info = {'info': ['Model: {}'.format(cfg.get('model_tag', '')),
                             'Train Epoch: {}({:.0f}%)'.format(
                                 epoch, 100.0 * epoch / max(total, 1)),
                             'Learning rate: {:.4g}'.format(lr)]}
            self.logger.append(info, 'train', mean=False)
            if self.dist_ctx is None or self.dist_ctx.is_main:
                flat = [m for group in cfg['metric_name']['train'].values()
                        for m in group]
                self.logger.write('train', flat)
        return user_idx

    # ------------------------------------------------------------------ stats
    def stats(self):
        """sBN statistics pass: rebuild the global model with tracked running
        stats and run the full train set in train mode under no_grad
        (reference: src/train_classifier_fed.py:127-138).

        Multi-rank (C2 of SURVEY §2b): the train set is sharded across ranks
        and the cumulative running stats are combined by a batch-count
        weighted all-reduce over RCCL — momentum=None BN keeps the running
        stats as the mean over batches, so the weighted mean over ranks
        reproduces the sequential pass (equal-size batches)."""
        cfg = self.cfg
        if self.is_lm:
            return self.global_model  # LM path has no sBN pass (reference:
            # src/train_transformer_fed.py:77)
        with torch.no_grad():
            test_model = make_model(cfg, model_rate=cfg['global_model_rate'],
                                    track=True).to(cfg['device'])
            test_model.load_state_dict(self.global_model.state_dict(), strict=False)
            if self._native_stats_ok():
                self._native_stats(test_model)
                return test_model
            test_model.train(True)
            # BN momentum=None is a cumulative average over batches, which is
            # batch-size invariant for equal-size batches — so the stats pass
            # can use large batches on GPU instead of the train batch of 10
            # (5000 launches -> ~100 per round)
            scfg = dict(cfg)
            scfg['shuffle'] = dict(cfg['shuffle'])
            scfg['shuffle']['train'] = False  # stats don't need a shuffle;
            # unshuffled batches make the pass deterministic and identical
            # across GPU counts
            if torch.cuda.is_available():
                scfg['batch_size'] = dict(cfg['batch_size'])
                scfg['batch_size']['train'] = 500
            train = self.dataset['train']
            n_batches = 0
            if self.dist_ctx is not None and self.dist_ctx.world_size > 1:
                rank, world = self.dist_ctx.rank, self.dist_ctx.world_size
                bs = scfg['batch_size']['train']
                # shard whole batches (incl. the partial tail batch) across
                # ranks: cumulative-BN (momentum=None) weighs every batch
                # equally, so giving the tail to exactly one rank and
                # weighting ranks by batch count reproduces the single-rank
                # pass exactly — tail included
                total_b = (len(train) + bs - 1) // bs
                my_b = list(range(rank, total_b, world))
                idx = [i for b in my_b
                       for i in range(b * bs, min((b + 1) * bs, len(train)))]
                train = SplitDataset(train, idx)
            loader = make_data_loader({'train': train}, scfg)['train']
            for input in loader:
                input = collate(input)
                input = to_device(input, cfg['device'])
                test_model(input)
                n_batches += 1
            if self.dist_ctx is not None and self.dist_ctx.world_size > 1:
                import torch.distributed as dist
                bufs = []
                for mod in test_model.modules():
                    if hasattr(mod, 'running_mean') and                             mod.running_mean is not None:
                        bufs.append(mod.running_mean)
                        bufs.append(mod.running_var)
                if bufs:
                    w = float(n_batches)
                    flat = torch.cat([b.reshape(-1).float() * w
                                      for b in bufs])
                    cnt = torch.tensor([w], device=flat.device)
                    dist.all_reduce(flat)
                    dist.all_reduce(cnt)
                    flat /= cnt
                    off = 0
                    for b in bufs:
                        n = b.numel()
                        b.copy_(flat[off:off + n].view_as(b))
                        off += n
        return test_model

    # ------------------------------------------------------ native sBN pass
    def _native_stats_ok(self):
        """Route the sBN statistics pass through the batched native engine
        (MFMA convs + fused sBN kernels) instead of the eager per-client
        model.  bn only (other norms track no running stats) and vision only;
        HETEROFL_NATIVE_STATS=0 forces the eager pass."""
        from .batched import BatchedClientTrainer
        return (self.cfg['norm'] == 'bn' and not self.is_lm
                and isinstance(self.trainer, BatchedClientTrainer)
                and os.environ.get('HETEROFL_NATIVE_STATS', '1') == '1')

    def _native_stats(self, test_model):
        """Run the full train set (normalized, augmented, unshuffled,
        batch 500 on GPU) through a 1-group batched model at the global rate
        and harvest each fused BN kernel's per-batch mean/var into the
        cumulative (momentum=None) running stats of test_model
        (reference semantics: src/train_classifier_fed.py:127-138 with
        BatchNorm2d(momentum=None), src/models/resnet.py:16-17).

        Multi-rank: whole batches (incl. the partial tail) shard across
        ranks; per-module stats merge with a batch-count-weighted
        all-reduce (C2)."""
        from .batched import pack_states, BNormReLU
        cfg = self.cfg
        device = torch.device(cfg['device'])
        trainer = self.trainer
        bmodel = trainer._batched_model(cfg['global_model_rate'], 1)
        pack_states(bmodel, [dict(self.federation.global_parameters)])
        bmodel.train(True)

        state = {}  # module -> [mean, unbiased_var, n_batches]

        def sink(mod, mean, invstd, x):
            n, _, h, w = x.shape
            m = n * h * w
            mu = mean.float()
            biased = invstd.float().pow(-2) - 1e-5
            unb = biased * (m / max(m - 1, 1))
            s = state.get(mod)
            if s is None:
                state[mod] = [mu.clone(), unb.clone(), 1]
            else:
                s[2] += 1
                k = s[2]
                s[0] += (mu - s[0]) / k
                s[1] += (unb - s[1]) / k

        bn_mods = [(name, m) for name, m in bmodel.named_modules()
                   if isinstance(m, BNormReLU) and m.norm == 'bn']
        for _, m in bn_mods:
            m.stats_sink = sink
        try:
            train = self.dataset['train']
            img = getattr(self, '_stats_img', None)
            if img is None:
                img = train.img
                if img.dim() == 3:
                    img = img.unsqueeze(-1)
                img = img.to(device)
                self._stats_img = img
            n_total = img.size(0)
            # cumulative BN is batch-size invariant for equal-size batches,
            # so the GPU pass uses large batches (fewer launches, fuller
            # kernels); HETEROFL_STATS_BS tunes it
            bs = int(os.environ.get('HETEROFL_STATS_BS', '2500')) \
                if device.type == 'cuda' else cfg['batch_size']['train']
            total_b = (n_total + bs - 1) // bs
            rank, world = (0, 1) if self.dist_ctx is None else \
                (self.dist_ctx.rank, self.dist_ctx.world_size)
            # the reference's stats pass iterates the TRAIN loader, i.e.
            # with train-time augmentation active (train_classifier_fed.py:
            # 127-138 over the train dataset's transforms).  Activations run
            # in the engine's compute dtype: fp32 MFMA is 1/16 the bf16 rate,
            # and the model trains in bf16 anyway (BN statistics accumulate
            # in fp32 inside the kernel either way).
            amp = getattr(trainer, '_amp', False)
            for b in range(rank, total_b, world):
                batch = img[b * bs:min((b + 1) * bs, n_total)]
                x = trainer.augment(batch, train=True)
                if amp:
                    x = x.to(torch.bfloat16)
                bmodel(x)
        finally:
            for _, m in bn_mods:
                m.stats_sink = None
        if self.dist_ctx is not None and self.dist_ctx.world_size > 1:
            import torch.distributed as dist
            parts, weights = [], []
            for _, m in bn_mods:
                if m not in state:  # rank had zero batches (tiny dataset)
                    z = torch.zeros(m.weight.numel(), device=device)
                    state[m] = [z, z.clone(), 0]
                mu, var, k = state[m]
                parts.append(mu * k)
                parts.append(var * k)
                weights.append(float(k))
            flat = torch.cat([p.reshape(-1) for p in parts])
            wt = torch.tensor(weights, dtype=torch.float32, device=flat.device)
            dist.all_reduce(flat)
            dist.all_reduce(wt)
            off = 0
            for i, (_, m) in enumerate(bn_mods):
                mu, var, k = state[m]
                n = mu.numel()
                state[m] = [flat[off:off + n] / wt[i].clamp(min=1),
                            flat[off + n:off + 2 * n] / wt[i].clamp(min=1),
                            int(wt[i].item())]
                off += 2 * n
        named = dict(test_model.named_modules())
        for name, m in bn_mods:
            mu, var, k = state[m]
            tgt = named[name]
            tgt.running_mean.copy_(mu.to(tgt.running_mean.dtype))
            tgt.running_var.copy_(var.to(tgt.running_var.dtype))
            if tgt.num_batches_tracked is not None:
                tgt.num_batches_tracked.fill_(k)
        test_model.train(False)

    # -------------------------------------------------------- fast GPU eval
    def _fast_vision_eval(self, test_model, metric, logger, cfg, rank, world):
        """Loader-free evaluation: stage the raw uint8 test set on device
        once, run ONE pass of large-batch forwards for the UNMASKED logits,
        then derive every user's Local (label-split-masked, reference:
        src/models/resnet.py:152-157) and the Global metrics from those
        logits.  Same math as the per-user loader loop (the mask only
        rewrites logits), ~100x fewer host round-trips.  Users and samples
        shard across ranks exactly like the loader path."""
        from ..models.functional import masked_cross_entropy
        from .batched import DeviceAugment
        import torch.nn.functional as F
        ds = self.dataset['test']
        device = torch.device(cfg['device'])
        staged = getattr(self, '_eval_staged', None)
        if staged is None:
            img = ds.img
            if img.dim() == 3:
                img = img.unsqueeze(-1)
            aug = DeviceAugment(cfg['data_name'], device)
            labels = torch.tensor(ds.target, device=device)
            staged = (img.to(device), labels, aug)
            self._eval_staged = staged
        img, labels, aug = staged
        n = img.size(0)
        bs = 2048
        scores = []
        for b0 in range(0, n, bs):
            x = aug(img[b0:b0 + bs], train=False)
            out = test_model({'img': x, 'label': labels[b0:b0 + bs]})
            scores.append(out['score'].float())
        scores = torch.cat(scores)
        names = cfg['metric_name']['test']
        # Local metrics: this rank's share of users
        for m in range(rank, cfg['num_users'], world):
            idx = torch.as_tensor(self.data_split['test'][m],
                                  dtype=torch.long, device=device)
            if idx.numel() == 0:
                continue
            y = labels[idx]
            ls = torch.tensor(self.label_split[m], device=device)
            sc = scores[idx]
            if cfg['mask']:
                sc, loss = masked_cross_entropy(sc, y, ls,
                                                cfg['classes_size'])
            else:
                loss = F.cross_entropy(sc, y)
            ev = metric.evaluate(names['Local'],
                                 {'label': y, 'label_split': ls},
                                 {'score': sc, 'loss': loss})
            if logger:
                logger.append(ev, 'test', idx.numel())
        # Global metrics: this rank's share of samples, unmasked logits
        gidx = torch.arange(rank, n, world, device=device)
        gy = labels[gidx]
        gs = scores[gidx]
        ev = metric.evaluate(names['Global'], {'label': gy},
                             {'score': gs,
                              'loss': F.cross_entropy(gs, gy)})
        if logger:
            logger.append(ev, 'test', gidx.numel())

    # ------------------------------------------------------------------- test
    def test(self, test_model, epoch):
        """Per-client Local metrics + Global metrics
        (reference: src/train_classifier_fed.py:141-169).  On GPU the eval
        batch is enlarged (metric means are batch-size invariant under the
        logger's count weighting) so evaluation is not launch-bound.

        Multi-rank (C3 of SURVEY §2b): the per-user Local loop is sharded by
        user and the Global set by sample across ranks; logger.sync() then
        merges the shards' weighted sums so every rank sees the sequential
        run's exact means (weighted means are additive over disjoint
        shards)."""
        cfg = self.cfg
        if torch.cuda.is_available() and not self.is_lm:
            cfg = dict(cfg)
            cfg['batch_size'] = dict(cfg['batch_size'])
            cfg['batch_size']['test'] = max(cfg['batch_size']['test'], 500)
        rank, world = (0, 1) if self.dist_ctx is None else \
            (self.dist_ctx.rank, self.dist_ctx.world_size)
        metric = Metric()
        logger = self.logger
        with torch.no_grad():
            test_model.train(False)
            if not self.is_lm and torch.cuda.is_available() and \
                    os.environ.get('HETEROFL_FAST_EVAL', '1') == '1' and \
                    hasattr(self.dataset['test'], 'img'):
                self._fast_vision_eval(test_model, metric, logger, cfg,
                                       rank, world)
            elif not self.is_lm:
                for m in range(rank, cfg['num_users'], world):
                    loader = make_data_loader(
                        {'test': SplitDataset(self.dataset['test'],
                                              self.data_split['test'][m])}, cfg)['test']
                    for input in loader:
                        input = collate(input)
                        input_size = input['label'].size(0)
                        input['label_split'] = torch.tensor(self.label_split[m])
                        input = to_device(input, cfg['device'])
                        output = test_model(input)
                        ev = metric.evaluate(cfg['metric_name']['test']['Local'],
                                             input, output)
                        if logger:
                            logger.append(ev, 'test', input_size)
                global_test = self.dataset['test']
                if world > 1:
                    global_test = SplitDataset(
                        global_test, list(range(rank, len(global_test), world)))
                loader = make_data_loader({'test': global_test}, cfg)['test']
                for input in loader:
                    input = collate(input)
                    input_size = input['label'].size(0)
                    input = to_device(input, cfg['device'])
                    output = test_model(input)
                    ev = metric.evaluate(cfg['metric_name']['test']['Global'],
                                         input, output)
                    if logger:
                        logger.append(ev, 'test', input_size)
            else:
                ds = BatchDataset(self.dataset['test'], cfg['bptt'])
                for i in range(rank, len(ds), world):
                    input = ds[i]
                    input_size = input['label'].size(0)
                    input = to_device(input, cfg['device'])
                    output = test_model(input)
                    ev = metric.evaluate(cfg['metric_name']['test']['Global'],
                                         input, output)
                    if logger:
                        logger.append(ev, 'test', input_size)
            if logger:
                if self.dist_ctx is not None:
                    logger.sync()
                info = {'info': ['Test Epoch: {}({:.0f}%)'.format(epoch, 100.)]}
                logger.append(info, 'test', mean=False)
                if self.dist_ctx is None or self.dist_ctx.is_main:
                    names = cfg['metric_name']['test']
                    flat = sum(names.values(), []) if isinstance(names, dict) \
                        else names
                    logger.write('test', flat)
