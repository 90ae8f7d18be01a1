"""Trainer driving the batched transformer engine: the round's active LM
clients grouped by (rate, rows, windows) and trained together.

Reference local loop: src/train_transformer_fed.py:158-176 — each client
iterates the bptt windows of its own rows of the batchified token matrix for
num_epochs['local'] epochs with clip(1.0)+momentum-SGD.
"""
import torch

from .. import ops as native_ops
from .batched import pack_states, unpack_states, per_client_clip_
from .batched_lm import (make_batched_transformer, lm_masked_ce,
                         enable_bf16_shadows, refresh_shadows)


class BatchedLMClientTrainer:
    def __init__(self, cfg):
        self.cfg = cfg
        self.device = torch.device(cfg['device'])
        self.token = None
        self.data_split = None
        self._amp = (cfg.get('compute_dtype') == 'bfloat16'
                     and self.device.type == 'cuda')
        self._model_cache = {}
        self._opt_cache = {}

    def set_data(self, dataset, data_split):
        self.token = dataset['train'].token.to(self.device)
        self.data_split = data_split['train']

    def _model(self, rate, R):
        key = (rate, R)
        if key not in self._model_cache:
            self._model_cache[key] = make_batched_transformer(
                self.cfg, rate, R).to(self.device)
        return self._model_cache[key]

    def train_clients(self, client_slots, user_idx, local_parameters,
                      model_rate, make_loader, label_split, lr, logger=None):
        cfg = self.cfg
        groups = {}
        for m in client_slots:
            u = user_idx[m]
            groups.setdefault((model_rate[u], len(self.data_split[u])),
                              []).append(m)
        out = []
        for (rate, n_rows), slots in groups.items():
            out.extend(self._train_group(rate, slots, user_idx,
                                         [local_parameters[m] for m in slots],
                                         label_split, lr, logger))
        return out

    def _train_group(self, rate, slots, user_idx, locals_list, label_split,
                     lr, logger=None):
        cfg = self.cfg
        R = len(slots)
        device = self.device
        bptt = cfg['bptt']
        if device.type == 'cuda' and cfg.get('hip_graphs', True):
            return self._train_group_graphed(rate, slots, user_idx,
                                             locals_list, label_split, lr,
                                             logger)
        model = self._model(rate, R)
        native = native_ops.use_native(device)
        use_shadows = native and self._amp
        if use_shadows:
            smap = enable_bf16_shadows(model)
        pack_states(model, locals_list)
        if use_shadows:
            refresh_shadows(model)
        model.train(True)
        params = [p for p in model.parameters() if p.requires_grad]
        if native:
            from ..ops.fused import FusedClipSGD
            cached = self._opt_cache.get((rate, R))
            if cached is None or cached[0] is not model:
                for p in params:
                    p.grad = torch.zeros_like(p)
                bufs = [torch.zeros_like(p) for p in params]
                shadows = [smap.get(id(p)) for p in params] \
                    if use_shadows else None
                cached = (model, FusedClipSGD(params,
                                              [p.grad for p in params],
                                              bufs, R, device,
                                              shadows=shadows))
                self._opt_cache[(rate, R)] = cached
            fopt = cached[1]
            for b in fopt.bufs:
                b.zero_()
        else:
            opt = torch.optim.SGD(params, lr=lr, momentum=cfg['momentum'],
                                  weight_decay=cfg['weight_decay'])
        masks = None
        if cfg['mask']:
            masks = torch.zeros(R, cfg['num_tokens'], device=device)
            for i, m in enumerate(slots):
                masks[i, label_split[user_idx[m]]] = 1
        # client rows of the batchified token matrix, stacked (R, B, L)
        rows = torch.stack([self.token[self.data_split[user_idx[m]]]
                            for m in slots])
        L = rows.size(2)
        n_win = (L + bptt - 1) // bptt
        for _ in range(cfg['num_epochs']['local']):
            for w in range(n_win):
                tokens = rows[:, :, w * bptt:(w + 1) * bptt]
                if native:
                    torch._foreach_zero_([p.grad for p in params])
                else:
                    opt.zero_grad(set_to_none=True)
                with torch.autocast('cuda', torch.bfloat16,
                                    enabled=self._amp and not use_shadows):
                    logits = model(tokens)
                losses = lm_masked_ce(logits, tokens, masks)
                losses.sum().backward()
                if native:
                    fopt.step(1.0, lr, cfg['momentum'], cfg['weight_decay'])
                    if use_shadows:
                        from ..ops.fused import bump_rng
                        bump_rng(device)
                else:
                    per_client_clip_(params, R, 1.0)
                    opt.step()
                if logger is not None:
                    with torch.no_grad():
                        # weight = window row count, the reference's n
                        # (src/train_transformer_fed.py:168-171)
                        n = tokens.size(1)
                        for i in range(R):
                            l = losses[i].item()
                            logger.append({'Local-Loss': l,
                                           'Local-Perplexity':
                                               float(torch.exp(torch.tensor(
                                                   min(l, 30.0))))},
                                          'train', n=n)
        template_keys = list(locals_list[0].keys())
        states = unpack_states(model, template_keys)
        return list(zip(slots, [dict(st) for st in states]))


    def _train_group_graphed(self, rate, slots, user_idx, locals_list,
                             label_split, lr, logger=None):
        """hipGraph path: stage token windows once, replay the captured step
        for every full window per epoch; a ragged last window runs eagerly."""
        from .graphs import LMGraphedStep
        cfg = self.cfg
        R = len(slots)
        device = self.device
        bptt = cfg['bptt']
        rows = torch.stack([self.token[self.data_split[user_idx[m]]]
                            for m in slots])
        L = rows.size(2)
        n_full = L // bptt
        key = ('g', rate, R, rows.size(1), n_full, lr)
        if not hasattr(self, '_graph_cache'):
            self._graph_cache = {}
        if key not in self._graph_cache:
            model = make_batched_transformer(cfg, rate, R).to(device)
            model.train(True)
            if self._amp and native_ops.use_native(device):
                enable_bf16_shadows(model)
            self._graph_cache[key] = LMGraphedStep(
                model, R, rows.size(1), bptt, n_full, lr, cfg['momentum'],
                cfg['weight_decay'], cfg['num_tokens'], self._amp, device)
        gs = self._graph_cache[key]
        pack_states(gs.model, locals_list)
        refresh_shadows(gs.model)
        masks = None
        if cfg['mask']:
            masks = torch.zeros(R, cfg['num_tokens'], device=device)
            for i, m in enumerate(slots):
                masks[i, label_split[user_idx[m]]] = 1
        gs.begin_round(rows, masks)
        tail = rows[:, :, n_full * bptt:] if L % bptt else None
        for _ in range(cfg['num_epochs']['local']):
            gs.run_window_pass()
            if tail is not None and tail.size(2) > 0:
                gs.tail_step(tail)
        if logger is not None:
            m = gs.metrics.detach().cpu()
            for i in range(R):
                cnt = max(int(m[i, 2].item()), 1)
                # row-weighted means of per-window loss and exp(loss) — the
                # reference's aggregation (mean-of-exp, weighted by
                # input['label'].size(0); src/train_transformer_fed.py:168-171)
                logger.append({'Local-Loss': m[i, 0].item() / cnt,
                               'Local-Perplexity': m[i, 1].item() / cnt},
                              'train', n=cnt)
        template_keys = list(locals_list[0].keys())
        states = unpack_states(gs.model, template_keys)
        return list(zip(slots, [dict(st) for st in states]))
