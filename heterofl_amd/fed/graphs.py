"""hipGraph-captured local-training steps.

HeteroFL's hot loop is thousands of tiny steps (batch 10 on 32x32 images,
~400 kernels each) — on MI355X that is launch-latency bound, not FLOP bound
(SURVEY §7 'tiny-work kernels').  We capture ONE training step of the
batched group model in a hipGraph (torch.cuda.CUDAGraph == hipGraph on ROCm)
with a DEVICE-side step counter: the graph gathers its own batch from the
round's pre-staged epoch buffer via the counter, so an entire local-training
run (5 epochs x 50 steps) is 250 back-to-back graph replays with no host
work in between.

With the native extension the step is: MFMA grouped-conv + fused norm+relu
+ fused head kernels in the model, fused masked-CE (which also accumulates
device-side metrics), autograd.grad into graph-pool buffers, then the fused
per-client clip(1.0)+momentum-SGD (ops/csrc/clip_sgd.hip) over a chunk table
whose pointers are bound after capture (GraphClipSGD).  Without the
extension (CPU/debug) the pure-torch step runs.

Graphs are cached per (rate, R, batch, lr, n_steps_capacity); weights,
momentum and the label masks live in stable buffers the graph reads, so a
new round only repacks buffers and replays.
"""
import torch

from .. import ops as native_ops
from .batched import batched_masked_ce


class GraphedGroupStep:
    """One captured training step for a (rate, R) client group."""

    def __init__(self, model, R, batch, lr, momentum, weight_decay, classes,
                 capacity, amp, device):
        self.model = model
        self.R = R
        self.batch = batch
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.amp = amp
        self.device = device
        self.params = [p for p in model.parameters() if p.requires_grad]
        self.bufs = [torch.zeros_like(p) for p in self.params]
        self.capacity = capacity  # max samples in the epoch buffer
        shape = self._input_shape()
        # bf16 local-training: inputs staged in bf16 so the native kernels
        # (which bypass torch autocast) really compute in bf16
        xdt = torch.bfloat16 if amp else torch.float32
        self.x_all = torch.zeros(capacity, *shape, device=device, dtype=xdt)
        self.y_all = torch.zeros(capacity, R, dtype=torch.long, device=device)
        self.masks = torch.ones(R, classes, device=device)
        self.counter = torch.zeros(1, dtype=torch.long, device=device)
        # device metric accumulators: [loss_sum, correct, samples] per client
        self.metrics = torch.zeros(R, 3, device=device)
        self.graph = None
        self._arange = torch.arange(batch, device=device)
        self._native = (device.type == 'cuda' and native_ops.use_native(device))
        self._capturing = False
        self._captured_grads = None
        self._shadows = False  # bf16 shadow weights are an LM-path feature
        if self._native:
            native_ops.require_native()
            from ..ops.fused import GraphClipSGD
            self.opt = GraphClipSGD(self.params, self.bufs, R, device)

    def _input_shape(self):
        m = self.model
        w = next(p for p in m.parameters() if p.dim() == 4)
        in_ch = w.size(1)
        hw = 32 if in_ch == 3 else 28
        return (self.R * in_ch, hw, hw)

    def _one_step(self):
        idx = self.counter * self.batch + self._arange
        xb = self.x_all.index_select(0, idx)
        yb = self.y_all.index_select(0, idx)
        if self._native:
            with torch.autocast('cuda', torch.bfloat16, enabled=self.amp):
                scores = self.model(xb)
            losses = batched_masked_ce(scores.float(), yb, self.masks,
                                       metrics=self.metrics)
            raw = torch.autograd.grad(losses.sum(), self.params)
            grads = [t if t.is_contiguous() else t.contiguous() for t in raw]
            self._captured_grads = grads
            if not self._capturing:
                self.opt.bind(grads)  # eager warmup: per-iteration pointers
            self.opt.launch(1.0, self.lr, self.momentum, self.weight_decay)
            with torch.no_grad():
                self.counter += 1
            return
        with torch.autocast('cuda', torch.bfloat16, enabled=self.amp):
            scores = self.model(xb)
            losses = batched_masked_ce(scores.float(), yb, self.masks)
        loss = losses.sum()
        raw = torch.autograd.grad(loss, self.params)
        with torch.no_grad():
            grads = [g.contiguous() for g in raw]
            # per-client global-L2 clip at 1.0
            sq = None
            views = [g.view(self.R, -1) for g in grads]
            for v in views:
                s = (v.float() ** 2).sum(dim=1)
                sq = s if sq is None else sq + s
            scale = (1.0 / (sq.sqrt() + 1e-6)).clamp(max=1.0)
            for v in views:
                v.mul_(scale.unsqueeze(1).to(v.dtype))
            # momentum SGD (wd before momentum, dampening 0 — torch semantics)
            torch._foreach_add_(grads, self.params, alpha=self.weight_decay)
            torch._foreach_mul_(self.bufs, self.momentum)
            torch._foreach_add_(self.bufs, grads)
            torch._foreach_add_(self.params, self.bufs, alpha=-self.lr)
            # metrics
            pred = scores.argmax(dim=2)
            correct = (pred == yb).float().sum(0)
            self.metrics[:, 0] += losses.detach() * self.batch
            self.metrics[:, 1] += correct
            self.metrics[:, 2] += self.batch
            self.counter += 1

    def capture(self):
        """Warmup + capture mutate params/bufs/metrics — snapshot and
        restore so lazy capture inside a round is state-neutral."""
        saved_p = [p.detach().clone() for p in self.params]
        saved_b = [b.clone() for b in self.bufs]
        saved_m = self.metrics.clone()
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):  # warmup allocations / autotune
                self.counter.zero_()
                self._one_step()
        torch.cuda.current_stream().wait_stream(s)
        self.counter.zero_()
        self.graph = torch.cuda.CUDAGraph()
        self._capturing = True
        with torch.cuda.graph(self.graph):
            self._one_step()
        self._capturing = False
        if self._native:
            # bind the captured grad-pool addresses (stable across replays)
            self.opt.bind(self._captured_grads)
        with torch.no_grad():
            for p, sp in zip(self.params, saved_p):
                p.copy_(sp)
            for b, sb in zip(self.bufs, saved_b):
                b.copy_(sb)
            self.metrics.copy_(saved_m)
            self.counter.zero_()
        if self._shadows:
            from .batched_lm import refresh_shadows
            refresh_shadows(self.model)
        torch.cuda.synchronize()

    # ------------------------------------------------------------------ API
    def begin_round(self, masks):
        """Load label masks, zero momentum and metrics.  (Weights are packed
        directly into self.model's parameters — stable graph addresses.)"""
        with torch.no_grad():
            if masks is not None:
                self.masks.copy_(masks)
            else:
                self.masks.fill_(1)
            for b in self.bufs:
                b.zero_()
            self.metrics.zero_()

    def run_epochs(self, x_epochs, y_epochs, steps_per_epoch):
        """x_epochs: (total, R*C, H, W) pre-augmented, all epochs concat."""
        total = x_epochs.size(0)
        with torch.no_grad():
            self.x_all[:total].copy_(x_epochs)
            self.y_all[:total].copy_(y_epochs)
            self.counter.zero_()
        n_steps = total // self.batch
        if self.graph is None:
            self.capture()
        for _ in range(n_steps):
            self.graph.replay()

    def ensure_captured(self):
        if self.graph is None:
            self.capture()


class LMGraphedStep:
    """hipGraph-captured masked-LM training step for a (rate, R) client
    group (reference local loop: src/train_transformer_fed.py:163-176).

    LM client data is static across local epochs (ordered bptt windows of
    the client's token rows, no augmentation), so the round stages the
    full-window tensor (n_win, R, B, S) once and the graph gathers window
    counter % n_win with a device-side counter.  torch's graph-private
    philox state makes the in-graph Bernoulli masking / dropout draw fresh
    randomness each replay."""

    def __init__(self, model, R, n_rows, bptt, n_win, lr, momentum,
                 weight_decay, num_tokens, amp, device):
        self.model = model
        self.R = R
        self.bptt = bptt
        self.n_win = n_win
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.amp = amp
        self.device = device
        self.params = [p for p in model.parameters() if p.requires_grad]
        self.bufs = [torch.zeros_like(p) for p in self.params]
        self.windows = torch.zeros(n_win, R, n_rows, bptt, dtype=torch.long,
                                   device=device)
        self.masks = torch.ones(R, num_tokens, device=device)
        self.counter = torch.zeros(1, dtype=torch.long, device=device)
        self.metrics = torch.zeros(R, 3, device=device)
        self.graph = None
        self._capturing = False
        self._captured_grads = None
        self._native = (device.type == 'cuda' and native_ops.use_native(device))
        smap = getattr(model, '_shadow_map', None) or {}
        self._shadows = bool(smap)
        if self._native:
            from ..ops.fused import GraphClipSGD
            self.opt = GraphClipSGD(self.params, self.bufs, R, device,
                                    shadows=[smap.get(id(p))
                                             for p in self.params]
                                    if smap else None)

    def _one_step(self):
        w = self.counter - (self.counter // self.n_win) * self.n_win
        tokens = self.windows.index_select(0, w).squeeze(0)
        self._step_on(tokens)
        with torch.no_grad():
            self.counter += 1

    def tail_step(self, tokens):
        """Eager step for a ragged last window (shorter S than the graph).
        The eager step rebinds the shared chunk table to its own grad
        tensors; restore the captured binding so later replays read the
        graph-pool grads again."""
        self._step_on(tokens)
        if self._native and self.graph is not None:
            self.opt.bind(self._graph_grads)

    def _step_on(self, tokens):
        from .batched_lm import lm_masked_ce
        # with bf16 shadow weights the model computes in bf16 natively and
        # autocast (with its per-use weight casts) is unnecessary
        with torch.autocast('cuda', torch.bfloat16,
                            enabled=self.amp and not self._shadows):
            logits = self.model(tokens)
        losses = lm_masked_ce(logits, tokens, self.masks)
        with torch.no_grad():
            # reference metric aggregation weights each bptt window by its
            # row count and logs per-window exp(loss)
            # (src/train_transformer_fed.py:168-171 with logger.append
            # n=input['label'].size(0)) — accumulate loss*rows, exp(loss)*rows
            # and rows device-side so the graphed round reproduces it
            n = tokens.size(1)
            ld = losses.detach()
            self.metrics[:, 0] += ld * n
            self.metrics[:, 1] += torch.exp(ld.clamp(max=30.0)) * n
            self.metrics[:, 2] += n
        if self._native:
            raw = torch.autograd.grad(losses.sum(), self.params)
            grads = [t if t.is_contiguous() else t.contiguous() for t in raw]
            self._captured_grads = grads
            if not self._capturing:
                self.opt.bind(grads)
            self.opt.launch(1.0, self.lr, self.momentum, self.weight_decay)
            if self._shadows:
                # one seed-cell advance per step: the fused dropout sites
                # draw (seed, salt, index)-keyed masks, fresh every replay
                from ..ops.fused import bump_rng
                bump_rng(self.device)
        else:
            raw = torch.autograd.grad(losses.sum(), self.params)
            with torch.no_grad():
                grads = [g.contiguous() for g in raw]
                sq = None
                views = [g.view(self.R, -1) for g in grads]
                for v in views:
                    s = (v.float() ** 2).sum(dim=1)
                    sq = s if sq is None else sq + s
                scale = (1.0 / (sq.sqrt() + 1e-6)).clamp(max=1.0)
                for v in views:
                    v.mul_(scale.unsqueeze(1).to(v.dtype))
                torch._foreach_add_(grads, self.params,
                                    alpha=self.weight_decay)
                torch._foreach_mul_(self.bufs, self.momentum)
                torch._foreach_add_(self.bufs, grads)
                torch._foreach_add_(self.params, self.bufs, alpha=-self.lr)

    def capture(self):
        saved_p = [p.detach().clone() for p in self.params]
        saved_b = [b.clone() for b in self.bufs]
        saved_m = self.metrics.clone()
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                self.counter.zero_()
                self._one_step()
        torch.cuda.current_stream().wait_stream(s)
        self.counter.zero_()
        self.graph = torch.cuda.CUDAGraph()
        self._capturing = True
        with torch.cuda.graph(self.graph):
            self._one_step()
        self._capturing = False
        if self._native:
            self._graph_grads = self._captured_grads
            self.opt.bind(self._graph_grads)
        with torch.no_grad():
            for p, sp in zip(self.params, saved_p):
                p.copy_(sp)
            for b, sb in zip(self.bufs, saved_b):
                b.copy_(sb)
            self.metrics.copy_(saved_m)
            self.counter.zero_()
        if self._shadows:
            from .batched_lm import refresh_shadows
            refresh_shadows(self.model)
        torch.cuda.synchronize()

    def begin_round(self, rows, masks):
        """rows (R, B, L): client token rows; stages the window tensor."""
        with torch.no_grad():
            for w in range(self.n_win):
                self.windows[w].copy_(
                    rows[:, :, w * self.bptt:(w + 1) * self.bptt])
            if masks is not None:
                self.masks.copy_(masks)
            else:
                self.masks.fill_(1)
            for b in self.bufs:
                b.zero_()
            self.metrics.zero_()
            self.counter.zero_()

    def run_window_pass(self):
        """One epoch's worth of full windows (counter wraps modulo n_win)."""
        if self.graph is None:
            self.capture()
        for _ in range(self.n_win):
            self.graph.replay()
