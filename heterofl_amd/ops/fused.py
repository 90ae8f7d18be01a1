"""Autograd wrappers for the hand-written gfx950 kernels, with pure-torch
reference implementations (the CPU path and the numerics oracle in tests).

Semantics mirrored from the reference:
- Scaler -> norm -> ReLU prefix of every block (src/models/resnet.py:44-50,
  src/models/conv.py:29-33).  With a train-mode norm the Scaler division
  cancels exactly (norm(x/r) == norm(x)), so the fused kernels omit it; the
  eager path keeps the explicit division as the oracle.
- masked cross-entropy (src/models/resnet.py:152-157).
- per-client clip(1.0) + momentum-SGD (src/train_classifier_fed.py:205-206).
"""
import torch
import torch.nn.functional as F

from . import require_native, use_native


class _FusedNormReLU(torch.autograd.Function):
    """y = relu(norm(x) * w + b) with batch (kind='bn') or group stats."""

    @staticmethod
    def forward(ctx, x, weight, bias, kind, groups, eps):
        ext = require_native()
        x = x.contiguous()
        if kind == 'bn':
            y, mean, invstd = ext.bn_relu_fwd(x, weight, bias, eps)
        else:
            y, mean, invstd = ext.gn_relu_fwd(x, weight, bias, groups, eps)
        ctx.save_for_backward(x, weight, bias, mean, invstd)
        ctx.kind = kind
        ctx.groups = groups
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_native()
        x, weight, bias, mean, invstd = ctx.saved_tensors
        if ctx.kind == 'bn':
            dx, dgamma, dbeta = ext.bn_relu_bwd(dy, x, weight, bias, mean,
                                                invstd)
        else:
            dx, dgamma, dbeta = ext.gn_relu_bwd(dy, x, weight, bias, mean,
                                                invstd, ctx.groups)
        return dx, dgamma, dbeta, None, None, None


def fused_norm_relu(x, weight, bias, kind, groups=1, eps=1e-5):
    return _FusedNormReLU.apply(x, weight, bias, kind, groups, eps)


def eager_scaler_norm_relu(x, weight, bias, kind, groups, rate, eps=1e-5):
    """Reference path: scaler -> norm -> relu with explicit torch ops."""
    if rate != 1.0:
        x = x / rate
    if kind == 'bn':
        x = F.batch_norm(x, None, None, weight, bias, training=True, eps=eps)
    else:
        x = F.group_norm(x, groups, weight, bias, eps=eps)
    return F.relu(x)


class _FusedMaskedCE(torch.autograd.Function):
    """Per-client mean masked-CE over scores (N, R, C); optionally
    accumulates (sum-loss, correct, count) into a device metrics buffer."""

    @staticmethod
    def forward(ctx, scores, labels, mask, metrics):
        ext = require_native()
        scores = scores.contiguous().float()
        (losses,) = ext.masked_ce_fwd(
            scores, labels.contiguous(),
            mask if mask is not None else torch.Tensor(),
            metrics if metrics is not None else torch.Tensor())
        ctx.save_for_backward(scores, labels,
                              mask if mask is not None else torch.Tensor())
        ctx.has_mask = mask is not None
        return losses

    @staticmethod
    def backward(ctx, up):
        ext = require_native()
        scores, labels, mask = ctx.saved_tensors
        dscores = ext.masked_ce_bwd(scores, labels,
                                    mask if ctx.has_mask else torch.Tensor(),
                                    up)
        return dscores, None, None, None


def fused_masked_ce(scores, labels, mask=None, metrics=None):
    return _FusedMaskedCE.apply(scores, labels, mask, metrics)


class FusedClipSGD:
    """Chunk-table driven per-client clip + momentum-SGD over the batched
    model's parameters.  Build once per captured graph (pointer stability is
    guaranteed by holding refs to grads/params/bufs)."""

    CHUNK = 65536

    def __init__(self, params, grads, bufs, R, device, shadows=None):
        ext = require_native()
        self.params = list(params)
        self.grads = list(grads)
        self.bufs = list(bufs)
        self.shadows = [s if s is not None else torch.Tensor()
                        for s in (shadows or [])]
        blob, n, clients = ext.build_chunk_table(
            self.grads, self.params, self.bufs, R, self.CHUNK, self.shadows)
        self.table = blob
        self.n_chunks = int(n.item())
        self.chunk_client = clients
        self.partials = torch.zeros(self.n_chunks, dtype=torch.float32,
                                    device=device)
        self.normsq = torch.zeros(R, dtype=torch.float32, device=device)

    def step(self, max_norm, lr, momentum, weight_decay):
        ext = require_native()
        ext.clip_sgd_step(self.table, self.n_chunks, self.chunk_client,
                          self.partials, self.normsq, max_norm, lr, momentum,
                          weight_decay)


class _GroupedConv(torch.autograd.Function):
    """MFMA implicit-GEMM grouped conv (ops/csrc/conv_mfma.hip): bf16/fp32
    activations, fp32 master weights, fp32 weight gradients."""

    @staticmethod
    def forward(ctx, x, weight, bias, residual, groups, stride, pad):
        from . import fp8_enabled
        ext = require_native()
        fp8 = 1 if fp8_enabled() else 0
        x = x.contiguous()
        y = ext.conv_fwd(x, weight,
                         bias if bias is not None else torch.Tensor(),
                         residual.contiguous() if residual is not None
                         else torch.Tensor(),
                         groups, stride, pad, fp8)
        ctx.save_for_backward(x, weight)
        ctx.meta = (groups, stride, pad, bias is not None, fp8)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_native()
        x, w = ctx.saved_tensors
        groups, stride, pad, has_bias, fp8 = ctx.meta
        dy = dy.contiguous()
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = ext.conv_bwd_data(dy, w, groups, stride, pad,
                                   x.size(2), x.size(3), fp8)
        if ctx.needs_input_grad[1]:
            dw = ext.conv_bwd_weight(dy, x, groups, stride, pad, w.size(2),
                                     fp8)
        if has_bias and ctx.needs_input_grad[2]:
            db = dy.float().sum(dim=(0, 2, 3))
        # residual gradient is dy itself (identity add in the epilogue)
        dres = dy if ctx.needs_input_grad[3] else None
        return dx, dw, db, dres, None, None, None


def grouped_conv(x, weight, bias, groups, stride, pad, residual=None):
    return _GroupedConv.apply(x, weight, bias, residual, groups, stride, pad)


class _FusedHead(torch.autograd.Function):
    """AdaptiveAvgPool(1) + flatten + per-client Linear in one kernel each
    way (ops/csrc/head.hip).  scores come back fp32 (they feed masked-CE)."""

    @staticmethod
    def forward(ctx, feat, weight, bias, R):
        ext = require_native()
        feat = feat.contiguous()
        scores, pooled = ext.head_fwd(feat, weight,
                                      bias if bias is not None
                                      else torch.Tensor(), R)
        ctx.save_for_backward(pooled, weight)
        ctx.meta = (R, feat.size(2), feat.size(3),
                    feat.dtype == torch.bfloat16, bias is not None)
        return scores

    @staticmethod
    def backward(ctx, dscores):
        ext = require_native()
        pooled, weight = ctx.saved_tensors
        R, H, W, bf16_feat, has_bias = ctx.meta
        dw, db, dfeat = ext.head_bwd(dscores.contiguous().float(), pooled,
                                     weight, R, H, W, bf16_feat, has_bias)
        return dfeat, dw, (db if has_bias else None), None


def fused_head(feat, weight, bias, R):
    return _FusedHead.apply(feat, weight, bias, R)


class GraphClipSGD:
    """hipGraph-capture variant of FusedClipSGD: the chunk LAYOUT is fixed by
    parameter shapes, so the device table is preallocated and the kernels are
    recorded against it; the grad POINTERS (graph-pool addresses captured by
    autograd.grad) are filled in afterwards with fill_chunk_table."""

    CHUNK = 65536
    CHUNK_BYTES = 40  # sizeof(Chunk): 4 pointers + 2 ints

    def __init__(self, params, bufs, R, device, shadows=None):
        require_native()
        self.params = list(params)
        self.bufs = list(bufs)
        self.shadows = [s if s is not None else torch.Tensor()
                        for s in (shadows or [])]
        self.R = R
        clients = []
        for p in self.params:
            per = p.numel() // R
            nch = (per + self.CHUNK - 1) // self.CHUNK
            for r in range(R):
                clients.extend([r] * nch)
        self.n_chunks = len(clients)
        self.table = torch.zeros(self.n_chunks * self.CHUNK_BYTES,
                                 dtype=torch.uint8, device=device)
        self.chunk_client = torch.tensor(clients, dtype=torch.int32).to(device)
        self.partials = torch.zeros(self.n_chunks, dtype=torch.float32,
                                    device=device)
        self.normsq = torch.zeros(R, dtype=torch.float32, device=device)

    def bind(self, grads):
        """Write (grad, param, buf, shadow) pointers into the device table."""
        ext = require_native()
        ext.fill_chunk_table(self.table, list(grads), self.params, self.bufs,
                             self.R, self.CHUNK, self.shadows)

    def launch(self, max_norm, lr, momentum, weight_decay):
        ext = require_native()
        ext.clip_sgd_step(self.table, self.n_chunks, self.chunk_client,
                          self.partials, self.normsq, max_norm, lr, momentum,
                          weight_decay)


class _FusedAttention(torch.autograd.Function):
    """Fused attention (ops/csrc/attention.hip), head_dim<=32.  S<=64 runs
    the whole-(batch*head) single-kernel path saving the bf16 softmax
    matrix; larger S runs the blockwise (flash-style) path — 64-row KV
    tiles with an online softmax, per-row LSE saved, and a deterministic
    two-pass backward that recomputes P (SURVEY §5 long-context
    readiness)."""

    @staticmethod
    def forward(ctx, q, k, v, temperature):
        ext = require_native()
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        if q.size(1) <= 64:
            out, p = ext.attn_fwd(q, k, v, temperature)
            ctx.save_for_backward(q, k, v, p)
            ctx.blockwise = False
        else:
            out, lse = ext.attn_fwd_block(q, k, v, temperature)
            ctx.save_for_backward(q, k, v, out, lse)
            ctx.blockwise = True
        ctx.temperature = temperature
        return out

    @staticmethod
    def backward(ctx, dout):
        ext = require_native()
        if ctx.blockwise:
            q, k, v, out, lse = ctx.saved_tensors
            dq, dk, dv = ext.attn_bwd_block(dout, q, k, v, out, lse,
                                            ctx.temperature)
        else:
            q, k, v, p = ctx.saved_tensors
            dq, dk, dv = ext.attn_bwd(dout, q, k, v, p, ctx.temperature)
        return dq, dk, dv, None


def fused_attention(q, k, v, temperature):
    return _FusedAttention.apply(q, k, v, temperature)


class _FusedLayerNorm(torch.autograd.Function):
    """Per-client LayerNorm over the last dim (ops/csrc/layernorm.hip)."""

    @staticmethod
    def forward(ctx, x, weight, bias, R, eps):
        ext = require_native()
        x = x.contiguous()
        y, mean, invstd = ext.ln_fwd(x, weight, bias, R, eps)
        ctx.save_for_backward(x, weight, mean, invstd)
        ctx.R = R
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_native()
        x, weight, mean, invstd = ctx.saved_tensors
        dx, dgamma, dbeta = ext.ln_bwd(dy, x, weight, mean, invstd, ctx.R)
        return dx, dgamma, dbeta, None, None


def fused_layernorm(x, weight, bias, R, eps=1e-5):
    return _FusedLayerNorm.apply(x, weight, bias, R, eps)


class _FusedLMCE(torch.autograd.Function):
    """Vocab-masked LM cross-entropy over (R, B, S, V) bf16/fp32 logits;
    per-client mean NLL over every position (ops/csrc/lm_ce.hip)."""

    @staticmethod
    def forward(ctx, logits, tokens, mask, R):
        ext = require_native()
        logits = logits.contiguous()
        labels = tokens.reshape(-1)
        losses, row_lse = ext.lm_ce_fwd(
            logits, labels, mask if mask is not None else torch.Tensor(), R)
        ctx.save_for_backward(logits, labels, row_lse,
                              mask if mask is not None else torch.Tensor())
        ctx.R = R
        ctx.has_mask = mask is not None
        return losses

    @staticmethod
    def backward(ctx, up):
        ext = require_native()
        logits, labels, row_lse, mask = ctx.saved_tensors
        d = ext.lm_ce_bwd(logits, labels,
                          mask if ctx.has_mask else torch.Tensor(), row_lse,
                          up, ctx.R)
        return d, None, None, None


def fused_lm_ce(logits, tokens, mask, R):
    return _FusedLMCE.apply(logits, tokens, mask, R)


# ------------------------------------------------------- LM glue fusions
_rng_cells = {}
_rng_salt = [0]


def _next_salt():
    """Per-call-site salt: decorrelates the many dropout sites inside one
    step without a bump launch per call.  In eager mode the counter also
    advances across steps; in a captured graph the salts freeze at capture
    and the per-step rng_bump (launched by the step driver) moves the seed
    cell instead."""
    _rng_salt[0] = (_rng_salt[0] + 1) & 0x7FFFFFFF
    return _rng_salt[0]


def bump_rng(device):
    """Advance the device RNG seed cell (once per training step)."""
    ext = require_native()
    ext.rng_bump(_rng_cell(device))


def _rng_cell(device):
    """Per-device u64 seed cell read by the fused dropout kernels and bumped
    device-side after each draw — a stable pointer, so hipGraph replays draw
    fresh masks (torch's philox handles its own ops; these are ours)."""
    key = str(device)
    if key not in _rng_cells:
        _rng_cells[key] = torch.tensor(
            [torch.initial_seed() & 0x7FFFFFFFFFFFFFFF],
            dtype=torch.int64, device=device)
    return _rng_cells[key]


class _FusedGeluDropout(torch.autograd.Function):
    """dropout(gelu(x / rate), p) in one kernel each way, mask saved
    (reference chain: src/models/transformer.py:116 —
    dropout1(GELU(scaler(linear1(src)))))."""

    @staticmethod
    def forward(ctx, x, rate, p):
        ext = require_native()
        x = x.contiguous()
        y, mask = ext.gelu_drop_fwd(x, _rng_cell(x.device), _next_salt(),
                                    rate, p)
        ctx.save_for_backward(x, mask)
        ctx.meta = (rate, p)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_native()
        x, mask = ctx.saved_tensors
        rate, p = ctx.meta
        return ext.gelu_drop_bwd(dy, x, mask, rate, p), None, None


def fused_gelu_dropout(x, rate, p):
    return _FusedGeluDropout.apply(x, rate, p)


class _FusedResDropout(torch.autograd.Function):
    """src + dropout(h / rate, p) in one kernel; backward is dt for src and
    one mask-scale kernel for h (the LN that follows is already fused)."""

    @staticmethod
    def forward(ctx, src, h, rate, p):
        ext = require_native()
        t, mask = ext.res_drop_fwd(src.contiguous(), h.contiguous(),
                                   _rng_cell(src.device), _next_salt(),
                                   rate, p)
        ctx.save_for_backward(mask)
        ctx.meta = (rate, p)
        return t

    @staticmethod
    def backward(ctx, dt):
        ext = require_native()
        (mask,) = ctx.saved_tensors
        rate, p = ctx.meta
        dh = ext.drop_scale_bwd(dt, mask, rate, p)
        return dt, dh, None, None


def fused_res_dropout(src, h, rate, p):
    return _FusedResDropout.apply(src, h, rate, p)


class _FusedScaler(torch.autograd.Function):
    """Standalone Scaler kernel (K5 of SURVEY §2b; reference
    src/modules/modules.py:9-11): y = x / rate in training.  Runs the
    drop_scale kernel with p=0 (pure 1/rate scale) both directions — used
    by the norm='none' ablation, where the Scaler cannot cancel into a
    following train-mode norm."""

    @staticmethod
    def forward(ctx, x, rate):
        ext = require_native()
        ctx.rate = rate
        return ext.drop_scale_bwd(x.contiguous(), torch.Tensor(), rate, 0.0)

    @staticmethod
    def backward(ctx, dy):
        ext = require_native()
        return ext.drop_scale_bwd(dy.contiguous(), torch.Tensor(),
                                  ctx.rate, 0.0), None


def fused_scaler(x, rate):
    return _FusedScaler.apply(x, rate)


def fused_token_mask(tokens, rate, mask_id):
    """Bernoulli(rate) token masking in one kernel (K11; reference:
    src/models/transformer.py:149-151) — (seed, salt, index)-keyed RNG,
    graph-replay safe."""
    ext = require_native()
    return ext.token_mask(tokens.contiguous(), _rng_cell(tokens.device),
                          _next_salt(), rate, mask_id)


class _FusedMaxPool2(torch.autograd.Function):
    """2x2/stride-2 MaxPool (K6; reference src/models/conv.py:33): forward
    saves the winning-corner index, backward scatters to it."""

    @staticmethod
    def forward(ctx, x):
        ext = require_native()
        y, arg = ext.maxpool2_fwd(x.contiguous())
        ctx.save_for_backward(arg)
        ctx.hw = (x.size(2), x.size(3))
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_native()
        (arg,) = ctx.saved_tensors
        return ext.maxpool2_bwd(dy, arg, *ctx.hw)


def fused_maxpool2(x):
    return _FusedMaxPool2.apply(x)


class _FusedEmbedPos(torch.autograd.Function):
    """Fused token-embedding + positional-embedding gather with the Scaler
    folded in (K9; reference: src/models/transformer.py:29-37).  Reads the
    bf16 shadow tables when present; the deterministic backward writes fp32
    grads straight for the fp32 masters (no shadow-upcast kernels, no
    atomics — bit-stable under graph replay)."""

    @staticmethod
    def forward(ctx, ids, table, pos, t16, p16, rate):
        ext = require_native()
        use16 = t16 is not None and p16 is not None
        t = t16 if use16 else table
        p = p16 if use16 else pos
        out = ext.embed_pos_fwd(ids.contiguous(), t.contiguous(),
                                p.contiguous(), rate)
        ctx.save_for_backward(ids)
        ctx.meta = (table.size(1), pos.size(1), rate)
        return out

    @staticmethod
    def backward(ctx, dy):
        ext = require_native()
        (ids,) = ctx.saved_tensors
        V, P, rate = ctx.meta
        dtable, dpos = ext.embed_pos_bwd(dy, ids, V, P, rate)
        return None, dtable, dpos, None, None, None


def fused_embed_pos(ids, table, pos, t16, p16, rate):
    return _FusedEmbedPos.apply(ids, table, pos, t16, p16, rate)
