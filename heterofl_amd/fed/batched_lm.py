"""Batched-client transformer engine: R same-rate clients' masked-LM models
trained as ONE model whose every parameter is the (R, *local_shape) stack of
the clients' parameters.  LM clients own whole rows of the batchified token
matrix (often a single row), so per-client work is tiny — batching clients
into bmm-shaped ops is what fills an MI355X (same argument as fed/batched.py
for vision).

State-dict keys mirror models/transformer.py exactly (stacked shapes), so
fed pack/unpack is mechanical (reference slicing rules: src/fed.py:104-156).
Reference model semantics: src/models/transformer.py:11-174.
"""
import math

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops as native_ops


class _ShadowParam(torch.autograd.Function):
    """Route a persistent bf16 shadow of an fp32 master parameter through
    autograd: forward returns the shadow with NO cast kernel (the fused
    clip+SGD kernel mirrors every update into it, ops/csrc/clip_sgd.hip),
    backward upcasts the bf16 grad for the fp32 master.  This removes the
    ~80 per-step autocast weight-cast launches the round-1 LM profile was
    dominated by (profiles/kstats_r01_lm.txt)."""

    @staticmethod
    def forward(ctx, param, shadow):
        return shadow

    @staticmethod
    def backward(ctx, g):
        return g.float(), None


class SLinear(nn.Module):
    """R stacked Linears: weight (R, out, in), x (R, B, in) -> (R, B, out)."""

    def __init__(self, R, in_f, out_f):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(R, out_f, in_f).normal_(0, 0.02))
        self.bias = nn.Parameter(torch.zeros(R, out_f))
        self.register_buffer('w16', None, persistent=False)
        self.register_buffer('b16', None, persistent=False)

    def forward(self, x):
        lead = x.shape[:-1]
        R = self.weight.size(0)
        if self.w16 is not None and x.is_cuda:
            w = _ShadowParam.apply(self.weight, self.w16)
            b = _ShadowParam.apply(self.bias, self.b16)
            if x.dtype != torch.bfloat16:
                x = x.to(torch.bfloat16)
        else:
            w, b = self.weight, self.bias
        flat = x.reshape(R, -1, x.size(-1))
        out = torch.baddbmm(b.unsqueeze(1), flat, w.transpose(1, 2))
        return out.reshape(*lead, w.size(1))


class SLayerNorm(nn.Module):
    """Per-client LayerNorm over the last dim: weight/bias (R, E)."""

    def __init__(self, R, E):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(R, E))
        self.bias = nn.Parameter(torch.zeros(R, E))

    def forward(self, x):
        # x (R, ..., E)
        if native_ops.use_native(x):
            from ..ops.fused import fused_layernorm
            return fused_layernorm(x, self.weight, self.bias,
                                   self.weight.size(0))
        mu = x.mean(-1, keepdim=True)
        var = x.var(-1, unbiased=False, keepdim=True)
        xhat = (x - mu) * torch.rsqrt(var + 1e-5)
        extra = x.dim() - 2
        w = self.weight.reshape(self.weight.size(0), *([1] * extra), -1)
        b = self.bias.reshape(self.bias.size(0), *([1] * extra), -1)
        return xhat * w + b


class SScaler(nn.Module):
    def __init__(self, rate):
        super().__init__()
        self.rate = rate

    def forward(self, x):
        if not self.training or self.rate == 1.0:
            return x
        if native_ops.use_native(x):
            from ..ops.fused import fused_scaler
            return fused_scaler(x, self.rate)
        return x / self.rate


class SEmbedding(nn.Module):
    """R stacked embedding tables (R, V, E); ids (R, B, S) -> (R, B, S, E)."""

    def __init__(self, R, V, E):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(R, V, E).normal_(0, 0.02))
        self.register_buffer('w16', None, persistent=False)

    def forward(self, ids):
        R, V, E = self.weight.shape
        w = self.weight
        if self.w16 is not None and ids.is_cuda:
            w = _ShadowParam.apply(self.weight, self.w16)
        flat = w.reshape(R * V, E)
        off = (torch.arange(R, device=ids.device) * V).view(R, *([1] * (ids.dim() - 1)))
        return F.embedding(ids + off, flat)


def enable_bf16_shadows(model):
    """Create bf16 shadow buffers for every SLinear / SEmbedding master
    parameter.  Returns {id(param): shadow}; pass the aligned shadow list to
    FusedClipSGD/GraphClipSGD so the optimizer kernel keeps the shadows
    current in-graph (no per-step cast launches)."""
    shadow_map = getattr(model, '_shadow_map', None)
    if shadow_map is not None:
        return shadow_map
    shadow_map = {}
    for mod in model.modules():
        if isinstance(mod, SLinear):
            names = ('weight', 'bias')
        elif isinstance(mod, SEmbedding):
            names = ('weight',)
        else:
            continue
        for pname in names:
            p = getattr(mod, pname)
            sh = p.detach().to(torch.bfloat16).contiguous()
            setattr(mod, 'w16' if pname == 'weight' else 'b16', sh)
            shadow_map[id(p)] = sh
    model._shadow_map = shadow_map
    return shadow_map


def refresh_shadows(model):
    """Re-cast every shadow from its fp32 master (after pack_states wrote
    new round parameters, or after graph-capture state restore)."""
    shadow_map = getattr(model, '_shadow_map', None)
    if not shadow_map:
        return
    with torch.no_grad():
        for p in model.parameters():
            sh = shadow_map.get(id(p))
            if sh is not None:
                sh.copy_(p)


class STransformerEmbedding(nn.Module):
    def __init__(self, R, num_tokens, bptt, E, dropout, rate):
        super().__init__()
        self.embedding = SEmbedding(R, num_tokens + 1, E)
        self.positional_embedding = nn.ModuleDict(
            {'positional_embedding': SEmbedding(R, bptt, E)})
        self.norm = SLayerNorm(R, E)
        self.dropout = nn.Dropout(dropout)
        self.scaler = SScaler(rate)

    def forward(self, src):
        R, B, S = src.shape
        if native_ops.use_native(src):
            from ..ops.fused import fused_embed_pos
            emb = self.embedding
            pe = self.positional_embedding['positional_embedding']
            r = self.scaler.rate if self.training else 1.0
            x = fused_embed_pos(src, emb.weight, pe.weight, emb.w16,
                                pe.w16, r)
            return self.dropout(self.norm(x))
        pos = torch.arange(S, device=src.device).view(1, 1, S).expand(R, B, S)
        x = self.scaler(self.embedding(src)) + \
            self.scaler(self.positional_embedding['positional_embedding'](pos))
        return self.dropout(self.norm(x))


class SMultiheadAttention(nn.Module):
    def __init__(self, R, E, num_heads, rate):
        super().__init__()
        self.num_heads = num_heads
        self.temperature = (E // num_heads) ** 0.5
        self.linear_q = SLinear(R, E, E)
        self.linear_k = SLinear(R, E, E)
        self.linear_v = SLinear(R, E, E)
        self.linear_o = SLinear(R, E, E)
        self.scaler = SScaler(rate)

    def forward(self, x):
        R, B, S, E = x.shape
        H = self.num_heads
        d = E // H
        q = self.scaler(self.linear_q(x)).reshape(R, B, S, H, d)
        k = self.scaler(self.linear_k(x)).reshape(R, B, S, H, d)
        v = self.scaler(self.linear_v(x)).reshape(R, B, S, H, d)
        q = q.permute(0, 1, 3, 2, 4).reshape(R * B * H, S, d)
        k = k.permute(0, 1, 3, 2, 4).reshape(R * B * H, S, d)
        v = v.permute(0, 1, 3, 2, 4).reshape(R * B * H, S, d)
        if native_ops.use_native(q) and d <= 32:
            # S<=64 one-kernel path; larger S blockwise flash-style
            from ..ops.fused import fused_attention
            out = fused_attention(q, k, v, self.temperature)
        else:
            scores = torch.bmm(q, k.transpose(1, 2)) / self.temperature
            attn = F.softmax(scores, dim=-1)
            out = torch.bmm(attn, v)
        out = out.reshape(R, B, H, S, d).permute(0, 1, 3, 2, 4).reshape(R, B, S, E)
        return self.scaler(self.linear_o(out))


class STransformerEncoderLayer(nn.Module):
    def __init__(self, R, E, num_heads, hidden, dropout, rate):
        super().__init__()
        self.mha = SMultiheadAttention(R, E, num_heads, rate)
        self.dropout = nn.Dropout(dropout)
        self.norm1 = SLayerNorm(R, E)
        self.linear1 = SLinear(R, E, hidden)
        self.dropout1 = nn.Dropout(dropout)
        self.linear2 = SLinear(R, hidden, E)
        self.dropout2 = nn.Dropout(dropout)
        self.norm2 = SLayerNorm(R, E)
        self.scaler = SScaler(rate)

    def forward(self, src):
        if native_ops.use_native(src):
            # fused glue (ops/csrc/lm_fused.hip): scaler+GELU+dropout and
            # scaler+dropout+residual collapse the layer's ~14 elementwise
            # launches into 4 (the LNs are already one kernel each)
            from ..ops.fused import fused_gelu_dropout, fused_res_dropout
            p = float(self.dropout.p) if self.training else 0.0
            r = self.scaler.rate if self.training else 1.0
            src = self.norm1(fused_res_dropout(src, self.mha(src), 1.0, p))
            h = self.linear2(fused_gelu_dropout(self.linear1(src), r, p))
            return self.norm2(fused_res_dropout(src, h, r, p))
        src = self.norm1(src + self.dropout(self.mha(src)))
        h = self.scaler(self.linear2(self.dropout1(
            F.gelu(self.scaler(self.linear1(src))))))
        return self.norm2(src + self.dropout2(h))


class SDecoder(nn.Module):
    def __init__(self, R, num_tokens, E, rate):
        super().__init__()
        self.linear1 = SLinear(R, E, E)
        self.scaler = SScaler(rate)
        self.norm1 = SLayerNorm(R, E)
        self.linear2 = SLinear(R, E, num_tokens)

    def forward(self, x):
        if native_ops.use_native(x):
            from ..ops.fused import fused_gelu_dropout
            r = self.scaler.rate if self.training else 1.0
            return self.linear2(self.norm1(
                fused_gelu_dropout(self.linear1(x), r, 0.0)))
        return self.linear2(self.norm1(F.gelu(self.scaler(self.linear1(x)))))


class BatchedTransformer(nn.Module):
    """R same-rate clients' Transformers.  Keys mirror models.transformer
    .Transformer with an (R, ...) leading stack dim on every tensor."""

    def __init__(self, R, num_tokens, bptt, E, num_heads, hidden, num_layers,
                 dropout, rate, mask_rate):
        super().__init__()
        self.R = R
        self.num_tokens = num_tokens
        self.mask_rate = mask_rate
        self.transformer_embedding = STransformerEmbedding(
            R, num_tokens, bptt, E, dropout, rate)
        self.transformer_encoder = nn.ModuleDict({'layers': nn.ModuleList(
            [STransformerEncoderLayer(R, E, num_heads, hidden, dropout, rate)
             for _ in range(num_layers)])})
        self.decoder = SDecoder(R, num_tokens, E, rate)

    def forward(self, tokens):
        """tokens (R, B, S) -> logits (R, B, S, V)."""
        if native_ops.use_native(tokens):
            # one-kernel Bernoulli masking (K11), graph-replay safe
            from ..ops.fused import fused_token_mask
            src = fused_token_mask(tokens, self.mask_rate, self.num_tokens)
        else:
            mask = torch.bernoulli(
                torch.full(tokens.shape, self.mask_rate,
                           device=tokens.device))
            src = tokens.masked_fill(mask == 1, self.num_tokens).detach()
        x = self.transformer_embedding(src)
        for layer in self.transformer_encoder['layers']:
            x = layer(x)
        return self.decoder(x)


def lm_masked_ce(logits, tokens, label_masks):
    """Per-client mean CE over all positions with optional vocab masking
    (reference: src/models/transformer.py:156-161).  logits (R, B, S, V),
    tokens (R, B, S), label_masks (R, V) in {0,1} or None.  Returns (R,).
    On GPU the fused large-vocab kernel runs (logits stay bf16)."""
    R, B, S, V = logits.shape
    if native_ops.use_native(logits):
        from ..ops.fused import fused_lm_ce
        return fused_lm_ce(logits, tokens, label_masks, R)
    if label_masks is not None:
        logits = logits.masked_fill(
            label_masks.view(R, 1, 1, V) == 0, 0)
    logp = F.log_softmax(logits, dim=-1)
    nll = -logp.gather(3, tokens.unsqueeze(3)).squeeze(3)
    return nll.reshape(R, -1).mean(1)


def make_batched_transformer(cfg, rate, R):
    tcfg = cfg['transformer']
    E = int(np.ceil(rate * tcfg['embedding_size']))
    hidden = int(np.ceil(rate * tcfg['hidden_size']))
    scaler_rate = rate / cfg['global_model_rate']
    return BatchedTransformer(R, cfg['num_tokens'], cfg['bptt'], E,
                              tcfg['num_heads'], hidden, tcfg['num_layers'],
                              tcfg['dropout'], scaler_rate, cfg['mask_rate'])
