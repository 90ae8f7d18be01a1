"""Masked-LM transformer (reference: src/models/transformer.py:11-174).

Learned positional embedding over bptt positions; token+positional embedding
-> LayerNorm -> Dropout; custom MultiheadAttention with separate
linear_q/k/v/o (the names are load-bearing for per-head slicing); post-norm
encoder layers with GELU FFN and Scaler around both linears; Decoder
linear1 -> Scaler -> GELU -> LN -> linear2-to-vocab.  Forward Bernoulli-masks
input tokens to the <mask> id (= num_tokens) and predicts all positions with
CE over (N, vocab, S), vocab-masked under cfg['mask'].
"""
import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from .modules import Scaler, init_param
from .functional import masked_cross_entropy


class PositionalEmbedding(nn.Module):
    def __init__(self, bptt, embedding_size):
        super().__init__()
        self.positional_embedding = nn.Embedding(bptt, embedding_size)

    def forward(self, x):
        N, S = x.size()
        position = torch.arange(S, dtype=torch.long, device=x.device)
        return self.positional_embedding(position).unsqueeze(0).expand(N, S, -1)


class TransformerEmbedding(nn.Module):
    def __init__(self, num_tokens, bptt, embedding_size, dropout, rate):
        super().__init__()
        self.num_tokens = num_tokens
        self.positional_embedding = PositionalEmbedding(bptt, embedding_size)
        self.embedding = nn.Embedding(num_tokens + 1, embedding_size)
        self.norm = nn.LayerNorm(embedding_size)
        self.dropout = nn.Dropout(dropout)
        self.scaler = Scaler(rate)

    def forward(self, src):
        src = self.scaler(self.embedding(src)) + self.scaler(self.positional_embedding(src))
        return self.dropout(self.norm(src))


class ScaledDotProduct(nn.Module):
    def __init__(self, temperature):
        super().__init__()
        self.temperature = temperature

    def forward(self, q, k, v, mask=None):
        scores = q.matmul(k.transpose(-2, -1)) / self.temperature
        if mask is not None:
            scores = scores.masked_fill(mask == 0, float('-inf'))
        attn = F.softmax(scores, dim=-1)
        return torch.matmul(attn, v), attn


class MultiheadAttention(nn.Module):
    def __init__(self, embedding_size, num_heads, rate):
        super().__init__()
        self.embedding_size = embedding_size
        self.num_heads = num_heads
        self.linear_q = nn.Linear(embedding_size, embedding_size)
        self.linear_k = nn.Linear(embedding_size, embedding_size)
        self.linear_v = nn.Linear(embedding_size, embedding_size)
        self.linear_o = nn.Linear(embedding_size, embedding_size)
        self.attention = ScaledDotProduct(temperature=(embedding_size // num_heads) ** 0.5)
        self.scaler = Scaler(rate)

    def _to_heads(self, x):
        N, S, E = x.size()
        d = E // self.num_heads
        return x.reshape(N, S, self.num_heads, d).permute(0, 2, 1, 3).reshape(N * self.num_heads, S, d)

    def _from_heads(self, x):
        NH, S, d = x.size()
        N = NH // self.num_heads
        return x.reshape(N, self.num_heads, S, d).permute(0, 2, 1, 3).reshape(N, S, self.num_heads * d)

    def forward(self, q, k, v, mask=None):
        q = self.scaler(self.linear_q(q))
        k = self.scaler(self.linear_k(k))
        v = self.scaler(self.linear_v(v))
        q, k, v = self._to_heads(q), self._to_heads(k), self._to_heads(v)
        q, attn = self.attention(q, k, v, mask)
        q = self._from_heads(q)
        q = self.scaler(self.linear_o(q))
        return q, attn


class TransformerEncoderLayer(nn.Module):
    def __init__(self, embedding_size, num_heads, hidden_size, dropout, rate):
        super().__init__()
        self.mha = MultiheadAttention(embedding_size, num_heads, rate)
        self.dropout = nn.Dropout(dropout)
        self.norm1 = nn.LayerNorm(embedding_size)
        self.linear1 = nn.Linear(embedding_size, hidden_size)
        self.dropout1 = nn.Dropout(dropout)
        self.linear2 = nn.Linear(hidden_size, embedding_size)
        self.dropout2 = nn.Dropout(dropout)
        self.norm2 = nn.LayerNorm(embedding_size)
        self.scaler = Scaler(rate)
        self.activation = nn.GELU()
        self._init_param()

    def _init_param(self):
        self.linear1.weight.data.normal_(mean=0.0, std=0.02)
        self.linear2.weight.data.normal_(mean=0.0, std=0.02)
        self.norm1.weight.data.fill_(1.0)
        self.norm1.bias.data.zero_()
        self.norm2.weight.data.fill_(1.0)
        self.norm2.bias.data.zero_()

    def forward(self, src):
        attn_output, _ = self.mha(src, src, src)
        src = self.norm1(src + self.dropout(attn_output))
        src2 = self.scaler(self.linear2(self.dropout1(self.activation(self.scaler(self.linear1(src))))))
        src = self.norm2(src + self.dropout2(src2))
        return src


class Decoder(nn.Module):
    def __init__(self, num_tokens, embedding_size, rate):
        super().__init__()
        self.linear1 = nn.Linear(embedding_size, embedding_size)
        self.scaler = Scaler(rate)
        self.activation = nn.GELU()
        self.norm1 = nn.LayerNorm(embedding_size)
        self.linear2 = nn.Linear(embedding_size, num_tokens)

    def forward(self, src):
        return self.linear2(self.norm1(self.activation(self.scaler(self.linear1(src)))))


class Transformer(nn.Module):
    def __init__(self, num_tokens, bptt, embedding_size, num_heads, hidden_size,
                 num_layers, dropout, rate, mask_rate, vocab_mask):
        super().__init__()
        self.num_tokens = num_tokens
        self.mask_rate = mask_rate
        self.vocab_mask = vocab_mask
        self.transformer_embedding = TransformerEmbedding(
            num_tokens, bptt, embedding_size, dropout, rate)
        # nn.TransformerEncoder deep-copies the prototype layer, matching the
        # reference's state-dict layout transformer_encoder.layers.N.*
        encoder_layer = TransformerEncoderLayer(embedding_size, num_heads,
                                                hidden_size, dropout, rate)
        self.transformer_encoder = nn.ModuleDict(
            {'layers': nn.ModuleList([_clone_layer(encoder_layer, i) for i in range(num_layers)])})
        self.decoder = Decoder(num_tokens, embedding_size, rate)

    def forward(self, input):
        output = {}
        src = input['label'].clone()
        N, S = src.size()
        mask = torch.bernoulli(torch.full((N, S), self.mask_rate, device=src.device))
        src = src.masked_fill(mask == 1, self.num_tokens).detach()
        src = self.transformer_embedding(src)
        for layer in self.transformer_encoder['layers']:
            src = layer(src)
        out = self.decoder(src)
        out = out.permute(0, 2, 1)
        score, loss = masked_cross_entropy(
            out, input['label'],
            input.get('label_split') if self.vocab_mask else None,
            self.num_tokens)
        output['score'] = score
        output['loss'] = loss
        return output


def _clone_layer(proto, i):
    # nn.TransformerEncoder deep-copies one prototype layer, so all layers
    # start from identical weights (reference: src/models/transformer.py:141).
    import copy
    return copy.deepcopy(proto)


def transformer(cfg, model_rate=1, track=False):
    num_tokens = cfg['num_tokens']
    tcfg = cfg['transformer']
    embedding_size = int(np.ceil(model_rate * tcfg['embedding_size']))
    hidden_size = int(np.ceil(model_rate * tcfg['hidden_size']))
    scaler_rate = model_rate / cfg['global_model_rate']
    model = Transformer(num_tokens, cfg['bptt'], embedding_size,
                        tcfg['num_heads'], hidden_size, tcfg['num_layers'],
                        tcfg['dropout'], scaler_rate, cfg['mask_rate'],
                        cfg['mask'])
    model.apply(init_param)
    return model
