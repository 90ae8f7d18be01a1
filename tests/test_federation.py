"""Federation slicing and aggregation tests.

Oracle: the slicing rules at reference src/fed.py:26-159 are
exactly specifiable — these tests pin them with hand-computed index maps,
the distribute->combine round-trip property, and count-weighted averaging
identities.  (The implementation was additionally verified once against the
reference's own Federation on identical state dicts: exact match for conv,
resnet18 and transformer.)
"""
import math

import numpy as np
import pytest
import torch

from heterofl_amd.fed import Federation
from heterofl_amd.fed.federation import PREFIX, FULL, GATHER
from heterofl_amd.models import make_model
from tests.conftest import make_cfg


def _fed(cfg, model, rates=None, label_split=None):
    rates = rates if rates is not None else cfg['model_rate']
    num_users = cfg['num_users']
    if label_split is None:
        label_split = {i: list(range(10)) for i in range(num_users)}
    return Federation(model.state_dict(), rates, label_split, cfg)


# ------------------------------------------------------------------- resnet
def test_resnet_split_rules(base_cfg):
    cfg = make_cfg(base_cfg, '1_4_1_iid_fix_a1-b1-c1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, model)
    idx = fed.split_model([1])  # user 1 -> rate 0.5
    m = idx[0]
    # stem conv: out prefix ceil(64*0.5)=32, input full 3
    ps = m['conv1.weight']
    assert ps.out.kind == PREFIX and ps.out.n == 32
    assert ps.inp.kind == FULL and ps.inp.n == 3
    # first block conv1: 32 in, 32 out
    ps = m['layer1.0.conv1.weight']
    assert ps.inp.n == 32 and ps.out.n == 32
    # layer2.0 downsamples 64->128: sliced to 32->64
    ps = m['layer2.0.conv1.weight']
    assert ps.inp.n == 32 and ps.out.n == 64
    # shortcut reuses conv1's INPUT index; out follows conv2's out
    ps = m['layer2.0.shortcut.weight']
    assert ps.inp.n == 32 and ps.out.n == 64
    # classifier keeps full rows, input sliced
    ps = m['linear.weight']
    assert ps.out.kind == FULL and ps.out.n == 10
    assert ps.inp.n == 256  # ceil(512*0.5)
    assert m['linear.bias'].out.kind == FULL
    # norm weight follows the running index of its block input
    ps = m['layer1.0.n1.weight']
    assert ps.out.n == 32


def test_resnet_split_tiny_rate(base_cfg):
    # ragged widths: rate 1/16 -> ceil(64/16)=4 channels
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['global_model_rate'] = 1.0  # slice from a full-width global model
    model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, model, rates=[0.0625, 0.0625])
    m = fed.split_model([0])[0]
    assert m['conv1.weight'].out.n == 4
    assert m['layer4.1.conv2.weight'].out.n == 32  # ceil(512/16)
    assert m['linear.weight'].inp.n == 32


# --------------------------------------------------------------------- conv
def test_conv_split_rules(base_cfg):
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1-b1_bn_1_1',
                   data_name='MNIST', model_name='conv')
    model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, model)
    m = fed.split_model([1])[0]  # rate 0.5
    assert m['blocks.0.weight'].out.n == 32 and m['blocks.0.weight'].inp.n == 1
    assert m['blocks.0.bias'].out.n == 32
    # norm (BN) weight/bias follow
    assert m['blocks.2.weight'].out.n == 32
    # final linear: full out rows, sliced input
    keys = [k for k in model.state_dict() if 'weight' in k]
    ps = m[keys[-1]]
    assert ps.out.kind == FULL and ps.inp.n == 256


# -------------------------------------------------------------- transformer
def test_transformer_split_rules(base_cfg):
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1-b1_bn_1_1',
                   data_name='WikiText2', model_name='transformer',
                   num_tokens=100)
    model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, model)
    m = fed.split_model([1])[0]  # rate 0.5
    # embedding: full vocab rows, embedding dim sliced to 128
    ps = m['transformer_embedding.embedding.weight']
    assert ps.out.kind == FULL and ps.out.n == 101
    assert ps.inp.n == 128
    # positional embedding also slices the embedding dim
    ps = m['transformer_embedding.positional_embedding.positional_embedding.weight']
    assert ps.out.kind == FULL and ps.inp.n == 128
    # q/k/v: per-head slicing — head_dim 32 -> 16 per head, 8 heads
    ps = m['transformer_encoder.layers.0.mha.linear_q.weight']
    assert ps.out.kind == GATHER
    expect = torch.arange(256).reshape(8, 32)[:, :16].reshape(-1)
    assert torch.equal(ps.out.idx, expect)
    assert ps.inp.n == 128
    # after q bias the running index resets to the layer input (fed.py:147-151):
    # linear_k input must be the embedding prefix, not q's per-head gather
    ps = m['transformer_encoder.layers.0.mha.linear_k.weight']
    assert ps.inp.kind == PREFIX and ps.inp.n == 128
    # linear_o input = v's per-head out index
    ps = m['transformer_encoder.layers.0.mha.linear_o.weight']
    assert ps.inp.kind == GATHER and ps.inp.n == 128
    # decoder.linear2 keeps full vocab rows
    ps = m['decoder.linear2.weight']
    assert ps.out.kind == FULL and ps.out.n == 100
    assert m['decoder.linear2.bias'].out.kind == FULL


# ---------------------------------------------------------------- transport
@pytest.mark.parametrize('model_name,data_name', [
    ('conv', 'MNIST'), ('resnet18', 'CIFAR10'), ('resnet50', 'CIFAR10'),
    ('transformer', 'WikiText2')])
def test_distribute_shapes_match_local_model(base_cfg, model_name, data_name):
    """Each distributed slice must load into a model built at that rate."""
    cfg = make_cfg(base_cfg, '1_4_1_iid_fix_a1-b1-c1-e1_bn_1_1',
                   data_name=data_name, model_name=model_name, num_tokens=100)
    global_model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, global_model)
    user_idx = [0, 1, 2, 3]
    local_parameters, _ = fed.distribute(user_idx)
    for m, u in enumerate(user_idx):
        local = make_model(cfg, model_rate=fed.model_rate[u])
        local.load_state_dict(local_parameters[m])  # raises on shape mismatch


def test_rate1_round_trip(base_cfg):
    """distribute -> combine with a single rate-1 client is an exact
    round-trip (SURVEY §4 property 2)."""
    cfg = make_cfg(base_cfg, '1_1_1_iid_fix_a1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    model = make_model(cfg, model_rate=1)
    before = {k: v.clone() for k, v in model.state_dict().items()}
    fed = _fed(cfg, model, rates=[1.0])
    local, pidx = fed.distribute([0])
    fed.combine(local, pidx, [0])
    after = fed.global_parameters
    for k in before:
        assert torch.allclose(before[k].float(), after[k].float(), atol=1e-6), k


def test_combine_count_weighted_average(base_cfg):
    """Two clients at different rates: overlapping entries average, the
    full-width client owns the tail alone."""
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1-b1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, model, rates=[1.0, 0.5])
    user_idx = [0, 1]
    local, pidx = fed.distribute(user_idx)
    # client 0 (full) contributes 1s, client 1 (half) contributes 3s
    for m, fill in [(0, 1.0), (1, 3.0)]:
        for k in local[m]:
            if local[m][k].is_floating_point():
                local[m][k] = torch.full_like(local[m][k], fill)
    fed.combine(local, pidx, user_idx)
    w = fed.global_parameters['layer1.0.conv1.weight']  # (64,64,3,3)
    assert torch.allclose(w[:32, :32], torch.full_like(w[:32, :32], 2.0))
    assert torch.allclose(w[32:, :], torch.full_like(w[32:, :], 1.0))
    assert torch.allclose(w[:32, 32:], torch.full_like(w[:32, 32:], 1.0))


def test_combine_label_split_filtering(base_cfg):
    """Output-layer rows are aggregated only over clients holding that label
    (reference: src/fed.py:193-198)."""
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1-a1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    model = make_model(cfg, model_rate=1)
    label_split = {0: [0, 1], 1: [1, 2]}
    fed = _fed(cfg, model, rates=[1.0, 1.0], label_split=label_split)
    user_idx = [0, 1]
    local, pidx = fed.distribute(user_idx)
    before = fed.global_parameters['linear.weight'].clone()
    for m, fill in [(0, 1.0), (1, 3.0)]:
        for k in local[m]:
            if local[m][k].is_floating_point():
                local[m][k] = torch.full_like(local[m][k], fill)
    fed.combine(local, pidx, user_idx)
    w = fed.global_parameters['linear.weight']
    assert torch.allclose(w[0], torch.full_like(w[0], 1.0))   # only client 0
    assert torch.allclose(w[1], torch.full_like(w[1], 2.0))   # both -> mean
    assert torch.allclose(w[2], torch.full_like(w[2], 3.0))   # only client 1
    # rows 3..9: nobody contributed -> unchanged
    assert torch.equal(w[3:], before[3:])


def test_dynamic_resampling_with_generator(base_cfg):
    cfg = make_cfg(base_cfg, '1_20_0.5_iid_dynamic_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, model, rates=cfg['model_rate'])
    g1 = torch.Generator().manual_seed(7)
    fed.make_model_rate(generator=g1)
    r1 = list(fed.model_rate)
    g2 = torch.Generator().manual_seed(7)
    fed.make_model_rate(generator=g2)
    assert r1 == fed.model_rate
    assert set(fed.model_rate) <= {1, 0.0625}


def test_combine_tolerates_missing_clients(base_cfg):
    """A client that never reports contributes nothing; entries only it
    covered keep their previous global values (reference tolerance
    semantics, src/fed.py:180-298)."""
    import torch
    from heterofl_amd.fed.federation import Federation
    from heterofl_amd.models import make_model
    from tests.conftest import make_cfg
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    torch.manual_seed(0)
    model = make_model(cfg, model_rate=1)
    fed = Federation(model.state_dict(), cfg['model_rate'],
                     {0: list(range(10)), 1: list(range(10))}, cfg)
    user_idx = [0, 1]   # user 0 rate 1, user 1 rate 1/16
    local, pidx = fed.distribute(user_idx)
    before = {k: v.clone() for k, v in fed.global_parameters.items()}
    # only the tiny client (slot 1) reports, with modified weights
    trained = {1: {k: v + 1.0 if v.is_floating_point() else v
                   for k, v in local[1].items()}}
    tmp_d, cnt_d = fed.accumulate(trained, pidx, user_idx,
                                  slots=sorted(trained.keys()))
    fed.finalize(tmp_d, cnt_d)
    w = fed.global_parameters['conv1.weight']
    n1 = pidx[1]['conv1.weight'].out.n
    # rows the tiny client covers were updated; the rest untouched
    assert (w[:n1] - (before['conv1.weight'][:n1] + 1.0)).abs().max() < 1e-6
    assert (w[n1:] - before['conv1.weight'][n1:]).abs().max() < 1e-6


def test_combine_label_split_filtering_transformer(base_cfg):
    """Transformer output layers (embedding vocab rows + decoder.linear2
    rows incl. bias) aggregate only over clients holding that token
    (reference: src/fed.py:263-288)."""
    cfg = make_cfg(base_cfg, '1_2_1_iid_fix_a1-a1_bn_1_1',
                   data_name='WikiText2', model_name='transformer')
    cfg['transformer'] = {'embedding_size': 16, 'num_heads': 2,
                          'hidden_size': 16, 'num_layers': 1, 'dropout': 0.0}
    cfg['bptt'] = 8
    cfg['classes_size'] = 12
    cfg['num_tokens'] = 12
    model = make_model(cfg, model_rate=1)
    label_split = {0: [0, 1], 1: [1, 2]}
    fed = _fed(cfg, model, rates=[1.0, 1.0], label_split=label_split)
    user_idx = [0, 1]
    local, pidx = fed.distribute(user_idx)
    emb_key = next(k for k in fed.global_parameters
                   if k.split('.')[-2] == 'embedding' and k.endswith('weight'))
    dec_key = next(k for k in fed.global_parameters
                   if 'decoder' in k and 'linear2' in k and k.endswith('weight'))
    dec_bias = dec_key.replace('weight', 'bias')
    before = {k: fed.global_parameters[k].clone()
              for k in (emb_key, dec_key, dec_bias)}
    for m, fill in [(0, 1.0), (1, 3.0)]:
        for k in local[m]:
            if local[m][k].is_floating_point():
                local[m][k] = torch.full_like(local[m][k], fill)
    fed.combine(local, pidx, user_idx)
    for k in (emb_key, dec_key, dec_bias):
        w = fed.global_parameters[k]
        assert torch.allclose(w[0], torch.full_like(w[0], 1.0)), k
        assert torch.allclose(w[1], torch.full_like(w[1], 2.0)), k
        assert torch.allclose(w[2], torch.full_like(w[2], 3.0)), k
        assert torch.equal(w[3:], before[k][3:]), k
    # a non-output layer aggregates over BOTH clients everywhere
    q = next(k for k in fed.global_parameters if 'linear_q.weight' in k)
    wq = fed.global_parameters[q]
    assert torch.allclose(wq, torch.full_like(wq, 2.0))


def test_single_client_partial_rate_combine(base_cfg):
    """One client at rate<1: combine writes exactly that client's slice and
    leaves every untouched entry of the global tensors unchanged."""
    cfg = make_cfg(base_cfg, '1_1_1_iid_fix_a1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    model = make_model(cfg, model_rate=1)
    fed = _fed(cfg, model, rates=[0.5])
    before = {k: v.clone() for k, v in fed.global_parameters.items()}
    local, pidx = fed.distribute([0])
    filled = {k: torch.full_like(v, 5.0) if v.is_floating_point() else v
              for k, v in local[0].items()}
    fed.combine([filled], pidx, [0])
    w = fed.global_parameters['layer1.0.conv1.weight']  # (64,64,3,3)
    assert torch.allclose(w[:32, :32], torch.full_like(w[:32, :32], 5.0))
    assert torch.equal(w[32:], before['layer1.0.conv1.weight'][32:])
    assert torch.equal(w[:32, 32:], before['layer1.0.conv1.weight'][:32, 32:])


def test_count_map_matches_accumulate_counts(base_cfg):
    """Federation.count_map (the local half-payload combine, K14/VERDICT r1
    item 9) must equal the count tensors that accumulate() derives from the
    same clients — including label-split-filtered output rows."""
    import torch
    from heterofl_amd.fed.federation import Federation
    from heterofl_amd.models import make_model
    from tests.conftest import make_cfg
    for model_name, data_name in (('resnet18', 'CIFAR10'),
                                  ('conv', 'MNIST'),
                                  ('transformer', 'WikiText2')):
        cfg = make_cfg(base_cfg, '1_4_1_non-iid-2_fix_a1-e1_bn_1_1'
                       if model_name != 'transformer'
                       else '1_4_1_iid_fix_a1-e1_bn_1_1',
                       data_name=data_name, model_name=model_name)
        if model_name == 'transformer':
            cfg['num_tokens'] = 120
            cfg['bptt'] = 16
        torch.manual_seed(0)
        model = make_model(cfg, model_rate=1.0)
        if model_name == 'transformer':
            label_split = {i: list(range(0, 100, 3)) for i in range(4)}
        else:
            label_split = {0: [0, 1], 1: [2, 3], 2: [4, 5], 3: [1, 9]}
        fed = Federation(model.state_dict(), cfg['model_rate'], label_split,
                         cfg)
        user_idx = [3, 0, 2, 1]
        locals_, pidx = fed.distribute(user_idx, resample=False)
        _, cnt_acc = fed.accumulate(locals_, pidx, user_idx)
        cnt_map = fed.count_map(pidx, user_idx)
        for k in cnt_acc:
            assert torch.equal(cnt_acc[k], cnt_map[k]), (model_name, k)
