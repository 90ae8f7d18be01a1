"""GPU tests (MI355X).  Run with: pytest tests -m gpu"""
import numpy as np
import pytest
import torch

from tests.conftest import make_cfg

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason='no GPU')


@needs_gpu
def test_smoke_forward_backward(base_cfg):
    cfg = make_cfg(base_cfg, '1_10_0.2_iid_fix_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    from heterofl_amd.models import make_model
    model = make_model(cfg).to('cuda:0')
    model.train(True)
    x = {'img': torch.randn(10, 3, 32, 32, device='cuda:0'),
         'label': torch.randint(0, 10, (10,), device='cuda:0'),
         'label_split': torch.arange(10, device='cuda:0')}
    out = model(x)
    out['loss'].backward()
    torch.cuda.synchronize()
    assert torch.isfinite(out['loss'])


@needs_gpu
def test_batched_equivalence_gpu(base_cfg):
    """Batched grouped-model step == per-client steps, on device."""
    from heterofl_amd.fed.batched import (BatchedResNet, pack_states,
                                          unpack_states, batched_masked_ce,
                                          per_client_clip_)
    from heterofl_amd.models import make_model
    from heterofl_amd.models.functional import masked_cross_entropy
    cfg = make_cfg(base_cfg, '1_3_1_iid_fix_b1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['global_model_rate'] = 1.0
    R, rate, lr, steps, n = 3, 0.5, 0.1, 2, 6
    dev = 'cuda:0'
    locals_ = []
    for i in range(R):
        torch.manual_seed(100 + i)
        locals_.append(make_model(cfg, model_rate=rate).to(dev))
    torch.manual_seed(0)
    data = [torch.randn(n, R, 3, 32, 32, device=dev) for _ in range(steps)]
    labels = [torch.randint(0, 10, (n, R), device=dev) for _ in range(steps)]
    seq_states = []
    for r in range(R):
        m = locals_[r]
        m.train(True)
        opt = torch.optim.SGD(m.parameters(), lr=lr, momentum=0.9,
                              weight_decay=5e-4)
        for s in range(steps):
            opt.zero_grad()
            score = m.linear(m.features(data[s][:, r]))
            _, loss = masked_cross_entropy(score, labels[s][:, r], None, 10)
            loss.backward()
            torch.nn.utils.clip_grad_norm_(m.parameters(), 1)
            opt.step()
        seq_states.append(m.state_dict())
    locals2 = []
    for i in range(R):
        torch.manual_seed(100 + i)
        locals2.append(make_model(cfg, model_rate=rate).state_dict())
    hidden = [int(np.ceil(rate * h)) for h in cfg['resnet']['hidden_size']]
    bm = BatchedResNet(R, [3, 32, 32], hidden, [2, 2, 2, 2], 10, rate,
                       'bn', True).to(dev)
    pack_states(bm, locals2)
    bm.train(True)
    params = list(bm.parameters())
    opt = torch.optim.SGD(params, lr=lr, momentum=0.9, weight_decay=5e-4)
    for s in range(steps):
        xb = data[s].reshape(n, R * 3, 32, 32)
        opt.zero_grad()
        scores = bm(xb)
        losses = batched_masked_ce(scores, labels[s], None)
        losses.sum().backward()
        per_client_clip_(params, R, 1.0)
        opt.step()
    outs = unpack_states(bm, list(seq_states[0].keys()))
    for r in range(R):
        for k in seq_states[r]:
            a, b = seq_states[r][k].cpu(), outs[r][k].cpu()
            diff = (a - b).abs().max().item()
            assert diff / max(a.abs().max().item(), 1.0) < 5e-4, (r, k, diff)


@needs_gpu
def test_bf16_batched_step(base_cfg):
    from heterofl_amd.fed.batched import BatchedResNet, batched_masked_ce
    dev = 'cuda:0'
    bm = BatchedResNet(4, [3, 32, 32], [64, 128, 256, 512], [2, 2, 2, 2],
                       10, 1.0, 'bn', True).to(dev)
    xb = torch.randn(10, 12, 32, 32, device=dev)
    yb = torch.randint(0, 10, (10, 4), device=dev)
    with torch.autocast('cuda', torch.bfloat16):
        scores = bm(xb)
        losses = batched_masked_ce(scores.float(), yb, None)
    losses.sum().backward()
    torch.cuda.synchronize()
    assert all(torch.isfinite(l) for l in losses)


@needs_gpu
def test_graph_path_matches_eager(base_cfg):
    """The hipGraph-captured step must train identically to the eager
    batched path (same device RNG seed -> same data order/augmentation)."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed.batched import BatchedClientTrainer
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset
    cfg = make_cfg(base_cfg, '1_4_1_iid_fix_a1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['device'] = 'cuda:0'
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = 'float32'
    cfg['num_epochs'] = {'global': 1, 'local': 2}
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=160)
    process_dataset(ds, cfg)
    torch.manual_seed(1)
    data_split, label_split = split_dataset(ds, 4, 'iid', 10)
    model = make_model(cfg).to('cuda:0')
    gp = model.state_dict()
    user_idx = [0, 1, 2, 3]
    locals_ = [{k: v.clone() for k, v in gp.items()} for _ in user_idx]
    rates = [1.0] * 4

    results = {}
    for use_graph in (False, True):
        cfg2 = dict(cfg)
        cfg2['hip_graphs'] = use_graph
        tr = BatchedClientTrainer(cfg2)
        tr.set_data(ds, data_split)
        torch.manual_seed(42)
        torch.cuda.manual_seed_all(42)
        out = tr.train_clients(list(range(4)), user_idx,
                               [dict(l) for l in locals_], rates,
                               None, label_split, 0.1)
        results[use_graph] = dict(out)
    for m in range(4):
        for k in results[False][m]:
            a = results[False][m][k].float().cpu()
            b = results[True][m][k].float().cpu()
            diff = (a - b).abs().max().item()
            assert diff < 1e-4, (m, k, diff)


@needs_gpu
def test_fed_round_gpu(base_cfg):
    """One full federated round on GPU with the batched engine."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset, make_optimizer
    cfg = make_cfg(base_cfg, '1_10_0.3_iid_fix_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['device'] = 'cuda:0'
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = 'bfloat16'
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=400)
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 10, 'iid', cfg['classes_size'])
    model = make_model(cfg).to('cuda:0')
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)
    runner.train_round(1)
    for v in runner.federation.global_parameters.values():
        if v.is_floating_point():
            assert torch.isfinite(v).all()


# ---------------------------------------------------------- native kernels
@needs_gpu
def test_fused_bn_relu_matches_torch(base_cfg):
    """HIP fused sBN+ReLU fwd/bwd vs plain torch fp32 reference."""
    from heterofl_amd.ops.fused import fused_norm_relu
    torch.manual_seed(0)
    for N, C, HW in [(10, 64, 32 * 32), (10, 2560, 16), (3, 20, 49)]:
        x = torch.randn(N, C, int(HW ** 0.5) if int(HW ** 0.5) ** 2 == HW else 1,
                        HW // (int(HW ** 0.5) if int(HW ** 0.5) ** 2 == HW else 1),
                        device='cuda', requires_grad=True)
        w = (torch.rand(C, device='cuda') + 0.5).requires_grad_()
        b = torch.randn(C, device='cuda', requires_grad=True)
        y = fused_norm_relu(x, w, b, 'bn', 0)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        import torch.nn.functional as F
        ref = F.relu(F.batch_norm(x2, None, None, w2, b2, training=True,
                                  eps=1e-5))
        assert (y - ref).abs().max().item() < 1e-4
        g = torch.randn_like(y)
        y.backward(g)
        ref.backward(g)
        assert (x.grad - x2.grad).abs().max().item() < 1e-4, (N, C, HW)
        assert (w.grad - w2.grad).abs().max().item() < 2e-3
        assert (b.grad - b2.grad).abs().max().item() < 2e-3


@needs_gpu
def test_fused_gn_relu_matches_torch(base_cfg):
    from heterofl_amd.ops.fused import fused_norm_relu
    import torch.nn.functional as F
    torch.manual_seed(0)
    for N, C, H, G in [(10, 40, 8, 20), (10, 64, 16, 4), (4, 16, 4, 16)]:
        x = torch.randn(N, C, H, H, device='cuda', requires_grad=True)
        w = (torch.rand(C, device='cuda') + 0.5).requires_grad_()
        b = torch.randn(C, device='cuda', requires_grad=True)
        y = fused_norm_relu(x, w, b, 'gn', G)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        b2 = b.detach().clone().requires_grad_(True)
        ref = F.relu(F.group_norm(x2, G, w2, b2, eps=1e-5))
        assert (y - ref).abs().max().item() < 1e-4
        g = torch.randn_like(y)
        y.backward(g)
        ref.backward(g)
        assert (x.grad - x2.grad).abs().max().item() < 1e-4, (N, C, G)
        assert (w.grad - w2.grad).abs().max().item() < 2e-3
        assert (b.grad - b2.grad).abs().max().item() < 2e-3


@needs_gpu
def test_fused_masked_ce_matches_torch(base_cfg):
    from heterofl_amd.ops.fused import fused_masked_ce
    import torch.nn.functional as F
    torch.manual_seed(0)
    N, R, C = 10, 5, 10
    scores = torch.randn(N, R, C, device='cuda', requires_grad=True)
    labels = torch.randint(0, C, (N, R), device='cuda')
    mask = (torch.rand(R, C, device='cuda') > 0.3).float()
    for r in range(R):  # labels must be unmasked (reference invariant)
        mask[r, labels[:, r]] = 1
    metrics = torch.zeros(R, 3, device='cuda')
    losses = fused_masked_ce(scores, labels, mask, metrics)
    s2 = scores.detach().clone().requires_grad_(True)
    ref_masked = s2.masked_fill(mask.unsqueeze(0) == 0, 0)
    logp = F.log_softmax(ref_masked, dim=2)
    ref = -logp.gather(2, labels.unsqueeze(2)).squeeze(2).mean(0)
    assert (losses - ref).abs().max().item() < 1e-5
    losses.sum().backward()
    ref.sum().backward()
    assert (scores.grad - s2.grad).abs().max().item() < 1e-5
    # metrics: loss sums and counts
    assert (metrics[:, 0] - ref.detach() * N).abs().max().item() < 1e-4
    assert (metrics[:, 2] == N).all()
    correct = (ref_masked.argmax(2) == labels).float().sum(0)
    assert (metrics[:, 1] - correct).abs().max().item() < 1e-4


@needs_gpu
def test_fused_clip_sgd_matches_torch(base_cfg):
    from heterofl_amd.ops.fused import FusedClipSGD
    torch.manual_seed(0)
    R, lr, mom, wd = 3, 0.1, 0.9, 5e-4
    shapes = [(R * 8, 4, 3, 3), (R * 8,), (R, 10, 8), (R, 10)]
    params = [torch.randn(*s, device='cuda') for s in shapes]
    grads = [torch.randn(*s, device='cuda') for s in shapes]
    bufs = [torch.zeros_like(p) for p in params]
    # torch reference: per-client clip + SGD
    ref_p = [p.clone() for p in params]
    ref_b = [b.clone() for b in bufs]
    ref_g = [g.clone() for g in grads]
    sq = None
    for g in ref_g:
        s = (g.view(R, -1) ** 2).sum(1)
        sq = s if sq is None else sq + s
    scale = (1.0 / (sq.sqrt() + 1e-6)).clamp(max=1.0)
    for i, g in enumerate(ref_g):
        g.view(R, -1).mul_(scale.unsqueeze(1))
        g.add_(ref_p[i], alpha=wd)
        ref_b[i].mul_(mom).add_(g)
        ref_p[i].add_(ref_b[i], alpha=-lr)
    opt = FusedClipSGD(params, grads, bufs, R, torch.device('cuda'))
    for step in range(2):
        opt.step(1.0, lr, mom, wd)
        if step == 0:
            for i in range(len(params)):
                assert (params[i] - ref_p[i]).abs().max().item() < 1e-5, i
                assert (bufs[i] - ref_b[i]).abs().max().item() < 1e-5, i
    torch.cuda.synchronize()


@needs_gpu
def test_mfma_probe_layout():
    """Pin the 16x16x32 bf16 MFMA fragment layout empirically (asymmetric
    operands catch transposes, cdna guide §5.4 rule 16)."""
    from heterofl_amd.ops import require_native
    ext = require_native()
    torch.manual_seed(0)
    A = torch.randn(16, 32)
    B = torch.randn(32, 16)
    D = ext.mfma_probe(A, B).cpu()
    ref = (A.to(torch.bfloat16).float() @ B.to(torch.bfloat16).float())
    assert (D - ref).abs().max().item() < 0.15, \
        (D - ref).abs().max().item()


@needs_gpu
def test_mfma_conv_matches_torch():
    """MFMA grouped conv fwd/bwd vs F.conv2d across every HeteroFL shape
    family (3x3 s1/s2, 1x1 s2, stem, ragged channels)."""
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import grouped_conv
    torch.manual_seed(0)
    shapes = [
        # (G, N, Cin, H, Cout, k, stride, pad)
        (5, 10, 3, 32, 64, 3, 1, 1),      # stem
        (5, 10, 64, 32, 64, 3, 1, 1),     # layer1
        (5, 10, 64, 32, 128, 3, 2, 1),    # downsample
        (5, 10, 64, 32, 128, 1, 2, 0),    # 1x1 shortcut
        (5, 10, 512, 4, 512, 3, 1, 1),    # layer4
        (3, 10, 4, 32, 4, 3, 1, 1),       # rate-1/16 ragged
        (2, 7, 20, 16, 36, 3, 2, 1),      # odd everything
    ]
    for G, N, Cin, H, Cout, k, s, p in shapes:
        x = torch.randn(N, G * Cin, H, H, device='cuda', requires_grad=True)
        w = (torch.randn(G * Cout, Cin, k, k, device='cuda') * 0.1
             ).requires_grad_()
        y = grouped_conv(x, w, None, G, s, p)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        ref = F.conv2d(x2, w2, None, stride=s, padding=p, groups=G)
        ferr = (y - ref).abs().max().item()
        assert ferr < 5e-4 * Cin * k, ('fwd', G, Cin, H, Cout, k, s, ferr)
        g = torch.randn_like(y)
        y.backward(g)
        ref.backward(g)
        derr = (x.grad - x2.grad).abs().max().item()
        werr = (w.grad - w2.grad).abs().max().item()
        assert derr < 5e-4 * Cout * k, ('bwd_data', G, Cin, H, Cout, s, derr)
        assert werr < 5e-3 * N * H, ('bwd_w', G, Cin, H, Cout, s, werr)


@needs_gpu
def test_mfma_conv_bias_bf16():
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import grouped_conv
    torch.manual_seed(0)
    G, N, Cin, H, Cout = 3, 10, 16, 28, 32
    x = torch.randn(N, G * Cin, H, H, device='cuda', dtype=torch.bfloat16)
    w = torch.randn(G * Cout, Cin, 3, 3, device='cuda') * 0.1
    b = torch.randn(G * Cout, device='cuda')
    y = grouped_conv(x, w, b, G, 1, 1)
    ref = F.conv2d(x.float(), w, b, stride=1, padding=1, groups=G)
    rel = (y.float() - ref).abs().max().item() / ref.abs().max().item()
    assert y.dtype == torch.bfloat16
    assert rel < 0.05, rel


@needs_gpu
def test_mfma_conv_residual_fusion():
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import grouped_conv
    torch.manual_seed(0)
    G, N, C, H = 4, 10, 32, 16
    x = torch.randn(N, G * C, H, H, device='cuda', requires_grad=True)
    w = (torch.randn(G * C, C, 3, 3, device='cuda') * 0.1).requires_grad_()
    res = torch.randn(N, G * C, H, H, device='cuda', requires_grad=True)
    y = grouped_conv(x, w, None, G, 1, 1, residual=res)
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    r2 = res.detach().clone().requires_grad_(True)
    ref = F.conv2d(x2, w2, None, 1, 1, groups=G) + r2
    assert (y - ref).abs().max().item() < 0.05
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g)
    assert (res.grad - r2.grad).abs().max().item() < 1e-6
    assert (x.grad - x2.grad).abs().max().item() < 0.05


@needs_gpu
def test_fused_head_matches_torch():
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import fused_head
    torch.manual_seed(0)
    N, R, C, H, J = 10, 5, 512, 4, 10
    feat = torch.randn(N, R * C, H, H, device='cuda', requires_grad=True)
    w = (torch.randn(R, J, C, device='cuda') * 0.05).requires_grad_()
    b = torch.randn(R, J, device='cuda', requires_grad=True)
    scores = fused_head(feat, w, b, R)
    f2 = feat.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    pooled = F.adaptive_avg_pool2d(f2, 1).view(N, R, C)
    ref = torch.einsum('nri,roi->nro', pooled, w2) + b2
    assert (scores - ref).abs().max().item() < 1e-4
    g = torch.randn_like(scores)
    scores.backward(g)
    ref.backward(g)
    assert (feat.grad - f2.grad).abs().max().item() < 1e-5
    assert (w.grad - w2.grad).abs().max().item() < 1e-4
    assert (b.grad - b2.grad).abs().max().item() < 1e-5


@needs_gpu
def test_lm_graph_matches_eager(base_cfg):
    """LM hipGraph path == eager batched path (RNG-free config)."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed.batched_lm_trainer import BatchedLMClientTrainer
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset
    cfg = make_cfg(base_cfg, '1_3_1_iid_fix_a1_bn_1_1',
                   data_name='WikiText2', model_name='transformer')
    cfg['device'] = 'cuda:0'
    cfg['compute_dtype'] = 'float32'
    cfg['num_epochs'] = {'global': 1, 'local': 2}
    cfg['transformer']['dropout'] = 0.0
    cfg['mask_rate'] = 0.0
    torch.manual_seed(0)
    ds = fetch_dataset('WikiText2', synthetic=True, synthetic_size=60_000)
    ds['train'].vocab.itos = ds['train'].vocab.itos[:500]
    ds['train'].token = ds['train'].token % 500
    ds['test'].token = ds['test'].token % 500
    process_dataset(ds, cfg)
    cfg['num_tokens'] = 500
    data_split, label_split = split_dataset(ds, 3, 'iid')
    torch.manual_seed(1)
    gm = make_model(cfg).to('cuda:0')
    gp = gm.state_dict()
    locals_ = [{k: v.clone() for k, v in gp.items()} for _ in range(3)]
    user_idx = [0, 1, 2]
    rates = {u: 1.0 for u in user_idx}
    results = {}
    for use_graph in (False, True):
        cfg2 = dict(cfg)
        cfg2['hip_graphs'] = use_graph
        tr = BatchedLMClientTrainer(cfg2)
        tr.set_data(ds, data_split)
        torch.manual_seed(3)
        torch.cuda.manual_seed_all(3)
        out = tr.train_clients([0, 1, 2], user_idx,
                               [dict(l) for l in locals_], rates, None,
                               label_split, 0.1)
        results[use_graph] = dict(out)
    for m in range(3):
        for k in results[False][m]:
            a = results[False][m][k].float().cpu()
            b = results[True][m][k].float().cpu()
            diff = (a - b).abs().max().item()
            assert diff < 1e-4, (m, k, diff)


@needs_gpu
def test_fp8_conv_close_to_fp32():
    """fp8-MFMA conv (e4m3 x e4m3) tracks the fp32 reference within fp8
    quantization error; bwd paths (bf8/e4m3 mixed MFMA) stay finite and
    directionally correct."""
    import torch.nn.functional as F
    from heterofl_amd.ops import require_native, set_fp8
    from heterofl_amd.ops.fused import grouped_conv
    ext = require_native()
    torch.manual_seed(0)
    G, N, C, H = 3, 8, 32, 16
    x = torch.randn(N, G * C, H, H, device='cuda',
                    dtype=torch.bfloat16, requires_grad=True)
    w = (torch.randn(G * C, C, 3, 3, device='cuda') * 0.1).requires_grad_()
    try:
        set_fp8(True)
        y = grouped_conv(x, w, None, G, 1, 1)
        ref = F.conv2d(x.float(), w, None, 1, 1, groups=G)
        rel = (y.float() - ref).abs().mean().item() / ref.abs().mean().item()
        assert rel < 0.08, rel   # e4m3 quantization class error
        g = torch.randn_like(y)
        y.backward(g)
        assert torch.isfinite(x.grad.float()).all()
        assert torch.isfinite(w.grad).all()
        # grad direction agrees with fp32 reference
        x2 = x.detach().float().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        F.conv2d(x2, w2, None, 1, 1, groups=G).backward(g.float())
        cos = torch.nn.functional.cosine_similarity(
            w.grad.flatten(), w2.grad.flatten(), dim=0).item()
        assert cos > 0.98, cos
    finally:
        set_fp8(False)


@needs_gpu
def test_fused_attention_matches_torch():
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import fused_attention
    torch.manual_seed(0)
    for B, S, d in [(40, 64, 32), (12, 64, 16), (8, 24, 32), (6, 64, 2)]:
        q = torch.randn(B, S, d, device='cuda', requires_grad=True)
        k = torch.randn(B, S, d, device='cuda', requires_grad=True)
        v = torch.randn(B, S, d, device='cuda', requires_grad=True)
        temp = d ** 0.5
        out = fused_attention(q, k, v, temp)
        q2 = q.detach().clone().requires_grad_(True)
        k2 = k.detach().clone().requires_grad_(True)
        v2 = v.detach().clone().requires_grad_(True)
        ref = torch.bmm(F.softmax(torch.bmm(q2, k2.transpose(1, 2)) / temp,
                                  dim=-1), v2)
        assert (out - ref).abs().max().item() < 0.03, (B, S, d)
        g = torch.randn_like(out)
        out.backward(g)
        ref.backward(g)
        for a, b, name in [(q.grad, q2.grad, 'dq'), (k.grad, k2.grad, 'dk'),
                           (v.grad, v2.grad, 'dv')]:
            assert (a - b).abs().max().item() < 0.05, (B, S, d, name)


@needs_gpu
def test_fused_layernorm_matches_torch():
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import fused_layernorm
    torch.manual_seed(0)
    R, B, S, E = 5, 3, 64, 256
    x = torch.randn(R, B, S, E, device='cuda', requires_grad=True)
    w = (torch.rand(R, E, device='cuda') + 0.5).requires_grad_()
    b = torch.randn(R, E, device='cuda', requires_grad=True)
    y = fused_layernorm(x, w, b, R)
    x2 = x.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    xhat = F.layer_norm(x2, (E,), None, None, 1e-5)
    ref = xhat * w2.view(R, 1, 1, E) + b2.view(R, 1, 1, E)
    assert (y - ref).abs().max().item() < 1e-4
    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g)
    assert (x.grad - x2.grad).abs().max().item() < 1e-4
    assert (w.grad - w2.grad).abs().max().item() < 2e-3
    assert (b.grad - b2.grad).abs().max().item() < 2e-3


@needs_gpu
def test_fused_lm_ce_matches_torch():
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import fused_lm_ce
    torch.manual_seed(0)
    R, B, S, V = 4, 2, 16, 1000
    logits = torch.randn(R, B, S, V, device='cuda', requires_grad=True)
    tokens = torch.randint(0, V, (R, B, S), device='cuda')
    mask = (torch.rand(R, V, device='cuda') > 0.2).float()
    for r in range(R):
        mask[r].scatter_(0, tokens[r].reshape(-1), 1.0)
    losses = fused_lm_ce(logits, tokens, mask, R)
    l2 = logits.detach().clone().requires_grad_(True)
    masked = l2.masked_fill(mask.view(R, 1, 1, V) == 0, 0)
    logp = F.log_softmax(masked, dim=-1)
    ref = -logp.gather(3, tokens.unsqueeze(3)).squeeze(3).reshape(R, -1).mean(1)
    assert (losses - ref).abs().max().item() < 1e-5
    losses.sum().backward()
    ref.sum().backward()
    assert (logits.grad - l2.grad).abs().max().item() < 1e-5


@needs_gpu
def test_native_stack_converges(base_cfg):
    """End-to-end: the full native hipGraph stack (MFMA conv, fused norms,
    fused CE, fused clip+SGD) reduces global loss on a learnable synthetic
    problem over a few federated rounds."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset, make_optimizer
    cfg = make_cfg(base_cfg, '1_4_1_iid_fix_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['device'] = 'cuda:0'
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = 'bfloat16'
    cfg['num_epochs'] = {'global': 3, 'local': 2}
    cfg['lr'] = 0.02
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=200)
    # learnable: label = mean-brightness bucket
    img = ds['train'].img.float()
    ds['train'].target = (img.reshape(200, -1).mean(1) * 10 / 256
                          ).long().clamp(0, 9).tolist()
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 4, 'iid', 10)
    model = make_model(cfg).to('cuda:0')
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)

    def global_loss():
        model.load_state_dict(runner.federation.global_parameters)
        model.train(True)
        with torch.no_grad():
            batch = {'img': torch.stack([ds['train'][i]['img']
                                         for i in range(200)]).to('cuda:0'),
                     'label': torch.tensor(ds['train'].target,
                                           device='cuda:0')}
            return model(batch)['loss'].item()

    l0 = global_loss()
    for ep in range(1, 4):
        runner.train_round(ep)
    l1 = global_loss()
    assert l1 < l0, (l0, l1)


@needs_gpu
def test_noniid_dynamic_gn_round_gpu(base_cfg):
    """BASELINE config-3 shape on one GPU: non-IID-2 dynamic a1-e1 with
    GroupNorm on the native batched stack."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset, make_optimizer
    cfg = make_cfg(base_cfg, '1_10_0.3_non-iid-2_dynamic_a1-e1_gn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['device'] = 'cuda:0'
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = 'bfloat16'
    cfg['num_epochs'] = {'global': 2, 'local': 1}
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=400)
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 10, 'non-iid-2', 10)
    model = make_model(cfg).to('cuda:0')
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)
    for ep in (1, 2):
        runner.train_round(ep)
    for v in runner.federation.global_parameters.values():
        if v.is_floating_point():
            assert torch.isfinite(v).all()


@needs_gpu
def test_stats_and_eval_paths_gpu(base_cfg):
    """sBN stats pass + Local/Global evaluation run on GPU (large-batch
    stats/eval paths)."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.logger import Logger
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset, make_optimizer
    import tempfile
    cfg = make_cfg(base_cfg, '1_4_0.5_iid_fix_a1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['device'] = 'cuda:0'
    cfg['engine'] = 'batched'
    cfg['num_epochs'] = {'global': 1, 'local': 1}
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=80)
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 4, 'iid', 10)
    model = make_model(cfg).to('cuda:0')
    opt = make_optimizer(model, cfg['lr'], cfg)
    logger = Logger(tempfile.mkdtemp())
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt,
                       logger=logger)
    logger.safe(True)
    runner.train_round(1)
    tm = runner.stats()
    assert any('running_mean' in k for k in tm.state_dict())
    runner.test(tm, 1)
    logger.safe(False)
    assert 'test/Global-Accuracy' in logger.mean


@needs_gpu
def test_mfma_conv_bottleneck_shapes():
    """Bottleneck-specific conv geometries (1x1 s1 with 4x expansion /
    contraction, wide 1x1 s2 shortcuts) vs F.conv2d — the shapes the
    grouped BBottleneck engine adds over resnet18/34
    (fed/batched.py::BBottleneck; ROUND2.md item 6)."""
    import torch.nn.functional as F
    from heterofl_amd.ops.fused import grouped_conv
    torch.manual_seed(0)
    shapes = [
        # (G, N, Cin, H, Cout, k, stride, pad)
        (3, 10, 64, 32, 256, 1, 1, 0),    # conv3 expansion (L1)
        (3, 10, 256, 32, 64, 1, 1, 0),    # conv1 contraction (L1)
        (3, 10, 256, 16, 512, 1, 2, 0),   # shortcut s2 (L2)
        (2, 10, 512, 4, 2048, 1, 1, 0),   # conv3 expansion (L4)
        (2, 10, 2048, 4, 512, 1, 1, 0),   # conv1 contraction (L4)
        (3, 10, 4, 32, 16, 1, 1, 0),      # rate-1/16 ragged expansion
    ]
    for G, N, Cin, H, Cout, k, s, p in shapes:
        x = torch.randn(N, G * Cin, H, H, device='cuda', requires_grad=True)
        w = (torch.randn(G * Cout, Cin, k, k, device='cuda') * 0.1
             ).requires_grad_()
        y = grouped_conv(x, w, None, G, s, p)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        ref = F.conv2d(x2, w2, None, stride=s, padding=p, groups=G)
        ferr = (y - ref).abs().max().item()
        assert ferr < 5e-4 * Cin * k, ('fwd', G, Cin, H, Cout, k, s, ferr)
        g = torch.randn_like(y)
        y.backward(g)
        ref.backward(g)
        derr = (x.grad - x2.grad).abs().max().item()
        werr = (w.grad - w2.grad).abs().max().item()
        assert derr < 5e-4 * Cout * k, ('bwd_data', G, Cin, H, Cout, s, derr)
        assert werr < 5e-3 * N * H, ('bwd_w', G, Cin, H, Cout, s, werr)


@needs_gpu
def test_batched_bottleneck_block_gpu_matches_cpu():
    """BBottleneck forward/backward on the native GPU path vs the same
    module's CPU (torch oracle) path."""
    from heterofl_amd.fed.batched import BBottleneck
    torch.manual_seed(0)
    for stride in (1, 2):
        blk = BBottleneck(3, 32, 16, stride, 1.0, 'bn', True)
        blk_gpu = BBottleneck(3, 32, 16, stride, 1.0, 'bn', True).cuda()
        blk_gpu.load_state_dict(blk.state_dict())
        blk.train(True)
        blk_gpu.train(True)
        x = torch.randn(8, 3 * 32, 16, 16, requires_grad=True)
        xg = x.detach().clone().cuda().requires_grad_(True)
        y = blk(x)
        yg = blk_gpu(xg)
        ferr = (y - yg.cpu()).abs().max().item()
        assert ferr < 2e-2, (stride, ferr)
        g = torch.randn_like(y)
        y.backward(g)
        yg.backward(g.cuda())
        derr = (x.grad - xg.grad.cpu()).abs().max().item()
        assert derr < 2e-2, (stride, derr)
        for (k, p), (kg, pg) in zip(blk.named_parameters(),
                                    blk_gpu.named_parameters()):
            perr = (p.grad - pg.grad.cpu()).abs().max().item()
            scale = p.grad.abs().max().item() + 1e-6
            assert perr / max(scale, 1.0) < 5e-2, (stride, k, perr, scale)


@needs_gpu
def test_resnet50_fed_round_batched_default(base_cfg):
    """A resnet50 federated round runs on the batched BBottleneck engine by
    default (no env opt-in) and reduces global loss on a learnable synthetic
    problem."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.fed.batched import BatchedClientTrainer
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset, make_optimizer
    cfg = make_cfg(base_cfg, '1_4_1_iid_fix_a1-e1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet50')
    cfg['device'] = 'cuda:0'
    cfg['engine'] = 'batched'
    cfg['compute_dtype'] = 'bfloat16'
    cfg['num_epochs'] = {'global': 2, 'local': 1}
    cfg['lr'] = 0.02
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=120)
    img = ds['train'].img.float()
    ds['train'].target = (img.reshape(120, -1).mean(1) * 10 / 256
                          ).long().clamp(0, 9).tolist()
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 4, 'iid', 10)
    model = make_model(cfg).to('cuda:0')
    opt = make_optimizer(model, cfg['lr'], cfg)
    runner = FedRunner(cfg, ds, data_split, label_split, model, opt)
    assert isinstance(runner.trainer, BatchedClientTrainer)

    def global_loss():
        model.load_state_dict(runner.federation.global_parameters)
        model.train(True)
        with torch.no_grad():
            batch = {'img': torch.stack([ds['train'][i]['img']
                                         for i in range(120)]).to('cuda:0'),
                     'label': torch.tensor(ds['train'].target,
                                           device='cuda:0')}
            return model(batch)['loss'].item()

    l0 = global_loss()
    for ep in range(1, 3):
        runner.train_round(ep)
    l1 = global_loss()
    assert l1 < l0, (l0, l1)


@needs_gpu
def test_fused_head_large_batch():
    """The n-tiled head forward handles eval/stats batch sizes (N=500) that
    overflowed the old (N, C) LDS slab and poisoned the queue (r2 fix)."""
    from heterofl_amd.ops.fused import fused_head
    for C, N in [(512, 500), (64, 500), (512, 10), (512, 37)]:
        x = torch.randn(N, C, 4, 4, device='cuda:0')
        w = torch.randn(1, 10, C, device='cuda:0')
        b = torch.randn(1, 10, device='cuda:0')
        s = fused_head(x, w, b, 1)
        pooled = torch.nn.functional.adaptive_avg_pool2d(x, 1).view(N, 1, C)
        ref = torch.einsum('nrc,rjc->nrj', pooled, w) + b
        assert (s.float() - ref).abs().max().item() < 1e-2, (C, N)
        # canary: the queue must still accept launches after the head
        torch.zeros(8, device='cuda:0').add_(1.0)
        torch.cuda.synchronize()


@needs_gpu
def test_lm_bf16_shadow_weights():
    """bf16 shadow weights: the fused optimizer keeps every shadow equal to
    bf16(master) across steps, and the shadow forward matches the autocast
    forward at init."""
    from heterofl_amd.config import default_config
    from heterofl_amd.control import process_control, CONTROL_FIELDS
    from heterofl_amd.fed.batched_lm import (make_batched_transformer,
                                             enable_bf16_shadows,
                                             refresh_shadows, lm_masked_ce)
    from heterofl_amd.ops.fused import FusedClipSGD
    cfg = default_config()
    control = '1_4_1_iid_fix_a1_bn_1_1'
    cfg['control'] = dict(zip(CONTROL_FIELDS, control.split('_')))
    cfg['control_name'] = control
    cfg['data_name'] = 'WikiText2'
    cfg['model_name'] = 'transformer'
    cfg['device'] = 'cuda:0'
    process_control(cfg)
    cfg['num_tokens'] = 300
    torch.manual_seed(0)
    model = make_batched_transformer(cfg, 1.0, 2).to('cuda:0')
    model.train(True)
    smap = enable_bf16_shadows(model)
    refresh_shadows(model)
    tokens = torch.randint(0, 300, (2, 3, cfg['bptt']), device='cuda:0')
    # shadow forward ~ autocast forward (same bf16 weight values)
    torch.manual_seed(1)
    logits_sh = model(tokens)
    assert logits_sh.dtype == torch.bfloat16
    params = [p for p in model.parameters() if p.requires_grad]
    for p in params:
        p.grad = torch.zeros_like(p)
    bufs = [torch.zeros_like(p) for p in params]
    fopt = FusedClipSGD(params, [p.grad for p in params], bufs, 2, 'cuda:0',
                        shadows=[smap.get(id(p)) for p in params])
    losses = lm_masked_ce(logits_sh, tokens, None)
    losses.sum().backward()
    fopt.step(1.0, 0.1, 0.9, 5e-4)
    torch.cuda.synchronize()
    # every shadow must equal bf16(master) after the fused step
    for p in params:
        sh = smap.get(id(p))
        if sh is not None:
            assert torch.equal(sh, p.detach().to(torch.bfloat16)), 'stale shadow'
    assert torch.isfinite(losses).all()


@needs_gpu
def test_fused_gelu_res_dropout():
    """Fused scaler+GELU+dropout and scaler+dropout+residual kernels vs the
    eager torch chain: exact at p=0; at p>0, the saved mask must reproduce
    forward and backward bit-consistently and keep ~1-p of elements."""
    from heterofl_amd.ops.fused import fused_gelu_dropout, fused_res_dropout
    import torch.nn.functional as F
    dev = 'cuda:0'
    torch.manual_seed(0)
    for dt in (torch.float32, torch.bfloat16):
        x = torch.randn(4, 3, 64, 128, device=dev, dtype=dt,
                        requires_grad=True)
        x2 = x.detach().clone().requires_grad_(True)
        rate = 0.5
        y = fused_gelu_dropout(x, rate, 0.0)
        ref = F.gelu(x2 / rate)
        tol = 2e-2 if dt == torch.bfloat16 else 1e-5
        assert (y.float() - ref.float()).abs().max().item() < tol
        g = torch.randn_like(y)
        y.backward(g)
        ref.backward(g)
        err = (x.grad.float() - x2.grad.float()).abs().max().item()
        assert err < tol * 4, (dt, err)
        # residual fusion, p=0
        s = torch.randn(4, 3, 64, 128, device=dev, dtype=dt,
                        requires_grad=True)
        h = torch.randn_like(s).requires_grad_(True)
        t = fused_res_dropout(s, h, rate, 0.0)
        assert (t.float() - (s + h / rate).float()).abs().max().item() < tol
        t.sum().backward()
        assert (h.grad.float() - torch.full_like(h, 1 / rate).float()) \
            .abs().max().item() < tol
    # dropout statistics + fwd/bwd mask consistency
    x = torch.randn(200_000, device=dev, requires_grad=True)
    p = 0.2
    y = fused_gelu_dropout(x, 1.0, p)
    kept = (y != 0).float().mean().item()
    assert abs(kept - (1 - p)) < 0.02, kept
    y.sum().backward()
    # dropped positions must get exactly zero gradient
    zero_out = (y == 0)
    assert bool((x.grad[zero_out] == 0).all())
    # two calls draw different masks (device seed cell bumps)
    y2 = fused_gelu_dropout(x.detach(), 1.0, p)
    assert not torch.equal((y != 0), (y2 != 0))


@needs_gpu
def test_fast_eval_matches_loader_eval(base_cfg, monkeypatch):
    """The loader-free staged evaluation reproduces the per-user loader
    loop's logged means exactly (same logits math, different batching)."""
    from heterofl_amd.data import fetch_dataset, split_dataset
    from heterofl_amd.fed import FedRunner
    from heterofl_amd.logger import Logger
    from heterofl_amd.models import make_model
    from heterofl_amd.utils import process_dataset, make_optimizer
    cfg = make_cfg(base_cfg, '1_5_0.4_non-iid-2_fix_a1_bn_1_1',
                   data_name='CIFAR10', model_name='resnet18')
    cfg['device'] = 'cuda:0'
    cfg['engine'] = 'batched'
    torch.manual_seed(0)
    ds = fetch_dataset('CIFAR10', synthetic=True, synthetic_size=240)
    process_dataset(ds, cfg)
    data_split, label_split = split_dataset(ds, 5, 'non-iid-2', 10)
    model = make_model(cfg).to('cuda:0')
    runner = FedRunner(cfg, ds, data_split, label_split, model,
                       make_optimizer(model, cfg['lr'], cfg))
    tm = runner.stats()
    means = {}
    for fast in ('1', '0'):
        monkeypatch.setenv('HETEROFL_FAST_EVAL', fast)
        logger = Logger(None)
        runner.logger = logger
        runner.test(tm, 1)
        means[fast] = dict(logger.mean)
    for k, v in means['0'].items():
        assert k in means['1'], k
        assert abs(means['1'][k] - v) < 5e-3, (k, means['1'][k], v)


@needs_gpu
def test_fused_bn_relu_large_batch():
    """The grid-parallel large-batch BN path (stats/eval batches) matches
    the one-block-per-channel kernel and torch at N=2500."""
    from heterofl_amd.ops import require_native
    import torch.nn.functional as F
    ext = require_native()
    torch.manual_seed(0)
    for C, HW in [(64, 1024), (512, 16)]:
        N = 2500
        side = int(HW ** 0.5)
        x = torch.randn(N, C, side, side, device='cuda:0')
        g = torch.rand(C, device='cuda:0') + 0.5
        b = torch.randn(C, device='cuda:0')
        y, mean, invstd = ext.bn_relu_fwd(x, g, b, 1e-5)
        ref = F.relu(F.batch_norm(x, None, None, g, b, training=True,
                                  eps=1e-5))
        assert (y - ref).abs().max().item() < 1e-3, (C, HW)
        ref_m = x.mean(dim=(0, 2, 3))
        assert (mean - ref_m).abs().max().item() < 1e-4


@needs_gpu
def test_blockwise_attention_matches_torch():
    """Blockwise (flash-style) attention for S>64 vs plain torch softmax
    attention: forward and all three input grads, fp32 and bf16, including
    a ragged S (SURVEY §5 long-context readiness)."""
    from heterofl_amd.ops.fused import fused_attention
    torch.manual_seed(0)
    for S in (128, 100, 256):
        for dt, tol in ((torch.float32, 3e-2), (torch.bfloat16, 6e-2)):
            B, d = 6, 32
            q = torch.randn(B, S, d, device='cuda:0', dtype=dt,
                            requires_grad=True)
            k = torch.randn(B, S, d, device='cuda:0', dtype=dt,
                            requires_grad=True)
            v = torch.randn(B, S, d, device='cuda:0', dtype=dt,
                            requires_grad=True)
            temp = d ** 0.5
            out = fused_attention(q, k, v, temp)
            q2, k2, v2 = (t.detach().clone().float().requires_grad_(True)
                          for t in (q, k, v))
            ref = torch.softmax(
                torch.bmm(q2, k2.transpose(1, 2)) / temp, dim=-1).bmm(v2)
            err = (out.float() - ref).abs().max().item()
            assert err < tol, (S, dt, 'fwd', err)
            g = torch.randn_like(ref)
            out.backward(g.to(dt))
            ref.backward(g)
            for a, b, name in ((q, q2, 'dq'), (k, k2, 'dk'), (v, v2, 'dv')):
                derr = (a.grad.float() - b.grad).abs().max().item()
                assert derr < tol * 2, (S, dt, name, derr)


@needs_gpu
def test_blockwise_attention_agrees_with_small_s_path():
    """At S=64 both kernel paths exist; the blockwise path must agree with
    the single-kernel path on identical inputs."""
    from heterofl_amd.ops import require_native
    ext = require_native()
    torch.manual_seed(1)
    B, S, d = 4, 64, 32
    q = torch.randn(B, S, d, device='cuda:0')
    k = torch.randn(B, S, d, device='cuda:0')
    v = torch.randn(B, S, d, device='cuda:0')
    out1, _ = ext.attn_fwd(q, k, v, d ** 0.5)
    out2, _ = ext.attn_fwd_block(q, k, v, d ** 0.5)
    assert (out1 - out2).abs().max().item() < 2e-2


@needs_gpu
def test_fused_maxpool_and_token_mask():
    """K6 MaxPool kernel vs F.max_pool2d (fwd + exact scatter bwd) and K11
    in-kernel Bernoulli token masking statistics/value correctness."""
    from heterofl_amd.ops.fused import fused_maxpool2, fused_token_mask
    import torch.nn.functional as F
    torch.manual_seed(0)
    for dt in (torch.float32, torch.bfloat16):
        for shape in ((6, 40, 32, 32), (3, 8, 7, 7)):
            x = torch.randn(*shape, device='cuda:0', dtype=dt,
                            requires_grad=True)
            x2 = x.detach().clone().requires_grad_(True)
            y = fused_maxpool2(x)
            ref = F.max_pool2d(x2, 2)
            assert torch.equal(y, ref), (dt, shape)
            g = torch.randn_like(y)
            y.backward(g)
            ref.backward(g)
            assert torch.equal(x.grad, x2.grad), (dt, shape)
    toks = torch.randint(0, 1000, (10, 4, 64), device='cuda:0')
    out = fused_token_mask(toks, 0.15, 1000)
    frac = (out == 1000).float().mean().item()
    assert 0.10 < frac < 0.20, frac
    keep = out != 1000
    assert torch.equal(out[keep], toks[keep])
    out2 = fused_token_mask(toks, 0.15, 1000)
    assert not torch.equal(out == 1000, out2 == 1000)  # fresh draw (salt)


@needs_gpu
def test_fused_embed_pos_matches_eager():
    """K9 fused embedding+positional+Scaler gather vs the eager chain —
    forward and the deterministic fp32 master grads, fp32 and bf16-shadow
    routes."""
    from heterofl_amd.ops.fused import fused_embed_pos
    torch.manual_seed(0)
    R, B, S, E, V, P = 3, 2, 64, 96, 300, 64
    rate = 0.5
    for shadows in (False, True):
        table = torch.randn(R, V, E, device='cuda:0',
                            requires_grad=True)
        pos = torch.randn(R, P, E, device='cuda:0', requires_grad=True)
        ids = torch.randint(0, V, (R, B, S), device='cuda:0')
        t16 = table.detach().to(torch.bfloat16).contiguous() if shadows \
            else None
        p16 = pos.detach().to(torch.bfloat16).contiguous() if shadows \
            else None
        y = fused_embed_pos(ids, table, pos, t16, p16, rate)
        # eager reference on the same effective (possibly quantized) tables
        t_ref = (t16.float() if shadows else table.detach()) \
            .clone().requires_grad_(True)
        p_ref = (p16.float() if shadows else pos.detach()) \
            .clone().requires_grad_(True)
        pidx = torch.arange(S, device='cuda:0').view(1, 1, S).expand(R, B, S)
        off = (torch.arange(R, device='cuda:0') * V).view(R, 1, 1)
        poff = (torch.arange(R, device='cuda:0') * P).view(R, 1, 1)
        ref = (torch.nn.functional.embedding((ids + off).reshape(-1),
                                             t_ref.reshape(R * V, E))
               .reshape(R, B, S, E) / rate
               + torch.nn.functional.embedding((pidx + poff).reshape(-1),
                                               p_ref.reshape(R * P, E))
               .reshape(R, B, S, E) / rate)
        # bf16 route: the kernel rounds the fused (a+b)/rate once to bf16,
        # the fp32 reference does not — one output ulp at |y|<=8 is 0.0625
        tol = 7e-2 if shadows else 1e-4
        assert (y.float() - ref).abs().max().item() < tol, shadows
        g = torch.randn_like(ref)
        y.backward(g.to(y.dtype))
        ref.backward(g)
        terr = (table.grad - t_ref.grad).abs().max().item()
        perr = (pos.grad - p_ref.grad).abs().max().item()
        assert terr < tol * 4, (shadows, terr)
        assert perr < tol * 4, (shadows, perr)


@needs_gpu
def test_pack_distribute_matches_slice(base_cfg, monkeypatch):
    """K13 one-kernel slice pack == per-tensor narrow/clone distribute,
    resnet18 heterogeneous rates and transformer (per-head GATHER slices
    take the per-tensor path inside the packed route)."""
    from heterofl_amd.fed.federation import Federation
    from heterofl_amd.models import make_model
    for model_name, data_name in (('resnet18', 'CIFAR10'),
                                  ('transformer', 'WikiText2')):
        cfg = make_cfg(base_cfg, '1_4_1_iid_fix_a1-e1_bn_1_1',
                       data_name=data_name, model_name=model_name)
        if model_name == 'transformer':
            cfg['num_tokens'] = 300
            cfg['bptt'] = 64
        torch.manual_seed(0)
        model = make_model(cfg, model_rate=1.0).to('cuda:0')
        label_split = {i: list(range(10)) for i in range(4)}
        fed = Federation(model.state_dict(), cfg['model_rate'], label_split,
                         cfg)
        user_idx = [0, 1, 2, 3]
        lp_pack, pidx = fed.distribute(user_idx, resample=False)
        assert getattr(fed, '_pack_cache', None), 'pack path did not run'
        monkeypatch.setenv('HETEROFL_FORCE_EAGER', '1')
        lp_ref, _ = fed.distribute(user_idx, resample=False)
        monkeypatch.delenv('HETEROFL_FORCE_EAGER')
        for m in range(4):
            for k in lp_ref[m]:
                a, b = lp_pack[m][k], lp_ref[m][k]
                assert a.shape == b.shape, (model_name, m, k)
                assert torch.equal(a, b), (model_name, m, k)
