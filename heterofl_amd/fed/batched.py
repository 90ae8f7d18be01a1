"""Batched-client training engine — the MI355X-native hot path.

The reference trains the round's active clients strictly one at a time with
batch 10 on 32x32 images (reference: src/train_classifier_fed.py:106-121) —
on a 256-CU MI355X that leaves >95% of the chip idle and is launch-latency
bound.  Here all active clients of one width rate are trained in ONE model:

- every client becomes a GROUP of a grouped convolution: R clients' conv
  weights stack to (R*Cout, Cin, kh, kw), the input batch stacks channelwise
  to (N, R*Cin, H, W), and one conv kernel launch computes all R clients;
- static BatchNorm over R*C channels IS per-client sBN (batch stats are
  per-channel, channels never mix across groups);
- the classifier becomes a per-client bmm; cross-entropy is computed per
  client and summed (gradients stay exactly per-client because groups are
  disjoint);
- grad clipping is the per-client global-L2 clip at 1.0, vectorized over
  the client dimension;
- the batched state_dict keys MIRROR the local model's keys with stacked
  shapes, so pack/unpack between Federation slices and the batched model is
  mechanical.

Numerics: identical to the sequential engine up to fp reduction order
(tested in tests/test_batched.py).
"""
import math

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops as native_ops


# --------------------------------------------------------------------- ops
class BConv2d(nn.Module):
    """R clients' Conv2d as one grouped conv."""

    def __init__(self, R, in_ch, out_ch, kernel, stride, padding, bias):
        super().__init__()
        self.R, self.in_ch, self.out_ch = R, in_ch, out_ch
        self.stride, self.padding = stride, padding
        self.weight = nn.Parameter(torch.empty(R * out_ch, in_ch, kernel, kernel))
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        if bias:
            bound = 1 / math.sqrt(in_ch * kernel * kernel)
            self.bias = nn.Parameter(torch.empty(R * out_ch).uniform_(-bound, bound))
        else:
            self.bias = None

    def forward(self, x, residual=None):
        if native_ops.use_native(x) and native_ops.native_conv_enabled():
            from ..ops.fused import grouped_conv
            return grouped_conv(x, self.weight, self.bias, self.R,
                                self.stride, self.padding, residual=residual)
        y = F.conv2d(x, self.weight, self.bias, stride=self.stride,
                     padding=self.padding, groups=self.R)
        return y if residual is None else y + residual


class BBatchNorm2d(nn.Module):
    """Per-client static BatchNorm (momentum=None, no running stats)."""

    def __init__(self, R, ch):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(R * ch))
        self.bias = nn.Parameter(torch.zeros(R * ch))

    def forward(self, x):
        return F.batch_norm(x, None, None, self.weight, self.bias,
                            training=True, eps=1e-5)


class BGroupNorm(nn.Module):
    """Per-client GroupNorm: local GroupNorm(G, C) -> batched (R*G, R*C)."""

    def __init__(self, R, groups, ch):
        super().__init__()
        self.num_groups = R * groups
        self.weight = nn.Parameter(torch.ones(R * ch))
        self.bias = nn.Parameter(torch.zeros(R * ch))

    def forward(self, x):
        return F.group_norm(x, self.num_groups, self.weight, self.bias, eps=1e-5)


class BLinear(nn.Module):
    """R clients' Linear: weight (R, out, in), input (N, R, in)."""

    def __init__(self, R, in_f, out_f):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(R, out_f, in_f))
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        self.bias = nn.Parameter(torch.zeros(R, out_f))

    def forward(self, x):
        # x: (N, R, in) -> (N, R, out)
        return torch.einsum('nri,roi->nro', x, self.weight) + self.bias


def batched_norm(norm, R, ch):
    if norm == 'bn':
        return BBatchNorm2d(R, ch)
    if norm == 'in':
        return BGroupNorm(R, ch, ch)
    if norm == 'ln':
        return BGroupNorm(R, 1, ch)
    if norm == 'gn':
        return BGroupNorm(R, 4, ch)
    if norm == 'none':
        return nn.Identity()
    raise ValueError('Not valid norm')


class BNormReLU(nn.Module):
    """Fused Scaler -> per-client norm -> ReLU (the prefix of every block,
    reference: src/models/resnet.py:44-50).  On GPU this is ONE hand-written
    HIP kernel forward and one backward (ops/csrc/fused_norm.hip); on CPU the
    explicit scaler/norm/relu torch ops run (the oracle).  The Scaler's x/rate
    cancels inside a train-mode norm, so the kernel omits it; with norm='none'
    the explicit ops always run."""

    def __init__(self, R, ch, norm, rate, scale):
        super().__init__()
        self.norm = norm
        self.rate = rate if scale else 1.0
        # set by the native sBN statistics pass (runner.stats): a callable
        # sink(module, batch_mean, batch_invstd, x) harvesting per-batch BN
        # stats from the fused kernel's own reduction
        self.stats_sink = None
        if norm == 'in':
            self.groups = R * ch
        elif norm == 'ln':
            self.groups = R
        elif norm == 'gn':
            self.groups = R * 4
        else:
            self.groups = 0
        if norm != 'none':
            self.weight = nn.Parameter(torch.ones(R * ch))
            self.bias = nn.Parameter(torch.zeros(R * ch))

    def forward(self, x):
        if self.norm == 'none':
            if self.training and self.rate != 1.0:
                if native_ops.use_native(x):
                    from ..ops.fused import fused_scaler
                    x = fused_scaler(x, self.rate)
                else:
                    x = x / self.rate
            return F.relu(x)
        kind = 'bn' if self.norm == 'bn' else 'gn'
        if native_ops.use_native(x):
            if self.stats_sink is not None and kind == 'bn':
                ext = native_ops.require_native()
                y, mean, invstd = ext.bn_relu_fwd(x.contiguous(), self.weight,
                                                  self.bias, 1e-5)
                self.stats_sink(self, mean, invstd, x)
                return y
            from ..ops.fused import fused_norm_relu
            return fused_norm_relu(x, self.weight, self.bias, kind,
                                   self.groups)
        if self.stats_sink is not None and kind == 'bn':
            mean = x.mean(dim=(0, 2, 3))
            var = x.var(dim=(0, 2, 3), unbiased=False)
            self.stats_sink(self, mean, (var + 1e-5).rsqrt(), x)
        from ..ops.fused import eager_scaler_norm_relu
        r = self.rate if self.training else 1.0
        return eager_scaler_norm_relu(x, self.weight, self.bias, kind,
                                      self.groups, r)


class BMaxPool2(nn.Module):
    """2x2/stride-2 MaxPool: hand kernel on GPU (K6), torch on CPU."""

    def forward(self, x):
        if native_ops.use_native(x):
            from ..ops.fused import fused_maxpool2
            return fused_maxpool2(x)
        return F.max_pool2d(x, 2)


class BScaler(nn.Module):
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


# ------------------------------------------------------------------ models
class BBlock(nn.Module):
    expansion = 1

    def __init__(self, R, in_planes, planes, stride, rate, norm, scale):
        super().__init__()
        self.n1 = BNormReLU(R, in_planes, norm, rate, scale)
        self.conv1 = BConv2d(R, in_planes, planes, 3, stride, 1, bias=False)
        self.n2 = BNormReLU(R, planes, norm, rate, scale)
        self.conv2 = BConv2d(R, planes, planes, 3, 1, 1, bias=False)
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = BConv2d(R, in_planes, self.expansion * planes,
                                    1, stride, 0, bias=False)

    def forward(self, x):
        out = self.n1(x)
        shortcut = self.shortcut(out) if hasattr(self, 'shortcut') else x
        out = self.conv1(out)
        # residual add fused into conv2's epilogue on the native path
        return self.conv2(self.n2(out), residual=shortcut)


class BBottleneck(nn.Module):
    """Pre-activation Bottleneck for R grouped clients (local counterpart:
    models/resnet.py:47-72; reference: src/models/resnet.py:53-101).  The
    1x1 convs ride the grouped-conv GEMM fast path; the residual is fused
    into conv3's epilogue on the native path."""
    expansion = 4

    def __init__(self, R, in_planes, planes, stride, rate, norm, scale):
        super().__init__()
        self.n1 = BNormReLU(R, in_planes, norm, rate, scale)
        self.conv1 = BConv2d(R, in_planes, planes, 1, 1, 0, bias=False)
        self.n2 = BNormReLU(R, planes, norm, rate, scale)
        self.conv2 = BConv2d(R, planes, planes, 3, stride, 1, bias=False)
        self.n3 = BNormReLU(R, planes, norm, rate, scale)
        self.conv3 = BConv2d(R, planes, self.expansion * planes, 1, 1, 0,
                             bias=False)
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = BConv2d(R, in_planes, self.expansion * planes,
                                    1, stride, 0, bias=False)

    def forward(self, x):
        out = self.n1(x)
        shortcut = self.shortcut(out) if hasattr(self, 'shortcut') else x
        out = self.conv1(out)
        out = self.conv2(self.n2(out))
        return self.conv3(self.n3(out), residual=shortcut)


class BatchedResNet(nn.Module):
    """R same-rate clients' ResNet (basic Block or Bottleneck) in one
    grouped model.  state_dict keys mirror models.resnet.ResNet with
    stacked shapes."""

    def __init__(self, R, data_shape, hidden_size, num_blocks, num_classes,
                 rate, norm, scale, block=BBlock):
        super().__init__()
        self.R = R
        self.num_classes = num_classes
        self.data_ch = data_shape[0]
        self.in_planes = hidden_size[0]
        self.conv1 = BConv2d(R, data_shape[0], hidden_size[0], 3, 1, 1, bias=False)
        self.layer1 = self._make_layer(R, block, hidden_size[0], num_blocks[0], 1, rate, norm, scale)
        self.layer2 = self._make_layer(R, block, hidden_size[1], num_blocks[1], 2, rate, norm, scale)
        self.layer3 = self._make_layer(R, block, hidden_size[2], num_blocks[2], 2, rate, norm, scale)
        self.layer4 = self._make_layer(R, block, hidden_size[3], num_blocks[3], 2, rate, norm, scale)
        self.n4 = BNormReLU(R, hidden_size[3] * block.expansion, norm, rate, scale)
        self.linear = BLinear(R, hidden_size[3] * block.expansion, num_classes)

    def _make_layer(self, R, block, planes, num_blocks, stride, rate, norm, scale):
        strides = [stride] + [1] * (num_blocks - 1)
        layers = []
        for s in strides:
            layers.append(block(R, self.in_planes, planes, s, rate, norm, scale))
            self.in_planes = planes * block.expansion
        return nn.Sequential(*layers)

    def forward(self, x):
        # x: (N, R*data_ch, H, W) -> scores (N, R, classes)
        out = self.conv1(x)
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = self.n4(out)
        if native_ops.use_native(out):
            from ..ops.fused import fused_head
            return fused_head(out, self.linear.weight, self.linear.bias,
                              self.R)
        out = F.adaptive_avg_pool2d(out, 1)
        out = out.view(out.size(0), self.R, -1)
        return self.linear(out)


class BatchedConv(nn.Module):
    """R same-rate clients' 4-block CNN.  Keys mirror models.conv.Conv
    (blocks.i.*) with stacked shapes."""

    def __init__(self, R, data_shape, hidden_size, num_classes, rate, norm, scale):
        super().__init__()
        self.R = R
        self.num_classes = num_classes
        blocks = []
        in_ch = data_shape[0]
        for i, out_ch in enumerate(hidden_size):
            blocks.append(BConv2d(R, in_ch, out_ch, 3, 1, 1, bias=True))
            # placeholder keeps the norm at sequential index 4k+2 so keys
            # mirror the local model (src/models/conv.py:29-33)
            blocks.append(nn.Identity())
            blocks.append(BNormReLU(R, out_ch, norm, rate, scale))
            blocks.append(nn.Identity())
            if i != len(hidden_size) - 1:
                blocks.append(BMaxPool2())
            in_ch = out_ch
        self.blocks = nn.Sequential(*blocks)
        # final linear occupies the tail indices like the local model
        self._feat = hidden_size[-1]
        self.head = BLinear(R, hidden_size[-1], num_classes)

    def forward(self, x):
        out = self.blocks(x)
        if native_ops.use_native(out):
            from ..ops.fused import fused_head
            return fused_head(out, self.head.weight, self.head.bias, self.R)
        out = F.adaptive_avg_pool2d(out, 1).view(out.size(0), self.R, -1)
        return self.head(out)


# ------------------------------------------------------- pack / unpack
def pack_states(batched, locals_list):
    """Load R per-client state dicts into the batched model, by key."""
    R = batched.R
    bsd = batched.state_dict()
    key_map = _key_map(batched, locals_list[0])
    with torch.no_grad():
        for bk, lk in key_map.items():
            bt = bsd[bk]
            vals = [locals_list[r][lk] for r in range(R)]
            if bt.dim() == 3:            # BLinear weight (R, O, I)
                src = torch.stack(vals, 0)
            elif bt.dim() == 2:          # BLinear bias (R, O)
                src = torch.stack(vals, 0)
            else:                        # stacked along channel dim 0
                src = torch.cat(vals, 0)
            bt.copy_(src.to(bt.device, bt.dtype))


def unpack_states(batched, template_keys):
    """Split the batched model back into R per-client state dicts."""
    R = batched.R
    bsd = batched.state_dict()
    key_map = _key_map_from_keys(batched, template_keys)
    outs = [dict() for _ in range(R)]
    for bk, lk in key_map.items():
        bt = bsd[bk].detach()
        # BLinear params are (R, ...) stacks; everything else is a dim-0 cat
        # (conv weights (R*O, I, kh, kw), norm weights/biases (R*C,)).
        if bt.dim() in (2, 3):
            for r in range(R):
                outs[r][lk] = bt[r].clone()
        else:
            per = bt.size(0) // R
            for r in range(R):
                outs[r][lk] = bt[r * per:(r + 1) * per].clone()
    return outs


def _key_map(batched, local_sd):
    return _key_map_from_keys(batched, list(local_sd.keys()))


def _key_map_from_keys(batched, local_keys):
    """Map batched state_dict keys -> local model keys.  BatchedResNet keys
    match exactly; BatchedConv maps head.* to the local blocks tail linear."""
    bkeys = list(batched.state_dict().keys())
    if isinstance(batched, BatchedConv):
        lin_w = [k for k in local_keys if 'weight' in k][-1]
        lin_b = [k for k in local_keys if 'bias' in k][-1]
        m = {}
        for bk in bkeys:
            if bk == 'head.weight':
                m[bk] = lin_w
            elif bk == 'head.bias':
                m[bk] = lin_b
            else:
                m[bk] = bk
        return m
    return {bk: bk for bk in bkeys}


# ------------------------------------------------------------- loss / clip
def batched_masked_ce(scores, labels, label_masks, metrics=None):
    """scores (N, R, C), labels (N, R), label_masks (R, C) in {0,1} or None.
    Returns per-client mean-CE losses (R,).  Matches the reference's
    mask-then-CE order (src/models/resnet.py:152-157).  On GPU this is the
    hand-written kernel (ops/csrc/masked_ce.hip), which also accumulates
    device-side (loss-sum, correct, count) into `metrics` (R, 3) if given."""
    if native_ops.use_native(scores):
        from ..ops.fused import fused_masked_ce
        return fused_masked_ce(scores, labels, label_masks, metrics)
    N, R, C = scores.shape
    if label_masks is not None:
        scores = scores.masked_fill(label_masks.unsqueeze(0) == 0, 0)
    logp = F.log_softmax(scores, dim=2)
    nll = -logp.gather(2, labels.unsqueeze(2)).squeeze(2)  # (N, R)
    losses = nll.mean(0)
    if metrics is not None:
        with torch.no_grad():
            correct = (scores.argmax(2) == labels).float().sum(0)
            metrics[:, 0] += losses.detach() * N
            metrics[:, 1] += correct
            metrics[:, 2] += N
    return losses


def per_client_clip_(params, R, max_norm=1.0):
    """Vectorized per-client global-L2 grad clip at max_norm
    (reference: src/train_classifier_fed.py:205)."""
    sq = None
    views = []
    for p in params:
        if p.grad is None:
            continue
        g = p.grad
        if g.size(0) % R != 0:
            raise RuntimeError(f'grad shape {tuple(g.shape)} not divisible by R={R}')
        v = g.view(R, -1)
        views.append((g, v))
        s = (v.float() ** 2).sum(dim=1)
        sq = s if sq is None else sq + s
    if sq is None:
        return None
    total = sq.sqrt()
    scale = (max_norm / (total + 1e-6)).clamp(max=1.0)
    for g, v in views:
        v.mul_(scale.unsqueeze(1).to(v.dtype))
    return total


# ------------------------------------------------------------ augmentation
class DeviceAugment:
    """GPU-side per-sample augmentation + normalization matching the
    reference's torchvision pipeline distributionally
    (reference: src/data.py:14-27)."""

    def __init__(self, data_name, device):
        self.data_name = data_name
        if data_name in ('MNIST', 'FashionMNIST'):
            mean, std = (0.1307,), (0.3081,)
            self.crop = False
        else:
            mean, std = (0.4914, 0.4822, 0.4465), (0.2023, 0.1994, 0.2010)
            self.crop = True
        self.mean = torch.tensor(mean, device=device).view(1, -1, 1, 1)
        self.std = torch.tensor(std, device=device).view(1, -1, 1, 1)

    def __call__(self, img_u8, train=True):
        """img_u8: (n, H, W, C) uint8 -> (n, C, H, W) float normalized."""
        x = img_u8.permute(0, 3, 1, 2).float().div_(255.0)
        n, C, H, W = x.shape
        if train and self.crop:
            pad = 4
            xp = F.pad(x, [pad] * 4)
            dy = torch.randint(0, 2 * pad + 1, (n,), device=x.device)
            dx = torch.randint(0, 2 * pad + 1, (n,), device=x.device)
            rows = dy.view(n, 1) + torch.arange(H, device=x.device).view(1, H)
            idx_r = rows.view(n, 1, H, 1).expand(n, C, H, W + 2 * pad)
            xp = xp.gather(2, idx_r)
            cols = dx.view(n, 1) + torch.arange(W, device=x.device).view(1, W)
            idx_c = cols.view(n, 1, 1, W).expand(n, C, H, W)
            x = xp.gather(3, idx_c)
            flip = torch.rand(n, device=x.device) < 0.5
            x = torch.where(flip.view(n, 1, 1, 1), x.flip(-1), x)
        return (x - self.mean) / self.std


# ----------------------------------------------------------------- trainer
class BatchedClientTrainer:
    """Trains the round's active clients grouped by (rate, batch schedule).

    Interface matches SequentialClientTrainer.train_clients; data access goes
    through shard tensors staged on device (set_data must be called by the
    runner before the first round)."""

    def __init__(self, cfg):
        self.cfg = cfg
        self.device = torch.device(cfg['device'])
        self.dataset = None
        self.data_split = None
        self.augment = None
        # bf16/fp8 local training (master weights fp32, compute bf16, and
        # in fp8 mode the conv GEMMs on fp8 MFMA)
        self._amp = (cfg.get('compute_dtype') in ('bfloat16', 'fp8')
                     and self.device.type == 'cuda')
        if self.device.type == 'cuda':
            native_ops.set_fp8(cfg.get('compute_dtype') == 'fp8')
        self._shard_cache = {}
        self._model_cache = {}
        self._graph_cache = {}

    # data staging ---------------------------------------------------------
    def set_data(self, dataset, data_split):
        self.dataset = dataset['train']
        self.data_split = data_split['train']
        self.augment = DeviceAugment(self.cfg['data_name'], self.device)
        self._shard_cache.clear()

    def _shard(self, user):
        if user not in self._shard_cache:
            idx = torch.tensor(self.data_split[user], dtype=torch.long)
            img = self.dataset.img[idx]          # uint8 (n, H, W[, C])
            if img.dim() == 3:
                img = img.unsqueeze(-1)
            labels = torch.tensor([self.dataset.target[i] for i in idx.tolist()],
                                  dtype=torch.long)
            self._shard_cache[user] = (img.to(self.device, non_blocking=True),
                                       labels.to(self.device, non_blocking=True))
        return self._shard_cache[user]

    # model construction ---------------------------------------------------
    def _batched_model(self, rate, R):
        key = (rate, R)
        if key not in self._model_cache:
            cfg = self.cfg
            name = cfg['model_name']
            scaler_rate = rate / cfg['global_model_rate']
            if 'resnet' in name:
                hidden = [int(np.ceil(rate * h)) for h in cfg['resnet']['hidden_size']]
                nb, blk = {'resnet18': ([2, 2, 2, 2], BBlock),
                           'resnet34': ([3, 4, 6, 3], BBlock),
                           'resnet50': ([3, 4, 6, 3], BBottleneck),
                           'resnet101': ([3, 4, 23, 3], BBottleneck),
                           'resnet152': ([3, 8, 36, 3], BBottleneck)}[name]
                model = BatchedResNet(R, cfg['data_shape'], hidden, nb,
                                      cfg['classes_size'], scaler_rate,
                                      cfg['norm'], cfg['scale'], block=blk)
            elif name == 'conv':
                hidden = [int(np.ceil(rate * h)) for h in cfg['conv']['hidden_size']]
                model = BatchedConv(R, cfg['data_shape'], hidden,
                                    cfg['classes_size'], scaler_rate,
                                    cfg['norm'], cfg['scale'])
            else:
                raise NotImplementedError(name)
            self._model_cache[key] = model.to(self.device)
        return self._model_cache[key]

    # training -------------------------------------------------------------
    def train_clients(self, client_slots, user_idx, local_parameters,
                      model_rate, make_loader, label_split, lr, logger=None):
        if self.dataset is None:
            raise RuntimeError('BatchedClientTrainer.set_data was not called')
        cfg = self.cfg
        B = cfg['batch_size']['train']
        # group slots by (rate, batch schedule)
        groups = {}
        for m in client_slots:
            u = user_idx[m]
            n = len(self.data_split[u])
            sched = tuple([B] * (n // B) + ([n % B] if n % B else []))
            groups.setdefault((model_rate[u], sched), []).append(m)
        out = []
        # optional: split each rate group into HETEROFL_GROUP_SPLIT
        # subgroups so independent client chains replay on extra concurrent
        # streams (experimental; subgroups get distinct graph tags so
        # equal-size subgroups never share a captured graph)
        import os as _os
        gsplit = int(_os.environ.get('HETEROFL_GROUP_SPLIT', '1'))
        split_groups = {}
        for (rate, sched), slots in groups.items():
            if gsplit > 1 and self.device.type == 'cuda' \
                    and len(slots) >= 2 * gsplit:
                per = (len(slots) + gsplit - 1) // gsplit
                for si in range(0, len(slots), per):
                    split_groups[(rate, sched, si)] = slots[si:si + per]
            else:
                split_groups[(rate, sched, 0)] = slots
        # graph-eligible groups are prepared on the default stream, then
        # their replay trains run CONCURRENTLY on per-group HIP streams (the
        # rate-e group's tiny latency-bound steps hide under rate-a's)
        graphed = []
        for (rate, sched, tag), slots in split_groups.items():
            use_graph = (self.device.type == 'cuda'
                         and cfg.get('hip_graphs', True)
                         and len(set(sched)) == 1)
            if use_graph:
                graphed.append((rate, sched, slots, tag))
            else:
                out.extend(self._train_group(rate, sched, slots, user_idx,
                                             local_parameters, label_split,
                                             lr, logger))
        if graphed:
            preps = [self._prepare_graphed(rate, sched, slots, user_idx,
                                           [local_parameters[m]
                                            for m in slots],
                                           label_split, lr, tag)
                     for rate, sched, slots, tag in graphed]
            from .runner import _phase_timer
            with _phase_timer('2d.replay'):
                if len(preps) > 1:
                    if not hasattr(self, '_streams') or \
                            len(self._streams) < len(preps):
                        self._streams = [torch.cuda.Stream()
                                         for _ in range(len(preps))]
                    cur = torch.cuda.current_stream()
                    events = []
                    for (gs, x_cat, y_cat, nspe), st in zip(preps,
                                                            self._streams):
                        st.wait_stream(cur)
                        with torch.cuda.stream(st):
                            gs.run_epochs(x_cat, y_cat, nspe)
                        ev = torch.cuda.Event()
                        ev.record(st)
                        events.append(ev)
                    for ev in events:
                        cur.wait_event(ev)
                else:
                    gs, x_cat, y_cat, nspe = preps[0]
                    gs.run_epochs(x_cat, y_cat, nspe)
            for (rate, sched, slots, tag), (gs, _, _, _) in zip(graphed,
                                                                preps):
                out.extend(self._finish_graphed(
                    gs, slots, [local_parameters[m] for m in slots], logger))
        return out

    def _graph_step(self, rate, sched, R, lr, tag=0):
        """Get/create the captured training step for this group shape.  The
        tag keeps concurrently-replayed subgroups (HETEROFL_GROUP_SPLIT) on
        distinct graphs/buffers."""
        from .graphs import GraphedGroupStep
        key = (rate, R, tuple(sched), lr, tag)
        if key not in self._graph_cache:
            cfg = self.cfg
            model = self._batched_model(rate, R)
            # graphs own their model instance: evict from the eager cache so
            # shapes don't alias
            self._model_cache.pop((rate, R), None)
            model.train(True)
            capacity = sum(sched) * cfg['num_epochs']['local']
            self._graph_cache[key] = GraphedGroupStep(
                model, R, sched[0], lr, cfg['momentum'], cfg['weight_decay'],
                cfg['classes_size'], capacity, self._amp, self.device)
        return self._graph_cache[key]

    def _train_group(self, rate, sched, slots, user_idx, local_parameters,
                     label_split, lr, logger=None):
        cfg = self.cfg
        R = len(slots)
        device = self.device
        use_graph = (device.type == 'cuda' and cfg.get('hip_graphs', True)
                     and len(set(sched)) == 1)
        locals_list = [local_parameters[m] for m in slots]
        if use_graph:
            return self._train_group_graphed(rate, sched, slots, user_idx,
                                             locals_list, label_split, lr,
                                             logger)
        model = self._batched_model(rate, R)
        pack_states(model, locals_list)
        model.train(True)
        params = [p for p in model.parameters() if p.requires_grad]
        native = native_ops.use_native(device)
        if native:
            # identical kernels to the graphed path: backward into stable
            # .grad buffers + fused per-client clip+momentum-SGD
            if not hasattr(self, '_opt_cache'):
                self._opt_cache = {}
            cached = self._opt_cache.get((rate, R))
            if cached is None or cached[0] is not model:
                from ..ops.fused import FusedClipSGD
                for p in params:
                    p.grad = torch.zeros_like(p)
                bufs = [torch.zeros_like(p) for p in params]
                cached = (model, FusedClipSGD(params,
                                              [p.grad for p in params],
                                              bufs, R, device))
                self._opt_cache[(rate, R)] = cached
            fopt = cached[1]
            for b in fopt.bufs:
                b.zero_()
            opt = None
        else:
            opt = torch.optim.SGD(params, lr=lr, momentum=cfg['momentum'],
                                  weight_decay=cfg['weight_decay'])
        # label masks (R, classes)
        masks = None
        if cfg['mask']:
            masks = torch.zeros(R, cfg['classes_size'], device=device)
            for i, m in enumerate(slots):
                masks[i, label_split[user_idx[m]]] = 1
        shards = [self._shard(user_idx[m]) for m in slots]
        n = sum(sched)
        for _ in range(cfg['num_epochs']['local']):
            # per-client shuffle + epoch-wide augmented gather
            xs, ys = [], []
            for img, lab in shards:
                perm = torch.randperm(img.size(0), device=device)
                xs.append(self.augment(img[perm], train=True))   # (n, C, H, W)
                ys.append(lab[perm])
            x_all = torch.stack(xs, 1)   # (n, R, C, H, W)
            x_all = x_all.reshape(n, -1, x_all.size(-2), x_all.size(-1))
            if self._amp:
                x_all = x_all.to(torch.bfloat16)
            y_all = torch.stack(ys, 1)   # (n, R)
            off = 0
            for bs in sched:
                xb = x_all[off:off + bs]
                yb = y_all[off:off + bs]
                off += bs
                if native:
                    torch._foreach_zero_([p.grad for p in params])
                else:
                    opt.zero_grad(set_to_none=True)
                with torch.autocast('cuda', torch.bfloat16, enabled=self._amp):
                    scores = model(xb)
                losses = batched_masked_ce(scores.float(), yb, masks)
                losses.sum().backward()
                if native:
                    fopt.step(1.0, lr, cfg['momentum'], cfg['weight_decay'])
                else:
                    per_client_clip_(params, R, 1.0)
                    opt.step()
                if logger is not None:
                    with torch.no_grad():
                        # accuracy over the MASKED scores, like the
                        # reference metric (the model's forward rewrites
                        # score before Metric sees it,
                        # src/models/resnet.py:152-157)
                        sc = scores
                        if masks is not None:
                            sc = sc.masked_fill(masks.unsqueeze(0) == 0, 0)
                        pred = sc.argmax(dim=2)
                        acc = (pred == yb).float().mean(0) * 100.0
                        for i in range(R):
                            logger.append({'Local-Loss': losses[i].item(),
                                           'Local-Accuracy': acc[i].item()},
                                          'train', n=bs)
        template_keys = list(locals_list[0].keys())
        states = unpack_states(model, template_keys)
        cpu_or_dev = [{k: v for k, v in st.items()} for st in states]
        return list(zip(slots, cpu_or_dev))

    def _prepare_graphed(self, rate, sched, slots, user_idx, locals_list,
                         label_split, lr, tag=0):
        """Default-stream phase of the hipGraph path: build/capture, pack,
        masks, stage all epochs' augmented data.  Returns (gs, x, y, nspe)."""
        from .runner import _phase_timer
        cfg = self.cfg
        R = len(slots)
        device = self.device
        with _phase_timer('2a.graph_build'):
            gs = self._graph_step(rate, sched, R, lr, tag)
            if gs.graph is None:
                gs.ensure_captured()
        with _phase_timer('2b.pack'):
            pack_states(gs.model, locals_list)
        masks = None
        if cfg['mask']:
            masks = torch.zeros(R, cfg['classes_size'], device=device)
            for i, m in enumerate(slots):
                masks[i, label_split[user_idx[m]]] = 1
        gs.begin_round(masks)
        shards = [self._shard(user_idx[m]) for m in slots]
        n = sum(sched)
        E = cfg['num_epochs']['local']
        with _phase_timer('2c.stage_augment'):
            xs_ep, ys_ep = [], []
            for _ in range(E):
                xs, ys = [], []
                for img, lab in shards:
                    perm = torch.randperm(img.size(0), device=device)
                    xs.append(self.augment(img[perm], train=True))
                    ys.append(lab[perm])
                x_all = torch.stack(xs, 1)
                xs_ep.append(x_all.reshape(n, -1, x_all.size(-2), x_all.size(-1)))
                ys_ep.append(torch.stack(ys, 1))
            x_cat = torch.cat(xs_ep, 0)
            if self._amp:
                x_cat = x_cat.to(torch.bfloat16)
            y_cat = torch.cat(ys_ep, 0)
        return gs, x_cat, y_cat, len(sched)

    def _finish_graphed(self, gs, slots, locals_list, logger=None):
        from .runner import _phase_timer
        R = len(slots)
        if logger is not None:
            m = gs.metrics.detach().cpu()
            for i in range(R):
                cnt = max(int(m[i, 2].item()), 1)
                logger.append({'Local-Loss': m[i, 0].item() / cnt,
                               'Local-Accuracy': m[i, 1].item() / cnt * 100.0},
                              'train', n=cnt)
        with _phase_timer('2e.unpack'):
            template_keys = list(locals_list[0].keys())
            states = unpack_states(gs.model, template_keys)
        return list(zip(slots, [dict(st) for st in states]))
