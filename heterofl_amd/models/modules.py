"""Shared building blocks: Scaler and norm selection.

Scaler (reference: src/modules/modules.py:4-11): divide activations by the
width rate during training only; identity at eval.  rate here is
model_rate / global_model_rate (reference: src/models/resnet.py:165).

On GPU with the native extension, Scaler is folded into the neighboring
kernels' epilogues by the fused ops layer; this module is the semantic
definition and the CPU path.
"""
import torch.nn as nn


class Scaler(nn.Module):
    def __init__(self, rate):
        super().__init__()
        self.rate = rate

    def forward(self, input):
        return input / self.rate if self.training else input

    def extra_repr(self):
        return f'rate={self.rate}'


def make_norm(norm, num_channels, track=False):
    """Select the norm layer by cfg['norm'] (reference: src/models/resnet.py:14-31).

    'bn' is static BatchNorm (sBN): momentum=None means cumulative moving
    average when running stats are tracked; track=False during federated
    training, True for the post-round statistics pass.
    """
    if norm == 'bn':
        return nn.BatchNorm2d(num_channels, momentum=None, track_running_stats=track)
    if norm == 'in':
        return nn.GroupNorm(num_channels, num_channels)
    if norm == 'ln':
        return nn.GroupNorm(1, num_channels)
    if norm == 'gn':
        return nn.GroupNorm(4, num_channels)
    if norm == 'none':
        return nn.Identity()
    raise ValueError('Not valid norm')


def init_param(m):
    """BN/IN weight<-1 bias<-0, Linear bias<-0 (reference: src/models/utils.py:4-10)."""
    if isinstance(m, (nn.BatchNorm2d, nn.InstanceNorm2d)):
        m.weight.data.fill_(1)
        m.bias.data.zero_()
    elif isinstance(m, nn.Linear):
        m.bias.data.zero_()
    return m
