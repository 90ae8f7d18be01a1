"""Model factories.  Every factory takes (cfg, model_rate, track) and returns
an nn.Module whose forward(dict) -> {'score', 'loss'}
(reference: src/models/resnet.py:161-168).

make_model replaces the reference's string-eval construction
(reference: src/train_classifier_fed.py:54) with an explicit registry.
"""
from .conv import conv
from .resnet import resnet18, resnet34, resnet50, resnet101, resnet152
from .transformer import transformer
from .modules import Scaler, make_norm, init_param

_REGISTRY = {
    'conv': conv,
    'resnet18': resnet18,
    'resnet34': resnet34,
    'resnet50': resnet50,
    'resnet101': resnet101,
    'resnet152': resnet152,
    'transformer': transformer,
}


def make_model(cfg, model_rate=None, track=False):
    name = cfg['model_name']
    if name not in _REGISTRY:
        raise ValueError(f'Not valid model name: {name}')
    if model_rate is None:
        model_rate = cfg['global_model_rate']
    return _REGISTRY[name](cfg, model_rate=model_rate, track=track)


__all__ = ['conv', 'resnet18', 'resnet34', 'resnet50', 'resnet101',
           'resnet152', 'transformer', 'make_model', 'Scaler', 'make_norm',
           'init_param']
