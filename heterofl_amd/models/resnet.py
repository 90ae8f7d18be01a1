"""Pre-activation ResNet with width-rate slicing hooks.

Same architecture and state-dict layout as the reference
(reference: src/models/resnet.py:9-208): pre-activation Block
(Scaler->n1->ReLU->[shortcut]->conv1->Scaler->n2->ReLU->conv2->+),
stem conv + 4 stages + n4/Scaler/ReLU + avgpool + linear, masked
cross-entropy in forward.  Module names (conv1, conv2, shortcut, linear)
are load-bearing: the federation slicing rules match on them.

Differences from the reference implementation (not behavior):
explicit constructor arguments instead of a global cfg; factories read cfg.
"""
import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

from .modules import Scaler, make_norm, init_param
from .functional import masked_cross_entropy


class Block(nn.Module):
    expansion = 1

    def __init__(self, in_planes, planes, stride, rate, track, norm, scale):
        super().__init__()
        self.n1 = make_norm(norm, in_planes, track)
        self.conv1 = nn.Conv2d(in_planes, planes, kernel_size=3, stride=stride,
                               padding=1, bias=False)
        self.n2 = make_norm(norm, planes, track)
        self.conv2 = nn.Conv2d(planes, planes, kernel_size=3, stride=1,
                               padding=1, bias=False)
        self.scaler = Scaler(rate) if scale else nn.Identity()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Conv2d(in_planes, self.expansion * planes,
                                      kernel_size=1, stride=stride, bias=False)

    def forward(self, x):
        out = F.relu(self.n1(self.scaler(x)))
        shortcut = self.shortcut(out) if hasattr(self, 'shortcut') else x
        out = self.conv1(out)
        out = self.conv2(F.relu(self.n2(self.scaler(out))))
        out += shortcut
        return out


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, in_planes, planes, stride, rate, track, norm, scale):
        super().__init__()
        self.n1 = make_norm(norm, in_planes, track)
        self.conv1 = nn.Conv2d(in_planes, planes, kernel_size=1, bias=False)
        self.n2 = make_norm(norm, planes, track)
        self.conv2 = nn.Conv2d(planes, planes, kernel_size=3, stride=stride,
                               padding=1, bias=False)
        self.n3 = make_norm(norm, planes, track)
        self.conv3 = nn.Conv2d(planes, self.expansion * planes, kernel_size=1,
                               bias=False)
        self.scaler = Scaler(rate) if scale else nn.Identity()
        if stride != 1 or in_planes != self.expansion * planes:
            self.shortcut = nn.Conv2d(in_planes, self.expansion * planes,
                                      kernel_size=1, stride=stride, bias=False)

    def forward(self, x):
        out = F.relu(self.n1(self.scaler(x)))
        shortcut = self.shortcut(out) if hasattr(self, 'shortcut') else x
        out = self.conv1(out)
        out = self.conv2(F.relu(self.n2(self.scaler(out))))
        out = self.conv3(F.relu(self.n3(self.scaler(out))))
        out += shortcut
        return out


class ResNet(nn.Module):
    def __init__(self, data_shape, hidden_size, block, num_blocks, num_classes,
                 rate, track, norm, scale, mask):
        super().__init__()
        self.num_classes = num_classes
        self.mask = mask
        self.in_planes = hidden_size[0]
        self.conv1 = nn.Conv2d(data_shape[0], hidden_size[0], kernel_size=3,
                               stride=1, padding=1, bias=False)
        self.layer1 = self._make_layer(block, hidden_size[0], num_blocks[0], 1, rate, track, norm, scale)
        self.layer2 = self._make_layer(block, hidden_size[1], num_blocks[1], 2, rate, track, norm, scale)
        self.layer3 = self._make_layer(block, hidden_size[2], num_blocks[2], 2, rate, track, norm, scale)
        self.layer4 = self._make_layer(block, hidden_size[3], num_blocks[3], 2, rate, track, norm, scale)
        self.n4 = make_norm(norm, hidden_size[3] * block.expansion, track)
        self.scaler = Scaler(rate) if scale else nn.Identity()
        self.linear = nn.Linear(hidden_size[3] * block.expansion, num_classes)

    def _make_layer(self, block, planes, num_blocks, stride, rate, track, norm, scale):
        strides = [stride] + [1] * (num_blocks - 1)
        layers = []
        for s in strides:
            layers.append(block(self.in_planes, planes, s, rate, track, norm, scale))
            self.in_planes = planes * block.expansion
        return nn.Sequential(*layers)

    def features(self, x):
        out = self.conv1(x)
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = F.relu(self.n4(self.scaler(out)))
        out = F.adaptive_avg_pool2d(out, 1)
        return out.view(out.size(0), -1)

    def forward(self, input):
        output = {}
        out = self.linear(self.features(input['img']))
        score, loss = masked_cross_entropy(
            out, input['label'],
            input.get('label_split') if self.mask else None,
            self.num_classes)
        output['score'] = score
        output['loss'] = loss
        return output


def _make_resnet(cfg, model_rate, track, block, num_blocks):
    data_shape = cfg['data_shape']
    classes_size = cfg['classes_size']
    hidden_size = [int(np.ceil(model_rate * x)) for x in cfg['resnet']['hidden_size']]
    scaler_rate = model_rate / cfg['global_model_rate']
    model = ResNet(data_shape, hidden_size, block, num_blocks, classes_size,
                   scaler_rate, track, cfg['norm'], cfg['scale'], cfg['mask'])
    model.apply(init_param)
    return model


def resnet18(cfg, model_rate=1, track=False):
    return _make_resnet(cfg, model_rate, track, Block, [2, 2, 2, 2])


def resnet34(cfg, model_rate=1, track=False):
    return _make_resnet(cfg, model_rate, track, Block, [3, 4, 6, 3])


def resnet50(cfg, model_rate=1, track=False):
    return _make_resnet(cfg, model_rate, track, Bottleneck, [3, 4, 6, 3])


def resnet101(cfg, model_rate=1, track=False):
    return _make_resnet(cfg, model_rate, track, Bottleneck, [3, 4, 23, 3])


def resnet152(cfg, model_rate=1, track=False):
    return _make_resnet(cfg, model_rate, track, Bottleneck, [3, 8, 36, 3])
