"""4-block CNN (reference: src/models/conv.py:10-81):
[Conv3x3 -> Scaler -> Norm -> ReLU -> MaxPool2] x4 (last pool dropped),
AdaptiveAvgPool -> Flatten -> Linear, masked cross-entropy in forward.
"""
import numpy as np
import torch.nn as nn

from .modules import Scaler, make_norm, init_param
from .functional import masked_cross_entropy


class Conv(nn.Module):
    def __init__(self, data_shape, hidden_size, classes_size, rate, track,
                 norm, scale, mask):
        super().__init__()
        self.classes_size = classes_size
        self.mask = mask
        blocks = []
        in_ch = data_shape[0]
        for i, out_ch in enumerate(hidden_size):
            blocks.append(nn.Conv2d(in_ch, out_ch, 3, 1, 1))
            blocks.append(Scaler(rate) if scale else nn.Identity())
            blocks.append(make_norm(norm, out_ch, track))
            blocks.append(nn.ReLU(inplace=True))
            if i != len(hidden_size) - 1:
                blocks.append(nn.MaxPool2d(2))
            in_ch = out_ch
        blocks.extend([nn.AdaptiveAvgPool2d(1),
                       nn.Flatten(),
                       nn.Linear(hidden_size[-1], classes_size)])
        self.blocks = nn.Sequential(*blocks)

    def forward(self, input):
        out = self.blocks(input['img'])
        score, loss = masked_cross_entropy(
            out, input['label'],
            input.get('label_split') if self.mask else None,
            self.classes_size)
        return {'score': score, 'loss': loss}


def conv(cfg, model_rate=1, track=False):
    data_shape = cfg['data_shape']
    hidden_size = [int(np.ceil(model_rate * x)) for x in cfg['conv']['hidden_size']]
    classes_size = cfg['classes_size']
    scaler_rate = model_rate / cfg['global_model_rate']
    model = Conv(data_shape, hidden_size, classes_size, scaler_rate, track,
                 cfg['norm'], cfg['scale'], cfg['mask'])
    model.apply(init_param)
    return model
