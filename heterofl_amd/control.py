"""control_name grammar and hyperparameter expansion.

Replicates the semantics of the reference's ``process_control``
(reference: src/utils.py:113-215): the 9-field underscore-joined control
string ``fed_numusers_frac_datasplit_modelsplit_modelmode_norm_scale_mask``
expands into the model-rate assignment per user and the per-dataset
hyperparameters.
"""
import numpy as np

# Width level -> rate (reference: src/utils.py:114)
MODEL_SPLIT_RATE = {'a': 1, 'b': 0.5, 'c': 0.25, 'd': 0.125, 'e': 0.0625}

CONTROL_FIELDS = ['fed', 'num_users', 'frac', 'data_split_mode',
                  'model_split_mode', 'model_mode', 'norm', 'scale', 'mask']


def parse_control_name(control_name):
    """Split a control_name string into the control dict (string values)."""
    parts = control_name.split('_')
    if len(parts) != len(CONTROL_FIELDS):
        raise ValueError(
            f'control_name {control_name!r} has {len(parts)} fields, expected '
            f'{len(CONTROL_FIELDS)} ({"_".join(CONTROL_FIELDS)})')
    return dict(zip(CONTROL_FIELDS, parts))


def parse_model_mode(model_mode):
    """Parse the '-'-separated ``<level><weight>`` terms, e.g. 'a1-e1'.

    Returns (mode_rate list, proportion list of ints).
    """
    mode_rate, proportion = [], []
    for m in model_mode.split('-'):
        level, weight = m[0], m[1:]
        if level not in MODEL_SPLIT_RATE:
            raise ValueError(f'unknown model level {level!r} in {model_mode!r}')
        mode_rate.append(MODEL_SPLIT_RATE[level])
        proportion.append(int(weight))
    return mode_rate, proportion


def process_control(cfg):
    """Expand cfg['control'] into derived keys, in place.

    Matches reference src/utils.py:113-215 exactly: fix mode partitions the
    user list by integer proportions (remainder users get the last rate);
    dynamic mode stores the normalized proportions for per-round multinomial
    resampling.
    """
    cfg['model_split_rate'] = dict(MODEL_SPLIT_RATE)
    ctl = cfg['control']
    cfg['fed'] = int(ctl['fed'])
    cfg['num_users'] = int(ctl['num_users'])
    cfg['frac'] = float(ctl['frac'])
    cfg['data_split_mode'] = ctl['data_split_mode']
    cfg['model_split_mode'] = ctl['model_split_mode']
    cfg['model_mode'] = ctl['model_mode']
    cfg['norm'] = ctl['norm']
    cfg['scale'] = bool(int(ctl['scale']))
    cfg['mask'] = bool(int(ctl['mask']))
    cfg['global_model_mode'] = cfg['model_mode'][0]
    cfg['global_model_rate'] = cfg['model_split_rate'][cfg['global_model_mode']]
    mode_rate, proportion = parse_model_mode(cfg['model_mode'])
    if cfg['model_split_mode'] == 'dynamic':
        cfg['model_rate'] = mode_rate
        cfg['proportion'] = (np.array(proportion) / sum(proportion)).tolist()
    elif cfg['model_split_mode'] == 'fix':
        num_users_proportion = cfg['num_users'] // sum(proportion)
        rates = []
        for r, p in zip(mode_rate, proportion):
            rates += [r] * (num_users_proportion * p)
        rates += [rates[-1]] * (cfg['num_users'] - len(rates))
        cfg['model_rate'] = rates
    else:
        raise ValueError('Not valid model split mode')
    # architecture widths (reference: src/utils.py:147-149)
    cfg['conv'] = {'hidden_size': [64, 128, 256, 512]}
    cfg['resnet'] = {'hidden_size': [64, 128, 256, 512]}
    cfg['transformer'] = {'embedding_size': 256, 'num_heads': 8,
                          'hidden_size': 512, 'num_layers': 4, 'dropout': 0.2}
    _dataset_hyperparameters(cfg)
    return cfg


def _dataset_hyperparameters(cfg):
    """Per-dataset optimizer/schedule/shape defaults
    (reference: src/utils.py:150-212)."""
    name = cfg['data_name']
    split = cfg['data_split_mode']
    if name in ('MNIST', 'FashionMNIST', 'EMNIST'):
        cfg['data_shape'] = [1, 28, 28]
        cfg.update(optimizer_name='SGD', lr=1e-2, momentum=0.9,
                   weight_decay=5e-4, scheduler_name='MultiStepLR', factor=0.1)
        if split == 'iid':
            cfg['num_epochs'] = {'global': 200, 'local': 5}
            cfg['batch_size'] = {'train': 10, 'test': 50}
            cfg['milestones'] = [100]
        elif 'non-iid' in split:
            cfg['num_epochs'] = {'global': 400, 'local': 5}
            cfg['batch_size'] = {'train': 10, 'test': 50}
            cfg['milestones'] = [200]
        elif split == 'none':
            cfg['num_epochs'] = 200
            cfg['batch_size'] = {'train': 100, 'test': 500}
            cfg['milestones'] = [100]
        else:
            raise ValueError('Not valid data_split_mode')
    elif name in ('CIFAR10', 'CIFAR100'):
        cfg['data_shape'] = [3, 32, 32]
        cfg.update(optimizer_name='SGD', lr=1e-1, momentum=0.9,
                   weight_decay=5e-4, scheduler_name='MultiStepLR', factor=0.1)
        if split == 'iid':
            cfg['num_epochs'] = {'global': 400, 'local': 5}
            cfg['batch_size'] = {'train': 10, 'test': 50}
            cfg['milestones'] = [150, 250]
        elif 'non-iid' in split:
            cfg['num_epochs'] = {'global': 800, 'local': 5}
            cfg['batch_size'] = {'train': 10, 'test': 50}
            cfg['milestones'] = [300, 500]
        elif split == 'none':
            cfg['num_epochs'] = 400
            cfg['batch_size'] = {'train': 100, 'test': 500}
            cfg['milestones'] = [150, 250]
        else:
            raise ValueError('Not valid data_split_mode')
    elif name in ('PennTreebank', 'WikiText2', 'WikiText103'):
        cfg.update(optimizer_name='SGD', lr=1e-1, momentum=0.9,
                   weight_decay=5e-4, scheduler_name='MultiStepLR', factor=0.1,
                   bptt=64, mask_rate=0.15)
        if split == 'iid':
            cfg['num_epochs'] = {'global': 200, 'local': 1}
            cfg['batch_size'] = {'train': 100, 'test': 10}
            cfg['milestones'] = [50, 100]
        elif split == 'none':
            cfg['num_epochs'] = 100
            cfg['batch_size'] = {'train': 100, 'test': 100}
            cfg['milestones'] = [25, 50]
        else:
            raise ValueError('Not valid data_split_mode')
    else:
        raise ValueError('Not valid dataset')
