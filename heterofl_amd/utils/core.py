"""Generic utilities: persistence, tree-mapped device moves, optimizer and
scheduler factories, checkpoint resume (reference: src/utils.py:27-357).
"""
import errno
import os

import torch
import torch.optim as optim


def makedir_exist_ok(path):
    try:
        os.makedirs(path)
    except OSError as e:
        if e.errno != errno.EEXIST:
            raise


def save(obj, path, protocol=2):
    dirname = os.path.dirname(path)
    if dirname:
        makedir_exist_ok(dirname)
    torch.save(obj, path, pickle_protocol=protocol)


def load(path):
    return torch.load(path, map_location=lambda storage, loc: storage,
                      weights_only=False)


def recur(fn, x, *args):
    """Apply fn to every tensor / leaf of a nested structure
    (reference: src/utils.py:56-97)."""
    if isinstance(x, torch.Tensor):
        return fn(x, *args)
    if isinstance(x, (list, tuple)):
        out = [recur(fn, v, *args) for v in x]
        return type(x)(out) if isinstance(x, tuple) else out
    if isinstance(x, dict):
        return {k: recur(fn, v, *args) for k, v in x.items()}
    if isinstance(x, (str, int, float, complex)) or x is None:
        return x
    return x


def to_device(x, device):
    return recur(lambda t: t.to(device, non_blocking=True), x)


def collate(input):
    """Stack the dict-of-lists produced by input_collate
    (reference: src/utils.py:347-350)."""
    for k in input:
        if isinstance(input[k], list) and input[k] and isinstance(input[k][0], torch.Tensor):
            input[k] = torch.stack(input[k], 0)
    return input


def batchify(dataset, batch_size):
    """Fold an LM token stream to (batch_size, -1)
    (reference: src/utils.py:353-357)."""
    num_batch = len(dataset) // batch_size
    dataset.token = dataset.token.narrow(0, 0, num_batch * batch_size)
    dataset.token = dataset.token.reshape(batch_size, -1)
    return dataset


def process_dataset(dataset, cfg):
    """Record classes_size / vocab on cfg and batchify LM streams
    (reference: src/utils.py:100-110)."""
    if cfg['data_name'] in ('MNIST', 'FashionMNIST', 'EMNIST', 'CIFAR10',
                            'CIFAR100'):
        cfg['classes_size'] = dataset['train'].classes_size
    elif cfg['data_name'] in ('PennTreebank', 'WikiText2', 'WikiText103'):
        cfg['vocab'] = dataset['train'].vocab
        cfg['num_tokens'] = len(dataset['train'].vocab)
        for split in dataset:
            dataset[split] = batchify(dataset[split], cfg['batch_size'][split])
    else:
        raise ValueError('Not valid data name')


def make_optimizer(model, lr, cfg):
    """SGD / RMSprop / Adam / Adamax factory (reference: src/utils.py:260-273)."""
    params = model.parameters() if hasattr(model, 'parameters') else model
    name = cfg['optimizer_name']
    if name == 'SGD':
        return optim.SGD(params, lr=lr, momentum=cfg['momentum'],
                         weight_decay=cfg['weight_decay'])
    if name == 'RMSprop':
        return optim.RMSprop(params, lr=lr, momentum=cfg['momentum'],
                             weight_decay=cfg['weight_decay'])
    if name == 'Adam':
        return optim.Adam(params, lr=lr, betas=(0.9, 0.999),
                          weight_decay=cfg['weight_decay'])
    if name == 'Adamax':
        return optim.Adamax(params, lr=lr, betas=(0.9, 0.999),
                            weight_decay=cfg['weight_decay'])
    raise ValueError('Not valid optimizer name')


def make_scheduler(optimizer, cfg):
    """Seven scheduler kinds (reference: src/utils.py:276-297)."""
    name = cfg['scheduler_name']
    if name == 'None':
        return optim.lr_scheduler.MultiStepLR(optimizer, milestones=[65535])
    if name == 'StepLR':
        return optim.lr_scheduler.StepLR(optimizer, step_size=cfg['step_size'],
                                         gamma=cfg['factor'])
    if name == 'MultiStepLR':
        return optim.lr_scheduler.MultiStepLR(optimizer,
                                              milestones=cfg['milestones'],
                                              gamma=cfg['factor'])
    if name == 'ExponentialLR':
        return optim.lr_scheduler.ExponentialLR(optimizer, gamma=0.99)
    if name == 'CosineAnnealingLR':
        num_epochs = cfg['num_epochs']
        if isinstance(num_epochs, dict):
            num_epochs = num_epochs['global']
        return optim.lr_scheduler.CosineAnnealingLR(optimizer, T_max=num_epochs,
                                                    eta_min=0)
    if name == 'ReduceLROnPlateau':
        return optim.lr_scheduler.ReduceLROnPlateau(
            optimizer, mode='max', factor=cfg['factor'],
            patience=cfg['patience'], threshold=cfg['threshold'],
            threshold_mode='rel', min_lr=cfg['min_lr'])
    if name == 'CyclicLR':
        return optim.lr_scheduler.CyclicLR(optimizer, base_lr=cfg['lr'],
                                           max_lr=10 * cfg['lr'])
    raise ValueError('Not valid scheduler name')


def model_tag_of(seed, cfg):
    """'{seed}_{data}_{subset}_{model}_{control_name}'
    (reference: src/train_classifier_fed.py:41-42)."""
    parts = [str(seed), cfg['data_name'], cfg['subset'], cfg['model_name'],
             cfg.get('control_name', '')]
    return '_'.join([p for p in parts if p])


def resume(model, model_tag, optimizer=None, scheduler=None, load_tag='checkpoint',
           strict=True, verbose=True):
    """Load ./output/model/{tag}_{load_tag}.pt
    (reference: src/utils.py:300-344).  Returns
    (last_epoch, data_split, label_split, model, optimizer, scheduler, logger).
    """
    from ..logger import Logger
    path = './output/model/{}_{}.pt'.format(model_tag, load_tag)
    if os.path.exists(path):
        ckpt = load(path)
        last_epoch = ckpt['epoch']
        data_split = ckpt['data_split']
        label_split = ckpt['label_split']
        model.load_state_dict(ckpt['model_dict'], strict=strict)
        if optimizer is not None:
            optimizer.load_state_dict(ckpt['optimizer_dict'])
        if scheduler is not None:
            scheduler.load_state_dict(ckpt['scheduler_dict'])
        logger = ckpt['logger']
        if verbose:
            print('Resume from {}'.format(last_epoch))
    else:
        last_epoch = 1
        data_split = None
        label_split = None
        logger = Logger(os.path.join('output', 'runs', 'train_{}'.format(model_tag)))
        if verbose:
            print('Not exists model tag: {}, start from scratch'.format(model_tag))
    return last_epoch, data_split, label_split, model, optimizer, scheduler, logger


class Stats:
    """Running count/mean/std accumulator (reference: src/utils.py:231-257)."""

    def __init__(self, dim=0):
        self.dim = dim
        self.n_samples = 0
        self.mean = None
        self.std = None

    def update(self, data):
        collapse = [i for i in range(data.dim()) if i != self.dim]
        n = int(torch.tensor(data.size())[collapse].prod().item()) if collapse else data.size(0)
        mean = data.mean(dim=collapse) if collapse else data
        if self.n_samples == 0:
            self.n_samples = n
            self.mean = mean
            self.std = data.std(dim=collapse, unbiased=False) if collapse else torch.zeros_like(mean)
        else:
            m = self.n_samples
            old_mean = self.mean
            new_mean = (m * old_mean + n * mean) / (m + n)
            var_new = data.var(dim=collapse, unbiased=False) if collapse else torch.zeros_like(mean)
            var = (m * (self.std ** 2) + n * var_new + m * (old_mean - new_mean) ** 2
                   + n * (mean - new_mean) ** 2) / (m + n)
            self.mean = new_mean
            self.std = var.sqrt()
            self.n_samples = m + n
        return self
