"""Data layer: dataset fetch, federated splitting, loaders.

Reference semantics (src/data.py:10-150, src/datasets/): MNIST/CIFAR vision
datasets with torchvision-style normalize/augment, WikiText-style token
streams with a Vocab, `iid` / `non-iid-N` federated splits that record each
user's label set, `SplitDataset` index views and `BatchDataset` bptt windows
over the batchified token matrix.

MI355X-native design notes: the fast training path (fed/batched.py) stages
each client's shard as a device-resident uint8 tensor and augments on-GPU, so
the Dataset classes here store raw uint8 `img` / int64 `token` tensors and
only the sequential oracle path pays per-item host transforms.  There is no
network in the build/bench environment, so every dataset has a deterministic
synthetic mode of the real shape (bench.py runs on synthetic data); real
on-disk raw files (MNIST idx / CIFAR pickle / WikiText tokens) are parsed
when present under ./data/<name>.
"""
import gzip
import os
import pickle
import struct

import numpy as np
import torch
from torch.utils.data import Dataset
from torch.utils.data.dataloader import default_collate

_NORM = {
    'MNIST': ((0.1307,), (0.3081,)),
    'FashionMNIST': ((0.2860,), (0.3530,)),
    'CIFAR10': ((0.4914, 0.4822, 0.4465), (0.2023, 0.1994, 0.2010)),
    'CIFAR100': ((0.5071, 0.4865, 0.4409), (0.2673, 0.2564, 0.2762)),
}

_REAL_SIZE = {  # (train, test) sample counts of the real datasets
    'MNIST': (60000, 10000), 'FashionMNIST': (60000, 10000),
    'EMNIST': (112800, 18800),  # balanced split
    'CIFAR10': (50000, 10000), 'CIFAR100': (50000, 10000),
}


class Vocab:
    """Token vocabulary (reference: src/datasets/lm.py Vocab)."""

    def __init__(self, tokens=None):
        self.itos = list(tokens) if tokens is not None else []
        self.stoi = {t: i for i, t in enumerate(self.itos)}

    def add(self, token):
        if token not in self.stoi:
            self.stoi[token] = len(self.itos)
            self.itos.append(token)
        return self.stoi[token]

    def __len__(self):
        return len(self.itos)

    def __getitem__(self, token):
        return self.stoi.get(token, self.stoi.get('<unk>', 0))


class VisionDataset(Dataset):
    """Image dataset holding raw uint8 pixels.

    `img`: uint8 tensor (N, H, W) or (N, H, W, C); `target`: list[int];
    items are normalized float CHW dicts {'img','label'} with train-time
    random-crop(pad 4)+flip for CIFAR (reference: src/data.py:14-27).
    """

    def __init__(self, data_name, img, target, classes_size, train):
        self.data_name = data_name
        self.img = img
        self.target = target
        self.classes_size = classes_size
        self.train = train
        mean, std = _NORM.get(data_name, ((0.5,), (0.5,)))
        self._mean = torch.tensor(mean).view(-1, 1, 1)
        self._std = torch.tensor(std).view(-1, 1, 1)
        self._augment = train and data_name in ('CIFAR10', 'CIFAR100')

    def __len__(self):
        return self.img.size(0)

    def __getitem__(self, index):
        img = self.img[index]
        if img.dim() == 2:
            img = img.unsqueeze(-1)
        x = img.permute(2, 0, 1).float().div(255.0)
        if self._augment:
            C, H, W = x.shape
            pad = 4
            xp = torch.nn.functional.pad(x, [pad] * 4)
            dy = int(torch.randint(0, 2 * pad + 1, (1,)))
            dx = int(torch.randint(0, 2 * pad + 1, (1,)))
            x = xp[:, dy:dy + H, dx:dx + W]
            if torch.rand(1).item() < 0.5:
                x = x.flip(-1)
        x = (x - self._mean) / self._std
        return {'img': x, 'label': torch.tensor(self.target[index])}


class LanguageModeling(Dataset):
    """Token-stream dataset.  Before utils.batchify, `token` is the 1-D
    stream; after, it is the (batch_size, stream_len) matrix and items are
    its rows (reference: src/datasets/lm.py + src/utils.py:353-357)."""

    def __init__(self, data_name, token, vocab):
        self.data_name = data_name
        self.token = token
        self.vocab = vocab

    def __len__(self):
        return self.token.size(0)

    def __getitem__(self, index):
        return {'label': self.token[index]}


# ----------------------------------------------------------------- parsers
def _read_idx(path):
    """MNIST idx file (optionally gzipped) -> numpy array
    (reference behavior: src/datasets/mnist.py:12-180)."""
    opener = gzip.open if path.endswith('.gz') else open
    with opener(path, 'rb') as f:
        magic = struct.unpack('>I', f.read(4))[0]
        ndim = magic & 0xFF
        shape = struct.unpack('>' + 'I' * ndim, f.read(4 * ndim))
        data = np.frombuffer(f.read(), dtype=np.uint8)
    return data.reshape(shape)


def _load_mnist_raw(root, split, data_name='MNIST'):
    prefix = 'train' if split == 'train' else 't10k'
    raw = os.path.join(root, 'raw')

    def find(stems):
        for stem in stems:
            for ext in ('', '.gz'):
                p = os.path.join(raw, stem + ext)
                if os.path.exists(p):
                    return p
        return None

    img_stems = [f'{prefix}-images-idx3-ubyte']
    lbl_stems = [f'{prefix}-labels-idx1-ubyte']
    if data_name == 'EMNIST':
        # EMNIST raw files are named emnist-<subset>-{train,test}-*
        # (the reference uses the balanced split, src/datasets/mnist.py)
        word = 'train' if split == 'train' else 'test'
        for subset in ('balanced', 'byclass', 'bymerge', 'digits',
                       'letters', 'mnist'):
            img_stems.append(f'emnist-{subset}-{word}-images-idx3-ubyte')
            lbl_stems.append(f'emnist-{subset}-{word}-labels-idx1-ubyte')
    ip = find(img_stems)
    lp = find(lbl_stems)
    if ip is None or lp is None:
        return None
    img = torch.from_numpy(_read_idx(ip).copy())
    target = _read_idx(lp).astype(np.int64).tolist()
    return img, target


def _load_cifar_raw(root, split, name):
    """CIFAR python-pickle batches (reference: src/datasets/cifar.py:12-143)."""
    sub = {'CIFAR10': 'cifar-10-batches-py', 'CIFAR100': 'cifar-100-python'}[name]
    base = os.path.join(root, sub)
    if not os.path.isdir(base):
        return None
    if name == 'CIFAR10':
        files = [f'data_batch_{i}' for i in range(1, 6)] if split == 'train' \
            else ['test_batch']
        label_key = b'labels'
    else:
        files = ['train'] if split == 'train' else ['test']
        label_key = b'fine_labels'
    imgs, targets = [], []
    for fn in files:
        p = os.path.join(base, fn)
        if not os.path.exists(p):
            return None
        with open(p, 'rb') as f:
            d = pickle.load(f, encoding='bytes')
        imgs.append(d[b'data'].reshape(-1, 3, 32, 32).transpose(0, 2, 3, 1))
        targets.extend(int(t) for t in d[label_key])
    return torch.from_numpy(np.concatenate(imgs).copy()), targets


_IMG_EXTENSIONS = ('.png', '.jpg', '.jpeg', '.bmp', '.ppm', '.pgm', '.webp')


def _load_folder_raw(root, split, size=None):
    """Class-per-subdirectory image tree -> (uint8 (N,H,W,C) tensor, targets,
    classes).  Covers the reference's ImageFolder/ImageNet/Omniglot family
    (reference: src/datasets/folder.py:9-61, src/datasets/imagenet.py:14-60):
    labels come from the sorted subdirectory names of ``root/<split>``; images
    are decoded with PIL and resized to a common size (``size`` or the first
    image's size)."""
    base = os.path.join(root, split)
    if not os.path.isdir(base):
        return None
    try:
        from PIL import Image
    except ImportError:
        return None
    classes = sorted(d for d in os.listdir(base)
                     if os.path.isdir(os.path.join(base, d)))
    if not classes:
        return None
    imgs, targets = [], []
    for label, cls in enumerate(classes):
        cdir = os.path.join(base, cls)
        for fn in sorted(os.listdir(cdir)):
            if not fn.lower().endswith(_IMG_EXTENSIONS):
                continue
            with Image.open(os.path.join(cdir, fn)) as im:
                im = im.convert('RGB')
                if size is None:
                    size = im.size
                if im.size != size:
                    im = im.resize(size)
                imgs.append(np.asarray(im, dtype=np.uint8))
            targets.append(label)
    if not imgs:
        return None
    return torch.from_numpy(np.stack(imgs)), targets, classes


def _load_wikitext_raw(root, split):
    for fn in (f'wiki.{split}.tokens', f'{split}.txt'):
        p = os.path.join(root, fn)
        if os.path.exists(p):
            with open(p, encoding='utf-8') as f:
                return f.read().split()
    return None


# --------------------------------------------------------------- synthetic
def _synthetic_vision(data_name, n, classes_size, seed):
    if os.environ.get('HETEROFL_SYNTHETIC_MODE') == 'learnable':
        return _learnable_vision(data_name, n, classes_size, seed)
    g = torch.Generator().manual_seed(seed)
    if data_name in ('MNIST', 'FashionMNIST', 'EMNIST'):
        img = torch.randint(0, 256, (n, 28, 28), dtype=torch.uint8, generator=g)
    else:
        img = torch.randint(0, 256, (n, 32, 32, 3), dtype=torch.uint8, generator=g)
    target = torch.randint(0, classes_size, (n,), generator=g).tolist()
    return img, target


def _learnable_vision(data_name, n, classes_size, seed):
    """Learnable class-conditional synthetic images (the no-network proxy
    for the BASELINE accuracy study — there is no way to fetch real CIFAR
    raw files in this environment).

    Each class owns a bank of 64 smooth random templates (low-pass noise
    upsampled to full resolution, shared between train/test); a sample is a
    random class template with amplitude jitter, a random circular shift and
    dense pixel noise.  A CNN must learn the 640 template shapes under
    noise/shift/crop — pixel means alone do not separate classes — so the
    federated training dynamics (sBN, lr schedule, heterogeneous widths,
    masked CE, augmentation) all matter, while the problem stays solvable to
    high accuracy when training is healthy."""
    import torch.nn.functional as F
    if data_name in ('MNIST', 'FashionMNIST', 'EMNIST'):
        C, H, W = 1, 28, 28
    else:
        C, H, W = 3, 32, 32
    T = 64
    gb = torch.Generator().manual_seed(777)   # class bank: split-invariant
    bank = torch.randn(classes_size * T, C, 8, 8, generator=gb)
    bank = F.interpolate(bank, size=(H, W), mode='bilinear',
                         align_corners=False)
    bank = (bank - bank.mean(dim=(1, 2, 3), keepdim=True)) \
        / bank.std(dim=(1, 2, 3), keepdim=True).clamp(min=1e-6)
    bank = bank.view(classes_size, T, C, H, W)
    gs = torch.Generator().manual_seed(10_000 + seed)  # draws: per split
    y = torch.randint(0, classes_size, (n,), generator=gs)
    t = torch.randint(0, T, (n,), generator=gs)
    amp = 0.7 + 0.6 * torch.rand(n, generator=gs)
    dy = torch.randint(-4, 5, (n,), generator=gs)
    dx = torch.randint(-4, 5, (n,), generator=gs)
    out = torch.empty(n, C, H, W, dtype=torch.uint8)
    chunk = 4096
    for s in range(0, n, chunk):
        e = min(s + chunk, n)
        x = bank[y[s:e], t[s:e]] * amp[s:e].view(-1, 1, 1, 1)
        x = x + 0.75 * torch.randn(x.shape, generator=gs)
        # group samples by shift so each group rolls at once
        key = (dy[s:e] + 4) * 9 + (dx[s:e] + 4)
        for kv in key.unique().tolist():
            sel = (key == kv).nonzero(as_tuple=True)[0]
            sdy, sdx = kv // 9 - 4, kv % 9 - 4
            if sdy or sdx:
                x[sel] = torch.roll(x[sel], shifts=(sdy, sdx), dims=(2, 3))
        out[s:e] = (128 + 48 * x).clamp(0, 255).to(torch.uint8)
    img = out.permute(0, 2, 3, 1)
    if C == 1:
        img = img.squeeze(-1)
    return img.contiguous(), y.tolist()


def fetch_dataset(data_name, subset='label', synthetic=False,
                  synthetic_size=None, root=None):
    """Return {'train': ds, 'test': ds} (reference: src/data.py:10-34).

    synthetic=True builds deterministic random data of the real dataset's
    shape (synthetic_size overrides the train-split sample count; test gets
    max(size//5, 50) samples).  Real raw files are used when present under
    root (default ./data/<name>)."""
    root = root or os.path.join('.', 'data', data_name)
    dataset = {}
    if data_name in ('MNIST', 'FashionMNIST', 'EMNIST', 'CIFAR10', 'CIFAR100'):
        classes_size = {'CIFAR100': 100, 'EMNIST': 47}.get(data_name, 10)
        for split in ('train', 'test'):
            raw = None
            if not synthetic:
                if data_name in ('MNIST', 'FashionMNIST', 'EMNIST'):
                    raw = _load_mnist_raw(root, split, data_name)
                else:
                    raw = _load_cifar_raw(root, split, data_name)
                if raw is None:
                    raise FileNotFoundError(
                        f'{data_name} raw files not found under {root}; '
                        f'pass synthetic=True in network-less environments')
            if raw is None:
                n_train, n_test = _REAL_SIZE[data_name]
                if synthetic_size is not None:
                    n_train = synthetic_size
                    n_test = max(synthetic_size // 5, 50)
                n = n_train if split == 'train' else n_test
                raw = _synthetic_vision(data_name, n, classes_size,
                                        seed=0 if split == 'train' else 1)
            img, target = raw
            dataset[split] = VisionDataset(data_name, img, target,
                                           classes_size, split == 'train')
    elif data_name in ('ImageFolder', 'ImageNet', 'Omniglot'):
        # folder-tree image datasets (reference: src/datasets/folder.py,
        # imagenet.py, omniglot.py — defined but unused by the three
        # benchmark tasks; supported here for parity)
        train_raw = _load_folder_raw(root, 'train')
        if train_raw is None:
            if not synthetic:
                raise FileNotFoundError(
                    f'{data_name} folder tree not found under {root}; expects '
                    f'{root}/train/<class>/*.png (pass synthetic=True '
                    f'otherwise)')
            classes_size = 10
            for split in ('train', 'test'):
                n = synthetic_size or 1000
                if split == 'test':
                    n = max(n // 5, 50)
                img, target = _synthetic_vision('CIFAR10', n, classes_size,
                                                seed=0 if split == 'train' else 1)
                dataset[split] = VisionDataset(data_name, img, target,
                                               classes_size, split == 'train')
        else:
            img, target, classes = train_raw
            classes_size = len(classes)
            dataset['train'] = VisionDataset(data_name, img, target,
                                             classes_size, True)
            test_raw = _load_folder_raw(root, 'test',
                                        size=(img.size(2), img.size(1)))
            if test_raw is None:
                test_raw = train_raw
            timg, ttarget = test_raw[0], test_raw[1]
            dataset['test'] = VisionDataset(data_name, timg, ttarget,
                                            classes_size, False)
    elif data_name in ('PennTreebank', 'WikiText2', 'WikiText103'):
        vocab = None
        for split in ('train', 'test'):
            tokens = None if synthetic else _load_wikitext_raw(root, split)
            if tokens is not None:
                if vocab is None:
                    vocab = Vocab(['<unk>'])
                    for t in tokens:
                        vocab.add(t)
                stream = torch.tensor([vocab[t] for t in tokens], dtype=torch.long)
            else:
                if not synthetic:
                    raise FileNotFoundError(
                        f'{data_name} raw files not found under {root}; '
                        f'pass synthetic=True in network-less environments')
                if vocab is None:
                    # real WikiText2 vocab is ~33k; keep that scale by default
                    v = 33278 if data_name == 'WikiText2' else 10000
                    vocab = Vocab([f'tok{i}' for i in range(v)])
                n = synthetic_size or 2_000_000
                if split == 'test':
                    n = max(n // 8, 4096)
                g = torch.Generator().manual_seed(0 if split == 'train' else 1)
                stream = torch.randint(0, len(vocab), (n,), dtype=torch.long,
                                       generator=g)
            dataset[split] = LanguageModeling(data_name, stream, vocab)
    else:
        raise ValueError('Not valid dataset name')
    return dataset


# ------------------------------------------------------------------ splits
def input_collate(batch):
    """dict-of-lists collate (reference: src/data.py:37-45); utils.collate
    stacks the lists afterwards."""
    if isinstance(batch[0], dict):
        out = {k: [] for k in batch[0]}
        for b in batch:
            for k in b:
                out[k].append(b[k])
        return out
    return default_collate(batch)


def split_dataset(dataset, num_users, data_split_mode, classes_size=None):
    """Federated split (reference: src/data.py:48-110).  Returns
    (data_split {'train': {user: [idx]}, 'test': {...}}, label_split
    {user: [labels]})."""
    data_split = {}
    if data_split_mode == 'iid':
        data_split['train'], label_split = _iid(dataset['train'], num_users)
        data_split['test'], _ = _iid(dataset['test'], num_users)
    elif data_split_mode.startswith('non-iid'):
        shard_per_user = int(data_split_mode.split('-')[-1])
        data_split['train'], label_split = _non_iid(
            dataset['train'], num_users, shard_per_user, classes_size)
        data_split['test'], _ = _non_iid(
            dataset['test'], num_users, shard_per_user, classes_size,
            label_split)
    else:
        raise ValueError('Not valid data split mode')
    return data_split, label_split


def _labels_of(dataset):
    if isinstance(dataset, LanguageModeling):
        return dataset.token  # rows of the batchified matrix
    return torch.tensor(dataset.target)


def _iid(dataset, num_users):
    """Equal random shards; records each user's label set
    (reference: src/data.py:61-76)."""
    label = _labels_of(dataset)
    num_items = len(dataset) // num_users
    idx = torch.randperm(len(dataset))
    data_split, label_split = {}, {}
    for i in range(num_users):
        take = idx[i * num_items:(i + 1) * num_items]
        data_split[i] = take.tolist()
        label_split[i] = torch.unique(label[take]).tolist()
    return data_split, label_split


def _non_iid(dataset, num_users, shard_per_user, classes_size,
             label_split=None):
    """'non-iid-N': each user draws N class shards; per-class index pools
    are cut into shard_per_class equal shards (reference: src/data.py:79-110).
    """
    target = np.array(dataset.target)
    by_class = {}
    for i, t in enumerate(target):
        by_class.setdefault(int(t), []).append(i)
    shard_per_class = int(shard_per_user * num_users / classes_size)
    shards = {}
    for c, idxs in by_class.items():
        n_keep = (len(idxs) // shard_per_class) * shard_per_class
        cut = np.array(idxs[:n_keep]).reshape(shard_per_class, -1).tolist()
        for j, extra in enumerate(idxs[n_keep:]):
            cut[j % len(cut)].append(extra)
        shards[c] = cut
    if label_split is None:
        pool = torch.tensor(list(range(classes_size)) * shard_per_class)
        pool = pool[torch.randperm(len(pool))].tolist()
        label_split = np.array(pool).reshape((num_users, -1)).tolist()
        label_split = [np.unique(ls).tolist() for ls in label_split]
    data_split = {i: [] for i in range(num_users)}
    for i in range(num_users):
        for c in label_split[i]:
            if shards.get(c):
                j = int(torch.randint(0, len(shards[c]), (1,)))
                data_split[i].extend(shards[c].pop(j))
    return data_split, label_split


def make_data_loader(dataset, cfg):
    """DataLoaders with dict collate (reference: src/data.py:113-119)."""
    out = {}
    for k in dataset:
        out[k] = torch.utils.data.DataLoader(
            dataset=dataset[k], shuffle=cfg['shuffle'][k],
            batch_size=cfg['batch_size'][k], pin_memory=False,
            num_workers=cfg['num_workers'], collate_fn=input_collate)
    return out


class SplitDataset(Dataset):
    """Index view over a parent dataset (reference: src/data.py:122-133).
    For LM the index may be a slice/list (BatchDataset slices rows)."""

    def __init__(self, dataset, idx):
        self.dataset = dataset
        self.idx = idx

    def __len__(self):
        return len(self.idx)

    def __getitem__(self, index):
        return self.dataset[self.idx[index]]


class BatchDataset(Dataset):
    """bptt windows over the batchified token matrix
    (reference: src/data.py:136-150)."""

    def __init__(self, dataset, seq_length):
        self.dataset = dataset
        self.seq_length = seq_length
        self.S = dataset[0]['label'].size(0)
        self.idx = list(range(0, self.S, seq_length))

    def __len__(self):
        return len(self.idx)

    def __getitem__(self, index):
        start = self.idx[index]
        seq = min(self.seq_length, self.S - start)
        return {'label': self.dataset[:]['label'][:, start:start + seq]}
