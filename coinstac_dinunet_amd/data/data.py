"""Data layer: dataset, data handle (stateful loader cursor), padded sampler.

API-parity with /root/reference/coinstac_dinunet/data/data.py:23-242
(safe_collate, COINNDataset, COINNDataHandle, COINNPaddedDataSampler):
same constructor/method signatures and cache-key protocol ('cursor',
'data_len', 'train_loader_iter', cache['dataset'] registry), so reference
user computations drop in unchanged. Implementation is our own; the padded
sampler additionally accepts an explicit total_size so the RCCL engine can
equalize batch COUNT across ranks with unequal per-site data.
"""
import json as _json
import math as _math
import os as _os
import random as _random

import numpy as _np
import torch as _torch
from torch.utils.data import DataLoader as _DataLoader, Dataset as _Dataset
from torch.utils.data._utils.collate import default_collate as _default_collate

import coinstac_dinunet_amd.config as _conf
from ..config.keys import Mode
from ..utils import FrozenDict as _FrozenDict
from ..utils.logger import success
from .datautils import init_k_folds as _kfolds

_sep = _os.sep


def safe_collate(batch):
    """Skip falsy (corrupt) items before collating."""
    return _default_collate([b for b in batch if b])


def _seed_worker(worker_id):
    seed = (int(_torch.initial_seed()) + worker_id) % (2 ** 32 - 1)
    _np.random.seed(seed)
    _random.seed(seed)


class COINNDataset(_Dataset):
    """Abstract dataset; user implements load_index(file) and __getitem__."""

    def __init__(self, mode='init', cache=None, input=None, state=None,
                 limit=_conf.max_size):
        self.mode = mode
        self.limit = limit if limit is not None else _conf.max_size
        self.cache = cache if cache is not None else {}
        self.input = input if input is not None else {}
        self.state = state if state is not None else {}
        self.indices = []

    def load_index(self, file):
        """One file -> >=1 entries in self.indices (override for patches)."""
        self.indices.append([file])

    def _load_indices(self, files, **kw):
        for file in files:
            if len(self) >= self.limit:
                break
            self.load_index(file)
        if kw.get('verbose'):
            print(f'{self.mode}, {len(self)} indices loaded')

    def __getitem__(self, index):
        raise NotImplementedError('Must be implemented by child class.')

    def __len__(self):
        return len(self.indices)

    def transforms(self, **kw):
        return None

    def path(self, root_dir='baseDirectory', cache_key='_N/A_'):
        return _os.path.join(self.state[root_dir], self.cache.get(cache_key, ''))

    def add(self, files):
        self._load_indices(files=files, verbose=False)


class COINNPaddedDataSampler:
    """Wrap-pad indices to a whole number of batches (optionally to an
    externally agreed total_size) so lock-step sites emit equal batch counts.
    Seeded shuffle per epoch."""

    def __init__(self, dataset, batch_size, seed=0, shuffle=False,
                 drop_last=False, total_size=None):
        self.dataset = dataset
        batch_size = int(batch_size)
        if total_size is not None:
            self.total_size = int(total_size)
        elif drop_last:
            self.total_size = _math.floor(len(dataset) / batch_size) * batch_size
        else:
            self.total_size = _math.ceil(len(dataset) / batch_size) * batch_size
        self.drop_last = drop_last
        self.shuffle = shuffle
        self.epoch = 0
        self.seed = seed if seed is not None else 0

    def __iter__(self):
        if self.shuffle:
            g = _torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = _torch.randperm(len(self.dataset), generator=g).tolist()
        else:
            indices = list(range(len(self.dataset)))
        if len(indices) == 0:
            return iter(())
        if not self.drop_last:
            pad = self.total_size - len(indices)
            if pad > 0:
                reps = _math.ceil(pad / len(indices))
                indices += (indices * reps)[:pad]
        indices = indices[:self.total_size]
        return iter(indices)

    def set_epoch(self, epoch):
        self.epoch = epoch

    def __len__(self):
        return int(self.total_size)


class COINNDataHandle:
    """Owns the cache['dataset'] registry, split plumbing and the stateful
    cursor iterator driving gradient-accumulation rounds."""

    def __init__(self, cache=None, input=None, state=None, dataloader_args=None,
                 **kw):
        self.cache = cache if cache is not None else {}
        self.input = input if input is not None else {}
        self.state = state if state is not None else {}
        self.dataset = self.cache.setdefault('dataset', {})
        self.dataloader_args = _FrozenDict(
            self.cache.get('dataloader_args') or dataloader_args or {})

    # ---- datasets ------------------------------------------------------
    def get_dataset(self, handle_key, files, dataset_cls=None):
        dataset = dataset_cls(mode=handle_key, cache=self.cache,
                              input=self.input, state=self.state,
                              limit=self.cache.get('load_limit', _conf.max_size))
        dataset.add(files=files)
        self.dataset[handle_key] = dataset if len(dataset) > 0 else None
        return self.dataset[handle_key]

    def _split(self):
        with open(_os.path.join(self.cache['split_dir'],
                                self.cache['split_file'])) as f:
            return _json.load(f)

    def get_train_dataset(self, dataset_cls):
        injected = self.dataloader_args.get('train', {}).get('dataset')
        if dataset_cls is None or injected:
            return injected
        return self.get_dataset('train', self._split().get('train', []),
                                dataset_cls=dataset_cls)

    def get_validation_dataset(self, dataset_cls):
        injected = self.dataloader_args.get('validation', {}).get('dataset')
        if dataset_cls is None or injected:
            return injected
        d = self.get_dataset('validation', self._split().get('validation', []),
                             dataset_cls=dataset_cls)
        if d and len(d) > 0:
            return d

    def get_test_dataset(self, dataset_cls):
        injected = self.dataloader_args.get('test', {}).get('dataset')
        if dataset_cls is None or injected:
            return injected
        files = self._split().get('test', [])[:self.cache.get('load_limit',
                                                              _conf.max_size)]
        if self.cache.get('load_sparse') and len(files) > 1:
            datasets = [self.get_dataset('test', [f], dataset_cls=dataset_cls)
                        for f in files]
            success(f'{len(datasets)} sparse datasets loaded.',
                    self.cache.get('verbose'))
        else:
            datasets = self.get_dataset('test', files, dataset_cls=dataset_cls)
        if datasets is None:
            return None
        lens = [len(t) for t in (datasets if isinstance(datasets, list)
                                 else [datasets]) if t]
        if sum(lens) > 0:
            return datasets

    # ---- loaders -------------------------------------------------------
    def get_loader(self, handle_key='', use_padded_sampler=False, **kw):
        args = {**self.cache}
        args.update(self.dataloader_args.get(handle_key, {}))
        args.update(**kw)

        loader_args = {'dataset': None, 'batch_size': 1, 'sampler': None,
                       'shuffle': False, 'batch_sampler': None,
                       'num_workers': 0, 'pin_memory': False,
                       'drop_last': False, 'timeout': 0,
                       'worker_init_fn': _seed_worker if args.get('seed_all') else None}
        for k in loader_args:
            loader_args[k] = args.get(k, loader_args.get(k))

        if loader_args['dataset'] is None:
            return None
        if use_padded_sampler:
            loader_args['drop_last'] = False
            loader_args['shuffle'] = False
            # DOCUMENTED DEVIATION: the requested shuffle flag reaches the
            # sampler here (seeded, epoch-aware). The reference builds its
            # sampler AFTER forcing loader shuffle False (data.py:163-171),
            # so its padded train loader never shuffles despite the
            # sampler's seeded-shuffle support — train batches repeat in
            # file order every epoch.
            loader_args['sampler'] = COINNPaddedDataSampler(
                loader_args['dataset'], loader_args['batch_size'],
                seed=args.get('seed', 0), shuffle=bool(args.get('shuffle')),
                drop_last=bool(args.get('drop_last')),
                total_size=args.get('total_size'))
            # epoch-aware shuffle: the cursor rebuilds this loader every
            # epoch, so the epoch counter must come from outside or the
            # "seeded per-epoch" shuffle degenerates to one fixed order
            # (ADVICE r1). All sites share the counter (lock-step epochs),
            # so sequences stay identical across sites.
            loader_args['sampler'].set_epoch(int(args.get('epoch', 0)))
        return _DataLoader(collate_fn=safe_collate, **loader_args)

    def next_iter(self, handle_key=Mode.TRAIN, shuffle=True):
        """Cursor-stateful train iterator: one batch per call; at epoch end
        flips out['mode'] to VALIDATION_WAITING and resets the cursor."""
        out = {}
        if self.cache.get('cursor', 0) == 0:
            # registry first; fall back to an injected dataset
            # (dataloader_args['train']['dataset']) — the reference KeyErrors
            # here for injected train datasets (data.py:179 reads only the
            # registry that get_dataset fills, data.py:99-103).
            dataset = self.dataset.get(handle_key)
            if dataset is None:
                dataset = self.dataloader_args.get(handle_key, {}).get('dataset')
            loader = self.get_loader(handle_key=handle_key, shuffle=shuffle,
                                     dataset=dataset, use_padded_sampler=True,
                                     total_size=self.cache.get('lockstep_total_size'),
                                     epoch=self.cache.get('train_epoch', 0))
            if loader is None:
                raise RuntimeError(
                    f"'{handle_key}' split is empty on site "
                    f"{self.state.get('clientId', '?')} — with k-fold "
                    "splitting, k=2 leaves NO training files (test and "
                    "validation take both parts); use num_folds >= 3 or a "
                    "split_ratio.")
            self.cache['data_len'] = len(loader) * self.cache['batch_size']
            self.cache['train_loader_iter'] = iter(loader)

        batch = next(self.cache['train_loader_iter'])
        self.cache['cursor'] = self.cache.get('cursor', 0) + self.cache['batch_size']

        if self.cache['cursor'] >= self.cache['data_len']:
            out['mode'] = Mode.VALIDATION_WAITING
            self.cache['cursor'] = 0
            # next rebuild of the train loader gets a fresh shuffle order
            self.cache['train_epoch'] = self.cache.get('train_epoch', 0) + 1
        return batch, out

    # ---- discovery / splits --------------------------------------------
    def prepare_data(self):
        return _kfolds(self.list_files(), self.cache, self.state)

    def list_files(self):
        if self.cache.get('data_dir'):
            return _os.listdir(_os.path.join(self.state['baseDirectory'],
                                             self.cache['data_dir']))
        return []
