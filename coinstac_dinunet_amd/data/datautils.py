"""Split generation: ratio and k-fold, bit-for-bit reference-matching.

Parity: /root/reference/coinstac_dinunet/data/datautils.py:11-98. The
observable contract preserved exactly:
  - shuffle seeded by len(files) (so equal-sized sites derive the same
    permutation structure);
  - k-fold i: test = part i of np.array_split, val = part (i+1)%k,
    train = the rest in original (shuffled) order;
  - ratio split boundaries computed as int(cumsum(reversed ratio) * n)
    from the tail (the reference's reversed-split trick) — train absorbs
    the rounding remainder;
  - files land in outputDirectory/<task_id>/splits as SPLIT_<i>.json,
    indexed into cache['splits'].
"""
import json as _json
import os as _os
import random as _rd
import shutil as _shutil

import numpy as _np

_sep = _os.sep


def create_ratio_split(files, cache, shuffle_files=True, name='SPLIT'):
    save_to_dir = cache['split_dir']
    ratio = cache.get('split_ratio', (0.6, 0.2, 0.2))
    first_key = cache.get('first_key', 'train')

    files = list(files)
    if shuffle_files:
        _rd.seed(len(files))
        _rd.shuffle(files)

    keys = [first_key]
    if len(ratio) == 2:
        keys.append('test')
    elif len(ratio) == 3:
        keys.append('validation')
        keys.append('test')

    n = len(files)
    # boundaries measured from the tail: int(cumsum(reversed ratio) * n)
    rev = list(ratio[::-1])
    locs = [int(sum(rev[:i + 1]) * n) for i in range(len(ratio) - 1)]
    # splits from the front: train = files[:n-locs[-1]], ..., last = files[n-locs[0]:]
    bounds = [n - l for l in locs[::-1]] + [n]
    splits = {}
    start = 0
    for key, end in zip(keys, bounds):
        splits[key] = files[start:end]
        start = end

    if save_to_dir:
        with open(save_to_dir + _sep + f'{name}.json', 'w') as f:
            f.write(_json.dumps(splits))
    else:
        return splits


def create_k_fold_splits(files, cache, shuffle_files=True, name='SPLIT'):
    k = cache['num_folds']
    save_to_dir = cache['split_dir']
    files = list(files)
    if shuffle_files:
        _rd.seed(len(files))
        _rd.shuffle(files)

    file_ix = _np.arange(len(files))
    ix_splits = _np.array_split(file_ix, k)
    for i in range(len(ix_splits)):
        test_ix = ix_splits[i].tolist()
        val_ix = ix_splits[(i + 1) % len(ix_splits)].tolist()
        train_ix = _np.delete(file_ix.copy(), _np.array(test_ix + val_ix))

        splits = {'train': [files[ix] for ix in train_ix],
                  'validation': [files[ix] for ix in val_ix],
                  'test': [files[ix] for ix in test_ix]}

        if save_to_dir:
            with open(save_to_dir + _sep + f'{name}_{i}.json', 'w') as f:
                f.write(_json.dumps(splits))
        else:
            return splits


def split_place_holder(files, cache):
    save_to_dir = cache['split_dir']
    splits = {'train': [], 'validation': [], 'test': []}
    with open(save_to_dir + _sep + 'empty_split.json', 'w') as f:
        f.write(_json.dumps(splits))


def init_k_folds(files, cache, state):
    """Resolve splits with precedence: existing split dir in baseDirectory >
    split_files > num_folds > split_ratio > placeholder."""
    out = {}
    _dir = state['baseDirectory'] + _sep + cache.get('split_dir', 'splits')

    cache['split_dir'] = state['outputDirectory'] + _sep + cache['task_id'] + _sep + 'splits'
    _os.makedirs(cache['split_dir'], exist_ok=True)

    if _os.path.exists(_dir) and len(_os.listdir(_dir)) > 0:
        for f in _os.listdir(_dir):
            _shutil.copy(_dir + _sep + f, cache['split_dir'] + _sep + f)
    elif cache.get('split_files'):
        for f in cache['split_files']:
            _shutil.copy(state['baseDirectory'] + _sep + f,
                         cache['split_dir'] + _sep + _os.path.basename(f))
    elif cache.get('num_folds'):
        create_k_fold_splits(files, cache)
    elif cache.get('split_ratio'):
        create_ratio_split(files, cache)
    else:
        split_place_holder(None, cache)

    splits = sorted(_os.listdir(cache['split_dir']))
    cache['splits'] = dict(zip([str(i) for i in range(len(splits))], splits))
    return out
