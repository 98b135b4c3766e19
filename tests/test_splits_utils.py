import json
import os

import pytest

from coinstac_dinunet_amd.data import datautils
from coinstac_dinunet_amd.utils import FrozenDict, lazy_debug
from coinstac_dinunet_amd.utils.utils import (performance_improved_,
                                              stop_training_)


def test_frozen_dict_blocks_overwrite():
    d = FrozenDict({'a': 1})
    with pytest.raises(ValueError):
        d['a'] = 2
    d['b'] = 3
    assert d['b'] == 3


def test_performance_improved_maximize():
    cache = {'metric_direction': 'maximize', 'best_val_score': 0.5,
             'best_val_epoch': 0}
    assert performance_improved_(3, 0.6, cache)
    assert cache['best_val_epoch'] == 3 and cache['best_val_score'] == 0.6
    # within score_delta => not improved
    assert not performance_improved_(4, 0.60005, cache)


def test_performance_improved_minimize():
    cache = {'metric_direction': 'minimize', 'best_val_score': 1.0,
             'best_val_epoch': 0}
    assert performance_improved_(2, 0.5, cache)
    assert not performance_improved_(3, 0.4999, cache)


def test_stop_training_patience():
    cache = {'best_val_epoch': 2, 'patience': 3}
    assert not stop_training_(5, cache)
    assert stop_training_(6, cache)


def test_kfold_split_structure(tmp_path):
    files = [f'f{i}.npy' for i in range(10)]
    cache = {'num_folds': 5, 'split_dir': str(tmp_path)}
    datautils.create_k_fold_splits(list(files), cache)
    names = sorted(os.listdir(tmp_path))
    assert len(names) == 5
    all_test = []
    for name in names:
        with open(tmp_path / name) as f:
            sp = json.load(f)
        # disjoint, complete
        assert set(sp['train']) | set(sp['validation']) | set(sp['test']) == set(files)
        assert not (set(sp['train']) & set(sp['test']))
        assert not (set(sp['validation']) & set(sp['test']))
        all_test += sp['test']
    # every file is a test file exactly once across folds
    assert sorted(all_test) == sorted(files)


def test_kfold_split_deterministic_by_len(tmp_path):
    files = [f'f{i}.npy' for i in range(12)]
    c1 = {'num_folds': 3, 'split_dir': str(tmp_path / 'a')}
    c2 = {'num_folds': 3, 'split_dir': str(tmp_path / 'b')}
    os.makedirs(c1['split_dir']); os.makedirs(c2['split_dir'])
    datautils.create_k_fold_splits(list(files), c1)
    datautils.create_k_fold_splits(list(files), c2)
    for name in os.listdir(c1['split_dir']):
        with open(os.path.join(c1['split_dir'], name)) as f1, \
                open(os.path.join(c2['split_dir'], name)) as f2:
            assert json.load(f1) == json.load(f2)


def test_ratio_split_boundaries(tmp_path):
    files = [f'f{i}.npy' for i in range(10)]
    cache = {'split_ratio': (0.6, 0.2, 0.2), 'split_dir': str(tmp_path)}
    datautils.create_ratio_split(list(files), cache)
    with open(tmp_path / 'SPLIT.json') as f:
        sp = json.load(f)
    assert len(sp['train']) == 6 and len(sp['validation']) == 2 \
        and len(sp['test']) == 2
    assert set(sp['train']) | set(sp['validation']) | set(sp['test']) == set(files)


def test_init_k_folds_precedence_placeholder(tmp_path):
    base = tmp_path / 'base'
    outd = tmp_path / 'out'
    base.mkdir(); outd.mkdir()
    cache = {'task_id': 't'}
    state = {'baseDirectory': str(base), 'outputDirectory': str(outd)}
    datautils.init_k_folds([], cache, state)
    assert cache['splits'] == {'0': 'empty_split.json'}


def test_lazy_debug_monotone():
    hits = [x for x in range(1, 200) if lazy_debug(x)]
    assert len(hits) > 5
    assert len(hits) < 150


def test_init_k_folds_existing_dir_wins(tmp_path):
    """Pre-existing split dir in baseDirectory overrides everything."""
    import json
    from coinstac_dinunet_amd.data import datautils
    base = tmp_path / 'base'
    out = tmp_path / 'out'
    (base / 'splits').mkdir(parents=True)
    custom = {'train': ['a'], 'validation': ['b'], 'test': ['c']}
    (base / 'splits' / 'SPLIT_custom.json').write_text(json.dumps(custom))
    cache = {'task_id': 'tsk', 'num_folds': 4,          # would lose
             'split_ratio': (0.8, 0.1, 0.1)}            # would lose
    state = {'baseDirectory': str(base), 'outputDirectory': str(out)}
    datautils.init_k_folds(['a', 'b', 'c'], cache, state)
    assert cache['splits'] == {'0': 'SPLIT_custom.json'}
    got = json.load(open(os.path.join(cache['split_dir'],
                                      'SPLIT_custom.json')))
    assert got == custom


def test_init_k_folds_split_files_beat_num_folds(tmp_path):
    import json
    from coinstac_dinunet_amd.data import datautils
    base = tmp_path / 'base'
    out = tmp_path / 'out'
    base.mkdir()
    custom = {'train': ['x'], 'validation': ['y'], 'test': ['z']}
    (base / 'given.json').write_text(json.dumps(custom))
    cache = {'task_id': 'tsk', 'split_files': ['given.json'],
             'num_folds': 3}
    state = {'baseDirectory': str(base), 'outputDirectory': str(out)}
    datautils.init_k_folds(['x', 'y', 'z'], cache, state)
    assert cache['splits'] == {'0': 'given.json'}
