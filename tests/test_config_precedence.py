"""Args resolution: platform input > nested task/engine/data
blocks > constructor defaults, cached once behind ARGS_CACHED."""
import pytest

from coinstac_dinunet_amd import COINNLocal
from coinstac_dinunet_amd.config.keys import Key, Mode


def _mk(input_dict, **ctor_kw):
    cache = {}
    state = {'clientId': 'local0', 'baseDirectory': '/tmp',
             'transferDirectory': '/tmp', 'outputDirectory': '/tmp'}
    kw = dict(task_id='t', mode=Mode.TRAIN, split_ratio=(0.8, 0.1, 0.1))
    kw.update(ctor_kw)
    COINNLocal(cache=cache, input=input_dict, state=state, **kw)
    return cache


def test_ctor_defaults_fill_missing():
    cache = _mk({})
    assert cache['batch_size'] == 8
    assert cache['agg_engine'] == 'dSGD'
    assert cache['patience'] == cache['epochs']
    assert cache[Key.ARGS_CACHED] is True


def test_input_overrides_ctor():
    cache = _mk({'batch_size': 32, 'learning_rate': 0.5})
    assert cache['batch_size'] == 32
    assert cache['learning_rate'] == 0.5


def test_nested_task_and_engine_blocks():
    cache = _mk({'task_id': 'vbm', 'agg_engine': 'powerSGD',
                 'vbm_args': {'batch_size': 64},
                 'powerSGD_args': {'matrix_approximation_rank': 3},
                 'vbm_data_conf': {'data_dir': 'imgs', 'batch_size': 999}},
                task_id='vbm')
    assert cache['batch_size'] == 64        # task block wins over data_conf
    assert cache['matrix_approximation_rank'] == 3
    assert cache['data_dir'] == 'imgs'      # data_conf fills non-conflicting


def test_args_cached_only_once():
    cache = _mk({'batch_size': 16})
    # a second construction with different input must NOT re-resolve
    state = {'clientId': 'local0', 'baseDirectory': '/tmp',
             'transferDirectory': '/tmp', 'outputDirectory': '/tmp'}
    COINNLocal(cache=cache, input={'batch_size': 999}, state=state,
               task_id='t', mode=Mode.TRAIN, split_ratio=(0.8, 0.1, 0.1))
    assert cache['batch_size'] == 16


def test_mode_validation():
    with pytest.raises(AssertionError):
        _mk({}, mode='bogus')
    with pytest.raises(AssertionError):
        _mk({}, mode=Mode.TRAIN, split_ratio=None, num_folds=None)


def test_explicit_patience_kept():
    cache = _mk({}, patience=5, epochs=50)
    assert cache['patience'] == 5
