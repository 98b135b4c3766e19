"""Early-stop / improvement predicates.

Parity: /root/reference/coinstac_dinunet/utils/utils.py:7-31
(performance_improved_, stop_training_, duration). Exact comparison
semantics preserved (BASELINE parity anchor): improvement requires
beating cache['best_val_score'] by more than score_delta in
cache['metric_direction']; early stop when epoch - best_val_epoch
exceeds patience.
"""
import time as _time

from .. import config as _conf


def performance_improved_(epoch, score, cache):
    delta = cache.get('score_delta', _conf.score_delta)
    improved = False
    if cache['metric_direction'] == 'maximize':
        improved = score > cache['best_val_score'] + delta
    elif cache['metric_direction'] == 'minimize':
        improved = score < cache['best_val_score'] - delta
    if improved:
        cache['best_val_epoch'] = epoch
        cache['best_val_score'] = score
    return bool(improved)


def stop_training_(epoch, cache):
    return epoch - cache['best_val_epoch'] > cache.get('patience', cache.get('epochs', 1))


def duration(cache, begin, key):
    """Append wall-clock seconds since `begin` into cache[key]."""
    seconds = _time.time() - begin
    cache.setdefault(key, []).append(seconds)
    return seconds
