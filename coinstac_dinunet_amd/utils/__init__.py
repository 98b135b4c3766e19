"""Foundation utilities: write-once dict, cache/score persistence, log thinning.

Parity: /root/reference/coinstac_dinunet/utils/__init__.py:8-80 (FrozenDict,
save_scores, save_cache, jsonable, lazy_debug).
"""
import json as _json
import math as _math
import os as _os


class FrozenDict(dict):
    """Dict that refuses to overwrite an existing key (args/input/state guard)."""

    def __init__(self, *args, **kwargs):
        super().__init__(*args, **kwargs)

    def prompt(self, key):
        raise ValueError(f"*** '{key}' key is frozen once set. ***")

    def __setitem__(self, key, value):
        if key in self:
            self.prompt(key)
        super().__setitem__(key, value)

    def update(self, *args, **kwargs):
        for d in args:
            for k, v in dict(d).items():
                self[k] = v
        for k, v in kwargs.items():
            self[k] = v


def save_scores(cache, log_dir=None, file_keys=None):
    """Write cache[key] rows (lists) as CSV files with the cache's log_header."""
    log_dir = log_dir if log_dir else cache.get('log_dir', '.')
    _os.makedirs(log_dir, exist_ok=True)
    for fk in (file_keys or []):
        rows = cache.get(fk, [])
        path = _os.path.join(log_dir, f'{fk}.csv')
        with open(path, 'w') as f:
            header = cache.get('log_header', '')
            if header:
                f.write(header + '\n')
            for row in rows:
                if isinstance(row, (list, tuple)):
                    f.write(','.join(str(r) for r in row) + '\n')
                else:
                    f.write(str(row) + '\n')


def jsonable(obj):
    """Return a JSON-serializable rendition of obj (stringify what isn't)."""
    try:
        _json.dumps(obj)
        return obj
    except (TypeError, ValueError):
        if isinstance(obj, dict):
            return {str(k): jsonable(v) for k, v in obj.items()}
        if isinstance(obj, (list, tuple, set)):
            return [jsonable(v) for v in obj]
        return str(obj)


def clean_recursive(obj):
    """In-place best-effort conversion of a nested structure to JSON-able types."""
    if isinstance(obj, dict):
        for k in list(obj.keys()):
            obj[k] = jsonable(obj[k])
    return obj


def save_cache(cache, log_dir=None, name='logs'):
    log_dir = log_dir if log_dir else cache.get('log_dir', '.')
    _os.makedirs(log_dir, exist_ok=True)
    with open(_os.path.join(log_dir, f'{name}.json'), 'w') as f:
        _json.dump(jsonable(dict(cache)), f, indent=2)


def lazy_debug(x, add=1):
    """Log-frequency thinner: true ~logarithmically often in x."""
    return x % int(_math.log(x + 1) + add) == 0

# re-exports for parity with the reference utils namespace
from .utils import duration, performance_improved_, stop_training_  # noqa: E402,F401
