"""Minimal gated console logger (parity: reference utils/logger.py:4-24)."""
import sys as _sys
import time as _time


def _emit(tag, msg, verbose):
    if verbose:
        print(f'[{tag}] {msg}', file=_sys.stderr if tag == 'ERROR' else _sys.stdout, flush=True)


def error(msg, verbose=True):
    _emit('ERROR', msg, verbose)


def warn(msg, verbose=True):
    _emit('WARN', msg, verbose)


def info(msg, verbose=True):
    _emit('INFO', msg, verbose)


def success(msg, verbose=True):
    _emit('OK', msg, verbose)


class duration:
    """Context manager appending wall-clock seconds into cache[key] (profiling aid)."""

    def __init__(self, cache, key):
        self.cache, self.key = cache, key

    def __enter__(self):
        self.t0 = _time.time()
        return self

    def __exit__(self, *exc):
        self.cache.setdefault(self.key, []).append(_time.time() - self.t0)
        return False


def lazy_debug(x, add=1):
    """Log-frequency thinner (also exported from utils; reference
    logger.py:23-24)."""
    from . import lazy_debug as _ld
    return _ld(x, add)
