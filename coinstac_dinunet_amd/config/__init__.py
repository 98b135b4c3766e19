"""Framework-wide constants.

Functional parity with the reference's config surface
(/root/reference/coinstac_dinunet/config/__init__.py:1-30): wire-file names,
metric precision/eps, score delta, GPU probing, seed — re-designed for a
persistent one-process-per-GPU MI355X runtime (the file names survive as the
loopback-transport artifact names and checkpoint names).
"""
import random as _random
import sys as _sys

import torch as _torch

# Artifact names (loopback transport + checkpoints).
grad_file_ext = '.npy'
grads_file = f'grads{grad_file_ext}'
avg_grads_file = f'avg_grads{grad_file_ext}'
weights_file = 'weights.tar'

# Metrics formatting.
metrics_eps = 1e-5
metrics_num_precision = 5

# Minimum improvement for "performance improved" checkpoint selection.
score_delta = 1e-4
score_high = 1.0
score_low = 0.0

# Unbounded sentinel (load_limit default, worst-possible minimize score).
max_size = _sys.maxsize

# Probed once at import (False/0 in the CPU container; True/N on an MI355X box).
CUDA_AVAILABLE = _torch.cuda.is_available()
NUM_GPUS = _torch.cuda.device_count() if CUDA_AVAILABLE else 0

# Process-wide default seed when the user supplies none.
current_seed = _random.randint(0, 2 ** 24)

# gfx950 / MI355X layout constants used by the ops layer.
GFX_ARCH = 'gfx950'
WAVE_SIZE = 64
NUM_CUS = 256
NUM_XCDS = 8


def boolean_string(s):
    if isinstance(s, bool):
        return s
    if str(s).lower() not in ('true', 'false'):
        raise ValueError(f'Not a valid boolean string: {s}')
    return str(s).lower() == 'true'
