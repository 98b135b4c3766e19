"""A/B: fused power-iteration kernel (K10) vs the torch/rocBLAS matvec
chain, at representative rankDAD layer shapes.

Run on a GPU box:  python tools/bench_rankdad.py
Writes gpurun_out/rankdad_ab.json when run under gpurun.
"""
import json
import sys
import os
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from coinstac_dinunet_amd import ops
from coinstac_dinunet_amd.distrib import rankdad as rd

# (n, m, k, rank, iters): n=out-features, m=in-features, k=flat batch
SHAPES = [
    ('mlp_h1', 256, 67, 16, 10, 5),     # FreeSurfer MLP hidden
    ('mlp_h2', 128, 257, 16, 10, 5),
    ('mlp_out', 2, 129, 16, 10, 5),
    ('wide', 512, 513, 64, 10, 5),
    ('recompress', 256, 67, 80, 10, 5),  # 8-site factor concat re-compress
]


def run(fused):
    torch.manual_seed(0)
    res = {}
    for name, n, m, k, rank, iters in SHAPES:
        B = torch.randn(n, k, device='cuda')
        C = torch.randn(m, k, device='cuda')
        gen = torch.Generator().manual_seed(1)

        if fused:
            call = lambda: rd.power_iteration_BC(
                B, C, rank, iters, 1e-3,
                generator=torch.Generator().manual_seed(1))
        else:
            import unittest.mock as mock

            def call():
                with mock.patch.object(ops, 'native_available',
                                       lambda: False):
                    return rd.power_iteration_BC(
                        B, C, rank, iters, 1e-3,
                        generator=torch.Generator().manual_seed(1))

        for _ in range(5):
            call()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        reps = 50
        for _ in range(reps):
            call()
        torch.cuda.synchronize()
        res[name] = (time.perf_counter() - t0) / reps * 1e6  # us per call
    return res


def main():
    assert torch.cuda.is_available()
    fused = run(True)
    chain = run(False)
    out = {'unit': 'us per power_iteration_BC call',
           'shapes': {name: {'fused': round(fused[name], 1),
                             'torch_chain': round(chain[name], 1),
                             'speedup': round(chain[name] / fused[name], 2)}
                      for name, *_ in SHAPES}}
    print(json.dumps(out, indent=2))
    os.makedirs('gpurun_out', exist_ok=True)
    with open('gpurun_out/rankdad_ab.json', 'w') as f:
        json.dump(out, f, indent=2)


if __name__ == '__main__':
    main()
