"""Per-op conv3d micro-bench: our HIP kernels vs library conv per layer."""
import os
import sys
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))
import time

import torch

from coinstac_dinunet_amd import ops

C = ops.require_native()

LAYERS = [  # (Cin, Cout, D, stride) at batch B, cube side D
    (1, 32, 64, 1), (32, 32, 64, 1),
    (32, 64, 64, 2), (64, 64, 32, 1),
    (64, 128, 32, 2), (128, 128, 16, 1),
    (128, 256, 16, 2), (256, 256, 8, 1),
]
B = 64


def t(fn, iters=5):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1000


def main():
    dev = torch.device('cuda:0')
    print(f"{'layer':28s} {'fwd':>8s} {'dgrad':>8s} {'wgrad':>8s} "
          f"{'lib_fwd':>8s} {'lib_bwd':>8s} {'TF_fwd':>7s}")
    tot = [0.0] * 5
    for Cin, Cout, D, s in LAYERS:
        x = torch.randn(B, Cin, D, D, D, device=dev, dtype=torch.bfloat16)
        w = torch.randn(Cout, Cin, 3, 3, 3, device=dev,
                        dtype=torch.bfloat16) * 0.1
        OD = (D + 2 - 3) // s + 1
        go = torch.randn(B, Cout, OD, OD, OD, device=dev,
                         dtype=torch.bfloat16)
        ow = (D + 2 - 3) // s + 1
        if ow % 8 == 0 and Cin >= 16 and ow * ow >= 64:
            ms_f = t(lambda: C.conv3d_fwd_spatial(x, w, s))
        else:
            ms_f = t(lambda: C.conv3d_fwd(x, w, s))
        wsub = (D + 1) // 2
        if s == 1 and D % 8 == 0 and Cout >= 16 and D * D >= 64:
            ms_d = t(lambda: C.conv3d_dgrad_spatial(go, w, list(x.shape)))
        elif s == 2 and wsub % 8 == 0 and Cout >= 32 and wsub * wsub >= 128:
            ms_d = t(lambda: C.conv3d_dgrad_s2_spatial(go, w, list(x.shape)))
        else:
            ms_d = t(lambda: C.conv3d_dgrad(go, w, list(x.shape), s))
        ms_w = t(lambda: C.conv3d_wgrad(x, go, s))
        xr = x.clone().requires_grad_(True)
        ms_lf = t(lambda: torch.nn.functional.conv3d(x, w, stride=s, padding=1))
        def lib_bwd():
            out = torch.nn.functional.conv3d(xr, w, stride=s, padding=1)
            out.backward(go)
            xr.grad = None
        ms_lb = t(lib_bwd, iters=3)
        flop = 2 * B * OD ** 3 * Cout * Cin * 27
        tf = flop / (ms_f / 1000) / 1e12
        print(f"ci{Cin:3d} co{Cout:3d} d{D:2d} s{s} "
              f"{ms_f:8.2f} {ms_d:8.2f} {ms_w:8.2f} "
              f"{ms_lf:8.2f} {ms_lb:8.2f} {tf:7.1f}")
        for i, v in enumerate([ms_f, ms_d, ms_w, ms_lf, ms_lb]):
            tot[i] += v
    print(f"{'TOTAL':28s} {tot[0]:8.2f} {tot[1]:8.2f} {tot[2]:8.2f} "
          f"{tot[3]:8.2f} {tot[4]:8.2f}")


if __name__ == '__main__':
    main()
