"""Tensor utilities: gradient (de)serialization, weight init, safe concat.

Parity: /root/reference/coinstac_dinunet/utils/tensorutils.py:10-55.
MI355X note: `extract_grads`/`save_arrays` survive only for the loopback
(CPU plumbing-test) transport; the GPU path replaces them with the fused
flat-bucket pack/unpack in coinstac_dinunet_amd.ops (device-resident, no
host round trip) feeding one RCCL all-reduce.
"""
import numpy as np
import torch


def safe_concat(large, small):
    """Center-crop `large` to `small`'s trailing spatial dims and concat on C.

    U-Net skip-connection helper for odd spatial sizes (2D or 3D).
    """
    diff = [l - s for l, s in zip(large.shape[2:], small.shape[2:])]
    starts = [d // 2 for d in diff]
    sl = [slice(None), slice(None)] + [
        slice(st, st + s) for st, s in zip(starts, small.shape[2:])
    ]
    return torch.cat([large[tuple(sl)], small], dim=1)


def initialize_weights(module):
    """Seeded Kaiming init for Conv2d/Conv3d/Linear; BN weight=1, bias=0.

    (The reference misses Conv3d — tensorutils.py:28-37; covered here since
    the flagship workload is a 3D CNN.)
    """
    for m in module.modules():
        if isinstance(m, (torch.nn.Conv2d, torch.nn.Conv3d, torch.nn.Linear)):
            # kaiming defaults (fan_in/leaky_relu) — bit-parity with the
            # reference's init (tensorutils.py:32), pinned by
            # tests/test_reference_parity.py::test_initialize_weights_bitwise
            torch.nn.init.kaiming_normal_(m.weight)
            if m.bias is not None:
                torch.nn.init.constant_(m.bias, 0)
        elif isinstance(m, (torch.nn.BatchNorm1d, torch.nn.BatchNorm2d, torch.nn.BatchNorm3d)):
            torch.nn.init.constant_(m.weight, 1)
            torch.nn.init.constant_(m.bias, 0)


def extract_grads(model, dtype='float32'):
    """Per-parameter gradients as CPU numpy arrays (loopback wire format)."""
    return [p.grad.detach().cpu().numpy().astype(dtype)
            for p in model.parameters() if p.grad is not None]


def assign_grads(model, grads, device=None):
    """Inverse of extract_grads: write arrays back into param.grad."""
    params = [p for p in model.parameters()]
    for p, g in zip(params, grads):
        t = torch.as_tensor(np.asarray(g), dtype=p.dtype)
        if device is not None:
            t = t.to(device, non_blocking=True)
        p.grad = t


def _to_obj(a):
    if isinstance(a, (list, tuple)):
        inner = np.empty(len(a), dtype=object)
        for i, x in enumerate(a):
            inner[i] = _to_obj(x)
        return inner
    return np.asarray(a)


def save_arrays(path, arrays):
    """np.save of an object array of (possibly nested) arrays (wire format)."""
    obj = np.empty(len(arrays), dtype=object)
    for i, a in enumerate(arrays):
        obj[i] = _to_obj(a)
    np.save(path, obj)


def load_arrays(path):
    return np.load(str(path) if str(path).endswith('.npy') else str(path) + '.npy',
                   allow_pickle=True)


def flatten_params(model):
    """Total element count and per-param (shape, numel) manifest for bucketing."""
    manifest = [(tuple(p.shape), p.numel()) for p in model.parameters()]
    return sum(n for _, n in manifest), manifest


def caste_ndarray(a, dtype='float32'):
    """Cast helper kept for API parity (reference tensorutils.py:40-41)."""
    return a.astype(dtype)
