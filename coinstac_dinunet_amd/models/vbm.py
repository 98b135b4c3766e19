"""VBM 3D-CNN classifier (BASELINE.json configs 3/4 — the flagship bench).

The reference's GPU example workload is a 3D CNN over ~64^3 (up to
121x145x121) brain volumes with small channel counts.
This is our own architecture sized for that shape: conv3x3x3 stacks with
BN/ReLU and stride-2 downsampling, global average pool, linear head.

MI355X notes: NCDHW layout (w-contiguous rows feed the spatial-slab conv
kernels' 16-byte staging loads); conv runs on OpsConv3d (implicit-GEMM
MFMA) and BN+ReLU is one fused pass (OpsBatchNorm3d).
"""
import os

import torch.nn as nn

from ..ops.bnorm import OpsBatchNorm3d
from ..ops.conv import OpsConv3d, can_fuse_bn_conv, conv_bn3d

# Cross-block BN fusion (normalize-on-load): block i's BN+ReLU is applied
# inside block i+1's conv kernels during slab staging, so the normalized
# activation never round-trips HBM. COINN_FUSE_BN=0 restores the separate
# fused-BN-pass path.
_FUSE_BN = os.environ.get('COINN_FUSE_BN', '1') == '1'


class _ConvBlock(nn.Module):
    def __init__(self, cin, cout, stride=1):
        super().__init__()
        self.conv = OpsConv3d(cin, cout, 3, stride=stride, padding=1, bias=False)
        self.bn = OpsBatchNorm3d(cout, relu=True)  # ReLU fused into BN pass

    def forward(self, x):
        return self.bn(self.conv(x))


class VBMNet(nn.Module):
    """~9-layer 3D CNN: 1 -> widths[0] -> ... with stride-2 stages."""

    def __init__(self, in_channels=1, num_class=2, widths=(32, 64, 128, 256)):
        super().__init__()
        stages = []
        cin = in_channels
        for i, w in enumerate(widths):
            stages.append(_ConvBlock(cin, w, stride=1 if i == 0 else 2))
            stages.append(_ConvBlock(w, w))
            cin = w
        self.features = nn.Sequential(*stages)
        self.pool = nn.AdaptiveAvgPool3d(1)
        self.head = nn.Linear(cin, num_class)

    def forward(self, x):
        blocks = list(self.features)
        if (_FUSE_BN and len(blocks) >= 2
                and all(can_fuse_bn_conv(blocks[i].bn, blocks[i + 1].conv, x)
                        for i in range(len(blocks) - 1))):
            # fused chain: conv_0 -> [bn_i folded into conv_{i+1}'s load]
            # -> final bn materialized once before the pool
            x = blocks[0].conv(x)
            for prev, blk in zip(blocks[:-1], blocks[1:]):
                x = conv_bn3d(x, prev.bn, blk.conv)
            x = blocks[-1].bn(x)
        else:
            x = self.features(x)
        x = self.pool(x).flatten(1)
        return self.head(x)
