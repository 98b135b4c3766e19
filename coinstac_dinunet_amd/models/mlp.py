"""FreeSurfer-style tabular MLP classifier (BASELINE.json config 1/2).

The reference ships no models in-library (they live in the external
dinunet_implementations repos, README.md:30-33); the FreeSurfer workload is
[B, ~66 features] -> hidden -> 2 classes. On GPU the
linear layers route through the MFMA HIP GEMM (ops.linear) when the
extension is loaded.
"""
import torch.nn as nn


class _OpsLinear(nn.Linear):
    """nn.Linear that runs the hand-written MFMA GEMM on GPU."""

    def forward(self, x):
        from .. import ops
        if x.is_cuda and ops.native_available():
            return ops.linear(x, self.weight, self.bias)
        return super().forward(x)


class FreeSurferMLP(nn.Module):
    def __init__(self, in_features=66, hidden_sizes=(256, 128, 64),
                 num_class=2, dropout=0.3):
        super().__init__()
        layers = []
        prev = in_features
        for h in hidden_sizes:
            layers += [_OpsLinear(prev, h), nn.BatchNorm1d(h), nn.ReLU(inplace=True),
                       nn.Dropout(dropout)]
            prev = h
        layers.append(_OpsLinear(prev, num_class))
        self.net = nn.Sequential(*layers)

    def forward(self, x):
        return self.net(x)
