"""Inference-time conv+BN folding.

At eval the BatchNorm is a fixed per-channel affine
(y = (x - mean)/sqrt(var+eps) * gamma + beta), so it folds into the
preceding convolution's weights host-side:

    w' = w * gamma / sqrt(var + eps)        (per output channel)
    b' = beta + (b - mean) * gamma / sqrt(var + eps)

`fuse_conv_bn_eval(model)` returns a COPY of the model where every
(ConvNd -> BatchNormNd) pair inside an nn.Sequential, and the conv/bn
attribute pairs of this package's block modules (models/vbm._ConvBlock,
models/unet._Block), are folded; a fused-ReLU BN (OpsBatchNorm3d
relu=True) leaves a plain ReLU behind. Validation/test forwards then
skip every BN kernel — each of which costs a full activation-tensor
read+write at HBM speed on MI355X.
"""
import copy

import torch
import torch.nn as nn


def fold_bn(conv, bn):
    """Mutate `conv` in place so conv(x) == bn(conv(x)) at eval."""
    scale = bn.weight.detach() / torch.sqrt(bn.running_var + bn.eps)
    conv.weight.data.mul_(scale.reshape(-1, *([1] * (conv.weight.dim() - 1))))
    shift = bn.bias.detach() - bn.running_mean * scale
    if conv.bias is None:
        conv.bias = nn.Parameter(shift.clone())
    else:
        conv.bias.data.mul_(scale).add_(shift)
    return conv


def _bn_tail(bn):
    """What remains where the BN stood."""
    return nn.ReLU(inplace=True) if getattr(bn, 'relu', False) \
        else nn.Identity()


def _is_pair(a, b):
    return isinstance(a, (nn.Conv1d, nn.Conv2d, nn.Conv3d)) and \
        isinstance(b, (nn.BatchNorm1d, nn.BatchNorm2d, nn.BatchNorm3d)) and \
        b.track_running_stats and b.running_mean is not None


def _fuse_module(mod):
    # this package's conv blocks: explicit (conv, bn) attribute pairs
    for conv_name, bn_name in (('conv', 'bn'), ('c1', 'b1'), ('c2', 'b2')):
        conv = getattr(mod, conv_name, None)
        bn = getattr(mod, bn_name, None)
        if conv is not None and bn is not None and _is_pair(conv, bn):
            fold_bn(conv, bn)
            setattr(mod, bn_name, _bn_tail(bn))
    # generic sequential scan
    if isinstance(mod, nn.Sequential):
        for i in range(len(mod) - 1):
            if _is_pair(mod[i], mod[i + 1]):
                fold_bn(mod[i], mod[i + 1])
                mod[i + 1] = _bn_tail(mod[i + 1])
    for child in mod.children():
        _fuse_module(child)


def fuse_conv_bn_eval(model):
    """Return a fused deep copy in eval mode (the original is untouched)."""
    fused = copy.deepcopy(model).eval()
    _fuse_module(fused)
    return fused
