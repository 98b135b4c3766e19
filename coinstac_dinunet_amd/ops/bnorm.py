"""BatchNorm3d(+fused ReLU) on the HIP streaming kernels.

Training fwd: one reduce + one normalize pass (HBM-roofline); the ReLU
folds into the normalize so conv->bn->relu is 2 passes instead of
MIOpen's BN + a separate clamp kernel. Running-stat semantics match
nn.BatchNorm3d (biased var normalizes, unbiased updates running_var).
"""
import torch
import torch.nn as nn
import torch.nn.functional as F

from . import native_available, require_native


class _BN3dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, momentum,
                eps, relu):
        C = require_native()
        xb = x.to(torch.bfloat16)
        y, mean, var, mean_rstd = C.bn3d_fwd(xb, gamma, beta, eps, relu)
        if running_mean is not None:
            with torch.no_grad():
                n = xb.numel() // xb.size(1)
                unbiased = var * (n / max(n - 1, 1))
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
        ctx.save_for_backward(xb, mean_rstd, gamma, beta)
        ctx.relu = relu
        ctx.in_dtype = x.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        C = require_native()
        xb, mean_rstd, gamma, beta = ctx.saved_tensors
        dx, dgamma, dbeta = C.bn3d_bwd(dy.to(torch.bfloat16), xb, mean_rstd,
                                       gamma, beta, ctx.relu)
        return (dx.to(ctx.in_dtype), dgamma.to(gamma.dtype),
                dbeta.to(beta.dtype), None, None, None, None, None)


class _OpsBNMixin:
    """Shared HIP-kernel forward for the 2D/3D BatchNorm modules — the
    streaming kernels only see [N, C, spatial] and are rank-agnostic."""

    def _ops_forward(self, x, super_forward):
        if not (x.is_cuda and native_available()):
            y = super_forward(x)
            return F.relu(y, inplace=True) if self.relu else y
        if self.training:
            if self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
            if getattr(x, '_coinn_bn_stats', None) is not None:
                # stats came free from the producing conv's epilogue
                from .conv import bn_stats_of
                # pass x itself: .to() would drop the attached stats
                mean, var, mean_rstd = bn_stats_of(x, self.eps)
                if self.running_mean is not None:
                    with torch.no_grad():
                        n = x.numel() // x.size(1)
                        unbiased = var * (n / max(n - 1, 1))
                        self.running_mean.mul_(1 - self.momentum).add_(
                            mean, alpha=self.momentum)
                        self.running_var.mul_(1 - self.momentum).add_(
                            unbiased, alpha=self.momentum)
                return _BN3dPreFn.apply(x, self.weight, self.bias,
                                        mean_rstd, self.relu)
            return _BN3dFn.apply(x, self.weight, self.bias,
                                 self.running_mean, self.running_var,
                                 self.momentum, self.eps, self.relu)
        C = require_native()
        return C.bn3d_infer(x.to(torch.bfloat16), self.weight, self.bias,
                            self.running_mean, self.running_var, self.eps,
                            self.relu)


class _BN3dPreFn(torch.autograd.Function):
    """BN normalize with externally computed batch statistics (the
    producing conv's epilogue-stats path): one normalize pass instead of
    reduce + normalize. Backward is the standard bn3d_bwd."""

    @staticmethod
    def forward(ctx, x, gamma, beta, mean_rstd, relu):
        C = require_native()
        xb = x.to(torch.bfloat16)
        y = C.bn3d_normalize(xb, mean_rstd, gamma, beta, relu)
        ctx.save_for_backward(xb, mean_rstd, gamma, beta)
        ctx.relu = relu
        ctx.in_dtype = x.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        C = require_native()
        xb, mean_rstd, gamma, beta = ctx.saved_tensors
        dx, dgamma, dbeta = C.bn3d_bwd(dy.to(torch.bfloat16), xb, mean_rstd,
                                       gamma, beta, ctx.relu)
        return (dx.to(ctx.in_dtype), dgamma.to(gamma.dtype),
                dbeta.to(beta.dtype), None, None)


class _BNAddReluFn(torch.autograd.Function):
    """BN(x) + residual -> ReLU as ONE pass each way (ResNet BasicBlock
    tail: bn2 -> +identity -> ReLU was three elementwise round-trips).
    Backward recomputes the mask from (affine(x) + res) and emits the
    residual-branch gradient (masked dy) from the same kernel."""

    @staticmethod
    def forward(ctx, x, res, gamma, beta, running_mean, running_var,
                momentum, eps):
        C = require_native()
        xb = x.to(torch.bfloat16)
        rb = res.to(torch.bfloat16)
        y, mean, var, mean_rstd = C.bn3d_fwd_res(xb, rb, gamma, beta, eps)
        if running_mean is not None:
            with torch.no_grad():
                n = xb.numel() // xb.size(1)
                unbiased = var * (n / max(n - 1, 1))
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)
        ctx.save_for_backward(xb, rb, mean_rstd, gamma, beta)
        ctx.in_dtype = x.dtype
        return y

    @staticmethod
    def backward(ctx, dy):
        C = require_native()
        xb, rb, mean_rstd, gamma, beta = ctx.saved_tensors
        dx, dgamma, dbeta, dres = C.bn3d_bwd_res(
            dy.to(torch.bfloat16), xb, rb, mean_rstd, gamma, beta)
        return (dx.to(ctx.in_dtype), dres.to(ctx.in_dtype),
                dgamma.to(gamma.dtype), dbeta.to(beta.dtype),
                None, None, None, None)


def bn_add_relu(x, residual, bn):
    """relu(bn(x) + residual) fused on the HIP path (train mode);
    falls back to the composed ops elsewhere."""
    if x.is_cuda and native_available() and bn.training:
        if bn.num_batches_tracked is not None:
            bn.num_batches_tracked.add_(1)
        if residual.dtype != torch.bfloat16:
            residual = residual.to(torch.bfloat16)
        return _BNAddReluFn.apply(x, residual, bn.weight, bn.bias,
                                  bn.running_mean, bn.running_var,
                                  bn.momentum, bn.eps)
    y = bn(x)
    if y.dtype != residual.dtype:
        residual = residual.to(y.dtype)
    return F.relu(y + residual)


class OpsBatchNorm3d(_OpsBNMixin, nn.BatchNorm3d):
    """BatchNorm3d with optional fused ReLU; HIP kernels on GPU."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False,
                 **kw):
        super().__init__(num_features, eps=eps, momentum=momentum, **kw)
        self.relu = relu

    def forward(self, x):
        return self._ops_forward(x, super().forward)


class OpsBatchNorm2d(_OpsBNMixin, nn.BatchNorm2d):
    """BatchNorm2d with optional fused ReLU on the same streaming kernels
    (ResNet-18 config)."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, relu=False,
                 **kw):
        super().__init__(num_features, eps=eps, momentum=momentum, **kw)
        self.relu = relu

    def forward(self, x):
        return self._ops_forward(x, super().forward)
