"""Conv3d module on the hand-written implicit-GEMM MFMA kernels (K1).

GPU path (bf16): ops._hip_ops.conv3d_{fwd,dgrad,wgrad}; CPU falls back to
torch.nn.functional.conv3d. Restricted to the VBM workload's conv family:
3x3x3 kernels, padding 1, stride 1 or 2, NCDHW.
"""
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from . import native_available, require_native

# Cin<16 stride-1 forwards route through the CTILE=1 spatial instances
# instead of the igemm fallback. DEFAULT ON since round 2: measured A/B on
# MI355X (profiles/r2_scaffold_ab.md) — 43.79 vs 44.46 ms/step on the VBM
# flagship. COINN_SPATIAL_CI1=0 restores the igemm routing.
_SPATIAL_CI1 = os.environ.get('COINN_SPATIAL_CI1', '1') == '1'
# Double-buffered CTILE=16 stride-1 forward instances — numerically
# validated on hardware but measured FLAT on the flagship (44.59 vs 44.46
# ms/step, profiles/r2_scaffold_ab.md): block-level overlap already hides
# the staging at multi-block occupancy. Kept compiled behind the env flag.
_SPATIAL_DB = os.environ.get('COINN_SPATIAL_DB', '0') == '1'
# Double-buffered stride-1 wgrad: numerically validated, measured WORSE
# (45.89 ms/step — LDS doubling costs co-residency more than intra-block
# overlap buys). Kept compiled behind the env flag.
_WGRAD_DB = os.environ.get('COINN_WGRAD_DB', '0') == '1'


class _Conv3dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, stride):
        C = require_native()
        xb = x.to(torch.bfloat16)
        wb = weight.to(torch.bfloat16)
        # spatial tap-reuse pays when the channel tile fills and chunks
        # are dense; else the igemm kernel wins (profiled in tools/bench_conv)
        ow = (xb.size(4) + 2 - 3) // stride + 1
        oh = (xb.size(3) + 2 - 3) // stride + 1
        min_chunk = 64
        ci_ok = xb.size(1) >= 16 or (_SPATIAL_CI1 and stride == 1)
        if (ow % 8 == 0 and ci_ok and oh * ow >= min_chunk):
            if xb.size(1) < 16:
                ctile_opt = 1
            elif _SPATIAL_DB and stride == 1:
                ctile_opt = 16
            else:
                ctile_opt = 0
            out = C.conv3d_fwd_spatial(xb, wb, stride, ctile_opt)
        else:
            out = C.conv3d_fwd(xb, wb, stride)
        if bias is not None:
            out = out + bias.to(out.dtype).view(1, -1, 1, 1, 1)
        ctx.save_for_backward(xb, wb)
        ctx.stride = stride
        ctx.has_bias = bias is not None
        ctx.x_requires = x.requires_grad
        ctx.in_dtype = x.dtype
        ctx.w_dtype = weight.dtype
        return out

    @staticmethod
    def backward(ctx, grad_out):
        C = require_native()
        xb, wb = ctx.saved_tensors
        go = grad_out.to(torch.bfloat16).contiguous()
        gx = gw = gb = None
        if ctx.needs_input_grad[0]:
            wsub = (xb.size(4) + 1) // 2
            hsub = (xb.size(3) + 1) // 2
            if (ctx.stride == 1 and xb.size(4) % 8 == 0
                    and go.size(1) >= 16
                    and xb.size(3) * xb.size(4) >= 64):
                gx = C.conv3d_dgrad_spatial(go, wb,
                                            list(xb.shape)).to(ctx.in_dtype)
            elif (ctx.stride == 2 and wsub % 8 == 0 and go.size(1) >= 32
                    and hsub * wsub >= 128):
                gx = C.conv3d_dgrad_s2_spatial(
                    go, wb, list(xb.shape)).to(ctx.in_dtype)
            else:
                gx = C.conv3d_dgrad(go, wb, list(xb.shape),
                                    ctx.stride).to(ctx.in_dtype)
        if ctx.needs_input_grad[1]:
            variant = 1 if (_WGRAD_DB and ctx.stride == 1) else 0
            gw = C.conv3d_wgrad(xb, go, ctx.stride,
                                variant).to(ctx.w_dtype)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = C.channel_sum(go)
        return gx, gw, gb, None


class _ConvPw3dFn(torch.autograd.Function):
    """1x1x1 (pointwise) conv — a per-position channel GEMM on the
    bandwidth-shaped pointwise kernels (UNet3D segmentation head)."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        C = require_native()
        xb = x.to(torch.bfloat16)
        wb = weight.to(torch.bfloat16)
        empty = torch.empty(0, device=x.device)
        out = C.conv3d_pw_fwd(xb, wb.view(wb.size(0), wb.size(1)),
                              bias if bias is not None else empty)
        ctx.save_for_backward(xb, wb)
        ctx.has_bias = bias is not None
        ctx.in_dtype = x.dtype
        ctx.w_dtype = weight.dtype
        return out

    @staticmethod
    def backward(ctx, grad_out):
        C = require_native()
        xb, wb = ctx.saved_tensors
        go = grad_out.to(torch.bfloat16).contiguous()
        gx = gw = gb = None
        w2d = wb.view(wb.size(0), wb.size(1))
        if ctx.needs_input_grad[0]:
            gx = C.conv3d_pw_dgrad(go, w2d).to(ctx.in_dtype)
        if ctx.needs_input_grad[1]:
            gw = C.conv3d_pw_wgrad(xb, go).view_as(wb).to(ctx.w_dtype)
        if ctx.has_bias and ctx.needs_input_grad[2]:
            gb = C.channel_sum(go)
        return gx, gw, gb


def conv3d(x, weight, bias=None, stride=1):
    if x.is_cuda and native_available() and weight.shape[2:] == (3, 3, 3):
        return _Conv3dFn.apply(x, weight, bias, int(stride))
    return F.conv3d(x, weight, bias, stride=stride, padding=1)


class OpsConv3d(nn.Conv3d):
    """nn.Conv3d that runs the MFMA implicit-GEMM kernels on GPU.

    Only 3x3x3/pad-1/stride-{1,2} routes to the HIP path; anything else
    falls through to the library conv.
    """

    def forward(self, x):
        if (x.is_cuda and native_available()
                and self.kernel_size == (3, 3, 3)
                and self.padding == (1, 1, 1)
                and self.stride[0] in (1, 2)
                and self.stride[0] == self.stride[1] == self.stride[2]
                and self.dilation == (1, 1, 1) and self.groups == 1):
            return _Conv3dFn.apply(x, self.weight, self.bias,
                                   int(self.stride[0]))
        if (x.is_cuda and native_available()
                and self.kernel_size == (1, 1, 1)
                and self.padding == (0, 0, 0)
                and self.stride == (1, 1, 1)
                and self.dilation == (1, 1, 1) and self.groups == 1):
            return _ConvPw3dFn.apply(x, self.weight, self.bias)
        # library fallback must still accept the bf16 activations the fused
        # kernels upstream emit (bf16 act into fp32 head crashed)
        if x.is_cuda and x.dtype != self.weight.dtype:
            x = x.to(self.weight.dtype)
        return super().forward(x)
