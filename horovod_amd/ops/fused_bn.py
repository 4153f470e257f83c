"""Fused BatchNorm(+Add)+ReLU modules backed by the CDNA4 kernels in
csrc/bn_kernels.hip.

Training-mode NHWC path: one HBM pass for normalize+affine+residual+relu
(vs MIOpen BN + separate add + relu kernels) and one fused reduction pass in
backward.  Eval mode, CPU tensors or non-channels-last inputs fall back to
the stock PyTorch ops, so the modules are drop-in (state_dict-compatible
with nn.BatchNorm2d).
"""
import torch
import torch.nn.functional as F

from horovod_amd import _core


def _fusable(x):
    return (x.is_cuda and x.dim() == 4 and
            x.is_contiguous(memory_format=torch.channels_last) and
            x.size(1) % 8 == 0 and x.size(1) <= 4096 and
            x.dtype in (torch.float32, torch.float16, torch.bfloat16))


class _FusedBNReLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, running_mean, running_var,
                momentum, eps):
        y, mean, invstd = _core.fused_bn_relu_forward(
            x, residual, weight, bias, running_mean, running_var, momentum,
            eps)
        ctx.save_for_backward(x, y, mean, invstd, weight)
        ctx.has_residual = residual is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y, mean, invstd, weight = ctx.saved_tensors
        outs = _core.fused_bn_relu_backward(x, y, dy, mean, invstd, weight,
                                            ctx.has_residual)
        dx, dgamma, dbeta = outs[0], outs[1], outs[2]
        dres = outs[3] if ctx.has_residual else None
        return dx, dres, dgamma, dbeta, None, None, None, None


class FusedBNReLU(torch.nn.BatchNorm2d):
    """BatchNorm2d + ReLU in one kernel pass (training, NHWC)."""

    def forward(self, input):
        if self.training and _fusable(input):
            return _FusedBNReLUFn.apply(
                input.contiguous(memory_format=torch.channels_last), None,
                self.weight, self.bias, self.running_mean, self.running_var,
                self.momentum if self.momentum is not None else 0.1, self.eps)
        return F.relu(super().forward(input))


class FusedBNAddReLU(torch.nn.BatchNorm2d):
    """relu(bn(x) + residual) in one kernel pass (the ResNet block tail)."""

    def forward(self, input, residual):
        if self.training and _fusable(input) and _fusable(residual) and \
                residual.shape == input.shape and residual.dtype == input.dtype:
            return _FusedBNReLUFn.apply(
                input.contiguous(memory_format=torch.channels_last),
                residual.contiguous(memory_format=torch.channels_last),
                self.weight, self.bias, self.running_mean, self.running_var,
                self.momentum if self.momentum is not None else 0.1, self.eps)
        return F.relu(super().forward(input) + residual)
