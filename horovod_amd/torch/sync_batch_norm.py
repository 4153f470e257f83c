"""Synchronized batch normalization across ranks.

Reference: horovod/torch/sync_batch_norm.py:40-218 (allgather of per-rank
count/mean/var + custom backward).  Statistics here travel through a single
fused allreduce of [sum, sqsum, count] per forward — one collective instead
of the reference's three allgathers — with the same mathematics, including
uneven per-rank batch sizes.
"""
import itertools

import torch
from torch.nn.modules.batchnorm import _BatchNorm

from horovod_amd.torch.mpi_ops import allreduce, size
from horovod_amd.common.process_sets import global_process_set


class SyncBatchNorm(_BatchNorm):
    """Applies synchronized BatchNorm: statistics computed over all ranks of
    the process set (drop-in for torch.nn.BatchNorm*d)."""

    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True, process_set=global_process_set):
        super().__init__(num_features, eps, momentum, affine,
                         track_running_stats)
        self.process_set = process_set

    def _check_input_dim(self, input):
        if input.dim() < 2:
            raise ValueError(
                f"expected at least 2D input (got {input.dim()}D input)")

    def forward(self, input):
        if not (self.training and
                (self.process_set.process_set_id == 0 and size() > 1 or
                 (self.process_set.process_set_id != 0 and
                  len(self.process_set.ranks) > 1))):
            return super().forward(input)
        self._check_input_dim(input)
        if self.momentum is None:
            momentum = 0.0
        else:
            momentum = self.momentum
        if self.training and self.track_running_stats:
            if self.num_batches_tracked is not None:
                self.num_batches_tracked.add_(1)
                if self.momentum is None:
                    momentum = 1.0 / float(self.num_batches_tracked)
        return _SyncBatchNormFn.apply(input, self.weight, self.bias,
                                      self.running_mean, self.running_var,
                                      self.eps, momentum, self.process_set)


# names must be identical across ranks: use a call counter (SPMD order), not
# object ids (rank-divergent).
_sbn_counter = itertools.count()


class _SyncBatchNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, bias, running_mean, running_var, eps,
                momentum, process_set):
        input = input.contiguous()
        reduce_dims = [0] + list(range(2, input.dim()))
        count = input.numel() // input.size(1)

        local_sum = input.sum(dim=reduce_dims)
        local_sqsum = (input * input).sum(dim=reduce_dims)
        stats = torch.cat([local_sum, local_sqsum,
                           local_sum.new_tensor([float(count)])])
        stats = allreduce(stats, average=False, process_set=process_set,
                          name=f"sync_batch_norm.fw.{next(_sbn_counter)}")
        c = input.size(1)
        total_count = stats[-1]
        mean = stats[:c] / total_count
        var = stats[c:2 * c] / total_count - mean * mean

        if running_mean is not None:
            with torch.no_grad():
                running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
                unbiased = var * (total_count / (total_count - 1).clamp(min=1))
                running_var.mul_(1 - momentum).add_(unbiased, alpha=momentum)

        invstd = torch.rsqrt(var + eps)
        shape = [1, c] + [1] * (input.dim() - 2)
        xhat = (input - mean.reshape(shape)) * invstd.reshape(shape)
        out = xhat
        if weight is not None:
            out = out * weight.reshape(shape)
        if bias is not None:
            out = out + bias.reshape(shape)
        ctx.save_for_backward(input, weight, mean, invstd)
        ctx.process_set = process_set
        ctx.total_count = total_count
        return out

    @staticmethod
    def backward(ctx, grad_output):
        input, weight, mean, invstd = ctx.saved_tensors
        process_set = ctx.process_set
        grad_output = grad_output.contiguous()
        reduce_dims = [0] + list(range(2, input.dim()))
        c = input.size(1)
        shape = [1, c] + [1] * (input.dim() - 2)

        xhat = (input - mean.reshape(shape)) * invstd.reshape(shape)
        g_sum = grad_output.sum(dim=reduce_dims)
        gx_sum = (grad_output * xhat).sum(dim=reduce_dims)

        grad_weight = gx_sum if weight is not None else None
        grad_bias = g_sum if weight is not None else None

        # cross-rank reduction of the two statistics used by dL/dx
        stats = torch.cat([g_sum, gx_sum])
        stats = allreduce(stats, average=False, process_set=process_set,
                          name=f"sync_batch_norm.bw.{next(_sbn_counter)}")
        g_sum_all, gx_sum_all = stats[:c], stats[c:]

        n = ctx.total_count
        gscale = weight.reshape(shape) if weight is not None else 1.0
        grad_input = (grad_output - (g_sum_all / n).reshape(shape) -
                      xhat * (gx_sum_all / n).reshape(shape)) * \
            invstd.reshape(shape) * gscale
        return (grad_input, grad_weight, grad_bias, None, None, None, None,
                None)
