"""Synchronized BatchNorm over bagua collectives
(reference: bagua/torch_api/contrib/sync_batchnorm.py:24-150, itself
Horovod-derived).

Forward reduces per-channel (sum, sqsum, count) with one allreduce;
backward reduces (sum_dy, sum_dy_xmu). Works on CPU/gloo and GPU/RCCL
through the same communication layer.
"""

import torch
import torch.nn.functional as F
from torch.nn.modules.batchnorm import _BatchNorm

from ..communication import ReduceOp, allreduce_inplace

__all__ = ["SyncBatchNorm"]


class SyncBatchNorm(_BatchNorm):
    def __init__(self, num_features, eps=1e-5, momentum=0.1, affine=True,
                 track_running_stats=True):
        super().__init__(num_features, eps, momentum, affine,
                         track_running_stats)

    def _check_input_dim(self, input):
        if input.dim() < 2:
            raise ValueError("expected at least 2D input (got %dD)"
                             % input.dim())

    def forward(self, input):
        self._check_input_dim(input)
        if not self.training and self.track_running_stats:
            return F.batch_norm(
                input, self.running_mean, self.running_var, self.weight,
                self.bias, False, 0.0, self.eps)
        return _SyncBatchNormFn.apply(
            input, self.weight, self.bias, self.running_mean,
            self.running_var, self.eps, self.momentum)

    @classmethod
    def convert_sync_batchnorm(cls, module: torch.nn.Module):
        """Replace every BatchNorm*d in ``module`` with SyncBatchNorm
        (reference: sync_batchnorm.py:109+)."""
        module_output = module
        if isinstance(module, _BatchNorm):
            module_output = SyncBatchNorm(
                module.num_features, module.eps, module.momentum,
                module.affine, module.track_running_stats)
            if module.affine:
                with torch.no_grad():
                    module_output.weight = module.weight
                    module_output.bias = module.bias
            module_output.running_mean = module.running_mean
            module_output.running_var = module.running_var
            module_output.num_batches_tracked = module.num_batches_tracked
        for name, child in module.named_children():
            module_output.add_module(name,
                                     cls.convert_sync_batchnorm(child))
        del module
        return module_output


class _SyncBatchNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, bias, running_mean, running_var, eps,
                momentum):
        input = input.contiguous()
        C = input.shape[1]
        reduce_dims = [0] + list(range(2, input.dim()))
        local_count = input.numel() // C

        stats = torch.empty(2 * C + 1, dtype=torch.float32,
                            device=input.device)
        stats[:C] = input.float().sum(dim=reduce_dims)
        stats[C:2 * C] = (input.float() ** 2).sum(dim=reduce_dims)
        stats[2 * C] = local_count
        allreduce_inplace(stats, op=ReduceOp.SUM)

        count = stats[2 * C]
        mean = stats[:C] / count
        var = stats[C:2 * C] / count - mean ** 2
        invstd = torch.rsqrt(var + eps)

        if running_mean is not None:
            with torch.no_grad():
                unbiased = var * count / (count - 1)
                running_mean.mul_(1 - momentum).add_(
                    mean.to(running_mean.dtype), alpha=momentum)
                running_var.mul_(1 - momentum).add_(
                    unbiased.to(running_var.dtype), alpha=momentum)

        shape = [1, C] + [1] * (input.dim() - 2)
        xhat = (input.float() - mean.view(shape)) * invstd.view(shape)
        out = xhat
        if weight is not None:
            out = out * weight.float().view(shape)
        if bias is not None:
            out = out + bias.float().view(shape)

        ctx.save_for_backward(input, weight, mean, invstd, count)
        return out.to(input.dtype)

    @staticmethod
    def backward(ctx, grad_output):
        input, weight, mean, invstd, count = ctx.saved_tensors
        grad_output = grad_output.contiguous().float()
        C = input.shape[1]
        reduce_dims = [0] + list(range(2, input.dim()))
        shape = [1, C] + [1] * (input.dim() - 2)

        xmu = input.float() - mean.view(shape)
        sum_dy = grad_output.sum(dim=reduce_dims)
        sum_dy_xmu = (grad_output * xmu).sum(dim=reduce_dims)

        grad_weight = grad_bias = None
        if weight is not None and ctx.needs_input_grad[1]:
            grad_weight = (sum_dy_xmu * invstd).to(weight.dtype)
        if ctx.needs_input_grad[2]:
            grad_bias = sum_dy.clone().to(input.dtype)

        packed = torch.cat([sum_dy, sum_dy_xmu])
        allreduce_inplace(packed, op=ReduceOp.SUM)
        g_sum_dy = packed[:C]
        g_sum_dy_xmu = packed[C:]

        w = weight.float().view(shape) if weight is not None else 1.0
        term1 = grad_output
        term2 = g_sum_dy.view(shape) / count
        term3 = xmu * invstd.view(shape) ** 2 * g_sum_dy_xmu.view(shape) \
            / count
        grad_input = ((term1 - term2 - term3) * invstd.view(shape)
                      * w).to(input.dtype)
        return grad_input, grad_weight, grad_bias, None, None, None, None
