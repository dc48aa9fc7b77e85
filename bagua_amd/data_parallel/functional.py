"""Autograd-differentiable collectives
(reference: bagua/torch_api/data_parallel/functional.py:56-79)."""

import torch

from ..communication import ReduceOp, allreduce_inplace


class _AllReduce(torch.autograd.Function):
    @staticmethod
    def forward(ctx, op, comm, tensor):
        ctx.op = op
        ctx.comm = comm
        out = tensor.detach().clone()
        allreduce_inplace(out, op=op, comm=comm)
        return out

    @staticmethod
    def backward(ctx, grad_output):
        out = grad_output.detach().clone()
        allreduce_inplace(out, op=ctx.op, comm=ctx.comm)
        return (None, None, out)


def all_reduce(tensor, op=ReduceOp.SUM, comm=None):
    return _AllReduce.apply(op, comm, tensor)
