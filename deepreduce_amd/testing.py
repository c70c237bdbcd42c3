"""Test-support utilities.

`stage_collectives_via_cpu()` — CPU-staged collective shim for exercising
the full GPU compression pipeline across MULTIPLE PROCESSES on a machine
with fewer GPUs than ranks.  RCCL refuses co-located ranks ("Duplicate GPU
detected", librccl init check), and gloo does not implement all_gather on
CUDA tensors — so neither backend alone can run a world-size-2 exchange on
a 1-GPU box.  The shim wraps torch.distributed's collectives to stage CUDA
tensors through pinned CPU copies over gloo: every HIP kernel, wire layout
and multi-rank decode runs exactly as in production; only the transport is
substituted.  NOT for production use (the staging copies serialize on the
PCIe link); the real path is backend "nccl" (= RCCL) with one rank per GPU.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

__all__ = ["stage_collectives_via_cpu"]


class _DoneWork:
    def wait(self, timeout=None):
        return True

    def is_completed(self):
        return True


def stage_collectives_via_cpu():
    """Monkeypatch dist.{all_gather, all_gather_into_tensor, all_reduce,
    broadcast} to stage CUDA tensors through CPU + gloo.  Idempotent.
    Returns a restore() callable."""
    if getattr(dist, "_deepreduce_staged", False):
        return lambda: None
    orig = {
        "all_gather": dist.all_gather,
        "all_gather_into_tensor": getattr(dist, "all_gather_into_tensor", None),
        "all_reduce": dist.all_reduce,
        "broadcast": dist.broadcast,
    }

    def all_gather(tensor_list, tensor, group=None, async_op=False):
        if tensor.is_cuda:
            cpu_outs = [torch.empty(t.shape, dtype=t.dtype) for t in tensor_list]
            orig["all_gather"](cpu_outs, tensor.cpu(), group=group)
            for d, s in zip(tensor_list, cpu_outs):
                d.copy_(s)
            return _DoneWork() if async_op else None
        return orig["all_gather"](tensor_list, tensor, group=group,
                                  async_op=async_op)

    def all_gather_into_tensor(output, input, group=None, async_op=False):
        if input.is_cuda:
            world = dist.get_world_size(group)
            outs = list(output.view(world, -1).unbind(0))
            all_gather(outs, input.reshape(-1), group=group)
            return _DoneWork() if async_op else None
        return orig["all_gather_into_tensor"](output, input, group=group,
                                              async_op=async_op)

    def all_reduce(tensor, op=dist.ReduceOp.SUM, group=None, async_op=False):
        if tensor.is_cuda:
            cpu = tensor.cpu()
            orig["all_reduce"](cpu, op=op, group=group)
            tensor.copy_(cpu)
            return _DoneWork() if async_op else None
        return orig["all_reduce"](tensor, op=op, group=group, async_op=async_op)

    def broadcast(tensor, src, group=None, async_op=False):
        if tensor.is_cuda:
            cpu = tensor.cpu()
            orig["broadcast"](cpu, src, group=group)
            tensor.copy_(cpu)
            return _DoneWork() if async_op else None
        return orig["broadcast"](tensor, src, group=group, async_op=async_op)

    dist.all_gather = all_gather
    if orig["all_gather_into_tensor"] is not None:
        dist.all_gather_into_tensor = all_gather_into_tensor
    dist.all_reduce = all_reduce
    dist.broadcast = broadcast
    dist._deepreduce_staged = True

    def restore():
        dist.all_gather = orig["all_gather"]
        if orig["all_gather_into_tensor"] is not None:
            dist.all_gather_into_tensor = orig["all_gather_into_tensor"]
        dist.all_reduce = orig["all_reduce"]
        dist.broadcast = orig["broadcast"]
        dist._deepreduce_staged = False

    return restore
