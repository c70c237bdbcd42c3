"""Small shared helpers (GRACE-`helper` equivalent).

Reference parity: grace_dl.dist.helper.tensor_bits (used by
/root/reference/pytorch/deepreduce.py:8,93-95) and the flat params-dict
contract of /root/reference/README.md:31-37.
"""
from __future__ import annotations

import torch

__all__ = ["tensor_bits", "tensor_bytes", "world_size", "rank"]


def tensor_bits(tensors) -> int:
    """Total wire size, in bits, of a list of tensors.

    Matches the reference's volume accounting: each element costs its
    storage dtype's width (float32 -> 32, int8/uint8 -> 8, bool -> 8).
    """
    total = 0
    for t in tensors:
        if t is None:
            continue
        if isinstance(t, (tuple, list)):
            total += tensor_bits(t)
            continue
        total += t.numel() * t.element_size() * 8
    return total


def tensor_bytes(tensors) -> int:
    return tensor_bits(tensors) // 8


def world_size() -> int:
    if torch.distributed.is_available() and torch.distributed.is_initialized():
        return torch.distributed.get_world_size()
    return 1


def rank() -> int:
    if torch.distributed.is_available() and torch.distributed.is_initialized():
        return torch.distributed.get_rank()
    return 0
