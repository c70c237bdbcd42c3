"""Sparsifying compressors (GRACE-`Compressor` equivalent layer).

This is the host-runtime layer the reference delegates to GRACE for
(`grace_dl.dist`, see /root/reference/pytorch/deepreduce.py:7 and SURVEY.md
sect. 2.5).  Re-implemented natively for PyTorch-ROCm: the compressor
duck-type used by the DeepReduce wrappers is

    compress(tensor, name) -> ((vals, idxs), ctx)      # ctx = torch.Size
    decompress(tensors, ctx) -> dense tensor
    tensors_size_are_same : bool   # uniform allgather payloads?

The sparsifiers run on whatever device the gradient lives on.  On MI355X
the top-k select is served by the HIP kernel library when available
(deepreduce_amd.ops); torch.topk otherwise.
"""
from __future__ import annotations

import torch

__all__ = [
    "Compressor",
    "NoneCompressor",
    "TopKCompressor",
    "ThresholdCompressor",
    "RandomKCompressor",
    "sparsifier_registry",
]


class Compressor:
    """Base interface for gradient compressors."""

    def __init__(self, average: bool = True, tensors_size_are_same: bool = True):
        self.average = average
        self.tensors_size_are_same = tensors_size_are_same

    def compress(self, tensor: torch.Tensor, name: str):
        raise NotImplementedError

    def decompress(self, tensors, ctx) -> torch.Tensor:
        raise NotImplementedError

    def aggregate(self, tensors):
        """Sum a list of decompressed tensors."""
        return sum(tensors)


class NoneCompressor(Compressor):
    """Identity: dense gradient travels as-is."""

    def compress(self, tensor, name):
        return (tensor,), tensor.size()

    def decompress(self, tensors, ctx):
        return tensors[0].view(ctx)


class TopKCompressor(Compressor):
    """Top-k magnitude sparsification.

    Reference behavior: GRACE `topk` sparsifier used by every DeepReduce
    experiment (/root/reference/run_deepreduce.sh:35,51).  k = max(1,
    round(numel * compress_ratio)); values keep their sign; indices are
    int64 positions into the flattened tensor.
    """

    def __init__(self, compress_ratio: float = 0.01):
        super().__init__(tensors_size_are_same=True)
        self.compress_ratio = compress_ratio

    def compress(self, tensor, name):
        shape = tensor.size()
        flat = tensor.reshape(-1)
        k = max(1, int(round(flat.numel() * self.compress_ratio)))
        from .ops import topk_select

        vals, idxs = topk_select(flat, k)
        # int32 on the wire (universe < 2^31 always): 8 bytes/entry like
        # the reference's int32 keys, not 12
        return (vals, idxs.int()), shape

    def decompress(self, tensors, ctx):
        vals, idxs = tensors
        shape = ctx
        numel = int(torch.Size(shape).numel())
        dense = torch.zeros(numel, dtype=vals.dtype, device=vals.device)
        dense.scatter_(0, idxs.long(), vals)
        return dense.view(shape)


class ThresholdCompressor(Compressor):
    """Keep entries with |g| >= threshold.  Payload sizes differ per rank."""

    def __init__(self, threshold: float = 0.01):
        super().__init__(tensors_size_are_same=False)
        self.threshold = threshold

    def compress(self, tensor, name):
        shape = tensor.size()
        flat = tensor.reshape(-1)
        mask = flat.abs() >= self.threshold
        idxs = mask.nonzero(as_tuple=False).reshape(-1)
        vals = flat[idxs]
        return (vals, idxs.int()), shape

    decompress = TopKCompressor.decompress


class RandomKCompressor(Compressor):
    """Random-k sparsification, seeded by (name, step) so every rank picks
    the same positions for the same tensor in the same step.

    Reference behavior: tensorflow/deepreduce.py:290-298 (seed =
    hash(name) + step).
    """

    def __init__(self, compress_ratio: float = 0.01):
        super().__init__(tensors_size_are_same=True)
        self.compress_ratio = compress_ratio
        self.step = 0

    def compress(self, tensor, name):
        shape = tensor.size()
        flat = tensor.reshape(-1)
        numel = flat.numel()
        k = max(1, int(round(numel * self.compress_ratio)))
        # deterministic across ranks: CPU generator seeded from (name, step)
        seed = (hash(name) & 0x7FFFFFFF) + self.step
        g = torch.Generator(device="cpu")
        g.manual_seed(seed)
        idxs = torch.randperm(numel, generator=g)[:k].to(flat.device)
        self.step += 1
        vals = flat[idxs]
        return (vals, idxs.int()), shape

    decompress = TopKCompressor.decompress


sparsifier_registry = {
    "none": NoneCompressor,
    "topk": TopKCompressor,
    "threshold": ThresholdCompressor,
    "randomk": RandomKCompressor,
}
