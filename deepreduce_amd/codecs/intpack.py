"""n-bit integer packing + the FastPFor-equivalent 'pfor' index codec.

Replaces (a) the reference's cupy byte/bit-plane packer
(/root/reference/pytorch/deepreduce.py:193-248) with a plain LSB-first
bitstream, and (b) the FastPFor TF ops
(/root/reference/tensorflow/integer_compression.cc) with a delta +
bit-packing codec of the same family, GPU-resident.

Header layout (uint8): [4B count LE][1B nbits][payload bitstream].
"""
from __future__ import annotations

import torch

from .. import ops
from . import SparseCompressor


def pack_with_header(values: torch.Tensor, nbits: int | None = None) -> torch.Tensor:
    v = values.long()
    n = v.numel()
    if nbits is None:
        mx = int(v.max().item()) if n else 0
        nbits = max(1, mx.bit_length())
    if nbits > 57:
        # a value may straddle 9 bytes beyond this width, which the
        # 8-byte-window unpack kernels do not support; real wires
        # (indices/gaps/runs/mappings) never exceed 31 bits
        raise ValueError(f"nbits={nbits} unsupported (max 57)")
    stream = ops.pack_ints(v, nbits)
    header = torch.tensor(
        [n & 255, (n >> 8) & 255, (n >> 16) & 255, (n >> 24) & 255, nbits],
        dtype=torch.uint8,
        device=values.device,
    )
    return torch.cat([header, stream])


def unpack_with_header(wire: torch.Tensor) -> torch.Tensor:
    header = wire[:5].cpu()
    n = int(header[0]) | (int(header[1]) << 8) | (int(header[2]) << 16) | (int(header[3]) << 24)
    nbits = int(header[4])
    return ops.unpack_ints(wire[5:], n, nbits)


class PFor(SparseCompressor):
    """Sorted-delta + bit-pack index codec ('pfor').

    compress: sort idxs ascending (vals permuted to match), deltas =
    [idx0, idx1-idx0-1, ...] (gaps), bit-packed at the max gap width.
    Order-preserving w.r.t. ascending index order only.
    """

    order_preserving = False

    @staticmethod
    def compress(sparse_tensor, params):
        vals, idxs, shape = sparse_tensor
        s, perm = idxs.long().sort()
        vals = vals[perm]
        if s.numel():
            gaps = torch.cat([s[:1], s[1:] - s[:-1] - 1])
        else:
            gaps = s
        wire = pack_with_header(gaps)
        return vals, wire, shape

    @staticmethod
    def decompress(sparse_tensor, params):
        vals, wire, shape = sparse_tensor
        gaps = unpack_with_header(wire).to(vals.device)
        if gaps.numel():
            idxs = torch.cumsum(gaps + 1, 0) - 1
        else:
            idxs = gaps
        return vals, idxs, shape
