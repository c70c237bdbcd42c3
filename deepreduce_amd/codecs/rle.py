"""Run-length index codec — vectorized bitmap RLE.

Reference behavior: /root/reference/pytorch/deepreduce.py:808-846 (python
loop over the full bitmap).  Same wire semantics — alternating run lengths
starting with a zero-run (first length may be 0) — computed with tensor ops:
boundary detection + diff on device, run lengths bit-packed via intpack.
"""
from __future__ import annotations

import torch

from . import SparseCompressor
from .intpack import pack_with_header, unpack_with_header


class RunLength(SparseCompressor):
    order_preserving = False

    @staticmethod
    def compress(sparse_tensor, params):
        vals, idxs, shape = sparse_tensor
        d = int(torch.Size(shape).numel())
        s, perm = idxs.long().sort()
        vals = vals[perm]

        bm = torch.zeros(d, dtype=torch.int8, device=idxs.device)
        bm[s] = 1
        change = (bm[1:] != bm[:-1]).nonzero(as_tuple=False).reshape(-1) + 1
        bounds = torch.cat(
            [
                torch.zeros(1, dtype=torch.int64, device=bm.device),
                change,
                torch.tensor([d], dtype=torch.int64, device=bm.device),
            ]
        )
        runs = bounds[1:] - bounds[:-1]
        if bool(bm[0].item()):  # must start with a zero-run
            runs = torch.cat([torch.zeros(1, dtype=torch.int64, device=bm.device), runs])
        wire = pack_with_header(runs)
        return vals, wire, shape

    @staticmethod
    def decompress(rle_sparse_tensor, params):
        vals, wire, shape = rle_sparse_tensor
        runs = unpack_with_header(wire).to(vals.device)
        parity = torch.arange(runs.numel(), device=runs.device) % 2
        bm = torch.repeat_interleave(parity.to(torch.int8), runs)
        idxs = bm.nonzero(as_tuple=False).reshape(-1)
        return vals, idxs, shape
