"""Deflate value codec (order-preserving, CPU).

Reference behavior: /root/reference/pytorch/deepreduce.py:742-764 — zlib
Deflate over the raw float32 bytes of the values.
"""
from __future__ import annotations

import zlib

import numpy as np
import torch

from . import SparseCompressor


class Gzip(SparseCompressor):
    order_preserving = True

    @staticmethod
    def compress(sparse_tensor, params):
        vals, idxs, shape = sparse_tensor
        raw = vals.float().cpu().numpy().tobytes()
        packed = zlib.compress(raw)
        wire = torch.from_numpy(np.frombuffer(packed, dtype=np.uint8).copy()).to(idxs.device)
        return wire, idxs, shape

    @staticmethod
    def decompress(gzip_sparse_tensor, params):
        wire, idxs, shape = gzip_sparse_tensor
        raw = zlib.decompress(wire.cpu().numpy().tobytes())
        vals = torch.from_numpy(np.frombuffer(raw, dtype=np.float32).copy()).to(idxs.device)
        return vals, idxs, shape
