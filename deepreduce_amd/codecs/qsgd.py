"""QSGD value codec — bucketed stochastic quantization, order-preserving.

Reference behavior: /root/reference/pytorch/deepreduce.py:852-907 (python
loop over 512-element buckets; norm packed as 4 offset int8 bytes appended
per bucket).  Here the whole tensor quantizes in one fused op
(ops.qsgd_quantize -> HIP kernel: one wave per bucket) and the wire layout
is preserved: per bucket `bucket_size` int8 levels followed by the 4 bytes
of the bucket's float32 L2 norm (each byte - 128).
"""
from __future__ import annotations

import torch

from .. import ops
from . import SparseCompressor


def _norm_bytes(norms: torch.Tensor) -> torch.Tensor:
    """float32[nb] -> int8[nb,4] (little-endian bytes, offset by -128)."""
    b = norms.float().contiguous().view(torch.uint8).view(-1, 4)
    return (b.to(torch.int16) - 128).to(torch.int8)


def _bytes_norm(b8: torch.Tensor) -> torch.Tensor:
    b = (b8.to(torch.int16) + 128).to(torch.uint8).contiguous()
    return b.view(-1).view(torch.float32)


class QSGD(SparseCompressor):
    order_preserving = True

    @staticmethod
    def compress(sparse_tensor, params):
        vals, idxs, shape = sparse_tensor
        quantum_num = int(params.get("quantum_num", 127))
        bucket_size = int(params.get("bucket_size", 512))

        levels, norms = ops.qsgd_quantize(vals.float(), quantum_num, bucket_size)
        nb = norms.numel()
        n = levels.numel()
        nbytes = _norm_bytes(norms)  # [nb, 4]

        bits = int(quantum_num * 2 + 1).bit_length()
        if params.get("qsgd_pack", False) and bits < 8:
            # sub-byte level packing (paper Table 2 runs QSGD at 7 bits):
            # wire = [levels+q packed at `bits` bits][norm bytes]
            packed = ops.pack_ints((levels.to(torch.int64) + quantum_num), bits)
            return torch.cat([packed.view(torch.int8),
                              nbytes.reshape(-1)]), idxs, shape

        full = (nb - 1) if n % bucket_size else nb
        pieces = []
        if full > 0:
            body = torch.cat(
                [levels[: full * bucket_size].view(full, bucket_size), nbytes[:full]], dim=1
            ).reshape(-1)
            pieces.append(body)
        if full < nb:  # ragged tail bucket
            pieces.append(levels[full * bucket_size :])
            pieces.append(nbytes[full].reshape(-1))
        wire = torch.cat(pieces)
        return wire, idxs, shape

    @staticmethod
    def decompress(sparse_tensor, params):
        wire, idxs, shape = sparse_tensor
        quantum_num = int(params.get("quantum_num", 127))
        bucket_size = int(params.get("bucket_size", 512))
        stride = bucket_size + 4

        bits = int(quantum_num * 2 + 1).bit_length()
        if params.get("qsgd_pack", False) and bits < 8:
            n = int(idxs.numel())
            nb = (n + bucket_size - 1) // bucket_size
            nbits_bytes = (n * bits + 7) // 8
            packed, nb8 = wire.split([nbits_bytes, 4 * nb])
            levels = (ops.unpack_ints(packed.view(torch.uint8).contiguous(), n, bits)
                      - quantum_num).to(torch.int8)
            norms = _bytes_norm(nb8)
            vals = ops.qsgd_dequantize(levels, norms, quantum_num, bucket_size)
            return vals, idxs, shape

        total = wire.numel()
        full = total // stride
        rem = total - full * stride
        if rem:
            body, tail = wire.split([full * stride, rem])
        else:
            body, tail = wire, None
        levels_parts, norms_parts = [], []
        if full:
            b = body.view(full, stride)
            levels_parts.append(b[:, :bucket_size].reshape(-1))
            norms_parts.append(_bytes_norm(b[:, bucket_size:]))
        if tail is not None:
            levels_parts.append(tail[: rem - 4])
            norms_parts.append(_bytes_norm(tail[rem - 4 :]))
        levels = torch.cat(levels_parts)
        norms = torch.cat(norms_parts)
        vals = ops.qsgd_dequantize(levels, norms, quantum_num, bucket_size)
        return vals, idxs, shape
