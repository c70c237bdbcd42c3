"""Huffman index codec (order-preserving, CPU).

Reference behavior: /root/reference/pytorch/deepreduce.py:770-802 — a
Huffman codec built DETERMINISTICALLY from the byte stream of
range(grad_size) int32s, so both sides derive the same code table from the
shape alone and nothing but the encoded index bytes travels.

Native replacement for `dahuffman` (unavailable): canonical Huffman with an
ANALYTIC byte histogram of the int32-LE encoding of 0..d-1 (no O(d) buffer),
numpy-vectorized encode, table-walk decode.
"""
from __future__ import annotations

import heapq
from functools import lru_cache

import numpy as np
import torch

from . import SparseCompressor


@lru_cache(maxsize=64)
def _range_byte_freq(d: int) -> tuple:
    """Byte histogram (256,) of the little-endian int32 bytes of range(d)."""
    freq = np.zeros(256, dtype=np.int64)
    for p in range(4):
        block = 1 << (8 * p)          # count of consecutive i sharing a byte value
        period = block * 256
        full, rem = divmod(d, period)
        freq += full * block          # every byte value appears `block` times per period
        vals = np.minimum(np.maximum(rem - np.arange(256) * block, 0), block)
        freq += vals
    return tuple(freq.tolist())


@lru_cache(maxsize=64)
def _canonical_code(d: int):
    """(codes uint32[256], lengths uint8[256]) canonical Huffman for d."""
    freq = _range_byte_freq(d)
    heap = [(f, sym, sym) for sym, f in enumerate(freq) if f > 0]
    # (freq, tiebreak, id); tree via heapq, deterministic tie-break by id
    heap = [(f, i, ("leaf", s)) for i, (f, s, _) in enumerate(sorted(heap, key=lambda t: t[1]))]
    heapq.heapify(heap)
    nxt = len(heap)
    if len(heap) == 1:
        lengths = np.zeros(256, dtype=np.uint8)
        lengths[heap[0][2][1]] = 1
    else:
        while len(heap) > 1:
            f1, _, n1 = heapq.heappop(heap)
            f2, _, n2 = heapq.heappop(heap)
            heapq.heappush(heap, (f1 + f2, nxt, ("node", n1, n2)))
            nxt += 1
        lengths = np.zeros(256, dtype=np.uint8)

        def walk(node, depth):
            if node[0] == "leaf":
                lengths[node[1]] = max(depth, 1)
            else:
                walk(node[1], depth + 1)
                walk(node[2], depth + 1)

        walk(heap[0][2], 0)
    # canonical assignment: sort by (length, symbol)
    codes = np.zeros(256, dtype=np.uint32)
    order = sorted([s for s in range(256) if lengths[s] > 0], key=lambda s: (lengths[s], s))
    code = 0
    prev_len = 0
    for s in order:
        code <<= int(lengths[s]) - prev_len
        codes[s] = code
        prev_len = int(lengths[s])
        code += 1
    return codes, lengths


class Huffman(SparseCompressor):
    order_preserving = True

    @staticmethod
    def compress(sparse_tensor, params):
        vals, idxs, shape = sparse_tensor
        d = int(torch.Size(shape).numel())
        codes, lengths = _canonical_code(d)

        data = idxs.cpu().numpy().astype(np.int32).tobytes()
        syms = np.frombuffer(data, dtype=np.uint8)
        clens = lengths[syms].astype(np.int64)
        ccodes = codes[syms].astype(np.int64)
        offsets = np.cumsum(clens) - clens
        total = int(offsets[-1] + clens[-1]) if len(syms) else 0
        bits = np.zeros(total, dtype=np.uint8)
        maxlen = int(clens.max()) if len(syms) else 0
        for j in range(maxlen):  # MSB-first within each code
            sel = clens > j
            bits[offsets[sel] + j] = (ccodes[sel] >> (clens[sel] - 1 - j)) & 1
        nbytes = (total + 7) // 8
        padded = np.zeros(nbytes * 8, dtype=np.uint8)
        padded[:total] = bits
        stream = (padded.reshape(-1, 8) << np.arange(8, dtype=np.uint8)).sum(axis=1).astype(np.uint8)
        n = len(syms)
        header = np.array([n & 255, (n >> 8) & 255, (n >> 16) & 255, (n >> 24) & 255], dtype=np.uint8)
        wire = torch.from_numpy(np.concatenate([header, stream])).to(vals.device)
        return vals, wire, shape

    @staticmethod
    def decompress(sparse_tensor, params):
        vals, wire, shape = sparse_tensor
        d = int(torch.Size(shape).numel())
        codes, lengths = _canonical_code(d)
        raw = wire.cpu().numpy()
        n = int(raw[0]) | (int(raw[1]) << 8) | (int(raw[2]) << 16) | (int(raw[3]) << 24)
        stream = raw[4:]
        bits = ((stream[:, None] >> np.arange(8, dtype=np.uint8)) & 1).reshape(-1)

        # canonical decode tables: for each length, (first_code, first_index)
        order = sorted([s for s in range(256) if lengths[s] > 0], key=lambda s: (lengths[s], s))
        sym_by_rank = np.array(order, dtype=np.uint8)
        first_code, first_rank = {}, {}
        rank = 0
        code = 0
        prev_len = 0
        for s in order:
            L = int(lengths[s])
            code <<= L - prev_len
            if L not in first_code:
                first_code[L] = code
                first_rank[L] = rank
            prev_len = L
            code += 1
            rank += 1
        max_code_end = {}  # last code value per length + 1
        code = 0
        prev_len = 0
        for s in order:
            L = int(lengths[s])
            code <<= L - prev_len
            prev_len = L
            code += 1
            max_code_end[L] = code

        from .. import ops

        if ops._cpu_native():
            from deepreduce_amd import _hip_ops

            out = _hip_ops.huffman_decode_cpu(
                torch.from_numpy(stream.copy()), n,
                torch.from_numpy(codes.astype(np.int64)),
                torch.from_numpy(lengths.astype(np.int64)),
            ).numpy()
        else:
            out = np.empty(n, dtype=np.uint8)
            pos = 0
            for i in range(n):
                acc = 0
                L = 0
                while True:
                    acc = (acc << 1) | int(bits[pos])
                    pos += 1
                    L += 1
                    if L in first_code and acc < max_code_end[L]:
                        out[i] = sym_by_rank[first_rank[L] + (acc - first_code[L])]
                        break
        idxs_np = np.frombuffer(out.tobytes(), dtype=np.int32).copy() if n else np.empty(0, np.int32)
        idxs = torch.from_numpy(idxs_np).long().to(vals.device)
        return vals, idxs, shape
