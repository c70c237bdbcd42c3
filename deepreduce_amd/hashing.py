"""Deterministic device-side hashing for the Bloom codec.

The reference ships a PRECOMPUTED MurmurHash3 table (`hash_table_18m_int.pt`,
~1 GB GPU-resident for NCF — /root/reference/pytorch/deepreduce.py:32,43 and
paper App. E).  The MI355X build replaces the table with in-register hashing:
Kirsch-Mitzenmacher double hashing over the murmur3 finalizer (fmix32), so a
query costs 2 fmix + k fused-multiply-adds instead of k gathers from a 1 GB
table.

CRITICAL INVARIANT: decompress re-derives indices independently on every rank
(SURVEY.md sect. 7 "Determinism across ranks"), so this module and the HIP
kernels in ops/src/bloom.hip MUST be bit-identical.  All torch arithmetic is
int64 emulating uint32 wraparound; any change here must be mirrored in the
kernel and covered by tests/test_bloom.py parity tests.
"""
from __future__ import annotations

import torch

MASK32 = 0xFFFFFFFF
_FM1 = 0x85EBCA6B
_FM2 = 0xC2B2AE35
H2_SALT = 0x6B43A9B5


def _mul32(a: torch.Tensor, m: int) -> torch.Tensor:
    """(a * m) mod 2**32 for int64 tensors holding uint32 values.

    Split into 16-bit halves to keep every intermediate below 2**63.
    """
    lo = (a & 0xFFFF) * m
    hi = (((a >> 16) * m) & 0xFFFF) << 16
    return (lo + hi) & MASK32


def fmix32(h: torch.Tensor) -> torch.Tensor:
    """MurmurHash3 32-bit finalizer (avalanche) on int64-as-uint32 tensors."""
    h = (h ^ (h >> 16)) & MASK32
    h = _mul32(h, _FM1)
    h = (h ^ (h >> 13)) & MASK32
    h = _mul32(h, _FM2)
    h = (h ^ (h >> 16)) & MASK32
    return h


def double_hash_bases(items: torch.Tensor):
    """Per-item (h1, h2) for Kirsch-Mitzenmacher double hashing.

    hash_j(item) = mulshift(h1 + j * h2, m) — see bloom_positions.  h2 is
    forced odd so the 32-bit sequence h1 + j*h2 cycles all residues.
    `items` may be any integer dtype; computed as int64.
    """
    x = items.long() & MASK32
    h1 = fmix32((x + 1) & MASK32)
    h2 = fmix32(h1 ^ H2_SALT) | 1
    return h1, h2


def bloom_positions(items: torch.Tensor, num_hash: int, m: int) -> torch.Tensor:
    """[n, k] int64 bit positions in a Bloom filter of m bits.

    Position = ((h1 + j*h2 mod 2^32) * m) >> 32 — Lemire multiply-shift
    range reduction instead of `mod m`.  Uniform on [0, m) for m < 2^31,
    and on the GPU it replaces two 64-bit integer divisions per candidate
    (the dominant VALU cost of the universe query kernel) with one 32x32
    multiply-high per probe.  MUST stay bit-identical to hash_bases/
    position math in ops/src/hip_ops.hip (parity-tested).
    """
    h1, h2 = double_hash_bases(items)
    j = torch.arange(num_hash, device=items.device, dtype=torch.int64)
    x = (h1.unsqueeze(1) + j.unsqueeze(0) * h2.unsqueeze(1)) & MASK32
    return (x * m) >> 32
