"""Pure-PyTorch reference implementations of every hot op.

These define the SEMANTICS (each HIP kernel in ops/src/ is tested against
these, per SURVEY.md sect. 4) and serve as the CPU execution path for the
gloo-based tests.  They are written in chunked tensor ops so the CPU path
stays usable at real gradient sizes.

Wire bit order (shared with the HIP kernels): bit b of a Bloom filter lives
in byte b>>3 at bit position b&7 (LSB-first); equivalently in little-endian
uint32 word b>>5 at bit b&31.
"""
from __future__ import annotations

import torch

from ..hashing import bloom_positions

_QUERY_CHUNK = 1 << 22  # elements per chunk in full-universe queries


def topk_select(flat: torch.Tensor, k: int):
    """(vals, idxs) of the k largest-|.| entries.  idxs int64, unordered."""
    _, idxs = torch.topk(flat.abs(), k, sorted=False)
    return flat[idxs], idxs


def bloom_insert(idxs: torch.Tensor, m: int, num_hash: int) -> torch.Tensor:
    """Build an m-bit Bloom filter; returns packed uint8 tensor (ceil(m/8))."""
    device = idxs.device
    bits = torch.zeros(m, dtype=torch.bool, device=device)
    pos = bloom_positions(idxs, num_hash, m).reshape(-1)
    bits[pos] = True
    return pack_bitarray(bits)


def pack_bitarray(bits: torch.Tensor) -> torch.Tensor:
    """bool[m] -> uint8[ceil(m/8)], LSB-first."""
    m = bits.numel()
    nbytes = (m + 7) // 8
    padded = torch.zeros(nbytes * 8, dtype=torch.uint8, device=bits.device)
    padded[:m] = bits.to(torch.uint8)
    weights = (1 << torch.arange(8, device=bits.device, dtype=torch.int16)).to(torch.uint8)
    return (padded.view(-1, 8) * weights).sum(dim=1, dtype=torch.uint8)


def unpack_bitarray(packed: torch.Tensor, m: int) -> torch.Tensor:
    """uint8[ceil(m/8)] -> bool[m]."""
    shifts = torch.arange(8, device=packed.device, dtype=torch.uint8)
    bits = (packed.unsqueeze(1) >> shifts.unsqueeze(0)) & 1
    return bits.reshape(-1)[:m].bool()


def bloom_query_positives(
    packed: torch.Tensor, m: int, num_hash: int, universe: int
) -> torch.Tensor:
    """Sorted int64 indices i in [0, universe) whose k bits are all set.

    This is the #1 hot op of the whole framework (SURVEY.md sect. 7): O(d*k)
    over the full gradient size on every decompress.
    """
    device = packed.device
    out = []
    for start in range(0, universe, _QUERY_CHUNK):
        end = min(start + _QUERY_CHUNK, universe)
        items = torch.arange(start, end, device=device, dtype=torch.int64)
        pos = bloom_positions(items, num_hash, m)  # [n, k]
        byte = packed[(pos >> 3)]
        bit = ((byte >> (pos & 7).to(torch.uint8)) & 1).bool()
        mask = bit.all(dim=1)
        out.append(items[mask])
    return torch.cat(out) if out else torch.empty(0, dtype=torch.int64, device=device)


def bloom_query_members(
    packed: torch.Tensor, m: int, num_hash: int, items: torch.Tensor
) -> torch.Tensor:
    """bool[n]: membership test of arbitrary items (used by tests/policies)."""
    pos = bloom_positions(items, num_hash, m)
    byte = packed[(pos >> 3)]
    bit = ((byte >> (pos & 7).to(torch.uint8)) & 1).bool()
    return bit.all(dim=1)


# ---------------------------------------------------------------------------
# n-bit integer packing (replaces the reference's cupy byte/bit-plane packer,
# /root/reference/pytorch/deepreduce.py:193-248, with a plain LSB-first
# bitstream: value v_i occupies bits [i*nbits, (i+1)*nbits) of a uint8 stream)
# ---------------------------------------------------------------------------

def pack_ints(values: torch.Tensor, nbits: int) -> torch.Tensor:
    """Non-negative ints < 2**nbits -> uint8 bitstream (no header)."""
    n = values.numel()
    device = values.device
    v = values.long()
    total_bits = n * nbits
    nbytes = (total_bits + 7) // 8
    # bit-plane approach: nbits scatters of one bit-plane each
    out_bits = torch.zeros(nbytes * 8, dtype=torch.uint8, device=device)
    base = torch.arange(n, device=device, dtype=torch.int64) * nbits
    for b in range(nbits):
        out_bits[base + b] = ((v >> b) & 1).to(torch.uint8)
    weights = (1 << torch.arange(8, device=device, dtype=torch.int16)).to(torch.uint8)
    return (out_bits.view(-1, 8) * weights).sum(dim=1, dtype=torch.uint8)


def unpack_ints(stream: torch.Tensor, n: int, nbits: int) -> torch.Tensor:
    device = stream.device
    shifts = torch.arange(8, device=device, dtype=torch.uint8)
    bits = ((stream.unsqueeze(1) >> shifts.unsqueeze(0)) & 1).reshape(-1)
    base = torch.arange(n, device=device, dtype=torch.int64) * nbits
    out = torch.zeros(n, dtype=torch.int64, device=device)
    for b in range(nbits):
        out |= bits[base + b].long() << b
    return out


# ---------------------------------------------------------------------------
# QSGD bucketed stochastic quantization
# (reference: /root/reference/pytorch/deepreduce.py:852-907; python loop over
#  512-element buckets -> vectorized here, HIP kernel fuses it per-wave)
# ---------------------------------------------------------------------------

def qsgd_quantize(vals: torch.Tensor, quantum_num: int, bucket_size: int):
    """-> (levels int8[n], norms float32[nb]); levels = signed quantized."""
    n = vals.numel()
    nb = (n + bucket_size - 1) // bucket_size
    pad = nb * bucket_size - n
    v = torch.nn.functional.pad(vals.float(), (0, pad)).view(nb, bucket_size)
    norms = v.norm(dim=1)
    safe = torch.where(norms == 0, torch.ones_like(norms), norms)
    level_float = quantum_num / safe.unsqueeze(1) * v.abs()
    previous = level_float.floor()
    prob = torch.rand_like(v)
    new_level = previous + (prob < (level_float - previous)).float()
    levels = (new_level * v.sign()).to(torch.int8)
    return levels.reshape(-1)[:n].contiguous(), norms


def qsgd_dequantize(levels: torch.Tensor, norms: torch.Tensor, quantum_num: int, bucket_size: int):
    n = levels.numel()
    nb = norms.numel()
    pad = nb * bucket_size - n
    l = torch.nn.functional.pad(levels.float(), (0, pad)).view(nb, bucket_size)
    v = norms.unsqueeze(1) / quantum_num * l
    return v.reshape(-1)[:n].contiguous()
