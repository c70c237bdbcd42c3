"""Op dispatch: HIP/CDNA4 kernels on GPU, torch reference on CPU.

The HIP extension (`deepreduce_amd._hip_ops`, built in-tree from ops/src/ by
setup.py / __graft_entry__.build) is the ONLY execution path on a GPU: if a
tensor lives on a CUDA (ROCm) device and the extension failed to load, the op
raises instead of silently falling back to eager PyTorch.  Set
DEEPREDUCE_ALLOW_EAGER=1 to override (debug only).
"""
from __future__ import annotations

import os

import torch

from . import reference as _ref

_hip = None
_hip_err: str | None = None


def _load_hip():
    global _hip, _hip_err
    if _hip is not None or _hip_err is not None:
        return _hip
    try:
        from deepreduce_amd import _hip_ops  # built in-tree .so

        _hip = _hip_ops
    except Exception as e:  # noqa: BLE001
        _hip_err = str(e)
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _want_hip(t: torch.Tensor) -> bool:
    if not t.is_cuda:
        return False
    if _load_hip() is not None:
        return True
    if os.environ.get("DEEPREDUCE_ALLOW_EAGER") == "1":
        return False
    raise RuntimeError(
        "deepreduce_amd: tensor is on a GPU but the HIP extension "
        f"deepreduce_amd._hip_ops is not loadable ({_hip_err}). Build it with "
        "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950), or "
        "set DEEPREDUCE_ALLOW_EAGER=1 to force the eager fallback (slow)."
    )


# --------------------------------------------------------------------------
# public ops
# --------------------------------------------------------------------------

def topk_select(flat: torch.Tensor, k: int):
    """(vals, idxs) of the k largest-magnitude entries.

    GPU: deterministic two-level radix select (ops/src/hip_ops.hip) — exact
    top-k up to ties within the top 22 bits of |x| (< 0.012% relative),
    resolved to the lowest index.  CPU: torch.topk.
    DEEPREDUCE_TORCH_TOPK=1 forces torch.topk on GPU too.
    """
    if (
        flat.is_cuda
        and flat.dtype == torch.float32
        and 1 <= k <= flat.numel()
        and flat.numel() >= int(os.environ.get("DEEPREDUCE_RADIX_MIN", "4096"))
        # tiny tensors: torch.topk is fewer launches
        and os.environ.get("DEEPREDUCE_TORCH_TOPK") != "1"
        and _want_hip(flat)
    ):
        out = _hip.topk_select(flat, k)
        return out[0], out[1]
    return _ref.topk_select(flat, k)


def _cpu_native() -> bool:
    """C++ CPU paths (same .so) are used opportunistically when loadable."""
    return _load_hip() is not None and os.environ.get("DEEPREDUCE_FORCE_TORCH_CPU") != "1"


def bloom_insert(idxs: torch.Tensor, m: int, num_hash: int) -> torch.Tensor:
    if _want_hip(idxs):
        return _hip.bloom_insert(idxs, m, num_hash)
    if _cpu_native():
        return _hip.bloom_insert_cpu(idxs, m, num_hash)
    return _ref.bloom_insert(idxs, m, num_hash)


def bloom_query_positives(packed: torch.Tensor, m: int, num_hash: int, universe: int):
    if _want_hip(packed):
        return _hip.bloom_query_positives(packed, m, num_hash, universe)
    if _cpu_native():
        return _hip.bloom_query_positives_cpu(packed, m, num_hash, universe)
    return _ref.bloom_query_positives(packed, m, num_hash, universe)


def bloom_query_positives_multi(packed2d: torch.Tensor, m: int, num_hash: int, universe: int):
    """R stacked filters (same m/k/universe) -> (positives concatenated
    rank-major, per-rank counts int64[R]).  On GPU the hash computation is
    amortized across ranks (the probe positions are filter-independent)."""
    if _want_hip(packed2d):
        return _hip.bloom_query_positives_multi(packed2d, m, num_hash, universe)
    outs = [
        _ref.bloom_query_positives(packed2d[r].contiguous(), m, num_hash, universe)
        for r in range(packed2d.shape[0])
    ]
    counts = torch.tensor([o.numel() for o in outs], dtype=torch.int64)
    return torch.cat(outs) if outs else torch.empty(0, dtype=torch.int64), counts


def bloom_query_leftmost(packed2d: torch.Tensor, m: int, num_hash: int, universe: int, k: int):
    """First k positives per rank, int64 [R, k], ascending; NO host sync on
    the GPU path.  packed2d may be 1-D (single filter).  Requires k <= the
    number of inserted distinct items (Bloom has no false negatives, so the
    positive count is always >= that)."""
    p2 = packed2d if packed2d.dim() == 2 else packed2d.unsqueeze(0)
    if _want_hip(p2):
        return _hip.bloom_query_leftmost(p2, m, num_hash, universe, k)
    rows = []
    for r in range(p2.shape[0]):
        pos = _ref.bloom_query_positives(p2[r].contiguous(), m, num_hash, universe)[:k]
        if pos.numel() < k:
            pos = torch.nn.functional.pad(pos, (0, k - pos.numel()))
        rows.append(pos)
    return torch.stack(rows)


def bloom_query_members(packed: torch.Tensor, m: int, num_hash: int, items: torch.Tensor):
    if _want_hip(packed):
        return _hip.bloom_query_members(packed, m, num_hash, items)
    if _cpu_native():
        return _hip.bloom_query_members_cpu(packed, m, num_hash, items)
    return _ref.bloom_query_members(packed, m, num_hash, items)


def pack_ints(values: torch.Tensor, nbits: int) -> torch.Tensor:
    if _want_hip(values):
        return _hip.pack_ints(values, nbits)
    if _cpu_native():
        return _hip.pack_ints_cpu(values, nbits)
    return _ref.pack_ints(values, nbits)


def unpack_ints(stream: torch.Tensor, n: int, nbits: int) -> torch.Tensor:
    if _want_hip(stream):
        return _hip.unpack_ints(stream, n, nbits)
    if _cpu_native():
        return _hip.unpack_ints_cpu(stream, n, nbits)
    return _ref.unpack_ints(stream, n, nbits)


def cholesky_solve_small(G: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Batched SPD solve for [S, d<=8, d] systems (polyfit normal equations).
    GPU: one-thread-per-system in-register Cholesky; CPU: torch."""
    if _want_hip(G):
        return _hip.cholesky_solve_small(G, b)
    return torch.linalg.solve(G, b.unsqueeze(-1)).squeeze(-1)


def dexp_fit(y: torch.Tensor, offs: torch.Tensor, lens: torch.Tensor) -> torch.Tensor:
    """Fused DoubleExp cumulative-integral fit: y float32 (sorted ascending
    per tensor), offs/lens int64 [B] -> coeffs float64 [B, 4] = (a, b, c, d).
    GPU: one block per tensor (dexp_fit_kernel); CPU callers use the torch
    fp64 reference in codecs/doubleexp.py directly."""
    if _want_hip(y):
        return _hip.dexp_fit(y, offs, lens)
    raise RuntimeError("dexp_fit is a GPU op; use codecs.doubleexp torch path on CPU")


def qsgd_quantize(vals: torch.Tensor, quantum_num: int, bucket_size: int):
    if _want_hip(vals):
        return _hip.qsgd_quantize(vals, quantum_num, bucket_size)
    return _ref.qsgd_quantize(vals, quantum_num, bucket_size)


def qsgd_dequantize(levels: torch.Tensor, norms: torch.Tensor, quantum_num: int, bucket_size: int):
    if _want_hip(levels):
        return _hip.qsgd_dequantize(levels, norms, quantum_num, bucket_size)
    return _ref.qsgd_dequantize(levels, norms, quantum_num, bucket_size)


pack_bitarray = _ref.pack_bitarray
unpack_bitarray = _ref.unpack_bitarray
