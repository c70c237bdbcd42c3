"""PolySeg value codec — piecewise polynomial fit over FIXED segments.

Reference behavior: tensorflow/deepreduce.py:446-557 (PolySegCompressor)
used hard-coded per-model break tables (:182-219) so every rank produces
the same payload size and the allgather sees one uniform tensor (:511-513).

MI355X-native redesign: segments are derived from N alone (geometric split
of the descending curve into `num_segments` pieces, heavier at the head
where the curve bends), so payload size depends only on k — uniform across
ranks, `tensors_size_are_same` stays True — with no per-model tables.
Shares the batched on-device fit/eval machinery with PolyFit.
"""
from __future__ import annotations


from . import SparseCompressor
from .polyfit import _eval_segments, _fit_segments


def fixed_segments(N: int, num_segments: int = 10):
    """Deterministic geometric segmentation from N alone.

    Head segments (largest values, steepest curve) get geometrically fewer
    elements: ratios 2^-s normalized.  Every rank with the same k derives
    the same list.
    """
    num_segments = max(1, min(num_segments, N))
    weights = [2.0 ** (i / 2) for i in range(num_segments)]
    total = sum(weights)
    segs = [max(1, int(N * w / total)) for w in weights]
    segs[-1] += N - sum(segs)
    if segs[-1] < 1:  # pathological tiny N
        return [N]
    return segs


class PolySeg(SparseCompressor):
    order_preserving = False

    @staticmethod
    def compress(sparse_tensor, params):
        degree = int(params.get("poly_degree", 5))
        nseg = int(params.get("num_segments", 10))
        vals, idxs, shape = sparse_tensor
        N = int(idxs.numel())
        y, mapping = vals.float().sort(descending=True)
        idxs = idxs[mapping]
        segments = fixed_segments(N, nseg)
        coeffs = _fit_segments(y, segments, degree)
        return coeffs.reshape(-1), idxs, shape

    @staticmethod
    def decompress(fitted, params):
        payload, idxs, shape = fitted
        degree = int(params.get("poly_degree", 5))
        nseg = int(params.get("num_segments", 10))
        N = int(idxs.numel())
        segments = fixed_segments(N, nseg)
        coeffs = payload.reshape(len(segments), degree + 1)
        vals = _eval_segments(coeffs, segments, payload.device).float()
        return vals, idxs, shape
