"""PolyFit CPU variant — chord-distance knot search + numpy polyfit.

Reference behavior: /root/reference/pytorch/deepreduce.py:566-688.  Values
sorted descending; breaks found recursively at the point of max distance to
the chord (paper Lemma 1), positives searched in reversed (ascending) order;
per-segment numpy polynomial fits in float64.

Wire: the reference returned a NESTED tuple (coeffs, breaks) which cannot
travel as one tensor (pytorch/deepreduce.py:672 'todo: encode ... into one
tensor') — here the payload is one float64 tensor:
    [n_breaks+2, break_0..break_last, coeff...] (breaks include 0 and N).
"""
from __future__ import annotations

import numpy as np
import torch

from . import SparseCompressor


def find_breaks(curve: np.ndarray, num_of_breaks: int = 10):
    """Greedy chord-split knot selection over successive suffixes.

    Behavioral contract (same algorithm as the reference layer,
    pytorch/deepreduce.py:566-582 — paper Lemma 1 — expressed in this
    codebase's idiom): walk the sorted curve left to right.  At each step,
    draw the straight segment joining the current window's first and last
    points (NOTE: the reference anchors the chord's right end at the
    *global* last sample via y[-1] of the suffix — the suffix always ends
    at the curve's end, so these coincide), place a knot where the curve
    deviates most from that segment, then continue on the tail to the
    right of the knot.  A window shorter than 20*num_of_breaks samples
    never receives a knot (min-segment guard), and the walk also stops
    once the remaining tail falls under the same guard.  Knots are
    absolute indices into `curve`, strictly increasing.
    """
    n = len(curve)
    guard = 20 * num_of_breaks
    knots: list[int] = []
    start = 0
    for _ in range(num_of_breaks):
        window = n - start
        if window < guard:
            break
        left, right = curve[start], curve[n - 1]
        # chord sampled at the window's integer positions
        chord = left + (right - left) * (np.arange(window) / max(window - 1, 1))
        k = start + int(np.argmax(np.abs(chord - curve[start:])))
        if n - k < guard:
            break
        knots.append(k)
        start = k
    return knots


def _fit(curve: np.ndarray, breaks, degree: int):
    bounds = [0] + list(breaks) + [len(curve)]
    x = np.arange(len(curve), dtype=np.float64)
    coeffs = []
    for i in range(1, len(bounds)):
        lo, hi = bounds[i - 1], bounds[i]
        if hi <= lo:
            coeffs.append(np.zeros(degree + 1))
            continue
        z = np.polynomial.polynomial.polyfit(x[lo:hi], curve[lo:hi], degree)
        if len(z) < degree + 1:
            z = np.pad(z, (0, degree + 1 - len(z)))
        coeffs.append(z)
    return np.concatenate(coeffs), bounds


def _restore(coeffs: np.ndarray, bounds, degree: int):
    n_seg = len(bounds) - 1
    coeffs = coeffs.reshape(n_seg, degree + 1)
    N = bounds[-1]
    x = np.arange(N, dtype=np.float64)
    out = np.empty(N, dtype=np.float64)
    for i in range(n_seg):
        lo, hi = bounds[i], bounds[i + 1]
        out[lo:hi] = np.polynomial.polynomial.polyval(x[lo:hi], coeffs[i])
    return out


class PolyFitCPU(SparseCompressor):
    order_preserving = False

    @staticmethod
    def compress(sparse_tensor, params):
        vals, idxs, shape = sparse_tensor
        degree = int(params.get("poly_degree", 5))
        num_of_breaks = int(params.get("num_breaks", 5))

        vals_sorted, mapping = torch.sort(vals.float(), descending=True)
        idxs_sorted = idxs[mapping]
        y = vals_sorted.cpu().numpy().astype(np.float64)
        num_pos = int(np.sum(y > 0))

        if num_pos == 0:
            breaks = find_breaks(y, num_of_breaks)
        elif num_pos == len(y):
            rev = y[::-1]
            b = find_breaks(rev, num_of_breaks)
            breaks = [len(y) - x for x in b[::-1]]
        else:
            pos_rev = y[:num_pos][::-1]
            b = find_breaks(pos_rev, num_of_breaks)
            breaks_pos = [num_pos - x for x in b[::-1]]
            b_neg = find_breaks(y[num_pos:], num_of_breaks)
            breaks_neg = [num_pos + x for x in b_neg]
            breaks = breaks_pos + [num_pos] + breaks_neg
        breaks = sorted(set(b for b in breaks if 0 < b < len(y)))

        coeffs, bounds = _fit(y, breaks, degree)
        payload = np.concatenate([[len(bounds)], bounds, coeffs]).astype(np.float64)
        wire = torch.from_numpy(payload).to(idxs.device)
        return wire, idxs_sorted, shape

    @staticmethod
    def decompress(sparse_tensor, params):
        wire, idxs, shape = sparse_tensor
        degree = int(params.get("poly_degree", 5))
        payload = wire.cpu().numpy()
        nb = int(payload[0])
        bounds = payload[1 : 1 + nb].astype(np.int64).tolist()
        coeffs = payload[1 + nb :]
        vals = _restore(coeffs, bounds, degree)
        return torch.from_numpy(vals.astype(np.float32)).to(idxs.device), idxs, shape
