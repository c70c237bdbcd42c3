"""Double-exponential value codec (Fit-DExp).

Reference behavior: /root/reference/tensorflow/deepreduce.py:377-442 — sort
values ascending, fold the sign into the index stream as (idx+1)*sign, fit
|y| with y = a*e^(bx) + c*e^(dx) via the cumulative-integral linearization
(tensorflow/deepreduce.py:67-144), and transmit only the 4 coefficients.

Method (Jacquelin): with S = cumint(y), SS = cumint(S), regress
    y ~ A*SS + B*S + C*x + D
then b, d are the roots of t^2 - B t - A = 0, and a, c come from the 2x2
least-squares of y on (e^{bx}, e^{dx}).  All in float64 on-device.

Wire: vals' = float64[4] (a,b,c,d); idxs' = signed mapping (idx+1)*sign in
value-sorted order.  Not order-preserving.
"""
from __future__ import annotations

import torch

from . import SparseCompressor


def _double_exp_fit(y: torch.Tensor):
    """y: float64[N] -> (a, b, c, d) scalars."""
    N = y.numel()
    device = y.device
    x = torch.arange(1, N + 1, dtype=torch.float64, device=device)
    dx = torch.ones_like(y)
    # trapezoid cumulative integrals, S[0] = SS[0] = 0
    S = torch.zeros_like(y)
    S[1:] = torch.cumsum((y[1:] + y[:-1]) * 0.5 * dx[1:], 0)
    SS = torch.zeros_like(y)
    SS[1:] = torch.cumsum((S[1:] + S[:-1]) * 0.5 * dx[1:], 0)

    ones = torch.ones_like(y)
    M = torch.stack([SS, S, x, ones], dim=1)  # [N,4]
    G = M.T @ M
    rhs = M.T @ y
    G = G + torch.eye(4, dtype=torch.float64, device=device) * (G.diagonal().abs().max() * 1e-12 + 1e-30)
    theta = torch.linalg.solve(G, rhs)
    A, B = theta[0], theta[1]
    disc = torch.clamp(B * B + 4 * A, min=0.0)
    r = torch.sqrt(disc)
    b = 0.5 * (B + r)
    d = 0.5 * (B - r)
    # guard against exp overflow at x = N
    cap = 650.0 / max(float(N), 1.0)
    b = torch.clamp(b, min=-cap, max=cap)
    d = torch.clamp(d, min=-cap, max=cap)
    eb = torch.exp(b * x)
    ed = torch.exp(d * x)
    E = torch.stack([eb, ed], dim=1)
    G2 = E.T @ E + torch.eye(2, dtype=torch.float64, device=device) * 1e-12
    ac = torch.linalg.solve(G2, E.T @ y)
    return ac[0], b, ac[1], d


class DoubleExp(SparseCompressor):
    order_preserving = False

    @staticmethod
    def compress(sparse_tensor, params):
        vals, idxs, shape = sparse_tensor
        y_abs = vals.double().abs()
        # stable: deterministic tie order, identical on CPU and GPU
        order = torch.argsort(y_abs, stable=True)  # ascending |value|
        y_sorted = y_abs[order]
        sign = torch.sign(vals.double()[order])
        sign = torch.where(sign == 0, torch.ones_like(sign), sign)
        signed_map = ((idxs[order].double() + 1.0) * sign).long()

        if vals.is_cuda:
            # fused device fit: one kernel replaces the
            # ~20-op fp64 chain incl. two hipSolver solves
            from .. import ops

            dev = vals.device
            payload = ops.dexp_fit(
                y_sorted.float(),
                torch.zeros(1, dtype=torch.int64, device=dev),
                torch.tensor([y_sorted.numel()], dtype=torch.int64, device=dev),
            ).reshape(-1)
        else:
            a, b, c, d = _double_exp_fit(y_sorted)
            payload = torch.stack([a, b, c, d])
        return payload, signed_map, shape

    @staticmethod
    def decompress(sparse_tensor, params):
        payload, signed_map, shape = sparse_tensor
        N = signed_map.numel()
        device = payload.device
        a, b, c, d = payload[0], payload[1], payload[2], payload[3]
        x = torch.arange(1, N + 1, dtype=torch.float64, device=device)
        y = a * torch.exp(b * x) + c * torch.exp(d * x)
        sign = torch.sign(signed_map.double())
        vals = (y * sign).float()
        idxs = signed_map.abs().long() - 1
        return vals, idxs, shape
