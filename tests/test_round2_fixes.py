"""Round-2 regression tests for the advisor findings and the
find_breaks in-house rewrite.

These run on CPU.  The overlap-reducer batched-pipeline path is CUDA-only in
production (ops/batched.maybe_pipeline requires CUDA tensors), so the residual
-guard regressions are exercised through a fake pipeline object that mimics
`compress_and_own`'s contract.
"""
from __future__ import annotations

import numpy as np
import pytest
import torch
import torch.nn as nn

from deepreduce_amd import deepreduce_from_params
from deepreduce_amd.parallel import OverlappedReducer


def _model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(1200, 50), nn.ReLU(), nn.Linear(50, 4))


class _FakeBP:
    """Mimics ops.batched pipeline: identity 'compression' whose own-decode
    equals the compensated input (wire = raw bytes)."""

    def __init__(self, total_values):
        self.total_values = total_values

    def compress_and_own(self, c_flat):
        wire = c_flat.detach().clone().view(torch.uint8)
        return wire, c_flat.detach().clone()

    def decode_sum(self, gathered):  # pragma: no cover (world=1 in tests)
        return gathered[0].view(torch.float32)


def _patch_pipeline(monkeypatch):
    from deepreduce_amd.ops import batched as _bt

    def fake_maybe_pipeline(grc, comp, named):
        total = sum(t.numel() for _, t in named)
        return _FakeBP(total)

    monkeypatch.setattr(_bt, "maybe_pipeline", fake_maybe_pipeline)


def test_overlap_pipeline_none_memory_does_not_crash(monkeypatch):
    """ADVICE high: with memory='none' the pipeline path used to hit
    grc.memory._flat_r unconditionally -> AttributeError."""
    _patch_pipeline(monkeypatch)
    model = _model()
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "none", "communicator": "allgather",
        "compress_ratio": 0.05,
    })
    reducer = OverlappedReducer(model, grc, num_buckets=2)
    x = torch.randn(8, 1200)
    loss = model(x).sum()
    loss.backward()
    reducer.finalize()  # must not raise
    for p in model.parameters():
        assert p.grad is not None
        assert torch.isfinite(p.grad).all()


def test_overlap_pipeline_single_tensor_bucket_residual(monkeypatch):
    """ADVICE high: a bucket whose compensate_many takes the non-homogeneous
    path (len==1) has no matching flat pool; the residual update must fall
    back to update_many instead of writing another bucket's _flat_r."""
    _patch_pipeline(monkeypatch)
    model = _model()
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.05,
    })
    # one bucket per large tensor -> every bucket has a single tensor
    reducer = OverlappedReducer(model, grc, num_buckets=8)
    x = torch.randn(8, 1200)
    model(x).sum().backward()
    reducer.finalize()
    # FakeBP own-decode == compensated, so every residual must be ~0
    for name, r in grc.memory.residuals.items():
        assert torch.allclose(r, torch.zeros_like(r), atol=1e-6), name


def test_overlap_partial_bucket_resets_and_exchanges():
    """ADVICE medium: a step that produces only part of a bucket's grads
    must not leave a stale arrival count (next step would launch the bucket
    mid-backward)."""
    torch.manual_seed(3)
    # two large params that END UP IN THE SAME BUCKET, but only one gets a
    # grad: route the input through branch `a` only.
    class Branchy(nn.Module):
        def __init__(self):
            super().__init__()
            self.a = nn.Linear(1100, 4, bias=False)
            self.b = nn.Linear(1100, 4, bias=False)

        def forward(self, x, use_b=False):
            return self.b(x) if use_b else self.a(x)

    model = Branchy()
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.05,
    })
    reducer = OverlappedReducer(model, grc, num_buckets=1)
    assert len(reducer._buckets) == 1 and len(reducer._buckets[0]) == 2

    x = torch.randn(8, 1100)
    model(x, use_b=False).sum().backward()
    assert reducer._arrived[0] == 1  # partial
    reducer.finalize()
    assert reducer._arrived[0] == 0  # counter reset
    # the present grad was exchanged: topk 5% own-decode zeroes most entries
    ga = model.a.weight.grad
    assert ga is not None
    sparsity = (ga == 0).float().mean().item()
    assert sparsity > 0.5, "partial bucket's grad was not exchanged"

    # next step, full backward: bucket must launch exactly once, cleanly
    model.zero_grad(set_to_none=False)
    (model(x, use_b=False).sum() + model(x, use_b=True).sum()).backward()
    reducer.finalize()
    assert reducer._arrived[0] == 0
    for p in (model.a.weight, model.b.weight):
        assert torch.isfinite(p.grad).all()


def test_allreduce_fused_residual_matches_per_tensor():
    """ADVICE low: Allreduce.step_many residual update must follow the
    per-tensor semantics (compensated - own decode), not
    (compensated - averaged global)."""
    from deepreduce_amd.factory import grace_from_params

    params = {"compressor": "topk", "memory": "residual",
              "communicator": "allreduce", "compress_ratio": 0.05}
    torch.manual_seed(7)
    tensors = {f"t{i}": torch.randn(2000) for i in range(3)}

    grc_a = grace_from_params(dict(params))
    outs_a = {}
    for n, t in tensors.items():
        outs_a[n] = grc_a.step(t.clone(), n)

    grc_b = grace_from_params(dict(params))
    named = [(n, t.clone()) for n, t in tensors.items()]
    outs_b = dict(zip([n for n, _ in named], grc_b.step_many(named)))

    for n in tensors:
        assert torch.allclose(outs_a[n], outs_b[n], atol=1e-6), n
        ra = grc_a.memory.residuals[n]
        rb = grc_b.memory.residuals[n]
        assert torch.allclose(ra, rb, atol=1e-6), f"residual mismatch: {n}"


def test_allreduce_none_compressor_zero_residual():
    from deepreduce_amd.factory import grace_from_params

    grc = grace_from_params({"compressor": "none", "memory": "residual",
                             "communicator": "allreduce"})
    named = [("w", torch.randn(500)), ("v", torch.randn(600))]
    outs = grc.step_many([(n, t.clone()) for n, t in named])
    for (n, t), o in zip(named, outs):
        assert torch.allclose(o, t, atol=1e-6)
        r = grc.memory.residuals[n]
        assert torch.allclose(r, torch.zeros_like(r), atol=1e-7)


def test_optimizer_snapshot_restore_roundtrip():
    """ADVICE medium (capture double-processing): the snapshot/restore pair
    must return p.grad and the residual state to pre-warmup values."""
    from deepreduce_amd import DistributedOptimizer

    model = _model()
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.05,
    })
    opt = DistributedOptimizer(torch.optim.SGD(model.parameters(), lr=0.1),
                               grc, model)
    x = torch.randn(8, 1200)
    model(x).sum().backward()
    # seed residual state with one real exchange
    from deepreduce_amd.optimizer import reduce_gradients
    reduce_gradients(model, grc)

    model.zero_grad(set_to_none=False)
    model(x * 2).sum().backward()
    grads, gsnap, rsnap = opt._snapshot_state()
    pre_g = [g.clone() for g in grads]
    pre_r = {k: v.clone() for k, v in grc.memory.residuals.items()}

    reduce_gradients(model, grc)  # the "warmup": mutates grads + residuals
    mutated = any(not torch.equal(g, s) for g, s in zip(grads, pre_g))
    assert mutated, "warmup did not mutate grads (test vacuous)"

    opt._restore_state(grads, gsnap, rsnap)
    for g, s in zip(grads, pre_g):
        assert torch.equal(g, s)
    for k, v in pre_r.items():
        assert torch.allclose(grc.memory.residuals[k], v, atol=0), k


# ---- find_breaks rewrite ---------------------------------

def _spec_find_breaks(curve, num_of_breaks=10):
    """Independent straight-line spec of the published algorithm: recursive
    suffix processing with an explicit linspace chord (used only as a test
    oracle)."""
    out = []
    lo = 0
    n = len(curve)
    for _ in range(num_of_breaks):
        suffix = curve[lo:]
        if len(suffix) < 20 * num_of_breaks:
            break
        chord = np.linspace(suffix[0], suffix[-1], len(suffix))
        j = int(np.argmax(np.abs(chord - suffix)))
        k = lo + j
        if n - k < 20 * num_of_breaks:
            break
        out.append(k)
        lo = k
    return out


@pytest.mark.parametrize("seed", [0, 1, 2, 3, 4])
def test_find_breaks_matches_spec(seed):
    from deepreduce_amd.codecs.polyfit_cpu import find_breaks

    rng = np.random.default_rng(seed)
    # sorted ascending curve, exponential-ish like sorted topk magnitudes
    n = int(rng.integers(300, 5000))
    curve = np.sort(rng.standard_normal(n) * np.exp(rng.uniform(0, 3, n)))
    got = find_breaks(curve, num_of_breaks=5)
    want = _spec_find_breaks(curve, num_of_breaks=5)
    assert got == want
    # structural properties
    assert got == sorted(got)
    assert all(0 < k < n for k in got)
    assert len(set(got)) == len(got)


def test_find_breaks_golden():
    """Pin concrete knot choices so future edits can't silently change
    behavior."""
    from deepreduce_amd.codecs.polyfit_cpu import find_breaks

    x = np.arange(1000, dtype=np.float64)
    curve = np.where(x < 700, 0.01 * x,
                     0.01 * 700 + np.clip(x - 700, 0, None) ** 1.5)
    got = find_breaks(curve, num_of_breaks=4)
    assert got == _spec_find_breaks(curve, num_of_breaks=4)
    assert got, "kink curve must yield at least one knot"
    assert 650 <= got[0] <= 900  # knot lands at/after the kink region
    # short curves: guarded, no knots
    assert find_breaks(np.arange(50, dtype=np.float64), 10) == []


# ---- block-PFoR ------------------------------------------

class TestBlockPFor:
    def test_roundtrip_shapes_and_outliers(self):
        from deepreduce_amd.codecs.intpack import pfor_decode, pfor_encode

        torch.manual_seed(0)
        for n in [0, 1, 127, 128, 129, 1000]:
            v = torch.randint(0, 1 << 24, (n,), dtype=torch.int64)
            assert torch.equal(pfor_decode(pfor_encode(v)), v)
        # outlier patching: one huge value must not inflate the block width
        g = torch.randint(0, 64, (4096,), dtype=torch.int64)
        g[100] = 1 << 27
        g[3000] = (1 << 31) - 1
        w = pfor_encode(g)
        assert torch.equal(pfor_decode(w), g)
        bits_per_int = w.numel() * 8 / g.numel()
        assert bits_per_int < 9, f"outliers not patched: {bits_per_int:.1f} b/int"

    def test_beats_fixed_width_on_outlier_gaps(self):
        from deepreduce_amd.codecs.intpack import pack_with_header, pfor_encode

        torch.manual_seed(1)
        g = torch.randint(0, 100, (5000,), dtype=torch.int64)
        g[torch.randint(0, 5000, (40,))] = torch.randint(
            1 << 20, 1 << 28, (40,), dtype=torch.int64)
        assert pfor_encode(g).numel() * 3 < pack_with_header(g).numel(), \
            "block-PFoR should be >=3x smaller than global fixed width here"

    def test_codec_reports_bits_per_int(self):
        from deepreduce_amd.codecs import compressor as registry

        torch.manual_seed(2)
        t = torch.randn(100000)
        k = 1000
        _, idx = t.abs().topk(k)
        vals = t[idx]
        params = {}
        v, w, s = registry["pfor"].compress((vals, idx, t.size()), params)
        assert 0 < params["_pfor_bits_per_int"] < 32
        v2, i2, _ = registry["pfor"].decompress((v, w, t.size()), params)
        assert torch.equal(i2.sort().values, idx.sort().values)
        got = torch.zeros_like(t).scatter_(0, i2, v2)
        want = torch.zeros_like(t).scatter_(0, idx, vals)
        assert torch.allclose(got, want)


def test_tensor_wire_bytes_measured_not_estimated():
    """VERDICT weak item 7: after one fused exchange, the stats logger
    must see the MEASURED per-tensor payload size for non-pipelined codecs
    (rle/gzip/huffman), not the 8k fp32+int32 estimate."""
    from deepreduce_amd.factory import grace_from_params

    grc = grace_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.02,
        "deepreduce": "index", "index": "rle",
    })
    torch.manual_seed(1)
    named = [("w", torch.randn(50_000)), ("v", torch.randn(20_000))]
    grc.step_many([(n, t.clone()) for n, t in named])
    total = grc.last_wire_bytes
    per_tensor = sum(grc._tensor_wire_bytes(n, t) for n, t in named)
    assert per_tensor == total, (per_tensor, total)


def test_auto_fpr():
    """fpr='auto' picks the closed-form wire-minimizing FPR and beats both
    the default and a too-large FPR on total wire bytes."""
    from deepreduce_amd.codecs import compressor as registry
    from deepreduce_amd.codecs.bloom import auto_fpr, get_bf_config

    d, k = 1_000_000, 10_000
    f = auto_fpr(k, d)
    assert 1e-5 <= f <= 0.5

    def wire_bytes(fpr):
        nh, m = get_bf_config(k, fpr)
        return m / 8 + (d - k) * fpr * 4  # filter + FP-carried values

    assert wire_bytes(f) <= wire_bytes(0.1 * k / d) + 1
    assert wire_bytes(f) <= wire_bytes(0.2) + 1

    # end-to-end: codec accepts the string and produces a valid roundtrip
    torch.manual_seed(31)
    t = torch.randn(d)
    _, idx = t.abs().topk(k)
    params = {"fpr": "auto", "policy": "leftmost", "dense_tensor": t}
    v, bits, shape = registry["bloom"].compress((t[idx], idx, t.size()), params)
    params.pop("dense_tensor")
    params.pop("_own_decoded", None)
    v2, i2, _ = registry["bloom"].decompress((v, bits, shape), params)
    assert v2.numel() == i2.numel() == k
    assert torch.allclose(t[i2], v2)

    from deepreduce_amd.params import validate
    validate({"fpr": "auto"})
    with pytest.raises(ValueError):
        validate({"fpr": "wrong"})


def test_generic_decompress_batch_chunking_cpu():
    """IndexCompressor.decompress_batch chunks >16 payloads (the MAXR
    kernel cap) on the CPU native path too."""
    from deepreduce_amd import deepreduce_from_params

    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "none",
        "communicator": "allgather", "compress_ratio": 0.02,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    })
    comp = grc.compressor
    torch.manual_seed(41)
    N = 20_000
    payloads, ref = [], torch.zeros(N)
    ctx = None
    for r in range(18):
        t = torch.randn(N)
        tc, ctx = comp.compress(t, f"r{r}")
        payloads.append(tc)
        ref += comp.decompress(tc, ctx).reshape(-1)
    got = comp.decompress_batch(payloads, ctx)
    assert got is not None, "fast path must engage for 18 ranks"
    assert torch.allclose(got.reshape(-1), ref, atol=1e-5), \
        (got.reshape(-1) - ref).abs().max()
