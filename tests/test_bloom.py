"""Bloom codec: no-false-negative property, FPR vs theory, policy
determinism across 'ranks', FP-aware value re-read."""
import math

import pytest
import torch

from deepreduce_amd.codecs import compressor
from deepreduce_amd.codecs.bloom import get_bf_config
from deepreduce_amd.ops import (
    bloom_insert,
    bloom_query_members,
    bloom_query_positives,
    topk_select,
)


@pytest.fixture
def data():
    torch.manual_seed(3)
    d = 100_000
    t = torch.randn(d)
    vals, idxs = topk_select(t, 1000)
    return t, vals, idxs


def test_no_false_negatives(data):
    t, vals, idxs = data
    nh, m = get_bf_config(1000, 0.001)
    packed = bloom_insert(idxs, m, nh)
    assert bloom_query_members(packed, m, nh, idxs).all()


def test_fpr_close_to_theory(data):
    t, vals, idxs = data
    d = t.numel()
    fpr = 0.001
    nh, m = get_bf_config(1000, fpr)
    packed = bloom_insert(idxs, m, nh)
    pos = bloom_query_positives(packed, m, nh, d)
    true = set(idxs.tolist())
    fp = len(set(pos.tolist()) - true)
    measured = fp / (d - len(true))
    assert measured < fpr * 3  # generous: binomial noise
    assert set(pos.tolist()) >= true  # superset (no false negatives)


def test_positives_sorted_ascending(data):
    t, vals, idxs = data
    nh, m = get_bf_config(1000, 0.01)
    packed = bloom_insert(idxs, m, nh)
    pos = bloom_query_positives(packed, m, nh, t.numel())
    assert torch.equal(pos, pos.sort().values)


def test_config_matches_reference_formula():
    # pytorch/deepreduce.py:495-500
    nh, m = get_bf_config(1000, 0.001)
    assert nh == math.ceil(math.log2(1000))
    assert m == math.ceil(math.log2(1000) * 1000 / 0.693147180)


@pytest.mark.parametrize("policy", ["leftmost", "random", "p0", "conflict_sets"])
def test_compress_decompress_determinism(data, policy):
    """decompress must re-derive identical indices on every 'rank'."""
    t, vals, idxs = data
    params = {"policy": policy, "dense_tensor": t}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    params2 = {"policy": policy}  # decompress side has no dense tensor
    out1 = compressor["bloom"].decompress((v.clone(), bits.clone(), shape), params2)
    out2 = compressor["bloom"].decompress((v.clone(), bits.clone(), shape), params2)
    assert torch.equal(out1[1], out2[1])
    assert torch.equal(out1[0], out2[0])


@pytest.mark.parametrize("policy", ["leftmost", "p0"])
def test_fp_aware_values_exact(data, policy):
    """with FP-aware re-read, every (val, idx) pair matches the dense tensor."""
    t, vals, idxs = data
    params = {"policy": policy, "dense_tensor": t}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    v2, i2, _ = compressor["bloom"].decompress((v, bits, shape), {"policy": policy})
    assert torch.allclose(t[i2], v2)


def test_recall_with_fp_aware(data):
    t, vals, idxs = data
    params = {"policy": "leftmost", "dense_tensor": t, "fpr": 0.001}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    v2, i2, _ = compressor["bloom"].decompress((v, bits, shape), {"policy": "leftmost", "fpr": 0.001})
    true = set(idxs.tolist())
    rec = set(i2.tolist())
    # leftmost drops ~#false-positives of the rightmost true indices:
    # expected recall ~ 1 - fpr*d/k = 0.9 here (the paper's motivation for P0)
    assert len(true & rec) / len(true) > 0.85


def test_p0_returns_all_positives(data):
    t, vals, idxs = data
    params = {"policy": "p0", "dense_tensor": t}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    assert v.numel() >= idxs.numel() + 1  # count + all positives
    v2, i2, _ = compressor["bloom"].decompress((v, bits, shape), {"policy": "p0"})
    assert set(i2.tolist()) >= set(idxs.tolist())
    assert v2.numel() == i2.numel()


def test_conflict_sets_covers_k(data):
    t, vals, idxs = data
    params = {"policy": "conflict_sets", "dense_tensor": t}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    v2, i2, _ = compressor["bloom"].decompress((v, bits, shape), {"policy": "conflict_sets"})
    assert i2.numel() == idxs.numel()
    assert torch.allclose(t[i2], v2)


def test_wire_volume_beats_raw_indices(data):
    t, vals, idxs = data
    params = {"policy": "leftmost", "fpr": 0.01}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    assert bits.numel() < idxs.numel() * 4  # paper: ~50% of int32 keys


def test_decompress_batch_matches_loop(data):
    """the fused multi-rank decompress must equal the per-rank loop sum."""
    from deepreduce_amd import TopKCompressor
    from deepreduce_amd.wrappers import IndexCompressor

    t, vals, idxs = data
    sp = TopKCompressor(0.01)
    wc = IndexCompressor(sp, {"index": "bloom", "policy": "leftmost"})
    # three 'ranks' with different gradients
    payloads, ctx = [], None
    for r in range(3):
        torch.manual_seed(500 + r)
        g = torch.randn(t.numel())
        payload, ctx = wc.compress(g, f"w")
        payloads.append(payload)
    fused = wc.decompress_batch(payloads, ctx)
    assert fused is not None
    loop = sum(wc.decompress(p, ctx) for p in payloads)
    assert torch.allclose(fused, loop)


def test_query_multi_cpu_fallback(data):
    from deepreduce_amd.ops import bloom_insert, bloom_query_positives, bloom_query_positives_multi

    t, vals, idxs = data
    nh, m = 8, 200_003
    b1 = bloom_insert(idxs, m, nh)
    b2 = bloom_insert(idxs + 1, m, nh)
    stacked = torch.stack([b1, b2])
    pos, counts = bloom_query_positives_multi(stacked, m, nh, t.numel())
    p1 = bloom_query_positives(b1, m, nh, t.numel())
    p2 = bloom_query_positives(b2, m, nh, t.numel())
    assert counts.tolist() == [p1.numel(), p2.numel()]
    assert torch.equal(pos[: p1.numel()], p1)
    assert torch.equal(pos[p1.numel() :], p2)


def test_fp16_wire_values_roundtrip():
    """wire_dtype=fp16 halves the value bytes; decode returns float32 with
    half-precision error, and the own-decode cache matches the wire decode
    (so the residual absorbs the quantization)."""
    import torch

    from deepreduce_amd.codecs import compressor

    torch.manual_seed(0)
    t = torch.randn(50_000)
    k = 500
    vals, idxs = t.abs().topk(k)
    vals = t[idxs]
    params = {"policy": "leftmost", "dense_tensor": t, "wire_dtype": "fp16"}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    assert v.dtype == torch.float16
    own = params.pop("_own_decoded")
    v2, i2, _ = compressor["bloom"].decompress((v, bits, shape),
                                               {"policy": "leftmost",
                                                "wire_dtype": "fp16"})
    assert v2.dtype == torch.float32
    assert torch.equal(own[1], i2)
    assert torch.equal(own[0], v2)  # cache == wire decode (residual exact)
    assert torch.allclose(v2, t[i2], rtol=1e-3, atol=1e-4)


def test_conflict_sets_deterministic_across_ranks():
    """conflict_sets policy must re-derive the identical selection on every
    rank from the wire alone (policies.hpp:136-146 determinism contract)."""
    import torch

    from deepreduce_amd.codecs import compressor

    torch.manual_seed(4)
    t = torch.randn(20_000)
    k = 200
    _, idxs = t.abs().topk(k)
    params = {"policy": "conflict_sets", "policy_seed": 123, "dense_tensor": t}
    v, bits, shape = compressor["bloom"].compress((t[idxs], idxs, t.size()), params)
    params.pop("dense_tensor")
    params.pop("_own_decoded", None)
    outs = [compressor["bloom"].decompress((v.clone(), bits.clone(), shape),
                                           dict(params)) for _ in range(3)]
    for _, i2, _ in outs[1:]:
        assert torch.equal(i2, outs[0][1])
    assert outs[0][1].numel() == k


def test_measured_fpr_ground_truth():
    """measured_fpr == brute-force false-positive rate of the filter."""
    import torch

    from deepreduce_amd.codecs.bloom import Bloom
    from deepreduce_amd.metrics import measured_fpr, policy_errors
    from deepreduce_amd.ops import bloom_query_members

    torch.manual_seed(5)
    universe = 30_000
    idxs = torch.randperm(universe)[:300]
    params = {"policy": "leftmost", "fpr": 0.01}
    v, bits, shape = Bloom.compress((torch.randn(300), idxs,
                                     torch.Size([universe])), params)
    from deepreduce_amd.codecs.bloom import get_bf_config

    num_hash, m = get_bf_config(300, 0.01)
    got = measured_fpr(bits, m, num_hash, universe, idxs)
    members = bloom_query_members(bits, m, num_hash, torch.arange(universe))
    true = torch.zeros(universe, dtype=torch.bool)
    true[idxs] = True
    brute = float((members & ~true).sum()) / float((~true).sum())
    assert abs(got - brute) < 1e-9
    # policy errors: leftmost selection vs true set
    _, rec, _ = Bloom.decompress((v, bits, shape), params)
    errs = policy_errors(rec, idxs)
    assert errs == len(set(rec.tolist()) - set(idxs.tolist()))
