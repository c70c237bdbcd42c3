"""Wrapper layer: value/index/'both' composition, small-tensor bypass,
mapping round trip (the subtle call stack of SURVEY.md sect. 3.4)."""
import pytest
import torch

from deepreduce_amd import TopKCompressor
from deepreduce_amd.wrappers import DeepReduce, IndexCompressor, ValueCompressor


@pytest.fixture
def grad():
    torch.manual_seed(11)
    return torch.randn(300, 200)  # 60k elements


def _recovered_quality(dense_in, dense_out, idxs_true):
    """fraction of true top-k positions whose recovered value is close."""
    flat_in, flat_out = dense_in.reshape(-1), dense_out.reshape(-1)
    close = torch.isclose(flat_out[idxs_true], flat_in[idxs_true], atol=0.2, rtol=0.2)
    return close.float().mean().item()


def test_value_compressor_polyfit(grad):
    sp = TopKCompressor(0.01)
    wc = ValueCompressor(sp, {"value": "polyfit"})
    payload, ctx = wc.compress(grad, "w")
    assert payload[0].dtype == torch.float64  # coefficients travel
    out = wc.decompress(payload, ctx)
    assert out.shape == grad.shape
    (v, i), _ = sp.compress(grad, "w")
    # values land at the right positions with bounded error
    err = (out.reshape(-1)[i] - v).abs().mean() / v.abs().mean()
    assert err < 0.15


def test_value_compressor_qsgd(grad):
    sp = TopKCompressor(0.01)
    wc = ValueCompressor(sp, {"value": "qsgd"})
    payload, ctx = wc.compress(grad, "w")
    out = wc.decompress(payload, ctx)
    (v, i), _ = sp.compress(grad, "w")
    # QSGD per-element bound: bucket_norm / quantum_num
    bound = v.norm() / 127 + 1e-5  # whole-tensor norm >= any bucket norm
    assert ((out.reshape(-1)[i] - v).abs() <= bound).all()


def test_index_compressor_bloom(grad):
    sp = TopKCompressor(0.01)
    wc = IndexCompressor(sp, {"index": "bloom", "policy": "leftmost"})
    payload, ctx = wc.compress(grad, "w")
    (vals, bits) = payload
    assert bits.dtype == torch.uint8
    out = wc.decompress(payload, ctx)
    assert out.shape == grad.shape
    # FP-aware: every nonzero of the output matches the input gradient
    nz = out.reshape(-1).nonzero().reshape(-1)
    assert torch.allclose(out.reshape(-1)[nz], grad.reshape(-1)[nz])


def test_index_compressor_rle_exact(grad):
    sp = TopKCompressor(0.01)
    wc = IndexCompressor(sp, {"index": "rle"})
    payload, ctx = wc.compress(grad, "w")
    out = wc.decompress(payload, ctx)
    # exact index codec -> decompressed == plain topk decompress
    (v, i), shape = sp.compress(grad, "w")
    expected = sp.decompress((v, i), shape)
    assert torch.allclose(out, expected)


def test_both_mode_mapping_roundtrip(grad):
    sp = TopKCompressor(0.01)
    wc = DeepReduce(sp, {"value": "polyfit", "index": "bloom", "policy": "leftmost"})
    payload, ctx = wc.compress(grad, "w")
    vals, bits, mapping = payload
    assert mapping.dtype == torch.uint8  # bit-packed (paper App. E)
    k = grad.numel() // 100
    assert mapping.numel() < k * 4  # beats the reference's int32 mapping
    assert vals.dtype == torch.float64
    out = wc.decompress(payload, ctx)
    assert out.shape == grad.shape
    (v, i), _ = sp.compress(grad, "w")
    assert _recovered_quality(grad, out, i) > 0.7


def test_small_tensor_bypass():
    sp = TopKCompressor(0.5)
    wc = DeepReduce(sp, {"value": "polyfit", "index": "bloom"})
    t = torch.randn(30, 30)  # 900 <= 1000 -> bypass
    payload, ctx = wc.compress(t, "b")
    assert len(payload) == 2  # raw (vals, idxs), no mapping
    out = wc.decompress(payload, ctx)
    (v, i), shape = sp.compress(t, "b")
    assert torch.allclose(out, sp.decompress((v, i), shape))


def test_micro_benchmark_prints(grad, capsys):
    sp = TopKCompressor(0.01)
    wc = ValueCompressor(sp, {"value": "qsgd", "micro-benchmark": True})
    payload, ctx = wc.compress(grad, "w")
    wc.decompress(payload, ctx)
    out = capsys.readouterr().out
    assert "val_compression time" in out
    assert "val_relative_volume" in out


def test_decompress_own_matches_decompress(grad):
    """the residual-update fast path must equal a full decompress."""
    sp = TopKCompressor(0.01)
    wc = IndexCompressor(sp, {"index": "bloom", "policy": "leftmost"})
    payload, ctx = wc.compress(grad, "w")
    own = wc.decompress_own(payload, ctx, "w")  # consumes cache
    full = wc.decompress(payload, ctx)
    assert torch.allclose(own, full)

    wb = DeepReduce(sp, {"value": "polyfit", "index": "bloom"})
    payload, ctx = wb.compress(grad, "w")
    own = wb.decompress_own(payload, ctx, "w")
    full = wb.decompress(payload, ctx)
    assert torch.allclose(own, full)


def test_residual_uses_own_decode(grad):
    from deepreduce_amd import ResidualMemory

    sp = TopKCompressor(0.01)
    wc = IndexCompressor(sp, {"index": "bloom", "policy": "leftmost"})
    mem = ResidualMemory()
    t = mem.compensate(grad, "w")
    payload, ctx = wc.compress(t, "w")
    mem.update(t, "w", wc, payload, ctx)
    # residual + own-decompressed == compensated
    assert torch.allclose(mem.residuals["w"] + wc.decompress(payload, ctx), t, atol=1e-6)
