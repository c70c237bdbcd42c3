"""Codec unit tests: round-trip exactness for lossless codecs, error bounds
and determinism for lossy ones (the test strategy SURVEY.md sect. 4 calls
for — the reference itself ships no tests)."""
import pytest
import torch

from deepreduce_amd.codecs import compressor
from deepreduce_amd.ops import topk_select


@pytest.fixture
def sparse():
    torch.manual_seed(7)
    d = 50_000
    t = torch.randn(d)
    vals, idxs = topk_select(t, 500)
    return t, vals, idxs


def _sorted_pair(vals, idxs):
    s, perm = idxs.sort()
    return vals[perm], s


class TestLossless:
    def test_rle_roundtrip(self, sparse):
        t, vals, idxs = sparse
        v, w, _ = compressor["rle"].compress((vals, idxs, t.size()), {})
        v2, i2, _ = compressor["rle"].decompress((v, w, t.size()), {})
        ev, ei = _sorted_pair(vals, idxs)
        assert torch.equal(i2, ei)
        assert torch.equal(v2, ev)

    def test_rle_volume(self, sparse):
        t, vals, idxs = sparse
        _, w, _ = compressor["rle"].compress((vals, idxs, t.size()), {})
        assert w.numel() < idxs.numel() * 4  # beats raw int32 indices

    def test_pfor_roundtrip(self, sparse):
        t, vals, idxs = sparse
        v, w, _ = compressor["pfor"].compress((vals, idxs, t.size()), {})
        v2, i2, _ = compressor["pfor"].decompress((v, w, t.size()), {})
        ev, ei = _sorted_pair(vals, idxs)
        assert torch.equal(i2, ei)
        assert torch.equal(v2, ev)
        assert w.numel() < idxs.numel() * 4

    def test_gzip_roundtrip(self, sparse):
        t, vals, idxs = sparse
        v, i, _ = compressor["gzip"].compress((vals, idxs, t.size()), {})
        assert v.dtype == torch.uint8
        v2, _, _ = compressor["gzip"].decompress((v, i, t.size()), {})
        assert torch.equal(v2, vals)

    def test_huffman_roundtrip(self, sparse):
        t, vals, idxs = sparse
        v, w, _ = compressor["huffman"].compress((vals, idxs, t.size()), {})
        v2, i2, _ = compressor["huffman"].decompress((v, w, t.size()), {})
        assert torch.equal(i2, idxs.long())
        assert w.numel() < idxs.numel() * 4

    def test_huffman_empty_and_small(self):
        t = torch.randn(2000)
        vals, idxs = topk_select(t, 3)
        v, w, _ = compressor["huffman"].compress((vals, idxs, t.size()), {})
        _, i2, _ = compressor["huffman"].decompress((v, w, t.size()), {})
        assert torch.equal(i2, idxs.long())


class TestLossy:
    def test_qsgd_error_bound(self, sparse):
        t, vals, idxs = sparse
        params = {"quantum_num": 127, "bucket_size": 512}
        v, i, _ = compressor["qsgd"].compress((vals, idxs, t.size()), params)
        assert v.dtype == torch.int8
        v2, i2, _ = compressor["qsgd"].decompress((v, i, t.size()), params)
        assert v2.numel() == vals.numel()
        assert torch.equal(i2, idxs)
        # per-bucket error bound: |err| <= norm/quantum per element
        for b in range(0, vals.numel(), 512):
            chunk = vals[b : b + 512]
            err = (v2[b : b + 512] - chunk).abs().max()
            assert err <= chunk.norm() / 127 + 1e-6

    def test_qsgd_ragged_tail(self):
        vals = torch.randn(700)
        idxs = torch.arange(700)
        params = {"bucket_size": 512}
        v, _, _ = compressor["qsgd"].compress((vals, idxs, torch.Size([100000])), params)
        assert v.numel() == 700 + 2 * 4
        v2, _, _ = compressor["qsgd"].decompress((v, idxs, torch.Size([100000])), params)
        assert v2.numel() == 700
        assert (v2 - vals).abs().max() < vals.norm() / 127 * 3

    def test_polyfit_roundtrip_shape_and_error(self, sparse):
        t, vals, idxs = sparse
        v, mapping, shape = compressor["polyfit"].compress((vals, idxs, t.size()), {})
        assert v.dtype == torch.float64
        v2, i2, _ = compressor["polyfit"].decompress((v, mapping, shape), {})
        sorted_desc = vals.sort(descending=True).values
        assert v2.numel() == vals.numel()
        # lossy: relative L2 error of the curve fit is bounded
        rel = (v2 - sorted_desc).norm() / sorted_desc.norm()
        assert rel < 0.05
        # mapping permutes original idxs
        assert torch.equal(i2.sort().values, idxs.sort().values)

    def test_polyfit_compresses(self, sparse):
        t, vals, idxs = sparse
        v, _, _ = compressor["polyfit"].compress((vals, idxs, t.size()), {})
        assert v.numel() * 8 < vals.numel() * 4  # payload smaller than fp32 vals

    def test_polyfit_smooth_curve_accurate(self):
        # on an actually-smooth sorted curve the fit should be tight
        N = 4000
        y = torch.linspace(1.0, 0.01, N) ** 2
        idxs = torch.arange(N)
        v, mapping, shape = compressor["polyfit"].compress((y, idxs, torch.Size([200000])), {})
        v2, _, _ = compressor["polyfit"].decompress((v, mapping, shape), {})
        assert (v2 - y.sort(descending=True).values).abs().max() < 1e-3

    def test_polyfit_all_negative(self):
        N = 2000
        y = -torch.rand(N)
        idxs = torch.arange(N)
        v, mapping, shape = compressor["polyfit"].compress((y, idxs, torch.Size([100000])), {})
        v2, _, _ = compressor["polyfit"].decompress((v, mapping, shape), {})
        assert v2.numel() == N

    def test_polyfit_cpu_roundtrip(self, sparse):
        t, vals, idxs = sparse
        v, i, shape = compressor["polyfit_cpu"].compress((vals, idxs, t.size()), {})
        v2, i2, _ = compressor["polyfit_cpu"].decompress((v, i, shape), {})
        sorted_desc = vals.sort(descending=True).values
        rel = (v2 - sorted_desc).norm() / sorted_desc.norm()
        assert rel < 0.05

    def test_doubleexp_roundtrip(self, sparse):
        t, vals, idxs = sparse
        v, m, shape = compressor["doubleexp"].compress((vals, idxs, t.size()), {})
        assert v.numel() == 4  # the whole point: 4 coefficients
        v2, i2, _ = compressor["doubleexp"].decompress((v, m, shape), {})
        assert torch.equal(i2.sort().values, idxs.sort().values)
        # signs must be preserved exactly
        orig_by_idx = dict(zip(idxs.tolist(), vals.tolist()))
        for val, ix in zip(v2.tolist(), i2.tolist()):
            assert (val >= 0) == (orig_by_idx[ix] >= 0) or abs(orig_by_idx[ix]) < 1e-6
        rel = (v2.abs().sort().values - vals.abs().sort().values).norm() / vals.norm()
        assert rel < 0.1


class TestPolySeg:
    def test_roundtrip_and_uniform_payload(self, sparse):
        t, vals, idxs = sparse
        v, i, shape = compressor["polyseg"].compress((vals, idxs, t.size()), {})
        v2, i2, _ = compressor["polyseg"].decompress((v, i, shape), {})
        sorted_desc = vals.sort(descending=True).values
        rel = (v2 - sorted_desc).norm() / sorted_desc.norm()
        assert rel < 0.1
        # payload size depends only on k, not on the value distribution
        vals_b = -vals.abs()  # all negative: different num_pos entirely
        vb, _, _ = compressor["polyseg"].compress((vals_b, idxs, t.size()), {})
        assert vb.numel() == v.numel()

    def test_fixed_segments_deterministic(self):
        from deepreduce_amd.codecs.polyseg import fixed_segments

        assert fixed_segments(500, 10) == fixed_segments(500, 10)
        assert sum(fixed_segments(12345, 10)) == 12345
        assert sum(fixed_segments(7, 10)) == 7


def test_qsgd_packed_levels_roundtrip():
    """qsgd_pack: sub-byte level packing (7 bits at quantum 63) halves...
    reduces the level bytes while reconstruction stays within the QSGD
    error bound."""
    import torch

    from deepreduce_amd.codecs import compressor

    torch.manual_seed(0)
    vals = torch.randn(5000)
    idxs = torch.arange(5000)
    shape = torch.Size([100_000])
    params = {"quantum_num": 63, "bucket_size": 512, "qsgd_pack": True}
    w, i, s = compressor["qsgd"].compress((vals.clone(), idxs, shape), params)
    params_u = dict(params, qsgd_pack=False)
    w8, _, _ = compressor["qsgd"].compress((vals.clone(), idxs, shape), params_u)
    assert w.numel() < w8.numel()  # packed is smaller
    v2, i2, _ = compressor["qsgd"].decompress((w, i, s), params)
    assert torch.equal(i2, idxs)
    pad = (-5000) % 512
    norms = torch.nn.functional.pad(vals, (0, pad)).view(-1, 512).norm(dim=1)
    bound = (norms / 63 * 1.001 + 1e-6).repeat_interleave(512)[:5000]
    assert ((v2 - vals).abs() <= bound).all()
