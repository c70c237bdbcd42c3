"""HIP kernel vs torch-reference parity (runs on an MI355X box only).

Every kernel in ops/src/hip_ops.hip is checked bit-for-bit (lossless ops) or
to an error bound (stochastic QSGD) against ops/reference.py, per the
SURVEY.md sect. 4 test plan.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hip():
    from deepreduce_amd.ops import hip_available

    assert hip_available(), "HIP extension must build+load on the GPU box"
    from deepreduce_amd import _hip_ops

    return _hip_ops


@pytest.fixture
def dev():
    return torch.device("cuda:0")


def test_bloom_insert_parity(hip, dev):
    from deepreduce_amd.ops import reference as ref

    torch.manual_seed(0)
    idxs = torch.randperm(1_000_000)[:5000].to(dev)
    m, k = 140_003, 7
    gpu = hip.bloom_insert(idxs, m, k)
    cpu = ref.bloom_insert(idxs.cpu(), m, k)
    assert torch.equal(gpu.cpu(), cpu)


def test_bloom_query_parity(hip, dev):
    from deepreduce_amd.ops import reference as ref

    torch.manual_seed(1)
    universe = 2_000_000
    idxs = torch.randperm(universe)[:20_000].to(dev)
    m, k = 500_009, 8
    packed = hip.bloom_insert(idxs, m, k)
    gpu_pos = hip.bloom_query_positives(packed, m, k, universe)
    cpu_pos = ref.bloom_query_positives(packed.cpu(), m, k, universe)
    assert torch.equal(gpu_pos.cpu(), cpu_pos)
    # ordered ascending (required by 'leftmost' policy determinism)
    assert torch.equal(gpu_pos, gpu_pos.sort().values)


def test_bloom_members_parity(hip, dev):
    from deepreduce_amd.ops import reference as ref

    idxs = torch.arange(0, 50_000, 7, device=dev)
    m, k = 100_003, 6
    packed = hip.bloom_insert(idxs, m, k)
    probe = torch.arange(0, 60_000, 3, device=dev)
    gpu = hip.bloom_query_members(packed, m, k, probe)
    cpu = ref.bloom_query_members(packed.cpu(), m, k, probe.cpu())
    assert torch.equal(gpu.cpu(), cpu)
    # no false negatives among probes that were actually inserted
    inserted = (probe % 7 == 0) & (probe < 50_000)
    assert gpu[inserted].all()


def test_pack_unpack_parity(hip, dev):
    from deepreduce_amd.ops import reference as ref

    torch.manual_seed(2)
    for nbits in [1, 3, 8, 13, 21, 31]:
        v = torch.randint(0, 2 ** min(nbits, 30), (10_000,), device=dev)
        gpu_stream = hip.pack_ints(v, nbits)
        cpu_stream = ref.pack_ints(v.cpu(), nbits)
        assert torch.equal(gpu_stream.cpu(), cpu_stream), f"nbits={nbits}"
        out = hip.unpack_ints(gpu_stream, v.numel(), nbits)
        assert torch.equal(out.cpu(), v.cpu().long())


def test_qsgd_roundtrip_error_bound(hip, dev):
    torch.manual_seed(3)
    vals = torch.randn(100_000, device=dev)
    levels, norms = hip.qsgd_quantize(vals, 127, 512)
    assert levels.dtype == torch.int8
    assert levels.abs().max() <= 127
    out = hip.qsgd_dequantize(levels, norms, 127, 512)
    # elementwise error <= bucket_norm / quantum
    v = torch.nn.functional.pad(vals, (0, norms.numel() * 512 - vals.numel())).view(-1, 512)
    err = (out - vals).abs().view(1, -1)
    bound = (norms / 127 * 1.001 + 1e-6).repeat_interleave(512)[: vals.numel()]
    assert (err <= bound).all()
    # unbiasedness-ish: mean error small
    assert (out - vals).mean().abs() < 1e-3


def test_qsgd_zero_bucket(hip, dev):
    vals = torch.zeros(1024, device=dev)
    levels, norms = hip.qsgd_quantize(vals, 127, 512)
    assert (levels == 0).all()
    out = hip.qsgd_dequantize(levels, norms, 127, 512)
    assert (out == 0).all()


def test_codec_end_to_end_gpu(dev):
    """Full bloom codec on GPU through the dispatch layer."""
    from deepreduce_amd.codecs import compressor
    from deepreduce_amd.ops import topk_select

    torch.manual_seed(4)
    t = torch.randn(1_000_000, device=dev)
    vals, idxs = topk_select(t, 10_000)
    params = {"policy": "leftmost", "dense_tensor": t}
    v, bits, shape = compressor["bloom"].compress((vals, idxs, t.size()), params)
    v2, i2, _ = compressor["bloom"].decompress((v, bits, shape), {"policy": "leftmost"})
    assert v2.is_cuda and i2.is_cuda
    assert torch.allclose(t[i2], v2)
    true = set(idxs.cpu().tolist())
    rec = set(i2.cpu().tolist())
    # leftmost policy: expected recall ~ 1 - fpr*d/k = 0.9 here
    assert len(true & rec) / len(true) > 0.85


def test_smoke_entrypoint():
    import __graft_entry__ as ge

    ge.smoke()


def test_bloom_query_multi_parity(hip, dev):
    """batched R-filter query == R independent queries, on GPU."""
    torch.manual_seed(9)
    universe = 1_500_000
    m, k = 400_009, 9
    filters, singles = [], []
    for r in range(8):
        idxs = torch.randperm(universe, device=dev)[:15_000]
        b = hip.bloom_insert(idxs, m, k)
        filters.append(b)
        singles.append(hip.bloom_query_positives(b, m, k, universe))
    pos, counts = hip.bloom_query_positives_multi(torch.stack(filters), m, k, universe)
    base = 0
    for r in range(8):
        n = int(counts[r])
        assert n == singles[r].numel(), f"rank {r}"
        assert torch.equal(pos[base : base + n], singles[r]), f"rank {r}"
        base += n


def test_bloom_query_leftmost_parity(hip, dev):
    """sync-free [R,k] leftmost query == first-k of the full query."""
    torch.manual_seed(10)
    universe = 800_000
    m, k = 250_007, 8
    filters, expect = [], []
    kk = 8000
    for r in range(4):
        idxs = torch.randperm(universe, device=dev)[:kk]
        b = hip.bloom_insert(idxs, m, k)
        filters.append(b)
        expect.append(hip.bloom_query_positives(b, m, k, universe)[:kk])
    out = hip.bloom_query_leftmost(torch.stack(filters), m, k, universe, kk)
    assert out.shape == (4, kk)
    for r in range(4):
        assert torch.equal(out[r], expect[r]), f"rank {r}"


def test_topk_select_kernel(hip, dev):
    """radix-select top-k: exact size/values, boundary within the 22-bit
    tie band of the true k-th magnitude, deterministic across calls."""
    from deepreduce_amd.ops import topk_select

    torch.manual_seed(11)
    for n, k in [(1_000_000, 10_000), (25_000_000, 250_000), (4096, 41), (2048, 2048)]:
        t = torch.randn(n, device=dev)
        vals, idxs = topk_select(t, k)
        assert vals.numel() == k and idxs.numel() == k
        assert torch.equal(t[idxs], vals)
        assert idxs.unique().numel() == k  # no duplicates
        # boundary: min selected |v| >= max unselected |v| within tie band
        sel_mask = torch.zeros(n, dtype=torch.bool, device=dev)
        sel_mask[idxs] = True
        min_sel = vals.abs().min()
        max_unsel = t.abs()[~sel_mask].max() if k < n else torch.tensor(0.0, device=dev)
        assert min_sel >= max_unsel * (1 - 2**-12), (min_sel, max_unsel)
        # determinism
        v2, i2 = topk_select(t, k)
        assert torch.equal(i2, idxs)
        # against torch.topk: selected magnitude sum must match to tie-band
        tv, _ = torch.topk(t.abs(), k)
        assert torch.allclose(vals.abs().sum(), tv.sum(), rtol=1e-4)


def test_cholesky_solve_small_parity(hip, dev):
    torch.manual_seed(12)
    S, d = 37, 6
    A = torch.randn(S, d, d, dtype=torch.float64, device=dev)
    G = A @ A.transpose(1, 2) + torch.eye(d, dtype=torch.float64, device=dev) * 1e-3
    b = torch.randn(S, d, dtype=torch.float64, device=dev)
    x = hip.cholesky_solve_small(G, b)
    x_ref = torch.linalg.solve(G.cpu(), b.cpu().unsqueeze(-1)).squeeze(-1)
    assert torch.allclose(x.cpu(), x_ref, rtol=1e-8, atol=1e-10)


def test_graph_vs_eager_training_parity(dev):
    """hipGraph-captured exchange must produce the same training trajectory
    as the eager path (same model, same data, 8 steps)."""
    import copy

    from deepreduce_amd import DistributedOptimizer, deepreduce_from_params

    def run(use_graph):
        torch.manual_seed(7)
        model = torch.nn.Sequential(
            torch.nn.Linear(512, 256), torch.nn.ReLU(), torch.nn.Linear(256, 10)
        ).to(dev)
        params = {
            "compressor": "topk",
            "memory": "residual",
            "communicator": "allgather",
            "compress_ratio": 0.01,
            "deepreduce": "index",
            "index": "bloom",
            "policy": "leftmost",
        }
        grc = deepreduce_from_params(params)
        opt = DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.05), grc, model,
            use_graph=use_graph, graph_warmup=2,
        )
        gen = torch.Generator(device="cpu").manual_seed(11)
        for _ in range(8):
            x = torch.randn(32, 512, generator=gen).to(dev)
            y = torch.randint(0, 10, (32,), generator=gen).to(dev)
            opt.zero_grad(set_to_none=False)
            loss = torch.nn.functional.cross_entropy(model(x), y)
            loss.backward()
            opt.step()
        torch.cuda.synchronize()
        return [p.detach().clone() for p in model.parameters()]

    eager = run(False)
    graphed = run(True)
    for a, b in zip(eager, graphed):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_batched_pipeline_matches_per_tensor(dev):
    """The whole-model batched pipeline (ops/batched.py, bt_* kernels) must
    produce bit-identical results and wire bytes to the per-tensor GPU
    path for the flagship config (topk + bloom + leftmost + residual)."""
    from deepreduce_amd import deepreduce_from_params

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }
    grc_b = deepreduce_from_params(dict(params))
    grc_p = deepreduce_from_params(dict(params))

    torch.manual_seed(42)
    named = [
        ("w1", torch.randn(262_144, device=dev)),
        ("w2", torch.randn(64, 512, device=dev)),
        ("w3", torch.randn(5_000, device=dev)),
    ]
    for step in range(3):
        tensors = [(n, t * (1.0 + step)) for n, t in named]
        fused = grc_b.step_many([(n, t.clone()) for n, t in tensors])
        # confirm the batched path actually ran
        assert getattr(grc_b, "_bt_pipeline", None) is not None, \
            "batched pipeline did not engage"
        loop = [grc_p.step(t.clone(), n) for n, t in tensors]
        for (n, _), f, l in zip(tensors, fused, loop):
            assert torch.equal(f.reshape(-1), l.reshape(-1)), \
                f"step {step} tensor {n}: max diff " \
                f"{(f.reshape(-1) - l.reshape(-1)).abs().max()}"
    torch.cuda.synchronize()


def test_batched_wire_bytes_match(dev):
    from deepreduce_amd import deepreduce_from_params

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }
    grc_b = deepreduce_from_params(dict(params))
    grc_p = deepreduce_from_params(dict(params))
    torch.manual_seed(1)
    named = [("a", torch.randn(100_000, device=dev)),
             ("b", torch.randn(30_000, device=dev))]
    grc_b.step_many([(n, t.clone()) for n, t in named])
    total = 0
    for n, t in named:
        grc_p.step(t.clone(), n)
        total += grc_p.last_wire_bytes
    assert grc_b.last_wire_bytes == total


def test_batched_multi_rank_decode(dev):
    """batched_decode_sum over R stacked wire buffers must equal the sum of
    per-rank generic decompressions (covers the world>1 kernels on one
    GPU: shared-hash qcount, R-plane scan, sequential scatter-adds)."""
    from deepreduce_amd import deepreduce_from_params
    from deepreduce_amd.ops.batched import BatchedPipeline

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }
    R = 4
    numels = [100_000, 40_000, 7_000]
    names = [f"t{i}" for i in range(len(numels))]
    bp = BatchedPipeline(names, numels, params, dev)

    torch.manual_seed(9)
    wires, dense_ref = [], torch.zeros(sum(numels), device=dev)
    grc = deepreduce_from_params(dict(params))
    for r in range(R):
        flat = torch.randn(sum(numels), device=dev)
        wire, out_idx = bp.compress(flat)
        wires.append(wire)
        dense_ref += bp.decode_own(wire, out_idx)
    got = bp.decode_sum(torch.stack(wires))
    assert torch.equal(got, dense_ref.reshape(-1)) or \
        torch.allclose(got, dense_ref.reshape(-1), atol=1e-5), \
        (got - dense_ref).abs().max()

    # cross-check rank-0's wire against the generic per-tensor Bloom
    # decompress (proves wire-format interop between batched and generic)
    from deepreduce_amd.codecs.bloom import Bloom

    flat0 = torch.randn(sum(numels), device=dev)
    wire0, out_idx0 = bp.compress(flat0)
    own0 = bp.decode_own(wire0, out_idx0)
    pad8 = lambda x: (x + 7) & ~7  # noqa: E731
    off_bytes = off_vals = 0
    for (_, k), (_, nb), n in zip(
        [m[0] for m in bp.metas], [m[1] for m in bp.metas], numels
    ):
        vals = wire0[off_bytes : off_bytes + 4 * k].clone().view(torch.float32)
        bits = wire0[off_bytes + pad8(4 * k) : off_bytes + pad8(4 * k) + nb]
        v2, i2, _ = Bloom.decompress((vals, bits, torch.Size([n])), params)
        dense_t = torch.zeros(n, device=dev)
        dense_t.scatter_(0, i2, v2)
        assert torch.equal(dense_t, own0[off_vals : off_vals + n])
        off_bytes += pad8(4 * k) + pad8(nb)
        off_vals += n
    torch.cuda.synchronize()


def test_polyfit_fused_kernel_parity(hip, dev):
    """Fused fit/eval kernels vs the torch float64 reference path."""
    from deepreduce_amd.codecs import polyfit as pf

    torch.manual_seed(3)
    y = torch.sort(torch.randn(20_000).abs(), descending=True).values
    segments = pf.get_segments(20_000, 20_000)  # all-positive curve
    deg = 5
    ref = pf._fit_segments(y.double(), segments, deg)          # CPU torch path
    got = hip.polyfit_fit(y.to(dev), pf._seg_starts(segments, dev), deg).cpu()
    assert torch.allclose(ref, got, rtol=1e-6, atol=1e-9), (ref - got).abs().max()

    ev_ref = pf._eval_segments(ref, segments, torch.device("cpu")).float()
    ev_got = hip.polyfit_eval(got.to(dev), pf._seg_starts(segments, dev), y.numel()).cpu()
    assert torch.allclose(ev_ref, ev_got, rtol=1e-5, atol=1e-6), \
        (ev_ref - ev_got).abs().max()


def test_polyfit_starts_kernel_parity(hip, dev):
    """Device-side padded segment boundaries == python get_segments."""
    from deepreduce_amd.codecs.polyfit import get_segments

    for N, num_pos in [(20_000, 7_345), (1_500, 0), (1_500, 1_500),
                       (2_000_000, 999_999), (1_001, 500)]:
        seg = get_segments(N, num_pos)
        expect = [0]
        for s in seg:
            expect.append(expect[-1] + s)
        npt = torch.tensor([float(num_pos)], dtype=torch.float64, device=dev)
        got = hip.polyfit_starts(npt, N).cpu().tolist()
        assert got == expect, (N, num_pos)


def test_polyfit_gpu_sync_free_roundtrip(dev):
    """GPU polyfit compress/decompress round trip via the codec API; payload
    size must be uniform (function of N only)."""
    from deepreduce_amd.codecs import compressor

    torch.manual_seed(21)
    N = 30_000
    sizes = set()
    for trial in range(3):
        vals = torch.randn(N, device=dev) * (trial + 1)
        idxs = torch.randperm(3_000_000, device=dev)[:N]
        p, m, shape = compressor["polyfit"].compress(
            (vals, idxs, torch.Size([3_000_000])), {"poly_degree": 5})
        sizes.add(p.numel())
        v2, i2, _ = compressor["polyfit"].decompress((p, m, shape), {"poly_degree": 5})
        assert v2.numel() == N
        # fit quality: relative L2 error of the sorted curve reconstruction
        ref = vals.sort(descending=True).values
        err = (v2 - ref).norm() / ref.norm()
        assert err < 0.15, float(err)
    assert len(sizes) == 1  # uniform payload size


def test_graph_vs_eager_both_mode(dev):
    """hipGraph replay of the full 'both' (bloom+polyfit) pipeline must
    match eager training step-for-step."""
    from deepreduce_amd import DistributedOptimizer, deepreduce_from_params

    def run(use_graph):
        torch.manual_seed(17)
        model = torch.nn.Sequential(
            torch.nn.Linear(256, 384), torch.nn.ReLU(), torch.nn.Linear(384, 10)
        ).to(dev)
        params = {
            "compressor": "topk", "memory": "residual",
            "communicator": "allgather", "compress_ratio": 0.02,
            "deepreduce": "both", "index": "bloom", "policy": "leftmost",
            "value": "polyfit",
        }
        grc = deepreduce_from_params(params)
        opt = DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.05), grc, model,
            use_graph=use_graph, graph_warmup=2,
        )
        gen = torch.Generator(device="cpu").manual_seed(23)
        for _ in range(7):
            x = torch.randn(64, 256, generator=gen).to(dev)
            y = torch.randint(0, 10, (64,), generator=gen).to(dev)
            opt.zero_grad(set_to_none=False)
            torch.nn.functional.cross_entropy(model(x), y).backward()
            opt.step()
        torch.cuda.synchronize()
        return [p.detach().clone() for p in model.parameters()]

    eager = run(False)
    graphed = run(True)
    for a, b in zip(eager, graphed):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_batched_both_pipeline_matches_per_tensor(dev):
    """BothPipeline (whole-model bloom+polyfit+mapping) must match the
    per-tensor DeepReduce wrapper path: same wire bytes, same training
    results across steps (residual evolution included)."""
    from deepreduce_amd import deepreduce_from_params

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "both", "index": "bloom", "policy": "leftmost",
        "value": "polyfit",
    }
    grc_b = deepreduce_from_params(dict(params))
    grc_p = deepreduce_from_params(dict(params))
    torch.manual_seed(77)
    named = [
        ("w1", torch.randn(200_000, device=dev)),
        ("w2", torch.randn(48_000, device=dev)),
        ("w3", torch.randn(4_000, device=dev)),
    ]
    for step in range(3):
        tensors = [(n, t * (1.0 + 0.3 * step)) for n, t in named]
        fused = grc_b.step_many([(n, t.clone()) for n, t in tensors])
        assert getattr(grc_b, "_bt_pipeline", None) is not None \
            and grc_b._bt_pipeline[1].kind == "both", "BothPipeline did not engage"
        total = 0
        loop = []
        for n, t in tensors:
            loop.append(grc_p.step(t.clone(), n))
            total += grc_p.last_wire_bytes
        if step == 0:
            assert grc_b.last_wire_bytes == total, \
                (grc_b.last_wire_bytes, total)
        for (n, _), f, l in zip(tensors, fused, loop):
            d = (f.reshape(-1) - l.reshape(-1)).abs().max()
            assert torch.allclose(f.reshape(-1), l.reshape(-1), atol=1e-5), \
                f"step {step} tensor {n}: max diff {d}"
    torch.cuda.synchronize()


def test_batched_value_pipeline_matches_per_tensor(dev):
    """ValuePipeline (polyfit coeffs + int32 idxs, no bloom) vs the
    per-tensor ValueCompressor path: same wire bytes, same results."""
    from deepreduce_amd import deepreduce_from_params

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "value", "value": "polyfit",
    }
    grc_b = deepreduce_from_params(dict(params))
    grc_p = deepreduce_from_params(dict(params))
    torch.manual_seed(88)
    named = [("w1", torch.randn(150_000, device=dev)),
             ("w2", torch.randn(30_000, device=dev)),
             ("w3", torch.randn(5_500, device=dev))]
    for step in range(3):
        tensors = [(n, t * (1.0 + 0.2 * step)) for n, t in named]
        fused = grc_b.step_many([(n, t.clone()) for n, t in tensors])
        assert getattr(grc_b, "_bt_pipeline", None) is not None \
            and grc_b._bt_pipeline[1].kind == "value", "ValuePipeline did not engage"
        total = 0
        loop = []
        for n, t in tensors:
            loop.append(grc_p.step(t.clone(), n))
            total += grc_p.last_wire_bytes
        if step == 0:
            assert grc_b.last_wire_bytes == total, (grc_b.last_wire_bytes, total)
        for (n, _), f, l in zip(tensors, fused, loop):
            assert torch.allclose(f.reshape(-1), l.reshape(-1), atol=1e-5), \
                f"step {step} {n}: {(f - l).abs().max()}"
    torch.cuda.synchronize()


def test_batched_fp16_wire_matches_generic(dev):
    """Index pipeline with wire_dtype=fp16: batched and per-tensor paths
    agree; wire is ~38% smaller than fp32."""
    from deepreduce_amd import deepreduce_from_params

    base = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
        "wire_dtype": "fp16",
    }
    grc_b = deepreduce_from_params(dict(base))
    grc_p = deepreduce_from_params(dict(base))
    grc_32 = deepreduce_from_params({**base, "wire_dtype": "fp32"})
    torch.manual_seed(5)
    named = [("a", torch.randn(180_000, device=dev)),
             ("b", torch.randn(20_000, device=dev))]
    fused = grc_b.step_many([(n, t.clone()) for n, t in named])
    half_bytes = grc_b.last_wire_bytes
    loop = [grc_p.step(t.clone(), n) for n, t in named]
    total = sum([grc_p.last_wire_bytes for _ in [0]])  # last call only
    for f, l in zip(fused, loop):
        assert torch.allclose(f.reshape(-1), l.reshape(-1), atol=1e-6), \
            (f - l).abs().max()
    grc_32.step_many([(n, t.clone()) for n, t in named])
    assert half_bytes < grc_32.last_wire_bytes * 0.75
    _ = total
    torch.cuda.synchronize()


def test_overlapped_reducer_gpu_matches_sync(dev):
    """Bucketed overlapped reducer (batched pipelines on the comm stream)
    must match the synchronous optimizer step-for-step on GPU."""
    from deepreduce_amd import DistributedOptimizer, deepreduce_from_params
    from deepreduce_amd.parallel import OverlappedReducer

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.02,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }

    def make_model():
        torch.manual_seed(31)
        return torch.nn.Sequential(
            torch.nn.Linear(512, 384), torch.nn.ReLU(),
            torch.nn.Linear(384, 384), torch.nn.ReLU(),
            torch.nn.Linear(384, 10),
        ).to(dev)

    def batch(s):
        g = torch.Generator().manual_seed(700 + s)
        return (torch.randn(32, 512, generator=g).to(dev),
                torch.randint(0, 10, (32,), generator=g).to(dev))

    model_a = make_model()
    grc_a = deepreduce_from_params(dict(params))
    opt_a = DistributedOptimizer(torch.optim.SGD(model_a.parameters(), lr=0.1),
                                 grc_a, model_a, use_graph=False)
    for s in range(5):
        x, y = batch(s)
        opt_a.zero_grad(set_to_none=False)
        torch.nn.functional.cross_entropy(model_a(x), y).backward()
        opt_a.step()

    model_b = make_model()
    grc_b = deepreduce_from_params(dict(params))
    reducer = OverlappedReducer(model_b, grc_b, num_buckets=2)
    sgd = torch.optim.SGD(model_b.parameters(), lr=0.1)
    for s in range(5):
        x, y = batch(s)
        sgd.zero_grad(set_to_none=False)
        reducer.zero_wire_counter()
        torch.nn.functional.cross_entropy(model_b(x), y).backward()
        reducer.finalize()
        sgd.step()
    torch.cuda.synchronize()
    for a, b in zip(model_a.parameters(), model_b.parameters()):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_graph_recaptures_on_grad_realloc(dev):
    """zero_grad(set_to_none=True) reallocates gradient storages; the
    captured exchange must detect the stale pointers and re-capture
    instead of silently replaying old buffers."""
    from deepreduce_amd import DistributedOptimizer, deepreduce_from_params

    torch.manual_seed(41)
    model = torch.nn.Sequential(torch.nn.Linear(2048, 512), torch.nn.ReLU(),
                                torch.nn.Linear(512, 8)).to(dev)
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.02,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    })
    opt = DistributedOptimizer(torch.optim.SGD(model.parameters(), lr=0.05),
                               grc, model, use_graph=True, graph_warmup=2)
    gen = torch.Generator().manual_seed(5)

    def one_step(set_to_none):
        x = torch.randn(16, 2048, generator=gen).to(dev)
        y = torch.randint(0, 8, (16,), generator=gen).to(dev)
        opt.zero_grad(set_to_none=set_to_none)
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()

    for _ in range(4):
        one_step(False)
    assert opt._graph is not None
    g1 = opt._graph
    one_step(True)   # grads reallocated -> must recapture (not crash/corrupt)
    one_step(True)
    torch.cuda.synchronize()
    assert opt._graph is not g1 or opt._graph is None
    # training remains finite and sane
    assert all(torch.isfinite(p).all() for p in model.parameters())


def test_dexp_fit_kernel_parity(hip, dev):
    """Fused DoubleExp fit kernel vs the torch fp64 reference math
    (codecs/doubleexp._double_exp_fit): same coefficients to fp32-input
    tolerance, same reconstruction."""
    from deepreduce_amd.codecs.doubleexp import _double_exp_fit

    torch.manual_seed(5)
    for N in [1200, 9000, 50_000]:
        # exponential-ish sorted magnitudes (what sorted topk values look like)
        y = torch.sort(torch.randn(N, device=dev).abs() ** 2.0).values
        ref = _double_exp_fit(y.double())
        ref = torch.stack(list(ref)).cpu()
        got = hip.dexp_fit(
            y.float(),
            torch.zeros(1, dtype=torch.int64, device=dev),
            torch.tensor([N], dtype=torch.int64, device=dev),
        ).cpu().reshape(-1)

        x = torch.arange(1, N + 1, dtype=torch.float64)
        rec_ref = ref[0] * torch.exp(ref[1] * x) + ref[2] * torch.exp(ref[3] * x)
        rec_got = got[0] * torch.exp(got[1] * x) + got[2] * torch.exp(got[3] * x)
        denom = y.double().cpu().norm() + 1e-12
        err_ref = (rec_ref - y.double().cpu()).norm() / denom
        err_got = (rec_got - y.double().cpu()).norm() / denom
        # the kernel's fit must be as good as the reference fit (small slack
        # for fp32 input + different fp64 summation order)
        assert err_got <= err_ref + 0.02, (N, float(err_got), float(err_ref))


def test_dexp_fit_batched_tensors(hip, dev):
    """One launch fits B tensors (one block each)."""
    torch.manual_seed(6)
    lens = [3000, 12_000, 700]
    ys = [torch.sort(torch.randn(n, device=dev).abs()).values for n in lens]
    flat = torch.cat(ys).float()
    offs = torch.tensor([0, 3000, 15_000], dtype=torch.int64, device=dev)
    lent = torch.tensor(lens, dtype=torch.int64, device=dev)
    out = hip.dexp_fit(flat, offs, lent)
    assert out.shape == (3, 4)
    for i, (y, n) in enumerate(zip(ys, lens)):
        c = out[i].cpu()
        single = hip.dexp_fit(
            y.float(), torch.zeros(1, dtype=torch.int64, device=dev),
            torch.tensor([n], dtype=torch.int64, device=dev)).cpu().reshape(-1)
        assert torch.allclose(c, single, rtol=1e-10, atol=1e-12), i


def test_doubleexp_codec_gpu_matches_cpu(dev):
    """DoubleExp codec end-to-end on GPU (device fit kernel) vs CPU (torch
    fp64 path): same decompressed tensor to fp32 tolerance."""
    from deepreduce_amd.codecs import compressor as registry

    torch.manual_seed(7)
    t = torch.randn(200_000)
    k = 20_000
    _, idx = t.abs().topk(k)
    vals = t[idx]

    codec = registry["doubleexp"]
    v_c, m_c, s_c = codec.compress((vals, idx, t.size()), {})
    out_c = codec.decompress((v_c, m_c, s_c), {})

    v_g, m_g, s_g = codec.compress((vals.to(dev), idx.to(dev), t.size()), {})
    out_g = codec.decompress((v_g, m_g, s_g), {})

    assert torch.equal(m_g.cpu(), m_c)  # mapping identical
    dense_c = torch.zeros(t.numel()).scatter_(0, out_c[1], out_c[0])
    dense_g = torch.zeros(t.numel(), device=dev).scatter_(0, out_g[1], out_g[0])
    scale = vals.abs().max()
    assert torch.allclose(dense_g.cpu(), dense_c, atol=2e-3 * float(scale)), \
        (dense_g.cpu() - dense_c).abs().max()


def test_batched_multi_rank_decode_r8_interleaved(dev):
    """R=8 (the SCALE run's world size) through the interleaved universe
    query must equal the sum of own-decodes."""
    from deepreduce_amd import deepreduce_from_params
    from deepreduce_amd.ops.batched import BatchedPipeline

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }
    R = 8
    numels = [300_000, 50_000, 9_000, 123_456]
    names = [f"t{i}" for i in range(len(numels))]
    bp = BatchedPipeline(names, numels, params, dev)
    assert bp.total_mw > 0

    torch.manual_seed(11)
    wires, dense_ref = [], torch.zeros(sum(numels), device=dev)
    for r in range(R):
        flat = torch.randn(sum(numels), device=dev)
        wire, out_idx = bp.compress(flat)
        wires.append(wire)
        dense_ref += bp.decode_own(wire, out_idx)
    got = bp.decode_sum(torch.stack(wires))
    assert torch.allclose(got, dense_ref.reshape(-1), atol=1e-5), \
        (got - dense_ref).abs().max()


def test_batched_large_filter_lds_fallback(dev, monkeypatch):
    """With the LDS cache enabled (opt-in, DEEPREDUCE_LDSQ_MAX), a filter
    larger than the budget must fall back to global word loads inside
    bt_qcount (and still match the per-tensor path)."""
    from deepreduce_amd import deepreduce_from_params
    from deepreduce_amd.ops import batched as bt
    from deepreduce_amd.ops.batched import BatchedPipeline

    monkeypatch.setattr(bt, "LDSQ_MAX", 65536)
    params = {
        "compressor": "topk", "memory": "none",
        "communicator": "allgather", "compress_ratio": 0.05,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }
    numels = [4_000_000, 60_000]   # k=200k -> filter ~0.4 MB >> LDSQ_MAX
    names = ["big", "small"]
    bp = BatchedPipeline(names, numels, params, dev)
    mws = [(int(bp.desc[t, 4]) + 31) // 32 for t in range(2)]
    assert mws[0] * 4 > bt.LDSQ_MAX, "test premise: big filter exceeds budget"
    assert mws[1] * 4 <= bt.LDSQ_MAX
    assert bp.ldsq_bytes > 0, "LDS cache should be armed for the small filter"

    torch.manual_seed(13)
    flat = torch.randn(sum(numels), device=dev)
    wire, out_idx = bp.compress(flat)
    own = bp.decode_own(wire, out_idx)

    # generic per-tensor path on the same inputs
    grc = deepreduce_from_params(dict(params))
    off = 0
    for n, nel in zip(names, numels):
        t = flat[off : off + nel]
        tc, ctx = grc.compressor.compress(t, n)
        dec = grc.compressor.decompress(tc, ctx)
        assert torch.allclose(own[off : off + nel], dec.reshape(-1), atol=1e-6), n
        off += nel


def test_block_pfor_gpu_roundtrip(dev):
    """Block-PFoR runs GPU-resident (torch-vectorized): roundtrip on
    device, wire identical to the CPU encode."""
    from deepreduce_amd.codecs.intpack import pfor_decode, pfor_encode

    torch.manual_seed(17)
    g = torch.randint(0, 64, (50_000,), dtype=torch.int64)
    g[torch.randint(0, 50_000, (64,))] = torch.randint(
        1 << 20, 1 << 28, (64,), dtype=torch.int64)
    w_cpu = pfor_encode(g)
    w_gpu = pfor_encode(g.to(dev))
    assert w_gpu.is_cuda
    assert torch.equal(w_gpu.cpu(), w_cpu), "wire must be device-independent"
    out = pfor_decode(w_gpu)
    assert out.is_cuda
    assert torch.equal(out.cpu(), g)


def test_batched_decode_r20_chunked(dev):
    """World sizes beyond the 16-filter kernel cap decode in chunks of 16
    (VERDICT r1 weak item 3): R=20 must equal the sum of own-decodes."""
    from deepreduce_amd.ops.batched import BatchedPipeline

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }
    numels = [80_000, 23_456]
    bp = BatchedPipeline(["a", "b"], numels, params, dev)
    torch.manual_seed(23)
    wires, ref = [], torch.zeros(sum(numels), device=dev)
    for r in range(20):
        flat = torch.randn(sum(numels), device=dev)
        wire, out_idx = bp.compress(flat)
        wires.append(wire)
        ref += bp.decode_own(wire, out_idx)
    got = bp.decode_sum(torch.stack(wires))
    assert torch.allclose(got, ref, atol=1e-4), (got - ref).abs().max()


def test_generic_decompress_batch_r20(dev):
    """IndexCompressor.decompress_batch chunks >16 payloads too."""
    from deepreduce_amd import deepreduce_from_params

    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "none",
        "communicator": "allgather", "compress_ratio": 0.01,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    })
    comp = grc.compressor
    torch.manual_seed(29)
    N = 120_000
    payloads, ref = [], torch.zeros(N, device=dev)
    ctx = None
    for r in range(20):
        t = torch.randn(N, device=dev)
        tc, ctx = comp.compress(t, f"r{r}")
        payloads.append(tc)
        ref += comp.decompress(tc, ctx).reshape(-1)
    got = comp.decompress_batch(payloads, ctx)
    assert got is not None, "fast path must engage for 20 ranks"
    assert torch.allclose(got.reshape(-1), ref, atol=1e-4), \
        (got.reshape(-1) - ref).abs().max()
