"""Host runtime: sparsifiers, residual memory, factory/params contract,
packing helpers, hashing invariants."""
import torch

from deepreduce_amd import (
    DeepReduce,
    IndexCompressor,
    ResidualMemory,
    TopKCompressor,
    ValueCompressor,
    deepreduce_from_params,
    grace_from_params,
    tensor_bits,
)
from deepreduce_amd.communicator import Allgather, _flatten_payload, _unflatten_payload
from deepreduce_amd.compressors import ThresholdCompressor, RandomKCompressor
from deepreduce_amd.hashing import fmix32


def test_topk_sparsifier_roundtrip():
    t = torch.randn(4, 100)
    c = TopKCompressor(0.05)
    (vals, idxs), ctx = c.compress(t, "w")
    assert vals.numel() == 20
    dense = c.decompress((vals, idxs), ctx)
    assert dense.shape == t.shape
    # kept entries exact, others zero
    mask = torch.zeros(400, dtype=torch.bool)
    mask[idxs] = True
    assert torch.equal(dense.reshape(-1)[mask], t.reshape(-1)[mask])
    assert (dense.reshape(-1)[~mask] == 0).all()


def test_threshold_sparsifier():
    t = torch.randn(1000)
    c = ThresholdCompressor(0.5)
    (vals, idxs), ctx = c.compress(t, "w")
    assert (vals.abs() >= 0.5).all()
    assert not c.tensors_size_are_same


def test_randomk_deterministic_across_ranks():
    t1, t2 = torch.randn(1000), torch.randn(1000)
    c1, c2 = RandomKCompressor(0.1), RandomKCompressor(0.1)
    (_, i1), _ = c1.compress(t1, "w")
    (_, i2), _ = c2.compress(t2, "w")
    assert torch.equal(i1, i2)  # same name+step -> same positions


def test_residual_memory_error_feedback():
    mem = ResidualMemory()
    c = TopKCompressor(0.1)
    t = torch.randn(500)
    comp = mem.compensate(t, "w")
    assert torch.equal(comp, t)  # first step: no residual
    payload, ctx = c.compress(comp, "w")
    mem.update(comp, "w", c, payload, ctx)
    # residual + decompressed == compensated (exact for topk)
    assert torch.allclose(mem.residuals["w"] + c.decompress(payload, ctx), comp)
    # second step: compensation adds the residual
    t2 = torch.randn(500)
    comp2 = mem.compensate(t2, "w")
    assert torch.allclose(comp2, mem.residuals["w"] + t2)


def test_memory_checkpoint_roundtrip():
    mem = ResidualMemory()
    mem.residuals["w"] = torch.randn(10)
    sd = mem.state_dict()
    mem2 = ResidualMemory()
    mem2.load_state_dict(sd)
    assert torch.equal(mem2.residuals["w"], mem.residuals["w"])


def test_grace_from_params_readme_contract():
    """The README.md:37 params dict must work unchanged."""
    params = {
        "compressor": "topk",
        "memory": "residual",
        "communicator": "allgather",
        "compress_ratio": 0.01,
        "deepreduce": "index",
        "index": "bloom",
    }
    grc = grace_from_params(params)
    assert isinstance(grc, Allgather)
    assert isinstance(grc.compressor, TopKCompressor)
    # wrap-after-build, exactly as README.md:44-48 does
    from deepreduce_amd.wrappers import deepreduce_wrapper

    grc.compressor = deepreduce_wrapper[params["deepreduce"]](grc.compressor, params)
    assert isinstance(grc.compressor, IndexCompressor)


def test_deepreduce_from_params_modes():
    base = {"compressor": "topk", "memory": "none", "communicator": "allgather",
            "compress_ratio": 0.02}
    assert isinstance(deepreduce_from_params({**base, "deepreduce": "value"}).compressor, ValueCompressor)
    assert isinstance(deepreduce_from_params({**base, "deepreduce": "index"}).compressor, IndexCompressor)
    assert isinstance(deepreduce_from_params({**base, "deepreduce": "both"}).compressor, DeepReduce)
    assert isinstance(deepreduce_from_params(base).compressor, TopKCompressor)
    # hash_table key accepted (and ignored) for reference compatibility
    deepreduce_from_params({**base, "deepreduce": "index", "hash_table": None})


def test_single_rank_step_end_to_end():
    params = {"compressor": "topk", "memory": "residual", "communicator": "allgather",
              "compress_ratio": 0.05, "deepreduce": "index", "index": "bloom"}
    grc = deepreduce_from_params(params)
    g = torch.randn(5000)
    out = grc.step(g, "layer.weight")
    assert out.shape == g.shape
    assert (out != 0).sum() > 0


def test_tensor_bits():
    assert tensor_bits([torch.zeros(10, dtype=torch.float32)]) == 320
    assert tensor_bits([torch.zeros(10, dtype=torch.uint8)]) == 80
    assert tensor_bits([torch.zeros(3, dtype=torch.int64), torch.zeros(2, dtype=torch.int8)]) == 208


def test_payload_flatten_roundtrip():
    payload = (torch.randn(7), torch.arange(5, dtype=torch.int64),
               torch.zeros(3, dtype=torch.uint8))
    buf, metas = _flatten_payload(payload)
    assert buf.dtype == torch.uint8
    out = _unflatten_payload(buf, metas)
    for a, b in zip(payload, out):
        assert torch.equal(a, b)


def test_fmix32_known_values():
    # murmur3 fmix32 reference vectors
    import numpy as np

    def ref(h):
        h ^= h >> 16
        h = (h * 0x85EBCA6B) & 0xFFFFFFFF
        h ^= h >> 13
        h = (h * 0xC2B2AE35) & 0xFFFFFFFF
        h ^= h >> 16
        return h

    xs = torch.tensor([0, 1, 2, 12345, 0xDEADBEEF, 0xFFFFFFFF], dtype=torch.int64)
    out = fmix32(xs)
    for x, o in zip(xs.tolist(), out.tolist()):
        assert o == ref(x)


def test_cpu_native_matches_torch_reference():
    """C++ CPU ops (if the extension is loadable) must be bit-identical to
    the torch reference implementations."""
    import deepreduce_amd.ops as ops
    from deepreduce_amd.ops import reference as ref

    if not ops.hip_available():
        import pytest

        pytest.skip("extension not built")
    from deepreduce_amd import _hip_ops

    idxs = torch.randperm(200_000)[:2000]
    m, k = 40_009, 7
    assert torch.equal(_hip_ops.bloom_insert_cpu(idxs, m, k), ref.bloom_insert(idxs, m, k))
    packed = ref.bloom_insert(idxs, m, k)
    assert torch.equal(
        _hip_ops.bloom_query_positives_cpu(packed, m, k, 200_000),
        ref.bloom_query_positives(packed, m, k, 200_000),
    )
    probe = torch.arange(0, 200_000, 17)
    assert torch.equal(
        _hip_ops.bloom_query_members_cpu(packed, m, k, probe),
        ref.bloom_query_members(packed, m, k, probe),
    )
    for nbits in (1, 5, 13, 21):
        v = torch.randint(0, 2**nbits, (5000,))
        assert torch.equal(_hip_ops.pack_ints_cpu(v, nbits), ref.pack_ints(v, nbits))
        assert torch.equal(
            _hip_ops.unpack_ints_cpu(ref.pack_ints(v, nbits), 5000, nbits), v.long()
        )


def test_checkpoint_resume_matches_uninterrupted():
    """Residual memory travels through DistributedOptimizer.state_dict:
    train 3+3 steps with a checkpoint/restore in the middle == 6 straight
    (exercises the flat-pool rebuild in ResidualMemory.load_state_dict)."""
    import torch.nn as nn

    from deepreduce_amd import DistributedOptimizer, deepreduce_from_params

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.05,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    }

    def make():
        torch.manual_seed(3)
        model = nn.Sequential(nn.Linear(500, 30), nn.ReLU(), nn.Linear(30, 5))
        grc = deepreduce_from_params(dict(params))
        opt = DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9), grc, model)
        return model, opt

    def step(model, opt, s):
        g = torch.Generator().manual_seed(42 + s)
        x = torch.randn(8, 500, generator=g)
        y = torch.randint(0, 5, (8,), generator=g)
        opt.zero_grad(set_to_none=False)
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()

    # uninterrupted
    m1, o1 = make()
    for s in range(6):
        step(m1, o1, s)

    # interrupted + resumed
    m2, o2 = make()
    for s in range(3):
        step(m2, o2, s)
    ckpt = {"model": m2.state_dict(), "opt": o2.state_dict()}
    m3, o3 = make()
    m3.load_state_dict(ckpt["model"])
    o3.load_state_dict(ckpt["opt"])
    for s in range(3, 6):
        step(m3, o3, s)

    for a, b in zip(m1.parameters(), m3.parameters()):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_model_zoo_forward_backward():
    """Every registry model runs a forward+backward on CPU-shaped synthetic
    data (reference benchmark matrix + the paper's FL backbones)."""
    import torch.nn.functional as F

    from deepreduce_amd.models import registry

    batches = {
        "resnet20": lambda: ((torch.randn(2, 3, 32, 32),),
                             torch.randint(0, 10, (2,))),
        "mobilenet": lambda: ((torch.randn(2, 3, 32, 32),),
                              torch.randint(0, 10, (2,))),
        "rnn": lambda: ((torch.randint(0, 10_004, (2, 20)),),
                        torch.randint(0, 10_004, (2, 20))),
    }
    for name in ("resnet20", "mobilenet", "rnn"):
        model = registry[name]()
        inputs, target = batches[name]()
        out = model(*inputs)
        if name == "rnn":
            loss = F.cross_entropy(out.flatten(0, 1), target.flatten())
        else:
            loss = F.cross_entropy(out, target)
        loss.backward()
        assert all(p.grad is not None for p in model.parameters()
                   if p.requires_grad)


def test_params_validation():
    import pytest as _pytest

    from deepreduce_amd import deepreduce_from_params

    with _pytest.raises(ValueError, match="unknown params key"):
        deepreduce_from_params({"compresor": "topk"})  # typo
    with _pytest.raises(ValueError, match="not in"):
        deepreduce_from_params({"deepreduce": "bot"})
    with _pytest.raises(ValueError, match="compress_ratio"):
        deepreduce_from_params({"compress_ratio": 3.0})
    # reference-compat key accepted and ignored
    deepreduce_from_params({"compressor": "topk", "hash_table": "/tmp/x.pt"})


def test_log_stats_wiring(tmp_path):
    """params['log_stats'] produces per-step per-tensor JSONL records."""
    import json

    from deepreduce_amd import deepreduce_from_params

    out = str(tmp_path / "stats")
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.05,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
        "log_stats": out,
    })
    for s in range(2):
        grc.step_many([("w", torch.randn(5000)), ("b", torch.randn(64))])
    fn = grc._stats.dump()
    recs = [json.loads(l) for l in open(fn)]
    assert len(recs) == 4  # 2 tensors x 2 steps
    assert {r["tensor"] for r in recs} == {"w", "b"}
    assert all(r["wire_bytes"] > 0 and r["dense_bytes"] > 0 for r in recs)


def test_randomk_same_positions_across_ranks():
    """RandomK must pick identical positions for the same (name, step) on
    every rank (seed = hash(name)+step, tensorflow/deepreduce.py:290-298),
    or decompressed averages diverge."""
    from deepreduce_amd.compressors import RandomKCompressor

    a, b = RandomKCompressor(0.05), RandomKCompressor(0.05)
    t1 = torch.randn(4000)
    t2 = torch.randn(4000)  # different values, same positions expected
    (_, ia), _ = a.compress(t1, "layer.weight")
    (_, ib), _ = b.compress(t2, "layer.weight")
    assert torch.equal(ia, ib)
    # next step: different positions than step 0, still rank-consistent
    (_, ia2), _ = a.compress(t1, "layer.weight")
    (_, ib2), _ = b.compress(t2, "layer.weight")
    assert torch.equal(ia2, ib2)
    assert not torch.equal(ia, ia2)


def test_codec_edge_cases_tiny_k():
    """k=1 and k=2 survive every registered codec round trip."""
    from deepreduce_amd.codecs import compressor

    shape = torch.Size([5000])
    for name in ["bloom", "rle", "huffman", "pfor", "qsgd", "gzip"]:
        for k in (1, 2):
            vals = torch.randn(k)
            idxs = torch.tensor(list(range(17, 17 + k)), dtype=torch.int64)
            params = {"policy": "p0"} if name == "bloom" else {}
            v, w, s = compressor[name].compress((vals.clone(), idxs.clone(), shape), params)
            v2, i2, _ = compressor[name].decompress((v, w, s), params)
            if name == "bloom":
                assert set(idxs.tolist()) <= set(i2.tolist()), name
            elif name in ("gzip", "qsgd"):
                assert torch.equal(i2, idxs), name
            else:
                assert set(i2.tolist()) == set(idxs.tolist()), (name, k)


def test_frozen_params_are_skipped():
    """Models with frozen (requires_grad=False) or grad-less parameters
    exchange only the live gradients."""
    import torch.nn as nn

    from deepreduce_amd import DistributedOptimizer, deepreduce_from_params

    torch.manual_seed(1)
    model = nn.Sequential(nn.Linear(2000, 50), nn.ReLU(), nn.Linear(50, 4))
    model[0].weight.requires_grad_(False)
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.05,
        "deepreduce": "index", "index": "bloom", "policy": "leftmost",
    })
    opt = DistributedOptimizer(
        torch.optim.SGD([p for p in model.parameters() if p.requires_grad],
                        lr=0.1), grc, model)
    frozen_before = model[0].weight.detach().clone()
    for s in range(3):
        x = torch.randn(8, 2000)
        y = torch.randint(0, 4, (8,))
        opt.zero_grad(set_to_none=False)
        torch.nn.functional.cross_entropy(model(x), y).backward()
        opt.step()
    assert torch.equal(model[0].weight, frozen_before)
    assert model[2].weight.grad is not None


def test_bench_json_contract_single_process():
    """bench.py's rank-0 JSON must carry every key the round driver
    parses, with sane values (single-process CPU invocation)."""
    import json
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "resnet20", "--steps", "2",
         "--warmup", "1", "--batch", "4", "--device", "cpu"],
        capture_output=True, text=True, timeout=600,
        cwd=__import__("os").path.dirname(__import__("os").path.dirname(
            __import__("os").path.abspath(__file__))),
    )
    assert out.returncode == 0, out.stderr[-1500:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1  # exactly ONE JSON line
    d = json.loads(lines[0])
    for key in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"]:
        assert key in d, key
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["data"] == "synthetic" and d["n_gpus"] == 1
    for key in ["model", "global_batch", "seq_len", "parallelism"]:
        assert key in d["config"], key
    assert d["value"] > 0 and d["ms_per_step"] > 0


def test_broadcast_communicator_and_logger_csv(tmp_path):
    """Broadcast communicator round trip (world=1) and the Logger-op
    equivalent CSV dump (logger.cc parity)."""
    from deepreduce_amd import deepreduce_from_params
    from deepreduce_amd.metrics import StatsLogger

    grc = deepreduce_from_params({
        "compressor": "none", "memory": "none", "communicator": "broadcast"})
    t = torch.randn(128)
    out = grc.step(t.clone(), "w")
    assert torch.equal(out, t)

    st = StatsLogger(str(tmp_path))
    fn = st.dump_values("conv1/weight", torch.tensor([1.5, -2.25]),
                        torch.tensor([0.1, 0.2, 0.3]))
    lines = open(fn).read().strip().splitlines()
    assert lines[0] == "1.5,-2.25"
    assert lines[1].startswith("0.1")


def test_intpack_wide_values():
    """n-bit pack/unpack at widths up to 63 bits."""
    from deepreduce_amd.codecs.intpack import pack_with_header, unpack_with_header

    for nbits in (1, 7, 24, 33, 48, 57):
        v = torch.randint(0, 2, (257,), dtype=torch.int64) * ((1 << (nbits - 1)) - 1)
        wire = pack_with_header(v, nbits=nbits)
        out = unpack_with_header(wire)
        assert torch.equal(out, v), nbits
    import pytest as _pytest

    with _pytest.raises(ValueError, match="max 57"):
        pack_with_header(torch.tensor([1]), nbits=63)
