"""Multi-process plumbing tests on gloo CPU, world_size=2 — BASELINE.json
config 1 (ResNet-20/CIFAR-10 top-k 1% + residual over allgather) plus the
ragged-payload path.  No GPU required."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from deepreduce_amd.models import resnet20


def _init(rank, world):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ.setdefault("MASTER_PORT", "29611")
    dist.init_process_group("gloo", rank=rank, world_size=world)


def _run_allgather_step(rank, world, mode, q):
    try:
        _init(rank, world)
        from deepreduce_amd import deepreduce_from_params

        params = {
            "compressor": "topk",
            "memory": "residual",
            "communicator": "allgather",
            "compress_ratio": 0.01,
        }
        if mode == "index":
            params.update({"deepreduce": "index", "index": "bloom", "policy": "leftmost"})
        elif mode == "both":
            params.update({"deepreduce": "both", "index": "bloom", "value": "polyfit"})
        elif mode == "value":
            params.update({"deepreduce": "value", "value": "qsgd"})
        grc = deepreduce_from_params(params)

        torch.manual_seed(100 + rank)  # DIFFERENT grads per rank
        g = torch.randn(8000)
        out = grc.step(g, "w")
        # every rank must compute the identical averaged tensor
        outs = [torch.empty_like(out) for _ in range(world)]
        dist.all_gather(outs, out)
        same = all(torch.allclose(outs[0], o, atol=1e-5) for o in outs)
        q.put((rank, bool(same), float(out.abs().sum())))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e!r}", 0.0))


@pytest.mark.parametrize("mode", ["plain", "index", "value", "both"])
def test_allgather_identical_average_across_ranks(mode):
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = str(29620 + hash(mode) % 50)
    procs = [ctx.Process(target=_run_allgather_step, args=(r, world, mode, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, same, s in results:
        assert same is True, f"rank {rank}: {same}"
        assert s > 0


def _run_resnet20_training(rank, world, q):
    try:
        _init(rank, world)
        from deepreduce_amd import (
            DistributedOptimizer,
            broadcast_parameters,
            deepreduce_from_params,
        )

        torch.manual_seed(0)  # same init everywhere, then broadcast anyway
        model = resnet20()
        broadcast_parameters(model)
        params = {
            "compressor": "topk",
            "memory": "residual",
            "communicator": "allgather",
            "compress_ratio": 0.01,
            "deepreduce": "index",
            "index": "bloom",
        }
        grc = deepreduce_from_params(params)
        opt = DistributedOptimizer(torch.optim.SGD(model.parameters(), lr=0.05), grc, model)

        torch.manual_seed(1000 + rank)  # different data per rank
        losses = []
        for _ in range(3):
            x = torch.randn(16, 3, 32, 32)
            y = torch.randint(0, 10, (16,))
            opt.zero_grad()
            loss = torch.nn.functional.cross_entropy(model(x), y)
            loss.backward()
            opt.step()
            losses.append(loss.item())
        # parameters must stay in sync across ranks
        p0 = next(model.parameters()).data.reshape(-1)[:100]
        ps = [torch.empty_like(p0) for _ in range(world)]
        dist.all_gather(ps, p0)
        in_sync = all(torch.allclose(ps[0], p, atol=1e-5) for p in ps)
        q.put((rank, bool(in_sync), losses, opt.last_wire_bytes))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e!r}", [], 0))


def test_resnet20_cifar_topk_training_stays_in_sync():
    """BASELINE.json config 1."""
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = "29701"
    procs = [ctx.Process(target=_run_resnet20_training, args=(r, world, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=600) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, in_sync, losses, wire in results:
        assert in_sync is True, f"rank {rank}: {in_sync}"
        assert len(losses) == 3
        assert wire > 0


def _run_ragged(rank, world, q):
    try:
        _init(rank, world)
        from deepreduce_amd import deepreduce_from_params

        params = {
            "compressor": "threshold",
            "threshold": 0.5 + 0.3 * rank,  # DIFFERENT payload sizes per rank
            "memory": "none",
            "communicator": "allgather",
        }
        grc = deepreduce_from_params(params)
        torch.manual_seed(55 + rank)
        g = torch.randn(5000)
        out = grc.step(g, "w")
        outs = [torch.empty_like(out) for _ in range(world)]
        dist.all_gather(outs, out)
        same = all(torch.allclose(outs[0], o, atol=1e-5) for o in outs)
        q.put((rank, bool(same)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e!r}"))


def test_ragged_allgather_threshold():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = "29702"
    procs = [ctx.Process(target=_run_ragged, args=(r, world, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, same in results:
        assert same is True, f"rank {rank}: {same}"


def _run_fused_equivalence(rank, world, mode, q):
    try:
        _init(rank, world)
        from deepreduce_amd import deepreduce_from_params

        params = {
            "compressor": "topk",
            "memory": "residual",
            "communicator": "allgather",
            "compress_ratio": 0.01,
            "small_dense": False,  # strict per-tensor parity with grc.step
        }
        if mode == "index":
            params.update({"deepreduce": "index", "index": "bloom", "policy": "leftmost"})
        elif mode == "fp16":
            params.update({"deepreduce": "index", "index": "bloom",
                           "policy": "leftmost", "wire_dtype": "fp16"})
        elif mode == "both":
            params.update({"deepreduce": "both", "index": "bloom", "value": "polyfit"})
        elif mode == "value":
            params.update({"deepreduce": "value", "value": "polyfit"})
        elif mode == "bothq":
            params.update({"deepreduce": "both", "index": "bloom", "value": "qsgd"})
        elif mode == "dexp":
            params.update({"deepreduce": "value", "value": "doubleexp"})
        elif mode == "dense":
            params = {"compressor": "none", "memory": "none", "communicator": "allreduce"}
        grc_a = deepreduce_from_params(dict(params))
        grc_b = deepreduce_from_params(dict(params))

        torch.manual_seed(300 + rank)
        named = [("a", torch.randn(9000)), ("b", torch.randn(4, 700)),
                 ("c", torch.randn(64))]
        ok = True
        for step in range(3):  # multiple steps: residuals must evolve identically
            grads = [(n, (t * (step + 1)).clone()) for n, t in named]
            # qsgd rounds stochastically: align the RNG draws of both paths
            torch.manual_seed(9000 + step)
            fused = grc_a.step_many([(n, t.clone()) for n, t in grads])
            torch.manual_seed(9000 + step)
            loop = [grc_b.step(t.clone(), n) for n, t in grads]
            for f, l in zip(fused, loop):
                if not torch.allclose(f.reshape(-1), l.reshape(-1), atol=1e-5):
                    ok = False
        q.put((rank, ok))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.parametrize("mode", ["plain", "index", "fp16", "both", "value",
                                  "bothq", "dexp", "dense"])
def test_fused_step_many_matches_per_tensor(mode):
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = str(29680 + hash("f" + mode) % 50)
    procs = [ctx.Process(target=_run_fused_equivalence, args=(r, world, mode, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok in results:
        assert ok is True, f"rank {rank}: {ok}"


def _run_small_dense(rank, world, q):
    try:
        _init(rank, world)
        from deepreduce_amd import deepreduce_from_params

        grc = deepreduce_from_params({
            "compressor": "topk", "memory": "residual",
            "communicator": "allgather", "compress_ratio": 0.01,
            "deepreduce": "index", "index": "bloom", "policy": "leftmost",
        })
        torch.manual_seed(500 + rank)
        small = torch.randn(64)
        large = torch.randn(8192)
        out = grc.step_many([("w", large.clone()), ("b", small.clone())])
        # small tensor must be the EXACT dense average across ranks
        gathered = [torch.empty_like(small) for _ in range(world)]
        dist.all_gather(gathered, small)
        expect = torch.stack(gathered).mean(0)
        ok = torch.allclose(out[1], expect, atol=1e-6)
        q.put((rank, bool(ok)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e!r}"))


def test_small_tensors_travel_dense_exact():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = "29745"
    procs = [ctx.Process(target=_run_small_dense, args=(r, world, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok in results:
        assert ok is True, f"rank {rank}: {ok}"


def _run_overlap_parity(rank, world, q):
    try:
        _init(rank, world)
        import torch.nn as nn

        from deepreduce_amd import (DistributedOptimizer, broadcast_parameters,
                                    deepreduce_from_params)
        from deepreduce_amd.parallel import OverlappedReducer

        params = {
            "compressor": "topk", "memory": "residual",
            "communicator": "allgather", "compress_ratio": 0.05,
            "deepreduce": "index", "index": "bloom", "policy": "leftmost",
        }

        def make_model():
            torch.manual_seed(1)
            return nn.Sequential(nn.Linear(600, 40), nn.ReLU(), nn.Linear(40, 4))

        def batch(step):
            g = torch.Generator().manual_seed(1000 + 13 * step + rank)
            return (torch.randn(8, 600, generator=g),
                    torch.randint(0, 4, (8,), generator=g))

        # path A: synchronous DistributedOptimizer
        model_a = make_model()
        broadcast_parameters(model_a)
        grc_a = deepreduce_from_params(dict(params))
        opt_a = DistributedOptimizer(torch.optim.SGD(model_a.parameters(), lr=0.1),
                                     grc_a, model_a)
        for s in range(4):
            x, y = batch(s)
            opt_a.zero_grad(set_to_none=False)
            torch.nn.functional.cross_entropy(model_a(x), y).backward()
            opt_a.step()

        # path B: hook-driven OverlappedReducer
        model_b = make_model()
        broadcast_parameters(model_b)
        grc_b = deepreduce_from_params(dict(params))
        reducer = OverlappedReducer(model_b, grc_b)
        sgd_b = torch.optim.SGD(model_b.parameters(), lr=0.1)
        for s in range(4):
            x, y = batch(s)
            sgd_b.zero_grad(set_to_none=False)
            reducer.zero_wire_counter()
            torch.nn.functional.cross_entropy(model_b(x), y).backward()
            reducer.finalize()
            sgd_b.step()

        ok = all(
            torch.allclose(pa, pb, atol=1e-5)
            for pa, pb in zip(model_a.parameters(), model_b.parameters())
        )
        q.put((rank, bool(ok)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e!r}"))


def test_overlapped_reducer_matches_sync_optimizer():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = "29753"
    procs = [ctx.Process(target=_run_overlap_parity, args=(r, world, q)) for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok in results:
        assert ok is True, f"rank {rank}: {ok}"


def test_bench_driver_contract_2proc_cpu():
    """Run bench.py exactly as the driver does (torch.distributed.run,
    nproc=2), on CPU/gloo: the full integration path incl. rank-0 JSON."""
    import json
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
        "--master-port", "29770", os.path.join(repo, "bench.py"),
        "--gpus", "2", "--steps", "2", "--warmup", "1", "--batch", "4",
        "--model", "resnet20", "--device", "cpu",
    ]
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                         cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2 and d["metric"] == "images/sec"
    assert d["config"]["parallelism"] == "dp2"
    assert 0 < d["config"]["rel_volume"] < 0.2


def _run_threshold_ragged(rank, world, q):
    try:
        _init(rank, world)
        from deepreduce_amd import deepreduce_from_params

        grc = deepreduce_from_params({
            "compressor": "threshold", "threshold": 0.8,
            "memory": "residual", "communicator": "allgather",
        })
        torch.manual_seed(800 + rank)  # different sparsity per rank -> ragged
        g = torch.randn(6000)
        out = grc.step_many([("w", g.clone())])[0]
        outs = [torch.empty_like(out) for _ in range(world)]
        dist.all_gather(outs, out)
        ok = all(torch.allclose(outs[0], o, atol=1e-6) for o in outs)
        q.put((rank, bool(ok)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, f"ERROR: {e!r}"))


def test_threshold_ragged_allgather():
    """Per-rank payload sizes differ (threshold sparsifier): the two-phase
    ragged exchange must still produce identical averages everywhere."""
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    os.environ["MASTER_PORT"] = "29781"
    procs = [ctx.Process(target=_run_threshold_ragged, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for rank, ok in results:
        assert ok is True, f"rank {rank}: {ok}"
