#!/usr/bin/env python3
"""Two-rank RCCL validation on co-located GPUs.

Launched by tests/test_gpu_multirank.py as:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 --master-port <P> scripts/two_rank_check.py

Both ranks sit on cuda:0 (modulo device mapping) so a single-GPU box can
exercise everything the 8-GPU SCALE run will hit: all_gather_into_tensor
under the nccl(=RCCL) backend, the two-phase ragged path, multi-rank
batched_decode_sum on real gathered wires, the overlap reducer's async
collectives, and fp16 wire payloads.

Checks per config:
  * cross-rank identity: reduced grads + final params bit-identical on both
    ranks (decode re-derives indices independently per rank — any hash/
    policy/segment divergence corrupts this);
  * independent own-decode cross-check (cfg index): the exchanged result
    must equal mean_r(decompress_r(compress_r(compensated_r))) where each
    term is recomputed OUTSIDE the communicator on fresh wrapper instances;
  * wire accounting: last_wire_bytes > 0 and rel volume sane.

Prints "CFG <name> OK" per config and "TWO_RANK_ALL_OK" at the end (rank 0).
"""
from __future__ import annotations

import os
import sys

import torch
import torch.distributed as dist
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def log(rank, msg):
    if rank == 0:
        print(msg, flush=True)


def make_model(seed=11):
    torch.manual_seed(seed)
    # two >1000-element tensors (batched-pipeline path) + small dense ones
    return nn.Sequential(
        nn.Conv2d(3, 64, 3, padding=1), nn.ReLU(),
        nn.Conv2d(64, 128, 3, padding=1), nn.ReLU(),
        nn.AdaptiveAvgPool2d(1), nn.Flatten(), nn.Linear(128, 10),
    )


def batch(rank, step, device):
    g = torch.Generator().manual_seed(500 + 31 * step + rank)
    x = torch.randn(8, 3, 32, 32, generator=g).to(device)
    y = torch.randint(0, 10, (8,), generator=g).to(device)
    return x, y


def assert_cross_rank_identical(t: torch.Tensor, what: str):
    world = dist.get_world_size()
    flat = t.reshape(-1).float().contiguous()
    gathered = [torch.empty_like(flat) for _ in range(world)]
    dist.all_gather(gathered, flat)
    for r in range(1, world):
        if not torch.equal(gathered[0], gathered[r]):
            diff = (gathered[0] - gathered[r]).abs().max().item()
            raise AssertionError(
                f"{what}: rank0 vs rank{r} differ (max abs {diff})")


def run_config(name, params, device, rank, steps=3, overlap=False,
               check_own_decode=False):
    from deepreduce_amd import (DistributedOptimizer, broadcast_parameters,
                                deepreduce_from_params, grace_from_params)

    model = make_model().to(device)
    broadcast_parameters(model)
    grc = deepreduce_from_params(dict(params))
    sgd = torch.optim.SGD(model.parameters(), lr=0.05)
    reducer = None
    if overlap:
        from deepreduce_amd.parallel import OverlappedReducer

        reducer = OverlappedReducer(model, grc)
        opt = sgd
    else:
        opt = DistributedOptimizer(sgd, grc, model)

    for s in range(steps):
        (opt if not overlap else sgd).zero_grad(set_to_none=False)
        x, y = batch(rank, s, device)
        torch.nn.functional.cross_entropy(model(x), y).backward()

        expected = None
        if check_own_decode and s == 0:
            # independent recompute: own-decode per rank, averaged over ranks,
            # must equal what the communicator returns.  Fresh wrapper (no
            # shared cache), same residual state (compensate does not mutate).
            sim = deepreduce_from_params(dict(params))
            world = dist.get_world_size()
            expected = {}
            for n, p in model.named_parameters():
                g32 = p.grad.data.float()
                if g32.numel() <= 1000:
                    own = g32.clone()   # small tensors travel dense & exact
                else:
                    t = grc.memory.compensate(g32, n)
                    tc, ctx = sim.compressor.compress(t, n)
                    od = getattr(sim.compressor, "decompress_own", None)
                    own = (od(tc, ctx, n) if od is not None
                           else sim.compressor.decompress(tc, ctx)).view_as(g32)
                dist.all_reduce(own)
                expected[n] = own / world

        if overlap:
            reducer.finalize()
        else:
            opt._exchange()

        if expected is not None:
            for n, p in model.named_parameters():
                got = p.grad.data.float()
                if not torch.allclose(got, expected[n], atol=1e-5):
                    diff = (got - expected[n]).abs().max().item()
                    raise AssertionError(
                        f"{name}: own-decode cross-check failed for {n} "
                        f"(max abs {diff})")

        for n, p in model.named_parameters():
            assert_cross_rank_identical(p.grad.data, f"{name} step{s} grad {n}")
        sgd.step()

    wire = (reducer.last_wire_bytes if overlap
            else getattr(opt, "last_wire_bytes", 0)
            or getattr(grc, "last_wire_bytes", 0))
    n_params = sum(p.numel() for p in model.parameters())
    assert wire > 0, f"{name}: wire accounting is zero"
    rel = wire / (n_params * 4)
    assert rel < 0.9, f"{name}: rel volume {rel:.3f} suspicious"
    for n, p in model.named_parameters():
        assert_cross_rank_identical(p.data, f"{name} final param {n}")
    log(rank, f"CFG {name} OK (wire={wire}B rel={rel:.4f})")


def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    if torch.cuda.is_available():
        ndev = max(1, torch.cuda.device_count())
        device = torch.device(f"cuda:{rank % ndev}")
        torch.cuda.set_device(device)
        if world <= ndev:
            dist.init_process_group("nccl")
        else:
            # RCCL refuses co-located ranks ("Duplicate GPU detected"):
            # full GPU pipeline per rank, CPU-staged gloo transport
            from deepreduce_amd.testing import stage_collectives_via_cpu

            dist.init_process_group("gloo")
            stage_collectives_via_cpu()
            log(rank, "transport: CPU-staged gloo (ranks > GPUs)")
    else:  # logic dry-run on CPU (gloo)
        device = torch.device("cpu")
        dist.init_process_group("gloo")

    base = {"compressor": "topk", "memory": "residual",
            "communicator": "allgather", "compress_ratio": 0.01}

    run_config("index_bloom_leftmost",
               {**base, "deepreduce": "index", "index": "bloom",
                "policy": "leftmost"},
               device, rank, check_own_decode=True)
    run_config("both_bloom_polyfit",
               {**base, "deepreduce": "both", "index": "bloom",
                "value": "polyfit", "policy": "leftmost"},
               device, rank)
    run_config("threshold_ragged",
               {"compressor": "threshold", "memory": "residual",
                "communicator": "allgather", "threshold": 0.01},
               device, rank)
    run_config("fp16_wire",
               {**base, "deepreduce": "index", "index": "bloom",
                "policy": "leftmost", "wire_dtype": "fp16"},
               device, rank)
    run_config("overlap_reducer",
               {**base, "deepreduce": "index", "index": "bloom",
                "policy": "leftmost"},
               device, rank, overlap=True, check_own_decode=False)
    if os.environ.get("DEEPREDUCE_GRAPH_DIST") == "1":
        run_config("graph_dist_capture",
                   {**base, "deepreduce": "index", "index": "bloom",
                    "policy": "leftmost"},
                   device, rank, steps=8)

    dist.barrier()
    log(rank, "TWO_RANK_ALL_OK")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
