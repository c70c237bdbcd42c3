#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50 top-k 1% data-parallel training step.

Measures the BASELINE.json headline metric — images/sec (whole-job) and
bytes/step for ResNet-50 with top-k 1% sparsification + residual memory +
Bloom index compression over RCCL allgather — on synthetic ImageNet-shaped
data with random-init weights (no network access).

Launch contract (driver):
    python bench.py --gpus N --steps K --warmup W            # N=1
    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
        --master-addr 127.0.0.1 bench.py --gpus N ...        # N>1
Rank 0 prints ONE JSON line.
"""
from __future__ import annotations

import argparse
import json
import os
import time

import torch
import torch.distributed as dist


def get_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=5)
    # 512/GPU: MI355X has 288 GB HBM3E — big per-GPU batches amortize the
    # fixed per-step comm/compression cost (measured: 5554/6315/6687 img/s
    # at 128/256/512, gpurun_out/batch_sweep.jsonl) and raise weak-scaling
    # efficiency at N>1 (collective cost is batch-independent)
    p.add_argument("--batch", type=int, default=512, help="per-GPU batch size")
    p.add_argument("--model", default="resnet50",
                   choices=["resnet50", "resnet20", "ncf", "bert", "mobilenet", "rnn"])
    p.add_argument("--compress-ratio", type=float, default=0.01)
    p.add_argument("--deepreduce", default="index", choices=["none", "value", "index", "both", "dense"])
    p.add_argument("--value", default="polyfit")
    p.add_argument("--index", default="bloom")
    p.add_argument("--policy", default="leftmost")
    p.add_argument("--wire-dtype", default="fp32", choices=["fp32", "fp16"])
    p.add_argument("--device", default=None)
    p.add_argument("--overlap", action="store_true",
                   help="hook-driven compression overlapped with backward + "
                        "fused dense exchange of small tensors")
    return p.parse_args()


def build_grc(args):
    from deepreduce_amd import deepreduce_from_params

    if args.deepreduce == "dense":
        params = {"compressor": "none", "memory": "none", "communicator": "allreduce"}
    else:
        params = {
            "compressor": "topk",
            "memory": "residual",
            "communicator": "allgather",
            "compress_ratio": args.compress_ratio,
            "deepreduce": None if args.deepreduce == "none" else args.deepreduce,
            "value": args.value,
            "index": args.index,
            "policy": args.policy,
            "wire_dtype": args.wire_dtype,
        }
    return deepreduce_from_params(params), params


def make_batch(args, device):
    if args.model in ("resnet50", "resnet20", "mobilenet"):
        res = 224 if args.model == "resnet50" else 32
        ncls = 1000 if args.model == "resnet50" else 10
        x = torch.randn(args.batch, 3, res, res, device=device)
        y = torch.randint(0, ncls, (args.batch,), device=device)
        return (x,), y
    if args.model == "rnn":
        ids = torch.randint(0, 10_004, (args.batch, 20), device=device)
        y = torch.randint(0, 10_004, (args.batch, 20), device=device)
        return (ids,), y
    if args.model == "ncf":
        u = torch.randint(0, 138_493, (args.batch,), device=device)
        i = torch.randint(0, 26_744, (args.batch,), device=device)
        y = torch.rand(args.batch, device=device).round()
        return (u, i), y
    if args.model == "bert":
        ids = torch.randint(0, 30522, (args.batch, 128), device=device)
        y = torch.randint(0, 30522, (args.batch, 128), device=device)
        return (ids,), y
    raise ValueError(args.model)


def main():
    args = get_args()
    env_world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    use_cuda = torch.cuda.is_available() and args.device != "cpu"
    # let MIOpen pick tuned conv algorithms during warmup (fixed shapes)
    torch.backends.cudnn.benchmark = True
    # modulo wrap: ranks beyond the device count co-locate (RCCL supports
    # multiple ranks per GPU), so `--gpus 2` is testable on a 1-GPU box
    ndev = max(1, torch.cuda.device_count()) if use_cuda else 1
    device = torch.device(f"cuda:{local_rank % ndev}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)

    if env_world > 1:
        if use_cuda and env_world <= torch.cuda.device_count():
            dist.init_process_group(backend="nccl")  # = RCCL on ROCm
        elif use_cuda:
            # more ranks than GPUs: RCCL refuses co-located ranks, so run
            # the GPU pipeline with CPU-staged gloo transport (launch-
            # qualification on small boxes; not a performance mode)
            from deepreduce_amd.testing import stage_collectives_via_cpu

            dist.init_process_group(backend="gloo")
            stage_collectives_via_cpu()
        else:
            dist.init_process_group(backend="gloo")
    world = env_world

    from deepreduce_amd.models import registry
    from deepreduce_amd import DistributedOptimizer, broadcast_parameters

    torch.manual_seed(1234 + rank)
    model = registry[args.model]().to(device)
    if args.model in ("resnet50", "resnet20", "mobilenet") and use_cuda:
        model = model.to(memory_format=torch.channels_last)
    broadcast_parameters(model)

    grc, params = build_grc(args)
    sgd = torch.optim.SGD(model.parameters(), lr=0.1, momentum=0.9, weight_decay=1e-4)
    reducer = None
    if args.overlap and args.deepreduce != "dense":
        from deepreduce_amd.parallel import OverlappedReducer

        reducer = OverlappedReducer(model, grc)
        opt = sgd  # reducer handles the exchange; plain SGD steps

        class _Opt:
            last_wire_bytes = 0

            def zero_grad(self, set_to_none=False):
                sgd.zero_grad(set_to_none=set_to_none)
                reducer.zero_wire_counter()

            def step(self):
                reducer.finalize()
                self.last_wire_bytes = reducer.last_wire_bytes
                sgd.step()

        opt = _Opt()
    else:
        opt = DistributedOptimizer(sgd, grc, model)
    loss_fn = (
        torch.nn.BCEWithLogitsLoss() if args.model == "ncf" else torch.nn.CrossEntropyLoss()
    )

    inputs, target = make_batch(args, device)
    if args.model in ("resnet50", "resnet20", "mobilenet") and use_cuda:
        inputs = (inputs[0].to(memory_format=torch.channels_last),)

    amp_dtype = torch.bfloat16
    autocast = torch.autocast(device_type="cuda", dtype=amp_dtype, enabled=use_cuda)

    def step():
        opt.zero_grad(set_to_none=False)
        with autocast:
            out = model(*inputs)
            if args.model in ("bert", "rnn"):
                loss = loss_fn(out.float().flatten(0, 1), target.flatten())
            else:
                loss = loss_fn(out.float(), target.float() if args.model == "ncf" else target)
        loss.backward()
        opt.step()
        return loss

    for _ in range(args.warmup):
        step()

    wire_bytes = opt.last_wire_bytes

    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64, device=device if use_cuda else "cpu")
    if world > 1:
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    if args.model in ("resnet50", "resnet20", "mobilenet"):
        unit, metric = "images/sec", "images/sec"
        value = args.batch * world * args.steps / elapsed
    elif args.model == "ncf":
        unit, metric = "samples/sec", "samples/sec"
        value = args.batch * world * args.steps / elapsed
    else:
        seq = 128 if args.model == "bert" else 20
        unit, metric = "tokens/sec", "tokens/sec"
        value = args.batch * seq * world * args.steps / elapsed

    n_params = sum(p.numel() for p in model.parameters())
    dense_bytes = n_params * 4
    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": metric,
                    "value": round(value, 2),
                    "unit": unit,
                    "n_gpus": world,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": round(ms_per_step, 3),
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": None,
                    "dtype": "bf16",
                    "data": "synthetic",
                    "config": {
                        "model": args.model,
                        "global_batch": args.batch * world,
                        "seq_len": {"bert": 128, "rnn": 20}.get(args.model),
                        "parallelism": f"dp{world}",
                        "deepreduce": args.deepreduce,
                        "compress_ratio": args.compress_ratio,
                        "bytes_per_step_per_rank": wire_bytes,
                        "dense_bytes_per_step": dense_bytes,
                        "rel_volume": round(wire_bytes / dense_bytes, 5) if dense_bytes else None,
                    },
                }
            )
        )
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
