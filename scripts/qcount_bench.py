#!/usr/bin/env python3
"""A/B microbench for the bloom universe-query memory-path work
(VERDICT r1 item 3): LDS-cached vs global word-load compress query (R=1),
and interleaved vs strided multi-rank decode query (R=8).

Runs on a GPU box:  python scripts/qcount_bench.py [--iters 30]
Prints one JSON line per arm (mean ms over iters, after warmup).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepreduce_amd.ops.batched import BatchedPipeline

# ResNet-50's large-tensor layout (the flagship's actual numels, >1000 only)
RESNET50_NUMELS = [
    9408, 4096, 16384, 36864, 16384, 16384, 16384, 36864, 16384, 16384,
    36864, 16384, 32768, 131072, 32768, 65536, 73728, 32768, 32768, 32768,
    73728, 32768, 32768, 73728, 32768, 32768, 73728, 32768, 131072, 524288,
    131072, 262144, 294912, 131072, 131072, 131072, 294912, 131072, 131072,
    294912, 131072, 131072, 294912, 131072, 131072, 294912, 131072, 524288,
    2097152, 524288, 1048576, 2359296, 524288, 524288, 524288, 2359296,
    524288, 524288, 2359296, 524288, 2048000,
]


def time_op(fn, iters, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=30)
    args = ap.parse_args()
    dev = torch.device("cuda:0")
    params = {"compressor": "topk", "memory": "residual",
              "communicator": "allgather", "compress_ratio": 0.01,
              "deepreduce": "index", "index": "bloom", "policy": "leftmost"}
    names = [f"t{i}" for i in range(len(RESNET50_NUMELS))]
    bp = BatchedPipeline(names, RESNET50_NUMELS, params, dev)
    torch.manual_seed(0)
    flat = torch.randn(bp.total_values, device=dev)

    results = {}
    # --- compress query: LDS vs global ---
    saved = bp.ldsq_bytes
    ms_lds = time_op(lambda: bp.compress(flat), args.iters)
    bp.ldsq_bytes = 0
    ms_glob = time_op(lambda: bp.compress(flat), args.iters)
    bp.ldsq_bytes = saved
    results["compress_lds_ms"] = round(ms_lds, 3)
    results["compress_global_ms"] = round(ms_glob, 3)

    # --- decode R=8: interleaved vs strided ---
    wires = []
    for r in range(8):
        w, _ = bp.compress(torch.randn(bp.total_values, device=dev))
        wires.append(w)
    stacked = torch.stack(wires)
    ms_inter = time_op(lambda: bp.decode_sum(stacked), args.iters)
    saved_mw = bp.total_mw
    bp.total_mw = 0   # plain strided word-load path
    ms_strided = time_op(lambda: bp.decode_sum(stacked), args.iters)
    bp.total_mw = saved_mw
    results["decode8_interleaved_ms"] = round(ms_inter, 3)
    results["decode8_strided_ms"] = round(ms_strided, 3)

    # parity while we're here
    bp.total_mw = 0
    a = bp.decode_sum(stacked)
    bp.total_mw = saved_mw
    b = bp.decode_sum(stacked)
    results["decode_parity"] = bool(torch.equal(a, b))
    results["ldsq_bytes"] = saved
    results["qhead"] = os.environ.get("DEEPREDUCE_QHEAD", "2")
    print(json.dumps(results))


if __name__ == "__main__":
    main()
