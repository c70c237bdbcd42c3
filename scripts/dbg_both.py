#!/usr/bin/env python3
"""Repro harness for the batched 'both' pipeline on real model shapes.

Runs BothPipeline vs the generic per-tensor path on the large-tensor
numels of a real model (resnet50 by default), eager, step by step.

    python scripts/dbg_both.py [resnet50|bert] [ratio]
"""
from __future__ import annotations

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepreduce_amd import deepreduce_from_params
from deepreduce_amd.models import registry


def main():
    model_name = sys.argv[1] if len(sys.argv) > 1 else "resnet50"
    ratio = float(sys.argv[2]) if len(sys.argv) > 2 else (
        0.001 if model_name == "bert" else 0.01)
    dev = torch.device("cuda:0")
    model = registry[model_name]()
    numels = [p.numel() for p in model.parameters() if p.numel() > 1000]
    names = [n for n, p in model.named_parameters() if p.numel() > 1000]
    del model
    print(f"{model_name}: {len(numels)} large tensors, "
          f"min {min(numels)}, max {max(numels)}, ratio {ratio}", flush=True)

    params = {
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": ratio,
        "deepreduce": "both", "index": "bloom", "policy": "leftmost",
        "value": "polyfit",
    }
    grc_b = deepreduce_from_params(dict(params))
    grc_p = deepreduce_from_params(dict(params))
    torch.manual_seed(5)
    base = [torch.randn(n, device=dev) for n in numels]
    for step in range(3):
        tensors = [(nm, t * (1.0 + 0.1 * step)) for nm, t in zip(names, base)]
        fused = grc_b.step_many([(n, t.clone()) for n, t in tensors])
        torch.cuda.synchronize()
        print(f"step {step}: batched ok, wire={grc_b.last_wire_bytes}", flush=True)
        worst = 0.0
        for (n, t), f in zip(tensors, fused):
            l = grc_p.step(t.clone(), n)
            d = float((f - l).abs().max())
            worst = max(worst, d)
            if d > 1e-4:
                print(f"  MISMATCH {n} (numel {t.numel()}): {d}", flush=True)
        torch.cuda.synchronize()
        print(f"step {step}: per-tensor ok, worst diff {worst:.3e}", flush=True)
    print("DONE", flush=True)


if __name__ == "__main__":
    main()
