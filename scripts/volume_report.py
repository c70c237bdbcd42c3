#!/usr/bin/env python3
"""Relative-data-volume report — the reference's headline metric.

Reproduces the paper's Table-2-style comparison (deepreduce.nips21.pdf
p.8: rel. volume of Top-r vs DR variants) on real
gradients: trains ResNet-20 on synthetic CIFAR-shaped data for a few steps
and accounts the exact wire bytes each configuration transmits, relative
to the dense float32 gradient.

Usage: python scripts/volume_report.py [--steps 3] [--ratio 0.01] [--out docs/VOLUME.md]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepreduce_amd import deepreduce_from_params
from deepreduce_amd.models import resnet20

CONFIGS = [
    ("dense (baseline)", {"compressor": "none", "memory": "none",
                          "communicator": "allreduce"}),
    ("Top-r", {}),
    ("DR-BF (leftmost)", {"deepreduce": "index", "index": "bloom",
                          "policy": "leftmost"}),
    ("DR-BF-P0", {"deepreduce": "index", "index": "bloom", "policy": "p0"}),
    ("DR-BF-P0 fpr=auto", {"deepreduce": "index", "index": "bloom",
                           "policy": "p0", "fpr": "auto"}),
    ("DR-FitPoly", {"deepreduce": "value", "value": "polyfit"}),
    ("DR-QSGD", {"deepreduce": "value", "value": "qsgd"}),
    ("DR-RLE", {"deepreduce": "index", "index": "rle"}),
    ("DR-Huffman", {"deepreduce": "index", "index": "huffman"}),
    ("DR-Gzip", {"deepreduce": "value", "value": "gzip"}),
    ("DR-PFor", {"deepreduce": "index", "index": "pfor"}),
    ("DR-QSGD+BF-P0 7bit auto", {"deepreduce": "both", "value": "qsgd",
                                 "index": "bloom", "policy": "p0",
                                 "fpr": "auto", "qsgd_pack": True,
                                 "quantum_num": 63}),
    ("DR-QSGD+BF-P0 ('both')", {"deepreduce": "both", "value": "qsgd",
                                "index": "bloom", "policy": "p0"}),
    ("DR-FitPoly+BF ('both')", {"deepreduce": "both", "value": "polyfit",
                                "index": "bloom", "policy": "leftmost"}),
]


def measure(ratio: float, steps: int):
    torch.manual_seed(0)
    model = resnet20()
    x = torch.randn(32, 3, 32, 32)
    y = torch.randint(0, 10, (32,))
    # capture real gradients from a few steps
    grads_per_step = []
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    for _ in range(steps):
        opt.zero_grad()
        torch.nn.functional.cross_entropy(model(x), y).backward()
        grads_per_step.append(
            [(n, p.grad.detach().clone()) for n, p in model.named_parameters()]
        )
        opt.step()

    dense_bytes = sum(g.numel() * 4 for _, g in grads_per_step[0])
    rows = []
    for label, extra in CONFIGS:
        params = {"compressor": "topk", "memory": "residual",
                  "communicator": "allgather", "compress_ratio": ratio,
                  # codec-pure accounting: the runtime's small-tensor dense
                  # fusion (a latency optimization) is off here so the
                  # table isolates what each CODEC transmits
                  "small_dense": False}
        params.update(extra)
        grc = deepreduce_from_params(params)
        total = 0
        for gs in grads_per_step:
            grc.step_many([(n, g.clone()) for n, g in gs])
            total += grc.last_wire_bytes
        rel = total / (dense_bytes * steps)
        rows.append((label, total // steps, rel))
    return dense_bytes, rows


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--ratio", type=float, default=0.01)
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    dense_bytes, rows = measure(args.ratio, args.steps)
    lines = [
        "# Relative data volume (ResNet-20, real gradients, "
        f"Top-r {args.ratio:.0%} + residual)",
        "",
        f"Dense float32 gradient: {dense_bytes:,} bytes/step.  "
        "Rel. volume = transmitted / dense (reference's headline metric, "
        "BASELINE.md; paper Table 2 shows Top-r(10%) 0.2033 vs "
        "DR-QSGD-BF-P0 0.0621 on an RNN).",
        "",
        "| Config | bytes/step | rel. volume | vs Top-r |",
        "|---|---:|---:|---:|",
    ]
    topr = next(r[2] for r in rows if r[0] == "Top-r")
    for label, b, rel in rows:
        lines.append(f"| {label} | {b:,} | {rel:.5f} | {rel/topr:.3f}x |")
    text = "\n".join(lines) + "\n"
    print(text)
    if args.out:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
