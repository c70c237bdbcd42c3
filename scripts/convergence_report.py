#!/usr/bin/env python3
"""Convergence comparison: dense vs DR-compressed training.

The reference validated compression by final accuracy (WANDB runs,
README.md:53; paper Figs. 15-17 show DR variants matching or beating
plain Top-r).  Offline equivalent: train an MLP on a fixed synthetic
classification task (separable clusters + noise) under each compression
config with identical seeds, and report final train/test accuracy.

Usage: python scripts/convergence_report.py [--steps 300] [--out docs/CONVERGENCE.md]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepreduce_amd import DistributedOptimizer, deepreduce_from_params

CONFIGS = [
    ("dense (no compression)", {"compressor": "none", "memory": "none",
                                "communicator": "allreduce"}),
    ("Top-r 1%", {}),
    ("DR-BF leftmost", {"deepreduce": "index", "index": "bloom",
                        "policy": "leftmost"}),
    ("DR-BF fp16 wire", {"deepreduce": "index", "index": "bloom",
                         "policy": "leftmost", "wire_dtype": "fp16"}),
    ("DR-BF-P0", {"deepreduce": "index", "index": "bloom", "policy": "p0"}),
    ("DR-FitPoly", {"deepreduce": "value", "value": "polyfit"}),
    ("DR-QSGD", {"deepreduce": "value", "value": "qsgd"}),
    ("DR-FitPoly+BF 'both'", {"deepreduce": "both", "value": "polyfit",
                              "index": "bloom", "policy": "leftmost"}),
]


def make_data(n, d, classes, seed, centers):
    g = torch.Generator().manual_seed(seed)
    y = torch.randint(0, classes, (n,), generator=g)
    x = centers[y] + torch.randn(n, d, generator=g)
    return x, y


def run(steps: int):
    d, classes = 64, 10
    centers = torch.randn(classes, d, generator=torch.Generator().manual_seed(5)) * 0.28
    xtr, ytr = make_data(4096, d, classes, 0, centers)
    xte, yte = make_data(1024, d, classes, 1, centers)
    rows = []
    for label, extra in CONFIGS:
        params = {"compressor": "topk", "memory": "residual",
                  "communicator": "allgather", "compress_ratio": 0.01}
        params.update(extra)
        torch.manual_seed(7)
        model = torch.nn.Sequential(
            torch.nn.Linear(d, 512), torch.nn.ReLU(),
            torch.nn.Linear(512, 512), torch.nn.ReLU(),
            torch.nn.Linear(512, classes),
        )
        grc = deepreduce_from_params(params)
        opt = DistributedOptimizer(
            torch.optim.SGD(model.parameters(), lr=0.05, momentum=0.9), grc, model)
        g = torch.Generator().manual_seed(99)
        for s in range(steps):
            idx = torch.randint(0, xtr.shape[0], (128,), generator=g)
            opt.zero_grad(set_to_none=False)
            loss = torch.nn.functional.cross_entropy(model(xtr[idx]), ytr[idx])
            loss.backward()
            opt.step()
        with torch.no_grad():
            tr_acc = (model(xtr).argmax(1) == ytr).float().mean().item()
            te_acc = (model(xte).argmax(1) == yte).float().mean().item()
        rows.append((label, float(loss), tr_acc, te_acc))
        print(f"{label:24s} loss {loss:.4f}  train {tr_acc:.4f}  test {te_acc:.4f}",
              flush=True)
    return rows


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=150)
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    rows = run(args.steps)
    dense_te = rows[0][3]
    lines = [
        "# Convergence: dense vs DR-compressed training",
        "",
        f"MLP on a fixed synthetic 10-class task, {args.steps} steps, "
        "identical seeds/batches across configs; Top-r 1% + residual error "
        "feedback.  (The paper's Figs. 15-17 show DR variants matching "
        "plain Top-r's accuracy at a third of the bytes.)",
        "",
        "| Config | final loss | train acc | test acc | vs dense |",
        "|---|---:|---:|---:|---:|",
    ]
    for label, loss, tr, te in rows:
        lines.append(f"| {label} | {loss:.4f} | {tr:.4f} | {te:.4f} | "
                     f"{te - dense_te:+.4f} |")
    text = "\n".join(lines) + "\n"
    print(text)
    if args.out:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
