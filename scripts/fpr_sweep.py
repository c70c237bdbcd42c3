#!/usr/bin/env python3
"""Bloom FPR sweep — paper Figs. 15/17 equivalent (BASELINE.md row:
"rel. volume 0.013-0.020 at top-1 0.88-0.91 across FPR in {0.001..0.02}",
ResNet-20 top-k 1%).

For each FPR, accounts wire bytes on real ResNet-20 gradients and the
recall of the recovered index set (leftmost policy: higher FPR -> smaller
filter but more false positives displacing true indices).

Usage: python scripts/fpr_sweep.py [--out docs/FPR_SWEEP.md]
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepreduce_amd import deepreduce_from_params
from deepreduce_amd.models import resnet20


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default=None)
    args = ap.parse_args()

    torch.manual_seed(0)
    model = resnet20()
    x = torch.randn(32, 3, 32, 32)
    y = torch.randint(0, 10, (32,))
    torch.nn.functional.cross_entropy(model(x), y).backward()
    grads = [(n, p.grad.detach().clone()) for n, p in model.named_parameters()]
    dense_bytes = sum(g.numel() * 4 for _, g in grads)

    rows = []
    for fpr in [0.001, 0.002, 0.005, 0.01, 0.02, 0.05, 0.1]:
        params = {"compressor": "topk", "memory": "residual",
                  "communicator": "allgather", "compress_ratio": 0.01,
                  "deepreduce": "index", "index": "bloom",
                  "policy": "leftmost", "fpr": fpr, "small_dense": False}
        grc = deepreduce_from_params(params)
        grc.step_many([(n, g.clone()) for n, g in grads])
        wire = grc.last_wire_bytes
        # recall: fraction of true top-k indices recovered by the policy
        hits = total = 0
        comp = grc.compressor
        for n, g in grads:
            if g.numel() <= 1000:
                continue
            (vals, idxs), ctx = comp.sparsifier.compress(g.reshape(-1), n)
            p2 = dict(params)
            v, bits, s = comp.idx_codec.compress(
                (vals, idxs, g.reshape(-1).size()), p2)
            _, rec, _ = comp.idx_codec.decompress((v, bits, s), p2)
            true = set(idxs.tolist())
            hits += len(true & set(rec.tolist()))
            total += len(true)
        rows.append((fpr, wire, wire / dense_bytes, hits / total))
        print(f"fpr={fpr:<6} wire={wire:>8,}  rel={wire/dense_bytes:.5f}  "
              f"recall={hits/total:.4f}", flush=True)

    lines = [
        "# Bloom FPR sweep (ResNet-20 top-k 1%, leftmost policy)",
        "",
        f"Dense gradient {dense_bytes:,} bytes.  Reference row (BASELINE.md):"
        " rel. volume 0.013-0.020 across FPR in {0.001..0.02} on 8xV100.",
        "",
        "| FPR | wire bytes/step | rel. volume | index recall |",
        "|---:|---:|---:|---:|",
    ]
    for fpr, wire, rel, recall in rows:
        lines.append(f"| {fpr} | {wire:,} | {rel:.5f} | {recall:.4f} |")
    text = "\n".join(lines) + "\n"
    print(text)
    if args.out:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
