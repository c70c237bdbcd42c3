#!/usr/bin/env python3
"""Per-codec encode+decode micro-benchmark — paper Fig. 8b equivalent
(conv gradient d=36,864, Top-r 1%; the paper quotes <19 ms absolute for
all DR variants on a T4 and ~380x over SKCompress).

Times codec.compress+decompress (GPU-synchronized) over the codec layer
only — no model, no collective.

Usage: python scripts/codec_bench.py [--size 36864] [--iters 50] [--out docs/CODEC_BENCH.md]
"""
from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepreduce_amd.codecs import compressor
from deepreduce_amd.ops import topk_select

CONFIGS = [
    ("bloom (leftmost)", "bloom", {"policy": "leftmost"}, True),
    ("bloom (p0)", "bloom", {"policy": "p0"}, True),
    ("polyfit", "polyfit", {"poly_degree": 5}, False),
    ("polyseg", "polyseg", {"poly_degree": 5}, False),
    ("qsgd", "qsgd", {"quantum_num": 127, "bucket_size": 512}, False),
    ("doubleexp", "doubleexp", {}, False),
    ("rle", "rle", {}, False),
    ("pfor", "pfor", {}, False),
    ("gzip", "gzip", {}, False),
    ("huffman", "huffman", {}, False),
]


def sync(dev):
    if dev.type == "cuda":
        torch.cuda.synchronize()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--size", type=int, default=36_864)
    ap.add_argument("--ratio", type=float, default=0.01)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    dev = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    torch.manual_seed(0)
    t = torch.randn(args.size, device=dev)
    k = max(1, int(round(args.size * args.ratio)))
    vals, idxs = topk_select(t, k)
    shape = t.size()

    rows = []
    for label, name, extra, needs_dense in CONFIGS:
        codec = compressor[name]
        params = dict(extra)
        if needs_dense:
            params["dense_tensor"] = t

        def enc():
            return codec.compress((vals.clone(), idxs.clone(), shape), params)

        payload = enc()
        for _ in range(5):
            enc()
            codec.decompress(tuple(x.clone() if torch.is_tensor(x) else x
                                   for x in payload), params)
        sync(dev)
        t0 = time.perf_counter()
        for _ in range(args.iters):
            enc()
        sync(dev)
        t_enc = (time.perf_counter() - t0) / args.iters * 1e3
        sync(dev)
        t0 = time.perf_counter()
        for _ in range(args.iters):
            codec.decompress(tuple(x.clone() if torch.is_tensor(x) else x
                                   for x in payload), params)
        sync(dev)
        t_dec = (time.perf_counter() - t0) / args.iters * 1e3
        wire = sum(x.numel() * x.element_size() for x in payload[:2]
                   if torch.is_tensor(x))
        rows.append((label, t_enc, t_dec, wire))
        print(f"{label:18s} enc {t_enc:8.3f} ms  dec {t_dec:8.3f} ms  "
              f"wire {wire:,} B", flush=True)

    lines = [
        "# Codec encode/decode micro-benchmark",
        "",
        f"Gradient d={args.size:,}, Top-r {args.ratio:.0%} (k={k}), "
        f"device {dev.type}, {args.iters} iters.  Paper Fig. 8b quotes "
        "<19 ms absolute encode+decode for all DR variants on a T4 at this "
        "shape.",
        "",
        "| codec | encode ms | decode ms | wire bytes |",
        "|---|---:|---:|---:|",
    ]
    for label, te, td, w in rows:
        lines.append(f"| {label} | {te:.3f} | {td:.3f} | {w:,} |")
    text = "\n".join(lines) + "\n"
    print(text)
    if args.out:
        os.makedirs(os.path.dirname(args.out), exist_ok=True)
        with open(args.out, "w") as f:
            f.write(text)


if __name__ == "__main__":
    main()
