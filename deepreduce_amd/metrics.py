"""Observability: per-rank/step/tensor compression statistics.

Native replacement for the reference's C++ file-dump logging
(/root/reference/tensorflow/compression_utils.hpp:96-217 — fpr.txt /
policy_errors.txt / stats.txt keyed by rank/step/gradient — and
tensorflow/logger.cc's values/coefficients CSV op), re-designed as a
lightweight in-process collector:

  * `StatsLogger` accumulates per-tensor records (wire bytes, dense bytes,
    measured false positives, policy errors, stage timings) and can dump
    them as CSV/JSONL per rank.
  * measurement helpers compute the same ground-truth quantities the C++
    ops logged: measured FPR (Compute_False_Positives equivalent) and
    policy errors vs the true index set
    (bloom_filter_compression.cc:144-151, policies.hpp:32-41).
"""
from __future__ import annotations

import json
import os
import time
from dataclasses import asdict, dataclass, field

import torch

from .helper import rank

__all__ = ["StatsLogger", "measured_fpr", "policy_errors", "global_logger"]


def measured_fpr(packed, m, num_hash, universe, true_idxs) -> float:
    """Fraction of non-member universe items that query positive
    (Compute_False_Positives equivalent)."""
    from .ops import bloom_query_positives

    pos = bloom_query_positives(packed, m, num_hash, universe)
    true = set(true_idxs.cpu().tolist())
    fp = len(set(pos.cpu().tolist()) - true)
    denom = universe - len(true)
    return fp / denom if denom else 0.0


def policy_errors(selected_idxs, true_idxs) -> int:
    """How many selected indices are not true indices (policies.hpp:32-41)."""
    true = set(true_idxs.cpu().tolist())
    return sum(1 for i in selected_idxs.cpu().tolist() if i not in true)


@dataclass
class Record:
    step: int
    tensor: str
    rank: int
    wire_bytes: int
    dense_bytes: int
    extra: dict = field(default_factory=dict)
    t: float = field(default_factory=time.time)


class StatsLogger:
    """Accumulates compression stats; dump-to-file per rank on request.

    Enabled via params['log_stats'] = directory, or construct directly.
    `frequency` mirrors the C++ ops' logging-frequency attr
    (bloom_filter_compression.cc:28-30).
    """

    def __init__(self, out_dir: str | None = None, frequency: int = 1, verbosity: int = 1):
        self.out_dir = out_dir
        self.frequency = max(1, frequency)
        self.verbosity = verbosity
        self.records: list[Record] = []
        self.step = 0

    def tick(self):
        self.step += 1

    def log(self, tensor: str, wire_bytes: int, dense_bytes: int, **extra):
        if self.step % self.frequency:
            return
        self.records.append(
            Record(self.step, tensor, rank(), int(wire_bytes), int(dense_bytes), extra)
        )

    def rel_volume(self) -> float:
        """Aggregate transmitted / dense bytes over all records."""
        wire = sum(r.wire_bytes for r in self.records)
        dense = sum(r.dense_bytes for r in self.records)
        return wire / dense if dense else 0.0

    def dump(self, path: str | None = None):
        out_dir = path or self.out_dir
        if not out_dir:
            return None
        os.makedirs(out_dir, exist_ok=True)
        fn = os.path.join(out_dir, f"stats_rank{rank()}.jsonl")
        with open(fn, "w") as f:
            for r in self.records:
                f.write(json.dumps(asdict(r)) + "\n")
        return fn

    def dump_values(self, tensor_name: str, values: torch.Tensor, coefficients=None):
        """Logger-op equivalent: CSV dump of values (+ fitted coefficients)
        for curve-fit research (logger.cc:37-52)."""
        if not self.out_dir:
            return None
        os.makedirs(self.out_dir, exist_ok=True)
        fn = os.path.join(
            self.out_dir, f"values_r{rank()}_s{self.step}_{tensor_name.replace('/', '_')}.csv"
        )
        with open(fn, "w") as f:
            f.write(",".join(f"{v:.8g}" for v in values.detach().cpu().reshape(-1).tolist()))
            f.write("\n")
            if coefficients is not None:
                f.write(",".join(f"{c:.17g}" for c in coefficients.detach().cpu().reshape(-1).tolist()))
                f.write("\n")
        return fn


global_logger = StatsLogger()
