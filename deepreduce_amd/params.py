"""Typed validation for the flat params dict (same contract
as the reference's `--grace_config` literal — /root/reference/README.md:31-37
— with schema checking behind it).

`validate(params)` raises ValueError on unknown keys or ill-typed values,
returning the dict unchanged otherwise.  Keys the reference accepted but
this build handles differently (e.g. `hash_table`: hashing is in-kernel)
are allowed and ignored.
"""
from __future__ import annotations

from .codecs import compressor as codec_registry
from .communicator import communicator_registry
from .compressors import sparsifier_registry
from .memory import memory_registry

_SCHEMA = {
    "compressor": (str, sorted(sparsifier_registry)),
    "memory": (str, sorted(memory_registry)),
    "communicator": (str, sorted(communicator_registry)),
    "deepreduce": ((str, type(None)), [None, "value", "index", "both"]),
    "value": (str, sorted(codec_registry)),
    "index": (str, sorted(codec_registry)),
    "policy": (str, ["leftmost", "random", "p0", "conflict_sets"]),
    "compress_ratio": ((int, float), None),
    "threshold": ((int, float), None),
    "fpr": ((int, float, str), None),  # number or "auto"
    "sort": (bool, None),
    "poly_degree": (int, None),
    "num_segments": (int, None),
    "quantum_num": (int, None),
    "bucket_size": (int, None),
    "qsgd_pack": (bool, None),
    "policy_seed": (int, None),
    "pack_mapping": (bool, None),
    "fp_aware": (bool, None),
    "wire_dtype": (str, ["fp32", "fp16"]),
    "small_dense": (bool, None),
    "micro-benchmark": (bool, None),
    "log_stats": (str, None),
    # accepted for reference compatibility, unused here:
    "hash_table": (object, None),   # in-kernel hashing replaces the table
    "dense_tensor": (object, None),  # internal side-channel
}


def validate(params: dict) -> dict:
    if not isinstance(params, dict):
        raise ValueError(f"params must be a dict, got {type(params).__name__}")
    for key, value in params.items():
        if key.startswith("_"):
            continue  # internal side-channels
        spec = _SCHEMA.get(key)
        if spec is None:
            raise ValueError(
                f"unknown params key {key!r} (known: {sorted(_SCHEMA)})")
        typ, allowed = spec
        if typ is not object and not isinstance(value, typ):
            raise ValueError(
                f"params[{key!r}] expects {typ}, got {type(value).__name__}")
        if allowed is not None and value not in allowed:
            raise ValueError(
                f"params[{key!r}] = {value!r} not in {allowed}")
    ratio = params.get("compress_ratio")
    if ratio is not None and not (0 < ratio <= 1):
        raise ValueError(f"compress_ratio must be in (0, 1], got {ratio}")
    fpr = params.get("fpr")
    if isinstance(fpr, str) and fpr != "auto":
        raise ValueError(f"fpr must be a number in (0, 1) or 'auto', got {fpr!r}")
    if isinstance(fpr, (int, float)) and not (0 < fpr < 1):
        raise ValueError(f"fpr must be in (0, 1), got {fpr}")
    if params.get("policy") == "conflict_sets":
        # The conflict-sets policy (reference policies.hpp:43-146 semantics)
        # is a HOST-side sequential round-robin: inherently unvectorizable
        # (each pick depends on every prior pick through the shared LCG and
        # set-emptying), so it runs as python over the positive set with a
        # device sync per tensor.  Correct and deterministic, but not a
        # hot-path policy on MI355X — use 'leftmost' (sync-free fused
        # kernel) or 'p0' for production runs.
        import warnings

        warnings.warn(
            "policy='conflict_sets' is a debug/compatibility policy that "
            "runs on the host (one device sync per tensor per step); use "
            "'leftmost' or 'p0' on the hot path",
            RuntimeWarning, stacklevel=3)
    return params
