"""Factories: grace_from_params / deepreduce_from_params.

The flat params-dict contract of the reference works unchanged
(/root/reference/README.md:31-37, pytorch/deepreduce.py:28-48):

    params = {'compressor': 'topk', 'memory': 'residual',
              'communicator': 'allgather', 'compress_ratio': 0.01,
              'deepreduce': 'index', 'index': 'bloom'}
    grc = grace_from_params(params)          # GRACE-equivalent instance
    grc.compressor = DeepReduce(grc.compressor, params)   # or use
    grc = deepreduce_from_params(params)     # ... the one-call form

`grc` is a Communicator instance: grc.step(grad, name) runs
compensate -> compress -> residual update -> collective -> average.

No precomputed hash table is needed (the reference loads a ~1 GB .pt file
at pytorch/deepreduce.py:43; hashing is in-kernel here) — a 'hash_table'
key is accepted and ignored for compatibility.
"""
from __future__ import annotations

from .communicator import communicator_registry
from .compressors import sparsifier_registry
from .memory import memory_registry
from .wrappers import deepreduce_wrapper

_SPARSIFIER_KWARGS = {
    "topk": ("compress_ratio",),
    "randomk": ("compress_ratio",),
    "threshold": ("threshold",),
    "none": (),
}


def grace_from_params(params: dict):
    from .params import validate

    validate(params)
    comp_name = params.get("compressor", "topk")
    mem_name = params.get("memory", "none")
    comm_name = params.get("communicator", "allgather")

    cls = sparsifier_registry[comp_name]
    kwargs = {k: params[k] for k in _SPARSIFIER_KWARGS.get(comp_name, ()) if k in params}
    compressor = cls(**kwargs)
    memory = memory_registry[mem_name]()
    communicator = communicator_registry[comm_name](compressor, memory)
    communicator.params = params  # runtime knobs (small_dense, ...)
    return communicator


def deepreduce_from_params(params: dict):
    grc = grace_from_params(params)
    mode = params.get("deepreduce", None)  # None | 'value' | 'index' | 'both'
    if mode:
        grc.compressor = deepreduce_wrapper[mode](grc.compressor, params)
    return grc
