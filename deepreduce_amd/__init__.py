"""deepreduce_amd — MI355X-native sparse-gradient communication framework.

A from-scratch build with the capabilities of DeepReduce
(hangxu0304/DeepReduce) plus the GRACE host layer it depends on, redesigned
for MI355X: hand-written HIP/CDNA4 kernels for every hot op (in-register
MurmurHash3 Bloom insert/query, fused top-k/residual, bucketed QSGD,
batched on-device polyfit), RCCL collectives over xGMI with fused uint8
payload buffers, and a torch.distributed-native runtime.

Public API mirrors the reference contract (README.md:31-48):
    grace_from_params(params) / deepreduce_from_params(params)
    ValueCompressor / IndexCompressor / DeepReduce wrappers
    codec registry `compressor` (bloom, polyfit, qsgd, rle, gzip, huffman,
    bloom_cpu, polyfit_cpu, doubleexp, pfor)
"""

__version__ = "0.1.0"

from .codecs import SparseCompressor, compressor  # noqa: F401
from .communicator import Allgather, Allreduce, broadcast_parameters  # noqa: F401
from .compressors import (  # noqa: F401
    Compressor,
    NoneCompressor,
    RandomKCompressor,
    ThresholdCompressor,
    TopKCompressor,
)
from .factory import deepreduce_from_params, grace_from_params  # noqa: F401
from .helper import tensor_bits  # noqa: F401
from .memory import NoneMemory, ResidualMemory  # noqa: F401
from .optimizer import DistributedOptimizer, reduce_gradients  # noqa: F401
from .params import validate as validate_params  # noqa: F401
from .wrappers import DeepReduce, IndexCompressor, ValueCompressor, deepreduce_wrapper  # noqa: F401
