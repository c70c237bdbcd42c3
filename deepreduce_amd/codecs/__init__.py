"""Sparse-tensor codecs (value codecs + index codecs).

Registry parity with the reference
(/root/reference/pytorch/deepreduce.py:913-922): bloom, polyfit, bloom_cpu,
polyfit_cpu, gzip, huffman, rle, qsgd — plus the native additions: 'pfor'
(FastPFor-equivalent delta+bitpack integer index codec, replacing
tensorflow/integer_compression.cc) and 'doubleexp' (the TF DoubleExp value
codec, tensorflow/deepreduce.py:377-442).

Codec interface (`SparseCompressor`): static
    compress((vals, idxs, shape), params)   -> (vals', idxs', shape)
    decompress((vals', idxs', shape), params) -> (vals, idxs, shape)
`shape` is a torch.Size; vals'/idxs' are flat wire tensors.
"""
from __future__ import annotations


class SparseCompressor:
    """Interface for compressing/decompressing a sparse tensor."""

    order_preserving = True

    @staticmethod
    def compress(sparse_tensor, params):
        raise NotImplementedError

    @staticmethod
    def decompress(sparse_tensor, params):
        raise NotImplementedError


from .bloom import Bloom, BloomCPU  # noqa: E402
from .polyfit import PolyFit  # noqa: E402
from .polyfit_cpu import PolyFitCPU  # noqa: E402
from .doubleexp import DoubleExp  # noqa: E402
from .polyseg import PolySeg  # noqa: E402
from .qsgd import QSGD  # noqa: E402
from .rle import RunLength  # noqa: E402
from .gzipc import Gzip  # noqa: E402
from .huffman import Huffman  # noqa: E402
from .intpack import PFor  # noqa: E402

compressor = {
    "bloom": Bloom,
    "polyfit": PolyFit,
    "bloom_cpu": BloomCPU,
    "polyfit_cpu": PolyFitCPU,
    "doubleexp": DoubleExp,
    "polyseg": PolySeg,
    "gzip": Gzip,
    "huffman": Huffman,
    "rle": RunLength,
    "qsgd": QSGD,
    "pfor": PFor,
}

codec_registry = compressor  # alias
