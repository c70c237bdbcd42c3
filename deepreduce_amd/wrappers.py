"""DeepReduce wrapper layer — the core of the framework.

Wraps any sparsifying Compressor and post-compresses its (values, indices)
output with pluggable codecs.  Reference behavior:
/root/reference/pytorch/deepreduce.py:51-302 and README.md:44-48.

  ValueCompressor   post-compresses the VALUES array   ('deepreduce':'value')
  IndexCompressor   post-compresses the INDICES array  ('deepreduce':'index')
  DeepReduce        both, glued by an explicit `mapping` permutation
                    ('deepreduce':'both')

Small sparse tensors bypass compression entirely (numel <= 1000, matching
pytorch/deepreduce.py:68).  `micro-benchmark` in params enables per-stage
timing/volume prints (pytorch/deepreduce.py:74-95).
"""
from __future__ import annotations

import time

import torch

from torch.profiler import record_function

from .codecs import compressor as codec_registry
from .compressors import Compressor
from .helper import tensor_bits

_BYPASS_NUMEL = 1000


def _sync_if(t: torch.Tensor):
    if t.is_cuda:
        torch.cuda.synchronize()


class _WrapperBase(Compressor):
    def __init__(self, sparsifier, params=None):
        super().__init__(tensors_size_are_same=sparsifier.tensors_size_are_same)
        self.average = getattr(sparsifier, "average", True)
        self.sparsifier = sparsifier
        self.params = params or {}
        # single-slot cache: (name -> (vals, idxs, shape)) of THIS rank's
        # payload in decoded form, filled during compress so the residual
        # update (memory.update -> decompress_own) skips re-decoding —
        # for bloom that saves one full-universe query per tensor per step.
        self._own_cache: dict = {}

    def aggregate(self, tensors):
        return self.sparsifier.aggregate(tensors)

    def decompress_own(self, tensors, ctx, name):
        cached = self._own_cache.get(name)
        if cached is not None:
            vals, idxs, shape = cached
            return self.sparsifier.decompress((vals, idxs), shape)
        return self.decompress(tensors, ctx)


class ValueCompressor(_WrapperBase):
    """Sparsify, then post-compress the values array.

    Parity: pytorch/deepreduce.py:51-97.  Payload sizes: qsgd/polyseg
    depend only on k and polyfit's padded slot layout only on N (see
    codecs/polyfit.py get_segments), so those three gather uniformly;
    other value codecs (gzip byte streams, doubleexp is uniform but kept
    conservative) mark payloads ragged like the reference's note at
    pytorch/deepreduce.py:364-367.
    """

    def __init__(self, sparsifier, params=None):
        super().__init__(sparsifier, params)
        name = self.params.get("value", "polyfit")
        self.val_codec = codec_registry[name]
        if name not in ("qsgd", "polyseg", "polyfit", "doubleexp"):
            # qsgd/polyseg payload sizes depend only on k; polyfit's padded
            # 22-slot layout depends only on N; doubleexp is 4 coeffs + a
            # k-sized signed mapping -> all uniform across ranks
            self.tensors_size_are_same = False

    def compress(self, tensor, name):
        tensors, ctx = self.sparsifier.compress(tensor, name)
        vals, idxs = tensors
        shape = ctx
        if torch.Size(shape).numel() > _BYPASS_NUMEL:
            start = time.perf_counter()
            with record_function("dr::val_compress"):
                vals, idxs, shape_out = self.val_codec.compress((vals, idxs, tensor.size()), self.params)
            if self.params.get("micro-benchmark", False):
                _sync_if(tensor)
                print(f"val_compression time:{time.perf_counter() - start}")
            ctx = shape_out
        return (vals, idxs), ctx

    def decompress(self, tensors, ctx):
        shape = ctx
        vals, idxs = tensors
        if torch.Size(shape).numel() > _BYPASS_NUMEL:
            start = time.perf_counter()
            with record_function("dr::val_decompress"):
                vals, idxs, shape = self.val_codec.decompress((vals, idxs, shape), self.params)
            if self.params.get("micro-benchmark", False):
                print(f"val_decompression time:{time.perf_counter() - start}")
                dense_bits = torch.Size(shape).numel() * 32
                print(f"idx_relative_volume: {tensor_bits([tensors[1]]) / dense_bits:.4f}")
                print(f"val_relative_volume: {tensor_bits([tensors[0]]) / dense_bits:.4f}")
        return self.sparsifier.decompress((vals, idxs), shape)


class IndexCompressor(_WrapperBase):
    """Sparsify, then post-compress the indices array.

    Parity: pytorch/deepreduce.py:100-153.  The dense tensor rides along in
    params for the Bloom codec's FP-aware value re-read (:117).
    """

    def __init__(self, sparsifier, params=None):
        super().__init__(sparsifier, params)
        name = self.params.get("index", "bloom")
        self.idx_codec = codec_registry[name]
        if name not in ("bloom",) or self.params.get("policy") == "p0":
            self.tensors_size_are_same = False

    def compress(self, tensor, name):
        tensors, ctx = self.sparsifier.compress(tensor, name)
        vals, idxs = tensors
        shape = ctx
        if torch.Size(shape).numel() > _BYPASS_NUMEL:
            self.params["dense_tensor"] = tensor
            start = time.perf_counter()
            with record_function("dr::idx_compress"):
                vals, idxs, shape_out = self.idx_codec.compress((vals, idxs, tensor.size()), self.params)
            self.params.pop("dense_tensor", None)
            own = self.params.pop("_own_decoded", None)
            if own is not None:
                self._own_cache[name] = (own[0], own[1], shape)
            if self.params.get("micro-benchmark", False):
                _sync_if(tensor)
                print(f"idx_compression time:{time.perf_counter() - start}")
            ctx = shape_out
        return (vals, idxs), ctx

    def decompress(self, tensors, ctx):
        shape = ctx
        vals, idxs = tensors
        if torch.Size(shape).numel() > _BYPASS_NUMEL:
            start = time.perf_counter()
            with record_function("dr::idx_decompress"):
                vals, idxs, shape = self.idx_codec.decompress((vals, idxs, shape), self.params)
            if self.params.get("micro-benchmark", False):
                print(f"idx_decompression time:{time.perf_counter() - start}")
                dense_bits = torch.Size(shape).numel() * 32
                print(f"idx_relative_volume: {tensor_bits([tensors[1]]) / dense_bits:.4f}")
                print(f"val_relative_volume: {tensor_bits([tensors[0]]) / dense_bits:.4f}")
        return self.sparsifier.decompress((vals, idxs), shape)

    def decompress_batch(self, payloads, ctx):
        """Fused multi-rank decompress (bloom + leftmost only): ONE batched
        universe query over all ranks' filters — the probe positions are
        filter-independent, so hashing amortizes R-fold on the GPU — then a
        single fused scatter-add.  Returns the SUM of dense tensors, or
        None if this codec/policy combination has no fast path.
        """
        from .codecs.bloom import Bloom

        shape = ctx
        numel = int(torch.Size(shape).numel())
        if (
            self.idx_codec is not Bloom
            or self.params.get("policy", "leftmost") != "leftmost"
            or numel <= _BYPASS_NUMEL
            or len(payloads) < 2
        ):
            return None
        from . import ops

        vals0 = payloads[0][0]
        num_indices = int(vals0.numel())
        if any(int(p[0].numel()) != num_indices for p in payloads):
            return None  # leftmost assumes uniform k (topk sparsifier)
        num_hash, m = Bloom._config(num_indices, numel, self.params)
        dense = torch.zeros(numel, dtype=torch.float32, device=vals0.device)
        # the query kernel handles <=16 filters per launch (MAXR); larger
        # worlds decode in chunks of 16, summed sequentially (deterministic
        # order, identical on every rank)
        for c in range(0, len(payloads), 16):
            chunk = payloads[c : c + 16]
            bits = torch.stack([p[1].contiguous() for p in chunk])
            # sync-free: [R, k] leftmost positives, one fused scatter-add
            idxs = ops.bloom_query_leftmost(bits, m, num_hash, numel,
                                            num_indices)
            vals = torch.stack([p[0] for p in chunk]).float()  # fp16 wire ok
            dense.index_add_(0, idxs.reshape(-1), vals.reshape(-1))
        return dense.view(shape)


class DeepReduce(_WrapperBase):
    """Joint index+value compression glued by a mapping permutation.

    Parity: pytorch/deepreduce.py:156-302.  compress: index codec first
    (bloom bits), then the value codec over the Bloom-ordered values with
    new_idxs = arange — its non-order-preserving sort permutation becomes
    `mapping`.  decompress: values from coeffs (in sorted order), indices
    re-derived from the bloom bits (ascending), then idxs = idxs[mapping]
    undoes the value sort (:290).  Wire = (vals', bloom_bits, mapping).
    """

    def __init__(self, sparsifier, params=None):
        super().__init__(sparsifier, params)
        value = self.params.get("value", "polyfit")
        index = self.params.get("index", "bloom")
        self.val_codec = codec_registry[value]
        self.idx_codec = codec_registry[index]
        # polyfit (padded layout), qsgd, polyseg payloads depend only on
        # (N, k); bloom bits + bit-packed mapping likewise -> uniform when
        # the sparsifier is uniform and the policy is not P0
        self.tensors_size_are_same = (
            sparsifier.tensors_size_are_same
            and value in ("polyfit", "qsgd", "polyseg")
            and index == "bloom"
            and self.params.get("policy", "leftmost") != "p0"
        )

    def compress(self, tensor, name):
        tensors, ctx = self.sparsifier.compress(tensor, name)
        vals, idxs = tensors
        shape = ctx
        start = time.perf_counter()
        if torch.Size(shape).numel() > _BYPASS_NUMEL:
            # FP-aware re-read (improvement over the reference, which skips it
            # in 'both' mode): values are read from the dense tensor at the
            # positions decompress will deterministically re-derive, so false
            # positives no longer shift the index<->value alignment.
            if self.params.get("fp_aware", True):
                self.params["dense_tensor"] = tensor
            vals, idxs, _ = self.idx_codec.compress((vals, idxs, tensor.size()), self.params)
            self.params.pop("dense_tensor", None)
            own = self.params.pop("_own_decoded", None)
            new_idxs = torch.arange(vals.numel(), device=vals.device)
            vals, mapping, shape_out = self.val_codec.compress((vals, new_idxs, shape), self.params)
            ctx = shape_out
            if self._skip_mapping():
                # Order-preserving value codec (qsgd/gzip): the "mapping" is
                # the identity, so it never travels — decompress re-derives
                # the bloom positives in ascending order and the value codec
                # returns values in exactly that order.  This is the lever
                # behind the paper's DR-QSGD-BF-P0 headline volume (Table 2
                # 0.0621): without it the ceil(log2 k)-bit permutation
                # dominates the wire.
                if own is not None:
                    self._own_cache[name] = (own[1], None)
                if self.params.get("micro-benchmark", False):
                    _sync_if(tensor)
                    print(f"_compression time:{time.perf_counter() - start}")
                return (vals, idxs), ctx
            if own is not None:
                # cache (bloom indices, unpacked mapping): decompress_own
                # then skips both the universe query AND the mapping
                # header parse (a host sync per tensor)
                self._own_cache[name] = (own[1], mapping.long())
            if self.params.get("pack_mapping", True):
                # ceil(log2 k) bits per mapping entry instead of int32
                # (paper App. E; the reference left this commented out at
                # pytorch/deepreduce.py:264-265 — we ship it, GPU-packed).
                # nbits is host-known (mapping is a permutation of
                # arange(k)) — no device sync.
                from .codecs.intpack import pack_with_header

                nbits = max(1, (int(mapping.numel()) - 1).bit_length())
                mapping = pack_with_header(mapping.long(), nbits=nbits)
                tensors = (vals, idxs, mapping)
            else:
                tensors = (vals, idxs, mapping.int())
        if self.params.get("micro-benchmark", False):
            _sync_if(tensor)
            print(f"_compression time:{time.perf_counter() - start}")
        return tensors, ctx

    def _skip_mapping(self) -> bool:
        """True when the identity mapping need not travel: the value codec
        preserves order AND the decompress side can derive the positive
        count without it (p0 carries it in-band; topk/randomk imply k)."""
        if not getattr(self.val_codec, "order_preserving", False):
            return False
        # the FP-aware re-read puts values in ascending-position order —
        # exactly the order decompress re-derives; without it the values
        # stay in topk-magnitude order and the mapping must travel
        if not self.params.get("fp_aware", True):
            return False
        return (self.params.get("policy", "leftmost") == "p0"
                or getattr(self.sparsifier, "compress_ratio", None) is not None)

    def _expected_k(self, shape) -> int:
        numel = int(torch.Size(shape).numel())
        ratio = getattr(self.sparsifier, "compress_ratio")
        return max(1, int(round(numel * ratio)))

    def decompress(self, tensors, ctx):
        shape = ctx
        start = time.perf_counter()
        if torch.Size(shape).numel() > _BYPASS_NUMEL:
            if len(tensors) == 2:  # mapping-free (order-preserving) wire
                vals_w, packed = tensors
                if self.params.get("policy", "leftmost") == "p0":
                    placeholder = vals_w  # count travels in-band in the bits
                else:
                    placeholder = vals_w.new_empty(self._expected_k(shape))
                _, idxs, _ = self.idx_codec.decompress(
                    (placeholder, packed, shape), self.params)
                vals, _, _ = self.val_codec.decompress(
                    (vals_w, idxs, shape), self.params)
            else:
                vals, idxs, mapping = tensors
                mapping = self._unpack_mapping(mapping)
                vals, _, _ = self.val_codec.decompress((vals, mapping, shape), self.params)
                _, idxs, _ = self.idx_codec.decompress((mapping, idxs, shape), self.params)
                idxs = idxs[mapping]
        else:
            vals, idxs = tensors
        if self.params.get("micro-benchmark", False):
            print(f"_decompression time:{time.perf_counter() - start}")
            dense_bits = torch.Size(shape).numel() * 32
            print(f"idx_relative_volume: {tensor_bits(list(tensors[1:])) / dense_bits:.4f}")
            print(f"val_relative_volume: {tensor_bits([tensors[0]]) / dense_bits:.4f}")
        return self.sparsifier.decompress((vals, idxs), shape)

    def _unpack_mapping(self, mapping):
        if mapping.dtype == torch.uint8:  # bit-packed wire
            from .codecs.intpack import unpack_with_header

            return unpack_with_header(mapping).to(mapping.device)
        return mapping.long()

    def decompress_own(self, tensors, ctx, name):
        """Own-payload decode with the cached bloom indices and mapping:
        only the (cheap) value-codec eval runs; the full-universe query
        and the mapping header parse are both skipped."""
        cached = self._own_cache.get(name)
        shape = ctx
        if cached is None or torch.Size(shape).numel() <= _BYPASS_NUMEL:
            return self.decompress(tensors, ctx)
        cached_idxs, mapping = cached
        vals = tensors[0]
        if mapping is None:  # order-preserving value codec: identity mapping
            vals, _, _ = self.val_codec.decompress((vals, cached_idxs, shape),
                                                   self.params)
            return self.sparsifier.decompress((vals, cached_idxs), shape)
        vals, _, _ = self.val_codec.decompress((vals, mapping, shape), self.params)
        idxs = cached_idxs[mapping]
        return self.sparsifier.decompress((vals, idxs), shape)


deepreduce_wrapper = {
    "value": ValueCompressor,
    "index": IndexCompressor,
    "both": DeepReduce,
}
