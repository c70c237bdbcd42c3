"""Whole-model batched compression pipeline driver (MI355X hot path).

Builds the static per-model descriptor table consumed by the bt_* kernels
in ops/src/hip_ops.hip: every kernel walks a block->tensor map so all
tensors' chunks execute concurrently, and the compress side writes the
fused wire buffer (byte-identical to communicator._flatten_payload of the
per-tensor path: per tensor [float32 vals (k, padded to 8B)] [uint8 bloom
bits (ceil(m/8), padded to 8B)]) in ~12 kernels for the entire model.

Applies to the flagship configuration: top-k sparsifier + Bloom index codec
+ leftmost policy (sync-free), float32 CUDA gradients, tensors above the
1000-element codec bypass.  Anything else falls back to the generic
per-tensor path in communicator.step_many.
"""
from __future__ import annotations

import os

import torch

BT_CHUNK = 8192
# dynamic-LDS budget for the compress-side bloom query (bt_qcount R=1).
# MEASURED on MI355X (gpurun_out/qcount_ab.json, 30-iter A/B on the
# ResNet-50 layout): LDS-cached 0.533 ms vs plain 32-bit word loads
# 0.516 ms per whole-model compress — the occupancy cost of a 42 KB
# dynamic-LDS reservation slightly outweighs the LDS latency win, so the
# default is OFF (word loads through L1/L2, which hold the hot filter
# fine at R=1).  Set DEEPREDUCE_LDSQ_MAX=65536 to re-enable for
# experiments.  The multi-rank decode query uses the interleaved layout
# instead (measured 9% faster at R=8), which is always on.
LDSQ_MAX = int(os.environ.get("DEEPREDUCE_LDSQ_MAX", "0"))


def _pad8(x: int) -> int:
    return (x + 7) & ~7


def _ldsq_bytes(mws) -> int:
    """Launch-wide dynamic LDS reservation: the largest filter (in words)
    that fits the budget; blocks whose filter does not fit fall back to
    global word loads inside the kernel.  0 when disabled (the default —
    see LDSQ_MAX note)."""
    fit = [w * 4 for w in mws if w * 4 <= LDSQ_MAX] if LDSQ_MAX else []
    return (max(fit) + 255) & ~255 if fit else 0


class BatchedPipeline:
    """Index-mode (topk + bloom leftmost) whole-model pipeline."""

    kind = "index"

    def __init__(self, names, numels, params, device):
        from ..codecs.bloom import Bloom

        self.names = list(names)
        self.numels = list(numels)
        self.wire_half = params.get("wire_dtype") == "fp16"
        vb = 2 if self.wire_half else 4  # bytes per wire value
        vdt = torch.float16 if self.wire_half else torch.float32
        T = len(numels)
        ratio = params.get("compress_ratio", 0.01)

        desc = torch.zeros(T, 16, dtype=torch.int64)
        b2t = []
        voff = koff = wire_off = cntoff = mwoff = blkoff = iloff = 0
        mws = []
        self.metas = []
        for t, n in enumerate(numels):
            k = max(1, int(round(n * ratio)))
            num_hash, m = Bloom._config(k, n, params)
            nbytes = (m + 7) // 8
            mwords = (m + 31) // 32
            nb = (n + BT_CHUNK - 1) // BT_CHUNK
            desc[t, 0] = n
            desc[t, 1] = voff
            desc[t, 2] = k
            desc[t, 3] = koff
            desc[t, 4] = m
            desc[t, 5] = num_hash
            desc[t, 6] = wire_off + _pad8(vb * k)  # bits after padded vals
            desc[t, 7] = wire_off                  # vals first (payload order)
            desc[t, 8] = cntoff
            desc[t, 9] = mwoff
            desc[t, 10] = blkoff
            desc[t, 15] = iloff                    # interleaved word offset
            self.metas.append([(vdt, k), (torch.uint8, nbytes)])
            b2t.extend([t] * nb)
            mws.append(mwords)
            voff += n
            koff += k
            wire_off += _pad8(vb * k) + _pad8(nbytes)
            cntoff += nb
            mwoff += nb * (BT_CHUNK // 64)
            blkoff += nb
            iloff += mwords

        self.total_values = voff
        self.k_total = koff
        self.wire_bytes = wire_off          # multiple of 8 by construction
        self.mask_words = mwoff
        self.total_mw = iloff
        self.ldsq_bytes = _ldsq_bytes(mws)
        self.desc = desc.to(device)
        self.b2t = torch.tensor(b2t, dtype=torch.int32, device=device)

    # -- kernels ----------------------------------------------------------
    def compress(self, values_flat: torch.Tensor):
        from deepreduce_amd import _hip_ops

        wire, out_idx = _hip_ops.batched_compress(
            values_flat, self.desc, self.b2t, self.wire_bytes, self.k_total,
            self.mask_words, int(self.wire_half), self.ldsq_bytes,
        )
        return wire, out_idx

    def compress_and_own(self, values_flat: torch.Tensor):
        wire, out_idx = self.compress(values_flat)
        return wire, self.decode_own(wire, out_idx)

    def decode_own(self, wire, out_idx):
        from deepreduce_amd import _hip_ops

        return _hip_ops.batched_scatter_dense(wire, out_idx, self.desc,
                                              self.total_values,
                                              int(self.wire_half))

    def decode_sum(self, wires2d):
        return self._decode_chunked(wires2d, self._decode_sum16)

    def _decode_chunked(self, wires2d, fn):
        """The query kernels handle <=16 filters per launch (MAXR);
        larger worlds decode in sequential chunks of 16 (deterministic
        accumulation order, identical on every rank)."""
        R = int(wires2d.size(0))
        if R <= 16:
            return fn(wires2d)
        total = None
        for i in range(0, R, 16):
            part = fn(wires2d[i : i + 16].contiguous())
            total = part if total is None else total.add_(part)
        return total

    def _decode_sum16(self, wires2d):
        from deepreduce_amd import _hip_ops

        return _hip_ops.batched_decode_sum(wires2d, self.desc, self.b2t,
                                           self.total_values, self.mask_words,
                                           int(self.wire_half), self.total_mw,
                                           self.ldsq_bytes)


class BothPipeline(BatchedPipeline):
    """'both'-mode whole-model pipeline: bloom index + polyfit values +
    bit-packed mapping, wire-compatible with the generic DeepReduce
    wrapper payload (coeffs f64 | bloom bits | 5B-header packed mapping).
    """

    kind = "both"

    def __init__(self, names, numels, params, device):
        from ..codecs.bloom import Bloom
        from ..codecs.polyfit import s_pad

        self.names = list(names)
        self.numels = list(numels)
        self.degree = int(params.get("poly_degree", 5))
        d1 = self.degree + 1
        T = len(numels)
        ratio = params.get("compress_ratio", 0.01)

        desc = torch.zeros(T, 16, dtype=torch.int64)
        b2t, seg_t, seg_i = [], [], []
        voff = koff = wire_off = cntoff = mwoff = blkoff = iloff = 0
        kmax = 1
        mws = []
        self.metas = []
        for t, n in enumerate(numels):
            k = max(1, int(round(n * ratio)))
            kmax = max(kmax, k)
            num_hash, m = Bloom._config(k, n, params)
            nbytes = (m + 7) // 8
            mwords = (m + 31) // 32
            nb = (n + BT_CHUNK - 1) // BT_CHUNK
            sp = s_pad(k)
            nbits = max(1, (k - 1).bit_length())
            coeff_bytes = (sp * d1 + 1) * 8
            map_payload = 5 + (k * nbits + 7) // 8
            desc[t, 0] = n
            desc[t, 1] = voff
            desc[t, 2] = k
            desc[t, 3] = koff
            desc[t, 4] = m
            desc[t, 5] = num_hash
            desc[t, 6] = wire_off + coeff_bytes          # bloom bits
            desc[t, 8] = cntoff
            desc[t, 9] = mwoff
            desc[t, 10] = blkoff
            desc[t, 11] = sp
            desc[t, 12] = wire_off                       # coeffs
            desc[t, 13] = wire_off + coeff_bytes + _pad8(nbytes)  # mapping
            desc[t, 14] = nbits
            desc[t, 15] = iloff                          # interleaved words
            self.metas.append([(torch.float64, sp * d1 + 1),
                               (torch.uint8, nbytes),
                               (torch.uint8, map_payload)])
            b2t.extend([t] * nb)
            seg_t.extend([t] * sp)
            seg_i.extend(range(sp))
            mws.append(mwords)
            voff += n
            koff += k
            wire_off += coeff_bytes + _pad8(nbytes) + _pad8(map_payload)
            cntoff += nb
            mwoff += nb * (BT_CHUNK // 64)
            blkoff += nb
            iloff += mwords

        self.total_values = voff
        self.k_total = koff
        self.kmax = kmax
        self.wire_bytes = wire_off
        self.mask_words = mwoff
        self.total_mw = iloff
        self.ldsq_bytes = _ldsq_bytes(mws)
        self.desc = desc.to(device)
        self.b2t = torch.tensor(b2t, dtype=torch.int32, device=device)
        self.seg_t = torch.tensor(seg_t, dtype=torch.int32, device=device)
        self.seg_i = torch.tensor(seg_i, dtype=torch.int32, device=device)

    def compress_and_own(self, values_flat: torch.Tensor):
        from deepreduce_amd import _hip_ops

        wire, own = _hip_ops.batched_compress_both(
            values_flat, self.desc, self.b2t, self.seg_t, self.seg_i,
            self.wire_bytes, self.k_total, self.mask_words, self.kmax,
            self.degree, self.total_values, self.ldsq_bytes,
        )
        return wire, own

    def decode_sum(self, wires2d):
        return self._decode_chunked(wires2d, self._decode_sum16)

    def _decode_sum16(self, wires2d):
        from deepreduce_amd import _hip_ops

        return _hip_ops.batched_decode_both_sum(
            wires2d, self.desc, self.b2t, self.total_values, self.mask_words,
            self.k_total, self.degree, self.total_mw, self.ldsq_bytes,
        )


class ValuePipeline(BatchedPipeline):
    """Value-mode whole-model pipeline: polyfit coefficients + raw int32
    indices in value-sorted order (no bloom), wire-compatible with the
    generic ValueCompressor payload."""

    kind = "value"

    def __init__(self, names, numels, params, device):
        from ..codecs.polyfit import s_pad

        self.names = list(names)
        self.numels = list(numels)
        self.degree = int(params.get("poly_degree", 5))
        d1 = self.degree + 1
        T = len(numels)
        ratio = params.get("compress_ratio", 0.01)

        desc = torch.zeros(T, 16, dtype=torch.int64)
        b2t, seg_t, seg_i = [], [], []
        voff = koff = wire_off = cntoff = mwoff = blkoff = 0
        kmax = 1
        self.metas = []
        for t, n in enumerate(numels):
            k = max(1, int(round(n * ratio)))
            kmax = max(kmax, k)
            nb = (n + BT_CHUNK - 1) // BT_CHUNK
            sp = s_pad(k)
            coeff_bytes = (sp * d1 + 1) * 8
            desc[t, 0] = n
            desc[t, 1] = voff
            desc[t, 2] = k
            desc[t, 3] = koff
            desc[t, 6] = wire_off + coeff_bytes   # int32 idxs
            desc[t, 8] = cntoff
            desc[t, 9] = mwoff
            desc[t, 10] = blkoff
            desc[t, 11] = sp
            desc[t, 12] = wire_off                # coeffs
            desc[t, 13] = -1                      # no mapping chunk
            self.metas.append([(torch.float64, sp * d1 + 1), (torch.int32, k)])
            b2t.extend([t] * nb)
            seg_t.extend([t] * sp)
            seg_i.extend(range(sp))
            voff += n
            koff += k
            wire_off += coeff_bytes + _pad8(4 * k)
            cntoff += nb
            mwoff += nb * (BT_CHUNK // 64)
            blkoff += nb

        self.total_values = voff
        self.k_total = koff
        self.kmax = kmax
        self.wire_bytes = wire_off
        self.mask_words = mwoff
        self.desc = desc.to(device)
        self.b2t = torch.tensor(b2t, dtype=torch.int32, device=device)
        self.seg_t = torch.tensor(seg_t, dtype=torch.int32, device=device)
        self.seg_i = torch.tensor(seg_i, dtype=torch.int32, device=device)

    def compress_and_own(self, values_flat: torch.Tensor):
        from deepreduce_amd import _hip_ops

        wire, own = _hip_ops.batched_compress_value(
            values_flat, self.desc, self.b2t, self.seg_t, self.seg_i,
            self.wire_bytes, self.k_total, self.kmax, self.degree,
            self.total_values,
        )
        return wire, own

    def decode_sum(self, wires2d):
        return self._decode_chunked(wires2d, self._decode_sum16)

    def _decode_sum16(self, wires2d):
        from deepreduce_amd import _hip_ops

        return _hip_ops.batched_decode_value_sum(
            wires2d, self.desc, self.total_values, self.k_total, self.degree,
        )


def maybe_pipeline(communicator, comp, named_tensors):
    """Return a cached pipeline when the configuration qualifies:
    BatchedPipeline for IndexCompressor(topk+bloom+leftmost),
    BothPipeline for DeepReduce(topk+bloom+leftmost+polyfit)."""
    from . import hip_available
    from ..codecs.bloom import Bloom
    from ..codecs.polyfit import PolyFit
    from ..compressors import TopKCompressor
    from ..wrappers import DeepReduce, IndexCompressor, ValueCompressor

    cls = None
    if isinstance(comp, IndexCompressor) and comp.idx_codec is Bloom:
        cls = BatchedPipeline
    elif (isinstance(comp, DeepReduce) and comp.idx_codec is Bloom
          and comp.val_codec is PolyFit):
        cls = BothPipeline
    elif isinstance(comp, ValueCompressor) and comp.val_codec is PolyFit:
        cls = ValuePipeline
    if cls is None:
        return None
    params = comp.params
    if params.get("micro-benchmark"):
        return None
    if cls is not ValuePipeline and params.get("policy", "leftmost") != "leftmost":
        return None
    if cls in (BothPipeline, ValuePipeline) and params.get("sort", False):
        return None
    if cls is BothPipeline and (
        not params.get("pack_mapping", True)
        or not params.get("fp_aware", True)
    ):
        return None
    if not isinstance(comp.sparsifier, TopKCompressor):
        return None
    grads = [t for _, t in named_tensors]
    if not grads or not grads[0].is_cuda or not hip_available():
        return None
    if any(t.dtype != torch.float32 or not t.is_cuda for t in grads):
        return None
    if any(t.numel() <= 1000 for t in grads):  # codec bypass: generic path
        return None
    key = (
        cls.__name__,
        tuple(n for n, _ in named_tensors),
        tuple(t.numel() for t in grads),
        comp.sparsifier.compress_ratio,
        params.get("fpr"),
        params.get("poly_degree", 5),
        params.get("wire_dtype"),
        str(grads[0].device),
    )
    cache = getattr(communicator, "_bt_pipelines", None)
    if cache is None:
        cache = communicator._bt_pipelines = {}
    bp = cache.get(key)
    if bp is None:
        bp = cls([n for n, _ in named_tensors],
                 [t.numel() for t in grads], params, grads[0].device)
        cache[key] = bp
        # legacy single-slot attribute (tests introspect it)
        communicator._bt_pipeline = (key, bp)
    else:
        communicator._bt_pipeline = (key, bp)
    return bp
