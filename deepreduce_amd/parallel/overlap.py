"""Bucketed gradient exchange overlapped with backward (MI355X/xGMI-first).

Design (SURVEY.md §2.3 collective notes, BASELINE north star "decompress
overlapped with backward on a side HIP stream"): the model's large tensors
are partitioned — in reverse registration order, i.e. the order backward
produces gradients — into a few buckets of roughly equal element count.
As soon as every gradient of a bucket has been produced
(post-accumulate-grad hooks), the bucket's whole compression pipeline
(compensate → batched compress → residual update → collective) is enqueued
on a side HIP stream, overlapping the rest of backward. `finalize()`
drains the stream, decodes each bucket (multi-rank-batched), writes the
reduced gradients, and runs the fused dense exchange of the small
(≤1000-element) tensors.

Each bucket gets its own batched pipeline (ops/batched.py) and flat
residual pool; on CPU (or for non-qualifying codecs) the bucket falls back
to the communicator's generic step_many semantics — bit-identical to the
synchronous DistributedOptimizer path (tested on gloo world=2).

The residual-update ordering invariant (compensate → compress → update,
SURVEY.md §7) is preserved per bucket inside the hook.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

__all__ = ["OverlappedReducer"]


class OverlappedReducer:
    """Hook-driven bucketed reducer.  Usage:

        reducer = OverlappedReducer(model, grc, num_buckets=3)
        ...
        loss.backward()          # buckets launch as their grads complete
        reducer.finalize()       # drain, decode, write reduced grads
        optimizer.step()
    """

    def __init__(self, model: torch.nn.Module, grc, num_buckets: int = 3,
                 small_threshold: int = 1000):
        self.model = model
        self.grc = grc
        self.small_threshold = small_threshold
        self.last_wire_bytes = 0

        params = [(n, p) for n, p in model.named_parameters() if p.requires_grad]
        self._small = [(n, p) for n, p in params if p.numel() <= small_threshold]
        large = [(n, p) for n, p in reversed(params) if p.numel() > small_threshold]

        # equal-elements partition in backward production order
        total = sum(p.numel() for _, p in large)
        num_buckets = max(1, min(num_buckets, len(large)))
        target = total / num_buckets
        self._buckets: list[list] = [[]]
        acc = 0
        for n, p in large:
            if acc >= target * len(self._buckets) and len(self._buckets) < num_buckets:
                self._buckets.append([])
            self._buckets[-1].append((n, p))
            acc += p.numel()
        self._bucket_of = {n: b for b, bucket in enumerate(self._buckets)
                           for n, _ in bucket}

        self._use_cuda = any(p.is_cuda for _, p in params)
        self._comm_stream = torch.cuda.Stream() if self._use_cuda else None
        self._arrived = [0] * len(self._buckets)
        self._launched: dict[int, tuple] = {}
        self._hooks = []
        for n, p in large:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._make_hook(n)))

        n_small = sum(p.numel() for _, p in self._small)
        dev = params[0][1].device if params else torch.device("cpu")
        self._small_buf = torch.zeros(n_small, dtype=torch.float32, device=dev)

    def remove_hooks(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []

    # -- hooks -------------------------------------------------------------
    def _make_hook(self, name):
        b = self._bucket_of[name]

        def hook(_param):
            self._arrived[b] += 1
            if self._arrived[b] == len(self._buckets[b]):
                self._arrived[b] = 0
                self._launch(b)

        return hook

    # -- per-bucket pipeline -------------------------------------------------
    def _launch(self, b):
        grc = self.grc
        named = [(n, p.grad.data.float()) for n, p in self._buckets[b]]
        if self._comm_stream is not None:
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            ctx = torch.cuda.stream(self._comm_stream)
        else:
            ctx = _null_ctx()
        with ctx:
            from ..ops import batched as _bt

            comp = grc.compressor
            bp = _bt.maybe_pipeline(grc, comp, named)
            if bp is None:
                # generic fallback: synchronous bucket exchange now (still
                # on the comm stream); result ready at finalize
                impl = getattr(grc, "_step_many_impl", None)
                outs = impl(named) if impl else [grc.step(t, n) for n, t in named]
                self._launched[b] = ("done", outs, grc.last_wire_bytes)
                return
            from ..memory import ResidualMemory

            names = [n for n, _ in named]
            grads = [t for _, t in named]
            mem = grc.memory
            compensated = mem.compensate_many(grads, names)
            flat_c = getattr(mem, "_flat_c", None)
            if (flat_c is not None and compensated
                    and compensated[0].data_ptr() == flat_c.data_ptr()
                    and flat_c.numel() == bp.total_values):
                c_flat = flat_c
            else:
                c_flat = torch.cat([t.reshape(-1) for t in compensated])
            wire, own = bp.compress_and_own(c_flat)
            # residual <- compensated - own decode; same guards as
            # communicator._step_many_batched: the in-place fast path is only
            # valid when c_flat IS this bucket's flat pool (same storage, so
            # _flat_r is the matching residual buffer); otherwise fall back to
            # the generic per-name update.  NoneMemory: nothing to update.
            if isinstance(mem, ResidualMemory):
                if c_flat is flat_c:
                    torch.sub(c_flat, own, out=mem._flat_r)
                else:
                    offs = 0
                    decs = []
                    for t in grads:
                        decs.append(own[offs : offs + t.numel()].view(t.shape))
                        offs += t.numel()
                    mem.update_many(compensated, names, decs)
            world = grc.world_size
            if world == 1:
                self._launched[b] = ("own", bp, own, wire.numel())
                return
            gathered = torch.empty(world, wire.numel(), dtype=torch.uint8,
                                   device=wire.device)
            try:
                work = dist.all_gather_into_tensor(gathered.view(-1), wire,
                                                   async_op=True)
            except (AttributeError, RuntimeError):
                bufs = list(gathered.unbind(0))
                work = dist.all_gather(bufs, wire, async_op=True)
            self._launched[b] = ("gathered", bp, gathered, work, wire.numel())

    # -- finalize ------------------------------------------------------------
    def finalize(self):
        grc = self.grc
        world = grc.world_size
        ctx = (torch.cuda.stream(self._comm_stream)
               if self._comm_stream is not None else _null_ctx())
        with ctx:
            for b, bucket in enumerate(self._buckets):
                entry = self._launched.pop(b, None)
                if entry is None:
                    # Bucket never filled this step.  A PARTIAL arrival count
                    # (frozen layer / conditional branch produced only some of
                    # the bucket's grads) must not leak into the next step, or
                    # the bucket would launch mid-backward on a stale mix —
                    # exchange whatever grads exist now, synchronously, and
                    # reset the counter.
                    if self._arrived[b]:
                        self._arrived[b] = 0
                        present = [(n, p) for n, p in bucket
                                   if p.grad is not None]
                        for n, p in present:
                            r = grc.step(p.grad.data.float(), n)
                            p.grad.data.copy_(r.view_as(p.grad.data))
                            self.last_wire_bytes += getattr(
                                grc, "last_wire_bytes", 0)
                    continue  # no grads this step
                self._arrived[b] = 0  # defensive: full reset each step
                if entry[0] == "done":
                    _, outs, wb = entry
                    self.last_wire_bytes += wb
                elif entry[0] == "own":
                    _, bp, own, wb = entry
                    self.last_wire_bytes += wb
                    outs = _split(own, [p for _, p in bucket])
                else:
                    _, bp, gathered, work, wb = entry
                    self.last_wire_bytes += wb
                    if work is not None:
                        work.wait()
                    result = bp.decode_sum(gathered)
                    if grc.compressor.average:
                        result /= world
                    outs = _split(result, [p for _, p in bucket])
                for (n, p), r in zip(bucket, outs):
                    p.grad.data.copy_(r.view_as(p.grad.data))
            self._exchange_small(world)
        if self._comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self._comm_stream)

    def _exchange_small(self, world):
        if not self._small:
            return
        offset = 0
        for name, p in self._small:
            n = p.numel()
            if p.grad is not None:
                self._small_buf[offset : offset + n] = p.grad.data.reshape(-1).float()
            offset += n
        self.last_wire_bytes += self._small_buf.numel() * 4
        if world > 1:
            dist.all_reduce(self._small_buf)
            self._small_buf /= world
        offset = 0
        for name, p in self._small:
            n = p.numel()
            if p.grad is not None:
                p.grad.data.copy_(
                    self._small_buf[offset : offset + n].view_as(p.grad.data))
            offset += n

    def zero_wire_counter(self):
        self.last_wire_bytes = 0


def _split(flat, params):
    outs = []
    off = 0
    for p in params:
        n = p.numel()
        outs.append(flat[off : off + n].view(p.grad.shape
                                             if p.grad is not None else p.shape))
        off += n
    return outs


class _null_ctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
