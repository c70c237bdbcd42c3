"""Federated-learning driver — paper Algorithm 2 (deepreduce.nips21.pdf
p.34, sect. 6.2), which the reference describes but ships no code for
(SURVEY.md sect. 2.4 'Federated DP').

One round, bidirectional compression with error feedback:
  server: delta = x_t - x_0  ->  DR(delta)  -> broadcast to m clients
  client: x <- x_0 + DR^-1(delta); E local steps; push DR(sum of grads)
  server: x_t+1 <- x_t - lr * mean(DR^-1(client updates))
Residual error-feedback on both directions (Top-r 10% in the paper).

This is a single-process simulation driver (clients are model replicas) —
the codec/wrapper layer is identical to the datacenter path; only the
round structure differs.  Used by tests and the FL relative-volume bench.
"""
from __future__ import annotations

import copy

import torch

from .helper import tensor_bits
from .memory import ResidualMemory


class FederatedServer:
    def __init__(self, model: torch.nn.Module, compressor, lr: float = 1.0):
        self.model = model
        self.compressor = compressor
        self.lr = lr
        self.x0 = {n: p.detach().clone() for n, p in model.named_parameters()}
        self.s2c_memory = ResidualMemory()
        self.wire_bytes_s2c = 0
        self.wire_bytes_c2s = 0

    def broadcast_payload(self):
        """Compress x_t - x_0 per tensor; returns {name: (payload, ctx)}."""
        out = {}
        self.wire_bytes_s2c = 0
        for n, p in self.model.named_parameters():
            delta = p.detach() - self.x0[n]
            delta = self.s2c_memory.compensate(delta, n)
            payload, ctx = self.compressor.compress(delta, n)
            self.s2c_memory.update(delta, n, self.compressor, payload, ctx)
            out[n] = (payload, ctx)
            self.wire_bytes_s2c += tensor_bits(list(payload)) // 8
        return out

    def apply_client_updates(self, updates: list[dict]):
        """updates: list of {name: (payload, ctx)} from clients; average."""
        self.wire_bytes_c2s = 0
        with torch.no_grad():
            for n, p in self.model.named_parameters():
                total = None
                for u in updates:
                    payload, ctx = u[n]
                    d = self.compressor.decompress(payload, ctx)
                    total = d if total is None else total + d
                    self.wire_bytes_c2s += tensor_bits(list(payload)) // 8
                p.add_(total.view_as(p), alpha=-self.lr / len(updates))


class FederatedClient:
    def __init__(self, server_model: torch.nn.Module, compressor, local_lr: float = 0.05):
        self.template = server_model
        self.compressor = compressor
        self.local_lr = local_lr
        self.c2s_memory = ResidualMemory()
        self.x0 = {n: p.detach().clone() for n, p in server_model.named_parameters()}

    def round(self, broadcast, data_iter, epochs: int = 1, loss_fn=None):
        """Receive compressed delta, run E local steps, return compressed
        gradient-sum payload."""
        model = copy.deepcopy(self.template)
        with torch.no_grad():
            for n, p in model.named_parameters():
                payload, ctx = broadcast[n]
                delta = self.compressor.decompress(payload, ctx)
                p.copy_(self.x0[n] + delta.view_as(p))
        start = {n: p.detach().clone() for n, p in model.named_parameters()}

        opt = torch.optim.SGD(model.parameters(), lr=self.local_lr)
        loss_fn = loss_fn or torch.nn.functional.cross_entropy
        for _ in range(epochs):
            for x, y in data_iter():
                opt.zero_grad()
                loss_fn(model(x), y).backward()
                opt.step()

        out = {}
        for n, p in model.named_parameters():
            g_sum = (start[n] - p.detach()) / self.local_lr  # accumulated grad
            g_sum = self.c2s_memory.compensate(g_sum, n)
            payload, ctx = self.compressor.compress(g_sum, n)
            self.c2s_memory.update(g_sum, n, self.compressor, payload, ctx)
            out[n] = (payload, ctx)
        return out


def run_federated_round(server: FederatedServer, clients: list[FederatedClient],
                        data_iters, epochs: int = 1, loss_fn=None):
    broadcast = server.broadcast_payload()
    updates = [
        c.round(broadcast, di, epochs, loss_fn) for c, di in zip(clients, data_iters)
    ]
    server.apply_client_updates(updates)
    return server.wire_bytes_s2c, server.wire_bytes_c2s
