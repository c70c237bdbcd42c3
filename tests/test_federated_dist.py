"""Multi-process federated driver (gloo, 1 server + 2 clients) must match
the single-process simulation exactly (same seeds -> same final model)."""
from __future__ import annotations

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _make_model():
    torch.manual_seed(21)
    return torch.nn.Sequential(
        torch.nn.Linear(64, 128), torch.nn.ReLU(), torch.nn.Linear(128, 10))


def _client_data(client_id, rnd):
    def it():
        g = torch.Generator().manual_seed(1000 * rnd + client_id)
        for _ in range(2):
            yield (torch.randn(16, 64, generator=g),
                   torch.randint(0, 10, (16,), generator=g))
    return it


# deterministic end-to-end config (QSGD's stochastic rounding depends on
# each process's RNG stream, so exact-match equivalence uses polyfit)
_PARAMS = {"compressor": "topk", "memory": "residual",
           "communicator": "allgather", "compress_ratio": 0.10,
           "deepreduce": "both", "value": "polyfit", "index": "bloom",
           "policy": "leftmost"}


def _simulate_single_process(rounds=3, clients=2):
    from deepreduce_amd import deepreduce_from_params
    from deepreduce_amd.federated import (FederatedClient, FederatedServer,
                                          run_federated_round)

    grc = deepreduce_from_params(dict(_PARAMS))
    model = _make_model()
    server = FederatedServer(model, grc.compressor, lr=0.5)
    cls = [FederatedClient(model, grc.compressor) for _ in range(clients)]
    for r in range(rounds):
        run_federated_round(server, cls,
                            [_client_data(i + 1, r) for i in range(clients)])
    return [p.detach().clone() for p in model.parameters()]


def _dist_worker(rank, world, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = "29712"
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from deepreduce_amd import deepreduce_from_params
        from deepreduce_amd.federated_dist import FederatedDistRunner

        grc = deepreduce_from_params(dict(_PARAMS))
        model = _make_model()
        runner = FederatedDistRunner(model, grc.compressor, server_lr=0.5)
        for r in range(3):
            s2c, c2s = runner.round(_client_data(rank, r))
        if rank == 0:
            assert s2c > 0 and c2s > 0
            q.put(("params", [p.detach().clone() for p in model.parameters()],
                   s2c, c2s))
        else:
            q.put(("ok", None, 0, runner.wire_bytes_c2s))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put(("error", f"{e!r}\n{traceback.format_exc()[-1500:]}", 0, 0))


@pytest.mark.timeout(300)
def test_federated_dist_matches_single_process():
    world = 3  # server + 2 clients
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dist_worker, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    server_params = None
    for kind, payload, s2c, c2s in results:
        assert kind != "error", payload
        if kind == "params":
            server_params = payload
            assert s2c > 0 and c2s > 0
    assert server_params is not None
    expected = _simulate_single_process()
    for a, b in zip(expected, server_params):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max()


def test_payload_serialization_roundtrip():
    from deepreduce_amd.federated_dist import (deserialize_payloads,
                                               serialize_payloads)

    torch.manual_seed(3)
    payloads = {
        "a": (torch.randn(100), torch.randint(0, 1000, (100,))),
        "b": (torch.randn(17).double(), torch.randint(0, 256, (33,),
                                                      dtype=torch.uint8),
              torch.tensor([5], dtype=torch.int64)),
    }
    buf = serialize_payloads(payloads, ["a", "b"])
    out = deserialize_payloads(buf, ["a", "b"])
    for n in payloads:
        assert len(out[n]) == len(payloads[n])
        for x, y in zip(out[n], payloads[n]):
            assert x.dtype == y.dtype
            assert torch.equal(x, y.reshape(-1))


def _qsgd_worker(rank, world, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = "29713"
        dist.init_process_group("gloo", rank=rank, world_size=world)
        from deepreduce_amd import deepreduce_from_params
        from deepreduce_amd.federated_dist import FederatedDistRunner

        params = {"compressor": "topk", "memory": "residual",
                  "communicator": "allgather", "compress_ratio": 0.10,
                  "deepreduce": "both", "value": "qsgd", "index": "bloom",
                  "policy": "p0", "qsgd_pack": True, "quantum_num": 63}
        grc = deepreduce_from_params(params)
        torch.manual_seed(21)
        model = torch.nn.Sequential(torch.nn.Linear(64, 512),
                                    torch.nn.ReLU(),
                                    torch.nn.Linear(512, 10))
        runner = FederatedDistRunner(model, grc.compressor, server_lr=0.5)
        dense = sum(p.numel() * 4 for p in model.parameters())
        rels = []
        for r in range(2):
            s2c, c2s = runner.round(_client_data(rank, r))
            if rank == 0:
                rels.append(s2c / dense)
        q.put(("ok", rels, 0, 0))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put(("error", f"{e!r}\n{traceback.format_exc()[-1500:]}", 0, 0))


@pytest.mark.timeout(300)
def test_federated_dist_qsgd_bf_p0_volume():
    """The paper-headline config over real process boundaries: finite
    round-trip and a sane transmitted volume (well under Top-r's 0.20;
    round 1 of a random-init model has near-uniform value magnitudes, so
    the volume-report number 0.054 is measured there, not here)."""
    world = 3
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_qsgd_worker, args=(r, world, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
    for kind, payload, _, _ in results:
        assert kind != "error", payload
        if isinstance(payload, list) and payload:
            for rel in payload:
                assert 0 < rel < 0.15, payload
