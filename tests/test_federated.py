"""Federated round driver (paper Algorithm 2): loss decreases, volumes
tracked, bidirectional compression round-trips."""
import torch

from deepreduce_amd import TopKCompressor
from deepreduce_amd.federated import FederatedClient, FederatedServer, run_federated_round
from deepreduce_amd.wrappers import IndexCompressor


class TinyNet(torch.nn.Module):
    def __init__(self):
        super().__init__()
        self.fc1 = torch.nn.Linear(20, 64)
        self.fc2 = torch.nn.Linear(64, 4)

    def forward(self, x):
        return self.fc2(torch.relu(self.fc1(x)))


def _make_data(seed):
    torch.manual_seed(seed)
    w = torch.randn(20, 4)
    x = torch.randn(64, 20)
    y = (x @ w).argmax(dim=1)

    def it():
        for i in range(0, 64, 16):
            yield x[i : i + 16], y[i : i + 16]

    return it, (x, y)


def test_federated_rounds_reduce_loss():
    torch.manual_seed(0)
    model = TinyNet()
    comp = TopKCompressor(0.1)
    server = FederatedServer(model, comp, lr=0.5)
    clients = [FederatedClient(model, comp, local_lr=0.05) for _ in range(3)]
    data = [_make_data(s) for s in range(3)]
    iters = [d[0] for d in data]

    def total_loss():
        with torch.no_grad():
            return sum(
                torch.nn.functional.cross_entropy(model(x), y).item() for _, (x, y) in data
            )

    before = total_loss()
    for _ in range(5):
        s2c, c2s = run_federated_round(server, clients, iters, epochs=1)
        assert s2c > 0 and c2s > 0
    after = total_loss()
    assert after < before


def test_federated_with_bloom_wrapper():
    torch.manual_seed(1)
    model = TinyNet()
    sp = TopKCompressor(0.2)
    comp = IndexCompressor(sp, {"index": "bloom", "policy": "leftmost"})
    server = FederatedServer(model, comp, lr=0.5)
    clients = [FederatedClient(model, comp, local_lr=0.05) for _ in range(2)]
    iters = [_make_data(s)[0] for s in range(2)]
    s2c, c2s = run_federated_round(server, clients, iters, epochs=1)
    assert s2c > 0 and c2s > 0


def test_multi_round_loss_decreases():
    """Compressed federated rounds make training progress (paper Alg. 2
    end-to-end): server loss after 6 rounds < initial loss."""
    import torch

    from deepreduce_amd import deepreduce_from_params
    from deepreduce_amd.federated import (FederatedClient, FederatedServer,
                                          run_federated_round)

    torch.manual_seed(0)
    model = torch.nn.Sequential(torch.nn.Linear(32, 64), torch.nn.ReLU(),
                                torch.nn.Linear(64, 4))
    grc = deepreduce_from_params({
        "compressor": "topk", "memory": "residual",
        "communicator": "allgather", "compress_ratio": 0.10,
        "deepreduce": "index", "index": "bloom", "policy": "p0",
    })
    centers = torch.randn(4, 32, generator=torch.Generator().manual_seed(2)) * 2

    def data_iter(seed):
        def it():
            g = torch.Generator().manual_seed(seed)
            for _ in range(3):
                y = torch.randint(0, 4, (32,), generator=g)
                yield centers[y] + torch.randn(32, 32, generator=g), y
        return it

    def server_loss():
        g = torch.Generator().manual_seed(999)
        y = torch.randint(0, 4, (256,), generator=g)
        x = centers[y] + torch.randn(256, 32, generator=g)
        with torch.no_grad():
            return float(torch.nn.functional.cross_entropy(model(x), y))

    server = FederatedServer(model, grc.compressor, lr=0.3)
    clients = [FederatedClient(model, grc.compressor) for _ in range(3)]
    before = server_loss()
    for r in range(6):
        run_federated_round(server, clients,
                            [data_iter(10 * r + i) for i in range(3)])
    after = server_loss()
    assert after < before * 0.8, (before, after)
