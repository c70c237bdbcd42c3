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
