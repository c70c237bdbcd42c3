"""Benchmark model zoo (random-init, synthetic data — no network access).

Mirrors the reference's benchmark matrix (/root/reference/README.md:18-22,
run_deepreduce.sh): ResNet-20/CIFAR-10, ResNet-50/ImageNet, NCF/ML-20m,
plus BERT-base for the 'both'-mode config in BASELINE.json.
"""
from .resnet import resnet20, resnet50
from .ncf import NCF
from .bert import BertBase

registry = {
    "resnet20": resnet20,
    "resnet50": resnet50,
    "ncf": NCF,
    "bert": BertBase,
}

__all__ = ["resnet20", "resnet50", "NCF", "BertBase", "registry"]
