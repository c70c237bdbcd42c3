"""Benchmark model zoo (random-init, synthetic data — no network access).

Mirrors the reference's benchmark matrix (/root/reference/README.md:18-22,
run_deepreduce.sh): ResNet-20/CIFAR-10, ResNet-50/ImageNet, NCF/ML-20m,
plus the paper's FL backbones (MobileNetV2/CIFAR, LSTM next-word — pdf
p.8 Table 2, p.33 Table 5) and BERT-base for the 'both'-mode config in
BASELINE.json.
"""
from .resnet import resnet20, resnet50
from .ncf import NCF
from .bert import BertBase
from .mobilenet import mobilenet_v2
from .rnn_lm import RnnLM

registry = {
    "resnet20": resnet20,
    "resnet50": resnet50,
    "ncf": NCF,
    "bert": BertBase,
    "mobilenet": mobilenet_v2,
    "rnn": RnnLM,
}

__all__ = ["resnet20", "resnet50", "NCF", "BertBase", "mobilenet_v2",
           "RnnLM", "registry"]
