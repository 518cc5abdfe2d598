"""Model zoo for examples and benchmarks.

Self-contained (torchvision is not a dependency): ImageNet-style ResNets
(resnet50/101/152 — bench.py's flagship is resnet50), CIFAR option-A
ResNets (resnet20..1202, reference examples/vision/cifar_resnet.py), and
a Transformer LM (examples/language).
"""

from kfac_amd.models.cifar_resnet import cifar_resnet
from kfac_amd.models.resnet import resnet50
from kfac_amd.models.resnet import resnet101
from kfac_amd.models.resnet import resnet152
from kfac_amd.models.gptneox import gptneox_125m
from kfac_amd.models.gptneox import GPTNeoXModel
from kfac_amd.models.transformer import TransformerModel

__all__ = [
    'resnet50',
    'resnet101',
    'resnet152',
    'cifar_resnet',
    'TransformerModel',
    'GPTNeoXModel',
    'gptneox_125m',
]
