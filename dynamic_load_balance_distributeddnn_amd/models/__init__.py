"""Model factory — CLI name -> constructor (reference dbs.py:345-362).

The CLI selects the same flagship variant per family as the reference:
resnet -> ResNet-101, densenet -> DenseNet-121, regnet -> RegNetY-400MF.
"""

from __future__ import annotations

from .densenet import (DenseNet121, DenseNet161, DenseNet169,  # noqa: F401
                       DenseNet201)
from .googlenet import GoogLeNet  # noqa: F401
from .mnistnet import MnistNet  # noqa: F401
from .regnet import (RegNetX_200MF, RegNetX_400MF,  # noqa: F401
                     RegNetY_400MF)
from .resnet import (ResNet18, ResNet34, ResNet50, ResNet101,  # noqa: F401
                     ResNet152)
from .transformer import TransformerModel  # noqa: F401

# Transformer hyperparameters hardcoded by the reference driver
# (dbs.py:337-343).
LM_CONFIG = dict(ntokens=33_278, emsize=200, nhead=2, nhid=200,
                 nlayers=2, dropout=0.2, bptt=35)


def build_model(name: str, num_classes: int = 10):
    if name == "mnistnet":
        return MnistNet()
    if name == "resnet":
        return ResNet101(num_classes)
    if name == "densenet":
        return DenseNet121(num_classes)
    if name == "googlenet":
        return GoogLeNet(num_classes)
    if name == "regnet":
        return RegNetY_400MF(num_classes)
    if name == "transformer":
        c = LM_CONFIG
        return TransformerModel(c["ntokens"], c["emsize"], c["nhead"],
                                c["nhid"], c["nlayers"], c["dropout"])
    raise ValueError(f"unknown model {name!r}")
