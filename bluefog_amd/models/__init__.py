# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Self-contained model zoo for benchmarks/examples (no torchvision /
transformers dependency): ResNet family + BERT-base."""

from bluefog_amd.models.resnet import (  # noqa: F401
    ResNet,
    resnet18,
    resnet34,
    resnet50,
    resnet101,
    resnet152,
)
from bluefog_amd.models.bert import BertConfig, BertForMaskedLM, bert_base  # noqa: F401
from bluefog_amd.models.vgg import VGG, vgg11, vgg13, vgg16, vgg19  # noqa: F401,E402
