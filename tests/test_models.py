# Copyright 2026. Licensed under the Apache License, Version 2.0.
"""Model-zoo smoke tests: forward/backward shapes and gradient flow."""

import pytest
import torch

from bluefog_amd import models


@pytest.mark.parametrize("name", ["resnet18", "resnet50", "vgg11", "vgg16"])
def test_model_forward_backward(name):
    m = getattr(models, name)(num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    out = m(x)
    assert out.shape == (2, 10)
    out.sum().backward()
    assert all(p.grad is not None for p in m.parameters())


def test_vgg_batch_norm_variant():
    m = models.vgg11(num_classes=5, batch_norm=True)
    out = m(torch.randn(2, 3, 64, 64))
    assert out.shape == (2, 5)


def test_bert_forward_loss():
    m = models.bert_base()
    ids = torch.randint(0, 30522, (2, 16))
    loss = m(ids, labels=ids)
    assert loss.dim() == 0
    loss.backward()
