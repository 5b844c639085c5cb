"""Model zoo shape/grad checks (CPU)."""

import torch

from torch_cgx_amd.models import resnet50, resnet18, bert_tiny


def test_resnet50_forward_backward():
    m = resnet50(num_classes=10)
    x = torch.randn(2, 3, 64, 64)
    y = m(x)
    assert y.shape == (2, 10)
    y.sum().backward()
    assert all(p.grad is not None for p in m.parameters())
    n_params = sum(p.numel() for p in resnet50(num_classes=1000).parameters())
    assert 25_500_000 < n_params < 25_600_000  # canonical ResNet-50: 25.557M


def test_resnet18_forward():
    m = resnet18(num_classes=5)
    assert m(torch.randn(2, 3, 32, 32)).shape == (2, 5)


def test_bert_tiny_forward_backward():
    m = bert_tiny(vocab=100)
    ids = torch.randint(0, 100, (2, 16))
    out = m(ids)
    assert out.shape == (2, 16, 100)
    out.sum().backward()


def test_bert_large_param_count():
    from torch_cgx_amd.models import bert_large
    m = bert_large()
    n = sum(p.numel() for p in m.parameters())
    assert 300_000_000 < n < 360_000_000  # BERT-large ~335M
