"""Model zoo shape/backward sanity (CPU, small inputs)."""
import pytest
import torch

from dear_pytorch_amd import models


@pytest.mark.parametrize("name,res", [
    ("resnet50", 64), ("resnet18", 64), ("vgg16", 64), ("densenet121", 64),
    ("inceptionv4", 299),
])
def test_cnn_forward_backward(name, res):
    m = models.get_cnn(name, num_classes=10)
    x = torch.randn(2, 3, res, res)
    y = m(x)
    assert y.shape == (2, 10)
    y.sum().backward()
    assert all(p.grad is not None for p in m.parameters() if p.requires_grad)


def test_param_counts_match_standard():
    # well-known ImageNet parameter counts (tolerance for fc variants)
    n50 = sum(p.numel() for p in models.get_cnn("resnet50").parameters())
    assert abs(n50 - 25_557_032) < 1000, n50
    nv = sum(p.numel() for p in models.get_cnn("vgg16").parameters())
    assert abs(nv - 138_357_544) < 1000, nv
    nd = sum(p.numel() for p in models.get_cnn("densenet201").parameters())
    assert abs(nd - 20_013_928) < 1000, nd


def test_bert_pretraining_step():
    c = models.bert_base()
    c.num_hidden_layers = 2  # small for CPU
    m = models.BertForPreTraining(c)
    crit = models.BertPretrainingCriterion(c.vocab_size)
    B, S = 2, 16
    ids = torch.randint(0, c.vocab_size, (B, S))
    tt = torch.zeros(B, S, dtype=torch.long)
    mask = torch.ones(B, S, dtype=torch.long)
    mlm_labels = torch.full((B, S), -1)
    mlm_labels[:, 3] = ids[:, 3]
    nsp = torch.randint(0, 2, (B,))
    scores, seq_rel = m(ids, tt, mask)
    assert scores.shape == (B, S, c.vocab_size)
    loss = crit(scores, seq_rel, mlm_labels, nsp)
    loss.backward()
    assert torch.isfinite(loss)
    # tied decoder/embedding weight
    assert m.decoder.weight is m.bert.embeddings.word.weight


def test_bert_large_config():
    c = models.bert_large()
    assert (c.num_hidden_layers, c.hidden_size, c.num_attention_heads) == \
        (24, 1024, 16)
    assert c.vocab_size % 8 == 0


def test_mnistnet():
    m = models.MnistNet()
    y = m(torch.randn(4, 1, 28, 28))
    assert y.shape == (4, 10)
    assert torch.allclose(y.exp().sum(1), torch.ones(4), atol=1e-5)
