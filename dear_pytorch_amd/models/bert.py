"""BERT pretraining model (MLM + NSP heads), self-contained.

Capability parity with the reference's BERT benchmark path
(dear/bert_benchmark.py:72-112: transformers-2.11 BertForPreTraining built
from bert_config.json + BertPretrainingCriterion).  Configs: Base 12L/768h/12
heads, Large 24L/1024h/16 heads, vocab padded to a multiple of 8 (30528).
Attention runs through torch.nn.functional.scaled_dot_product_attention
(MIOpen/CK fused path on ROCm)."""
from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["BertConfig", "BertForPreTraining", "BertPretrainingCriterion",
           "bert_base", "bert_large"]


@dataclass
class BertConfig:
    vocab_size: int = 30528          # 30522 padded %8 (reference :77-78)
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    intermediate_size: int = 3072
    max_position_embeddings: int = 512
    type_vocab_size: int = 2
    hidden_dropout_prob: float = 0.1
    attention_probs_dropout_prob: float = 0.1
    layer_norm_eps: float = 1e-12


def bert_base():
    return BertConfig()


def bert_large():
    return BertConfig(hidden_size=1024, num_hidden_layers=24,
                      num_attention_heads=16, intermediate_size=4096)


class BertEmbeddings(nn.Module):
    def __init__(self, c: BertConfig):
        super().__init__()
        self.word = nn.Embedding(c.vocab_size, c.hidden_size)
        self.position = nn.Embedding(c.max_position_embeddings, c.hidden_size)
        self.token_type = nn.Embedding(c.type_vocab_size, c.hidden_size)
        self.ln = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.drop = nn.Dropout(c.hidden_dropout_prob)
        self.register_buffer("pos_ids",
                             torch.arange(c.max_position_embeddings)[None],
                             persistent=False)

    def forward(self, input_ids, token_type_ids):
        s = input_ids.size(1)
        e = self.word(input_ids) + self.position(self.pos_ids[:, :s]) \
            + self.token_type(token_type_ids)
        return self.drop(self.ln(e))


class BertSelfAttention(nn.Module):
    def __init__(self, c: BertConfig):
        super().__init__()
        self.nh = c.num_attention_heads
        self.hd = c.hidden_size // c.num_attention_heads
        self.qkv = nn.Linear(c.hidden_size, 3 * c.hidden_size)
        self.out = nn.Linear(c.hidden_size, c.hidden_size)
        self.p_drop = c.attention_probs_dropout_prob

    def forward(self, x, attn_mask):
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.nh, self.hd)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        o = F.scaled_dot_product_attention(
            q, k, v, attn_mask=attn_mask,
            dropout_p=self.p_drop if self.training else 0.0)
        return self.out(o.transpose(1, 2).reshape(B, S, H))


class BertLayer(nn.Module):
    def __init__(self, c: BertConfig):
        super().__init__()
        self.attn = BertSelfAttention(c)
        self.ln1 = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.fc1 = nn.Linear(c.hidden_size, c.intermediate_size)
        self.fc2 = nn.Linear(c.intermediate_size, c.hidden_size)
        self.ln2 = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.drop = nn.Dropout(c.hidden_dropout_prob)

    def forward(self, x, attn_mask):
        x = self.ln1(x + self.drop(self.attn(x, attn_mask)))
        h = self.fc2(F.gelu(self.fc1(x)))
        return self.ln2(x + self.drop(h))


class BertModel(nn.Module):
    def __init__(self, c: BertConfig):
        super().__init__()
        self.embeddings = BertEmbeddings(c)
        self.layers = nn.ModuleList(BertLayer(c) for _ in range(c.num_hidden_layers))
        self.pooler = nn.Linear(c.hidden_size, c.hidden_size)

    def forward(self, input_ids, token_type_ids, attention_mask=None):
        mask = None
        if attention_mask is not None:
            mask = (attention_mask[:, None, None, :].to(torch.bool))
        x = self.embeddings(input_ids, token_type_ids)
        for layer in self.layers:
            x = layer(x, mask)
        pooled = torch.tanh(self.pooler(x[:, 0]))
        return x, pooled


class BertForPreTraining(nn.Module):
    """Sequence output -> tied-embedding MLM head; pooled -> NSP head."""

    def __init__(self, c: BertConfig):
        super().__init__()
        self.config = c
        self.bert = BertModel(c)
        self.transform = nn.Linear(c.hidden_size, c.hidden_size)
        self.transform_ln = nn.LayerNorm(c.hidden_size, eps=c.layer_norm_eps)
        self.decoder = nn.Linear(c.hidden_size, c.vocab_size, bias=True)
        self.decoder.weight = self.bert.embeddings.word.weight  # tied
        self.nsp = nn.Linear(c.hidden_size, 2)
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids, token_type_ids, attention_mask=None):
        seq, pooled = self.bert(input_ids, token_type_ids, attention_mask)
        h = self.transform_ln(F.gelu(self.transform(seq)))
        return self.decoder(h), self.nsp(pooled)


class BertPretrainingCriterion(nn.Module):
    """MLM + NSP cross-entropy (reference dear/bert_benchmark.py:101-112)."""

    def __init__(self, vocab_size):
        super().__init__()
        self.vocab_size = vocab_size
        self.ce = nn.CrossEntropyLoss(ignore_index=-1)

    def forward(self, pred_scores, nsp_scores, mlm_labels, nsp_labels):
        mlm = self.ce(pred_scores.view(-1, self.vocab_size),
                      mlm_labels.view(-1))
        nsp = self.ce(nsp_scores.view(-1, 2), nsp_labels.view(-1))
        return mlm + nsp
