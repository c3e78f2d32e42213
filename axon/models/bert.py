"""BERT encoder + masked-LM head for the BERT-large pretrain config
(BASELINE.json config 3: BERT-large + LAMB + fp16 grad compression).

Own implementation of the standard architecture (Devlin et al., BERT):
post-LN transformer encoder, GELU FFN, tied MLM decoder. Attention uses
torch scaled_dot_product_attention (PyTorch-ROCm's fused path).
"""
from dataclasses import dataclass

import torch
import torch.nn as nn

from axon.ops.layernorm import FusedLayerNorm
import torch.nn.functional as F


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden: int = 1024
    layers: int = 24
    heads: int = 16
    ffn: int = 4096
    max_seq: int = 512
    type_vocab: int = 2
    dropout: float = 0.0  # pretrain-benchmark default: no dropout


def bert_large_config():
    return BertConfig()


def bert_base_config():
    return BertConfig(hidden=768, layers=12, heads=12, ffn=3072)


def bert_tiny_config():
    """For tests: same code path, tiny shapes."""
    return BertConfig(vocab_size=128, hidden=64, layers=2, heads=4, ffn=128,
                      max_seq=64)


class BertSelfAttention(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.heads = cfg.heads
        self.head_dim = cfg.hidden // cfg.heads
        self.qkv = nn.Linear(cfg.hidden, 3 * cfg.hidden)
        self.out = nn.Linear(cfg.hidden, cfg.hidden)

    def forward(self, x, attn_mask=None):
        b, s, h = x.shape
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        q = q.view(b, s, self.heads, self.head_dim).transpose(1, 2)
        k = k.view(b, s, self.heads, self.head_dim).transpose(1, 2)
        v = v.view(b, s, self.heads, self.head_dim).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v, attn_mask=attn_mask)
        return self.out(o.transpose(1, 2).reshape(b, s, h))


class BertLayer(nn.Module):
    def __init__(self, cfg):
        super().__init__()
        self.attn = BertSelfAttention(cfg)
        self.ln1 = FusedLayerNorm(cfg.hidden)
        self.fc1 = nn.Linear(cfg.hidden, cfg.ffn)
        self.fc2 = nn.Linear(cfg.ffn, cfg.hidden)
        self.ln2 = FusedLayerNorm(cfg.hidden)
        self.drop = nn.Dropout(cfg.dropout)

    def forward(self, x, attn_mask=None):
        x = self.ln1(x + self.drop(self.attn(x, attn_mask)))
        x = self.ln2(x + self.drop(self.fc2(F.gelu(self.fc1(x)))))
        return x


class BertForPreTraining(nn.Module):
    """Encoder + MLM head (tied embedding decoder)."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.tok = nn.Embedding(cfg.vocab_size, cfg.hidden)
        self.pos = nn.Embedding(cfg.max_seq, cfg.hidden)
        self.typ = nn.Embedding(cfg.type_vocab, cfg.hidden)
        self.ln_emb = FusedLayerNorm(cfg.hidden)
        self.layers = nn.ModuleList(BertLayer(cfg) for _ in range(cfg.layers))
        self.mlm_dense = nn.Linear(cfg.hidden, cfg.hidden)
        self.mlm_ln = FusedLayerNorm(cfg.hidden)
        self.mlm_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.apply(self._init)

    def _init(self, m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
            if isinstance(m, nn.Linear) and m.bias is not None:
                nn.init.zeros_(m.bias)

    def forward(self, input_ids, token_type_ids=None, attn_mask=None):
        b, s = input_ids.shape
        pos = torch.arange(s, device=input_ids.device)
        x = self.tok(input_ids) + self.pos(pos)
        if token_type_ids is not None:
            x = x + self.typ(token_type_ids)
        else:
            x = x + self.typ.weight[0]
        x = self.ln_emb(x)
        for layer in self.layers:
            x = layer(x, attn_mask)
        h = self.mlm_ln(F.gelu(self.mlm_dense(x)))
        # tied decoder: logits = h @ tok_emb^T + bias
        return h @ self.tok.weight.t() + self.mlm_bias

    def mlm_loss(self, input_ids, labels, **kw):
        """labels = -100 where not masked (standard MLM convention)."""
        logits = self.forward(input_ids, **kw)
        return F.cross_entropy(logits.float().view(-1, self.cfg.vocab_size),
                               labels.view(-1), ignore_index=-100)
