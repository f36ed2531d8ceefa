"""BERT-family encoder (the reference's original headline workload:
docs/_tutorials/bert-pretraining.md; its DeepSpeedTransformerLayer fused
encoder kernels are csrc/transformer — here the fused pieces are this
framework's FusedLayerNorm and torch SDPA, with hipBLASLt GEMMs).

Post-LN architecture (original BERT): Attn -> Add&LN -> FFN(GELU) ->
Add&LN. Pretraining head = masked-LM + next-sentence prediction.
"""

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.norms import FusedLayerNorm


@dataclass
class BertConfig:
    vocab_size: int = 30522
    hidden_size: int = 1024
    intermediate_size: int = 4096
    num_layers: int = 24
    num_heads: int = 16
    max_seq_len: int = 512
    type_vocab_size: int = 2
    dropout: float = 0.1
    ln_eps: float = 1e-12
    initializer_range: float = 0.02

    @property
    def head_dim(self):
        return self.hidden_size // self.num_heads


def bert_large():
    return BertConfig()


def bert_base():
    return BertConfig(hidden_size=768, intermediate_size=3072, num_layers=12,
                      num_heads=12)


def bert_tiny():
    return BertConfig(vocab_size=512, hidden_size=64, intermediate_size=128,
                      num_layers=2, num_heads=4, max_seq_len=64, dropout=0.0)


class BertEmbeddings(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.word_embeddings = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.position_embeddings = nn.Embedding(cfg.max_seq_len,
                                                cfg.hidden_size)
        self.token_type_embeddings = nn.Embedding(cfg.type_vocab_size,
                                                  cfg.hidden_size)
        self.LayerNorm = FusedLayerNorm(cfg.hidden_size, eps=cfg.ln_eps)
        self.dropout = nn.Dropout(cfg.dropout)

    def forward(self, input_ids, token_type_ids=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device).expand(B, S)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        x = (self.word_embeddings(input_ids) +
             self.position_embeddings(pos) +
             self.token_type_embeddings(token_type_ids))
        return self.dropout(self.LayerNorm(x))


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        h = cfg.hidden_size
        self.num_heads = cfg.num_heads
        self.head_dim = cfg.head_dim
        self.qkv = nn.Linear(h, 3 * h)
        self.attn_out = nn.Linear(h, h)
        self.attn_norm = FusedLayerNorm(h, eps=cfg.ln_eps)
        self.ffn_in = nn.Linear(h, cfg.intermediate_size)
        self.ffn_out = nn.Linear(cfg.intermediate_size, h)
        self.ffn_norm = FusedLayerNorm(h, eps=cfg.ln_eps)
        self.dropout = nn.Dropout(cfg.dropout)

    def forward(self, x, attention_mask=None):
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.num_heads, self.head_dim)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        o = F.scaled_dot_product_attention(q, k, v, attn_mask=attention_mask)
        o = o.transpose(1, 2).reshape(B, S, H)
        x = self.attn_norm(x + self.dropout(self.attn_out(o)))
        f = self.ffn_out(F.gelu(self.ffn_in(x), approximate="tanh"))
        return self.ffn_norm(x + self.dropout(f))


class BertModel(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.embeddings = BertEmbeddings(cfg)
        self.layers = nn.ModuleList([BertLayer(cfg)
                                     for _ in range(cfg.num_layers)])
        self.pooler = nn.Linear(cfg.hidden_size, cfg.hidden_size)

    def forward(self, input_ids, attention_mask=None, token_type_ids=None):
        if attention_mask is not None and attention_mask.dim() == 2:
            # [B, S] padding mask -> broadcastable bool [B, 1, 1, S]
            attention_mask = attention_mask[:, None, None, :].bool()
        x = self.embeddings(input_ids, token_type_ids)
        for layer in self.layers:
            x = layer(x, attention_mask)
        pooled = torch.tanh(self.pooler(x[:, 0]))
        return x, pooled


class BertForPreTraining(nn.Module):
    """Masked-LM + next-sentence heads (the BERT pretraining objective the
    reference's headline numbers were quoted on)."""

    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.cfg = cfg
        self.bert = BertModel(cfg)
        self.mlm_dense = nn.Linear(cfg.hidden_size, cfg.hidden_size)
        self.mlm_norm = FusedLayerNorm(cfg.hidden_size, eps=cfg.ln_eps)
        self.mlm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size)
        self.mlm_head.weight = self.bert.embeddings.word_embeddings.weight
        self.nsp_head = nn.Linear(cfg.hidden_size, 2)
        self.apply(self._init)

    def _init(self, m):
        std = self.cfg.initializer_range
        if isinstance(m, nn.Linear):
            m.weight.data.normal_(0.0, std)
            if m.bias is not None:
                m.bias.data.zero_()
        elif isinstance(m, nn.Embedding):
            m.weight.data.normal_(0.0, std)

    def forward(self, input_ids, attention_mask=None, token_type_ids=None,
                labels=None, next_sentence_label=None):
        hidden, pooled = self.bert(input_ids, attention_mask, token_type_ids)
        mlm = self.mlm_head(self.mlm_norm(
            F.gelu(self.mlm_dense(hidden), approximate="tanh")))
        nsp = self.nsp_head(pooled)
        if labels is None:
            return mlm, nsp
        loss = F.cross_entropy(mlm.float().view(-1, self.cfg.vocab_size),
                               labels.view(-1), ignore_index=-100)
        if next_sentence_label is not None:
            loss = loss + F.cross_entropy(nsp.float(),
                                          next_sentence_label.view(-1))
        return loss

    def num_parameters(self):
        return sum(p.numel() for p in self.parameters())
