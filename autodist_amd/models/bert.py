"""BERT encoder for the Parallax benchmark config.

Reference workload: examples/benchmark/bert.py (TF-model-garden BERT-large
pretraining). Implemented from the architecture (Devlin et al. 2018):
token/position/segment embeddings -> N transformer encoder layers ->
masked-LM head (weight-tied). BASELINE config #3 is BERT-base Parallax on
8x MI355X: dense params sync over bucketed RCCL all-reduce, the (sparse)
embedding tables go to load-balanced PS owners.

Attention uses torch.scaled_dot_product_attention (ROCm SDPA); params stay
fp32 with bf16 autocast compute.
"""

import torch
import torch.nn as nn

from autodist_amd.ops.fused_linear import FusedLinear, fused_linear


class BertConfig:
    def __init__(self, vocab_size=30522, hidden=768, layers=12, heads=12,
                 intermediate=3072, max_seq=512, type_vocab=2, dropout=0.1):
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.layers = layers
        self.heads = heads
        self.intermediate = intermediate
        self.max_seq = max_seq
        self.type_vocab = type_vocab
        self.dropout = dropout

    @classmethod
    def base(cls):
        return cls()

    @classmethod
    def large(cls):
        return cls(hidden=1024, layers=24, heads=16, intermediate=4096)

    @classmethod
    def tiny(cls):  # CI-sized
        return cls(vocab_size=1000, hidden=64, layers=2, heads=4,
                   intermediate=128, max_seq=64)


class BertSelfAttention(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        self.heads = cfg.heads
        self.head_dim = cfg.hidden // cfg.heads
        self.qkv = FusedLinear(cfg.hidden, 3 * cfg.hidden)
        self.out = FusedLinear(cfg.hidden, cfg.hidden)
        self.dropout = cfg.dropout

    def forward(self, x, attn_mask=None):
        B, S, H = x.shape
        qkv = self.qkv(x).view(B, S, 3, self.heads, self.head_dim)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        # serving path: the fused gfx950 MFMA attention kernel (falls back to
        # torch SDPA for training / masked / non-bf16 cases)
        from autodist_amd.ops.fused_attention import fused_sdpa
        o = fused_sdpa(q, k, v, attn_mask=attn_mask,
                       dropout_p=self.dropout if self.training else 0.0)
        o = o.transpose(1, 2).reshape(B, S, H)
        return self.out(o)


class BertLayer(nn.Module):
    def __init__(self, cfg: BertConfig):
        super().__init__()
        from autodist_amd.ops.fused_ln import FusedLayerNorm
        self.attn = BertSelfAttention(cfg)
        self.ln1 = FusedLayerNorm(cfg.hidden, eps=1e-12)
        self.fc1 = FusedLinear(cfg.hidden, cfg.intermediate)
        self.fc2 = FusedLinear(cfg.intermediate, cfg.hidden)
        self.ln2 = FusedLayerNorm(cfg.hidden, eps=1e-12)
        self.drop = nn.Dropout(cfg.dropout)

    def forward(self, x, attn_mask=None):
        # residual adds fold into the fused bf16 LN kernels (gfx950)
        x = self.ln1(self.drop(self.attn(x, attn_mask)), residual=x)
        h = self.fc2(torch.nn.functional.gelu(self.fc1(x)))
        return self.ln2(self.drop(h), residual=x)


class BertModel(nn.Module):
    def __init__(self, cfg: BertConfig, sparse_embeddings=False):
        super().__init__()
        self.cfg = cfg
        self.tok_emb = nn.Embedding(cfg.vocab_size, cfg.hidden,
                                    sparse=sparse_embeddings)
        self.pos_emb = nn.Embedding(cfg.max_seq, cfg.hidden)
        self.seg_emb = nn.Embedding(cfg.type_vocab, cfg.hidden)
        from autodist_amd.ops.fused_ln import FusedLayerNorm
        self.emb_ln = FusedLayerNorm(cfg.hidden, eps=1e-12)
        self.emb_drop = nn.Dropout(cfg.dropout)
        self.layers = nn.ModuleList(BertLayer(cfg) for _ in range(cfg.layers))
        self.apply(self._init)

    @staticmethod
    def _init(m):
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
        if isinstance(m, nn.Linear) and m.bias is not None:
            nn.init.zeros_(m.bias)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device).unsqueeze(0)
        seg = token_type_ids if token_type_ids is not None else \
            torch.zeros_like(input_ids)
        x = self.tok_emb(input_ids) + self.pos_emb(pos) + self.seg_emb(seg)
        x = self.emb_drop(self.emb_ln(x))
        mask = None
        if attention_mask is not None:
            mask = attention_mask[:, None, None, :].to(torch.bool)
        for layer in self.layers:
            x = layer(x, mask)
        return x


class BertForPreTraining(nn.Module):
    """MLM head (weight-tied to token embedding) + NSP head."""

    def __init__(self, cfg: BertConfig, sparse_embeddings=False):
        super().__init__()
        self.bert = BertModel(cfg, sparse_embeddings)
        from autodist_amd.ops.fused_ln import FusedLayerNorm
        self.mlm_dense = FusedLinear(cfg.hidden, cfg.hidden)
        self.mlm_ln = FusedLayerNorm(cfg.hidden, eps=1e-12)
        self.mlm_bias = nn.Parameter(torch.zeros(cfg.vocab_size))
        self.nsp = FusedLinear(cfg.hidden, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        h = self.bert(input_ids, token_type_ids, attention_mask)
        m = self.mlm_ln(torch.nn.functional.gelu(self.mlm_dense(h)))
        logits = fused_linear(m, self.bert.tok_emb.weight, self.mlm_bias)
        nsp = self.nsp(h[:, 0])
        return logits, nsp

    def loss(self, input_ids, mlm_labels, nsp_labels=None,
             token_type_ids=None, attention_mask=None):
        logits, nsp = self.forward(input_ids, token_type_ids, attention_mask)
        l = torch.nn.functional.cross_entropy(
            logits.view(-1, logits.size(-1)), mlm_labels.view(-1),
            ignore_index=-100)
        if nsp_labels is not None:
            l = l + torch.nn.functional.cross_entropy(nsp, nsp_labels)
        return l


def bert_base(sparse_embeddings=False):
    return BertForPreTraining(BertConfig.base(), sparse_embeddings)


def bert_large(sparse_embeddings=False):
    return BertForPreTraining(BertConfig.large(), sparse_embeddings)


def bert_tiny(sparse_embeddings=False):
    return BertForPreTraining(BertConfig.tiny(), sparse_embeddings)
