"""LM1B LSTM language model for the PartitionedAR config.

Reference workload: examples/lm1b/{lm1b_train.py, language_model.py} — LSTM
LM on the One Billion Word benchmark, trained via `autodist.function`
(lm1b_train.py:62). Architecture: embedding -> multi-layer LSTM -> tied
softmax projection. BASELINE config #5 runs it with a simulator-selected
PartitionedAR strategy (the big embedding/softmax matrices split axis-0 into
per-shard all-reduce groups).
"""
import torch
import torch.nn as nn


class LM1BModel(nn.Module):
    def __init__(self, vocab_size=793470, emb_dim=512, hidden=2048,
                 layers=2, proj=512, dropout=0.1, tie_weights=True,
                 sparse=False):
        super().__init__()
        self.emb = nn.Embedding(vocab_size, emb_dim, sparse=sparse)
        self.lstm = nn.LSTM(emb_dim, hidden, num_layers=layers,
                            proj_size=proj, batch_first=True,
                            dropout=dropout if layers > 1 else 0.0)
        self.drop = nn.Dropout(dropout)
        if tie_weights:
            assert proj == emb_dim
            self.out_weight = self.emb.weight
        else:
            self.out_weight = nn.Parameter(
                torch.randn(vocab_size, proj) * 0.02)
        self.out_bias = nn.Parameter(torch.zeros(vocab_size))
        nn.init.normal_(self.emb.weight, std=0.02)

    def forward(self, tokens, state=None):
        x = self.drop(self.emb(tokens))
        h, state = self.lstm(x, state)
        logits = torch.nn.functional.linear(self.drop(h), self.out_weight,
                                            self.out_bias)
        return logits, state

    def loss(self, tokens, targets, state=None):
        logits, _ = self.forward(tokens, state)
        return torch.nn.functional.cross_entropy(
            logits.reshape(-1, logits.size(-1)), targets.reshape(-1))


def lm1b_small(vocab_size=10000, **kw):
    kw.setdefault("emb_dim", 256)
    kw.setdefault("hidden", 512)
    kw.setdefault("proj", 256)
    return LM1BModel(vocab_size, **kw)


def lm1b_full(**kw):
    return LM1BModel(**kw)
