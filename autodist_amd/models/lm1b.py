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
                 sparse=False, sharded_softmax=False):
        super().__init__()
        self.sharded_softmax = sharded_softmax
        if sharded_softmax:
            # vocab-parallel path (reference partitioner seam,
            # partitioner.py:577-602): row-sharded tied table — input
            # lookup over all-to-all, output projection + CE over the
            # 3-collective sharded softmax (parallel/vocab_parallel.py)
            from autodist_amd.parallel.sharded_embedding import \
                ShardedEmbedding
            from autodist_amd.parallel.vocab_parallel import \
                VocabParallelProjection
            assert tie_weights and proj == emb_dim
            self.emb = ShardedEmbedding(vocab_size, emb_dim)
            self.out = VocabParallelProjection(
                vocab_size, proj, bias=True, tied_shard=self.emb.shard,
                rank=self.emb.rank, world_size=self.emb.world_size,
                process_group=self.emb.process_group)
        else:
            self.emb = nn.Embedding(vocab_size, emb_dim, sparse=sparse)
            if tie_weights:
                assert proj == emb_dim
                self.out_weight = self.emb.weight
            else:
                self.out_weight = nn.Parameter(
                    torch.randn(vocab_size, proj) * 0.02)
            self.out_bias = nn.Parameter(torch.zeros(vocab_size))
            nn.init.normal_(self.emb.weight, std=0.02)
        self.lstm = nn.LSTM(emb_dim, hidden, num_layers=layers,
                            proj_size=proj, batch_first=True,
                            dropout=dropout if layers > 1 else 0.0)
        self.drop = nn.Dropout(dropout)

    def _hidden(self, tokens, state=None):
        x = self.drop(self.emb(tokens))
        # run the LSTM OUTSIDE autocast in fp32: the decomposed ROCm LSTM
        # thrashes bf16<->fp32 casts around every pointwise op under
        # autocast (~170 cast kernels/step measured); a clean fp32 pass is
        # net faster and more precise. The projection/CE stay bf16.
        if torch.is_autocast_enabled() and x.is_cuda:
            with torch.autocast(device_type="cuda", enabled=False):
                h, state = self.lstm(x.float(), state)
        else:
            h, state = self.lstm(x, state)
        return self.drop(h), state

    def forward(self, tokens, state=None):
        h, state = self._hidden(tokens, state)
        if self.sharded_softmax:
            return self.out.full_logits(h), state
        logits = torch.nn.functional.linear(h, self.out_weight,
                                            self.out_bias)
        return logits, state

    def loss(self, tokens, targets, state=None):
        if self.sharded_softmax:
            h, _ = self._hidden(tokens, state)
            return self.out.loss(h, targets)
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
