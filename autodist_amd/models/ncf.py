"""Neural Collaborative Filtering (NeuMF) for the PartitionedPS config.

Reference workload: examples/benchmark/ncf.py (+utils/recommendation/) — NCF
on MovieLens with sparse embeddings routed to PartitionedPS. Architecture
from He et al. 2017: GMF tower (elementwise product of embeddings) + MLP
tower, fused prediction head.

The user/item tables use sparse gradients (torch sparse COO) so the engine's
sparse path (variable-length allgather + segment coalesce + rowwise apply)
or PS routing applies; with `sharded=True` the tables are row-sharded across
ranks via ShardedEmbedding (all-to-all id exchange over xGMI).
"""
import torch
import torch.nn as nn


class NeuMF(nn.Module):
    def __init__(self, num_users, num_items, mf_dim=64,
                 mlp_dims=(256, 128, 64), sparse=True, sharded=False):
        super().__init__()
        emb_cls = None
        if sharded:
            from autodist_amd.parallel.sharded_embedding import ShardedEmbedding
            emb_cls = ShardedEmbedding
        mlp_emb_dim = mlp_dims[0] // 2

        def make_emb(n, d):
            if emb_cls is not None:
                return emb_cls(n, d)
            return nn.Embedding(n, d, sparse=sparse)

        self.mf_user = make_emb(num_users, mf_dim)
        self.mf_item = make_emb(num_items, mf_dim)
        self.mlp_user = make_emb(num_users, mlp_emb_dim)
        self.mlp_item = make_emb(num_items, mlp_emb_dim)
        mlp = []
        in_dim = mlp_dims[0]
        for out_dim in mlp_dims[1:]:
            mlp += [nn.Linear(in_dim, out_dim), nn.ReLU()]
            in_dim = out_dim
        self.mlp = nn.Sequential(*mlp)
        self.head = nn.Linear(mf_dim + in_dim, 1)
        for m in self.modules():
            if isinstance(m, nn.Embedding):
                nn.init.normal_(m.weight, std=0.01)

    def forward(self, users, items):
        gmf = self.mf_user(users) * self.mf_item(items)
        mlp = self.mlp(torch.cat(
            [self.mlp_user(users), self.mlp_item(items)], dim=-1))
        return self.head(torch.cat([gmf, mlp], dim=-1)).squeeze(-1)

    def loss(self, users, items, labels):
        return torch.nn.functional.binary_cross_entropy_with_logits(
            self.forward(users, items), labels.float())


def ncf_movielens(num_users=138493, num_items=26744, **kw):
    """MovieLens-20M-sized NCF (the reference's benchmark scale)."""
    return NeuMF(num_users, num_items, **kw)
