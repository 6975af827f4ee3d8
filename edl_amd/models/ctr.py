"""Wide&Deep CTR model (parity: reference example/ctr — criteo-style
wide-and-deep with sparse embeddings; BASELINE config 5: elastic 1->8 GPUs
on the all-reduce path).

Sparse gradients all-reduce poorly; we use dense embedding bags of modest
vocab (hash-bucketed) so the elastic DP path applies unchanged."""
import torch
import torch.nn as nn


class WideAndDeep(nn.Module):
    def __init__(self, num_sparse_fields=26, vocab_size=100000, embed_dim=16,
                 num_dense=13, hidden=(400, 400, 400)):
        super().__init__()
        self.num_sparse_fields = num_sparse_fields
        self.vocab_size = vocab_size
        self.embedding = nn.EmbeddingBag(vocab_size, embed_dim, mode="sum")
        self.wide = nn.EmbeddingBag(vocab_size, 1, mode="sum")
        dims = [num_dense + num_sparse_fields * embed_dim] + list(hidden)
        mlp = []
        for a, b in zip(dims[:-1], dims[1:]):
            mlp += [nn.Linear(a, b), nn.ReLU(inplace=True)]
        mlp.append(nn.Linear(dims[-1], 1))
        self.deep = nn.Sequential(*mlp)

    def forward(self, dense, sparse_ids):
        """dense: [B, 13] float; sparse_ids: [B, 26] long (one id per field)."""
        B = sparse_ids.shape[0]
        flat = sparse_ids.reshape(-1) % self.vocab_size
        offsets = torch.arange(B * self.num_sparse_fields, device=sparse_ids.device)
        emb = self.embedding(flat, offsets).reshape(B, -1)
        wide = self.wide(flat, offsets).reshape(B, self.num_sparse_fields).sum(1, keepdim=True)
        deep = self.deep(torch.cat([dense, emb], dim=1))
        return wide + deep
