"""Trainer models for the shuffling-loader benchmarks/examples.

The reference's Horovod example defines an MNIST-ish conv ``Net``
(reference: examples/horovod/ray_torch_shuffle.py:124-140) but never runs it
— the train step is ``time.sleep(mock_train_step_time)`` with the real
fwd/bwd commented out (ray_torch_shuffle.py:209-218). Here the benchmark
model is a real MLP regression head shaped for the tabular batches the
loader yields, and the step executes a genuine forward+backward+optimizer
update on the GPU.
"""

import torch
import torch.nn as nn


class TabularMLP(nn.Module):
    """MLP over a [B, num_features] float batch -> scalar regression."""

    def __init__(
        self,
        num_features: int = 100,
        hidden: int = 512,
        depth: int = 3,
    ):
        super().__init__()
        layers = []
        d = num_features
        for i in range(depth):
            h = hidden // (2**i)
            layers += [nn.Linear(d, h), nn.ReLU()]
            d = h
        layers.append(nn.Linear(d, 1))
        self.net = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)


class EmbeddingMLP(nn.Module):
    """Model shaped for the reference DATA_SPEC batches: 19 int64
    categorical columns -> embeddings, concatenated -> MLP
    (the realistic counterpart of the reference example's unused Net)."""

    def __init__(self, cardinalities, embed_dim: int = 16, hidden: int = 256):
        super().__init__()
        self.embeddings = nn.ModuleList(
            [
                nn.Embedding(int(c), embed_dim)
                for c in cardinalities
            ]
        )
        d = embed_dim * len(cardinalities)
        self.mlp = nn.Sequential(
            nn.Linear(d, hidden),
            nn.ReLU(),
            nn.Linear(hidden, 1),
        )

    def forward(self, cat_feats) -> torch.Tensor:
        embs = [
            emb(x.clamp_min(0).squeeze(-1) % emb.num_embeddings)
            for emb, x in zip(self.embeddings, cat_feats)
        ]
        return self.mlp(torch.cat(embs, dim=1))
