"""DCN-v2 style dense tower (BASELINE.json config 4)."""
from typing import List

import torch
import torch.nn as nn


class CrossLayerV2(nn.Module):
    """x_{l+1} = x0 * (U(V x_l) + b) + x_l — DCN-v2 cross layer in its
    low-rank ("mixture of experts" degenerate) form, the production variant
    of the DCN-v2 paper; rank=0 selects the full-rank W.

    ``fused=True`` runs the two projections on the hand-written MFMA GEMM
    kernels (ops/dense.py FusedLinear) — requires dim % 32 == 0 (DCNv2
    pads its feature width)."""

    def __init__(self, dim: int, rank: int = 256, fused: bool = False):
        super().__init__()
        if fused:
            from persia_amd.ops.dense import FusedLinear

            mk = lambda i, o, b: FusedLinear(i, o, relu=False, bias=b)  # noqa: E731
        else:
            mk = lambda i, o, b: nn.Linear(i, o, bias=b)  # noqa: E731
        if rank and rank < dim:
            self.v = mk(dim, rank, False)
            self.u = mk(rank, dim, True)
            self.w = None
        else:
            self.w = mk(dim, dim, True)

    def forward(self, x0: torch.Tensor, xl: torch.Tensor) -> torch.Tensor:
        proj = self.w(xl) if self.w is not None else self.u(self.v(xl))
        return x0 * proj.to(x0.dtype) + xl


class DCNv2(nn.Module):
    def __init__(
        self,
        num_sparse: int = 26,
        num_dense: int = 13,
        dim: int = 64,
        num_cross: int = 3,
        deep: List[int] = (512, 256, 128),
        cross_rank: int = 256,
        fused: bool = False,
    ):
        super().__init__()
        in_dim = num_dense + num_sparse * dim
        # fused path: pad the wide feature vector to a multiple of 32 so the
        # MFMA kernels' K/N constraints hold end-to-end (zero features are
        # inert through cross layers, deep tower and head)
        self.fused = fused
        self.pad = (-in_dim) % 32 if fused else 0
        in_dim += self.pad
        self.in_dim = in_dim
        self.cross = nn.ModuleList(
            [CrossLayerV2(in_dim, rank=cross_rank, fused=fused)
             for _ in range(num_cross)]
        )
        if fused:
            from persia_amd.ops.dense import FusedMLP

            self.deep = FusedMLP([in_dim] + list(deep), last_relu=True)
        else:
            layers: List[nn.Module] = []
            sizes = [in_dim] + list(deep)
            for i in range(len(sizes) - 1):
                layers += [nn.Linear(sizes[i], sizes[i + 1]), nn.ReLU()]
            self.deep = nn.Sequential(*layers)
        self.head = nn.Linear(in_dim + list(deep)[-1], 1)

    def forward(self, non_id_tensors, embedding_tensors) -> torch.Tensor:
        dense = (
            non_id_tensors if torch.is_tensor(non_id_tensors) else non_id_tensors[0]
        )
        # keep the wide cross features in the compute dtype (bf16 on GPU):
        # the [B, 1677] elementwise chain in f32 doubles memory traffic and
        # inserts a cast per op
        dt = torch.bfloat16 if dense.is_cuda else torch.float32
        dense = dense.to(dt)
        if torch.is_tensor(embedding_tensors):
            # packed slot-major [S*B, D] from the engine's fused sum output
            B = dense.shape[0]
            S = embedding_tensors.shape[0] // B
            emb = (
                embedding_tensors.view(S, B, -1).permute(1, 0, 2).reshape(B, -1).to(dt)
            )
            x0 = torch.cat([dense, emb], dim=1)
        else:
            x0 = torch.cat(
                [dense] + [e.flatten(1).to(dt) for e in embedding_tensors], dim=1
            )
        if self.pad:
            x0 = torch.nn.functional.pad(x0, (0, self.pad))
        xl = x0
        for layer in self.cross:
            xl = layer(x0, xl)
        d = self.deep(x0)
        return self.head(torch.cat([xl, d.to(xl.dtype)], dim=1)).squeeze(1).float()
