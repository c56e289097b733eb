"""DLRM (Deep Learning Recommendation Model) dense tower.

The benchmark model of BASELINE.json configs 2-3 (26 sparse slots, 13 dense
features, dim=128 — the Criteo shape).  Bottom MLP embeds the dense
features, a dot-product feature interaction crosses all embedding vectors,
and a top MLP produces the CTR logit.
"""
from typing import List

import torch
import torch.nn as nn


class _BlasLinearFn(torch.autograd.Function):
    """nn.Linear math with the bias gradient on the coalesced split-M kernel
    (csrc/dense.hip bias_grad) instead of torch's generic strided reduce."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        return torch.nn.functional.linear(x, w, b)

    @staticmethod
    def backward(ctx, g):
        from persia_amd.ops import native

        x, w = ctx.saved_tensors
        g = g.contiguous()
        dx = g @ w
        dw = g.t() @ x
        db = native().bias_grad(g).to(w.dtype)
        return dx, dw, db


class PALinear(nn.Linear):
    """nn.Linear drop-in (same parameters/state_dict); routes the backward
    bias reduction to the native kernel on GPU.  Gated by PA_BLAS_LINEAR=1."""

    def forward(self, x):
        import os

        if (
            x.is_cuda and x.dim() == 2 and self.bias is not None
            and x.dtype == self.weight.dtype and x.dtype == torch.bfloat16
            and os.environ.get("PA_BLAS_LINEAR", "0") == "1"
        ):
            from persia_amd.ops import native_available

            if native_available():
                return _BlasLinearFn.apply(x, self.weight, self.bias)
        return super().forward(x)


def _mlp(sizes: List[int], last_relu: bool = False) -> nn.Sequential:
    layers: List[nn.Module] = []
    for i in range(len(sizes) - 1):
        layers.append(PALinear(sizes[i], sizes[i + 1]))
        if i < len(sizes) - 2 or last_relu:
            layers.append(nn.ReLU())
    return nn.Sequential(*layers)


class _InteractFn(torch.autograd.Function):
    """Fused pairwise-dot interaction (csrc/interact.hip): hipBLASLt runs
    this batched tiny GEMM at <10 TF; the fused wave-per-sample kernel is
    memory-bound on one read of V."""

    @staticmethod
    def forward(ctx, v):
        from persia_amd.ops import native

        v = v.to(torch.bfloat16).contiguous()
        ctx.save_for_backward(v)
        return native().interact_fwd(v)

    @staticmethod
    def backward(ctx, g):
        from persia_amd.ops import native

        (v,) = ctx.saved_tensors
        return native().interact_bwd(g.to(torch.bfloat16).contiguous(), v)


class _InteractPackedFn(torch.autograd.Function):
    """Packed-input interaction: consumes the bottom-MLP output x (bf16
    [B, D]) and the engine's slot-major f16 sum base ([S*B, D]) DIRECTLY —
    no [B, F, D] cat/permute/cast materialization on either pass."""

    @staticmethod
    def forward(ctx, x, base):
        from persia_amd.ops import native

        x = x.contiguous()
        base = base.contiguous()
        ctx.save_for_backward(x, base)
        return native().interact_fwd_packed(x, base)

    @staticmethod
    def backward(ctx, g):
        from persia_amd.ops import native

        x, base = ctx.saved_tensors
        dx, dbase = native().interact_bwd_packed(g, x, base)
        return dx, dbase


class DotInteraction(nn.Module):
    """Pairwise dot products of the (num_slots+1) feature vectors, strict
    lower triangle in torch.tril_indices order."""

    def forward(self, vectors: torch.Tensor) -> torch.Tensor:
        # vectors: [B, F, D]
        F, D = vectors.shape[1], vectors.shape[2]
        if vectors.is_cuda and D % 8 == 0:
            from persia_amd.ops import native, native_available

            if native_available() and native().interact_feasible(F, D):
                return _InteractFn.apply(vectors)
        B, F, _D = vectors.shape
        prod = torch.bmm(vectors, vectors.transpose(1, 2))  # [B, F, F]
        li, lj = torch.tril_indices(F, F, offset=-1, device=vectors.device)
        return prod[:, li, lj]  # [B, F*(F-1)/2]


class DLRM(nn.Module):
    def __init__(
        self,
        num_sparse: int = 26,
        num_dense: int = 13,
        dim: int = 128,
        bottom_mlp: List[int] = (512, 256, 128),
        top_mlp: List[int] = (1024, 1024, 512, 256),
        fused: bool = False,
    ):
        super().__init__()
        self.dim = dim
        self.num_sparse = num_sparse
        n_f = num_sparse + 1
        inter_out = n_f * (n_f - 1) // 2
        if fused:
            # hand-written CDNA4 MFMA GEMM+bias+ReLU layers (GPU only)
            from persia_amd.ops.dense import FusedMLP

            self.bottom = FusedMLP([num_dense] + list(bottom_mlp) + [dim], last_relu=True)
            self.top = FusedMLP([dim + inter_out] + list(top_mlp) + [1])
        else:
            self.bottom = _mlp([num_dense] + list(bottom_mlp) + [dim], last_relu=True)
            self.top = _mlp([dim + inter_out] + list(top_mlp) + [1])
        self.interaction = DotInteraction()

    def forward(
        self, non_id_tensors, embedding_tensors
    ) -> torch.Tensor:
        dense = (
            non_id_tensors if torch.is_tensor(non_id_tensors) else non_id_tensors[0]
        )
        # follow the tower's weight dtype: f32 normally (autocast inserts the
        # bf16 casts), bf16 when the whole model runs in bf16 with f32 master
        # weights (bench graph path — no autocast, no per-layer casts)
        dt = next(self.bottom.parameters()).dtype
        dense = dense.to(dt)
        x = self.bottom(dense)  # [B, D]
        if torch.is_tensor(embedding_tensors):
            B = x.shape[0]
            if (
                x.is_cuda
                and x.dtype == torch.bfloat16
                and embedding_tensors.dtype == torch.float16
                and self.dim % 8 == 0
            ):
                from persia_amd.ops import native, native_available

                if native_available() and native().interact_packed_feasible(
                    self.num_sparse + 1, self.dim
                ):
                    inter = _InteractPackedFn.apply(x, embedding_tensors)
                    # build the top input PRE-padded to the fused GEMM's
                    # K%128 by catting the zero tail here — one cat pass
                    # instead of cat + a full-width _pad32 copy inside the
                    # first top layer (zero K-columns hit zero-padded weight
                    # columns: bit-identical output)
                    k = x.shape[1] + inter.shape[1]
                    pad = (-k) % 128
                    parts = [x, inter]
                    if pad:
                        parts.append(x.new_zeros(B, pad))
                    return self.top(torch.cat(parts, dim=1)).squeeze(1)
            # packed slot-major [S*B, D] (the engine's fused sum output) —
            # ONE reshape instead of a 26-way stack
            emb = (
                embedding_tensors.view(self.num_sparse, B, self.dim)
                .permute(1, 0, 2)
                .to(x.dtype)
            )
            vectors = torch.cat([x.unsqueeze(1), emb], dim=1)
        else:
            vectors = torch.stack([x] + [e.to(x.dtype) for e in embedding_tensors], dim=1)
        inter = self.interaction(vectors)
        out = self.top(torch.cat([x, inter], dim=1))
        return out.squeeze(1)  # logits
