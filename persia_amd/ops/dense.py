"""Fused dense layers over the CDNA4 MFMA kernels (csrc/dense.hip).

``FusedLinear`` = bf16 GEMM + bias + ReLU in ONE kernel (forward), with
hand-written dgrad / wgrad / bias-grad kernels in backward — replacing a
hipBLASLt GEMM plus an elementwise cascade per layer.  f32 master weights,
bf16 activations (the bench's dtype contract).  Since the XOR-swizzled LDS
staging fix (wgrad 113 -> 252 TF) this path BEATS the graphed hipBLASLt
dense step end-to-end and is the bench default for dim-128 towers.

Constraints (asserted): hidden widths N % 32 == 0 (dgrad reuses the GEMM
kernel with K=N); input features are zero-padded to a multiple of 32.
"""
from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def _wgrad_via_blas() -> bool:
    # default: the hand-written wgrad (XOR-swizzled LDS staging) — it beats
    # hipBLASLt 2-4x on the DLRM shapes; "blas" kept as an A/B switch
    import os

    return os.environ.get("PA_FUSED_WGRAD", "own") == "blas"


def _pad32(x: torch.Tensor, dim: int, mult: int = 128) -> torch.Tensor:
    """Zero-pad dim to a multiple of `mult` (the glds-pipelined GEMM v2
    needs K % 64 and the tr-read wgrad K % 128; zero K-columns are inert —
    the extra flops are cheaper than falling off the v2 kernels)."""
    k = x.shape[dim]
    pad = (-k) % mult
    if pad == 0:
        return x
    padding = [0, 0] * (x.dim() - 1 - dim) + [0, pad]
    return F.pad(x, padding)


class FusedLinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, act: int):
        from persia_amd.ops import native

        C = native()
        x_bf = _pad32(x.to(torch.bfloat16), 1).contiguous()
        w_bf = _pad32(weight.to(torch.bfloat16), 1).contiguous()  # [N, Kp]
        out = C.gemm_nt_bias_act(x_bf, w_bf, bias.float(), act, 0, 0)
        ctx.save_for_backward(x_bf, w_bf, out)
        ctx.act = act
        ctx.k = x.shape[1]       # dx width (input may arrive pre-padded)
        ctx.kw = weight.shape[1]  # dw width (the stored, unpadded weight)
        ctx.w_dtype = weight.dtype
        ctx.b_dtype = bias.dtype
        # side-band gradient targets (set by the bench's flat-buffer scheme):
        # when present, backward ACCUMULATES dw/db straight into these f32
        # slots via atomics and returns None — no dw materialization, cast,
        # pad-slice or AccumulateGrad add per layer
        ctx.w_side = getattr(weight, "_sideband_grad", None)
        ctx.b_side = getattr(bias, "_sideband_grad", None)
        return out

    @staticmethod
    def backward(ctx, g):
        from persia_amd.ops import native

        C = native()
        x_bf, w_bf, out = ctx.saved_tensors
        g = g.to(torch.bfloat16).contiguous()
        N = g.shape[1]
        pow2 = N >= 8 and (N & (N - 1)) == 0
        if ctx.act == 1 and pow2 and ctx.b_side is not None:
            # one fused pass: relu mask + bias column sums straight into the
            # side-band slot (saves re-reading the 16 MB masked gradient)
            g = C.relu_bwd_bias(g, out, ctx.b_side)
            db = None
        else:
            if ctx.act == 1:
                g = C.relu_bwd(g, out)
            if ctx.b_side is not None:
                C.bias_grad_into(g, ctx.b_side)
                db = None
            else:
                db = C.bias_grad(g)
                if db.dtype != ctx.b_dtype:
                    db = db.to(ctx.b_dtype)
        # dX = g @ W: trans_b path consumes the [N, Kp] weight directly; at
        # the big square shape the transposed-staging kernel loses to an
        # explicit W^T + the glds NT path (51 vs 33+8 us measured,
        # tools/dgrad_probe.py)
        if w_bf.shape[0] >= 1024 and w_bf.shape[1] >= 1024:
            dx = C.gemm_nt_bias_act(
                g, w_bf.t().contiguous(), torch.empty(0, device=g.device),
                0, 0, 0,
            )
        else:
            dx = C.gemm_nt_bias_act(
                g, w_bf, torch.empty(0, device=g.device), 0, 0, 1
            )
        if dx.shape[1] != ctx.k:
            dx = dx[:, : ctx.k]
        if ctx.w_side is not None:
            C.wgrad_into(g, x_bf, ctx.w_side)
            dw = None
        elif _wgrad_via_blas():
            # hybrid: the hand-written fwd/dgrad kernels beat hipBLASLt on
            # these shapes but the wgrad kernel trails it (transpose-staging
            # bound) — let the library run the one GEMM it wins
            dw = torch.matmul(g.t(), x_bf).to(ctx.w_dtype)
        else:
            dw = C.wgrad(g, x_bf)
            if dw.dtype != ctx.w_dtype:
                dw = dw.to(ctx.w_dtype)
        if dw is not None and dw.shape[1] != ctx.kw:
            dw = dw[:, : ctx.kw].contiguous()
        return dx, dw, db, None


class FusedLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int, relu: bool = True,
                 bias: bool = True):
        super().__init__()
        assert out_features % 32 == 0, "FusedLinear needs out_features % 32 == 0"
        self.in_features = in_features
        self.out_features = out_features
        self.relu = relu
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        if bias:
            self.bias = nn.Parameter(torch.zeros(out_features))
        else:  # constant zero bias: buffer, so parameter lists align with
            # a bias-free nn.Linear twin
            self.register_buffer("bias", torch.zeros(out_features))
        nn.init.kaiming_uniform_(self.weight, a=5 ** 0.5)

    def _apply(self, fn, recurse=True):
        # keep the bias fp32 through .bfloat16()/.half(): the GEMM epilogue
        # adds it in f32 and bias_grad produces f32 — a low-precision bias
        # only inserts per-step cast kernels on BOTH passes (the profile
        # showed ~24 bias-sized launches/step from exactly this)
        r = super()._apply(fn, recurse)
        if self.bias.dtype != torch.float32:
            if isinstance(self.bias, nn.Parameter):
                self.bias.data = self.bias.data.float()
                if self.bias.grad is not None:
                    self.bias.grad = self.bias.grad.float()
            else:
                self.bias = self.bias.float()
        return r

    def forward(self, x):
        return FusedLinearFn.apply(x, self.weight, self.bias, 1 if self.relu else 0)


class FusedMLP(nn.Module):
    """Stack of FusedLinear layers (+ optional plain final projection when the
    last width isn't a multiple of 32, e.g. the 1-logit head)."""

    def __init__(self, sizes: List[int], last_relu: bool = False):
        super().__init__()
        layers: List[nn.Module] = []
        for i in range(len(sizes) - 1):
            is_last = i == len(sizes) - 2
            relu = (not is_last) or last_relu
            if sizes[i + 1] % 32 == 0:
                layers.append(FusedLinear(sizes[i], sizes[i + 1], relu=relu))
            else:
                lin = nn.Linear(sizes[i], sizes[i + 1])
                layers.append(lin if not relu else nn.Sequential(lin, nn.ReLU()))
        self.layers = nn.ModuleList(layers)

    def forward(self, x):
        for l in self.layers:
            if not isinstance(l, FusedLinear):
                # plain final projection (e.g. the 1-logit head): follow its
                # weight dtype — FusedLinear outputs bf16 even when the
                # module holds f32 weights (mixed-precision user models)
                w = next(l.parameters(), None)
                if w is not None and x.dtype != w.dtype:
                    x = x.to(w.dtype)
            x = l(x)
        return x


class FusedBCEFn(torch.autograd.Function):
    """BCE-with-logits as 2 kernels (fwd: loss + sigmoid cache, bwd: one
    elementwise) — torch's chain is ~10 launch-bound kernels inside the
    captured step."""

    @staticmethod
    def forward(ctx, logits, label):
        from persia_amd.ops import native

        C = native()
        loss, sig = C.bce_fwd(logits.float().contiguous(),
                              label.float().contiguous())
        ctx.save_for_backward(sig, label)
        ctx.in_dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, g):
        from persia_amd.ops import native

        C = native()
        sig, label = ctx.saved_tensors
        dz = C.bce_bwd(sig, label.float(), g.reshape(1).float().contiguous())
        if dz.dtype != ctx.in_dtype:
            dz = dz.to(ctx.in_dtype)
        return dz, None


def fused_bce_with_logits(logits: torch.Tensor, label: torch.Tensor) -> torch.Tensor:
    return FusedBCEFn.apply(logits, label)
