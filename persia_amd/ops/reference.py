"""Pure-torch fp32 reference implementations of every sparse op.

These define the numerics contract the HIP kernels are tested against
(tests/test_gpu_kernels.py compares each HIP kernel with these on the same
inputs).  They also power the CPU execution path (job on a machine without a
GPU — reference parity: the adult-income CPU run).

Math sources (reference implementation):
* segment sum + sqrt scaling  — embedding_worker_service/mod.rs:486-629
* gradient scatter + scaling  — embedding_worker_service/mod.rs:703-872
* sparse optimizers           — rust/persia-simd/src/lib.rs (exact formulas)
"""
from typing import Optional, Tuple

import torch


# ---------------------------------------------------------------- forward ops


def segment_sum_rows(
    rows: torch.Tensor,  # [U, dim] f32 — unique rows
    inverse: torch.Tensor,  # [nnz] i64 — position -> unique row index
    seg_offsets: torch.Tensor,  # [n_seg + 1] i64 — CSR offsets of positions per segment
    sqrt_scaling: bool = False,
    out_dtype: torch.dtype = torch.float16,
) -> torch.Tensor:
    """out[s] = sum_{k in segment s} rows[inverse[k]]; optionally scaled by
    1/sqrt(max(len(segment), 1))."""
    n_seg = seg_offsets.numel() - 1
    dim = rows.shape[1]
    lens = (seg_offsets[1:] - seg_offsets[:-1]).to(rows.device)
    seg_ids = torch.repeat_interleave(
        torch.arange(n_seg, device=rows.device, dtype=torch.int64), lens
    )
    out = torch.zeros(n_seg, dim, dtype=torch.float32, device=rows.device)
    out.index_add_(0, seg_ids, rows.float()[inverse])
    if sqrt_scaling:
        scale = lens.clamp(min=1).float().rsqrt().unsqueeze(1)
        out = out * scale
    return out.to(out_dtype)


def raw_embedding_tensors(
    rows: torch.Tensor,  # [U_slot, dim] f32 — this slot's distinct rows (order = first occurrence in sorted-unique order)
    inverse: torch.Tensor,  # [nnz_slot] i64 — position -> distinct row index (0-based)
    seg_offsets: torch.Tensor,  # [B+1] i64
    sample_fixed_size: int,
    scale: float = 1.0,
    out_dtype: torch.dtype = torch.float16,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Build the reference raw-slot tensor contract
    (embedding_worker_service/mod.rs:593-629 + persia-core forward.rs:333-394):

    returns (distinct_with_pad [U+1, dim] out_dtype — row 0 zeros,
             index [B*sfs] i64 — 0 = padding, else distinct_idx+1,
             non_empty_index [k] i64 — positions in index that are non-zero,
             sample_id_num [B] i64).

    Note: when one sample holds more than ``sample_fixed_size`` ids, only the
    first ``sample_fixed_size`` are indexed (reference truncation).  A
    distinct id appearing twice in one sample occupies both columns.
    """
    device = rows.device
    B = seg_offsets.numel() - 1
    U = rows.shape[0]
    dim = rows.shape[1]
    distinct = torch.zeros(U + 1, dim, dtype=torch.float32, device=device)
    distinct[1:] = rows.float() * scale
    lens = seg_offsets[1:] - seg_offsets[:-1]
    index = torch.zeros(B * sample_fixed_size, dtype=torch.int64, device=device)
    # column index of each position within its sample
    pos = torch.arange(inverse.numel(), device=device, dtype=torch.int64)
    col = pos - seg_offsets[:-1].repeat_interleave(lens)
    sample = torch.arange(B, device=device, dtype=torch.int64).repeat_interleave(lens)
    keep = col < sample_fixed_size
    index[sample[keep] * sample_fixed_size + col[keep]] = inverse[keep] + 1
    non_empty_index = torch.nonzero(index, as_tuple=False).view(-1)
    sample_id_num = lens.clamp(max=sample_fixed_size)
    return distinct.to(out_dtype), index, non_empty_index, sample_id_num


# --------------------------------------------------------------- backward ops


def segment_grad_scatter(
    grads: torch.Tensor,  # [n_seg, dim] f16/f32 — per-(slot,sample) grads
    inverse: torch.Tensor,  # [nnz] i64 — position -> unique row index
    seg_offsets: torch.Tensor,  # [n_seg+1] i64
    n_unique: int,
    scale_factor: float = 1.0,
    sqrt_scaling: bool = False,
) -> torch.Tensor:
    """grad_out[u] = sum_{k: inverse[k]==u} grads[segment(k)] * scale
    where scale = (1/scale_factor) * (sqrt_scaling ? 1/sqrt(len(segment(k))) : 1).

    Mirror of the reference's per-sign AVX2 grad accumulation
    (mod.rs:782-814) with loss-scale recip (mod.rs:751-755) and sqrt scaling
    (mod.rs:757-777)."""
    n_seg = seg_offsets.numel() - 1
    dim = grads.shape[1]
    lens = (seg_offsets[1:] - seg_offsets[:-1]).to(grads.device)
    seg_ids = torch.repeat_interleave(
        torch.arange(n_seg, device=grads.device, dtype=torch.int64), lens
    )
    g = grads.float()
    if scale_factor != 1.0:
        g = g / scale_factor
    if sqrt_scaling:
        g = g * lens.clamp(min=1).float().rsqrt().unsqueeze(1)
    out = torch.zeros(n_unique, dim, dtype=torch.float32, device=grads.device)
    out.index_add_(0, inverse, g[seg_ids])
    return out


# ------------------------------------------------------- sparse optimizer math
# Row layout (reference emb_entry.rs:17-23): inner = [emb(dim) | opt_state].


def sgd_update(
    emb: torch.Tensor, grad: torch.Tensor, lr: float, wd: float, weight_bound: float
) -> None:
    """w -= lr*(g + wd*w); clamp ±weight_bound (persia-simd lib.rs:124-144, 231)."""
    emb -= lr * (grad + wd * emb)
    if weight_bound > 0:
        emb.clamp_(-weight_bound, weight_bound)


def adagrad_update(
    emb: torch.Tensor,  # [n, dim] f32
    accum: torch.Tensor,  # [n, dim] f32 (or [n, 1] vectorwise-shared)
    grad: torch.Tensor,  # [n, dim] f32
    lr: float,
    g_square_momentum: float,
    eps: float,
    weight_bound: float,
    vectorwise_shared: bool = False,
) -> None:
    """w -= lr * g * rsqrt(acc_old + eps);
    acc = acc_old * momentum + g²  (per-dim) or mean(g²) (shared)
    (persia-simd lib.rs:21-121 — accumulator is read BEFORE its update)."""
    if vectorwise_shared:
        emb -= lr * grad * torch.rsqrt(accum + eps)
        gsq = (grad * grad).mean(dim=1, keepdim=True)
        accum.mul_(g_square_momentum).add_(gsq)
    else:
        emb -= lr * grad * torch.rsqrt(accum + eps)
        accum.mul_(g_square_momentum).add_(grad * grad)
    if weight_bound > 0:
        emb.clamp_(-weight_bound, weight_bound)


def adam_update(
    emb: torch.Tensor,
    m: torch.Tensor,
    v: torch.Tensor,
    grad: torch.Tensor,
    beta1_power: float,
    beta2_power: float,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_bound: float,
) -> None:
    """Bias-corrected Adam (persia-simd lib.rs:147-228); β-powers are
    maintained per feature group per update batch (persia-common optim.rs:147-216)."""
    m.mul_(beta1).add_(grad, alpha=1.0 - beta1)
    v.mul_(beta2).add_(grad * grad, alpha=1.0 - beta2)
    m_hat = m / (1.0 - beta1_power)
    v_hat = v / (1.0 - beta2_power)
    emb -= lr * m_hat / (eps + v_hat.sqrt())
    if weight_bound > 0:
        emb.clamp_(-weight_bound, weight_bound)
