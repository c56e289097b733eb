import numpy as np
import torch

from persia_amd.ops import reference as R


def test_segment_sum_rows_matches_naive():
    torch.manual_seed(0)
    U, dim = 20, 8
    rows = torch.randn(U, dim)
    inverse = torch.tensor([0, 1, 1, 5, 7, 7, 7, 2], dtype=torch.int64)
    seg_offsets = torch.tensor([0, 3, 3, 7, 8], dtype=torch.int64)  # segs of len 3,0,4,1
    out = R.segment_sum_rows(rows, inverse, seg_offsets, sqrt_scaling=False, out_dtype=torch.float32)
    assert out.shape == (4, dim)
    expected0 = rows[0] + rows[1] + rows[1]
    assert torch.allclose(out[0], expected0)
    assert torch.all(out[1] == 0)
    assert torch.allclose(out[2], rows[5] + rows[7] * 3)
    assert torch.allclose(out[3], rows[2])
    # sqrt scaling divides by sqrt(len)
    out_s = R.segment_sum_rows(rows, inverse, seg_offsets, sqrt_scaling=True, out_dtype=torch.float32)
    assert torch.allclose(out_s[0], expected0 / np.sqrt(3))
    assert torch.allclose(out_s[3], rows[2])


def test_raw_embedding_tensors_contract():
    rows = torch.arange(12, dtype=torch.float32).view(3, 4)  # 3 distinct rows
    inverse = torch.tensor([0, 1, 2, 1], dtype=torch.int64)
    seg_offsets = torch.tensor([0, 2, 2, 4], dtype=torch.int64)  # B=3: [2,0,2]
    distinct, index, non_empty, num = R.raw_embedding_tensors(
        rows, inverse, seg_offsets, sample_fixed_size=3, out_dtype=torch.float32
    )
    assert distinct.shape == (4, 4)
    assert torch.all(distinct[0] == 0)  # padding row
    assert torch.allclose(distinct[1:], rows)
    # sample 0: ids at cols 0,1 -> distinct 0,1 (+1)
    assert index.tolist() == [1, 2, 0, 0, 0, 0, 3, 2, 0]
    assert non_empty.tolist() == [0, 1, 6, 7]
    assert num.tolist() == [2, 0, 2]


def test_raw_truncation():
    rows = torch.ones(5, 2)
    inverse = torch.tensor([0, 1, 2, 3, 4], dtype=torch.int64)
    seg_offsets = torch.tensor([0, 5], dtype=torch.int64)
    _d, index, _ne, num = R.raw_embedding_tensors(
        rows, inverse, seg_offsets, sample_fixed_size=3, out_dtype=torch.float32
    )
    assert index.tolist() == [1, 2, 3]
    assert num.tolist() == [3]


def test_segment_grad_scatter_matches_naive():
    torch.manual_seed(1)
    n_seg, dim, U = 4, 8, 6
    grads = torch.randn(n_seg, dim)
    inverse = torch.tensor([0, 1, 1, 5, 3, 3, 3, 2], dtype=torch.int64)
    seg_offsets = torch.tensor([0, 3, 3, 7, 8], dtype=torch.int64)
    out = R.segment_grad_scatter(grads, inverse, seg_offsets, U)
    # position k in segment s contributes grads[s] to inverse[k]
    expected = torch.zeros(U, dim)
    seg_of = [0, 0, 0, 2, 2, 2, 2, 3]
    for k in range(8):
        expected[inverse[k]] += grads[seg_of[k]]
    assert torch.allclose(out, expected, atol=1e-6)
    # loss scale
    out2 = R.segment_grad_scatter(grads, inverse, seg_offsets, U, scale_factor=2.0)
    assert torch.allclose(out2, expected / 2.0, atol=1e-6)
    # sqrt scaling scales each contribution by 1/sqrt(seg_len)
    out3 = R.segment_grad_scatter(grads, inverse, seg_offsets, U, sqrt_scaling=True)
    expected3 = torch.zeros(U, dim)
    seg_len = [3, 0, 4, 1]
    for k in range(8):
        expected3[inverse[k]] += grads[seg_of[k]] / np.sqrt(seg_len[seg_of[k]])
    assert torch.allclose(out3, expected3, atol=1e-6)


def test_optimizer_updates_match_reference_formulas():
    # persia-simd/src/lib.rs formulas, exact-math variants
    torch.manual_seed(2)
    n, dim = 5, 4
    g = torch.randn(n, dim)

    w = torch.randn(n, dim)
    w0 = w.clone()
    R.sgd_update(w, g, lr=0.1, wd=0.01, weight_bound=10.0)
    assert torch.allclose(w, w0 - 0.1 * (g + 0.01 * w0), atol=1e-6)

    w = torch.randn(n, dim)
    acc = torch.full((n, dim), 0.01)
    w0, acc0 = w.clone(), acc.clone()
    R.adagrad_update(w, acc, g, lr=0.1, g_square_momentum=0.9, eps=1e-10, weight_bound=10.0)
    assert torch.allclose(w, w0 - 0.1 * g * torch.rsqrt(acc0 + 1e-10), atol=1e-5)
    assert torch.allclose(acc, acc0 * 0.9 + g * g, atol=1e-6)

    # vectorwise-shared adagrad: scalar accumulator, mean g^2
    w = torch.randn(n, dim)
    acc = torch.full((n, 1), 0.01)
    w0, acc0 = w.clone(), acc.clone()
    R.adagrad_update(
        w, acc, g, lr=0.1, g_square_momentum=1.0, eps=1e-10, weight_bound=10.0,
        vectorwise_shared=True,
    )
    assert torch.allclose(w, w0 - 0.1 * g * torch.rsqrt(acc0 + 1e-10), atol=1e-5)
    assert torch.allclose(acc, acc0 + (g * g).mean(dim=1, keepdim=True), atol=1e-6)

    w = torch.randn(n, dim)
    m = torch.zeros(n, dim)
    v = torch.zeros(n, dim)
    w0 = w.clone()
    b1, b2 = 0.9, 0.999
    R.adam_update(w, m, v, g, b1, b2, lr=0.001, beta1=b1, beta2=b2, eps=1e-8, weight_bound=10.0)
    m_exp = (1 - b1) * g
    v_exp = (1 - b2) * g * g
    assert torch.allclose(m, m_exp, atol=1e-6)
    assert torch.allclose(v, v_exp, atol=1e-6)
    step = 0.001 * (m_exp / (1 - b1)) / (1e-8 + (v_exp / (1 - b2)).sqrt())
    assert torch.allclose(w, w0 - step, atol=1e-5)


def test_weight_bound_clamps():
    w = torch.tensor([[100.0, -100.0]])
    g = torch.zeros(1, 2)
    R.sgd_update(w, g, lr=0.1, wd=0.0, weight_bound=10.0)
    assert w.abs().max() <= 10.0
