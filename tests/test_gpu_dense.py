"""Numerics of the fused MFMA dense kernels vs plain-torch fp32 references."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _dev():
    return torch.device("cuda", 0)


@pytest.mark.parametrize("shape", [(256, 64, 128), (512, 96, 256), (4096, 480, 1024),
                                   (8192, 512, 1024), (8192, 1024, 1024)])
@pytest.mark.parametrize("act", [0, 1])
def test_gemm_nt_bias_act_matches_torch(shape, act):
    from persia_amd.ops import native

    C = native()
    M, K, N = shape
    torch.manual_seed(0)
    # asymmetric operands (transpose-detecting — guide §5.4 rule 16)
    A = (torch.randn(M, K, device=_dev()) * 0.5).to(torch.bfloat16)
    B = (torch.randn(N, K, device=_dev()) * 0.5).to(torch.bfloat16)
    bias = torch.randn(N, device=_dev())
    out = C.gemm_nt_bias_act(A.contiguous(), B.contiguous(), bias, act, 0, 0)
    ref = A.float() @ B.float().t() + bias
    if act == 1:
        ref = torch.relu(ref)
    assert torch.allclose(out.float(), ref, atol=0.1 + 0.02 * np.sqrt(K), rtol=0.02), (
        f"max err {(out.float() - ref).abs().max()}"
    )
    # trans_b path: same product with B given as [K, N]
    Bt = B.t().contiguous()
    out2 = C.gemm_nt_bias_act(A.contiguous(), Bt, bias, act, 0, 1)
    assert torch.allclose(out2.float(), ref, atol=0.1 + 0.02 * np.sqrt(K), rtol=0.02), (
        f"trans_b max err {(out2.float() - ref).abs().max()}"
    )


def test_wgrad_matches_torch():
    from persia_amd.ops import native

    C = native()
    torch.manual_seed(1)
    # (8192, 1024, 1024) exercises the tr-read wgrad v2 (glds + transpose
    # reads); the smaller shapes stay on the split-M v1 kernel
    for M, N, K in [(4096, 512, 480), (1024, 64, 32), (4096, 1024, 1024),
                    (8192, 1024, 1024), (8192, 512, 512)]:
        dC = (torch.randn(M, N, device=_dev()) * 0.1).to(torch.bfloat16)
        A = (torch.randn(M, K, device=_dev()) * 0.1).to(torch.bfloat16)
        dW = C.wgrad(dC.contiguous(), A.contiguous())
        ref = dC.float().t() @ A.float()
        assert torch.allclose(dW, ref, atol=0.5, rtol=0.02), (
            f"{M}x{N}x{K}: max err {(dW - ref).abs().max()}"
        )


def test_bias_grad_and_relu_bwd():
    from persia_amd.ops import native

    C = native()
    torch.manual_seed(2)
    for M, N in ((1000, 256), (8192, 512), (777, 360), (4096, 4096)):
        dC = torch.randn(M, N, device=_dev()).to(torch.bfloat16)
        db = C.bias_grad(dC.contiguous())
        ref = dC.float().sum(0)
        assert torch.allclose(db, ref, atol=0.5 + M / 4000, rtol=0.02), (M, N)
    dC = torch.randn(1000, 256, device=_dev()).to(torch.bfloat16)
    out = torch.randn(1000, 256, device=_dev()).to(torch.bfloat16)
    geff = C.relu_bwd(dC.contiguous(), out.contiguous())
    ref = dC.float() * (out.float() > 0)
    assert torch.allclose(geff.float(), ref, atol=1e-2)


@pytest.mark.parametrize("act", [0, 1])
def test_fused_linear_autograd_matches_torch(act):
    from persia_amd.ops.dense import FusedLinearFn

    torch.manual_seed(3)
    M, K, N = 512, 200, 256  # K not %32: exercises padding
    x = torch.randn(M, K, device=_dev(), dtype=torch.bfloat16, requires_grad=True)
    w = (torch.randn(N, K, device=_dev()) * 0.05).requires_grad_(True)
    b = torch.randn(N, device=_dev(), requires_grad=True)
    out = FusedLinearFn.apply(x, w, b, act)
    g = torch.randn_like(out.float()).to(torch.bfloat16)
    out.backward(g)

    x2 = x.detach().float().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    ref = x2 @ w2.t() + b2
    if act == 1:
        ref = torch.relu(ref)
    ref.backward(g.float())
    assert torch.allclose(out.float(), ref, atol=0.3, rtol=0.05)
    if act == 0:
        # no mask ambiguity: grads must match tightly
        assert torch.allclose(x.grad.float(), x2.grad, atol=0.3, rtol=0.05)
        assert torch.allclose(w.grad, w2.grad, atol=0.5, rtol=0.05), (
            f"dw max err {(w.grad - w2.grad).abs().max()}"
        )
        assert torch.allclose(b.grad, b2.grad, atol=0.5, rtol=0.05)
    else:
        # relu mask flips at pre-activation ~0 differ legitimately between the
        # bf16 kernel and the f32 reference: compare directionally
        cos = torch.nn.functional.cosine_similarity(
            w.grad.flatten(), w2.grad.flatten(), dim=0
        )
        assert cos > 0.999, f"dw cosine {cos}"
        assert torch.allclose(x.grad.float(), x2.grad, atol=0.3, rtol=0.05)


def test_interaction_matches_torch():
    from persia_amd.ops import native

    C = native()
    torch.manual_seed(5)
    for B, F, D in [(64, 27, 128), (257, 9, 64), (1024, 27, 128), (512, 65, 8)]:
        V = (torch.randn(B, F, D, device=_dev()) * 0.5).to(torch.bfloat16).contiguous()
        out = C.interact_fwd(V)
        prod = torch.bmm(V.float(), V.float().transpose(1, 2))
        li, lj = torch.tril_indices(F, F, offset=-1, device=_dev())
        ref = prod[:, li, lj]
        assert torch.allclose(out.float(), ref, atol=0.2, rtol=0.02), (
            f"{B}x{F}x{D} fwd max err {(out.float() - ref).abs().max()}"
        )
        g = torch.randn_like(ref).to(torch.bfloat16)
        dV = C.interact_bwd(g.contiguous(), V)
        # reference dV via autograd
        V2 = V.float().detach().requires_grad_(True)
        prod2 = torch.bmm(V2, V2.transpose(1, 2))
        prod2[:, li, lj].backward(g.float())
        assert torch.allclose(dV.float(), V2.grad, atol=0.3, rtol=0.02), (
            f"{B}x{F}x{D} bwd max err {(dV.float() - V2.grad).abs().max()}"
        )


def test_interaction_packed_matches_reference():
    """Packed-input variant (x bf16 + slot-major f16 base) == the cat/cast
    construction it replaces."""
    from persia_amd.ops import native

    C = native()
    torch.manual_seed(9)
    for B, S, D in [(64, 26, 128), (512, 64, 8), (130, 12, 64)]:
        F = S + 1
        x = (torch.randn(B, D, device=_dev()) * 0.5).to(torch.bfloat16).contiguous()
        base = (torch.randn(S * B, D, device=_dev()) * 0.5).to(torch.float16).contiguous()
        out = C.interact_fwd_packed(x, base)
        emb = base.view(S, B, D).permute(1, 0, 2).float()
        V = torch.cat([x.float().unsqueeze(1), emb], dim=1)
        prod = torch.bmm(V, V.transpose(1, 2))
        li, lj = torch.tril_indices(F, F, offset=-1, device=_dev())
        ref = prod[:, li, lj]
        assert torch.allclose(out.float(), ref, atol=0.2, rtol=0.02), (
            f"{B}x{S}x{D} fwd max err {(out.float() - ref).abs().max()}"
        )
        g = torch.randn_like(ref).to(torch.bfloat16)
        dx, dbase = C.interact_bwd_packed(g.contiguous(), x, base)
        V2 = V.detach().requires_grad_(True)
        prod2 = torch.bmm(V2, V2.transpose(1, 2))
        prod2[:, li, lj].backward(g.float())
        ref_dx = V2.grad[:, 0]
        ref_dbase = V2.grad[:, 1:].permute(1, 0, 2).reshape(S * B, D)
        assert torch.allclose(dx.float(), ref_dx, atol=0.3, rtol=0.02), (
            f"{B}x{S}x{D} dx max err {(dx.float() - ref_dx).abs().max()}"
        )
        assert torch.allclose(dbase.float(), ref_dbase, atol=0.3, rtol=0.02), (
            f"{B}x{S}x{D} dbase max err {(dbase.float() - ref_dbase).abs().max()}"
        )


def test_interaction_autograd_in_dlrm_path():
    from persia_amd.models.dlrm import DotInteraction

    torch.manual_seed(6)
    B, F, D = 128, 9, 64
    v = torch.randn(B, F, D, device=_dev(), dtype=torch.bfloat16, requires_grad=True)
    out = DotInteraction()(v)
    out.sum().backward()
    v2 = v.detach().float().requires_grad_(True)
    prod = torch.bmm(v2, v2.transpose(1, 2))
    li, lj = torch.tril_indices(F, F, offset=-1, device=_dev())
    prod[:, li, lj].sum().backward()
    assert torch.allclose(v.grad.float(), v2.grad, atol=0.3, rtol=0.02)


def test_fused_mlp_matches_torch_training():
    """Train FusedMLP and an identically-initialized torch MLP on the same
    data: the loss trajectories must track each other (bf16-level agreement)."""
    from persia_amd.ops.dense import FusedMLP

    torch.manual_seed(4)
    mlp = FusedMLP([64, 256, 128, 32], last_relu=False).to(_dev())

    ref = torch.nn.Sequential(
        torch.nn.Linear(64, 256), torch.nn.ReLU(),
        torch.nn.Linear(256, 128), torch.nn.ReLU(),
        torch.nn.Linear(128, 32),
    ).to(_dev())
    with torch.no_grad():
        for fl, tl in zip(mlp.layers, [ref[0], ref[2], ref[4]]):
            tl.weight.copy_(fl.weight)
            tl.bias.copy_(fl.bias)

    opt = torch.optim.SGD(mlp.parameters(), lr=0.02)
    opt_ref = torch.optim.SGD(ref.parameters(), lr=0.02)
    x = torch.randn(2048, 64, device=_dev(), dtype=torch.bfloat16)
    proj = torch.randn(64, 32, device=_dev()) * 0.3
    target = x.float() @ proj
    losses, losses_ref = [], []
    for _ in range(60):
        out = mlp(x)
        loss = ((out.float() - target) ** 2).mean()
        opt.zero_grad(); loss.backward(); opt.step()
        losses.append(float(loss.detach()))

        out_r = ref(x.float())
        loss_r = ((out_r - target) ** 2).mean()
        opt_ref.zero_grad(); loss_r.backward(); opt_ref.step()
        losses_ref.append(float(loss_r.detach()))
    # same trajectory within bf16 tolerance; same total improvement
    assert abs(losses[-1] - losses_ref[-1]) < 0.15 * losses_ref[0], (
        losses[::15], losses_ref[::15]
    )
    assert losses[-1] < losses[0], losses[::15]


def test_fused_dcn_matches_eager():
    """DCNv2 fused (MFMA cross + deep tower) vs eager with IDENTICAL
    weights (copied; the fused model pads in-features with zero columns):
    same math within bf16 tolerance, gradients flow to all weights."""
    from persia_amd.models import DCNv2

    torch.manual_seed(0)
    B, S, D = 256, 6, 64
    dense = torch.randn(B, 13, device=_dev())
    embs = [torch.randn(B, D, device=_dev(), dtype=torch.float16) for _ in range(S)]
    me = DCNv2(num_sparse=S, num_dense=13, dim=D, fused=False).to(_dev())
    mf = DCNv2(num_sparse=S, num_dense=13, dim=D, fused=True).to(_dev())
    in_e, in_p = me.in_dim, mf.in_dim
    with torch.no_grad():
        pe, pf = list(me.parameters()), list(mf.parameters())
        assert len(pe) == len(pf)
        for a, b in zip(pe[:-2], pf[:-2]):  # all but the head
            # fused shapes pad BOTH axes that carry the feature width
            # (cross u outputs the padded width); zero rows/cols are inert
            b.zero_()
            if a.dim() == 2:
                b[: a.shape[0], : a.shape[1]].copy_(a)
            else:
                b[: a.shape[0]].copy_(a)
        # head: column blocks move with the pad (cat([xl_pad, d]))
        hw_e, hb_e = pe[-2], pe[-1]
        hw_f, hb_f = pf[-2], pf[-1]
        hw_f.zero_()
        hw_f[:, :in_e].copy_(hw_e[:, :in_e])
        hw_f[:, in_p:].copy_(hw_e[:, in_e:])
        hb_f.copy_(hb_e)
    me.bfloat16()
    mf.bfloat16()

    def run(m):
        out = m(dense, [e.clone().requires_grad_(True) for e in embs])
        out.sum().backward()
        grads = [p.grad.float().norm().item() for p in m.parameters()
                 if p.grad is not None]
        return out.detach().float(), grads

    o_e, g_e = run(me)
    o_f, g_f = run(mf)
    cos = torch.nn.functional.cosine_similarity(o_e.view(-1), o_f.view(-1), dim=0)
    assert cos > 0.99, f"fused/eager cosine {cos}"
    assert torch.allclose(o_e, o_f, atol=1.0, rtol=0.1), (
        (o_e - o_f).abs().max().item()
    )
    assert all(g > 0 for g in g_f), "fused DCN: some weight got no gradient"


def test_sideband_grad_accumulation_matches_standard():
    """FusedLinear with _sideband_grad targets: the atomically-accumulated
    flat f32 gradients must equal the standard autograd path's."""
    from persia_amd.ops.dense import FusedLinear

    torch.manual_seed(11)
    M, K, N = 1024, 480, 512
    x = torch.randn(M, K, device=_dev(), dtype=torch.bfloat16)
    layer = FusedLinear(K, N, relu=True).to(_dev())
    layer.bfloat16()  # bias stays f32 (_apply override)
    assert layer.bias.dtype == torch.float32

    # standard path
    out = layer(x)
    g = torch.randn_like(out.float()).to(torch.bfloat16)
    out.backward(g)
    dw_std = layer.weight.grad.float().clone()
    db_std = layer.bias.grad.clone()

    # sideband path
    layer.weight.grad = None
    layer.bias.grad = None
    wslot = torch.zeros(N, K, dtype=torch.float32, device=_dev())
    bslot = torch.zeros(N, dtype=torch.float32, device=_dev())
    layer.weight._sideband_grad = wslot
    layer.bias._sideband_grad = bslot
    out2 = layer(x)
    out2.backward(g)
    assert layer.weight.grad is None and layer.bias.grad is None
    assert torch.allclose(wslot, dw_std, atol=0.5, rtol=0.05), (
        (wslot - dw_std).abs().max().item()
    )
    assert torch.allclose(bslot, db_std, atol=0.5, rtol=0.02)
    # accumulation: second backward ADDS
    out3 = layer(x)
    out3.backward(g)
    assert torch.allclose(wslot, 2 * dw_std, atol=1.0, rtol=0.05)
    del layer.weight._sideband_grad, layer.bias._sideband_grad


def test_fused_bce_matches_torch():
    from persia_amd.ops.dense import fused_bce_with_logits

    torch.manual_seed(13)
    z = torch.randn(8192, device=_dev(), dtype=torch.bfloat16, requires_grad=True)
    y = (torch.rand(8192, device=_dev()) > 0.5).float()
    loss = fused_bce_with_logits(z, y)
    loss.backward()
    z2 = z.detach().float().requires_grad_(True)
    ref = torch.nn.functional.binary_cross_entropy_with_logits(z2, y)
    ref.backward()
    assert torch.allclose(loss, ref, atol=1e-4, rtol=1e-4)
    assert torch.allclose(z.grad.float(), z2.grad.to(torch.bfloat16).float(),
                          atol=1e-6)
