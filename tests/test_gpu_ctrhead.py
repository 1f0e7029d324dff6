"""GPU numerics: fused CTR head vs the plain fp32 torch composition
(forward values and every input gradient)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _torch_head(e_all, dense, w, b, use_fm):
    dim = e_all.shape[2] - 1
    e = e_all[..., :dim]
    lin = e_all[..., dim]
    deep_in = torch.cat([e.flatten(1), dense], dim=1)
    partial = lin.sum(dim=1) + dense @ w.reshape(-1) + b.reshape(())
    if use_fm:
        s = e.sum(dim=1)
        partial = partial + 0.5 * (s * s - (e * e).sum(dim=1)).sum(dim=1)
    return deep_in, partial


@pytest.mark.parametrize("use_fm", [True, False])
@pytest.mark.parametrize("out_bf16", [False, True])
@pytest.mark.parametrize("dim,nd", [(9, 13), (4, 13), (63, 7), (64, 13), (127, 5)])
def test_head_matches_torch(use_fm, out_bf16, dim, nd):
    from openembedding_amd.models.ctr import _FusedCTRHeadFn

    torch.manual_seed(0)
    B, F = 257, 26
    e_all = torch.randn(B, F, dim + 1, device=DEV, requires_grad=True)
    dense = torch.rand(B, nd, device=DEV, requires_grad=True)
    w = torch.randn(1, nd, device=DEV, requires_grad=True)
    b = torch.randn(1, device=DEV, requires_grad=True)

    deep_in, partial = _FusedCTRHeadFn.apply(e_all, dense, w, b, use_fm,
                                             out_bf16)
    ref_in, ref_p = _torch_head(e_all, dense, w, b, use_fm)

    tol = 2e-2 if out_bf16 else 1e-5
    assert torch.allclose(deep_in.float(), ref_in, atol=tol, rtol=tol)
    assert torch.allclose(partial, ref_p, atol=2e-4, rtol=1e-4)

    g_in = torch.randn_like(ref_in)
    g_p = torch.randn_like(ref_p)
    (deep_in.float() * g_in).sum().backward(retain_graph=True)
    (partial * g_p).sum().backward()
    got = [t.grad.clone() for t in (e_all, dense, w, b)]
    for t in (e_all, dense, w, b):
        t.grad = None
    (ref_in * g_in).sum().backward(retain_graph=True)
    (ref_p * g_p).sum().backward()
    names = ["e_all", "dense", "w", "b"]
    for name, gg, t in zip(names, got, (e_all, dense, w, b)):
        atol = 5e-2 if out_bf16 else 1e-3  # atomic order + bf16 grad
        assert torch.allclose(gg, t.grad, atol=atol, rtol=1e-3), name


def test_deepfm_fused_matches_plain():
    """Whole-model check: DeepFM forward with fused head equals the torch
    path bit-for... closely (fp32 head)."""
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM, synthetic_batch

    torch.manual_seed(0)
    model = DeepFM(dim=9).to(DEV)
    dense, sparse, labels = synthetic_batch(512, device=DEV)
    with torch.no_grad():
        # materialize rows once so both paths read identical weights
        model.embedding.variable.sparse_read(
            sparse + model.embedding.field_offsets)
    out_fused = model(dense, sparse)
    try:
        model._use_fused_head = lambda t: False
        out_plain = model(dense, sparse)
    finally:
        del model._use_fused_head
    assert torch.allclose(out_fused, out_plain, atol=1e-4, rtol=1e-4)
