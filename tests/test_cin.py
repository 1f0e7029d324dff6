"""CIN implicit-outer-product layer vs the einsum oracle (CPU).

The reference's CIN (DeepCTR, used by the xDeepFM benchmark row) computes
xk+1 = relu(conv1d(outer(x0, xk))); _CINLayerFn computes the same
multilinear map without materializing the outer-product tensor. Checked
here for forward values and all three gradients."""

import torch

from openembedding_amd.models.ctr import CIN, _CINLayerFn


def _oracle(x0, xk, W):
    B, F, d = x0.shape
    H = xk.shape[1]
    z = torch.einsum("bfd,bhd->bfhd", x0, xk).reshape(B, F * H, d)
    return torch.einsum("ok,bkd->bod", W, z)


def test_cin_layer_matches_einsum():
    g = torch.Generator().manual_seed(0)
    B, F, H, O, d = 32, 6, 11, 7, 5
    x0 = torch.randn(B, F, d, generator=g, requires_grad=True)
    xk = torch.randn(B, H, d, generator=g, requires_grad=True)
    W = torch.randn(O, F * H, generator=g, requires_grad=True)

    out = _CINLayerFn.apply(x0, xk, W, torch.float32)
    ref = _oracle(x0, xk, W)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-6)

    dout = torch.randn(B, O, d, generator=g)
    out.backward(dout)
    g1 = (x0.grad.clone(), xk.grad.clone(), W.grad.clone())
    for t in (x0, xk, W):
        t.grad = None
    ref2 = _oracle(x0, xk, W)
    ref2.backward(dout)
    torch.testing.assert_close(g1[0], x0.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g1[1], xk.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(g1[2], W.grad, rtol=1e-4, atol=1e-5)


def test_cin_module_trains():
    torch.manual_seed(1)
    cin = CIN(n_fields=5, dim=4, layer_sizes=(8, 8))
    e = torch.randn(16, 5, 4, requires_grad=True)
    out = cin(e)
    assert out.shape == (16,)
    out.sum().backward()
    assert e.grad is not None and torch.isfinite(e.grad).all()
    for p in cin.parameters():
        assert p.grad is not None and torch.isfinite(p.grad).all()


def test_cin_x0_equals_xk_first_layer():
    """First layer uses x0 twice (xk is x0): gradient must combine both
    paths, matching autograd through the einsum."""
    g = torch.Generator().manual_seed(2)
    B, F, d, O = 8, 4, 3, 6
    e1 = torch.randn(B, F, d, generator=g, requires_grad=True)
    e2 = e1.detach().clone().requires_grad_(True)
    W = torch.randn(O, F * F, generator=g)
    out1 = _CINLayerFn.apply(e1, e1, W, torch.float32)
    z = torch.einsum("bfd,bhd->bfhd", e2, e2).reshape(B, F * F, d)
    out2 = torch.einsum("ok,bkd->bod", W, z)
    dout = torch.randn(B, O, d, generator=g)
    out1.backward(dout)
    out2.backward(dout)
    torch.testing.assert_close(e1.grad, e2.grad, rtol=1e-4, atol=1e-5)
