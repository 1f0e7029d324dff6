"""GPU numerics: fused 3-layer MLP kernel vs torch reference (forward
values, activations, and all gradients through the autograd Function)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _torch_mlp(x0, ws, bs):
    a = x0
    acts = []
    for i in range(3):
        a = torch.relu(a @ ws[i].t() + bs[i])
        acts.append(a)
    out = a @ ws[3].reshape(-1) + bs[3]
    return out.float(), acts


@pytest.mark.parametrize("M,K0", [(256, 247), (512, 256), (130, 64)])
def test_mlp3_fwd_matches_torch(M, K0):
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    torch.manual_seed(0)
    H = 400
    K0p = (K0 + 31) // 32 * 32
    Hp = (H + 31) // 32 * 32
    x0 = torch.zeros(M, K0p, device=DEV, dtype=torch.bfloat16)
    x0[:, :K0] = (torch.randn(M, K0, device=DEV) * 0.5).to(torch.bfloat16)
    ws = [(torch.randn(H, K0p, device=DEV) * 0.05).to(torch.bfloat16),
          (torch.randn(H, H, device=DEV) * 0.05).to(torch.bfloat16),
          (torch.randn(H, H, device=DEV) * 0.05).to(torch.bfloat16),
          (torch.randn(1, H, device=DEV) * 0.05).to(torch.bfloat16)]
    ws[0][:, K0:] = 0
    bs = [(torch.randn(H, device=DEV) * 0.1).to(torch.bfloat16)
          for _ in range(3)] + [(torch.randn(1, device=DEV) * 0.1
                                 ).to(torch.bfloat16)]
    w2p = torch.zeros(H, Hp, device=DEV, dtype=torch.bfloat16)
    w2p[:, :H] = ws[1]
    w3p = torch.zeros(H, Hp, device=DEV, dtype=torch.bfloat16)
    w3p[:, :H] = ws[2]
    out, a1, a2, a3 = ext.mlp3_fwd(x0, ws[0], bs[0], w2p, bs[1],
                                   w3p, bs[2], ws[3].reshape(-1), bs[3],
                                   None)
    # STAGE-WISE reference: each layer recomputed from the KERNEL's own
    # previous activation, so bf16 rounding differences don't compound
    r1 = torch.relu(x0 @ ws[0].t() + bs[0])
    r2 = torch.relu(a1 @ ws[1].t() + bs[1])
    r3 = torch.relu(a2 @ ws[2].t() + bs[2])
    rout = (a3 @ ws[3].reshape(-1) + bs[3]).float()
    for got, ref, name in [(a1.float(), r1.float(), "a1"),
                           (a2.float(), r2.float(), "a2"),
                           (a3.float(), r3.float(), "a3")]:
        assert torch.allclose(got, ref, atol=2e-2, rtol=2e-2), name
    assert torch.allclose(out, rout, atol=3e-2, rtol=2e-2)


def test_fused_mlp_function_grads():
    from openembedding_amd.models.ctr import _FusedMLP3Fn
    torch.manual_seed(1)
    M, K0, K0p, H = 384, 247, 256, 400
    x0 = torch.zeros(M, K0p, device=DEV, dtype=torch.bfloat16,
                     requires_grad=True)
    with torch.no_grad():
        x0[:, :K0] = (torch.randn(M, K0, device=DEV) * 0.5
                      ).to(torch.bfloat16)
    params = []
    for shape in [(H, K0), (H,), (H, H), (H,), (H, H), (H,), (1, H), (1,)]:
        t = (torch.randn(*shape, device=DEV)
             * (0.05 if len(shape) == 2 else 0.1)).to(torch.bfloat16)
        t.requires_grad_(True)
        params.append(t)
    Hp = (H + 31) // 32 * 32
    z = lambda *s: torch.zeros(*s, device=DEV, dtype=torch.bfloat16)  # noqa: E731
    bufs = {"w1p": z(H, K0p), "w2p": z(H, Hp), "w3p": z(H, Hp),
            "w3tp": z(H, Hp), "w2tp": z(H, Hp), "w1tp": z(K0p, Hp)}
    out = _FusedMLP3Fn.apply(x0, None, *params, bufs)
    g = torch.randn(M, device=DEV)
    (out * g).sum().backward()
    got = [t.grad.clone() for t in [x0] + params]

    # torch reference with the SAME bf16 weights/activations
    for t in [x0] + params:
        t.grad = None
    x0r = x0.detach().clone().requires_grad_(True)
    pr = [t.detach().clone().requires_grad_(True) for t in params]
    a = x0r[:, :K0]
    a = torch.relu(a @ pr[0].t() + pr[1])
    a = torch.relu(a @ pr[2].t() + pr[3])
    a = torch.relu(a @ pr[4].t() + pr[5])
    ref_out = a @ pr[6].reshape(-1) + pr[7]
    (ref_out.float() * g).sum().backward()
    ref = [x0r.grad] + [t.grad for t in pr]
    names = ["x0", "w1", "b1", "w2", "b2", "w3", "b3", "w4", "b4"]
    for n, gg, rr in zip(names, got, ref):
        assert torch.allclose(gg.float(), rr.float(), atol=0.5, rtol=5e-2), \
            (n, (gg.float() - rr.float()).abs().max())


def test_deepfm_native_fused_mlp_trains():
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM, synthetic_batch
    from openembedding_amd.models.ctr import convert_mlp_bf16

    torch.manual_seed(0)
    model = convert_mlp_bf16(DeepFM(dim=9).to(DEV))
    assert model.fused_mlp
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01),
        flatten_dense=True)
    lossf = torch.nn.BCEWithLogitsLoss()
    losses = []
    for _ in range(10):
        dense, sparse, labels = synthetic_batch(2048, device=DEV)
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    assert losses[-1] < losses[0]


def test_fused_mlp_partial_fold():
    # partial carried into the final-dot epilogue == explicit add outside
    from openembedding_amd.models.ctr import _FusedMLP3Fn
    torch.manual_seed(3)
    M, K0, H = 512, 256, 400
    K0p = (K0 + 31) // 32 * 32
    Hp = (H + 31) // 32 * 32
    x0 = torch.zeros(M, K0p, device=DEV, dtype=torch.bfloat16)
    x0[:, :K0] = (torch.randn(M, K0, device=DEV) * 0.1).to(torch.bfloat16)
    params = []
    for shape in [(H, K0), (H,), (H, H), (H,), (H, H), (H,), (1, H), (1,)]:
        t = (torch.randn(*shape, device=DEV)
             * (0.05 if len(shape) == 2 else 0.1)).to(torch.bfloat16)
        t.requires_grad_(True)
        params.append(t)
    z = lambda *s: torch.zeros(*s, device=DEV, dtype=torch.bfloat16)  # noqa: E731
    bufs = {"w1p": z(H, K0p), "w2p": z(H, Hp), "w3p": z(H, Hp),
            "w3tp": z(H, Hp), "w2tp": z(H, Hp), "w1tp": z(K0p, Hp)}
    partial = torch.randn(M, device=DEV, requires_grad=True)

    out_folded = _FusedMLP3Fn.apply(x0, partial, *params, bufs)
    base = _FusedMLP3Fn.apply(x0, None, *params, bufs)
    out_added = partial + base
    assert torch.allclose(out_folded, out_added, atol=1e-5, rtol=1e-5)

    g = torch.randn(M, device=DEV)
    (out_folded * g).sum().backward()
    assert torch.allclose(partial.grad, g, atol=1e-6)


def test_prebound_bias_kernel_matches_gemv_fallback():
    # the flat-optimizer path (pre-bound bf16 .grad views) computes bias
    # grads + head wgrad in k_mlp3_bias_bwd; it must match the GEMV
    # fallback (non-prebound branch) to bf16 rounding
    from openembedding_amd.models.ctr import _FusedMLP3Fn
    torch.manual_seed(9)
    M, K0, H = 1024, 247, 400
    K0p = (K0 + 31) // 32 * 32
    Hp = (H + 31) // 32 * 32
    x0 = torch.zeros(M, K0p, device=DEV, dtype=torch.bfloat16)
    x0[:, :K0] = (torch.randn(M, K0, device=DEV) * 0.1).to(torch.bfloat16)
    shapes = [(H, K0), (H,), (H, H), (H,), (H, H), (H,), (1, H), (1,)]
    base = [(torch.randn(*s, device=DEV)
             * (0.05 if len(s) == 2 else 0.1)).to(torch.bfloat16)
            for s in shapes]
    g = torch.randn(M, device=DEV)

    def run(prebind):
        params = [t.clone().requires_grad_(True) for t in base]
        if prebind:
            for p in params:
                p.grad = torch.zeros_like(p)
        z = lambda *s: torch.zeros(*s, device=DEV, dtype=torch.bfloat16)  # noqa: E731
        bufs = {"w1p": z(H, K0p), "w2p": z(H, Hp), "w3p": z(H, Hp),
                "w3tp": z(H, Hp), "w2tp": z(H, Hp), "w1tp": z(K0p, Hp)}
        out = _FusedMLP3Fn.apply(x0.clone().requires_grad_(True), None,
                                 *params, bufs)
        (out * g).sum().backward()
        return [p.grad.float() for p in params]

    fused = run(True)
    gemv = run(False)
    names = ["w1", "b1", "w2", "b2", "w3", "b3", "w4", "b4"]
    for n, a, b in zip(names, fused, gemv):
        assert torch.allclose(a, b, atol=5e-3, rtol=3e-2), (
            n, (a - b).abs().max().item())


def test_mlp3_wgrad_matches_addmm():
    """Fused 3-wgrad MFMA kernel vs the hipBLASLt addmm_ trio it replaces
    (same bf16 operands; fp32 accumulation both sides)."""
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    torch.manual_seed(4)
    M, H, K0, K0p = 1024, 400, 247, 256
    bf = torch.bfloat16
    dz1 = torch.randn(M, H, device=DEV).to(bf)
    dz2 = torch.randn(M, H, device=DEV).to(bf)
    dz3 = torch.randn(M, H, device=DEV).to(bf)
    x0 = torch.zeros(M, K0p, device=DEV, dtype=bf)
    x0[:, :K0] = torch.randn(M, K0, device=DEV).to(bf)
    a1 = torch.randn(M, H, device=DEV).to(bf)
    a2 = torch.randn(M, H, device=DEV).to(bf)
    base = [torch.randn(H, K0, device=DEV).to(bf).contiguous(),
            torch.randn(H, H, device=DEV).to(bf).contiguous(),
            torch.randn(H, H, device=DEV).to(bf).contiguous()]
    got = [b.clone() for b in base]
    ref = [b.clone() for b in base]
    scratch = torch.zeros(H * K0p + 2 * H * H, device=DEV)
    ext.mlp3_wgrad(dz1, dz2, dz3, x0, a1, a2, scratch, *got)
    ref[0].addmm_(dz1.t(), x0[:, :K0])
    ref[1].addmm_(dz2.t(), a1)
    ref[2].addmm_(dz3.t(), a2)
    for g, r in zip(got, ref):
        torch.testing.assert_close(g.float(), r.float(), rtol=2e-2,
                                   atol=2e-2)
    # scratch self-cleaned for the next (captured) step
    assert float(scratch.abs().sum()) == 0.0
    # second call accumulates again (beta=1 semantics)
    ext.mlp3_wgrad(dz1, dz2, dz3, x0, a1, a2, scratch, *got)
    ref[1].addmm_(dz2.t(), a1)
    torch.testing.assert_close(got[1].float(), ref[1].float(), rtol=3e-2,
                               atol=3e-2)


def test_mlp3_pack_matches_copy():
    """Fused weight repack kernel vs the aten copies it replaced: plain
    pads and transposed pads, including the padded-source (w1p -> w1tp)
    case and untouched zero tails."""
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    torch.manual_seed(7)
    bf = torch.bfloat16
    H, K0, K0p, Hp = 400, 247, 256, 416
    w1 = torch.randn(H, K0, device=DEV).to(bf)
    w2 = torch.randn(H, H, device=DEV).to(bf)
    w3 = torch.randn(H, H, device=DEV).to(bf)
    z = lambda *s: torch.zeros(*s, device=DEV, dtype=bf)  # noqa: E731
    w1p, w2p, w3p = z(H, K0p), z(H, Hp), z(H, Hp)
    ext.mlp3_pack(w1, w1p, False, w2, w2p, False, w3, w3p, False)
    assert torch.equal(w1p[:, :K0], w1) and float(w1p[:, K0:].abs().sum()) == 0
    assert torch.equal(w2p[:, :H], w2) and float(w2p[:, H:].abs().sum()) == 0
    assert torch.equal(w3p[:, :H], w3)

    w3tp, w2tp, w1tp = z(H, Hp), z(H, Hp), z(K0p, Hp)
    ext.mlp3_pack(w3, w3tp, True, w2, w2tp, True, w1p, w1tp, True)
    assert torch.equal(w3tp[:, :H], w3.t())
    assert torch.equal(w2tp[:, :H], w2.t())
    assert torch.equal(w1tp[:, :H], w1p.t())   # incl. zero K0..K0p rows
    assert float(w1tp[:, H:].abs().sum()) == 0
