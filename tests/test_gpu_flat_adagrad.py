"""GPU numerics: fused flat Adagrad kernels vs the fp32 torch reference,
both f32 and bf16-with-master forms; native-bf16 DeepFM trains."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_flat_adagrad_f32_matches_torch():
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    torch.manual_seed(0)
    n, lr, eps = 10000, 0.01, 1e-10
    p = torch.randn(n, device=DEV)
    accum = torch.rand(n, device=DEV)
    g = torch.randn(n, device=DEV)
    rp, ra = p.clone(), accum.clone()
    ext.flat_adagrad(p, accum, g, None, lr, eps)
    ra.addcmul_(g, g)
    rp.addcdiv_(g, ra.sqrt().add_(eps), value=-lr)
    assert torch.allclose(p, rp, atol=1e-7)
    assert torch.equal(accum, ra)


def test_flat_adagrad_bf16_master():
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    torch.manual_seed(1)
    n, lr, eps = 8192, 0.05, 1e-10
    master = torch.randn(n, device=DEV)
    p = master.to(torch.bfloat16)
    accum = torch.zeros(n, device=DEV)
    rm, ra = master.clone(), accum.clone()
    for step in range(3):
        g32 = torch.randn(n, device=DEV, generator=None)
        g = g32.to(torch.bfloat16)
        ext.flat_adagrad(p, accum, g, master, lr, eps)
        gf = g.to(torch.float32)   # reference uses the same bf16-read grad
        ra.addcmul_(gf, gf)
        rm.addcdiv_(gf, ra.sqrt().add(eps), value=-lr)
    assert torch.allclose(master, rm, atol=1e-6)
    assert torch.allclose(p.float(), rm.to(torch.bfloat16).float())
    assert torch.allclose(accum, ra, atol=1e-4, rtol=1e-5)


def test_native_bf16_deepfm_trains():
    import openembedding_amd.torch as embed
    from openembedding_amd.models import DeepFM, synthetic_batch
    from openembedding_amd.models.ctr import convert_mlp_bf16

    torch.manual_seed(0)
    model = convert_mlp_bf16(DeepFM(dim=9).to(DEV))
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad(model.parameters(), lr=0.01),
        flatten_dense=True)
    lossf = torch.nn.BCEWithLogitsLoss()
    losses = []
    for i in range(8):
        dense, sparse, labels = synthetic_batch(1024, device=DEV)
        opt.zero_grad()
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses)))
    # training moves the loss (random labels -> toward ~0.56 entropy region)
    assert losses[-1] < losses[0]


@pytest.mark.parametrize("kind", ["sgd", "sgd-mom", "sgd-nesterov", "adam"])
def test_flat_opt_kernels_match_torch(kind):
    """round-2 generic flat kernels (flat_opt): SGD/momentum/nesterov and
    Adam vs per-tensor torch.optim on cuda, f32 and bf16+master forms."""
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    torch.manual_seed(2)
    n, lr = 4096, 0.02
    opt_id = {"sgd": 0, "sgd-mom": 0, "sgd-nesterov": 0, "adam": 2}[kind]
    mom = 0.9 if "mom" in kind or "nesterov" in kind else 0.0
    nest = 1.0 if "nesterov" in kind else 0.0
    b1, b2, eps = 0.9, 0.999, 1e-8

    for bf16 in (False, True):
        master = torch.randn(n, device=DEV)
        p = master.to(torch.bfloat16) if bf16 else master.clone()
        ref = torch.nn.Parameter(master.clone())
        if opt_id == 0:
            topt = torch.optim.SGD([ref], lr=lr, momentum=mom,
                                   nesterov=bool(nest))
            cfg = (mom, nest, 0.0)
            s1 = torch.zeros(n, device=DEV) if mom else None
            s2, sc = None, None
        else:
            topt = torch.optim.Adam([ref], lr=lr, betas=(b1, b2), eps=eps)
            cfg = (b1, b2, eps)
            s1 = torch.zeros(n, device=DEV)
            s2 = torch.zeros(n, device=DEV)
            sc = torch.zeros(3, device=DEV)
        g = torch.Generator(device=DEV).manual_seed(5)
        for _ in range(4):
            grad32 = torch.randn(n, device=DEV, generator=g)
            grad = grad32.to(torch.bfloat16) if bf16 else grad32
            # snapshot BEFORE flat_opt: the kernel zeroes the grad buffer
            # after consuming it (zero_grad no-op contract)
            ref.grad = grad.to(torch.float32).clone()  # same bf16-read grad
            if sc is not None:
                ext.flat_step_scalars(sc, b1, b2)
            ext.flat_opt(opt_id, p, master if bf16 else None, s1, s2,
                         grad, sc, lr, *cfg)
            assert float(grad.abs().max()) == 0.0  # kernel re-zeroed grads
            topt.step()
        w = master if bf16 else p
        assert torch.allclose(w, ref.detach(), atol=1e-5, rtol=1e-5), kind
        if bf16:
            assert torch.equal(p, w.to(torch.bfloat16))
