"""Flat dense-parameter Adagrad path: numerically identical to the plain
torch.optim.Adagrad step on per-parameter tensors."""

import pytest
import torch

import openembedding_amd.torch as embed
from openembedding_amd.models import DeepFM, synthetic_batch


def _train(flatten, n=5, opt_ctor=None):
    torch.manual_seed(3)
    model = DeepFM(dim=4)
    if opt_ctor is None:
        opt_ctor = lambda ps: torch.optim.Adagrad(ps, lr=0.01)  # noqa: E731
    opt = embed.distributed_optimizer(
        opt_ctor(model.parameters()),
        flatten_dense=flatten,
        sparse_config=dict(category="adagrad", learning_rate=0.01))
    lossf = torch.nn.BCEWithLogitsLoss()
    gen = torch.Generator().manual_seed(9)
    losses = []
    for _ in range(n):
        dense, sparse, labels = synthetic_batch(64, generator=gen)
        opt.zero_grad(set_to_none=False)
        loss = lossf(model(dense, sparse), labels)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    sd = {k: v.clone() for k, v in model.state_dict().items()}
    return losses, sd, opt


def _reset():
    import openembedding_amd.context as cm
    import openembedding_amd.torch as api
    if cm._context is not None:
        cm._context.finalize()
        cm._context = None
    api._tracked.clear()


def test_flat_matches_plain_adagrad():
    plain_losses, plain_sd, _ = _train(False)
    _reset()
    flat_losses, flat_sd, _ = _train(True)
    assert len(plain_losses) == len(flat_losses)
    for a, b in zip(plain_losses, flat_losses):
        assert a == pytest.approx(b, rel=1e-5), (plain_losses, flat_losses)
    for k in plain_sd:
        if k.endswith("grad_hook"):
            continue
        assert torch.allclose(plain_sd[k], flat_sd[k], atol=1e-6), k


def test_flat_state_roundtrip():
    losses, sd, opt = _train(True, n=3)
    blob = opt.state_dict()
    assert "flat_dense" in blob
    opt.load_state_dict(blob)  # no-throw, idempotent


@pytest.mark.parametrize("ctor", [
    lambda ps: torch.optim.SGD(ps, lr=0.05),
    lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9),
    lambda ps: torch.optim.SGD(ps, lr=0.05, momentum=0.9, nesterov=True),
    lambda ps: torch.optim.Adam(ps, lr=0.003),
], ids=["sgd", "sgd-mom", "sgd-nesterov", "adam"])
def test_flat_matches_plain_sgd_adam(ctor):
    """round-2: the flat path must cover SGD/Adam too, identically to the
    per-param torch step (the round-1 Adagrad-only gate made every other
    optimizer fall into the slow cat/allreduce path)."""
    _reset()
    plain_losses, plain_sd, _ = _train(False, opt_ctor=ctor)
    _reset()
    flat_losses, flat_sd, _ = _train(True, opt_ctor=ctor)
    for a, b in zip(plain_losses, flat_losses):
        assert a == pytest.approx(b, rel=1e-5), (plain_losses, flat_losses)
    for k in plain_sd:
        if k.endswith("grad_hook"):
            continue
        assert torch.allclose(plain_sd[k], flat_sd[k], atol=1e-6), k
    _reset()


def test_flat_rejects_unsupported():
    model = torch.nn.Linear(4, 2)
    with pytest.raises(ValueError):
        embed.distributed_optimizer(
            torch.optim.Adagrad(model.parameters(), lr=0.1, weight_decay=0.1),
            flatten_dense=True)


def test_bce_with_logits_cpu_fallback_matches_torch():
    # the fused-loss dispatch must be a drop-in for BCEWithLogitsLoss off-GPU
    import torch
    from openembedding_amd.ops.dispatch import bce_with_logits
    g = torch.Generator().manual_seed(11)
    z = (torch.randn(513, generator=g) * 3).requires_grad_(True)
    y = (torch.rand(513, generator=g) < 0.25).float()
    z2 = z.detach().clone().requires_grad_(True)
    loss = bce_with_logits(z, y)
    ref = torch.nn.functional.binary_cross_entropy_with_logits(z2, y)
    assert torch.allclose(loss, ref)
    loss.backward(); ref.backward()
    assert torch.allclose(z.grad, z2.grad)


def test_bce_with_logits_int_labels():
    import torch
    from openembedding_amd.ops.dispatch import bce_with_logits
    z = torch.zeros(4, requires_grad=True)
    y = torch.tensor([0, 1, 0, 1])            # non-float labels accepted
    loss = bce_with_logits(z, y)
    assert torch.isfinite(loss)


def test_zero_grad_noop_after_step_still_correct():
    """step() re-zeroes the flat grad buffer, so the first zero_grad of the
    next iteration is skipped; training must match the non-flat path, and
    a zero_grad used to DISCARD an extra backward (second zero_grad in the
    same cycle) must really clear the buffer."""
    _reset()
    losses, sd, opt = _train(True)
    f = opt._flat
    # after the final step(): buffer zeroed, flag armed
    assert f._grads_zeroed
    assert all(float(g["flat_grad"].abs().max()) == 0 for g in f.groups)
    f.zero_grad()                       # consumes the flag (no fill needed)
    assert not f._grads_zeroed
    # stray backward -> second zero_grad must do the real fill
    g0 = f.groups[0]
    g0["flat_grad"].fill_(1.0)
    f.zero_grad()
    assert float(g0["flat_grad"].abs().max()) == 0
