"""GPU tests of the padded (sync-free) all-to-all route (all @gpu).

At world 1 with _force_remote the collectives are identity, so the full HIP
padded pipeline (k_bucketize_pad -> owner unique/gather -> k_scatter_out,
push via k_gather_pad / k_split_payload) is validatable on one GPU against
(a) the exact remote path and (b) the CPU torch oracle. The 2-process RCCL
run lives in scripts/rccl_2rank_1gpu.py (driven via gpurun)."""

import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"
DIM = 9
VOCAB = 100_000


def _sharded(device, padded, force_remote=True, vocab=VOCAB, dim=DIM,
             hash_mode=False, reserve=0):
    from openembedding_amd.context import Context
    ctx = Context(device=device)
    st = ctx.create_storage()
    var = st.create_variable((1 << 63) if hash_mode else vocab, dim)
    var.set_initializer("uniform", minval=-1.0, maxval=1.0)
    var.set_optimizer("adagrad", learning_rate=0.05,
                      initial_accumulator_value=0.1, epsilon=1e-10)
    var._force_remote = force_remote
    var._padded = padded
    if reserve:
        var.reserve_rows(reserve)
    return st, var


def _batches(n_steps=4, n=4096, vocab=VOCAB, seed=0):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n_steps):
        keys = torch.randint(0, vocab, (n,), dtype=torch.int64, generator=g)
        grads = torch.randn(n, DIM, generator=g)
        out.append((keys, grads))
    return out


def _train(st, var, batches, device):
    for keys, grads in batches:
        out, h = var.pull(keys.to(device))
        var.push(h, grads.to(device))
        st.update_weights()
    probe = torch.unique(torch.cat([k for k, _ in batches])).to(device)
    after, _ = var.pull(probe, readonly=True)
    return after.cpu(), probe.cpu()


@pytest.mark.parametrize("hash_mode", [False, True])
def test_padded_matches_exact_gpu(hash_mode):
    """GPU padded route == GPU exact remote route after 4 training steps."""
    batches = _batches()
    st_a, var_a = _sharded(DEV, padded=False, hash_mode=hash_mode)
    a, probe = _train(st_a, var_a, batches, DEV)
    st_b, var_b = _sharded(DEV, padded=True, hash_mode=hash_mode)
    b, _ = _train(st_b, var_b, batches, DEV)
    var_b.check_padded_overflow()
    torch.testing.assert_close(a, b, rtol=1e-5, atol=1e-6)


def test_padded_matches_cpu_oracle():
    """GPU padded route == CPU engine (local path) on the same stream of
    batches — ties the wire format to the reference semantics end-to-end."""
    batches = _batches(n_steps=3, n=2048)
    st_g, var_g = _sharded(DEV, padded=True)
    g, probe = _train(st_g, var_g, batches, DEV)
    st_c, var_c = _sharded("cpu", padded=False, force_remote=False)
    c, _ = _train(st_c, var_c, batches, "cpu")
    torch.testing.assert_close(g, c, rtol=1e-4, atol=1e-5)


def test_bucketize_pad_kernel_matches_torch():
    from openembedding_amd.ops import require_hip
    from openembedding_amd.parallel.sharded import _PaddedPlan
    ext = require_hip()
    world = 8
    n = 10000
    g = torch.Generator().manual_seed(7)
    keys = torch.randint(0, 1 << 40, (n,), dtype=torch.int64, generator=g)
    keys[::97] = -1                      # sprinkle reserved keys
    dkeys = keys.to(DEV)
    uk, inverse, u_dev = ext.unique_bounded(dkeys)
    plan = _PaddedPlan(world, n, torch.device(DEV))
    ext.bucketize_pad(uk, u_dev, world, plan.cap, plan.send_keys,
                      plan.send_src, plan.pos_of, plan.counts, plan.overflow)
    torch.cuda.synchronize()
    assert int(plan.overflow.item()) == 0
    u = int(u_dev.item())
    send_keys = plan.send_keys.cpu()
    send_src = plan.send_src.cpu()
    pos_of = plan.pos_of.cpu()
    uk_h = uk.cpu()
    # every valid unique key lands exactly once, in its owner's block
    for owner in range(world):
        blk = send_keys[owner * plan.cap:(owner + 1) * plan.cap]
        live = blk[blk >= 0]
        assert torch.all(live % world == owner)
    valid = [int(k) for k in uk_h[:u] if int(k) >= 0]
    shipped = sorted(int(k) for k in send_keys[send_keys >= 0])
    assert shipped == sorted(valid)
    # pos_of/send_src are mutually inverse on the shipped set
    for i in range(u):
        p = int(pos_of[i])
        if int(uk_h[i]) < 0:
            assert p == -1
        else:
            assert p >= 0 and int(send_src[p]) == i


def test_gather_split_payload_roundtrip():
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    n, dim, world, cap = 500, DIM, 4, 200
    g = torch.Generator().manual_seed(3)
    ugrads = torch.randn(n, dim, generator=g).to(DEV)
    counts = torch.randint(1, 9, (n,), dtype=torch.int64, generator=g).to(DEV)
    send_src = torch.full((world * cap,), -1, dtype=torch.int32, device=DEV)
    send_src[: n] = torch.arange(n, dtype=torch.int32, device=DEV)
    send_p = ext.gather_pad(ugrads, counts, send_src)
    assert send_p.shape == (world * cap, dim + 1)
    torch.testing.assert_close(send_p[:n, :dim], ugrads)
    assert torch.equal(send_p[:n, dim].long(), counts)
    assert float(send_p[n:].abs().sum()) == 0.0
    g2, c2 = ext.split_payload(send_p[:n].contiguous(), None)
    torch.testing.assert_close(g2, ugrads)
    assert torch.equal(c2, counts)


def test_scatter_out_kernel():
    from openembedding_amd.ops import require_hip
    ext = require_hip()
    dim = DIM
    rows = torch.randn(64, dim).to(DEV)
    # 3 uniques at wire positions 5, 9, 63; unique 1 never shipped
    pos_of = torch.tensor([5, -1, 9, 63], dtype=torch.int32, device=DEV)
    inverse = torch.tensor([0, 1, 2, 3, 0, 2], dtype=torch.int64, device=DEV)
    out = ext.scatter_out(rows, inverse, pos_of, inverse.numel())
    torch.testing.assert_close(out[0], rows[5])
    assert float(out[1].abs().sum()) == 0.0
    torch.testing.assert_close(out[2], rows[9])
    torch.testing.assert_close(out[3], rows[63])
    torch.testing.assert_close(out[4], rows[5])
    torch.testing.assert_close(out[5], rows[9])


def test_multi_pull_commit_merges_blocks():
    """Two pulls of overlapping keys committed together must apply the
    optimizer ONCE per unique key over summed grads (reference
    MpscGradientReducer semantics) — on both the CPU oracle and the GPU
    bounded path (which queues one block per pull and must merge them)."""
    batches = [(torch.arange(0, 64, dtype=torch.int64),
                torch.full((64, DIM), 0.5)),
               (torch.arange(32, 96, dtype=torch.int64),  # overlap 32..64
                torch.full((64, DIM), 0.25))]
    results = []
    for device in ("cpu", DEV):
        st, var = _sharded(device, padded=False, force_remote=False)
        hs = []
        for keys, grads in batches:
            _, h = var.pull(keys.to(device))
            hs.append((h, grads))
        for h, grads in hs:
            var.push(h, grads.to(device))
        st.update_weights()
        probe = torch.arange(0, 96, dtype=torch.int64, device=device)
        after, _ = var.pull(probe, readonly=True)
        results.append(after.cpu())
    torch.testing.assert_close(results[0], results[1], rtol=1e-5,
                               atol=1e-6)


def test_padded_step_graph_capturable():
    """The padded route must capture into a hipGraph (array mode; replays
    train fresh data through static buffers)."""
    st, var = _sharded(DEV, padded=True)
    keys = torch.randint(0, VOCAB, (4096,), dtype=torch.int64, device=DEV)
    grads = torch.randn(4096, DIM, device=DEV)

    def step():
        out, h = var.pull(keys)
        var.push(h, grads)
        st.update_weights()
        return out

    for _ in range(3):
        step()
    torch.cuda.synchronize()
    side = torch.cuda.Stream()
    side.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(side):
        for _ in range(2):
            step()
    torch.cuda.current_stream().wait_stream(side)
    graph = torch.cuda.CUDAGraph()
    with torch.cuda.graph(graph):
        out = step()
    before = out.clone()
    keys.copy_(torch.randint(0, VOCAB, (4096,), dtype=torch.int64,
                             device=DEV))
    graph.replay()
    torch.cuda.synchronize()
    assert not torch.equal(before, out)   # fresh keys -> fresh rows
    var.check_padded_overflow()
