"""Variable engine behavior (the reference c_api_test mirror-checking style):
a CPU mirror tracks expected values with the deterministic 'test' optimizer
and every pull is compared exactly."""

import numpy as np
import pytest
import torch

from openembedding_amd.core import (HASH_VOCAB_THRESHOLD, VariableMeta,
                                    VariableShard, make_optimizer)


def make_shard(vocab, dim=4, seed=3):
    meta = VariableMeta(variable_id=1, embedding_dim=dim,
                        vocabulary_size=vocab)
    s = VariableShard(meta, 0, 1, device="cpu", seed=seed)
    s.set_initializer("uniform", minval=-1.0, maxval=1.0)
    s.set_optimizer("test")
    return s


@pytest.mark.parametrize("vocab", [1000, HASH_VOCAB_THRESHOLD])
def test_mirror_training(vocab):
    """Random pulls/pushes with duplicated, shuffled keys vs a dict mirror
    (reference c_api_test.h:123-133 exact-check pattern)."""
    torch.manual_seed(0)
    dim = 4
    shard = make_shard(vocab, dim)
    opt = make_optimizer("test")
    mirror = {}          # key -> np weights
    mirror_state = {}    # key -> flip state

    rng = np.random.default_rng(1)
    for step in range(8):
        base = rng.integers(0, 999, size=12)
        keys = np.concatenate([base, rng.choice(base, size=6)])  # dups
        rng.shuffle(keys)
        kt = torch.tensor(keys, dtype=torch.int64)
        uk, inverse = torch.unique(kt, return_inverse=True)
        pulled = shard.pull(uk)
        # mirror lazy init from the engine's own deterministic pull values
        for i, k in enumerate(uk.tolist()):
            if k not in mirror:
                mirror[k] = pulled[i].numpy().copy()
                mirror_state[k] = 0.0
            np.testing.assert_array_equal(pulled[i].numpy(), mirror[k]), k

        grads = torch.randn(len(keys), dim)
        ug = torch.zeros(uk.numel(), dim)
        ug.index_add_(0, inverse, grads)
        counts = torch.bincount(inverse, minlength=uk.numel())
        shard.push(uk, ug, counts)
        shard.update_weights()
        for i, k in enumerate(uk.tolist()):
            s0 = 10000.0 - mirror_state[k]
            mirror_state[k] = s0
            # same association as the engine: w + (lr*g/count + s0)
            delta = (0.1 * ug[i].numpy() / np.float32(counts[i].item())
                     + np.float32(s0))
            mirror[k] = mirror[k] + delta

    all_keys = torch.tensor(sorted(mirror), dtype=torch.int64)
    final = shard.pull_readonly(all_keys)
    for i, k in enumerate(all_keys.tolist()):
        np.testing.assert_allclose(final[i].numpy(), mirror[k], rtol=1e-5)


def test_deterministic_lazy_init():
    a = make_shard(HASH_VOCAB_THRESHOLD, seed=9)
    b = make_shard(HASH_VOCAB_THRESHOLD, seed=9)
    keys = torch.tensor([5, 123456789, 7], dtype=torch.int64)
    torch.testing.assert_close(a.pull(keys), b.pull(torch.flip(keys, [0])).flip(0))
    c = make_shard(HASH_VOCAB_THRESHOLD, seed=10)
    assert not torch.equal(a.pull(keys), c.pull(keys))


def test_array_mode_rejects_out_of_range():
    shard = make_shard(100)
    with pytest.raises(IndexError):
        shard.pull(torch.tensor([150], dtype=torch.int64))


def test_array_mode_shard_ownership():
    meta = VariableMeta(variable_id=0, embedding_dim=2, vocabulary_size=100)
    s1 = VariableShard(meta, shard_id=1, shard_num=2, device="cpu", seed=0)
    s1.set_optimizer("sgd")
    with pytest.raises(ValueError):
        s1.pull(torch.tensor([4], dtype=torch.int64))  # owned by shard 0
    out = s1.pull(torch.tensor([5], dtype=torch.int64))
    assert out.shape == (1, 2)


def test_push_without_pull_creates_row():
    shard = make_shard(1000)
    k = torch.tensor([7], dtype=torch.int64)
    shard.push(k, torch.ones(1, 4), torch.tensor([1], dtype=torch.int64))
    shard.update_weights()
    assert shard.num_rows == 1
    got = shard.pull_readonly(k)[0]
    # init + one test-optimizer step: w0 + 0.1*1/1 + 10000
    assert torch.all(got > 9000)


def test_readonly_missing_is_zero():
    shard = make_shard(1000)
    out = shard.pull_readonly(torch.tensor([3], dtype=torch.int64))
    assert torch.all(out == 0)
    assert shard.num_rows == 0


def test_export_import_roundtrip():
    shard = make_shard(HASH_VOCAB_THRESHOLD)
    keys = torch.tensor([2, 9, 11], dtype=torch.int64)
    shard.pull(keys)
    shard.push(keys, torch.randn(3, 4),
               torch.tensor([1, 1, 1], dtype=torch.int64))
    shard.update_weights()
    k, w, s = shard.export_rows()
    assert k.numel() == 3 and s is not None

    other = make_shard(HASH_VOCAB_THRESHOLD, seed=99)
    other.import_rows(k, w, s)
    torch.testing.assert_close(other.pull_readonly(keys),
                               shard.pull_readonly(keys))
    # optimizer state must carry over: one more identical update -> equal
    g = torch.randn(3, 4)
    for sh in (shard, other):
        sh.push(keys, g.clone(), torch.tensor([1, 1, 1], dtype=torch.int64))
        sh.update_weights()
    torch.testing.assert_close(other.pull_readonly(keys),
                               shard.pull_readonly(keys))


def test_clear():
    shard = make_shard(1000)
    shard.pull(torch.tensor([1, 2], dtype=torch.int64))
    shard.clear()
    assert shard.num_rows == 0


def test_optimizer_reconfig_migrates():
    """Changing optimizer rebuilds state; same category keeps it
    (reference EmbeddingVariable.cpp:29-60)."""
    shard = make_shard(1000)
    keys = torch.tensor([1], dtype=torch.int64)
    shard.pull(keys)
    shard.set_optimizer("adagrad", initial_accumulator_value=0.3)
    assert shard.state_dim == 4
    assert torch.allclose(shard.state[shard._lookup_readonly(keys)][0],
                          torch.full((4,), 0.3))
    shard.set_optimizer("adam")
    assert shard.state_dim == 2 * 4 + 2


def test_empty_batch_pull_push():
    """n=0 pulls/pushes must be no-ops (collective paths call them on every
    rank even when a rank's batch is empty)."""
    from openembedding_amd.parallel.sharded import ShardedVariable

    sh = VariableShard(VariableMeta(variable_id=50, embedding_dim=4,
                                    vocabulary_size=100))
    sh.set_initializer("constant", value=1.0)
    sh.set_optimizer("adagrad", learning_rate=0.1)
    v = ShardedVariable(sh)
    out, h = v.pull(torch.empty(0, dtype=torch.int64))
    assert out.shape == (0, 4)
    v.push(h, torch.empty(0, 4))
    v.update_weights()
    assert sh.num_rows == 0


def test_update_without_push_is_noop():
    sh = VariableShard(VariableMeta(variable_id=51, embedding_dim=4,
                                    vocabulary_size=100))
    sh.set_initializer("constant", value=2.0)
    sh.set_optimizer("sgd", learning_rate=0.1, momentum=0.0, nesterov=False)
    before = sh.pull(torch.tensor([3])).clone()
    sh.update_weights()
    sh.update_weights()
    assert torch.equal(sh.pull(torch.tensor([3])), before)


def test_single_key_many_duplicates():
    """One key duplicated across a whole batch: gradient SUM + one
    optimizer step (the reference's counts semantics)."""
    sh = VariableShard(VariableMeta(variable_id=52, embedding_dim=2,
                                    vocabulary_size=10))
    sh.set_initializer("constant", value=0.0)
    sh.set_optimizer("default", learning_rate=1.0)
    keys = torch.zeros(64, dtype=torch.int64)
    from openembedding_amd.parallel.sharded import ShardedVariable
    v = ShardedVariable(sh)
    out, h = v.pull(keys)
    v.push(h, torch.ones(64, 2))
    v.update_weights()
    got = sh.pull_readonly(torch.tensor([0]))
    # default optimizer: w -= lr * summed_grad = -64
    assert torch.allclose(got, torch.full((1, 2), -64.0))


def test_hash_mode_key_minus_one_reserved():
    # the table's empty marker — reserved exactly like the reference's
    # empty_key = -1; the CPU oracle fails loudly (the GPU kernel resolves
    # it to a zero row instead, see ops/csrc/embops.hip k_ht_lookup)
    import pytest as _pytest
    from openembedding_amd.core import VariableMeta, VariableShard
    meta = VariableMeta(variable_id=1, embedding_dim=4,
                        vocabulary_size=1 << 63)
    s = VariableShard(meta, 0, 1, device="cpu", seed=1)
    s.set_initializer("uniform", minval=-1, maxval=1)
    with _pytest.raises(ValueError, match="reserved"):
        s.pull(torch.tensor([5, -1, 7]))
    # read-only path: defined miss (zeros), no raise
    out = s.pull_readonly(torch.tensor([-1]))
    assert torch.equal(out, torch.zeros(1, 4))
