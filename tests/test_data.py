"""Criteo TSV pipeline: parsing, key modes, missing values, background
loader equivalence, end-to-end train from file."""

import pytest
import torch

from openembedding_amd.data import BackgroundLoader, CriteoTSV, _hash_token
from openembedding_amd.models.criteo import N_DENSE

N_SPARSE = 26


def _write_sample(path, rows=10):
    lines = []
    for i in range(rows):
        label = i % 2
        dense = [str(i + d) if (i + d) % 7 else "" for d in range(N_DENSE)]
        cats = [format(i * 31 + c, "x") if (i + c) % 5 else ""
                for c in range(N_SPARSE)]
        lines.append("\t".join([str(label)] + dense + cats))
    lines.append("malformed line without enough fields")
    path.write_text("\n".join(lines) + "\n")


def test_parses_shapes_and_values(tmp_path):
    p = tmp_path / "sample.tsv"
    _write_sample(p, rows=10)
    batches = list(CriteoTSV(str(p), batch_size=4))
    # 10 valid rows (malformed skipped) -> 4+4+2
    assert [b[0].shape[0] for b in batches] == [4, 4, 2]
    dense, sparse, labels = batches[0]
    assert dense.shape == (4, N_DENSE) and dense.dtype == torch.float32
    assert sparse.shape == (4, N_SPARSE) and sparse.dtype == torch.int64
    assert labels.tolist() == [0.0, 1.0, 0.0, 1.0]
    # log1p normalization: value "1" -> log(2); empty -> 0
    assert torch.isclose(dense[1, 0], torch.log(torch.tensor(2.0)))


def test_bounded_vs_hash_keys(tmp_path):
    p = tmp_path / "sample.tsv"
    _write_sample(p, rows=6)
    fv = [17] * N_SPARSE
    bounded = next(iter(CriteoTSV(str(p), 6, field_vocabs=fv)))[1]
    hashed = next(iter(CriteoTSV(str(p), 6, hash_mode=True)))[1]
    assert int(bounded.max()) < 17 and int(bounded.min()) >= 0
    assert int(hashed.max()) > 2**32          # real 63-bit keys
    assert int(hashed.min()) >= 0             # int64-positive by contract
    # same token in the same field -> same key; across fields -> salted
    assert _hash_token(3, "ab") == _hash_token(3, "ab")
    assert _hash_token(3, "ab") != _hash_token(4, "ab")


def test_deterministic_across_runs(tmp_path):
    p = tmp_path / "sample.tsv"
    _write_sample(p)
    a = list(CriteoTSV(str(p), 3, hash_mode=True))
    b = list(CriteoTSV(str(p), 3, hash_mode=True))
    for (d1, s1, l1), (d2, s2, l2) in zip(a, b):
        assert torch.equal(s1, s2) and torch.equal(d1, d2)


def test_background_loader_equivalent(tmp_path):
    p = tmp_path / "sample.tsv"
    _write_sample(p, rows=9)
    direct = list(CriteoTSV(str(p), 2))
    threaded = list(BackgroundLoader(CriteoTSV(str(p), 2), depth=2))
    assert len(direct) == len(threaded)
    for (d1, s1, l1), (d2, s2, l2) in zip(direct, threaded):
        assert torch.equal(d1, d2) and torch.equal(s1, s2)


def test_end_to_end_train_from_file(tmp_path):
    import openembedding_amd.torch as embed
    from openembedding_amd.models import MODELS
    p = tmp_path / "train.tsv"
    _write_sample(p, rows=32)
    fv = [50] * N_SPARSE
    model = MODELS["deepfm"](field_vocabs=fv, dim=4)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad([q for q in model.parameters() if q.numel()],
                            lr=0.05))
    for dense, sparse, labels in BackgroundLoader(
            CriteoTSV(str(p), 8, field_vocabs=fv)):
        opt.zero_grad()
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            model(dense, sparse), labels)
        loss.backward()
        opt.step()
    assert torch.isfinite(loss)


def test_build_vocabs_frequency_threshold(tmp_path):
    from openembedding_amd.data import build_vocabs
    p = tmp_path / "v.tsv"
    rows = []
    for i in range(8):
        dense = ["1"] * N_DENSE
        # field 0: token "aa" appears 8x, "b<i>" once each
        cats = ["aa" if i < 8 else "x"] + [f"t{i}c{c}" for c in range(1, N_SPARSE)]
        rows.append("\t".join(["1"] + dense + cats))
    p.write_text("\n".join(rows) + "\n")
    sizes, maps = build_vocabs(str(p), min_count=2)
    assert maps[0] == {"aa": 1} and sizes[0] == 2        # rare tokens -> OOV 0
    assert maps[1] == {} and sizes[1] == 2               # all rare
    batch = next(iter(CriteoTSV(str(p), 8, field_vocabs=sizes,
                                vocab_maps=maps)))
    sparse = batch[1]
    assert set(sparse[:, 0].tolist()) == {1}             # "aa" -> id 1
    assert set(sparse[:, 1].tolist()) == {0}             # rare -> OOV
    assert int(sparse.max()) < max(sizes)


def test_pulling_over_file_loader(tmp_path):
    # the reference's full pipeline: dataset thread -> prefetched pull ->
    # train (pulling() + BackgroundLoader + CriteoTSV)
    import openembedding_amd.torch as embed
    from openembedding_amd.models import MODELS
    p = tmp_path / "train.tsv"
    _write_sample(p, rows=24)
    fv = [50] * N_SPARSE
    model = MODELS["wdl"](field_vocabs=fv, dim=4)
    opt = embed.distributed_optimizer(
        torch.optim.Adagrad([q for q in model.parameters() if q.numel()],
                            lr=0.05))
    n = 0
    for dense, sparse, labels in embed.pulling(
            BackgroundLoader(CriteoTSV(str(p), 8, field_vocabs=fv)), model):
        opt.zero_grad()
        loss = torch.nn.functional.binary_cross_entropy_with_logits(
            model(dense, sparse), labels)
        loss.backward()
        opt.step()
        n += 1
    assert n == 3 and torch.isfinite(loss)


def test_tfrecord_roundtrip(tmp_path):
    """write_tfrecord -> CriteoTFRecord round-trip, int64 ids."""
    from openembedding_amd.data import CriteoTFRecord, write_tfrecord
    rows = [(float(i % 2), [float(i + j) for j in range(13)],
             [(i * 7 + f) % 50 for f in range(26)]) for i in range(10)]
    p = str(tmp_path / "d.tfrecord")
    assert write_tfrecord(p, rows) == 10
    batches = list(CriteoTFRecord(p, batch_size=4,
                                  field_vocabs=[50] * 26))
    assert len(batches) == 3
    dense, sparse, labels = batches[0]
    assert dense.shape == (4, 13) and sparse.shape == (4, 26)
    assert labels.tolist() == [0.0, 1.0, 0.0, 1.0]
    assert sparse[2, 3].item() == (2 * 7 + 3) % 50
    assert dense[1, 2].item() == 3.0


def test_tfrecord_bytes_tokens_match_tsv_hashing(tmp_path):
    """String tokens hash exactly like the TSV pipeline (same keys from
    either input format, the reference grid's {tfrecord, csv} parity)."""
    from openembedding_amd.data import (CriteoTFRecord, CriteoTSV,
                                        write_tfrecord)
    toks = [f"tok{f}" for f in range(26)]
    dense = [1.0] * 13
    tsv = str(tmp_path / "d.tsv")
    with open(tsv, "w") as f:
        import math
        f.write("1\t" + "\t".join(str(int(math.expm1(1.0))) for _ in range(13))
                + "\t" + "\t".join(toks) + "\n")
    rec = str(tmp_path / "d.tfrecord")
    write_tfrecord(rec, [(1.0, dense, toks)])
    for hash_mode in (False, True):
        bt = list(CriteoTFRecord(rec, 1, hash_mode=hash_mode))[0]
        bc = list(CriteoTSV(tsv, 1, hash_mode=hash_mode))[0]
        assert torch.equal(bt[1], bc[1])   # sparse keys identical


def test_tfrecord_corruption_detected(tmp_path):
    from openembedding_amd.data import CriteoTFRecord, write_tfrecord
    p = str(tmp_path / "d.tfrecord")
    write_tfrecord(p, [(1.0, [0.0] * 13, [1] * 26)])
    blob = bytearray(open(p, "rb").read())
    blob[20] ^= 0xFF
    open(p, "wb").write(bytes(blob))
    with pytest.raises(ValueError, match="crc"):
        list(CriteoTFRecord(p, 1))
