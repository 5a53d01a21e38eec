"""Sharded dataset, dynamic masking, and chunked sampler tests."""

import numpy as np
import pytest
import torch

from bert_pytorch_amd.data import (
    DistributedSampler,
    ShardedPretrainingDataset,
    synth,
)


@pytest.fixture
def shards(tmp_path):
    return synth.make_dataset(
        str(tmp_path), num_shards=3, samples_per_shard=16, seq_len=32,
        vocab_size=2000, seed=1,
    )


def _dataset(shards, **kw):
    defaults = dict(
        mask_token_index=103, max_pred_per_seq=5, masked_lm_prob=0.15,
        vocab_size=2000, seed=0,
    )
    defaults.update(kw)
    return ShardedPretrainingDataset(shards, **defaults)


def test_len_and_sequential_access(shards):
    ds = _dataset(shards)
    assert len(ds) == 48
    for i in range(48):
        sample = ds[i]
        assert len(sample) == 5
        ids, seg, mask, labels, nsp = sample
        assert ids.shape == (32,)
        assert ids.dtype == np.int64
        assert nsp in (0, 1)


def test_masking_semantics(shards):
    ds = _dataset(shards)
    n_masked_total = 0
    for i in range(16):
        ids, seg, mask, labels, nsp = ds[i]
        masked = labels != -1
        n = int(masked.sum())
        assert 1 <= n <= 5
        n_masked_total += n
        # special tokens never masked: position 0 is [CLS]
        assert labels[0] == -1
        # labels hold the ORIGINAL token at masked positions
        raw = ds.data["input_ids"][i]
        np.testing.assert_array_equal(labels[masked], raw[masked])
        # padding region: mask 0, labels -1
        pad = mask == 0
        assert (labels[pad] == -1).all()
    assert n_masked_total > 16  # masking actually happened


def test_mask_count_distinct(shards):
    """replace=False: exactly mask_count distinct positions masked."""
    ds = _dataset(shards, masked_lm_prob=0.5, max_pred_per_seq=8)
    for i in range(8):
        _, _, mask, labels, _ = ds[i]
        n_valid = int(mask.sum()) - (3 if len(ds.data["special_token_positions"][i]) == 3 else 2)
        expect = min(8, max(1, int(n_valid * 0.5)))
        assert int((labels != -1).sum()) == expect


def test_segment_ids(shards):
    ds = _dataset(shards)
    for i in range(4):
        ids, seg, mask, labels, nsp = ds[i]
        special = ds.data["special_token_positions"][i]
        # segment 1 spans (sep1, sep2]
        assert seg[special[1]] == 0
        assert seg[special[1] + 1] == 1
        assert seg[special[2]] == 1
        if special[2] + 1 < len(seg):
            assert seg[special[2] + 1] == 0


def test_out_of_order_access_raises(shards):
    ds = _dataset(shards)
    _ = ds[0]
    with pytest.raises(RuntimeError):
        _ = ds[40]  # random access across shards is rejected
    # backward jump within walked-past shard also rejected after moving on


def test_shard_does_not_mutate(shards):
    """Dynamic masking must not corrupt the cached shard (reference bug)."""
    ds = _dataset(shards, masked_lm_prob=0.5, max_pred_per_seq=16)
    before = ds_copy = None
    _ = ds[0]
    before = ds.data["input_ids"][1].copy()
    _ = ds[1]
    np.testing.assert_array_equal(ds.data["input_ids"][1], before)


def test_sampler_chunked_and_stateful(shards):
    ds = _dataset(shards)
    s0 = DistributedSampler(ds, num_replicas=2, rank=0, seed=0)
    s1 = DistributedSampler(_dataset(shards), num_replicas=2, rank=1, seed=0)
    idx0 = [next(s0) for _ in range(len(s0))]
    idx1 = [next(s1) for _ in range(len(s1))]
    # contiguous chunks per rank
    assert idx0 == list(range(24))
    assert idx1 == list(range(24, 48))
    # state round-trip
    s0b = DistributedSampler(_dataset(shards), num_replicas=2, rank=0, seed=0)
    for _ in range(7):
        next(s0b)
    state = s0b.state_dict()
    s0c = DistributedSampler(_dataset(shards), num_replicas=2, rank=0, seed=0)
    s0c.load_state_dict(state)
    assert next(s0c) == idx0[7]


def test_sampler_reset_on_world_change(shards):
    ds = _dataset(shards)
    s = DistributedSampler(ds, num_replicas=2, rank=0, seed=0)
    state = s.state_dict()
    s2 = DistributedSampler(_dataset(shards), num_replicas=3, rank=0, seed=0)
    with pytest.warns(UserWarning):
        s2.load_state_dict(state)
    assert s2.index == 0


def test_dataloader_with_workers(shards):
    ds = _dataset(shards)
    sampler = DistributedSampler(ds, num_replicas=1, rank=0, seed=0)
    loader = torch.utils.data.DataLoader(
        ds, sampler=sampler, batch_size=4, num_workers=0, drop_last=True
    )
    batches = list(loader)
    assert len(batches) == 12
    assert batches[0][0].shape == (4, 32)


def test_legacy_premasked_format(tmp_path):
    from bert_pytorch_amd.data import h5lite

    n, s, p = 6, 16, 3
    rng = np.random.default_rng(0)
    input_ids = rng.integers(0, 100, (n, s)).astype(np.int32)
    positions = np.zeros((n, p), dtype=np.int32)
    mids = np.zeros((n, p), dtype=np.int32)
    positions[:, 0] = 2
    positions[:, 1] = 5
    mids[:, 0] = 7
    mids[:, 1] = 9
    path = str(tmp_path / "legacy.hdf5")
    h5lite.write(path, {
        "input_ids": input_ids,
        "segment_ids": np.zeros((n, s), np.int32),
        "input_mask": np.ones((n, s), np.int32),
        "masked_lm_positions": positions,
        "masked_lm_ids": mids,
        "next_sentence_labels": np.zeros(n, np.int8),
    })
    ds = _dataset([path])
    ids, seg, mask, labels, nsp = ds[0]
    assert labels[2] == 7 and labels[5] == 9
    assert (labels != -1).sum() == 2


def test_corrupt_and_missing_shards_are_skipped(shards, tmp_path):
    """Failure handling: unreadable / truncated / missing
    shards are skipped with a warning; valid ones still serve."""
    bad_garbage = tmp_path / "garbage.hdf5"
    bad_garbage.write_bytes(b"\x00\x01not an hdf5 file\xff" * 64)
    bad_trunc = tmp_path / "truncated.hdf5"
    bad_trunc.write_bytes(open(shards[0], "rb").read()[:200])
    missing = str(tmp_path / "never_written.hdf5")
    files = [shards[0], str(bad_garbage), str(bad_trunc), missing, shards[1]]
    with pytest.warns(UserWarning):
        ds = _dataset(files)
    assert len(ds) == 32  # the two valid shards
    ids, seg, mask, labels, nsp = ds[0]
    assert ids.shape[0] == 32


def test_all_shards_invalid_raises(tmp_path):
    bad = tmp_path / "junk.hdf5"
    bad.write_bytes(b"junk")
    with pytest.raises(RuntimeError), pytest.warns(UserWarning):
        _dataset([str(bad)])
