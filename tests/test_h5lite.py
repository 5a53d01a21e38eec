"""h5lite round-trip and format tests."""

import numpy as np
import pytest

from bert_pytorch_amd.data import h5lite


def test_roundtrip_int32(tmp_path):
    path = str(tmp_path / "a.hdf5")
    data = {
        "input_ids": np.arange(24, dtype=np.int32).reshape(4, 6),
        "special_token_positions": np.array(
            [[0, 2, 5]] * 4, dtype=np.int32
        ),
        "next_sentence_labels": np.array([0, 1, 1, 0], dtype=np.int8),
    }
    h5lite.write(path, data)
    out = h5lite.read(path)
    assert set(out) == set(data)
    for k in data:
        np.testing.assert_array_equal(out[k], data[k])
        assert out[k].dtype == data[k].dtype


def test_roundtrip_float_and_shapes(tmp_path):
    path = str(tmp_path / "b.hdf5")
    data = {
        "f32": np.random.default_rng(0).standard_normal((3, 5)).astype(np.float32),
        "i64": np.arange(7, dtype=np.int64),
        "u8": np.arange(11, dtype=np.uint8),
    }
    h5lite.write(path, data)
    out = h5lite.read(path)
    for k in data:
        np.testing.assert_array_equal(out[k], data[k])


def test_signature_check(tmp_path):
    path = str(tmp_path / "bad.hdf5")
    with open(path, "wb") as f:
        f.write(b"not an hdf5 file at all")
    with pytest.raises(ValueError):
        h5lite.H5LiteFile(path)


def test_lookup3_known_values():
    # self-consistency + stability of the checksum function
    assert h5lite.lookup3(b"") == h5lite.lookup3(b"")
    assert h5lite.lookup3(b"abc") != h5lite.lookup3(b"abd")
    assert h5lite.lookup3(b"x" * 100) == h5lite.lookup3(b"x" * 100)


def test_file_mmappable_layout(tmp_path):
    # contiguous datasets start 8-byte aligned
    path = str(tmp_path / "c.hdf5")
    h5lite.write(path, {"x": np.arange(10, dtype=np.int32)})
    f = h5lite.H5LiteFile(path)
    np.testing.assert_array_equal(f["x"], np.arange(10, dtype=np.int32))
