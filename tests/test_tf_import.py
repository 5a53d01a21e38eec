"""TF checkpoint-bundle reader/writer + BERT weight import (no TF).

Replaces the reference's tf.train-based load_tf_weights_in_bert
(src/modeling.py:58-116) with the in-repo bundle parser; tests are a
write->read round trip (the writer is the only TF-checkpoint producer
available in this image) plus an end-to-end model import.
"""

import numpy as np
import pytest
import torch

from bert_pytorch_amd.data.tf_bundle import (
    TFBundleReader,
    TFBundleWriter,
    crc32c,
    snappy_decompress,
)
from bert_pytorch_amd.models import (
    BertForPreTraining,
    load_tf_weights,
    tf_name_to_state_key,
)


def test_crc32c_known_vectors():
    # RFC 3720 test vectors
    assert crc32c(b"\x00" * 32) == 0x8A9136AA
    assert crc32c(b"\xff" * 32) == 0x62A8AB43
    assert crc32c(bytes(range(32))) == 0x46DD794E


def test_snappy_literals_and_copies():
    # hand-assembled stream: literal "abcd" + copy1(offset 4, len 4)
    # -> "abcdabcd"
    comp = bytes([8]) + bytes([0b00001100]) + b"abcd" + \
        bytes([0b00000001, 4])
    assert snappy_decompress(comp) == b"abcdabcd"


def test_bundle_roundtrip(tmp_path):
    prefix = str(tmp_path / "model.ckpt")
    w = TFBundleWriter(prefix)
    rng = np.random.default_rng(0)
    tensors = {
        "bert/embeddings/word_embeddings": rng.normal(
            size=(64, 16)).astype(np.float32),
        "bert/encoder/layer_0/attention/self/query/kernel": rng.normal(
            size=(16, 16)).astype(np.float32),
        "scalar/step": np.array(7, dtype=np.int64),
        "bias": rng.normal(size=(16,)).astype(np.float32),
    }
    for k, v in tensors.items():
        w.add(k, v)
    w.save()

    r = TFBundleReader(prefix)
    names = [n for n, _ in r.list_variables()]
    assert sorted(tensors) == names
    for k, v in tensors.items():
        got = r.load_variable(k)
        assert got.dtype == v.dtype
        assert got.shape == v.shape
        np.testing.assert_array_equal(got, v)
    with pytest.raises(KeyError):
        r.load_variable("nope")


def test_tf_name_mapping():
    cases = {
        "bert/embeddings/word_embeddings":
            ("bert.embeddings.word_embeddings.weight", False),
        "bert/embeddings/LayerNorm/gamma":
            ("bert.embeddings.LayerNorm.weight", False),
        "bert/encoder/layer_3/attention/self/query/kernel":
            ("bert.encoder.layer.3.attention.self.query.weight", True),
        "bert/encoder/layer_3/attention/output/dense/kernel":
            ("bert.encoder.layer.3.attention.output.dense.weight", True),
        "bert/encoder/layer_3/intermediate/dense/kernel":
            ("bert.encoder.layer.3.intermediate.dense_act.weight", True),
        "bert/pooler/dense/bias": ("bert.pooler.dense_act.bias", False),
        "cls/predictions/output_bias": ("cls.predictions.bias", False),
        "cls/predictions/transform/dense/kernel":
            ("cls.predictions.transform.dense_act.weight", True),
        "cls/seq_relationship/output_weights":
            ("cls.seq_relationship.weight", False),
    }
    for tf_name, expected in cases.items():
        assert tf_name_to_state_key(tf_name) == expected
    assert tf_name_to_state_key("bert/adam_v/whatever") is None
    assert tf_name_to_state_key("global_step") is None


def _export_tf_style(model, prefix):
    """Write the model's weights under TF names (kernels transposed
    back to TF's [in, out] layout) — the inverse of load_tf_weights."""
    inverse = {}
    for tf_like in _ALL_TF_NAMES(model):
        key, transpose = tf_name_to_state_key(tf_like)
        inverse[tf_like] = (key, transpose)
    sd = model.state_dict()
    w = TFBundleWriter(prefix)
    for tf_name, (key, transpose) in inverse.items():
        t = sd[key].detach().float()
        if transpose:
            t = t.t().contiguous()
        w.add(tf_name, t.numpy())
    w.save()


def _ALL_TF_NAMES(model):
    n_layers = len(model.bert.encoder.layer)
    names = [
        "bert/embeddings/word_embeddings",
        "bert/embeddings/position_embeddings",
        "bert/embeddings/token_type_embeddings",
        "bert/embeddings/LayerNorm/gamma",
        "bert/embeddings/LayerNorm/beta",
        "bert/pooler/dense/kernel",
        "bert/pooler/dense/bias",
        "cls/predictions/output_bias",
        "cls/predictions/transform/dense/kernel",
        "cls/predictions/transform/dense/bias",
        "cls/predictions/transform/LayerNorm/gamma",
        "cls/predictions/transform/LayerNorm/beta",
        "cls/seq_relationship/output_weights",
        "cls/seq_relationship/output_bias",
    ]
    for i in range(n_layers):
        p = f"bert/encoder/layer_{i}"
        for sub in ("attention/self/query", "attention/self/key",
                    "attention/self/value", "attention/output/dense",
                    "intermediate/dense", "output/dense"):
            names += [f"{p}/{sub}/kernel", f"{p}/{sub}/bias"]
        for ln in ("attention/output/LayerNorm", "output/LayerNorm"):
            names += [f"{p}/{ln}/gamma", f"{p}/{ln}/beta"]
    return names


def test_load_tf_weights_end_to_end(tiny_config, tmp_path):
    torch.manual_seed(0)
    src = BertForPreTraining(tiny_config).eval()
    prefix = str(tmp_path / "bert_model.ckpt")
    _export_tf_style(src, prefix)

    torch.manual_seed(1)  # different init
    dst = BertForPreTraining(tiny_config).eval()
    load_tf_weights(dst, prefix)

    for k, v in src.state_dict().items():
        torch.testing.assert_close(
            dst.state_dict()[k], v, rtol=0, atol=0, msg=k
        )
    ids = torch.randint(0, tiny_config.vocab_size, (2, 16))
    mask = torch.ones(2, 16, dtype=torch.long)
    with torch.no_grad():
        s1, r1 = src(ids, None, mask)
        s2, r2 = dst(ids, None, mask)
    torch.testing.assert_close(s1, s2)
    torch.testing.assert_close(r1, r2)


def test_from_pretrained_tf_prefix(tiny_config, tmp_path):
    """from_pretrained auto-detects a TF bundle prefix (reference's
    from_tf path) and loads it through the in-repo parser."""
    torch.manual_seed(2)
    src = BertForPreTraining(tiny_config).eval()
    prefix = str(tmp_path / "bert_model.ckpt")
    _export_tf_style(src, prefix)
    dst = BertForPreTraining.from_pretrained(prefix, config=tiny_config)
    torch.testing.assert_close(
        dst.state_dict()["bert.embeddings.word_embeddings.weight"],
        src.state_dict()["bert.embeddings.word_embeddings.weight"],
    )


def test_bundle_malformed_files(tmp_path):
    short = tmp_path / "short.ckpt.index"
    short.write_bytes(b"tiny")
    with pytest.raises(ValueError, match="too short"):
        TFBundleReader(str(tmp_path / "short.ckpt"))
    bad = tmp_path / "bad.ckpt.index"
    bad.write_bytes(b"\x00" * 64)
    with pytest.raises(ValueError, match="magic"):
        TFBundleReader(str(tmp_path / "bad.ckpt"))


def test_snappy_copy2_copy4():
    # literal "ab" then copy2(offset 2, len 6) -> "abababab"
    comp = bytes([8, 0b00000100]) + b"ab" + \
        bytes([0b00010110]) + (2).to_bytes(2, "little")
    assert snappy_decompress(comp) == b"abababab"
    # literal "xy" then copy4(offset 2, len 2) -> "xyxy"
    comp = bytes([4, 0b00000100]) + b"xy" + \
        bytes([0b00000111]) + (2).to_bytes(4, "little")
    assert snappy_decompress(comp) == b"xyxy"


def test_bundle_truncated_data_shard(tmp_path):
    prefix = str(tmp_path / "m.ckpt")
    w = TFBundleWriter(prefix)
    w.add("t", np.arange(64, dtype=np.float32))
    w.save()
    data = prefix + ".data-00000-of-00001"
    with open(data, "r+b") as f:
        f.truncate(16)
    r = TFBundleReader(prefix)
    with pytest.raises(ValueError, match="truncated"):
        r.load_variable("t")
