"""Finetune runners end-to-end on CPU: tokenization, SQuAD featurize/
train/predict/eval, NER train/eval (tiny synthetic data)."""

import json
import os

import pytest
import torch

import run_ner
import run_squad
from bert_pytorch_amd.data import squad as squad_data
from bert_pytorch_amd.data.ner_dataset import NERDataset
from bert_pytorch_amd.data.tokenization import (
    BasicTokenizer,
    BertTokenizer,
    get_wordpiece_tokenizer,
)

VOCAB = (
    "[PAD] [UNK] [CLS] [SEP] [MASK] the capital of france is paris a city "
    "in europe what country big mountain ##s ##ing and it ! ? . , berlin "
    "germany where located"
).split()


@pytest.fixture
def vocab_file(tmp_path):
    path = tmp_path / "vocab.txt"
    path.write_text("\n".join(VOCAB))
    return str(path)


@pytest.fixture
def model_cfg(tmp_path):
    cfg = {
        "vocab_size": len(VOCAB),
        "hidden_size": 64,
        "num_hidden_layers": 2,
        "num_attention_heads": 4,
        "intermediate_size": 128,
        "max_position_embeddings": 128,
        "type_vocab_size": 2,
        "hidden_act": "gelu",
        "hidden_dropout_prob": 0.1,
        "attention_probs_dropout_prob": 0.1,
        "initializer_range": 0.02,
        "next_sentence": True,
        "lowercase": True,
    }
    path = tmp_path / "model.json"
    path.write_text(json.dumps(cfg))
    return str(path)


@pytest.fixture
def squad_file(tmp_path):
    data = {
        "version": "1.1",
        "data": [{
            "title": "t",
            "paragraphs": [
                {
                    "context": "The capital of France is Paris.",
                    "qas": [{
                        "id": "q1",
                        "question": "What is the capital of France?",
                        "answers": [{"text": "Paris", "answer_start": 25}],
                    }],
                },
                {
                    "context": "Berlin is a city in Germany.",
                    "qas": [{
                        "id": "q2",
                        "question": "Where is Berlin located?",
                        "answers": [{"text": "Germany", "answer_start": 20}],
                    }],
                },
            ],
        }],
    }
    path = tmp_path / "squad.json"
    path.write_text(json.dumps(data))
    return str(path)


def test_basic_tokenizer():
    bt = BasicTokenizer(do_lower_case=True)
    assert bt.tokenize("Hello, World!") == ["hello", ",", "world", "!"]
    assert bt.tokenize("  a\tb\nc ") == ["a", "b", "c"]
    assert bt.tokenize("Café") == ["cafe"]  # accent stripped
    # CJK chars split into single-char tokens (reference
    # src/tokenization.py CJK padding semantics)
    assert bt.tokenize("ab\u4e2d\u6587cd") == ["ab", "\u4e2d", "\u6587", "cd"]
    # control chars removed, case preserved when lowercasing is off
    assert BasicTokenizer(do_lower_case=False).tokenize("A\x00B") == ["AB"]


def test_wordpiece_tokenizer_roundtrip(vocab_file):
    tok = get_wordpiece_tokenizer(vocab_file, lowercase=True)
    enc = tok.encode("The capital of France is Paris!")
    assert enc.tokens[0] == "[CLS]" and enc.tokens[-1] == "[SEP]"
    assert "paris" in enc.tokens
    assert tok.id_to_token(enc.ids[1]) == enc.tokens[1]
    legacy = BertTokenizer(vocab_file, do_lower_case=True)
    assert legacy.tokenize("the capital") == ["the", "capital"]


def test_squad_read_and_featurize(vocab_file, squad_file):
    tok = get_wordpiece_tokenizer(vocab_file, lowercase=True)
    examples = squad_data.read_squad_examples(squad_file, is_training=True)
    assert len(examples) == 2
    assert examples[0].orig_answer_text == "Paris"
    features = squad_data.convert_examples_to_features(
        examples, tok, max_seq_length=64, doc_stride=32, max_query_length=16,
        is_training=True,
    )
    assert len(features) >= 2
    f = features[0]
    assert len(f.input_ids) == 64
    assert f.tokens[f.start_position] == "paris"
    assert f.tokens[f.end_position] == "paris"


def test_squad_evaluator(squad_file):
    metrics = squad_data.evaluate_predictions(
        squad_file, {"q1": "Paris", "q2": "berlin"}
    )
    assert metrics["exact_match"] == 50.0
    assert 50.0 <= metrics["f1"] <= 100.0


def test_run_squad_end_to_end(tmp_path, vocab_file, model_cfg, squad_file):
    args = run_squad.parse_args([
        "--model_config_file", model_cfg,
        "--vocab_file", vocab_file,
        "--train_file", squad_file,
        "--predict_file", squad_file,
        "--output_dir", str(tmp_path / "out"),
        "--do_train", "--do_predict", "--do_eval",
        "--train_batch_size", "2",
        "--predict_batch_size", "2",
        "--num_train_epochs", "1",
        "--max_seq_length", "64",
        "--doc_stride", "32",
    ])
    results = run_squad.main(args)
    assert "exact_match" in results and "f1" in results
    assert os.path.exists(tmp_path / "out" / "predictions.json")
    preds = json.loads((tmp_path / "out" / "predictions.json").read_text())
    assert set(preds.keys()) == {"q1", "q2"}
    assert os.path.exists(tmp_path / "out" / "pytorch_model.bin")


@pytest.fixture
def conll_file(tmp_path):
    lines = []
    sents = [
        [("paris", "B-LOC"), ("is", "O"), ("big", "O")],
        [("berlin", "B-LOC"), ("and", "O"), ("france", "B-LOC")],
        [("the", "O"), ("city", "O"), ("of", "O"), ("paris", "B-LOC")],
    ]
    for sent in sents:
        for word, label in sent:
            lines.append(f"{word} {label}")
        lines.append("")
    path = tmp_path / "train.txt"
    path.write_text("\n".join(lines))
    return str(path)


def test_ner_dataset(vocab_file, conll_file):
    tok = get_wordpiece_tokenizer(vocab_file, lowercase=True)
    ds = NERDataset(conll_file, tok, max_seq_len=32)
    assert len(ds) == 3
    ids, mask, labels = ds[0]
    assert ids.shape == (32,)
    assert (labels != -100).sum() == 3  # one label per word


def test_run_ner_end_to_end(tmp_path, vocab_file, model_cfg, conll_file):
    args = run_ner.parse_args([
        "--model_config_file", model_cfg,
        "--vocab_file", vocab_file,
        "--data_dir", os.path.dirname(conll_file),
        "--train_file", os.path.basename(conll_file),
        "--eval_file", os.path.basename(conll_file),
        "--output_dir", str(tmp_path / "nerout"),
        "--batch_size", "2",
        "--epochs", "1",
        "--do_train", "--do_eval",
    ])
    run_ner.main(args)
    assert os.path.exists(tmp_path / "nerout" / "pytorch_model.bin")


def test_basic_tokenizer_never_split():
    """Special tokens pass through untouched (reference
    src/tokenization.py:64-65,74)."""
    bt = BasicTokenizer(do_lower_case=True)
    assert bt.tokenize("foo [UNK] bar") == ["foo", "[UNK]", "bar"]
    assert bt.tokenize("[CLS] Hi [SEP]") == ["[CLS]", "hi", "[SEP]"]
    # default list covers all five BERT specials
    for tok in ("[UNK]", "[SEP]", "[PAD]", "[CLS]", "[MASK]"):
        assert bt.tokenize(f"x {tok} y") == ["x", tok, "y"]
    # custom never_split overrides the default
    bt2 = BasicTokenizer(do_lower_case=True, never_split=("<KEEP>",))
    assert bt2.tokenize("a <KEEP> b") == ["a", "<KEEP>", "b"]
    assert bt2.tokenize("a [UNK] b") == ["a", "[", "unk", "]", "b"]


def test_metric_logger_csv_stable_schema(tmp_path):
    """A train+eval log with different metric keys parses whole with
    csv.DictReader (schema widening rewrites under a union header)."""
    import csv as _csv

    from bert_pytorch_amd.utils.logging import MetricLogger

    prefix = str(tmp_path / "run")
    ml = MetricLogger(log_prefix=prefix, verbose=False)
    ml.log("train", 1, loss=1.5, lr=0.01)
    ml.log("train", 2, loss=1.2, lr=0.01)
    ml.log("eval", 2, f1=0.7)  # new field mid-file
    ml.log("train", 3, loss=1.0, lr=0.009)
    ml.close()
    with open(prefix + "_metrics.csv", newline="") as f:
        rows = list(_csv.DictReader(f))
    assert len(rows) == 4
    assert all(set(r.keys()) == {"tag", "step", "loss", "lr", "f1"} for r in rows)
    assert rows[0]["loss"] == "1.5" and rows[2]["f1"] == "0.7"
    assert rows[2]["loss"] == "" and rows[0]["f1"] == ""
    # resume into the same file keeps one header and the schema
    ml2 = MetricLogger(log_prefix=prefix, verbose=False)
    ml2.log("train", 4, loss=0.9, lr=0.008)
    ml2.close()
    with open(prefix + "_metrics.csv", newline="") as f:
        content = f.read()
    assert content.count("tag,step") == 1
    with open(prefix + "_metrics.csv", newline="") as f:
        rows = list(_csv.DictReader(f))
    assert len(rows) == 5 and rows[4]["loss"] == "0.9"
