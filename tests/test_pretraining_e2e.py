"""End-to-end pretraining on CPU: config merge, train loop, checkpoint,
resume, two-phase hand-off (BASELINE config 1 — the plumbing slice)."""

import json
import os

import pytest
import torch

import run_pretraining
from bert_pytorch_amd.data import synth
from bert_pytorch_amd.utils import checkpoint as ckpt_io


@pytest.fixture
def workspace(tmp_path):
    data_dir = tmp_path / "data"
    synth.make_dataset(
        str(data_dir), num_shards=2, samples_per_shard=32, seq_len=32,
        vocab_size=512, seed=0,
    )
    model_cfg = {
        "vocab_size": 512,
        "hidden_size": 64,
        "num_hidden_layers": 2,
        "num_attention_heads": 4,
        "intermediate_size": 128,
        "max_position_embeddings": 64,
        "type_vocab_size": 2,
        "hidden_act": "gelu",
        "hidden_dropout_prob": 0.1,
        "attention_probs_dropout_prob": 0.1,
        "initializer_range": 0.02,
        "next_sentence": True,
    }
    cfg_path = tmp_path / "model.json"
    cfg_path.write_text(json.dumps(model_cfg))
    return tmp_path, str(data_dir), str(cfg_path)


def _args(tmp_path, data_dir, cfg_path, **overrides):
    argv = [
        "--model_config_file", cfg_path,
        "--input_dir", data_dir,
        "--output_dir", str(tmp_path / "out"),
        "--local_batch_size", "4",
        "--global_batch_size", "8",
        "--max_steps", "6",
        "--learning_rate", "1e-3",
        "--warmup_proportion", "0.2",
        "--num_steps_per_checkpoint", "2",
        "--seed", "7",
        "--num_workers", "0",
    ]
    for key, value in overrides.items():
        argv += [f"--{key}", str(value)]
    return run_pretraining.parse_arguments(argv)


def test_train_and_checkpoint(workspace):
    tmp_path, data_dir, cfg_path = workspace
    args = _args(tmp_path, data_dir, cfg_path)
    steps = run_pretraining.main(args)
    assert steps == 6
    latest = ckpt_io.find_latest(str(tmp_path / "out"))
    assert latest is not None
    path, step = latest
    assert step == 6
    state = ckpt_io.load(path)
    for key in ("model", "optimizer", "sampler", "epoch"):
        assert key in state
    assert state["optimizer"]["param_groups"][0]["step"] == 6
    # metrics CSV written
    csvs = list((tmp_path / "out").glob("*_metrics.csv"))
    assert csvs, "metrics CSV missing"


def test_resume_continues_from_checkpoint(workspace):
    tmp_path, data_dir, cfg_path = workspace
    args = _args(tmp_path, data_dir, cfg_path, max_steps=4)
    run_pretraining.main(args)
    # resume: same output dir, higher max_steps
    args2 = _args(tmp_path, data_dir, cfg_path, max_steps=8)
    steps = run_pretraining.main(args2)
    assert steps == 8
    _, step = ckpt_io.find_latest(str(tmp_path / "out"))
    assert step == 8


def test_two_phase_handoff(workspace):
    """Phase-2 resumes a phase-1 checkpoint with new hyperparameters
    (reference: run_pretraining.py:298-309)."""
    tmp_path, data_dir, cfg_path = workspace
    args1 = _args(tmp_path, data_dir, cfg_path, max_steps=4)
    run_pretraining.main(args1)
    args2 = _args(
        tmp_path, data_dir, cfg_path,
        max_steps=3, previous_phase_end_step=4, learning_rate="5e-4",
    )
    steps = run_pretraining.main(args2)
    assert steps == 3
    _, step = ckpt_io.find_latest(str(tmp_path / "out"))
    assert step == 7  # 4 (phase 1) + 3 (phase 2)


def test_config_file_merge(workspace, tmp_path):
    _, data_dir, cfg_path = workspace
    train_cfg = {
        "model_config_file": cfg_path,
        "learning_rate": 2e-3,
        "max_steps": 5,
        "local_batch_size": 2,
        "global_batch_size": 4,
    }
    cfg_file = tmp_path / "train.json"
    cfg_file.write_text(json.dumps(train_cfg))
    args = run_pretraining.parse_arguments(
        ["--config_file", str(cfg_file), "--input_dir", data_dir,
         "--max_steps", "9"]  # CLI overrides JSON
    )
    assert args.learning_rate == 2e-3  # from JSON
    assert args.max_steps == 9  # CLI wins
    assert args.local_batch_size == 2
    assert args.model_config_file == cfg_path


def test_roberta_path_adam_no_nsp(tmp_path):
    """RoBERTa single-phase: no NSP (2-entry special_token_positions),
    fused Adam, linear decay (BASELINE config 4 semantics on CPU)."""
    data_dir = tmp_path / "data"
    synth.make_dataset(
        str(data_dir), num_shards=1, samples_per_shard=32, seq_len=32,
        vocab_size=512, seed=0, nsp=False,
    )
    model_cfg = {
        "vocab_size": 512, "hidden_size": 64, "num_hidden_layers": 2,
        "num_attention_heads": 4, "intermediate_size": 128,
        "max_position_embeddings": 64, "type_vocab_size": 2,
        "hidden_act": "gelu", "hidden_dropout_prob": 0.1,
        "attention_probs_dropout_prob": 0.1, "initializer_range": 0.02,
        "next_sentence": False,
    }
    cfg_path = tmp_path / "roberta.json"
    cfg_path.write_text(json.dumps(model_cfg))
    args = _args(
        tmp_path, str(data_dir), str(cfg_path),
        optimizer="adam", lr_decay="linear", max_steps="4",
    )
    steps = run_pretraining.main(args)
    assert steps == 4
    ckpts = list((tmp_path / "out" / "pretrain_ckpts").glob("ckpt_*.pt"))
    assert ckpts
    state = torch.load(ckpts[0], map_location="cpu", weights_only=False)
    # NSP head absent from the checkpoint for the RoBERTa config
    assert not any("seq_relationship" in k for k in state["model"])


def test_pure_bf16_mode(workspace):
    """--pure_bf16: bf16 weights with fp32 masters trains and checkpoints."""
    tmp_path, data_dir, cfg_path = workspace
    args = _args(tmp_path, data_dir, cfg_path, max_steps="3")
    args.pure_bf16 = True
    steps = run_pretraining.main(args)
    assert steps == 3
    ckpts = sorted((tmp_path / "out" / "pretrain_ckpts").glob("ckpt_*.pt"))
    state = torch.load(ckpts[-1], map_location="cpu", weights_only=False)
    w = next(v for k, v in state["model"].items() if "word_embeddings" in k)
    assert w.dtype == torch.bfloat16
    # optimizer state keeps fp32 masters
    assert any(
        "master" in s
        for s in state["optimizer"]["state"].values()
    )
