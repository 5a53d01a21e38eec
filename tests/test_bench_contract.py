"""bench.py driver-contract test.

The round driver runs ``python bench.py --gpus N --steps K --warmup W``
and parses the single JSON line rank 0 prints (see bench.py docstring).
This test runs the real script on CPU with a tiny model config and
asserts the schema the driver depends on, plus cwd-independence of the
default config path.
"""

from __future__ import annotations

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

TINY = {
    "vocab_size": 512,
    "hidden_size": 64,
    "num_hidden_layers": 2,
    "num_attention_heads": 4,
    "intermediate_size": 128,
    "hidden_act": "gelu",
    "hidden_dropout_prob": 0.1,
    "attention_probs_dropout_prob": 0.1,
    "max_position_embeddings": 512,
    "type_vocab_size": 2,
    "initializer_range": 0.02,
}

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


@pytest.mark.timeout(300)
def test_bench_json_contract(tmp_path):
    cfg = tmp_path / "tiny.json"
    cfg.write_text(json.dumps(TINY))
    # run from a DIFFERENT cwd to cover default-path resolution too
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--gpus", "1", "--steps", "2", "--warmup", "1",
         "--local_batch", "2", "--accumulation", "2",
         "--model_config", str(cfg)],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=280,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    rec = json.loads(line)
    assert REQUIRED_KEYS.issubset(rec.keys()), sorted(REQUIRED_KEYS - set(rec))
    assert rec["n_gpus"] == 1 and rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["metric"] == "sequences/sec"
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    # whole-job value consistency: value = local_batch * 1e3 / ms_per_step
    assert rec["value"] == pytest.approx(2 * 1e3 / rec["ms_per_step"], rel=0.05)
    assert rec["data"] == "synthetic"
    assert rec["config"]["seq_len"] == 128
    # global_batch = world * local_batch * accumulation
    assert rec["config"]["global_batch"] == 4


@pytest.mark.timeout(300)
def test_bench_two_rank_gloo(tmp_path):
    """bench.py under the driver's exact torchrun launch pattern
    (--nnodes=1 --nproc-per-node N --master-addr 127.0.0.1) with
    world_size 2 on CPU/gloo: rank 0 prints one whole-job JSON line."""
    cfg = tmp_path / "tiny.json"
    cfg.write_text(json.dumps(TINY))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29518",
         os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--local_batch", "2", "--accumulation", "2",
         "--model_config", str(cfg)],
        cwd=str(tmp_path), capture_output=True, text=True, timeout=280,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, "exactly one JSON line (rank 0 only)"
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2
    assert rec["config"]["parallelism"] == "dp2"
    # whole-job value: world * local_batch * 1e3 / ms_per_step
    assert rec["value"] == pytest.approx(2 * 2 * 1e3 / rec["ms_per_step"], rel=0.05)
