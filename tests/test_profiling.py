"""Profiling utilities (StepTimer phases, torch_profile artifacts)."""

import glob
import time

import torch

from bert_pytorch_amd.utils.profiling import StepTimer, torch_profile


def test_step_timer_cpu_phases():
    timer = StepTimer(enabled=True, use_cuda=False)
    for _ in range(3):
        with timer.phase("data"):
            time.sleep(0.003)
        with timer.phase("forward"):
            time.sleep(0.001)
        timer.step_end()
    s = timer.summary()
    assert s["data"] > s["forward"] > 0
    assert timer.steps == 3
    assert "data=" in timer.format_summary()


def test_step_timer_disabled_noop():
    timer = StepTimer(enabled=False)
    with timer.phase("forward"):
        pass
    timer.step_end()
    assert timer.summary() == {}


def test_torch_profile_writes_artifacts(tmp_path):
    with torch_profile(str(tmp_path), enabled=True) as prof:
        x = torch.randn(64, 64)
        (x @ x).sum()
    assert prof is not None
    assert glob.glob(str(tmp_path / "trace_rank0.json"))
    ops_txt = (tmp_path / "ops_rank0.txt").read_text()
    assert "mm" in ops_txt or "matmul" in ops_txt


def test_torch_profile_disabled(tmp_path):
    with torch_profile(str(tmp_path), enabled=False) as prof:
        pass
    assert prof is None
    assert not glob.glob(str(tmp_path / "*"))
