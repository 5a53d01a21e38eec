"""Sanitizer harness test for the C++ tokenizer core."""

import os

import pytest




def test_tokenizer_core_sanitizers():
    """ASan+UBSan pass over the C++ tokenizer core with adversarial
    inputs (SURVEY §5 sanitizers row; scripts/sanitize_check.sh)."""
    import shutil
    import subprocess

    if shutil.which("g++") is None:
        pytest.skip("no host g++")
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    res = subprocess.run(
        ["bash", os.path.join(root, "scripts", "sanitize_check.sh")],
        capture_output=True, text=True, timeout=300,
    )
    assert res.returncode == 0, res.stdout + res.stderr
    assert "SANITIZE OK" in res.stdout
