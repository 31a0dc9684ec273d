"""Smoke tests: every example script runs end-to-end on CPU with tiny
arguments (the reference ships runnable examples/tutorials; bit-rot in
ours would be invisible otherwise)."""

import subprocess
import sys

import pytest

ARGS = {
    "train_simple_dnn.py": ["--iterations", "2", "--steps-per-iteration",
                            "6"],
    "customizing_adanet.py": ["--iterations", "2"],
    "adanet_objective.py": [],
    "train_improve_nas.py": ["--boosting-iterations", "1", "--train-steps",
                             "4", "--num-cells", "2", "--num-conv-filters",
                             "8", "--batch-size", "32"],
}


@pytest.mark.parametrize("script", sorted(ARGS))
def test_example_runs(script, tmp_path):
    args = list(ARGS[script])
    if script != "adanet_objective.py":
        args += ["--model-dir", str(tmp_path / "md")]
    r = subprocess.run(
        [sys.executable, "examples/%s" % script] + args,
        capture_output=True, text=True, timeout=600, cwd=".")
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
