"""Failure-path tests (reference tests/test_termination_log.py:16-36):
startup crashes must be written to the Kubernetes termination log."""

from __future__ import annotations

import os
import subprocess
import sys

import pytest


@pytest.mark.parametrize("argv, needle", [
    (["--model-name", "no-such-model-xyz"], "no-such-model-xyz"),
    (["--model-name", "tiny-llama", "--max-sequence-length", "10241024"],
     "10241024"),
])
def test_startup_crash_writes_termination_log(tmp_path, argv, needle):
    env = dict(os.environ)
    log = tmp_path / "termination-log"
    log.touch()  # the writer only writes where k8s pre-created the file
    env["TERMINATION_LOG_DIR"] = str(log)
    proc = subprocess.run(
        [sys.executable, "-m", "vllm_tgis_adapter_amd", *argv,
         "--grpc-port", "0", "--port", "0"],
        capture_output=True, text=True, env=env, timeout=120,
    )
    assert proc.returncode != 0
    assert needle in log.read_text(), proc.stderr[-1500:]
