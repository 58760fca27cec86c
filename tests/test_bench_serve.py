"""bench.py serve mode drives the BASELINE wire metric (driver contract).

Runs the flagship bench in CPU dry-run mode end to end: real dual-front-end
server subprocess, C GenerateStream clients, exactly-K-steps window armed
through the VTA_BENCH /bench/window endpoint, one JSON line on stdout.
"""

import json
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_bench_serve_cpu_dry_run():
    proc = subprocess.run(
        [sys.executable, str(REPO / "bench.py"),
         "--steps", "8", "--warmup", "2", "--batch", "16",
         "--grpc-port", "18913", "--http-port", "18914",
         "--server-log", "/tmp/test_bench_serve_server.log"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO),
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    line = proc.stdout.strip().splitlines()[-1]
    out = json.loads(line)
    assert out["metric"] == "grpc_stream_output_tokens_per_s"
    assert out["steps"] == 8 and out["warmup"] == 2
    assert out["value"] > 0
    assert out["ms_per_step"] > 0
    assert out["config"]["p50_ttft_ms"] > 0
    assert out["config"]["engine_tokens_per_s"] > 0
    assert out["data"] == "synthetic"
