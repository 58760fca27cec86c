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


def test_bench_serve_tp2_torchrun_cpu():
    """World-2 dry run of the DRIVER's exact invocation shape.

    The round-end scale bench launches ``torch.distributed.run --nnodes=1
    --nproc-per-node N ... bench.py --gpus N`` (one rank per GPU).  This
    runs that same shape on CPU/gloo with N=2: rank 0 spawns the dual
    server subprocess (which takes the rank-0 slot in the rendezvous and
    drives TP-2 sharded workers over broadcast), rank 1 enters
    serve_worker_rank, and the client window must deliver tokens.
    """
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29731", str(REPO / "bench.py"),
         "--gpus", "2", "--steps", "4", "--warmup", "2", "--batch", "8",
         "--grpc-port", "18915", "--http-port", "18916",
         "--server-log", "/tmp/test_bench_serve_tp2_server.log"],
        capture_output=True, text=True, timeout=420, cwd=str(REPO),
    )
    assert proc.returncode == 0, (proc.stdout[-1500:], proc.stderr[-2500:])
    json_lines = [ln for ln in proc.stdout.splitlines()
                  if ln.startswith("{") and '"metric"' in ln]
    assert json_lines, proc.stdout[-2000:]
    out = json.loads(json_lines[-1])
    assert out["metric"] == "grpc_stream_output_tokens_per_s"
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "tp2"
    assert out["value"] > 0


def test_bench_engine_tp2_torchrun_cpu():
    """Same world-2 torchrun shape for ``--mode engine`` (bare step loop)."""
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29732", str(REPO / "bench.py"),
         "--gpus", "2", "--mode", "engine",
         "--steps", "4", "--warmup", "2", "--batch", "8"],
        capture_output=True, text=True, timeout=420, cwd=str(REPO),
    )
    assert proc.returncode == 0, (proc.stdout[-1500:], proc.stderr[-2500:])
    json_lines = [ln for ln in proc.stdout.splitlines()
                  if ln.startswith("{") and '"metric"' in ln]
    assert json_lines, proc.stdout[-2000:]
    out = json.loads(json_lines[-1])
    assert out["metric"] == "engine_output_tokens_per_s"
    assert out["n_gpus"] == 2
    assert out["value"] > 0


def test_bench_serve_top_n_dry_run():
    """--top-n drives the wire logprob/token-detail path end to end."""
    proc = subprocess.run(
        [sys.executable, str(REPO / "bench.py"),
         "--steps", "6", "--warmup", "2", "--batch", "8", "--top-n", "5",
         "--grpc-port", "18919", "--http-port", "18920",
         "--server-log", "/tmp/test_bench_serve_topn_server.log"],
        capture_output=True, text=True, timeout=300, cwd=str(REPO),
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    out = json.loads(proc.stdout.strip().splitlines()[-1])
    assert out["value"] > 0
