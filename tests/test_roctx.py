"""rocTX range plumbing (SURVEY.md §5 tracing addendum).

Checks the module is a zero-overhead no-op when disabled (the default and
the CPU-CI case) and that the ctypes binding path activates when
``VTA_ROCTX=1`` and ``libroctx64.so`` is present.
"""

import subprocess
import sys


def test_disabled_by_default_noop():
    from vllm_tgis_adapter_amd.engine import roctx

    assert not roctx.enabled
    roctx.range_push("x")  # must not raise
    roctx.range_pop()
    with roctx.trace_range("y"):
        pass


def test_enabled_via_env_subprocess():
    # Fresh interpreter so the import-time env check runs with VTA_ROCTX=1.
    code = (
        "import ctypes, sys\n"
        "from vllm_tgis_adapter_amd.engine import roctx\n"
        "try:\n"
        "    ctypes.CDLL('librocprofiler-sdk-roctx.so')\n"
        "    have_lib = True\n"
        "except OSError:\n"
        "    have_lib = False\n"
        "assert roctx.enabled == have_lib, (roctx.enabled, have_lib)\n"
        "with roctx.trace_range('step'):\n"
        "    roctx.range_push('inner'); roctx.range_pop()\n"
        "print('ROCTX_OK', roctx.enabled)\n"
    )
    out = subprocess.run(
        [sys.executable, "-c", code],
        env={"VTA_ROCTX": "1", "PATH": "/usr/bin:/bin",
             "LD_LIBRARY_PATH": "/opt/rocm/lib"},
        capture_output=True, text=True, timeout=120, cwd="/root/repo",
    )
    assert out.returncode == 0, out.stderr
    assert "ROCTX_OK" in out.stdout
