"""In-tree build of the CDNA4 HIP extension.

Direct hipcc invocation (no hipify, no CUDA-compat layer): compiles every
kernels/*.hip plus the torch bindings for gfx950 and links
vllm_tgis_adapter_amd/_C.so.  The built .so travels with the repo snapshot to
GPU boxes (a JIT cache would not).
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
KERNELS = REPO / "kernels"
OUT = REPO / "vllm_tgis_adapter_amd" / "_C.so"
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def build(verbose: bool = True, force: bool = False) -> Path:
    import torch
    import torch.utils.cpp_extension as ce

    sources = sorted(KERNELS.glob("*.hip")) + [KERNELS / "bindings.cpp"]
    if OUT.exists() and not force:
        newest_src = max(p.stat().st_mtime for p in sources + [KERNELS / "common.h"])
        if OUT.stat().st_mtime > newest_src:
            if verbose:
                print(f"[kernels/build] {OUT} up to date")
            return OUT

    include_dirs = ce.include_paths() + [sysconfig.get_path("include")]
    torch_lib = Path(torch.__file__).parent / "lib"
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)

    build_dir = KERNELS / "build"
    build_dir.mkdir(exist_ok=True)

    common_flags = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1",
        "-DHIPBLAS_V2",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
        "-Wno-switch-bool",
    ] + [f"-I{d}" for d in include_dirs]

    objects = []
    for src in sources:
        obj = build_dir / (src.stem + ".o")
        objects.append(obj)
        if obj.exists() and not force and obj.stat().st_mtime > max(
            src.stat().st_mtime, (KERNELS / "common.h").stat().st_mtime
        ):
            continue
        cmd = ["hipcc", "-c", str(src), "-o", str(obj)] + common_flags
        if verbose:
            print("[kernels/build]", " ".join(cmd[:4]), "...")
        subprocess.run(cmd, check=True)

    link_cmd = (
        ["hipcc", "-shared", "-fPIC"]
        + [str(o) for o in objects]
        + [f"-L{torch_lib}", "-ltorch", "-ltorch_python", "-ltorch_hip",
           "-lc10", "-lc10_hip", "-lamdhip64",
           f"-Wl,-rpath,{torch_lib}"]
        + ["-o", str(OUT)]
    )
    if verbose:
        print("[kernels/build] linking", OUT.name)
    subprocess.run(link_cmd, check=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
