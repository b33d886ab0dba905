"""In-tree hipcc build for the quoracle_amd HIP extension (gfx950).

Builds quoracle_amd/ops/_quoracle_ops.so directly with hipcc (no JIT cache —
the .so must travel with the repo snapshot to the GPU box).  hipcc
cross-compiles gfx950 without a GPU, so this runs in CPU-only CI too.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
SO_PATH = os.path.join(HERE, "_quoracle_ops.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch
    from torch.utils import cpp_extension
    includes = cpp_extension.include_paths()
    lib_dir = os.path.join(os.path.dirname(torch.__file__), "lib")
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    return includes, lib_dir, abi


def build(force: bool = False, verbose: bool = True) -> str:
    sources = [os.path.join(CSRC, "bindings.cpp")]
    headers = [os.path.join(CSRC, f) for f in
               ("common.h", "forward.h", "attention.hip", "attention_mfma.hip", "elementwise.hip")]
    if not force and os.path.exists(SO_PATH):
        newest_src = max(os.path.getmtime(p) for p in sources + headers)
        if os.path.getmtime(SO_PATH) >= newest_src:
            return SO_PATH

    includes, lib_dir, abi = _torch_paths()
    py_include = sysconfig.get_paths()["include"]
    cmd = [
        "hipcc", "-x", "hip", f"--offload-arch={ARCH}",
        "-O3", "-std=c++17", "-fPIC", "-shared",
        "-DTORCH_EXTENSION_NAME=_quoracle_ops",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DUSE_ROCM=1", "-D__HIP_PLATFORM_AMD__=1",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-fno-gpu-rdc",
        "-Wno-unused-result",
    ]
    for inc in includes + [py_include, CSRC]:
        cmd.append(f"-I{inc}")
    cmd += sources
    cmd += [f"-L{lib_dir}", "-ltorch", "-ltorch_cpu", "-ltorch_hip",
            "-lc10", "-lc10_hip", "-ltorch_python", "-lamdhip64",
            f"-Wl,-rpath,{lib_dir}",
            "-o", SO_PATH]
    if verbose:
        print("[ops.build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
