"""In-tree build of the ``_gar_hip`` torch extension for gfx950.

(The reference's auto-compiling native-op infrastructure equivalent,
/root/reference/native/__init__.py:208-304: mtime-incremental compilation of
the native op library at import/build time; here a single hipcc invocation
produces one in-tree .so.)

Direct hipcc invocation (no hipify, no CUDA compatibility machinery): the
kernels are native HIP/CDNA4 and the glue uses the c10::hip API directly.
The resulting ``_gar_hip.so`` is written next to this file so it ships with
the repo snapshot to GPU boxes (it is .gitignore'd, not .gpurunignore'd).

Usage: ``python -m aggregathor_amd.ops.build`` (re-builds only when sources
are newer than the .so; ``--force`` to rebuild).
"""

import os
import pathlib
import subprocess
import sys
import sysconfig

OPS_DIR = pathlib.Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
OUT = OPS_DIR / "_gar_hip.so"

SOURCES = [CSRC / "gar_ops.cpp", CSRC / "gar_kernels.hip",
           CSRC / "bn_kernels.hip"]
HEADERS = [CSRC / "gar_kernels.h"]

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _torch_paths():
    import torch.utils.cpp_extension as ce
    return ce.include_paths("cuda"), ce.library_paths("cuda")


def needs_build():
    if not OUT.exists():
        return True
    out_mtime = OUT.stat().st_mtime
    return any(p.stat().st_mtime > out_mtime for p in SOURCES + HEADERS)


def build(force=False, verbose=True):
    if not force and not needs_build():
        if verbose:
            print(f"[ops.build] {OUT.name} up to date")
        return OUT
    import torch
    includes, libdirs = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)
    cmd = [
        "hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-fPIC", "-shared",
        # NaN ordering is load-bearing in every GAR: never fast-math.
        "-fno-fast-math",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1",
        "-DTORCH_EXTENSION_NAME=_gar_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
    ]
    for inc in includes + [py_inc]:
        cmd.append(f"-I{inc}")
    cmd += [str(s) for s in SOURCES]
    for lib in libdirs:
        cmd.append(f"-L{lib}")
        cmd.append(f"-Wl,-rpath,{lib}")
    cmd += ["-ltorch", "-ltorch_hip", "-lc10", "-lc10_hip", "-ltorch_python",
            "-lamdhip64", "-o", str(OUT)]
    if verbose:
        print("[ops.build]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    if verbose:
        print(f"[ops.build] built {OUT}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
