"""In-tree build of the _hipshuffle extension for gfx950.

Direct hipcc invocation (no JIT cache): the produced .so sits next to this
file so it travels to GPU boxes with the repo snapshot.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def ext_path() -> str:
    suffix = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
    return os.path.join(OPS_DIR, f"_hipshuffle{suffix}")


def sources():
    return [os.path.join(CSRC, f) for f in ("pool.cpp", "kernels.hip",
                                            "bindings.cpp")]


def needs_build() -> bool:
    out = ext_path()
    if not os.path.exists(out):
        return True
    out_mtime = os.path.getmtime(out)
    deps = sources() + [os.path.join(CSRC, h)
                        for h in ("common.h", "hipshuffle.h")]
    return any(os.path.getmtime(s) > out_mtime for s in deps)


def build(verbose: bool = True) -> str:
    import pybind11
    hipcc = os.environ.get("HIPCC", "hipcc")
    py_include = sysconfig.get_paths()["include"]
    cmd = [
        hipcc, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-shared", "-x", "hip",           # treat .cpp uniformly as HIP
        f"-I{CSRC}", f"-I{py_include}", f"-I{pybind11.get_include()}",
        "-fvisibility=hidden",
        *sources(),
        "-o", ext_path(),
    ]
    if verbose:
        print("+", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return ext_path()


def ensure_built(verbose: bool = False) -> str:
    if needs_build():
        build(verbose=verbose)
    return ext_path()


if __name__ == "__main__":
    build()
