"""In-tree build of the CDNA4 kernel library (libnxd_ops.so).

Hand-driven hipcc (no hipify, no CUDA shims): every ``csrc/*.hip`` is
gfx950 device code compiled with ``--offload-arch=gfx950`` and linked into
one shared library loaded via ctypes.  Built IN-TREE so the .so travels to
the GPU box with the repo snapshot.
"""

import os
import subprocess
import sys

_THIS = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(_THIS, "csrc")
LIB = os.path.join(_THIS, "libnxd_ops.so")

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("NXDA_GPU_ARCH", "gfx950")


def _sources():
    return sorted(
        os.path.join(CSRC, f) for f in os.listdir(CSRC) if f.endswith(".hip")
    )


def _needs_build():
    if not os.path.exists(LIB):
        return True
    lib_mtime = os.path.getmtime(LIB)
    deps = _sources() + [os.path.join(CSRC, h)
                         for h in ("common.h", "mfma.h")]
    return any(os.path.getmtime(s) > lib_mtime for s in deps)


def build(force: bool = False, verbose: bool = True) -> str:
    if not force and not _needs_build():
        return LIB
    srcs = _sources()
    objs = []
    for s in srcs:
        o = s.replace(".hip", ".o")
        cmd = [
            HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
            "-ffp-contract=fast", "-c", s, "-o", o,
        ]
        if verbose:
            print("[nxd_ops]", " ".join(cmd), file=sys.stderr)
        subprocess.check_call(cmd)
        objs.append(o)
    cmd = [HIPCC, f"--offload-arch={ARCH}", "-shared", "-fPIC", "-o", LIB] + objs
    if verbose:
        print("[nxd_ops]", " ".join(cmd), file=sys.stderr)
    subprocess.check_call(cmd)
    return LIB


if __name__ == "__main__":
    build(force="--force" in sys.argv)
