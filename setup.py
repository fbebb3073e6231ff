"""Build the mpi4jax_amd native HIP/RCCL extension in-tree for gfx950.

Drives hipcc directly (no hipify, no CUDA shims — the sources are native
HIP/CDNA4).  Usage:

    python setup.py build_ext --inplace

or programmatically: ``from setup import build_native; build_native()``
(used by __graft_entry__.build()).
"""

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

ROOT = Path(__file__).resolve().parent
CSRC = ROOT / "mpi4jax_amd" / "csrc"
OUT_SO = ROOT / "mpi4jax_amd" / "_rccl_C.so"
BUILD = ROOT / "build"

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def _torch_paths():
    import torch.utils.cpp_extension as ce

    return ce.include_paths("cuda"), ce.library_paths("cuda")


def _run(cmd):
    print("+", " ".join(str(c) for c in cmd), flush=True)
    subprocess.check_call([str(c) for c in cmd])


def _newer(target, sources):
    if not target.exists():
        return True
    t = target.stat().st_mtime
    return any(s.stat().st_mtime > t for s in sources)


def build_native(force=False, variant=""):
    """Build the in-tree extension.

    variant="nofma" builds ``_rccl_C_nofma.so`` with the device kernels
    compiled ``-ffp-contract=off`` — the FMA-contraction control used to
    pin the f64 fused-vs-eager divergence to FMA alone
    (tests/test_gpu_nofma.py); select it at runtime with
    ``MPI4JAX_AMD_SW_EXT=nofma``.
    """
    includes, libpaths = _torch_paths()
    py_inc = sysconfig.get_paths()["include"]
    suffix = f"_{variant}" if variant else ""
    ext_name = f"_rccl_C{suffix}"
    out_so = ROOT / "mpi4jax_amd" / f"{ext_name}.so"
    build_dir = BUILD / variant if variant else BUILD
    build_dir.mkdir(parents=True, exist_ok=True)

    common = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1",
        "-DHIPBLAS_V2",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        f"-DTORCH_EXTENSION_NAME={ext_name}",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-fvisibility=hidden",
        "-Wno-unused-result",
    ]
    inc_flags = [f"-I{p}" for p in includes] + [f"-I{py_inc}",
                                               f"-I{CSRC}"]

    objs = []
    for src in ["kernels.hip", "shallow_water.hip", "bridge.cpp"]:
        sp = CSRC / src
        op = build_dir / (src.replace(".", "_") + ".o")
        objs.append(op)
        deps = [sp, CSRC / "kernels.h"]
        variant_flags = []
        if variant == "nofma" and src.endswith(".hip"):
            variant_flags = ["-ffp-contract=off"]
        if force or _newer(op, deps):
            extra = ["-x", "hip"] if src.endswith(".cpp") else []
            # bridge.cpp is host-only but compiled as hip for runtime hdrs
            _run([HIPCC, "-c", *common, *variant_flags, *inc_flags, *extra,
                  sp, "-o", op])

    if force or _newer(out_so, objs):
        link = [
            HIPCC,
            "-shared",
            "-fPIC",
            *objs,
            "-o",
            out_so,
        ]
        for lp in libpaths:
            link += [f"-L{lp}", f"-Wl,-rpath,{lp}"]
        link += [
            "-ltorch",
            "-ltorch_cpu",
            "-ltorch_hip",
            "-ltorch_python",
            "-lc10",
            "-lc10_hip",
            "-lrccl",
            "-lrocprofiler-sdk-roctx",
            "-lamdhip64",
        ]
        _run(link)
    print(f"built {out_so}")
    return out_so


if __name__ == "__main__":
    if "build_ext" in sys.argv or len(sys.argv) == 1:
        build_native(force="--force" in sys.argv)
        if "--nofma" in sys.argv or os.environ.get(
                "MPI4JAX_AMD_BUILD_NOFMA") == "1":
            build_native(force="--force" in sys.argv, variant="nofma")
    else:
        print(__doc__)
