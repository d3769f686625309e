"""In-tree hipcc build for the simumax_hip extension (gfx950 only).

Drives hipcc directly — no torch hipify pass, no CUDA compatibility
layer: the .hip sources are native CDNA4 code. The built .so lands next
to this file so it travels to the GPU box with the repo snapshot.
"""

import os
import subprocess
import sys
import sysconfig

import torch
import torch.utils.cpp_extension as cpp_ext

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT = os.path.join(HERE, "simumax_hip.so")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNEL_SOURCES = ["rmsnorm.hip", "rope.hip", "swiglu.hip", "cross_entropy.hip", "fp8_cast.hip",
                  "attention.hip", "mfma_probe.hip", "gemm_bench.hip",
                  "grouped_gemm.hip"]


def _newest_mtime(paths):
    return max(os.path.getmtime(p) for p in paths if os.path.exists(p))


def build(verbose: bool = True, force: bool = False) -> str:
    sources = [os.path.join(CSRC, s) for s in KERNEL_SOURCES
               if os.path.exists(os.path.join(CSRC, s))]
    binding = os.path.join(CSRC, "binding.cpp")
    sources.append(os.path.join(CSRC, "wgrad.cpp"))
    all_src = sources + [binding, os.path.join(CSRC, "common.h")]
    if (not force and os.path.exists(OUT)
            and os.path.getmtime(OUT) > _newest_mtime(all_src)):
        return OUT

    inc = [f"-I{p}" for p in cpp_ext.include_paths()]
    inc.append(f"-I{sysconfig.get_paths()['include']}")
    torch_lib = cpp_ext.library_paths()[0]
    abi = int(torch._C._GLIBCXX_USE_CXX11_ABI)

    common = [
        f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1", "-DHIPBLAS_V2",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        "-DTORCH_EXTENSION_NAME=simumax_hip",
    ]

    objs = []
    for src in sources + [binding]:
        obj = os.path.join(HERE, os.path.basename(src) + ".o")
        if (not force and os.path.exists(obj)
                and os.path.getmtime(obj) > max(os.path.getmtime(src),
                                                os.path.getmtime(os.path.join(CSRC, "common.h")))):
            objs.append(obj)
            continue
        cmd = ["hipcc", "-c", src, "-o", obj] + common
        if src.endswith(".cpp"):
            cmd += inc
        if verbose:
            print("[simumax_hip]", " ".join(cmd[:4]), "...")
        subprocess.run(cmd, check=True)
        objs.append(obj)

    link = ["hipcc", "-shared", "-o", OUT] + objs + [
        f"-L{torch_lib}", "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10",
        "-lc10_hip", "-ltorch_python", "-lhipblas", f"-Wl,-rpath,{torch_lib}",
    ]
    if verbose:
        print("[simumax_hip] linking", OUT)
    subprocess.run(link, check=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
