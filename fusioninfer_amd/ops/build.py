"""In-tree build of the CDNA4 HIP extension (gfx950 only, no hipify).

Drives hipcc directly — no CUDA compatibility layer, no JIT cache: the
resulting ``_C.so`` lives next to this file so the gpurun snapshot carries it.

Usage:  python -m fusioninfer_amd.ops.build
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig

OPS_DIR = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(OPS_DIR, "csrc")
SO_PATH = os.path.join(OPS_DIR, "_C.so")

SOURCES = [
    "ops.hip",
    "norm.hip",
    "activation.hip",
    "quant.hip",
    "rope.hip",
    "cache.hip",
    "paged_attention.hip",
    "prefill_attention.hip",
    "moe_gemm.hip",
]


def build(verbose: bool = True, force: bool = False) -> str:
    import torch
    from torch.utils import cpp_extension as ce

    srcs = [os.path.join(CSRC, s) for s in SOURCES]
    hdrs = [os.path.join(CSRC, "common.h")]
    if not force and os.path.exists(SO_PATH):
        so_mtime = os.path.getmtime(SO_PATH)
        if all(os.path.getmtime(f) < so_mtime for f in srcs + hdrs):
            return SO_PATH

    hipcc = os.path.join(os.environ.get("ROCM_PATH", "/opt/rocm"), "bin", "hipcc")
    torch_lib = ce.library_paths()[0]
    cmd = [
        hipcc,
        "--offload-arch=gfx950",
        "-O3",
        "-std=c++17",
        "-shared",
        "-fPIC",
        "-DTORCH_EXTENSION_NAME=_C",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={int(torch._C._GLIBCXX_USE_CXX11_ABI)}",
        *getattr(ce, "COMMON_HIP_FLAGS", ["-D__HIP_PLATFORM_AMD__=1", "-DUSE_ROCM=1"]),
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS",
        *[f"-I{p}" for p in ce.include_paths()],
        f"-I{sysconfig.get_paths()['include']}",
        f"-I{CSRC}",
        *srcs,
        f"-L{torch_lib}",
        f"-Wl,-rpath,{torch_lib}",
        "-ltorch",
        "-ltorch_python",
        "-lc10",
        "-ltorch_hip",
        "-lc10_hip",
        "-lamdhip64",
        "-o",
        SO_PATH,
    ]
    if verbose:
        print("[fusioninfer build]", " ".join(cmd), file=sys.stderr)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
