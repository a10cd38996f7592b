"""In-tree build of the gfx950 HIP extension.

Drives hipcc directly (no hipify, no CUDA shims — the sources are native
HIP/CDNA4) and links against the installed PyTorch-ROCm. The resulting
ops/_C/vilbert_hip.so is committed-adjacent (git-ignored) and travels with
the repo snapshot to GPU boxes.

Usage: python -m vilbert_multi_task_amd.ops.build [--force]
"""

from __future__ import annotations

import os
import subprocess
import sys

import torch
import torch.utils.cpp_extension as ce

HERE = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(HERE, "csrc")
OUT_DIR = os.path.join(HERE, "_C")
OUT_SO = os.path.join(OUT_DIR, "vilbert_hip.so")

SOURCES = ["elementwise.hip", "attention.hip", "gemm_mfma.hip", "optim.hip", "attn_bwd.hip", "nms.hip", "roialign.hip", "linear_gelu.hip", "fp8_quant.hip",
    "train_bwd.hip", "bindings.cpp"]
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _newest_src_mtime() -> float:
    return max(
        os.path.getmtime(os.path.join(CSRC, s)) for s in SOURCES + ["common.h"]
    )


def needs_build() -> bool:
    if not os.path.exists(OUT_SO):
        return True
    return os.path.getmtime(OUT_SO) < _newest_src_mtime()


def build(force: bool = False, verbose: bool = True, asan: bool = False,
          ubsan: bool = False) -> str:
    if not force and not needs_build():
        return OUT_SO
    os.makedirs(OUT_DIR, exist_ok=True)
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    torch_inc = ce.include_paths()
    torch_lib = ce.library_paths()[0]
    abi = int(torch.compiled_with_cxx11_abi())
    cmd = [
        hipcc,
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fno-gpu-rdc",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DUSE_ROCM",
        "-D__HIP_PLATFORM_AMD__",
        "-DTORCH_EXTENSION_NAME=vilbert_hip",
    ]
    if asan:
        # host-side AddressSanitizer (scripts/sanitize.sh); device code is
        # not instrumented — -fsanitize applies to the x86 host pass only
        cmd += ["-fsanitize=address", "-shared-libsan", "-g1"]
    if ubsan:
        # UndefinedBehaviorSanitizer, host side: no allocator interception,
        # coexists with the HIP runtime (ASan's interceptors do not)
        cmd += ["-fsanitize=undefined", "-fno-sanitize=vptr,function",
                "-fno-sanitize-recover=undefined", "-shared-libsan", "-g1"]
    for inc in torch_inc:
        cmd.append(f"-I{inc}")
    cmd += [os.path.join(CSRC, s) for s in SOURCES]
    cmd += [
        f"-L{torch_lib}",
        f"-Wl,-rpath,{torch_lib}",
        "-ltorch",
        "-ltorch_hip",
        "-ltorch_cpu",
        "-lc10",
        "-lc10_hip",
        "-lamdhip64",
        "-lhipblaslt",
        "-o",
        OUT_SO,
    ]
    if verbose:
        print("[vilbert_amd.ops.build]", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True)
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv, asan="--asan" in sys.argv,
          ubsan="--ubsan" in sys.argv)
    print(f"built {OUT_SO}")
