#!/usr/bin/env python3
"""Build the in-tree native extension (agentcontrolplane_amd/_C.so).

hipcc cross-compiles gfx950 without a GPU, so this runs in CPU-only CI too.
The .so lands inside the package (NOT a JIT cache dir) so the gpurun
snapshot carries it to the GPU box.
"""
import os
import shutil
import sys

os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "csrc")
PKG = os.path.join(ROOT, "agentcontrolplane_amd")
BUILD = os.path.join(ROOT, "build", "ext")


def build(verbose: bool = False) -> str:
    from torch.utils import cpp_extension

    os.makedirs(BUILD, exist_ok=True)
    sources = [
        os.path.join(CSRC, "bindings.cpp"),
        os.path.join(CSRC, "block_manager.cpp"),
        os.path.join(CSRC, "elementwise.hip"),
        os.path.join(CSRC, "rope_cache.hip"),
        os.path.join(CSRC, "decode_attn.hip"),
        os.path.join(CSRC, "prefill_attn.hip"),
        os.path.join(CSRC, "prefill_attn_mfma.hip"),
        os.path.join(CSRC, "mfma_probe.hip"),
        os.path.join(CSRC, "gemm_bf16.hip"),
        os.path.join(CSRC, "sampling.hip"),
        os.path.join(CSRC, "sampling_fullvocab.hip"),
    ]
    module = cpp_extension.load(
        name="_C",
        sources=sources,
        build_directory=BUILD,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950", "-std=c++17"],
        verbose=verbose,
        is_python_module=False,
        is_standalone=False,
        keep_intermediates=True,
    )
    # cpp_extension.load with is_python_module=False loads into the process;
    # we want the artifact path:
    so_path = os.path.join(BUILD, "_C.so")
    if not os.path.exists(so_path):
        raise RuntimeError(f"build produced no _C.so in {BUILD}")
    dest = os.path.join(PKG, "_C.so")
    shutil.copy2(so_path, dest)
    print(f"built {dest}")
    return dest


if __name__ == "__main__":
    build(verbose="-v" in sys.argv)
