#!/usr/bin/env python3
"""Build the npf gfx950 HIP extension IN-TREE (npf/_hip_C.so).

hipcc cross-compiles for gfx950 without a GPU; the built .so travels with the
repo snapshot to the GPU box.  Usage: `python csrc/build.py`.
"""

import os
import shutil
import sys

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def build():
    # ALWAYS build clean: a stale ninja log (e.g. restored from a checkpoint
    # without its object files) or a snapshot taken mid-rebuild can produce a
    # .so that loads but memory-faults on the GPU — observed 2026-09-13.
    shutil.rmtree(os.path.join(REPO, "csrc", "_build"), ignore_errors=True)
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", str(min(os.cpu_count() or 4, 16)))

    from torch.utils.cpp_extension import BuildExtension, CUDAExtension

    sources = [
        os.path.join(REPO, "csrc", "npf_hip", "ext.cpp"),
        os.path.join(REPO, "csrc", "npf_hip", "attn.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "attn_mfma.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "setconv.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "gauss_ll.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "convblock.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "convblock2d.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "griddensity.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "gauss_kl.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "mlp_chain.hip"),
        os.path.join(REPO, "csrc", "npf_hip", "attender.hip"),
    ]
    ext = CUDAExtension(
        name="npf._hip_C",
        sources=sources,
        extra_compile_args={
            "cxx": ["-O3"],
            "nvcc": ["-O3", "--offload-arch=gfx950"],
        },
    )

    from setuptools import Distribution

    dist = Distribution(
        {"name": "npf_hip", "ext_modules": [ext]}
    )
    dist.script_name = "build.py"
    cmd = BuildExtension.with_options(no_python_abi_suffix=False)(dist)
    cmd.ensure_finalized()
    cmd.build_lib = os.path.join(REPO, "csrc", "_build", "lib")
    cmd.build_temp = os.path.join(REPO, "csrc", "_build", "tmp")
    cmd.run()

    # copy the built .so into the package (in-tree, ships with snapshots)
    built = None
    for root, _, files in os.walk(cmd.build_lib):
        for f in files:
            if f.startswith("_hip_C") and f.endswith(".so"):
                built = os.path.join(root, f)
    assert built, "extension build produced no .so"
    dest = os.path.join(REPO, "npf", os.path.basename(built))
    shutil.copy2(built, dest)
    print(f"built {dest}")
    return dest


if __name__ == "__main__":
    sys.exit(0 if build() else 1)
