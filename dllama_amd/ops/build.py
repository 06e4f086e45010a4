"""In-tree build of the HIP extension for gfx950.

The .so lands in dllama_amd/ops/_build/ (inside the repo) so it travels to
the GPU box with the snapshot — a JIT cache under ~/.cache would not.
Run `python -m dllama_amd.ops.build` or __graft_entry__.build().
"""

from __future__ import annotations

import os

_EXT_NAME = "dllama_hip"


def _paths():
    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "csrc", "dllama_kernels.hip")
    build_dir = os.path.join(here, "_build")
    return src, build_dir


def load_extension(verbose: bool = False):
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load
    src, build_dir = _paths()
    os.makedirs(build_dir, exist_ok=True)
    # -mavx2/-mfma vectorize the HOST-side CPU Q40 matmul (hipcc applies
    # x86 flags to the host pass only); AVX2 is safe on every EPYC this
    # project can land on
    return load(name=_EXT_NAME, sources=[src], build_directory=build_dir,
                extra_cuda_cflags=["-O3", "-std=c++17", "-mavx2", "-mfma",
                                   "-mf16c", "-fopenmp"],
                extra_ldflags=["-L/opt/rocm/lib/llvm/lib", "-lomp",
                               "-Wl,-rpath,/opt/rocm/lib/llvm/lib"],
                verbose=verbose, with_cuda=True)


if __name__ == "__main__":
    load_extension(verbose=True)
    print("dllama_hip extension built OK")
