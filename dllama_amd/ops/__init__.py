"""Op layer: hand-written gfx950 HIP kernels with torch fp32 references.

`reference` implements every op in plain PyTorch fp32 — the numerics oracle
(reference CPU kernels: src/nn/nn-cpu-ops.cpp) and the CPU backend.
`hip` loads the in-tree compiled HIP extension; on a GPU box the HIP path is
mandatory — ops fail loudly if the extension is missing.
"""

from . import reference  # noqa: F401

_hip = None
_hip_err = None


def hip_ops():
    """The compiled HIP extension module (raises if unavailable on GPU)."""
    global _hip, _hip_err
    if _hip is None and _hip_err is None:
        try:
            from .build import load_extension
            _hip = load_extension()
        except Exception as e:  # noqa: BLE001
            _hip_err = e
    if _hip is None:
        raise RuntimeError(
            f"dllama_amd HIP extension not available: {_hip_err}. "
            "Build it with `python -m dllama_amd.ops.build` (or __graft_entry__.build()).")
    return _hip


def hip_available() -> bool:
    try:
        hip_ops()
        return True
    except RuntimeError:
        return False
