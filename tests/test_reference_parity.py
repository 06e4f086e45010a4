"""Greedy-trajectory parity against the reference C++ runtime (built from
/root/reference). See tools/reference_parity.py and
profiles/reference_parity.md."""

import os
import shutil

import pytest


@pytest.mark.timeout(600)
@pytest.mark.skipif(not os.path.isdir("/root/reference")
                    or shutil.which("make") is None,
                    reason="reference sources or make unavailable")
def test_greedy_parity_with_reference(tmp_path_factory):
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "tools"))
    import reference_parity as rp
    # persistent workdir so the reference binary build is cached across runs
    workdir = "/tmp/dllama_parity_cache"
    os.makedirs(workdir, exist_ok=True)
    try:
        binary = rp.build_reference(workdir)
    except Exception as e:  # noqa: BLE001  (environment-dependent toolchain)
        pytest.skip(f"could not build the reference binary: {e}")
    for arch in ("llama", "qwen3", "qwen3_moe"):
        model, tok = rp.make_ascii_assets(workdir, arch=arch)
        ref = rp.run_reference(binary, model, tok, "hello world, this is", 48)
        ours = rp.run_ours(model, tok, "hello world, this is", 48)
        assert ref == ours, f"{arch}: {ref!r} != {ours!r}"
