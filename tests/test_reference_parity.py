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
        # per-arch vocab: see the comment in reference_parity.main()
        model, tok = rp.make_ascii_assets(
            workdir, vocab_size=101 if arch == "llama" else 128, arch=arch)
        ref = rp.run_reference(binary, model, tok, "hello world, this is", 48)
        ours = rp.run_ours(model, tok, "hello world, this is", 48)
        assert ref == ours, f"{arch}: {ref!r} != {ours!r}"

    # multi-turn chat parity (template generation, KV continuity across
    # turns, EOS detection) against the reference's interactive loop
    model, tok = rp.make_ascii_assets(workdir, arch="llama")
    sys_prompt, users = "keep it short", ["hello ab", "more cd"]
    ref_turns = rp.run_reference_chat(binary, model, tok, sys_prompt, users)
    our_turns = rp.run_ours_chat(model, tok, sys_prompt, users)
    assert len(ref_turns) == 2, ref_turns
    assert ref_turns == our_turns, (ref_turns, our_turns)
