"""End-to-end HIP transformer vs the CPU oracle (TP=1), decode==prefill,
and hipGraph replay == eager (reference validates Vulkan against CPU the
same way, nn-vulkan-test.cpp)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from dllama_amd import model_file as mf
from dllama_amd.models.config import ModelConfig
from dllama_amd.models.cpu_model import CpuTransformer
from dllama_amd.utils.testing import make_tiny_llama, make_tiny_qwen3


@pytest.fixture(scope="module")
def tiny_path(tmp_path_factory):
    p = str(tmp_path_factory.mktemp("m") / "tiny.m")
    make_tiny_llama(p, vocab_size=256)
    return p


def _rel_err(a, b):
    return (a - b).abs().max().item() / (b.abs().max().item() + 1e-9)


def test_hip_matches_cpu(tiny_path):
    from dllama_amd.models.hip_model import HipTransformer
    m = mf.ModelFile(tiny_path)
    cfg = ModelConfig.from_header(m.header)
    cpu = CpuTransformer(m, cfg)
    hip = HipTransformer.from_file(m, cfg)
    tokens = torch.tensor([3, 17, 101, 42])
    positions = torch.arange(4)
    want = cpu.forward(tokens, positions)
    got = hip.forward(tokens, positions).cpu()
    assert _rel_err(got, want) < 0.02, _rel_err(got, want)
    # argmax agreement on every row
    assert torch.equal(got.argmax(-1), want.argmax(-1))


def test_hip_decode_equals_prefill(tiny_path):
    from dllama_amd.models.hip_model import HipTransformer
    m = mf.ModelFile(tiny_path)
    cfg = ModelConfig.from_header(m.header)
    hip = HipTransformer.from_file(m, cfg)
    tokens = [5, 9, 33, 77, 120]
    batch = hip.forward(torch.tensor(tokens), torch.arange(5)).cpu().clone()
    hip2 = HipTransformer.from_file(m, cfg)
    for i, t in enumerate(tokens):
        one = hip2.forward(torch.tensor([t]), torch.tensor([i])).cpu()
    assert _rel_err(one[0], batch[-1]) < 0.01


def test_hip_graph_matches_eager(tiny_path):
    from dllama_amd.models.hip_model import HipTransformer
    m = mf.ModelFile(tiny_path)
    cfg = ModelConfig.from_header(m.header)
    eager = HipTransformer.from_file(m, cfg)
    graphed = HipTransformer.from_file(m, cfg)
    prompt = [1, 2, 3]
    want = eager.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    got = graphed.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    assert _rel_err(got, want) < 1e-5

    graphed.capture_decode_graph()
    for step in range(4):
        t = torch.tensor([10 + step])
        p = torch.tensor([3 + step])
        want = eager.forward(t, p).cpu().clone()
        got = graphed.forward(t, p).cpu().clone()
        assert _rel_err(got[0], want[0]) < 1e-5, f"step {step}"


def test_hip_qwen3_and_moe(tmp_path):
    from dllama_amd.models.hip_model import HipTransformer
    for moe in (False, True):
        p = str(tmp_path / f"q{int(moe)}.m")
        make_tiny_qwen3(p, moe=moe)
        m = mf.ModelFile(p)
        cfg = ModelConfig.from_header(m.header)
        cpu = CpuTransformer(m, cfg)
        hip = HipTransformer.from_file(m, cfg)
        tokens = torch.tensor([1, 2, 3])
        want = cpu.forward(tokens, torch.arange(3))
        got = hip.forward(tokens, torch.arange(3)).cpu()
        assert _rel_err(got, want) < 0.03, (moe, _rel_err(got, want))


def test_native_extension_is_loaded():
    """Guard against silent eager/PyTorch fallbacks: the in-tree .so must be
    what provides the ops."""
    from dllama_amd.ops import hip_ops
    k = hip_ops()
    assert "dllama_amd/ops/_build" in k.__file__, k.__file__


def test_hip_prefill_gemm_path(tiny_path):
    """Prefill batches >=8 run the int8-MFMA GEMM; decode==prefill must
    still hold across the GEMV/GEMM boundary."""
    from dllama_amd.models.hip_model import HipTransformer
    m = mf.ModelFile(tiny_path)
    cfg = ModelConfig.from_header(m.header)
    hip = HipTransformer.from_file(m, cfg)
    tokens = list(range(3, 23))  # B=20 -> padded 32 -> GEMM path
    batch = hip.forward(torch.tensor(tokens), torch.arange(len(tokens))).cpu().clone()
    hip2 = HipTransformer.from_file(m, cfg)
    for i, t in enumerate(tokens):
        one = hip2.forward(torch.tensor([t]), torch.tensor([i])).cpu()
    assert _rel_err(one[0], batch[-1]) < 0.02
    cpu = CpuTransformer(m, cfg)
    want = cpu.forward(torch.tensor(tokens), torch.arange(len(tokens)))
    assert _rel_err(batch, want) < 0.02


def test_adaptive_splits_recapture(tiny_path):
    """Crossing the adaptive threshold re-captures the decode graph with
    S=16; logits must keep matching eager decode at S=8."""
    from dllama_amd.models.hip_model import HipTransformer
    m = mf.ModelFile(tiny_path)
    cfg = ModelConfig.from_header(m.header)
    eager = HipTransformer.from_file(m, cfg)
    eager.adaptive_thresh = 0      # pin eager at the default S=8 pair
    adaptive = HipTransformer.from_file(m, cfg)
    adaptive.adaptive_thresh = 6
    prompt = [1, 2, 3]
    eager.forward(torch.tensor(prompt), torch.arange(3))
    adaptive.forward(torch.tensor(prompt), torch.arange(3))
    adaptive.capture_decode_graph()
    for step in range(8):  # crosses the threshold at pos 6
        t = torch.tensor([20 + step])
        p = torch.tensor([3 + step])
        want = eager.forward(t, p).cpu().clone()
        got = adaptive.forward(t, p).cpu().clone()
        assert _rel_err(got[0], want[0]) < 1e-4, f"step {step}"
    assert adaptive.attn_splits == 16


def test_tp_path_matches_plain_world1(tiny_path):
    """force_sync=True runs the FULL TP code path (Q80 sync pack -> gather ->
    merge-add, logits gather + concat kernel, argmax_token greedy) at
    world=1, where SingleComm collectives are identity — so its logits and
    greedy tokens must match the plain path. This validates every TP kernel
    and the graph-captured sync step on a 1-GPU box."""
    from dllama_amd.models.hip_model import HipTransformer
    m = mf.ModelFile(tiny_path)
    cfg = ModelConfig.from_header(m.header)
    plain = HipTransformer.from_file(m, cfg)
    tp = HipTransformer.from_file(m, cfg, force_sync=True)
    prompt = [3, 17, 101]
    want = plain.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    got = tp.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    # Q80 sync quantizes the partial sums, so allow quantization-level error
    assert _rel_err(got, want) < 0.02, _rel_err(got, want)
    assert torch.equal(got.argmax(-1), want.argmax(-1))
    # graph-captured TP decode (the sync + gather + argmax inside the graph)
    tp.greedy_feedback = True
    tp.capture_decode_graph()
    plain.greedy_feedback = True
    plain.capture_decode_graph()
    plain.pos.fill_(3)
    tp.pos.fill_(3)
    plain.tokens[0] = 7
    tp.tokens[0] = 7
    for step in range(4):
        plain._graph.replay()
        tp._graph.replay()
        assert int(plain.tokens[0]) == int(tp.tokens[0]), f"step {step}"


def test_tp_path_moe_world1(tmp_path):
    """force_sync TP path through the MoE FFN (scale_merge kernel into
    partial + Q80 sync) must match the fused world=1 scale_merge_add."""
    from dllama_amd.models.hip_model import HipTransformer
    p = str(tmp_path / "moe.m")
    make_tiny_qwen3(p, moe=True)
    m = mf.ModelFile(p)
    cfg = ModelConfig.from_header(m.header)
    plain = HipTransformer.from_file(m, cfg)
    tp = HipTransformer.from_file(m, cfg, force_sync=True)
    tokens = torch.tensor([1, 2, 3])
    want = plain.forward(tokens, torch.arange(3)).cpu().clone()
    got = tp.forward(tokens, torch.arange(3)).cpu().clone()
    assert _rel_err(got, want) < 0.02, _rel_err(got, want)
    assert torch.equal(got.argmax(-1), want.argmax(-1))


def test_deferred_quant_decode_matches_explicit(tiny_path):
    """The deferred-quant decode path (EPI_RESID_Q producers + PRO=2
    consumers, no norm_quant launches) must match the explicit-norm path:
    Q80 codes are scale-invariant, so only fp rounding differs."""
    from dllama_amd.models.hip_model import HipTransformer
    m = mf.ModelFile(tiny_path)
    cfg = ModelConfig.from_header(m.header)
    expl = HipTransformer.from_file(m, cfg)
    expl.use_deferred = False
    deferred = HipTransformer.from_file(m, cfg)
    prompt = [3, 17, 101]
    expl.forward(torch.tensor(prompt), torch.arange(3))
    deferred.forward(torch.tensor(prompt), torch.arange(3))
    for step in range(5):
        t = torch.tensor([20 + step])
        p = torch.tensor([3 + step])
        want = expl.forward(t, p).cpu().clone()
        got = deferred.forward(t, p).cpu().clone()
        assert _rel_err(got[0], want[0]) < 2e-3, (step, _rel_err(got[0], want[0]))
        assert torch.equal(got.argmax(-1), want.argmax(-1)), f"step {step}"


def test_deferred_quant_qwen3_dense(tmp_path):
    """Deferred path through the qwen3 qk-norm variant (non-fused-rope
    consumers with PRO=2) vs the CPU oracle."""
    from dllama_amd.models.hip_model import HipTransformer
    p = str(tmp_path / "q3.m")
    make_tiny_qwen3(p, moe=False)
    m = mf.ModelFile(p)
    cfg = ModelConfig.from_header(m.header)
    cpu = CpuTransformer(m, cfg)
    hip = HipTransformer.from_file(m, cfg)
    pr = [1, 2, 3]
    hip.forward(torch.tensor(pr), torch.arange(3))
    cpu.forward(torch.tensor(pr), torch.arange(3))
    for step in range(3):
        t = torch.tensor([9 + step])
        pp = torch.tensor([3 + step])
        want = cpu.forward(t, pp)
        got = hip.forward(t, pp).cpu()
        assert _rel_err(got[0], want[0]) < 0.03, f"step {step}"


def test_deferred_quant_moe(tmp_path):
    """MoE deferred-quant decode (fused router+norm, deferred grouped w13
    input, scale-merge quant emit) vs the explicit-norm path and the CPU
    oracle. Needs dim % 256 == 0 to activate."""
    from dllama_amd.models.hip_model import HipTransformer
    p = str(tmp_path / "moe256.m")
    make_tiny_qwen3(p, moe=True, dim=256)
    m = mf.ModelFile(p)
    cfg = ModelConfig.from_header(m.header)
    expl = HipTransformer.from_file(m, cfg)
    expl.use_deferred = False
    deferred = HipTransformer.from_file(m, cfg)
    cpu = CpuTransformer(m, cfg)
    prompt = [1, 2, 3]
    expl.forward(torch.tensor(prompt), torch.arange(3))
    deferred.forward(torch.tensor(prompt), torch.arange(3))
    cpu.forward(torch.tensor(prompt), torch.arange(3))
    for step in range(4):
        t = torch.tensor([9 + step])
        pp = torch.tensor([3 + step])
        want = expl.forward(t, pp).cpu().clone()
        got = deferred.forward(t, pp).cpu().clone()
        oracle = cpu.forward(t, pp)
        assert _rel_err(got[0], want[0]) < 2e-3, (step, _rel_err(got[0], want[0]))
        assert _rel_err(got[0], oracle[0]) < 0.03, f"step {step}"
        assert torch.equal(got.argmax(-1), want.argmax(-1)), f"step {step}"


def test_rccl_collectives_capture_in_graph(tmp_path):
    """Real RCCL (nccl backend) collectives inside a captured hipGraph on
    hardware: world=1 process group, force_sync TP path (all_gather of the
    Q80 wire + logits gather are genuine RCCL calls even at world 1).
    dim=256 activates the DEFERRED TP path (EPI_PACK wire emit +
    merge_add_q), the exact kernels the driver's 8-GPU decode replays.
    De-risks the 8-GPU graph-captured decode before it ever runs."""
    import os
    import subprocess
    import sys
    mpath = str(tmp_path / "l256r.m")
    make_tiny_llama(mpath, vocab_size=256, dim=256)
    code = f'''
import os, torch
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29581")
os.environ["WORLD_SIZE"] = "1"
os.environ["RANK"] = "0"
os.environ["LOCAL_RANK"] = "0"
import torch.distributed as dist
dist.init_process_group("nccl", rank=0, world_size=1)
from dllama_amd import model_file as mf
from dllama_amd.models.config import ModelConfig
from dllama_amd.models.hip_model import HipTransformer
from dllama_amd.parallel.comm import DistComm
m = mf.ModelFile({mpath!r})
cfg = ModelConfig.from_header(m.header)
plain = HipTransformer.from_file(m, cfg)
tp = HipTransformer.from_file(m, cfg, comm=DistComm(), force_sync=True)
prompt = [3, 17, 101]
import torch as T
want = plain.forward(T.tensor(prompt), T.arange(3)).cpu()
got = tp.forward(T.tensor(prompt), T.arange(3)).cpu()
assert T.equal(got.argmax(-1), want.argmax(-1))
tp.greedy_feedback = True
tp.capture_decode_graph()   # RCCL all-gathers captured inside the graph
plain.greedy_feedback = True
plain.capture_decode_graph()
plain.pos.fill_(3); tp.pos.fill_(3)
plain.tokens[0] = 7; tp.tokens[0] = 7
for step in range(4):
    plain._graph.replay()
    tp._graph.replay()
    T.cuda.synchronize()
    assert int(plain.tokens[0]) == int(tp.tokens[0]), step
dist.destroy_process_group()
print("RCCL_GRAPH_OK")
'''
    r = subprocess.run([sys.executable, "-c", code], capture_output=True,
                       text=True, timeout=300,
                       env={**os.environ, "HSA_ENABLE_IPC_MODE_LEGACY": "0"})
    assert "RCCL_GRAPH_OK" in r.stdout, (r.stdout[-2000:], r.stderr[-2000:])


def test_tp_deferred_path_world1(tmp_path):
    """TP deferred decode (EPI_PACK wire emit -> all-gather -> merge_add_q
    with deferred quant) vs the plain world=1 path, at a dim that activates
    it (dim % 256 == 0)."""
    from dllama_amd.models.hip_model import HipTransformer
    p = str(tmp_path / "l256.m")
    make_tiny_llama(p, vocab_size=256, dim=256)
    m = mf.ModelFile(p)
    cfg = ModelConfig.from_header(m.header)
    plain = HipTransformer.from_file(m, cfg)
    tp = HipTransformer.from_file(m, cfg, force_sync=True)
    prompt = [3, 17, 101]
    want = plain.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    got = tp.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    assert _rel_err(got, want) < 0.02, _rel_err(got, want)
    assert torch.equal(got.argmax(-1), want.argmax(-1))
    # graph-captured deferred TP decode with greedy feedback
    tp.greedy_feedback = True
    tp.capture_decode_graph()
    plain.greedy_feedback = True
    plain.capture_decode_graph()
    for mdl in (plain, tp):
        mdl.pos.fill_(3)
        mdl.tokens[0] = 7
    for step in range(5):
        plain._graph.replay()
        tp._graph.replay()
        assert int(plain.tokens[0]) == int(tp.tokens[0]), f"step {step}"


def test_tp_deferred_moe_world1(tmp_path):
    """TP deferred MoE decode (router+norm fused, scale_merge_pack wire
    emit, merge_add_q) vs the plain world=1 MoE path at dim=256."""
    from dllama_amd.models.hip_model import HipTransformer
    p = str(tmp_path / "m256.m")
    make_tiny_qwen3(p, moe=True, dim=256)
    m = mf.ModelFile(p)
    cfg = ModelConfig.from_header(m.header)
    plain = HipTransformer.from_file(m, cfg)
    tp = HipTransformer.from_file(m, cfg, force_sync=True)
    prompt = [1, 2, 3]
    plain.forward(torch.tensor(prompt), torch.arange(3))
    tp.forward(torch.tensor(prompt), torch.arange(3))
    for step in range(4):
        t = torch.tensor([9 + step])
        pp = torch.tensor([3 + step])
        want = plain.forward(t, pp).cpu().clone()
        got = tp.forward(t, pp).cpu().clone()
        assert _rel_err(got[0], want[0]) < 0.02, (step, _rel_err(got[0], want[0]))
        assert torch.equal(got.argmax(-1), want.argmax(-1)), f"step {step}"


def test_tp_path_f32_sync_world1(tiny_path):
    """force_sync with the f32 sync buffer (--buffer-float-type f32):
    all-reduce + add_ssq TP branch on GPU, vs the plain path."""
    from dllama_amd.models.hip_model import HipTransformer
    from dllama_amd.quants import F32
    m = mf.ModelFile(tiny_path, sync_type=F32)
    cfg = ModelConfig.from_header(m.header)
    cfg.sync_type = F32
    plain = HipTransformer.from_file(m, cfg)
    tp = HipTransformer.from_file(m, cfg, force_sync=True)
    prompt = [3, 17, 101]
    want = plain.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    got = tp.forward(torch.tensor(prompt), torch.arange(3)).cpu().clone()
    assert _rel_err(got, want) < 1e-4, _rel_err(got, want)
    assert torch.equal(got.argmax(-1), want.argmax(-1))


def test_experimental_paths_match_defaults(tmp_path, monkeypatch):
    """Env-gated experiment paths (read per model construction) must stay
    correct while not default: bf16 prefill shadow, K-split
    down-projections, fused FFN. (DLLAMA_GQA_ATTN and the kernel-level v1
    toggles are cached in C++ statics at first use, so they are validated
    by their dedicated kernel tests instead.) dim=256 so every variant's
    shape constraints are met."""
    from dllama_amd.models import hip_model as hm
    p = str(tmp_path / "l256e.m")
    make_tiny_llama(p, vocab_size=256, dim=256)
    m = mf.ModelFile(p)
    cfg = ModelConfig.from_header(m.header)
    ref = hm.HipTransformer.from_file(m, cfg)
    tokens = torch.tensor(list(range(3, 15)))  # 12 -> padded 32: prefill path
    want = ref.forward(tokens, torch.arange(12)).cpu().clone()
    dt = torch.tensor([5])
    dp = torch.tensor([12])
    want_d = ref.forward(dt, dp).cpu().clone()

    for env in ("DLLAMA_PREFILL_BF16", "DLLAMA_KSPLIT_RESID",
                "DLLAMA_FUSED_FFN"):
        monkeypatch.setenv(env, "2" if env == "DLLAMA_KSPLIT_RESID" else "1")
        alt = hm.HipTransformer.from_file(m, cfg)
        got = alt.forward(tokens, torch.arange(12)).cpu().clone()
        tol = 0.03 if env == "DLLAMA_PREFILL_BF16" else 2e-3
        assert _rel_err(got, want) < tol, (env, _rel_err(got, want))
        if env != "DLLAMA_PREFILL_BF16":  # bf16 flips near-ties on the
            # tiny random model (logit magnitudes ~1e-2); rel-err covers it
            assert torch.equal(got.argmax(-1), want.argmax(-1)), env
        got_d = alt.forward(dt, dp).cpu().clone()
        assert _rel_err(got_d[0], want_d[0]) < tol, (env, "decode")
        monkeypatch.delenv(env)

    # bf16 prefill through the qwen3 branch (qk-norm inside the chunk path)
    q3 = str(tmp_path / "q3e.m")
    make_tiny_qwen3(q3, moe=False, dim=256)
    mq = mf.ModelFile(q3)
    cq = ModelConfig.from_header(mq.header)
    refq = hm.HipTransformer.from_file(mq, cq)
    wantq = refq.forward(tokens, torch.arange(12)).cpu().clone()
    monkeypatch.setenv("DLLAMA_PREFILL_BF16", "1")
    altq = hm.HipTransformer.from_file(mq, cq)
    gotq = altq.forward(tokens, torch.arange(12)).cpu().clone()
    assert _rel_err(gotq, wantq) < 0.03, _rel_err(gotq, wantq)
    monkeypatch.delenv("DLLAMA_PREFILL_BF16")
