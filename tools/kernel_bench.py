"""Per-kernel microbench on Llama-3.1-8B decode shapes (isolated graphs)."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from dllama_amd.ops import hip_ops

k = hip_ops()
dev = "cuda"
g = torch.Generator(device=dev).manual_seed(0)

def bench(fn, n=256, reps=10):
    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3): fn()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    gr = torch.cuda.CUDAGraph()
    with torch.cuda.graph(gr):
        for _ in range(n): fn()
    for _ in range(3): gr.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps): gr.replay()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps / n * 1e6

def mklin(d, n):
    qs = torch.randint(0, 256, (d, n // 2), dtype=torch.uint8, device=dev, generator=g)
    sc = (torch.rand((d, n // 32), device=dev, generator=g) * 0.01).to(torch.float16)
    return qs, sc

def mkx(nb, n):
    q = torch.randint(-100, 100, (nb, n), dtype=torch.int8, device=dev, generator=g)
    s = torch.rand(nb, n // 32, device=dev, generator=g) * 0.01
    bs = torch.rand(nb, n // 32, device=dev, generator=g)
    return q, s, bs

NB = 1
dim, ffd, qkvd = 4096, 14336, 6144
x = torch.randn(32, dim, device=dev, generator=g)
w = torch.rand(dim, device=dev, generator=g)
ssq = torch.rand(32, 16, device=dev, generator=g) * 100

for d, n, tag in ((4096, 4096, "wo  4096x4096"), (4096, 14336, "w2  4096x14336"),
                  (6144, 4096, "qkv 6144x4096"), (28672, 4096, "w13 28672x4096"),
                  (128256, 4096, "cls 128256x4096")):
    qs, sc = mklin(d, n)
    xq, xs, xbs = mkx(NB, n)
    y = torch.zeros(NB, d, device=dev)
    t_plain = bench(lambda: k.q40_gemv(qs, sc, xq, xs, xbs, y, NB), n=64)
    ideal = (d * n * 0.5625) / 6.3e3 / 1000  # us at 6.3 TB/s
    extra = ""
    if d == dim:
        xr = torch.randn(NB, d, device=dev, generator=g)
        t_resid = bench(lambda: k.q40_gemv_resid(qs, sc, xq, xs, xbs, xr, ssq[0], NB), n=64)
        extra = f" resid={t_resid:6.2f}us"
    print(f"{tag:16s} plain={t_plain:7.2f}us ideal={ideal:6.2f}us eff={100*ideal/t_plain:4.0f}%{extra}")

# norm / small kernels at decode shapes
q8 = torch.zeros(1, dim, dtype=torch.int8, device=dev)
s8 = torch.zeros(1, dim // 32, device=dev)
bs8 = torch.zeros(1, dim // 32, device=dev)
print("norm_quant  [1,4096]: %6.2f us" % bench(lambda: k.norm_quant(x[:1], w, ssq, q8, s8, bs8, 1, 1e-5)))
ff = torch.randn(1, 2 * ffd, device=dev, generator=g)
dq = torch.zeros(1, ffd, dtype=torch.int8, device=dev)
ds = torch.zeros(1, ffd // 32, device=dev)
dbs = torch.zeros(1, ffd // 32, device=dev)
print("swiglu_q80  [1,14336]: %6.2f us" % bench(lambda: k.swiglu_q80(ff, ff[:, ffd:], 2 * ffd, ffd, 1, dq, ds, dbs)))

# attention decode at pos=1024, 8B shapes
H0, hd, kvd, seq = 32, 128, 1024, 4096
kc = torch.randn(seq, kvd, device=dev, generator=g)
vc = torch.randn(seq, kvd, device=dev, generator=g)
qv = torch.randn(1, H0 * hd, device=dev, generator=g)
z = torch.zeros(1, H0 * hd, device=dev)
pos = torch.tensor([1024], dtype=torch.int32, device=dev)
S = 32
ml = torch.zeros(1 * H0 * S * 2, device=dev)
osc = torch.zeros(1 * H0 * S * hd, device=dev)
cnt = torch.zeros(1 * H0, dtype=torch.int32, device=dev)
zq3 = mkx(1, H0 * hd)
print("attn(pos=1024) fused+quant: %6.2f us" %
      bench(lambda: k.attn(qv, H0 * hd, kc, vc, z, pos, 1, H0, 4, hd, S, ml, osc,
                           cnt, zq3[0], zq3[1], zq3[2])))
pos2 = torch.tensor([64], dtype=torch.int32, device=dev)
print("attn(pos=64)  fused+quant: %6.2f us" %
      bench(lambda: k.attn(qv, H0 * hd, kc, vc, z, pos2, 1, H0, 4, hd, S, ml, osc,
                           cnt, zq3[0], zq3[1], zq3[2])))
xr = torch.randn(1, dim, device=dev, generator=g)
pr = torch.randn(1, dim, device=dev, generator=g)
print("add_ssq [1,4096]: %6.2f us" % bench(lambda: k.add_ssq(xr, pr, ssq[0], 1)))

# PRO (fused norm+quant prologue) vs separate norm_quant + gemv, NB=1
for d, n, tag in ((6144, 4096, "qkv"), (28672, 4096, "w13"), (128256, 4096, "cls")):
    qs, sc = mklin(d, n)
    xq, xs, xbs = mkx(1, n)
    y = torch.zeros(1, d, device=dev)
    q8b = torch.zeros(1, n, dtype=torch.int8, device=dev)
    s8b = torch.zeros(1, n // 32, device=dev)
    bs8b = torch.zeros(1, n // 32, device=dev)
    wn = torch.rand(n, device=dev, generator=g)
    xf = torch.randn(1, n, device=dev, generator=g)
    sq = torch.rand(1, 16 * 32, device=dev, generator=g)
    def sep():
        k.norm_quant(xf, wn, sq, q8b, s8b, bs8b, 1, 1e-5)
        k.q40_gemv(qs, sc, q8b, s8b, bs8b, y, 1)
    t_sep = bench(sep, n=64)
    t_pro = bench(lambda: k.q40_gemv_nq(qs, sc, xf, wn, sq, 1e-5, y, 1), n=64)
    print(f"PRO {tag:4s}: norm_quant+gemv={t_sep:6.2f}us  fused-nq={t_pro:6.2f}us")

# prefill GEMM v1 vs v2 (round-2 DLLAMA_GEMM_V2 candidate), batch 32
print("\n-- GEMM v1 vs v2 (batch 32) --")
part = torch.zeros(16 * 32 * 28672, device=dev)
for d, n, tag in ((6144, 4096, "qkv"), (4096, 14336, "w2"), (28672, 4096, "w13")):
    qs, sc = mklin(d, n)
    xq, xs, _ = mkx(32, n)
    y = torch.zeros(32, d, device=dev)
    t1 = bench(lambda: k.q40_gemm(qs, sc, xq, xs, y, 32, part, variant=0), n=32)
    t2 = bench(lambda: k.q40_gemm(qs, sc, xq, xs, y, 32, part, variant=1), n=32)
    ideal = (d * n * 0.5625) / 6.3e3 / 1000
    print(f"gemm {tag:4s} v1={t1:7.2f}us v2={t2:7.2f}us ideal={ideal:6.2f}us "
          f"v2-eff={100*ideal/t2:4.0f}%")

# grouped GEMV v1 vs v2 at Qwen3-30B-A3B decode shapes
print("\n-- grouped GEMV v1 vs v2 (qwen3-30b shapes, 8 slots) --")
for d, n, ks, tag in ((1536, 2048, 8, "w13"), (2048, 768, 1, "w2")):
    E = 16
    qs = torch.randint(0, 256, (E, d, n // 2), dtype=torch.uint8, device=dev, generator=g)
    sc = (torch.rand((E, d, n // 32), device=dev, generator=g) * 0.01).to(torch.float16)
    nx = 8 // ks
    xq, xs, xbs = mkx(max(1, nx), n)
    idx = torch.arange(8, dtype=torch.int32, device=dev) % E
    y = torch.zeros(8, d, device=dev)
    t1 = bench(lambda: k.q40_gemv_grouped(qs, sc, xq, xs, xbs, idx, y, ks, variant=0), n=64)
    t2 = bench(lambda: k.q40_gemv_grouped(qs, sc, xq, xs, xbs, idx, y, ks, variant=1), n=64)
    ideal = (8 * d * n * 0.5625) / 6.3e3 / 1000
    print(f"grouped {tag:4s} v1={t1:7.2f}us v2={t2:7.2f}us ideal={ideal:6.2f}us "
          f"v2-eff={100*ideal/t2:4.0f}%")
