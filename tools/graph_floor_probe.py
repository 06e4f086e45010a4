"""In-graph small-kernel floor: a captured chain of N dependent norm_quant-
sized kernels vs the same kernels eager, to separate launch floor from
execution (explains the ~2-3 us in-graph premium every kernel pays)."""
import sys, time, torch
sys.path.insert(0, "/root/repo")
from dllama_amd.ops import hip_ops

k = hip_ops()
dev = "cuda"
n = 4096
x = torch.randn(1, n, device=dev)
w = torch.randn(n, device=dev).abs()
ssq = torch.zeros(1, 16 * 32, device=dev)
ssq[0, 0] = float(n)  # inv ~ 1
q = torch.zeros(1, n, dtype=torch.int8, device=dev)
s = torch.zeros(1, n // 32, device=dev)
bs = torch.zeros(1, n // 32, device=dev)

def chain(reps):
    for _ in range(reps):
        k.norm_quant(x, w, ssq, q, s, bs, 1, 1e-5)

# eager timing
for _ in range(3):
    chain(64)
torch.cuda.synchronize()
t0 = time.perf_counter(); chain(640); torch.cuda.synchronize()
eager = (time.perf_counter() - t0) / 640 * 1e6
print(f"eager chain: {eager:.2f} us/kernel")

# captured graph timing
g = torch.cuda.CUDAGraph()
st = torch.cuda.Stream(); st.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(st):
    chain(3)
torch.cuda.current_stream().wait_stream(st)
torch.cuda.synchronize()
with torch.cuda.graph(g):
    chain(256)
for _ in range(3):
    g.replay()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
    g.replay()
torch.cuda.synchronize()
ing = (time.perf_counter() - t0) / 20 / 256 * 1e6
print(f"in-graph chain: {ing:.2f} us/kernel")

# bigger-grid variant of the same op (more wgs per launch)
xb = torch.randn(8, n, device=dev)
qb = torch.zeros(8, n, dtype=torch.int8, device=dev)
sb = torch.zeros(8, n // 32, device=dev)
bsb = torch.zeros(8, n // 32, device=dev)
ssqb = torch.zeros(8, 16 * 32, device=dev); ssqb[:, 0] = float(n)
g2 = torch.cuda.CUDAGraph()
with torch.cuda.stream(st):
    for _ in range(3):
        k.norm_quant(xb, w, ssqb, qb, sb, bsb, 8, 1e-5)
torch.cuda.current_stream().wait_stream(st)
torch.cuda.synchronize()
with torch.cuda.graph(g2):
    for _ in range(256):
        k.norm_quant(xb, w, ssqb, qb, sb, bsb, 8, 1e-5)
g2.replay(); torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(20):
    g2.replay()
torch.cuda.synchronize()
print(f"in-graph chain (8-row grid): {(time.perf_counter()-t0)/20/256*1e6:.2f} us/kernel")
