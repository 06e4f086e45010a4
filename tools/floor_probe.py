import torch, time, sys
sys.path.insert(0, "/root/repo")
from dllama_amd.ops import hip_ops
k = hip_ops()
dev = "cuda"

def graph_of(fn, n):
    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3): fn()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        for _ in range(n): fn()
    torch.cuda.synchronize()
    for _ in range(3): g.replay()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20): g.replay()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/20/n*1e6

pos = torch.zeros(1, dtype=torch.int32, device=dev)
print("pos_inc (1 wg, dependent chain):      %.2f us/kernel" % graph_of(lambda: k.pos_inc(pos, 1), 512))

x = torch.randn(1, 4096, device=dev); y = torch.randn(1, 4096, device=dev)
print("add_ [1,4096] (16 wg, dep):           %.2f us/kernel" % graph_of(lambda: k.add_(x, y), 512))

w = torch.rand(4096, device=dev)
q = torch.zeros(1, 4096, dtype=torch.int8, device=dev)
s_ = torch.zeros(1, 128, device=dev); bs = torch.zeros(1, 128, device=dev)
print("add_rmsnorm_q80 [1,4096] (1 wg, dep): %.2f us/kernel" % graph_of(lambda: k.add_rmsnorm_q80(x, y, w, q, s_, bs, 1e-5), 256))

# independent big-ish kernels: does a wider kernel hide the floor?
xs = [torch.randn(64, 4096, device=dev) for _ in range(2)]
print("add_ [64,4096] (1024 wg, dep):        %.2f us/kernel" % graph_of(lambda: k.add_(xs[0], xs[1]), 256))

# q80_quantize small
print("q80_quantize [1,4096] (16wg):         %.2f us/kernel" % graph_of(lambda: k.q80_quantize(x, q, s_, bs), 256))
