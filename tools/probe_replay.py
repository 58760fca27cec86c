"""Is torch.cuda.CUDAGraph.replay() asynchronous on ROCm? Measure the host
time of replay() calls for a graph with ~20 ms of GEMM work."""

import time

import torch

assert torch.cuda.is_available()
a = torch.randn(8192, 8192, dtype=torch.bfloat16, device="cuda")
b = torch.randn(8192, 8192, dtype=torch.bfloat16, device="cuda")

s = torch.cuda.Stream()
with torch.cuda.stream(s):
    c = a @ b
torch.cuda.current_stream().wait_stream(s)
torch.cuda.synchronize()

g = torch.cuda.CUDAGraph()
with torch.cuda.graph(g):
    c = a @ b
    for _ in range(15):
        c = a @ c

torch.cuda.synchronize()
t0 = time.perf_counter()
g.replay()
t1 = time.perf_counter()
torch.cuda.synchronize()
t2 = time.perf_counter()
print(f"first replay call: {1e3*(t1-t0):.2f} ms; sync wait: {1e3*(t2-t1):.2f} ms")

# back-to-back replays without sync: if the call blocks on the previous
# instance, the second call takes ~the kernel time
t0 = time.perf_counter()
g.replay()
t1 = time.perf_counter()
g.replay()
t2 = time.perf_counter()
torch.cuda.synchronize()
t3 = time.perf_counter()
print(f"replay#1 call {1e3*(t1-t0):.2f} ms, replay#2 call {1e3*(t2-t1):.2f} ms, "
      f"final sync {1e3*(t3-t2):.2f} ms")

# replay + host work + sync: can host work overlap a single in-flight replay?
t0 = time.perf_counter()
g.replay()
t1 = time.perf_counter()
x = 0
while time.perf_counter() - t1 < 0.010:
    x += 1  # 10 ms of host spin
t2 = time.perf_counter()
torch.cuda.synchronize()
t3 = time.perf_counter()
print(f"replay call {1e3*(t1-t0):.2f} ms, host spin {1e3*(t2-t1):.2f} ms, "
      f"remaining sync {1e3*(t3-t2):.2f} ms (should be kernel-time minus spin)")
