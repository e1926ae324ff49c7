import torch, time
assert torch.cuda.is_available()
dev = 'cuda'
# small-grid workload (N=1024 GEMM ~ 64 workgroups) so two could co-run
a = torch.randn(1024, 1024, device=dev, dtype=torch.half)
b = torch.randn(1024, 1024, device=dev, dtype=torch.half)
c = torch.randn(1024, 1024, device=dev, dtype=torch.half)
d = torch.randn(1024, 1024, device=dev, dtype=torch.half)

sA, sB = torch.cuda.Stream(), torch.cuda.Stream()
def work(x, y, n=200):
    z = x
    for _ in range(n):
        z = z @ y
    return z
# warmup + capture two graphs on two streams
for s, (x, y) in ((sA, (a, b)), (sB, (c, d))):
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        work(x, y, 10)
    torch.cuda.current_stream().wait_stream(s)
g1, g2 = torch.cuda.CUDAGraph(), torch.cuda.CUDAGraph()
with torch.cuda.graph(g1, stream=sA): r1 = work(a, b)
with torch.cuda.graph(g2, stream=sB): r2 = work(c, d)

def bench(fn, n=20):
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / n * 1000

def serial():
    with torch.cuda.stream(sA): g1.replay()
    torch.cuda.current_stream().wait_stream(sA)
    with torch.cuda.stream(sA): g2.replay()

def parallel():
    with torch.cuda.stream(sA): g1.replay()
    with torch.cuda.stream(sB): g2.replay()

print("serial both-on-A ms:", round(bench(serial), 3))
print("parallel A+B ms:   ", round(bench(parallel), 3))

# also eager kernels on two streams (no graphs)
def eager_parallel():
    with torch.cuda.stream(sA): work(a, b, 50)
    with torch.cuda.stream(sB): work(c, d, 50)
def eager_serial():
    with torch.cuda.stream(sA):
        work(a, b, 50); work(c, d, 50)
print("eager serial ms:   ", round(bench(eager_serial, 10), 3))
print("eager parallel ms: ", round(bench(eager_parallel, 10), 3))
