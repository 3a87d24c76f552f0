import sys, os, time, torch
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from evotorch_amd import ops

def tb(name, fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter()-t0)/iters*1000:.3f} ms")

n = 4096
A = torch.randn(n, n, device="cuda:0"); C = A @ A.T + n * torch.eye(n, device="cuda:0")
info = torch.zeros(1, dtype=torch.int32, device="cuda:0")

# 32 sequential panel factorizations (fresh SPD each time to avoid NaN drift)
P = [ (torch.randn(128,128,device="cuda:0") @ torch.randn(128,128,device="cuda:0").T + 256*torch.eye(128,device="cuda:0")) for _ in range(32)]
def panels():
    for p in P: ops.potrf_tile_(p.clone(), info)
tb("32x potrf_tile_128", panels)

L = torch.linalg.cholesky(C)
def outer_trsm():
    for k in range(0, n, 512):
        e = k + 512
        if e < n:
            torch.linalg.solve_triangular(L[k:e, k:e].mT, C[e:, k:e], upper=True, left=False)
tb("8x outer trsm", outer_trsm)

def inner_trsm():
    for k in range(0, n, 512):
        for p in range(k, k+512, 128):
            q = p + 128
            if q < k+512:
                torch.linalg.solve_triangular(L[p:q, p:q].mT, C[q:k+512, p:q], upper=True, left=False)
tb("24x inner trsm", inner_trsm)

def gemms():
    for k in range(0, n, 512):
        e = k + 512
        if e < n:
            C[e:, k:e] @ C[e:, k:e].T
def igemms():
    for k in range(0, n, 512):
        for p in range(k, k+512, 128):
            q = p + 128
            if q < k+512:
                C[q:k+512, p:q] @ C[q:k+512, p:q].T
tb("outer gemm (approx)", gemms)
tb("inner gemm (approx)", igemms)
tb("torch.linalg.cholesky(512) x8", lambda: [torch.linalg.cholesky(C[k:k+512, k:k+512] + 0*C[k:k+512,k:k+512]) for k in range(0, n, 512)])
tb("one potrf_tile_128", lambda: ops.potrf_tile_(P[0].clone(), info))
