import os, time, torch, sys
sys.path.insert(0, "/root/repo")
from hpc_patterns_amd import ops
def t(fn, reps=6, warm=2):
    for _ in range(warm): fn()
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(reps):
        t0 = time.perf_counter(); fn(); torch.cuda.synchronize()
        best = min(best, time.perf_counter() - t0)
    return best
dev = torch.device("cuda", 0)
for sz in (8192, 16384):
    fl = 2.0 * sz**3
    a = (torch.rand(sz, sz, device=dev)*2-1).to(torch.bfloat16)
    b = (torch.rand(sz, sz, device=dev)*2-1).to(torch.bfloat16)
    ctrl = fl/t(lambda: torch.matmul(a, b.t()))/1e12
    a8, b8 = a.to(torch.float8_e4m3fn), b.to(torch.float8_e4m3fn)
    s1 = torch.full((sz, sz // 32), 127, dtype=torch.uint8, device=dev)
    c = torch.empty(sz, sz, dtype=torch.float32, device=dev)
    best = {}
    for rnd in range(3):
        for w in ("plain", "32"):
            os.environ["HPK_MX8_VARIANT"] = w
            best[w] = max(best.get(w, 0),
                          fl/t(lambda: ops.gemm_mxfp8(c, a8, b8, s1, s1))/1e12)
        os.environ["HPK_GEMM_VARIANT"] = "8ph"
        best["fp8_8ph"] = max(best.get("fp8_8ph", 0),
                              fl/t(lambda: ops.gemm_fp8(c, a8, b8))/1e12)
        os.environ.pop("HPK_GEMM_VARIANT", None)
        best["fp8_32"] = max(best.get("fp8_32", 0),
                             fl/t(lambda: ops.gemm_fp8(c, a8, b8))/1e12)
    print(f"{sz}: ctrl={ctrl:.0f}  mx8plain={best['plain']:.0f}  "
          f"mx8_32={best['32']:.0f}  fp8_8ph={best['fp8_8ph']:.0f}  "
          f"fp8_32={best['fp8_32']:.0f} TF", flush=True)
    del a, b, a8, b8, s1, c
    torch.cuda.empty_cache()
