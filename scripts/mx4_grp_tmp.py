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
sz = 16384
fl = 2.0 * sz**3
p4a = torch.randint(0, 256, (sz, sz // 2), dtype=torch.uint8, device=dev)
p4b = torch.randint(0, 256, (sz, sz // 2), dtype=torch.uint8, device=dev)
s1 = torch.full((sz, sz // 32), 127, dtype=torch.uint8, device=dev)
c = torch.empty(sz, sz, dtype=torch.float32, device=dev)
best = {}
for rnd in range(3):
    for gset in ("auto", "1", "4", "8", "16", "32"):
        if gset == "auto":
            os.environ.pop("HPK_GEMM_GROUP", None)
        else:
            os.environ["HPK_GEMM_GROUP"] = gset
        best[gset] = max(best.get(gset, 0),
                         fl/t(lambda: ops.gemm_mxfp4(c, p4a, p4b, s1, s1))/1e12)
    print("  ".join(f"g{k}={v:.0f}" for k, v in best.items()), flush=True)
