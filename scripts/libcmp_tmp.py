import os, time, torch, sys
sys.path.insert(0, "/root/repo")
from hpc_patterns_amd import ops
def t(fn, reps=5, warm=2):
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
a = (torch.rand(sz, sz, device=dev)*2-1).to(torch.bfloat16)
b = (torch.rand(sz, sz, device=dev)*2-1).to(torch.bfloat16)
c = torch.empty(sz, sz, dtype=torch.float32, device=dev)
a8, b8 = a.to(torch.float8_e4m3fn), b.to(torch.float8_e4m3fn)
s1 = torch.full((sz, sz // 32), 127, dtype=torch.uint8, device=dev)
print(f"hipBLASLt bf16 (start): {fl/t(lambda: torch.matmul(a, b.t()))/1e12:7.1f} TF", flush=True)
GS = (1, 4, 8, 16, 32)
best_bf = {g: 0.0 for g in GS}
best_mx = {g: 0.0 for g in GS}
for rnd in range(3):
    for g in GS:
        os.environ["HPK_GEMM_GROUP"] = str(g)
        best_bf[g] = max(best_bf[g], fl/t(lambda: ops.gemm_bf16(c, a, b))/1e12)
        best_mx[g] = max(best_mx[g], fl/t(lambda: ops.gemm_mxfp8(c, a8, b8, s1, s1))/1e12)
    print(f"round {rnd}: bf16 " +
          " ".join(f"g{g}={best_bf[g]:.0f}" for g in GS) +
          " | mx8 " + " ".join(f"g{g}={best_mx[g]:.0f}" for g in GS),
          flush=True)
print(f"hipBLASLt bf16 (end)  : {fl/t(lambda: torch.matmul(a, b.t()))/1e12:7.1f} TF")
