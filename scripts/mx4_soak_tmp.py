import sys, time, torch
sys.path.insert(0, "/root/repo")
from hpc_patterns_amd import ops
dev = torch.device("cuda", 0)
sz = 16384
g = torch.Generator(device="cpu").manual_seed(163)
p4a = torch.randint(0, 256, (sz, sz // 2), generator=g,
                    dtype=torch.uint8).cuda()
p4b = torch.randint(0, 256, (sz, sz // 2), generator=g,
                    dtype=torch.uint8).cuda()
sa = torch.randint(120, 135, (sz, sz // 32), generator=g,
                   dtype=torch.int16).to(torch.uint8).cuda()
sb = torch.randint(120, 135, (sz, sz // 32), generator=g,
                   dtype=torch.int16).to(torch.uint8).cuda()
c = torch.empty(sz, sz, dtype=torch.float32, device=dev)
ops.gemm_mxfp4(c, p4a, p4b, sa, sb)
torch.cuda.synchronize()
first = c.clone()
bad = 0
t0 = time.perf_counter()
N = 40
for i in range(N):
    ops.gemm_mxfp4(c, p4a, p4b, sa, sb)
    torch.cuda.synchronize()
    if not torch.equal(c, first):
        bad += 1
        print("MISMATCH iter", i)
dt = (time.perf_counter() - t0) / N
print(f"mx4 256^2 race screen: {N+1} runs at 16384^3 (random data AND "
      f"random e8m0 scales), {bad} mismatches, "
      f"{2.0*sz**3/dt/1e12:.0f} TF sustained")
