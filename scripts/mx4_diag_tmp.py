import os, sys, torch
sys.path.insert(0, "/root/repo")
os.environ["HPK_MX4_WAVES"] = "32"
from hpc_patterns_amd import ops
m = n = k = 128
g = torch.Generator(device="cpu").manual_seed(7)
vals = torch.tensor([0., 0.5, -0.5, 1., -1., 1.5, -1.5, 2., -2.])
fa = vals[torch.randint(0, 9, (m, k), generator=g)]
fb = vals[torch.randint(0, 9, (n, k), generator=g)]
pa, pb = ops.e2m1_pack(fa).cuda(), ops.e2m1_pack(fb).cuda()
sa = torch.full((m, k // 32), 127, dtype=torch.uint8, device="cuda")
c = torch.full((m, n), 12345.0, device="cuda")
ops.gemm_mxfp4(c, pa, pb, sa, sa.clone())
torch.cuda.synchronize()
ref = torch.matmul(fa, fb.t()).cuda()
unwritten = (c == 12345.0)
wrong = (c != ref) & ~unwritten
print("unwritten:", int(unwritten.sum()), "wrong:", int(wrong.sum()),
      "of", m * n)
uw = unwritten.nonzero()
wr = wrong.nonzero()
for name, t in (("unwritten", uw), ("wrong", wr)):
    if len(t):
        rows = sorted(set(t[:, 0].tolist()))[:8]
        cols = sorted(set(t[:, 1].tolist()))[:16]
        print(f"{name}: first rows {rows} cols {cols}")
# a couple of wrong-value samples
for i in range(min(4, len(wr))):
    r, cc = wr[i].tolist()
    print(f"  C[{r},{cc}]={c[r,cc].item():.4f} ref={ref[r,cc].item():.4f}")
