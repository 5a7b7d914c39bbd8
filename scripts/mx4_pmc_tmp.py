import sys, torch
sys.path.insert(0, "/root/repo")
from hpc_patterns_amd import ops
dev = torch.device("cuda", 0)
sz = 16384
p4a = torch.randint(0, 256, (sz, sz // 2), dtype=torch.uint8, device=dev)
p4b = torch.randint(0, 256, (sz, sz // 2), dtype=torch.uint8, device=dev)
s1 = torch.full((sz, sz // 32), 127, dtype=torch.uint8, device=dev)
c = torch.empty(sz, sz, dtype=torch.float32, device=dev)
for _ in range(3):
    ops.gemm_mxfp4(c, p4a, p4b, s1, s1)
torch.cuda.synchronize()
print("done")
