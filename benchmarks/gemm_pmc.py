import torch, sys, os
sys.path.insert(0, "/root/repo")
from bflc_amd.ops import functional as fn
hip = fn.hip_ops()
dev = torch.device("cuda:0")
A = torch.randn(4096, 4096, device=dev, dtype=torch.bfloat16)
B = torch.randn(4096, 4096, device=dev, dtype=torch.bfloat16)  # [N][K] for tb
for _ in range(20):
    y = hip.gemm_raw(A, B, False, True)
torch.cuda.synchronize()
print("pmc run done", y.shape)
