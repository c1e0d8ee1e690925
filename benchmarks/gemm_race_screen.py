"""Race screen for the 8-phase GEMM (sync-structure discipline):
repeat-determinism (bitwise) + fp32 reference check across shapes and
many iterations with fresh random data."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from bflc_amd.ops import functional as fn

hip = fn.hip_ops()
dev = torch.device("cuda:0")
shapes = [(256, 64, 256), (256, 128, 256), (512, 192, 256),
          (512, 2304, 256), (2048, 2048, 2048), (4096, 4096, 4096),
          (12544, 2304, 512)]
iters = int(sys.argv[1]) if len(sys.argv) > 1 else 20
for M, K, N in shapes:
    for it in range(iters):
        A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        B = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        y1 = hip.gemm_raw(A, B, False, True)
        y2 = hip.gemm_raw(A, B, False, True)
        assert torch.equal(y1, y2), f"non-deterministic at {(M,K,N)} it{it}"
        if it == 0:
            ref = A.float() @ B.float().t()
            scale = ref.abs().max().clamp_min(1.0)
            torch.testing.assert_close(y1.float(), ref, rtol=0.02,
                                       atol=float(scale) * 0.02)
    print(f"{M}x{K}x{N}: {iters} iters deterministic + ref ok", flush=True)
print("RACE SCREEN PASSED")
