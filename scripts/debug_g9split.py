import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from baton_amd.ops._ext import require_hip
ops = require_hip()
torch.manual_seed(32)
M = N = 768; K = 4096
A = torch.randn(K, M, device="cuda").bfloat16().contiguous()
B = torch.randn(K, N, device="cuda").bfloat16().contiguous()
ref = A.float().t() @ B.float()
# non-split g9 refcheck at an 8ph-eligible shape (>=160 blocks)
M2, N2, K2 = 4096, 2560, 512
A2 = torch.randn(M2, K2, device="cuda").bfloat16().contiguous()
B2 = torch.randn(N2, K2, device="cuda").bfloat16().contiguous()
C2 = ops.gemm(A2, B2, 0)
ref2 = A2.float() @ B2.float().t()
print(f"non-split g9 @{M2}x{N2}x{K2}: max err {(C2.float()-ref2).abs().max().item():.3e}")

for it in range(3):
    C = ops.gemm(A, B, 2)
    err = (C.float() - ref).abs()
    print(f"iter {it}: max {err.max().item():.3e}")
    bad = []
    for mt in range(3):
        for nt in range(3):
            e = err[mt*256:(mt+1)*256, nt*256:(nt+1)*256].max().item()
            if e > 1:
                # find worst row/col inside
                sub = err[mt*256:(mt+1)*256, nt*256:(nt+1)*256]
                r = int(sub.max(dim=1).values.argmax())
                bad.append((mt, nt, round(e,1), r))
    print("  bad tiles (mt, nt, err, worst_local_row):", bad)
