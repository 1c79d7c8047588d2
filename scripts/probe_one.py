import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from tfservingcache_amd.engine import _tfsc_engine as ext
dev = "cuda:0"
M, N, K = 128, 128, 64
A = (torch.randn(M, K, device=dev) * 0.5).to(torch.bfloat16).contiguous()
Bt = (torch.randn(N, K, device=dev) * 0.5).to(torch.bfloat16).contiguous()
C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
print("ptrs", hex(A.data_ptr()), hex(Bt.data_ptr()), hex(C.data_ptr()), flush=True)
plan = ext.ExecPlan([(ext.K_GEMM, [A.data_ptr(), Bt.data_ptr(), 0, 0, C.data_ptr()],
                      [M, N, K, ext.ACT_NONE], [1.0])])
plan.run()
torch.cuda.synchronize()
want = A.float() @ Bt.float().t()
err = (C.float() - want).abs().max().item()
print("gemm128 maxerr", err, flush=True)
