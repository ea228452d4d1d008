"""Run just the hand GEMM on one shape for rocprofv3 PMC profiling."""
import math
import sys, os
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch

M, K, N = (int(x) for x in (sys.argv[1:4] or [65536, 768, 3072]))
iters = int(sys.argv[4]) if len(sys.argv) > 4 else 5
from ravnest_amd.ops import get_ext
ext = get_ext(True)
dev = torch.device("cuda", 0)
torch.manual_seed(0)
a = (torch.randn(M, K, device=dev) / math.sqrt(K)).to(torch.bfloat16)
b = torch.randn(N, K, device=dev).to(torch.bfloat16)
for _ in range(2):
    ext.gemm_nt_bf16(a, b, None, 0)
torch.cuda.synchronize()
for _ in range(iters):
    ext.gemm_nt_bf16(a, b, None, 0)
torch.cuda.synchronize()
print("done")
