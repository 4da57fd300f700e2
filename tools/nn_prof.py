"""Isolated k_gemm_nn_splitk run for rocprofv3 capture (d_code shape)."""
import torch

from code2vec_amd.ops import hip_ext

ext = hip_ext(required=True)
torch.manual_seed(0)
N, M, K = 1024, 384, 261246
A = (torch.randn(N, K, device='cuda') * 0.1).to(torch.bfloat16)
B = (torch.randn(K, M, device='cuda') * 0.1).to(torch.bfloat16)
for _ in range(3):
    ext.gemm_nn_splitk(A, B)
torch.cuda.synchronize()
import time
t0 = time.perf_counter()
for _ in range(20):
    C = ext.gemm_nn_splitk(A, B)
torch.cuda.synchronize()
print('nn splitk: %.1f us/iter' % ((time.perf_counter() - t0) / 20 * 1e6))
