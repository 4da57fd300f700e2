"""Isolated gemm256 plain vs CE_PART for PMC comparison."""
import torch

from code2vec_amd.ops import hip_ext

ext = hip_ext(required=True)
torch.manual_seed(0)
B, V, D = 1024, 261246, 384
code = (torch.randn(B, D, device='cuda') * 0.1).to(torch.bfloat16)
shadow = (torch.randn(V, D, device='cuda') * 0.1).to(torch.bfloat16)
labels = torch.randint(0, V, (B,), device='cuda')
for _ in range(3):
    ext.gemm_bt_v(code, shadow, False, 2)
    ext.logits_ce_fused(code, shadow, labels)
torch.cuda.synchronize()
for _ in range(10):
    ext.gemm_bt_v(code, shadow, False, 2)
for _ in range(10):
    ext.logits_ce_fused(code, shadow, labels)
torch.cuda.synchronize()
