"""Isolate the dX-GEMM dropout-epilogue cost (k_gemm_bt<false,true>)."""
import time

import torch

from code2vec_amd.ops import hip_ext

ext = hip_ext(required=True)
torch.manual_seed(0)
N, D = 204800, 384
dz = (torch.randn(N, D, device='cuda') * 0.1).to(torch.bfloat16)
w = (torch.randn(D, D, device='cuda') * 0.1).to(torch.bfloat16)
seed_t = torch.tensor([1234], dtype=torch.int64, device='cuda')


def t(name, fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print('%-34s %8.1f us' % (name, (time.perf_counter() - t0) / iters * 1e6))


t('gemm_bt plain', lambda: ext.gemm_bt_bf16(dz, w))
t('gemm_bt tanh', lambda: ext.transform_tanh_fwd(dz, w))
t('gemm_bt dropout (seed scalar)',
  lambda: ext.gemm_bt_dropout(dz, w, 0.75, 1234, torch.empty(0)))
t('gemm_bt dropout (seed_t device)',
  lambda: ext.gemm_bt_dropout(dz, w, 0.75, 0, seed_t))
t('unfused: gemm_bt + dropout_bwd',
  lambda: ext.gather_concat_bwd(ext.gemm_bt_bf16(dz, w), 0.75, 1234, True,
                                torch.empty(0)))
