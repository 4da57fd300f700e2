"""Isolated timings for the remaining round-1 optimization candidates."""
import os
import shutil

_here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_tuned = os.path.join(_here, 'code2vec_amd', 'ops', 'tunableop_gfx950.csv')
if os.path.isfile(_tuned):
    for _i in range(8):
        dst = '/tmp/c2v_tunableop%d.csv' % _i
        if not os.path.isfile(dst):
            shutil.copy2(_tuned, dst)
    os.environ.setdefault('PYTORCH_TUNABLEOP_ENABLED', '1')
    os.environ.setdefault('PYTORCH_TUNABLEOP_TUNING', '0')
    os.environ.setdefault('PYTORCH_TUNABLEOP_FILENAME', '/tmp/c2v_tunableop.csv')

import time  # noqa: E402
import torch  # noqa: E402

from code2vec_amd.ops import hip_ext  # noqa: E402

ext = hip_ext(required=True)
torch.manual_seed(0)
B, C, D, V = 1024, 200, 384, 261246
N = B * C


def t(name, fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    print('%-38s %8.1f us' % (name, (time.perf_counter() - t0) / iters * 1e6))


ctx = (torch.randn(N, D, device='cuda') * 0.1).to(torch.bfloat16)
dz = (torch.randn(N, D, device='cuda') * 0.1).to(torch.bfloat16)
code = (torch.randn(B, D, device='cuda') * 0.1).to(torch.bfloat16)
shadow = (torch.randn(V, D, device='cuda') * 0.1).to(torch.bfloat16)
labels = torch.randint(0, V, (B,), device='cuda')

t('dW = ctx.T @ d_z (hipBLASLt)', lambda: (ctx.t() @ dz).float())
t('logits+CE fused fwd (gemm256)',
  lambda: ext.logits_ce_fused(code, shadow, labels))
t('logits plain (gemm256)', lambda: ext.gemm_bt_v(code, shadow, False, 2))

p = torch.randn(V, D, device='cuda')
g = torch.randn(V, D, device='cuda').to(torch.bfloat16)
m = torch.zeros_like(p)
v = torch.zeros_like(p)
sh = torch.empty_like(g)
t('adam_dense (target, with shadow)',
  lambda: ext.adam_dense_step(p, g, m, v, 1, 1e-3, 0.9, 0.999, 1e-8, sh,
                              torch.empty(0)))
lse = torch.randn(B, device='cuda')
t('ce_bwd', lambda: ext.ce_bwd(
    ext.gemm_bt_v(code, shadow, False, 2), lse, labels, 1.0 / B), iters=5)

t('dW our split-K tn', lambda: ext.gemm_tn_splitk(ctx, dz))

w = (torch.randn(D, D, device='cuda') * 0.1).to(torch.bfloat16)
t('transform GEMM 128-tile (current)', lambda: ext.gemm_bt_v(dz, w, False, 1))
t('transform GEMM 256-tile', lambda: ext.gemm_bt_v(dz, w, False, 2))
t('transform GEMM 128-tile tanh', lambda: ext.gemm_bt_v(dz, w, True, 1))
t('transform GEMM 256-tile tanh', lambda: ext.gemm_bt_v(dz, w, True, 2))
t('transform GEMM bt2 dbuf', lambda: ext.gemm_bt_v(dz, w, False, 3))
t('transform GEMM bt2 dbuf tanh', lambda: ext.gemm_bt_v(dz, w, True, 3))

# sparse embedding-update chain, java14m-ish shapes (uniform ids)
ids = torch.randint(1, 1301137, (2 * N,), device='cuda', dtype=torch.int32)
dctx = (torch.randn(N, D, device='cuda') * 0.1).to(torch.bfloat16)
tok_p = torch.randn(1301137, 128, device='cuda')
tok_m = torch.zeros_like(tok_p)
tok_v = torch.zeros_like(tok_p)
t('sparse adam ctx-direct (tok table)',
  lambda: ext.adam_sparse_rows_hash_ctx(
      tok_p, ids, dctx, 0, 256, 2, 128, tok_m, tok_v, 1, 1e-3, 0.9, 0.999,
      1e-8, torch.empty(0)))

comb = (torch.randn(B, C, D, device='cuda') * 0.1).to(torch.bfloat16)
av = torch.randn(D, device='cuda')
maskf = torch.ones(B, C, device='cuda')
codev, alphav = ext.attention_fwd(comb, av, maskf)
dcv = torch.randn(B, D, device='cuda')
t('attn_bwd', lambda: ext.attention_bwd(comb, av, alphav, dcv, True))
t('attn_fwd', lambda: ext.attention_fwd(comb, av, maskf))
