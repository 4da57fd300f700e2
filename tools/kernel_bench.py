#!/usr/bin/env python3
"""Micro-benchmarks for individual engine ops on MI355X (run via gpurun).
Prints per-op times so kernel changes can be A/B'd without full bench runs."""

import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import time

import torch

from code2vec_amd.ops import hip_ext

B, C, d, D, V = 1024, 200, 128, 384, 261246
N = B * C


def timeit(name, fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / iters * 1e6
    print('%-38s %10.1f us' % (name, us))
    return us


def main():
    ext = hip_ext(True)
    torch.manual_seed(0)
    dev = 'cuda'
    ctx = torch.randn(N, D, device=dev).to(torch.bfloat16)
    w_oi = torch.randn(D, D, device=dev).to(torch.bfloat16)
    comb = torch.randn(B, C, D, device=dev).to(torch.bfloat16)
    a = torch.randn(D, device=dev)
    mask = torch.ones(B, C, device=dev)
    code = torch.randn(B, D, device=dev).to(torch.bfloat16)
    shadow = torch.randn(V, D, device=dev).to(torch.bfloat16)
    logits = torch.randn(B, V, device=dev).to(torch.bfloat16)
    labels = torch.randint(0, V, (B,), device=dev)
    lse = torch.randn(B, device=dev)

    timeit('gather tok/path tables alloc', lambda: None, 1, 0)

    timeit('transform_tanh_fwd (our MFMA)', lambda: ext.transform_tanh_fwd(ctx, w_oi))
    timeit('gemm_bt (our MFMA, no tanh)', lambda: ext.gemm_bt_bf16(ctx, w_oi))
    timeit('torch.matmul same shape', lambda: ctx @ w_oi.t())

    timeit('logits hipBLASLt', lambda: code @ shadow.t())
    timeit('logits our gemm_bt', lambda: ext.gemm_bt_bf16(code, shadow))

    dl = torch.randn(B, V, device=dev).to(torch.bfloat16)
    timeit('d_code hipBLASLt', lambda: dl @ shadow)
    timeit('d_code our split-K nn', lambda: ext.gemm_nn_splitk(dl, shadow))
    timeit('d_target hipBLASLt', lambda: dl.t() @ code)
    timeit('d_target our tn', lambda: ext.gemm_tn_bf16(dl, code))

    timeit('attn_fwd', lambda: ext.attention_fwd(comb, a, mask))
    alpha = ext.attention_fwd(comb, a, mask)[1]
    dcode = torch.randn(B, D, device=dev)
    timeit('attn_bwd', lambda: ext.attention_bwd(comb, a, alpha, dcode, False))

    timeit('ce_fwd', lambda: ext.ce_fwd(logits, labels))
    timeit('ce_bwd', lambda: ext.ce_bwd(logits, lse, labels, 1.0 / B))

    p = torch.randn(V, D, device=dev)
    g = torch.randn(V, D, device=dev).to(torch.bfloat16)
    m = torch.zeros_like(p)
    v = torch.zeros_like(p)
    timeit('adam_dense target (w/ shadow)',
           lambda: ext.adam_dense_step(p, g, m, v, 1, 1e-3, 0.9, 0.999, 1e-8, shadow, torch.empty(0)))

    ids = torch.randint(0, 1301137, (2 * N,), device=dev)
    rows = torch.randn(2 * N, d, device=dev).to(torch.bfloat16)
    tok = torch.randn(1301137, d, device=dev)
    mm = torch.zeros_like(tok)
    vv = torch.zeros_like(tok)

    def sparse_step():
        ext.adam_sparse_rows_hash(tok, ids, rows, mm, vv, 1, 1e-3, 0.9,
                                  0.999, 1e-8, torch.empty(0), torch.empty(0))
    timeit('sparse adam (hash dedup, tok)', sparse_step)
    ids_pad = ids.clone(); ids_pad[::3] = 0   # heavy PAD duplication
    def sparse_step_pad():
        ext.adam_sparse_rows_hash(tok, ids_pad, rows, mm, vv, 1, 1e-3, 0.9,
                                  0.999, 1e-8, torch.empty(0), torch.empty(0))
    timeit('sparse adam (hash, 33% PAD ids)', sparse_step_pad)
    # realistic: PAD-slot grads are exactly zero (masked contexts produce no
    # gradient), and the accumulate kernel skips atomics for zero values
    rows_pad0 = rows.clone(); rows_pad0[::3] = 0
    def sparse_step_pad0():
        ext.adam_sparse_rows_hash(tok, ids_pad, rows_pad0, mm, vv, 1, 1e-3,
                                  0.9, 0.999, 1e-8, torch.empty(0),
                                  torch.empty(0))
    timeit('sparse adam (33% PAD, zero grads)', sparse_step_pad0)

    def zipf_ids(vocab, n, s=1.1, seed=7):
        p = 1.0 / torch.arange(1, vocab + 1, dtype=torch.float64) ** s
        cdf = torch.cumsum(p / p.sum(), 0).to(dev)
        g = torch.Generator(device=dev).manual_seed(seed)
        u = torch.rand(n, device=dev, generator=g, dtype=torch.float64)
        return torch.searchsorted(cdf, u).clamp_(0, vocab - 1)

    for s in (1.0, 1.1):
        ids_z = zipf_ids(1301137, 2 * N, s)
        uz = torch.unique(ids_z).numel()
        def sparse_step_zipf():
            ext.adam_sparse_rows_hash(tok, ids_z, rows, mm, vv, 1, 1e-3, 0.9,
                                      0.999, 1e-8, torch.empty(0),
                                      torch.empty(0))
        timeit('sparse adam (zipf s=%.1f, %dK uniq)' % (s, uz // 1000),
               sparse_step_zipf)
        timeit('dedup_sum rows (zipf s=%.1f)' % s,
               lambda: ext.sparse_dedup_sum_rows(ids_z, rows))
    timeit('dedup_sum rows (uniform ids)',
           lambda: ext.sparse_dedup_sum_rows(ids, rows))
    timeit('torch.unique only', lambda: torch.unique(ids, return_inverse=True))


if __name__ == '__main__':
    main()
