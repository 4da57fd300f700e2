"""HIP kernel parity vs the pure-torch fp32 reference (runs on MI355X only).

Every kernel is checked against ops/reference.py on the same inputs with
random ASYMMETRIC data (transpose-detecting — CDNA4 guide G9/errata #3).
Tolerances account for bf16 compute in the HIP path vs fp32 reference."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from code2vec_amd.ops import reference as R  # noqa: E402


def ext():
    from code2vec_amd.ops import hip_ext
    return hip_ext(required=True)


def randn(*shape, dtype=torch.float32, scale=1.0, seed=None):
    if seed is not None:
        torch.manual_seed(seed)
    return (torch.randn(*shape, dtype=torch.float32) * scale).to(dtype).cuda()


def rel_err(a, b):
    a, b = a.float(), b.float()
    return ((a - b).abs() / (b.abs().clamp_min(1e-3))).max().item()


# ---------------------------------------------------------------------------

def test_gather_concat_no_dropout():
    torch.manual_seed(0)
    Vt, Vp, d, B, C = 100, 80, 16, 4, 7
    tok = randn(Vt, d)
    path = randn(Vp, d)
    src = torch.randint(0, Vt, (B, C), dtype=torch.int32).cuda()
    pth = torch.randint(0, Vp, (B, C), dtype=torch.int32).cuda()
    tgt = torch.randint(0, Vt, (B, C), dtype=torch.int32).cuda()
    out = ext().gather_concat_fwd(tok, path, src, pth, tgt, 1.0, 7, True, torch.empty(0))
    ref = R.gather_concat_fwd(tok, path, src, pth, tgt, 1.0, 7, True)
    assert out.dtype == torch.bfloat16
    assert torch.equal(out, ref)


def test_gather_concat_dropout_matches_reference_hash():
    torch.manual_seed(1)
    Vt, Vp, d, B, C = 50, 40, 8, 3, 5
    tok, path = randn(Vt, d), randn(Vp, d)
    src = torch.randint(0, Vt, (B, C), dtype=torch.int32).cuda()
    pth = torch.randint(0, Vp, (B, C), dtype=torch.int32).cuda()
    tgt = torch.randint(0, Vt, (B, C), dtype=torch.int32).cuda()
    seed = 987654321
    out = ext().gather_concat_fwd(tok, path, src, pth, tgt, 0.75, seed, True, torch.empty(0))
    ref = R.gather_concat_fwd(tok, path, src, pth, tgt, 0.75, seed, True)
    assert torch.equal(out, ref)
    # eval: no dropout even with keep<1
    out_eval = ext().gather_concat_fwd(tok, path, src, pth, tgt, 0.75, seed, False, torch.empty(0))
    ref_eval = R.gather_concat_fwd(tok, path, src, pth, tgt, 0.75, seed, False)
    assert torch.equal(out_eval, ref_eval)


def test_dropout_bwd_mask():
    g = randn(64, 24, dtype=torch.bfloat16, seed=2)
    out = ext().gather_concat_bwd(g, 0.75, 555, True, torch.empty(0))
    ref = R.gather_concat_bwd(g, 0.75, 555, True)
    assert torch.equal(out, ref)


# ---------------------------------------------------------------------------

@pytest.mark.parametrize("N,M,K", [(128, 128, 32), (256, 384, 384),
                                   (204800 // 50, 384, 384), (100, 70, 64)])
def test_gemm_bt_vs_matmul(N, M, K):
    # random asymmetric inputs: a transposed output CANNOT pass this
    A = randn(N, K, dtype=torch.bfloat16, scale=0.5, seed=N + M)
    Bt = randn(M, K, dtype=torch.bfloat16, scale=0.5)
    C = ext().gemm_bt_bf16(A, Bt)
    ref = A.float() @ Bt.float().t()
    err = (C.float() - ref).abs().max().item()
    denom = ref.abs().max().item()
    assert err / denom < 0.02, 'max err %g vs scale %g' % (err, denom)


def test_gemm_bt_tanh_epilogue():
    A = randn(256, 64, dtype=torch.bfloat16, scale=0.5, seed=5)
    Bt = randn(128, 64, dtype=torch.bfloat16, scale=0.5)
    C = ext().transform_tanh_fwd(A, Bt)
    ref = torch.tanh(A.float() @ Bt.float().t())
    assert (C.float() - ref).abs().max().item() < 0.02


def test_tanh_bwd_mul():
    dy = randn(64, 48, dtype=torch.bfloat16, seed=6)
    y = randn(64, 48, dtype=torch.bfloat16, scale=0.9)
    dz = ext().tanh_bwd_mul(dy, y)
    ref = dy.float() * (1 - y.float() ** 2)
    assert (dz.float() - ref).abs().max().item() < 0.05


# ---------------------------------------------------------------------------

@pytest.mark.parametrize("B,C,D", [(4, 17, 64), (8, 200, 384), (2, 1, 128)])
def test_attention_fwd(B, C, D):
    comb = randn(B, C, D, dtype=torch.bfloat16, scale=0.5, seed=B * C)
    a = randn(D, scale=0.3)
    mask = (torch.rand(B, C) > 0.3).float().cuda()
    mask[:, 0] = 1.0
    code, alpha = ext().attention_fwd(comb, a, mask)
    code_ref, alpha_ref = R.attention_fwd(comb, a, mask)
    assert (alpha - alpha_ref).abs().max().item() < 1e-3
    assert (code - code_ref).abs().max().item() < 5e-3
    # alpha rows sum to 1 and masked entries are 0
    assert torch.allclose(alpha.sum(1), torch.ones(B, device='cuda'), atol=1e-4)
    assert torch.all(alpha[mask == 0] == 0)


def test_attention_fwd_all_masked_row():
    comb = randn(3, 10, 64, dtype=torch.bfloat16, seed=9)
    a = randn(64)
    mask = torch.ones(3, 10).cuda()
    mask[1] = 0.0
    code, alpha = ext().attention_fwd(comb, a, mask)
    assert torch.all(torch.isfinite(code))
    assert torch.all(code[1] == 0) and torch.all(alpha[1] == 0)


@pytest.mark.parametrize("B,C,D", [(4, 17, 64), (8, 200, 384)])
def test_attention_bwd(B, C, D):
    comb = randn(B, C, D, dtype=torch.bfloat16, scale=0.5, seed=B + C + D)
    a = randn(D, scale=0.3)
    mask = (torch.rand(B, C) > 0.3).float().cuda()
    mask[:, 0] = 1.0
    _, alpha = R.attention_fwd(comb, a, mask)
    d_code = randn(B, D, scale=0.5)
    d_comb, d_a = ext().attention_bwd(comb, a, alpha, d_code, False)
    d_comb_ref, d_a_ref = R.attention_bwd(comb, a, alpha, d_code)
    assert (d_comb.float() - d_comb_ref.float()).abs().max().item() < 1e-2
    assert (d_a - d_a_ref).abs().max().item() / max(d_a_ref.abs().max().item(), 1e-3) < 1e-2


# ---------------------------------------------------------------------------

@pytest.mark.parametrize("B,V", [(4, 1000), (8, 261246), (3, 77)])
def test_ce_fwd_bwd(B, V):
    logits = randn(B, V, dtype=torch.bfloat16, scale=2.0, seed=B + V)
    labels = torch.randint(0, V, (B,)).cuda()
    loss, lse = ext().ce_fwd(logits, labels)
    loss_ref, lse_ref = R.ce_fwd(logits, labels)
    assert (lse - lse_ref).abs().max().item() < 2e-3
    assert (loss - loss_ref).abs().max().item() < 4e-3
    d = ext().ce_bwd(logits, lse, labels, 1.0 / B)
    d_ref = R.ce_bwd(logits, lse_ref, labels, 1.0 / B)
    assert (d.float() - d_ref.float()).abs().max().item() < 2e-3


# ---------------------------------------------------------------------------

@pytest.mark.parametrize("B,V,k", [(4, 1000, 10), (32, 261246, 10), (2, 50, 32)])
def test_topk_vs_torch(B, V, k):
    logits = randn(B, V, dtype=torch.bfloat16, scale=3.0, seed=B + V)
    vals, idx = ext().topk(logits, k)
    ref_vals, ref_idx = torch.topk(logits.float(), k=k, dim=1)
    assert torch.equal(vals, ref_vals)
    # indices must point at the same values (tie order may differ from torch)
    picked = logits.float().gather(1, idx)
    assert torch.equal(picked, ref_vals)
    # our tie rule: strictly non-increasing values, index ascending on ties
    for b in range(B):
        for j in range(1, k):
            assert vals[b, j] < vals[b, j - 1] or idx[b, j] > idx[b, j - 1]


# ---------------------------------------------------------------------------

@pytest.mark.parametrize("B,S", [(8, 32), (1024, 8192)])
def test_sampled_ce_fwd_bwd(B, S):
    V = 261246
    logits = randn(B, B + S, dtype=torch.bfloat16, scale=2.0, seed=B)
    labels = torch.randint(0, V, (B,)).cuda()
    sampled = torch.randint(0, V, (S,)).cuda()
    sampled[0] = labels[0]  # accidental hit
    # the kernels compute the log-uniform corrections INLINE from (S, V);
    # the reference takes them explicitly — comparing validates that math
    ct = torch.log(R.log_uniform_probs(labels, V) * S)
    cs = torch.log(R.log_uniform_probs(sampled, V) * S)
    loss, lse = ext().sampled_ce_fwd(logits, labels, sampled, V)
    loss_ref, lse_ref = R.sampled_ce_fwd(logits, labels, sampled, ct, cs)
    assert (lse - lse_ref).abs().max().item() < 2e-3
    assert (loss - loss_ref).abs().max().item() < 4e-3
    d = ext().sampled_ce_bwd(logits, labels, sampled, V, lse, 1.0 / B)
    d_ref = R.sampled_ce_bwd(logits, labels, sampled, ct, cs, lse_ref, 1.0 / B)
    assert (d.float() - d_ref.float()).abs().max().item() < 2e-3


def test_adam_sparse_rows_with_shadow():
    torch.manual_seed(13)
    Vr, d, n = 32, 16, 100
    p = randn(Vr, d); p_ref = p.clone()
    m = torch.zeros_like(p); v = torch.zeros_like(p)
    m_ref = m.clone(); v_ref = v.clone()
    shadow = p.to(torch.bfloat16)
    shadow_ref = p_ref.to(torch.bfloat16)
    ids = torch.randint(0, Vr, (n,), dtype=torch.int64).cuda()
    rows = randn(n, d, dtype=torch.bfloat16)
    from code2vec_amd.ops import functional as F
    F.adam_sparse_rows_step(p, ids, rows, m, v, 1, 1e-3, 0.9, 0.999, 1e-8,
                            shadow=shadow)
    R.adam_sparse_rows_step(p_ref, ids, rows, m_ref, v_ref, 1, 1e-3, 0.9,
                            0.999, 1e-8, shadow=shadow_ref)
    assert (p - p_ref).abs().max().item() < 1e-5
    assert torch.equal(shadow, shadow_ref)


def test_adam_dense():
    torch.manual_seed(11)
    n = 10000
    p = randn(n)
    p_ref = p.clone()
    g = randn(n, dtype=torch.bfloat16)
    m = torch.zeros_like(p); v = torch.zeros_like(p)
    m_ref = m.clone(); v_ref = v.clone()
    shadow = torch.zeros(n, dtype=torch.bfloat16).cuda()
    for t in (1, 2, 3):
        ext().adam_dense_step(p, g, m, v, t, 1e-3, 0.9, 0.999, 1e-8, shadow, torch.empty(0))
        R.adam_dense_step(p_ref, g, m_ref, v_ref, t, 1e-3, 0.9, 0.999, 1e-8)
    assert (p - p_ref).abs().max().item() < 1e-6
    assert (m - m_ref).abs().max().item() < 1e-6
    assert torch.equal(shadow, p.to(torch.bfloat16))


def test_adam_sparse_rows():
    torch.manual_seed(12)
    Vr, d, n = 64, 16, 200
    p = randn(Vr, d); p_ref = p.clone()
    m = torch.zeros_like(p); v = torch.zeros_like(p)
    m_ref = m.clone(); v_ref = v.clone()
    ids = torch.randint(0, Vr, (n,), dtype=torch.int64).cuda()
    rows = randn(n, d, dtype=torch.bfloat16)
    from code2vec_amd.ops import functional as F
    F.adam_sparse_rows_step(p, ids, rows, m, v, 1, 1e-3, 0.9, 0.999, 1e-8)
    R.adam_sparse_rows_step(p_ref, ids, rows, m_ref, v_ref, 1, 1e-3, 0.9, 0.999, 1e-8)
    assert (p - p_ref).abs().max().item() < 1e-5
    assert (m - m_ref).abs().max().item() < 1e-5
    assert (v - v_ref).abs().max().item() < 1e-5


# ---------------------------------------------------------------------------

def test_full_train_step_gpu_vs_cpu():
    """Whole-engine integration: a few bf16 GPU steps track the fp32 CPU
    engine losses within bf16 tolerance on a small model."""
    from code2vec_amd.config import Config
    from code2vec_amd.models.network import Code2VecNetwork

    def build(device, dtype):
        cfg = Config(set_defaults=True)
        cfg.TRAIN_DATA_PATH_PREFIX = 'unused'
        cfg.MAX_CONTEXTS = 20
        cfg.TOKEN_EMBEDDINGS_SIZE = 64
        cfg.PATH_EMBEDDINGS_SIZE = 64
        cfg.CODE_VECTOR_SIZE = 192
        cfg.TARGET_EMBEDDINGS_SIZE = 192
        cfg.DROPOUT_KEEP_RATE = 1.0
        cfg.COMPUTE_DTYPE = dtype
        torch.manual_seed(99)
        return Code2VecNetwork(cfg, 500, 300, 200, device=device)

    net_gpu = build('cuda:0', 'bf16')
    net_cpu = build('cpu', 'fp32')
    # identical init (same CPU generator seeding path)
    for n in net_gpu.param_names():
        assert torch.allclose(net_gpu.get_param(n).cpu(), net_cpu.get_param(n))

    g = torch.Generator().manual_seed(17)
    B, C = 16, 20
    src = torch.randint(0, 500, (B, C), generator=g, dtype=torch.int32)
    pth = torch.randint(0, 300, (B, C), generator=g, dtype=torch.int32)
    tgt = torch.randint(0, 500, (B, C), generator=g, dtype=torch.int32)
    mask = torch.ones(B, C)
    labels = torch.randint(1, 200, (B,), generator=g)
    for step in range(5):
        lg = float(net_gpu.train_step(src.cuda(), pth.cuda(), tgt.cuda(),
                                      mask.cuda(), labels.cuda()))
        lc = float(net_cpu.train_step(src, pth, tgt, mask, labels))
        assert abs(lg - lc) < 0.05 * max(1.0, abs(lc)), (step, lg, lc)


# ---------------------------------------------------------------------------

@pytest.mark.parametrize("N,M,K", [(256, 256, 64), (512, 384, 384),
                                   (1024, 2048, 384), (300, 260, 128)])
def test_gemm256_vs_matmul(N, M, K):
    """256x256-tile 2-phase GEMM: random asymmetric refcheck (guide G9)."""
    A = randn(N, K, dtype=torch.bfloat16, scale=0.5, seed=N * 3 + M)
    Bt = randn(M, K, dtype=torch.bfloat16, scale=0.5)
    C = ext().gemm_bt_v(A, Bt, False, 2)
    ref = A.float() @ Bt.float().t()
    err = (C.float() - ref).abs().max().item()
    denom = ref.abs().max().item()
    assert err / denom < 0.02, 'max err %g vs scale %g' % (err, denom)
    Ct = ext().gemm_bt_v(A, Bt, True, 2)
    assert (Ct.float() - torch.tanh(ref)).abs().max().item() < 0.02


def test_adam_sparse_from_ctx_matches_rows_path():
    """ctx-direct scatter-Adam (reads grad rows straight from the (N,3d)
    activation layout) ≡ the materialized-rows path."""
    torch.manual_seed(21)
    Vr, dt, N = 96, 16, 64
    d_ctx = randn(N, 3 * dt, dtype=torch.bfloat16, seed=77)
    src = torch.randint(0, Vr, (N,), dtype=torch.int32).cuda()
    tgt = torch.randint(0, Vr, (N,), dtype=torch.int32).cuda()
    ids = torch.cat([src, tgt])
    p1 = randn(Vr, dt); p2 = p1.clone()
    m1 = torch.zeros_like(p1); v1 = torch.zeros_like(p1)
    m2 = m1.clone(); v2 = v1.clone()
    from code2vec_amd.ops import functional as F
    F.adam_sparse_rows_from_ctx(p1, ids, d_ctx, 0, 2 * dt, 2, dt, m1, v1,
                                1, 1e-3, 0.9, 0.999, 1e-8)
    rows = torch.cat([d_ctx[:, :dt], d_ctx[:, 2 * dt:]], 0).contiguous()
    F.adam_sparse_rows_step(p2, ids, rows, m2, v2, 1, 1e-3, 0.9, 0.999, 1e-8)
    assert (p1 - p2).abs().max().item() < 1e-6
    assert (m1 - m2).abs().max().item() < 1e-6
    # single-segment (path table) form
    pp1 = randn(Vr, dt, seed=5); pp2 = pp1.clone()
    mm1 = torch.zeros_like(pp1); vv1 = torch.zeros_like(pp1)
    mm2 = mm1.clone(); vv2 = vv1.clone()
    pids = torch.randint(0, Vr, (N,), dtype=torch.int32).cuda()
    F.adam_sparse_rows_from_ctx(pp1, pids, d_ctx, dt, dt, 1, dt, mm1, vv1,
                                1, 1e-3, 0.9, 0.999, 1e-8)
    F.adam_sparse_rows_step(pp2, pids, d_ctx[:, dt:2 * dt].contiguous(),
                            mm2, vv2, 1, 1e-3, 0.9, 0.999, 1e-8)
    assert (pp1 - pp2).abs().max().item() < 1e-6


def test_logits_ce_fused_matches_fp32_oracle():
    """Fused K8+K9 forward vs an fp32 matmul+logsumexp oracle. The fused lse
    comes from the fp32 MFMA accumulator (pre-bf16 rounding), so it must track
    the fp32 oracle tightly — NOT the bf16-rounded unfused path, whose lse
    differs by up to bf16-eps at logit scale (~0.06 at |logit|≈8)."""
    torch.manual_seed(31)
    B, D, V = 1024, 384, 261246
    code = randn(B, D, dtype=torch.bfloat16, scale=0.3, seed=31)
    shadow = randn(V, D, dtype=torch.bfloat16, scale=0.3)
    labels = torch.randint(0, V, (B,)).cuda()
    logits_f, loss_f, lse_f = ext().logits_ce_fused(code, shadow, labels)
    logits_u = ext().gemm_bt_v(code, shadow, False, 2)
    assert torch.equal(logits_f, logits_u)   # same bf16 logits out
    ref32 = code.float() @ shadow.float().t()
    lse32 = torch.logsumexp(ref32, dim=1)
    assert (lse_f - lse32).abs().max().item() < 2e-3
    # loss = fp32 lse − bf16-rounded label logit (the stored logits feed bwd)
    picked = logits_f.float().gather(1, labels.reshape(-1, 1)).squeeze(1)
    assert (loss_f - (lse32 - picked)).abs().max().item() < 3e-3


def test_attention_bwd_fused_tanh():
    from code2vec_amd.ops import functional as F
    B, C, D = 6, 33, 128
    comb = randn(B, C, D, dtype=torch.bfloat16, scale=0.7, seed=41)
    a = randn(D, scale=0.3)
    mask = (torch.rand(B, C) > 0.2).float().cuda()
    mask[:, 0] = 1.0
    _, alpha = R.attention_fwd(comb, a, mask)
    d_code = randn(B, D, scale=0.5)
    d_z, d_a = F.attention_bwd(comb, a, alpha, d_code, fuse_tanh_bwd=True)
    d_comb_ref, d_a_ref = R.attention_bwd(comb, a, alpha, d_code)
    d_z_ref = (d_comb_ref.float() * (1 - comb.float() ** 2))
    assert (d_z.float() - d_z_ref).abs().max().item() < 1e-2
    assert (d_a - d_a_ref).abs().max().item() < 1e-2


def test_gemm_bt_dropout_epilogue():
    from code2vec_amd.ops import functional as F
    N, K, M = 256, 64, 48
    dz = randn(N, K, dtype=torch.bfloat16, scale=0.5, seed=43)
    w = randn(M, K, dtype=torch.bfloat16, scale=0.5)
    ctx = randn(N, M, dtype=torch.bfloat16, scale=0.5)
    seed = 777
    d_ctx, d_w = F.linear_bwd_dropout(dz, w, ctx, 0.75, seed, training=True)
    plain = dz.float() @ w.float().t()
    mask = R.dropout_keep_mask(seed, N * M, 0.75, 'cuda').reshape(N, M)
    # dropped positions are exactly zero
    assert torch.all(d_ctx[~mask] == 0)
    # kept positions match plain/keep within bf16 noise of the value scale
    expected = plain / 0.75
    err = (d_ctx.float() - expected)[mask].abs().max().item()
    assert err < 0.02 * expected.abs().max().item(), err


@pytest.mark.parametrize('N,M,K', [
    (1024, 384, 261246),   # d_code shape: K = java14m target vocab (K%32!=0)
    (800, 384, 65536),     # row-guard path (N not a multiple of 256)
    (256, 192, 70030),     # narrow M (zero-padded cols) + K tail
])
def test_gemm_nn_splitk_vs_matmul(N, M, K):
    """split-K nn GEMM (d_code): random asymmetric refcheck (guide G9).
    fp32 split-K partial sums reassociate vs torch's fp32 matmul, so the
    check is relative to the output scale."""
    A = randn(N, K, dtype=torch.bfloat16, scale=0.1, seed=N + M)
    B = randn(K, M, dtype=torch.bfloat16, scale=0.1)
    C = ext().gemm_nn_splitk(A, B)
    assert C.dtype == torch.float32 and C.shape == (N, M)
    ref = A.float() @ B.float()
    err = (C - ref).abs().max().item()
    denom = ref.abs().max().item()
    assert err / denom < 0.01, 'max err %g vs scale %g' % (err, denom)


def test_logits_bwd_code_dispatch():
    """functional.logits_bwd_code routes to the split-K kernel on GPU and
    matches the eager matmul it replaces (network.py d_code)."""
    from code2vec_amd.ops import functional as F
    torch.manual_seed(7)
    d_logits = randn(256, 8192, dtype=torch.bfloat16, scale=0.01)
    shadow = randn(8192, 384, dtype=torch.bfloat16, scale=0.5)
    out = F.logits_bwd_code(d_logits, shadow)
    ref = (d_logits.float() @ shadow.float())
    assert out.dtype == torch.float32
    err = (out - ref).abs().max().item() / ref.abs().max().item()
    assert err < 0.01, err


@pytest.mark.parametrize('K,V,M', [
    (1024, 261246, 384),   # d_target shape (V tail: 261246 % 128 != 0)
    (1000, 65536, 384),    # ragged batch (K % 32 != 0)
    (256, 8200, 192),      # narrow M + V tail
])
def test_gemm_tn_vs_matmul(K, V, M):
    """tn GEMM (d_target): random asymmetric refcheck (guide G9)."""
    A = randn(K, V, dtype=torch.bfloat16, scale=0.1, seed=K + V)
    B = randn(K, M, dtype=torch.bfloat16, scale=0.1)
    C = ext().gemm_tn_bf16(A, B)
    assert C.dtype == torch.bfloat16 and C.shape == (V, M)
    ref = A.float().t() @ B.float()
    err = (C.float() - ref).abs().max().item()
    denom = ref.abs().max().item()
    assert err / denom < 0.02, 'max err %g vs scale %g' % (err, denom)


def test_logits_bwd_target_dispatch():
    from code2vec_amd.ops import functional as F
    torch.manual_seed(9)
    d_logits = randn(512, 8192, dtype=torch.bfloat16, scale=0.01)
    code_c = randn(512, 384, dtype=torch.bfloat16, scale=0.5)
    out = F.logits_bwd_target(d_logits, code_c)
    ref = d_logits.float().t() @ code_c.float()
    err = (out.float() - ref).abs().max().item() / ref.abs().max().item()
    assert err < 0.02, err


def test_ce_fused_consumers_match_materialized():
    """gemm_nn_splitk_ce / gemm_tn_ce vs k_ce_bwd + plain GEMMs: the fused
    staging computes bit-identical bf16 d_logit values, so the outputs must
    match the materialized two-kernel compositions near-exactly."""
    e = ext()
    torch.manual_seed(11)
    B, V, D = 512, 70030, 384
    code = randn(B, D, dtype=torch.bfloat16, scale=0.3)
    shadow = randn(V, D, dtype=torch.bfloat16, scale=0.3)
    labels = torch.randint(0, V, (B,), device='cuda')
    logits, _, lse = e.logits_ce_fused(code, shadow, labels)
    scale = 1.0 / B

    d_logits = e.ce_bwd(logits, lse, labels, scale)
    ref_code = e.gemm_nn_splitk(d_logits, shadow)
    fused_code = e.gemm_nn_splitk_ce(logits, shadow, lse, labels, scale)
    err = (fused_code - ref_code).abs().max().item()
    dn = ref_code.abs().max().item()
    assert err / dn < 1e-3, (err, dn)

    ref_tgt = e.gemm_tn_bf16(d_logits, code)
    fused_tgt = e.gemm_tn_ce(logits, code, lse, labels, scale)
    err = (fused_tgt.float() - ref_tgt.float()).abs().max().item()
    dn = ref_tgt.float().abs().max().item()
    assert err / dn < 1e-3, (err, dn)


def test_ce_fused_consumers_vs_fp32_oracle():
    """End math check: fused d_code/d_target vs an fp32 softmax-grad oracle."""
    e = ext()
    torch.manual_seed(12)
    B, V, D = 256, 8192, 384
    code = randn(B, D, dtype=torch.bfloat16, scale=0.3)
    shadow = randn(V, D, dtype=torch.bfloat16, scale=0.3)
    labels = torch.randint(0, V, (B,), device='cuda')
    logits, _, lse = e.logits_ce_fused(code, shadow, labels)
    scale = 1.0 / B
    lf = logits.float()
    dl = (torch.softmax(lf, dim=1)
          - torch.nn.functional.one_hot(labels, V).float()) * scale
    ref_code = dl @ shadow.float()
    ref_tgt = dl.t() @ code.float()
    fused_code = e.gemm_nn_splitk_ce(logits, shadow, lse, labels, scale)
    fused_tgt = e.gemm_tn_ce(logits, code, lse, labels, scale).float()
    assert (fused_code - ref_code).abs().max().item() \
        / ref_code.abs().max().item() < 0.02
    assert (fused_tgt - ref_tgt).abs().max().item() \
        / ref_tgt.abs().max().item() < 0.02


@pytest.mark.parametrize('K,N2,M', [
    (204800, 384, 384),    # dW shape (K = B*C)
    (65560, 384, 384),     # K % 32 != 0 tail
    (131072, 128, 192),    # narrow output
])
def test_gemm_tn_splitk_vs_matmul(K, N2, M):
    """split-K tn GEMM (dW): random asymmetric refcheck (guide G9)."""
    A = randn(K, N2, dtype=torch.bfloat16, scale=0.1, seed=K % 977)
    B = randn(K, M, dtype=torch.bfloat16, scale=0.1)
    C = ext().gemm_tn_splitk(A, B)
    assert C.dtype == torch.float32 and C.shape == (N2, M)
    ref = A.float().t() @ B.float()
    err = (C - ref).abs().max().item()
    dn = ref.abs().max().item()
    assert err / dn < 0.01, 'max err %g vs scale %g' % (err, dn)


def test_ce_fused_write_matches_ce_bwd():
    """gemm_nn_splitk_ce_write: the streamed-out d_logits must be
    bit-identical to k_ce_bwd's output, and d_code must match the
    materialized composition."""
    e = ext()
    torch.manual_seed(13)
    B, V, D = 512, 70030, 384
    code = randn(B, D, dtype=torch.bfloat16, scale=0.3)
    shadow = randn(V, D, dtype=torch.bfloat16, scale=0.3)
    labels = torch.randint(0, V, (B,), device='cuda')
    logits, _, lse = e.logits_ce_fused(code, shadow, labels)
    scale = 1.0 / B
    d_logits_ref = e.ce_bwd(logits, lse, labels, scale)
    d_code, d_logits = e.gemm_nn_splitk_ce_write(logits, shadow, lse,
                                                 labels, scale)
    assert torch.equal(d_logits, d_logits_ref), 'd_logits not bit-identical'
    ref_code = e.gemm_nn_splitk(d_logits_ref, shadow)
    err = (d_code - ref_code).abs().max().item() / ref_code.abs().max().item()
    assert err < 1e-3, err


@pytest.mark.parametrize('N,M,K', [(1024, 384, 384), (500, 200, 128)])
def test_gemm_bt2_vs_matmul(N, M, K):
    """double-buffered BK=64 128-tile GEMM (variant 3) refcheck."""
    A = randn(N, K, dtype=torch.bfloat16, scale=0.3, seed=N + M + K)
    Bt = randn(M, K, dtype=torch.bfloat16, scale=0.3)
    C = ext().gemm_bt_v(A, Bt, False, 3)
    ref = A.float() @ Bt.float().t()
    err = (C.float() - ref).abs().max().item() / ref.abs().max().item()
    assert err < 0.02, err
    Ct = ext().gemm_bt_v(A, Bt, True, 3)
    assert (Ct.float() - torch.tanh(ref)).abs().max().item() < 0.02


def test_adam_sparse_heavy_duplicates_vs_reference():
    """Count-aware accumulation (single-occurrence fast path + atomics for
    the rest) must match the fp32 lazy-Adam reference when ids repeat
    heavily and non-contiguously (the pattern that defeats run-length
    claim dedup)."""
    e = ext()
    torch.manual_seed(21)
    Vr, d, n = 3000, 64, 4096
    ids = torch.randint(1, Vr, (n,), device='cuda')
    ids[::3] = 7          # hot id as many non-contiguous single runs
    ids[1::5] = 7
    rows = (torch.randn(n, d, device='cuda') * 0.1).to(torch.bfloat16)
    p = torch.randn(Vr, d, device='cuda')
    m = torch.zeros_like(p)
    v = torch.zeros_like(p)
    p0, m0, v0 = p.cpu().clone(), m.cpu().clone(), v.cpu().clone()

    e.adam_sparse_rows_hash(p, ids, rows, m, v, 1, 1e-3, 0.9, 0.999, 1e-8,
                            torch.empty(0), torch.empty(0))

    from code2vec_amd.ops import reference as RR
    RR.adam_sparse_rows_step(p0, ids.cpu(), rows.cpu().float(), m0, v0,
                             1, 1e-3, 0.9, 0.999, 1e-8)
    # float-atomic accumulation order differs from the reference sum
    assert (p.cpu() - p0).abs().max().item() < 2e-5
    assert (m.cpu() - m0).abs().max().item() < 2e-5


def test_sparse_dedup_sum_matches_reference():
    """Rank-local dedup+sum (DP wire-volume reduction): unique set and
    per-id sums must match the torch.unique+index_add reference, for both
    the materialized-rows and the d_ctx-slicing entry points."""
    e = ext()
    torch.manual_seed(33)
    Vr, d, n = 500, 32, 2048
    ids = torch.randint(0, Vr, (n,), dtype=torch.int32, device='cuda')
    ids[::4] = 3          # hot id, non-contiguous
    rows = (torch.randn(n, d, device='cuda') * 0.1).to(torch.bfloat16)
    uniq, acc, cnt = e.sparse_dedup_sum_rows(ids, rows)
    c = int(cnt.item())
    ref_u, ref_acc, ref_c = R.sparse_dedup_sum(ids.cpu(), rows.cpu().float())
    assert c == ref_c
    order = torch.argsort(uniq[:c].cpu())
    assert torch.equal(uniq[:c].cpu()[order], ref_u)
    assert (acc[:c].cpu()[order] - ref_acc).abs().max().item() < 2e-2

    # d_ctx two-segment layout (token grads: src cols [0,d), tgt cols [2d,3d))
    N = 1024
    d_ctx = (torch.randn(N, 3 * d, device='cuda') * 0.1).to(torch.bfloat16)
    tok_ids = torch.randint(0, Vr, (2 * N,), dtype=torch.int32, device='cuda')
    uniq2, acc2, cnt2 = e.sparse_dedup_sum_ctx(tok_ids, d_ctx, 0, 2 * d, 2, d)
    c2 = int(cnt2.item())
    rows2 = torch.cat([d_ctx[:, :d], d_ctx[:, 2 * d:]], 0).cpu().float()
    ref_u2, ref_acc2, ref_c2 = R.sparse_dedup_sum(tok_ids.cpu(), rows2)
    assert c2 == ref_c2
    order2 = torch.argsort(uniq2[:c2].cpu())
    assert torch.equal(uniq2[:c2].cpu()[order2], ref_u2)
    assert (acc2[:c2].cpu()[order2] - ref_acc2).abs().max().item() < 2e-2


def test_gather_gemms_vs_reference():
    """Sampled-softmax gather GEMMs: the candidate-row gather fused into the
    bt (logits) and split-K nn (d_code) staging must match index_select +
    plain GEMM."""
    e = ext()
    torch.manual_seed(44)
    N, K, V, S = 300, 384, 5000, 1000
    A = randn(N, K, dtype=torch.bfloat16, scale=0.2)
    table = randn(V, K, dtype=torch.bfloat16, scale=0.2)
    idx = torch.randint(0, V, (S,), dtype=torch.int64).cuda()
    C = e.gemm_bt_gather(A, table, idx)
    ref = A.float() @ table.float()[idx].t()
    assert C.shape == (N, S)
    err = (C.float() - ref).abs().max().item() / ref.abs().max().item()
    assert err < 0.02, err

    d = randn(N, S, dtype=torch.bfloat16, scale=0.2)
    C2 = e.gemm_nn_splitk_gather(d, table, idx)
    ref2 = d.float() @ table.float()[idx]
    assert C2.dtype == torch.float32 and C2.shape == (N, K)
    err2 = (C2 - ref2).abs().max().item() / ref2.abs().max().item()
    assert err2 < 0.02, err2


def test_sampled_train_step_gpu_vs_cpu():
    """Sampled-softmax training path end to end on GPU (fused gather GEMMs +
    sampled CE kernels) vs the fp32 CPU engine. The negative sample draw is
    seeded identically on both engines."""
    from code2vec_amd.config import Config
    from code2vec_amd.models.network import Code2VecNetwork

    def build(device, dtype):
        cfg = Config(set_defaults=True)
        cfg.TRAIN_DATA_PATH_PREFIX = 'unused'
        cfg.MAX_CONTEXTS = 20
        cfg.TOKEN_EMBEDDINGS_SIZE = 64
        cfg.PATH_EMBEDDINGS_SIZE = 64
        cfg.CODE_VECTOR_SIZE = 192
        cfg.TARGET_EMBEDDINGS_SIZE = 192
        cfg.DROPOUT_KEEP_RATE = 1.0
        cfg.COMPUTE_DTYPE = dtype
        cfg.SAMPLED_SOFTMAX_SIZE = 96
        torch.manual_seed(7)
        return Code2VecNetwork(cfg, 500, 300, 600, device=device)

    net_gpu = build('cuda:0', 'bf16')
    net_cpu = build('cpu', 'fp32')
    g = torch.Generator().manual_seed(3)
    B, C = 16, 20
    src = torch.randint(0, 500, (B, C), generator=g, dtype=torch.int32)
    pth = torch.randint(0, 300, (B, C), generator=g, dtype=torch.int32)
    tgt = torch.randint(0, 500, (B, C), generator=g, dtype=torch.int32)
    mask = torch.ones(B, C)
    labels = torch.randint(1, 600, (B,), generator=g)
    # pin the negative draw so both engines see identical candidates
    # (cuda and cpu RNGs produce different sequences for the same seed)
    import code2vec_amd.ops.reference as RR2
    fixed = torch.randint(0, 600, (96,), generator=g)
    orig_draw = RR2.sample_log_uniform
    RR2.sample_log_uniform =         lambda n, V, device, generator=None: fixed.to(device)
    try:
        for step in range(4):
            lg = float(net_gpu.train_step(src.cuda(), pth.cuda(), tgt.cuda(),
                                          mask.cuda(), labels.cuda()))
            lc = float(net_cpu.train_step(src, pth, tgt, mask, labels))
            assert abs(lg - lc) < 0.06 * max(1.0, abs(lc)), (step, lg, lc)
    finally:
        RR2.sample_log_uniform = orig_draw


def test_sparse_hash_build_pre_matches_inline():
    """The phase-split dedup (sparse_hash_build + sparse_dedup_sum_ctx_pre,
    the DP path under the forward-overlap scheme) must produce the same
    (unique ids, summed rows) as the single-call sparse_dedup_sum_ctx."""
    e = ext()
    torch.manual_seed(77)
    N, d = 512, 32
    d_ctx = (torch.randn(N, 3 * d, device='cuda') * 0.1).to(torch.bfloat16)
    ids = torch.randint(0, 300, (2 * N,), dtype=torch.int32, device='cuda')
    ids[::5] = 7
    u1, acc1, c1 = e.sparse_dedup_sum_ctx(ids, d_ctx, 0, 2 * d, 2, d)
    state = e.sparse_hash_build(ids)
    u2, acc2, c2 = e.sparse_dedup_sum_ctx_pre(state[0], state[1], state[2],
                                              state[3], state[4], d_ctx,
                                              0, 2 * d, 2, d)
    n1, n2 = int(c1.item()), int(c2.item())
    assert n1 == n2
    o1 = torch.argsort(u1[:n1])
    o2 = torch.argsort(u2[:n2])
    assert torch.equal(u1[:n1][o1], u2[:n2][o2])
    assert (acc1[:n1][o1] - acc2[:n2][o2]).abs().max().item() < 1e-4
