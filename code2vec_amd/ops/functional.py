"""Backend dispatch: each engine op routes to the HIP extension on GPU and to
the pure-torch reference on CPU. The HIP path is mandatory on GPU (see
ops/__init__.backend_for) — no silent eager fallback."""

import math
from typing import Tuple

import torch

from . import backend_for, hip_ext
from . import reference as ref


def gather_concat_fwd(tok_table, path_table, src_ids, path_ids, tgt_ids,
                      keep_prob: float, seed: int, training: bool,
                      out_dtype=torch.bfloat16, seed_t=None) -> torch.Tensor:
    """seed_t: optional 1-elem int64 device tensor overriding `seed` — used
    under hipGraph capture so each replay reads a fresh seed."""
    if backend_for(tok_table) == 'hip':
        assert out_dtype == torch.bfloat16, "HIP gather_concat emits bf16"
        return hip_ext(True).gather_concat_fwd(
            tok_table, path_table, src_ids, path_ids, tgt_ids,
            float(keep_prob), int(seed), bool(training),
            seed_t if seed_t is not None else torch.empty(0))
    if seed_t is not None:
        seed = int(seed_t.item())
    return ref.gather_concat_fwd(tok_table, path_table, src_ids, path_ids,
                                 tgt_ids, keep_prob, seed, training,
                                 out_dtype=out_dtype)


def gather_concat_bwd(d_ctx, keep_prob: float, seed: int, training: bool,
                      seed_t=None) -> torch.Tensor:
    if backend_for(d_ctx) == 'hip':
        return hip_ext(True).gather_concat_bwd(
            d_ctx, float(keep_prob), int(seed), bool(training),
            seed_t if seed_t is not None else torch.empty(0))
    if seed_t is not None:
        seed = int(seed_t.item())
    return ref.gather_concat_bwd(d_ctx, keep_prob, seed, training)


def transform_tanh_fwd(ctx, w_oi_bf16) -> torch.Tensor:
    """ctx (N,K) bf16 × W — the HIP kernel takes the (out,in) bf16 shadow
    (row-major in K, the MFMA B^T layout); the reference path takes the same
    and transposes internally."""
    if backend_for(ctx) == 'hip':
        return hip_ext(True).transform_tanh_fwd(ctx, w_oi_bf16)
    return ref.transform_tanh_fwd(ctx, w_oi_bf16.t())


def transform_tanh_bwd(ctx, w_io_bf16, y, d_y) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (d_ctx, d_w (in,out) fp32). The GEMM for d_ctx = d_z·W^T runs on
    the MFMA kernel with the (in,out) shadow as its B^T operand; d_w runs on
    hipBLASLt (plain library GEMM)."""
    if backend_for(ctx) == 'hip':
        e = hip_ext(True)
        d_z = e.tanh_bwd_mul(d_y, y)          # d_z = d_y * (1 - y^2)
        d_ctx = e.gemm_bt_bf16(d_z, w_io_bf16)
        return d_ctx, grad_weight_gemm(ctx, d_z)
    d_zr = d_y * (1.0 - y.float() ** 2).to(d_y.dtype)
    d_ctx = d_zr @ w_io_bf16.to(d_zr.dtype).t()
    d_w = (ctx.t().float() @ d_zr.float())
    return d_ctx, d_w


def grad_weight_gemm(ctx, d_z):
    """d_w = ctx^T @ d_z (fp32 out). On GPU with the training-sized K
    (batch*contexts ~ 205K rows) this runs the split-K tn MFMA kernel
    (k_gemm_tn<false,true>: fp32 partials per XCD-grouped k-chunk +
    reduce); elsewhere hipBLASLt / plain matmul."""
    import os as _os
    if (backend_for(ctx) == 'hip' and ctx.dtype == torch.bfloat16
            and d_z.dtype == torch.bfloat16
            and ctx.shape[1] <= 384 and ctx.shape[1] % 8 == 0
            and d_z.shape[1] <= 384 and d_z.shape[1] % 8 == 0
            and ctx.shape[0] >= 65536
            and _os.environ.get('C2V_DW_GEMM', '1') == '1'):
        return hip_ext(True).gemm_tn_splitk(ctx, d_z)
    return (ctx.t().to(d_z.dtype) @ d_z).float()


def attention_fwd(comb, a, valid_mask) -> Tuple[torch.Tensor, torch.Tensor]:
    if backend_for(comb) == 'hip':
        return hip_ext(True).attention_fwd(comb, a, valid_mask)
    return ref.attention_fwd(comb, a, valid_mask)


def attention_bwd(comb, a, alpha, d_code,
                  fuse_tanh_bwd: bool = False) -> Tuple[torch.Tensor, torch.Tensor]:
    """With fuse_tanh_bwd, returns dL/dz (z = pre-tanh transform output)
    instead of dL/dcomb — the tanh' factor is applied inside the kernel's
    d_comb write where comb is already loaded."""
    if backend_for(comb) == 'hip':
        return hip_ext(True).attention_bwd(comb, a, alpha, d_code,
                                           bool(fuse_tanh_bwd))
    d_comb, d_a = ref.attention_bwd(comb, a, alpha, d_code)
    if fuse_tanh_bwd:
        d_comb = (d_comb.float() * (1.0 - comb.float() ** 2)).to(d_comb.dtype)
    return d_comb, d_a


def linear_bwd_dropout(d_z, w_io, ctx, keep_prob: float, seed: int,
                       seed_t=None,
                       training: bool = True) -> Tuple[torch.Tensor, torch.Tensor]:
    """Backward of [dropout -> Linear]: d_ctx = dropout_mask ⊙ (d_z @ W^T)
    with the mask fused into the GEMM epilogue on HIP; d_w = ctx^T @ d_z via
    hipBLASLt. (The tanh' factor is expected to already be in d_z — see
    attention_bwd(fuse_tanh_bwd=True).)"""
    drop = training and keep_prob < 1.0
    if backend_for(d_z) == 'hip':
        e = hip_ext(True)
        if drop:
            d_ctx = e.gemm_bt_dropout(d_z, w_io, float(keep_prob), int(seed),
                                      seed_t if seed_t is not None
                                      else torch.empty(0))
        else:
            d_ctx = e.gemm_bt_bf16(d_z, w_io)
        return d_ctx, grad_weight_gemm(ctx, d_z)
    d_ctx = d_z @ w_io.to(d_z.dtype).t()
    if drop:
        if seed_t is not None:
            seed = int(seed_t.item())
        d_ctx = ref.gather_concat_bwd(d_ctx, keep_prob, seed, True)
    d_w = (ctx.t().float() @ d_z.float())
    return d_ctx, d_w


def ce_fwd(logits, labels) -> Tuple[torch.Tensor, torch.Tensor]:
    if backend_for(logits) == 'hip':
        return hip_ext(True).ce_fwd(logits, labels)
    return ref.ce_fwd(logits, labels)


def ce_bwd(logits, lse, labels, scale: float) -> torch.Tensor:
    if backend_for(logits) == 'hip':
        return hip_ext(True).ce_bwd(logits, lse, labels, float(scale))
    return ref.ce_bwd(logits, lse, labels, scale)


def logits_gemm(code_c, shadow):
    """logits = code @ shadow^T. On GPU with java14m-like shapes this runs the
    256x256-tile MFMA kernel (column-major XCD-chunked grid streams the big
    shadow panel once per XCD); falls back to hipBLASLt elsewhere."""
    if (backend_for(code_c) == 'hip' and code_c.dtype == torch.bfloat16
            and code_c.shape[1] % 64 == 0 and code_c.shape[0] >= 256
            and shadow.shape[0] >= 4096):
        return hip_ext(True).gemm_bt_v(code_c, shadow, False, 2)
    return code_c @ shadow.t()


def logits_bwd_code(d_logits, shadow):
    """d_code = d_logits @ shadow (nn operands, fp32 out). On GPU with the
    vocab-sized K this runs the split-K MFMA kernel (k_gemm_nn_splitk:
    rotated row-major LDS tiles, XCD-grouped k-chunks, fp32 partials +
    reduce — 409 us vs tuned hipBLASLt's 469 us); elsewhere a plain matmul."""
    import os as _os
    if (backend_for(d_logits) == 'hip' and d_logits.dtype == torch.bfloat16
            and shadow.dtype == torch.bfloat16
            and shadow.shape[1] <= 384 and shadow.shape[1] % 8 == 0
            and shadow.shape[0] >= 4096
            and _os.environ.get('C2V_NN_GEMM', '1') == '1'):
        return hip_ext(True).gemm_nn_splitk(d_logits, shadow)
    return (d_logits @ shadow).float()


def logits_bwd_target(d_logits, code_c):
    """d_target = d_logits^T @ code (tn operands, bf16 out). C2V_TN_GEMM=1
    opts into the custom per-V-tile MFMA kernel (k_gemm_tn) — measured
    527 us vs hipBLASLt's 333 us on the java14m shape (K=batch is too small
    to amortize the staged-tile overheads at 1 wave/SIMD), so the library
    GEMM is the default; the kernel is kept as the starting point for a
    multi-K-tile revision."""
    import os as _os
    if (backend_for(d_logits) == 'hip' and d_logits.dtype == torch.bfloat16
            and code_c.dtype == torch.bfloat16
            and code_c.shape[1] <= 384 and code_c.shape[1] % 8 == 0
            and d_logits.shape[1] >= 4096
            and _os.environ.get('C2V_TN_GEMM', '0') == '1'):
        return hip_ext(True).gemm_tn_bf16(d_logits, code_c)
    return d_logits.t() @ code_c


def ce_bwd_mode(logits):
    """CE-backward fusion mode (C2V_FUSED_CEBWD). Default '0'
    (materialized k_ce_bwd + plain GEMMs) — BOTH fusion modes measured
    slower in-bench despite moving less HBM traffic, because the nn d_code
    kernel is latency-bound at 1 wave/SIMD and the fused exp()/store chain
    sits on its staging critical path (mode 2: 282K vs 302K ex/s; mode 1:
    252K — see profiles/r01_optimization_log.md):
    '2': gemm_nn_splitk_ce_write — CE backward fused into the d_code
        GEMM's staging, which also streams d_logits out (bijective tile
        coverage, bit-identical to k_ce_bwd); d_target keeps hipBLASLt.
    '1': fully virtual d_logits (also forces the tn d_target kernel)."""
    import os as _os
    if not (backend_for(logits) == 'hip' and logits.dtype == torch.bfloat16
            and logits.shape[1] >= 4096):
        return 0
    return int(_os.environ.get('C2V_FUSED_CEBWD', '0'))


def ce_bwd_fused_available(logits):
    return ce_bwd_mode(logits) == 1


def logits_bwd_code_ce(logits, shadow, lse, labels, scale):
    """d_code = ce_bwd(logits, lse, labels, scale) @ shadow with the CE
    backward computed inside the GEMM's A staging (bit-identical bf16
    d_logit values to k_ce_bwd; the 535 MB d_logits tensor is skipped)."""
    if ce_bwd_fused_available(logits) and shadow.shape[1] <= 384 \
            and shadow.shape[1] % 8 == 0:
        return hip_ext(True).gemm_nn_splitk_ce(logits, shadow, lse, labels,
                                               float(scale))
    d_logits = ce_bwd(logits, lse, labels, scale)
    return logits_bwd_code(d_logits, shadow)


def logits_bwd_code_ce_write(logits, shadow, lse, labels, scale):
    """(d_code fp32, d_logits bf16) with CE backward fused into the GEMM
    staging and d_logits streamed out as a byproduct (mode '2')."""
    if ce_bwd_mode(logits) != 0 and shadow.shape[1] <= 384 \
            and shadow.shape[1] % 8 == 0:
        out = hip_ext(True).gemm_nn_splitk_ce_write(logits, shadow, lse,
                                                    labels, float(scale))
        return out[0], out[1]
    d_logits = ce_bwd(logits, lse, labels, scale)
    return logits_bwd_code(d_logits, shadow), d_logits


def logits_bwd_target_ce(logits, code_c, lse, labels, scale):
    """d_target = ce_bwd(logits, lse, labels, scale)^T @ code, CE backward
    fused into the tn GEMM's staging (no d_logits materialization)."""
    if ce_bwd_fused_available(logits) and code_c.shape[1] <= 384 \
            and code_c.shape[1] % 8 == 0:
        return hip_ext(True).gemm_tn_ce(logits, code_c, lse, labels,
                                        float(scale))
    d_logits = ce_bwd(logits, lse, labels, scale)
    return logits_bwd_target(d_logits, code_c)


def logits_ce_fused(code_c, shadow, labels):
    """Fused K8+K9 forward: the 256-tile logits GEMM also emits per-tile
    (rowmax, sumexp) partials, folded into (loss, lse) by a small reduce —
    no separate full-vocab CE pass. Returns (logits, loss_rows, lse).
    The lse comes from the fp32 accumulator (pre-bf16-rounding), which is
    closer to the reference's fp32 TF math than the unfused bf16 path."""
    import os as _os
    if (backend_for(code_c) == 'hip' and code_c.dtype == torch.bfloat16
            and code_c.shape[1] % 64 == 0 and code_c.shape[0] >= 256
            and shadow.shape[0] >= 4096
            and _os.environ.get('C2V_FUSED_CE', '1') == '1'):
        logits, loss, lse = hip_ext(True).logits_ce_fused(code_c, shadow, labels)
        return logits, loss, lse
    logits = logits_gemm(code_c, shadow)
    loss, lse = ce_fwd(logits, labels)
    return logits, loss, lse


def topk(logits, k: int):
    """Per-row top-k (values fp32 desc, tie → lower index; int64 indices)."""
    if backend_for(logits) == 'hip' and logits.dtype == torch.bfloat16 and k <= 32:
        vals, idx = hip_ext(True).topk(logits, int(k))
        return vals, idx
    vals, idx = torch.topk(logits.float(), k=k, dim=1)
    return vals, idx


def adam_lrt(step_t, lr: float, beta1: float, beta2: float):
    """Bias-corrected lr_t = lr*sqrt(1-b2^t)/(1-b1^t) from the device step
    counter, as ONE kernel (the torch scalar chain is ~7 launches inside
    the captured graph). Falls back to the torch chain off-GPU."""
    if backend_for(step_t) == 'hip':
        return hip_ext(True).adam_lrt(step_t, float(lr), float(beta1),
                                      float(beta2))
    t = step_t.to(torch.float32)
    return (lr * torch.sqrt(1.0 - torch.exp(t * math.log(beta2)))
            / (1.0 - torch.exp(t * math.log(beta1)))).reshape(1)


def adam_dense_step(p, g, m, v, step: int, lr: float, beta1: float,
                    beta2: float, eps: float, shadow=None, lrt_t=None):
    """lrt_t: optional 1-elem fp32 device tensor with the precomputed
    bias-corrected lr_t (used under hipGraph capture)."""
    if backend_for(p) == 'hip':
        hip_ext(True).adam_dense_step(
            p, g, m, v, int(step), float(lr), float(beta1), float(beta2),
            float(eps), shadow if shadow is not None else torch.empty(0),
            lrt_t if lrt_t is not None else torch.empty(0))
        return
    ref.adam_dense_step(p, g, m, v, step, lr, beta1, beta2, eps, shadow)


def adam_sparse_rows_step(p, ids, grad_rows, m, v, step: int, lr: float,
                          beta1: float, beta2: float, eps: float, shadow=None,
                          lrt_t=None):
    if backend_for(p) == 'hip':
        # hash-based dedup + accumulate + lazy row update, no sort / no sync
        hip_ext(True).adam_sparse_rows_hash(
            p, ids, grad_rows, m, v, int(step), float(lr), float(beta1),
            float(beta2), float(eps),
            shadow if shadow is not None else torch.empty(0),
            lrt_t if lrt_t is not None else torch.empty(0))
        return
    ref.adam_sparse_rows_step(p, ids, grad_rows, m, v, step, lr, beta1, beta2,
                              eps, shadow)


def adam_sparse_rows_from_ctx(p, ids, d_ctx, off0, off1, n_seg, d, m, v,
                              step: int, lr: float, beta1: float,
                              beta2: float, eps: float, lrt_t=None):
    """Sparse-row Adam reading grad rows directly from the (N,3d) d_ctx
    layout (no cat/contiguous materialization). Single-process path; DP uses
    the gathered-rows variant."""
    if backend_for(p) == 'hip':
        hip_ext(True).adam_sparse_rows_hash_ctx(
            p, ids, d_ctx, int(off0), int(off1), int(n_seg), int(d), m, v,
            int(step), float(lr), float(beta1), float(beta2), float(eps),
            lrt_t if lrt_t is not None else torch.empty(0))
        return
    n = d_ctx.shape[0]
    if n_seg == 2:
        rows = torch.cat([d_ctx[:, off0:off0 + d], d_ctx[:, off1:off1 + d]], 0)
    else:
        rows = d_ctx[:, off0:off0 + d]
    ref.adam_sparse_rows_step(p, ids, rows, m, v, step, lr, beta1, beta2, eps)


def sparse_hash_build(ids):
    """Prebuild the sparse-grad dedup hash state from the ids alone (they
    are known at step START, so this launches on a side stream under the
    forward pass — the claim/compact/lookup passes leave the backward
    tail). Returns an opaque state tuple on GPU; None on CPU (the eager
    reference path recomputes its dedup inline, deterministically)."""
    if backend_for(ids) == 'hip':
        return tuple(hip_ext(True).sparse_hash_build(ids))
    return None


def adam_sparse_rows_from_ctx_pre(p, state, d_ctx, off0, off1, n_seg, d,
                                  m, v, step: int, lr: float, beta1: float,
                                  beta2: float, eps: float, lrt_t=None):
    """adam_sparse_rows_from_ctx consuming a prebuilt hash state."""
    hip_ext(True).adam_sparse_rows_hash_ctx_pre(
        p, state[0], state[1], state[2], state[3], state[4], d_ctx,
        int(off0), int(off1), int(n_seg), int(d), m, v, int(step), float(lr),
        float(beta1), float(beta2), float(eps),
        lrt_t if lrt_t is not None else torch.empty(0))


def sparse_dedup_sum_ctx_pre(state, d_ctx, off0: int, off1: int, n_seg: int,
                             d: int):
    """sparse_dedup_sum_ctx consuming a prebuilt hash state (GPU only)."""
    uniq, acc, n_uniq = hip_ext(True).sparse_dedup_sum_ctx_pre(
        state[0], state[1], state[2], state[3], state[4], d_ctx,
        int(off0), int(off1), int(n_seg), int(d))
    return uniq, acc, n_uniq


def sparse_dedup_sum_ctx(ids, d_ctx, off0: int, off1: int, n_seg: int, d: int):
    """Rank-local dedup+sum of embedding grad rows read straight from the
    (N,3d) d_ctx layout (DP wire-volume reduction — SURVEY §2.4). Returns
    (uniq_ids int64[capacity], acc fp32[capacity,d], count) where count is a
    1-elem int32 device tensor on GPU and a plain int on CPU; only the first
    `count` rows of the outputs are live."""
    if backend_for(d_ctx) == 'hip':
        uniq, acc, n_uniq = hip_ext(True).sparse_dedup_sum_ctx(
            ids, d_ctx, int(off0), int(off1), int(n_seg), int(d))
        return uniq, acc, n_uniq
    if n_seg == 2:
        rows = torch.cat([d_ctx[:, off0:off0 + d], d_ctx[:, off1:off1 + d]], 0)
    else:
        rows = d_ctx[:, off0:off0 + d]
    return ref.sparse_dedup_sum(ids, rows)


def sparse_dedup_sum(ids, rows):
    """As sparse_dedup_sum_ctx but for materialized (n,d) grad rows."""
    if backend_for(rows) == 'hip':
        uniq, acc, n_uniq = hip_ext(True).sparse_dedup_sum_rows(ids, rows)
        return uniq, acc, n_uniq
    return ref.sparse_dedup_sum(ids, rows)


def sampled_logits_gemm(code_c, shadow, cand):
    """Candidate logits for the sampled-softmax path: code @ gather(targets,
    cand)^T with the row gather fused into the GEMM's B staging (BASELINE
    config 4 deliverable: no torch GEMM and no w_cand materialization)."""
    if (backend_for(code_c) == 'hip' and code_c.dtype == torch.bfloat16
            and shadow.dtype == torch.bfloat16
            and code_c.shape[1] % 32 == 0):
        return hip_ext(True).gemm_bt_gather(code_c, shadow, cand)
    return code_c @ shadow.index_select(0, cand).t()


def sampled_bwd_code(d_cand, shadow, cand):
    """d_code = d_cand @ gather(targets, cand) (fp32 out), gather fused into
    the split-K nn GEMM's B staging."""
    if (backend_for(d_cand) == 'hip' and d_cand.dtype == torch.bfloat16
            and shadow.dtype == torch.bfloat16
            and shadow.shape[1] <= 384 and shadow.shape[1] % 8 == 0):
        return hip_ext(True).gemm_nn_splitk_gather(d_cand, shadow, cand)
    return (d_cand @ shadow.index_select(0, cand)).float()


def sampled_bwd_target_rows(d_cand, code_c):
    """Per-candidate target-table grad rows: d_cand^T @ code (tn MFMA)."""
    if (backend_for(d_cand) == 'hip' and d_cand.dtype == torch.bfloat16
            and code_c.dtype == torch.bfloat16
            and code_c.shape[1] <= 384 and code_c.shape[1] % 8 == 0
            and d_cand.shape[1] % 8 == 0):
        return hip_ext(True).gemm_tn_bf16(d_cand, code_c)
    return d_cand.t() @ code_c


def _lu_corrections(labels, sampled, vocab: int):
    """log(S*q(id)) corrections for the log-uniform candidate sampler
    (reference-path helper; the HIP kernels compute these inline)."""
    S = int(sampled.numel())
    ct = torch.log(ref.log_uniform_probs(labels, vocab) * S)
    cs = torch.log(ref.log_uniform_probs(sampled, vocab) * S)
    return ct, cs


def sampled_ce_fwd(logits_cand, labels, sampled, vocab: int):
    if backend_for(logits_cand) == 'hip':
        return hip_ext(True).sampled_ce_fwd(logits_cand, labels, sampled,
                                            int(vocab))
    ct, cs = _lu_corrections(labels, sampled, vocab)
    return ref.sampled_ce_fwd(logits_cand, labels, sampled, ct, cs)


def sampled_ce_bwd(logits_cand, labels, sampled, vocab: int, lse,
                   scale: float):
    if backend_for(logits_cand) == 'hip':
        return hip_ext(True).sampled_ce_bwd(logits_cand, labels, sampled,
                                            int(vocab), lse, float(scale))
    ct, cs = _lu_corrections(labels, sampled, vocab)
    return ref.sampled_ce_bwd(logits_cand, labels, sampled, ct, cs, lse,
                              scale)
