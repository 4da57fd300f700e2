"""Pure-PyTorch reference implementations of every engine op (fwd + bwd).

These are the semantic oracle for the HIP kernels and the CPU execution path.
The math matches the reference TF graph exactly:
- gather+concat+dropout   ≙ tensorflow_model.py:238-246
- transform+tanh          ≙ tensorflow_model.py:248-252
- masked softmax attention≙ tensorflow_model.py:254-263
- logits + CE loss        ≙ tensorflow_model.py:226-230
Dropout uses a counter-based splitmix64 hash so CPU and HIP backends generate
the identical mask from (seed, element-index).
"""

import math
from typing import Tuple

import torch

_SPLITMIX_GAMMA = -7046029254386353131  # 0x9E3779B97F4A7C15 as signed int64
_MIX1 = -4658895280553007687            # 0xBF58476D1CE4E5B9
_MIX2 = -7723592293110705685            # 0x94D049BB133111EB


def _splitmix64(x: torch.Tensor) -> torch.Tensor:
    """splitmix64 finalizer on int64 tensors (wrapping arithmetic)."""
    z = x * _SPLITMIX_GAMMA
    z = (z ^ (z >> 30).bitwise_and(0x3FFFFFFFF)) * _MIX1  # logical shift via mask
    z = (z ^ (z >> 27).bitwise_and(0x1FFFFFFFFF)) * _MIX2
    return z ^ (z >> 31).bitwise_and(0x1FFFFFFFF)


def dropout_keep_mask(seed: int, numel: int, keep_prob: float,
                      device) -> torch.Tensor:
    """Deterministic keep-mask: element i kept iff hash(seed, i) < keep_prob.
    Uses the top 24 bits of splitmix64 as a uniform in [0,1)."""
    idx = torch.arange(numel, dtype=torch.int64, device=device)
    h = _splitmix64(idx + seed)
    u24 = (h >> 40).bitwise_and(0xFFFFFF).to(torch.float32) / float(1 << 24)
    return u24 < keep_prob


# ---------------------------------------------------------------------------
# gather + concat (+ fused dropout)
# ---------------------------------------------------------------------------

def gather_concat_fwd(tok_table: torch.Tensor, path_table: torch.Tensor,
                      src_ids: torch.Tensor, path_ids: torch.Tensor,
                      tgt_ids: torch.Tensor, keep_prob: float, seed: int,
                      training: bool, out_dtype=torch.bfloat16) -> torch.Tensor:
    """(B,C) id triples → (B*C, 3d) context matrix in compute dtype.
    Fuses the reference's 3 embedding lookups + concat + train-time dropout
    (with 1/keep_prob scaling, tf.nn.dropout semantics)."""
    B, C = src_ids.shape
    d = tok_table.shape[1]
    src = tok_table.index_select(0, src_ids.reshape(-1).long())
    pth = path_table.index_select(0, path_ids.reshape(-1).long())
    tgt = tok_table.index_select(0, tgt_ids.reshape(-1).long())
    ctx = torch.cat([src, pth, tgt], dim=1)  # (B*C, 3d) fp32
    if training and keep_prob < 1.0:
        mask = dropout_keep_mask(seed, ctx.numel(), keep_prob, ctx.device)
        ctx = ctx * (mask.reshape(ctx.shape).to(ctx.dtype) / keep_prob)
    return ctx.to(out_dtype)


def gather_concat_bwd(d_ctx: torch.Tensor, keep_prob: float, seed: int,
                      training: bool) -> torch.Tensor:
    """Backward through the fused dropout only: the scatter into the embedding
    tables is performed by the sparse Adam step (SURVEY §2.3 K1 backward),
    which consumes (ids, d_rows) directly."""
    if training and keep_prob < 1.0:
        mask = dropout_keep_mask(seed, d_ctx.numel(), keep_prob, d_ctx.device)
        # fp32 scale then one rounding to the storage dtype — matches the
        # HIP kernel's bf2f -> *(1/keep) -> f2bf path exactly
        d_ctx = (d_ctx.float() * (mask.reshape(d_ctx.shape).float() / keep_prob)
                 ).to(d_ctx.dtype)
    return d_ctx


# ---------------------------------------------------------------------------
# transform (FC, no bias) + tanh
# ---------------------------------------------------------------------------

def transform_tanh_fwd(ctx: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Y = tanh(ctx @ W); W is (in=3d, out=D) like the reference TRANSFORM."""
    return torch.tanh(ctx @ w.to(ctx.dtype))


def transform_tanh_bwd(ctx: torch.Tensor, w: torch.Tensor, y: torch.Tensor,
                       d_y: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (d_ctx, d_w). d_z = d_y * (1 - y^2); d_ctx = d_z @ W^T;
    d_w = ctx^T @ d_z (accumulated in fp32)."""
    d_z = d_y * (1.0 - y.float() ** 2).to(d_y.dtype)
    d_ctx = d_z @ w.to(d_z.dtype).t()
    d_w = (ctx.t().float() @ d_z.float())
    return d_ctx, d_w


# ---------------------------------------------------------------------------
# masked softmax attention + weighted reduce
# ---------------------------------------------------------------------------

def attention_fwd(comb: torch.Tensor, a: torch.Tensor,
                  valid_mask: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """comb (B,C,D), a (D,), valid_mask (B,C) in {0,1} →
    (code_vectors (B,D) fp32, attention (B,C) fp32).

    scores = comb·a + log(mask); attention = softmax over C; code = Σ α·comb.
    An all-masked row yields zeros (the reference NaNs there — see SURVEY §7
    'masked softmax edge case'; train/eval never see such rows)."""
    scores = (comb.float() @ a.float().reshape(-1, 1)).squeeze(-1)  # (B,C)
    neg_inf = torch.finfo(torch.float32).min
    scores = torch.where(valid_mask > 0, scores, torch.full_like(scores, neg_inf))
    m = scores.max(dim=1, keepdim=True).values
    any_valid = (valid_mask > 0).any(dim=1, keepdim=True)
    m = torch.where(any_valid, m, torch.zeros_like(m))
    e = torch.exp(scores - m) * (valid_mask > 0).float()
    denom = e.sum(dim=1, keepdim=True)
    alpha = torch.where(denom > 0, e / denom, torch.zeros_like(e))  # (B,C)
    code = torch.einsum('bc,bcd->bd', alpha, comb.float())
    return code, alpha


def attention_bwd(comb: torch.Tensor, a: torch.Tensor, alpha: torch.Tensor,
                  d_code: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (d_comb (B,C,D) in comb.dtype, d_a (D,) fp32).

    dα_c = comb_c · dv ; de_c = α_c (dα_c − Σ_j α_j dα_j) ;
    d_comb_c = α_c dv + de_c a ; d_a = Σ de_c comb_c."""
    combf = comb.float()
    dv = d_code.float()                                    # (B,D)
    d_alpha = torch.einsum('bcd,bd->bc', combf, dv)        # (B,C)
    inner = (alpha * d_alpha).sum(dim=1, keepdim=True)     # (B,1)
    d_e = alpha * (d_alpha - inner)                        # (B,C)
    d_comb = alpha.unsqueeze(-1) * dv.unsqueeze(1) + d_e.unsqueeze(-1) * a.float().reshape(1, 1, -1)
    d_a = torch.einsum('bc,bcd->d', d_e, combf)
    return d_comb.to(comb.dtype), d_a


# ---------------------------------------------------------------------------
# cross-entropy over the target vocabulary
# ---------------------------------------------------------------------------

def ce_fwd(logits: torch.Tensor, labels: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (per-row loss (B,) fp32, logsumexp (B,) fp32).
    loss_b = logsumexp(logits_b) − logits_b[label_b]."""
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=1)
    picked = lf.gather(1, labels.reshape(-1, 1)).squeeze(1)
    return lse - picked, lse


def ce_bwd(logits: torch.Tensor, lse: torch.Tensor, labels: torch.Tensor,
           scale: float) -> torch.Tensor:
    """d_logits = scale * (softmax(logits) − onehot(label)), in logits.dtype."""
    p = torch.exp(logits.float() - lse.reshape(-1, 1))
    p.scatter_add_(1, labels.reshape(-1, 1),
                   torch.full((logits.shape[0], 1), -1.0, device=logits.device))
    return (p * scale).to(logits.dtype)


# ---------------------------------------------------------------------------
# sampled softmax (training-time approximation of the full-vocab CE)
# ---------------------------------------------------------------------------

def log_uniform_probs(ids: torch.Tensor, vocab_size: int) -> torch.Tensor:
    """Zipf/log-uniform sampling probability of index k:
    q(k) = log((k+2)/(k+1)) / log(V+1). The target vocabulary is sorted by
    frequency (vocabularies top-N by count), so index order ~ rank order."""
    idf = ids.float()
    return torch.log((idf + 2.0) / (idf + 1.0)) / math.log(vocab_size + 1.0)


def sample_log_uniform(n: int, vocab_size: int, device,
                       generator=None) -> torch.Tensor:
    """Draw n ids (with replacement) from the log-uniform distribution over
    [0, vocab_size)."""
    u = torch.rand(n, device=device, generator=generator)
    ids = (torch.exp(u * math.log(vocab_size + 1.0)) - 1.0).long()
    return ids.clamp_(0, vocab_size - 1)


def sampled_ce_fwd(logits_cand: torch.Tensor, labels: torch.Tensor,
                   sampled: torch.Tensor, corr_true: torch.Tensor,
                   corr_samp: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """logits_cand (B, B+S): candidate logits where column b is row b's true
    class and columns B.. are the shared sampled negatives.
    Row b's effective logit set: z0 = logits[b,b] - corr_true[b];
    z_j = logits[b,B+j] - corr_samp[j], with accidental hits
    (sampled[j] == labels[b]) masked to -inf.
    Returns (per-row loss, per-row lse over the effective set)."""
    B = logits_cand.shape[0]
    lf = logits_cand.float()
    z0 = lf.diagonal() - corr_true                          # (B,)
    zs = lf[:, B:] - corr_samp.reshape(1, -1)               # (B,S)
    hit = sampled.reshape(1, -1) == labels.reshape(-1, 1)   # (B,S)
    zs = torch.where(hit, torch.full_like(zs, -3.0e38), zs)
    z = torch.cat([z0.reshape(-1, 1), zs], dim=1)           # (B,1+S)
    lse = torch.logsumexp(z, dim=1)
    return lse - z0, lse


def sampled_ce_bwd(logits_cand: torch.Tensor, labels: torch.Tensor,
                   sampled: torch.Tensor, corr_true: torch.Tensor,
                   corr_samp: torch.Tensor, lse: torch.Tensor,
                   scale: float) -> torch.Tensor:
    """d_logits_cand (B, B+S) in logits_cand.dtype: softmax over the effective
    set minus the one-hot at the true class, times scale. Columns 0..B-1
    other than the diagonal get zero."""
    B, T = logits_cand.shape
    lf = logits_cand.float()
    out = torch.zeros_like(lf)
    z0 = lf.diagonal() - corr_true
    p0 = torch.exp(z0 - lse) - 1.0
    out.diagonal().copy_(p0 * scale)
    zs = lf[:, B:] - corr_samp.reshape(1, -1)
    hit = sampled.reshape(1, -1) == labels.reshape(-1, 1)
    ps = torch.exp(zs - lse.reshape(-1, 1))
    ps = torch.where(hit, torch.zeros_like(ps), ps)
    out[:, B:] = ps * scale
    return out.to(logits_cand.dtype)


# ---------------------------------------------------------------------------
# Adam (TF AdamOptimizer formulation) — dense and sparse-row
# ---------------------------------------------------------------------------

def sparse_dedup_sum(ids: torch.Tensor,
                     rows: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, int]:
    """Rank-local dedup+sum of sparse (id, grad-row) contributions before a
    DP all-gather: returns (unique ids int64, fp32 summed rows, count).
    Semantics-preserving vs shipping the raw rows — duplicate contributions
    are summed either way; only the fp addition order changes."""
    uniq, inverse = torch.unique(ids.long(), return_inverse=True)
    acc = torch.zeros(uniq.numel(), rows.shape[1], dtype=torch.float32,
                      device=rows.device)
    acc.index_add_(0, inverse, rows.float())
    return uniq, acc, int(uniq.numel())


def adam_dense_step(p: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
                    v: torch.Tensor, step: int, lr: float, beta1: float,
                    beta2: float, eps: float, shadow: torch.Tensor = None):
    """In-place dense Adam in the TF formulation (epsilon outside the
    bias-corrected sqrt): lr_t = lr·√(1−β2^t)/(1−β1^t); p −= lr_t·m/(√v+ε).
    Optionally refreshes a bf16 shadow copy used by bf16 GEMM consumers."""
    gf = g.float()
    m.mul_(beta1).add_(gf, alpha=1 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
    lr_t = lr * (1 - beta2 ** step) ** 0.5 / (1 - beta1 ** step)
    p.addcdiv_(m, v.sqrt().add_(eps), value=-lr_t)
    if shadow is not None:
        shadow.copy_(p.to(shadow.dtype).reshape(shadow.shape))


def adam_sparse_rows_step(p: torch.Tensor, ids: torch.Tensor,
                          grad_rows: torch.Tensor, m: torch.Tensor,
                          v: torch.Tensor, step: int, lr: float, beta1: float,
                          beta2: float, eps: float, shadow: torch.Tensor = None):
    """Lazy sparse-row Adam (TF AdamOptimizer._apply_sparse semantics: only
    touched rows update their moments). `ids` may contain duplicates; grads
    for duplicate rows are summed first."""
    uniq, inverse = torch.unique(ids.long(), return_inverse=True)
    acc = torch.zeros(uniq.numel(), grad_rows.shape[1],
                      dtype=torch.float32, device=grad_rows.device)
    acc.index_add_(0, inverse, grad_rows.float())
    m_rows = m.index_select(0, uniq)
    v_rows = v.index_select(0, uniq)
    m_rows.mul_(beta1).add_(acc, alpha=1 - beta1)
    v_rows.mul_(beta2).addcmul_(acc, acc, value=1 - beta2)
    lr_t = lr * (1 - beta2 ** step) ** 0.5 / (1 - beta1 ** step)
    p_rows = p.index_select(0, uniq)
    p_rows.addcdiv_(m_rows, v_rows.sqrt().add(eps), value=-lr_t)
    m.index_copy_(0, uniq, m_rows)
    v.index_copy_(0, uniq, v_rows)
    p.index_copy_(0, uniq, p_rows)
    if shadow is not None:
        shadow.index_copy_(0, uniq, p_rows.to(shadow.dtype))
