"""Sampled-softmax op + training-path tests (CPU).

The op's gradient is checked against autograd on the same candidate-logit
formulation; the engine path is checked for learning and for sparse-only
target-table updates."""


import torch

from code2vec_amd.config import Config
from code2vec_amd.models.network import Code2VecNetwork
from code2vec_amd.ops import reference as R

B, S, V, D = 6, 12, 50, 24


def make_inputs(seed=0):
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(B, B + S, generator=g) * 2
    labels = torch.randint(0, V, (B,), generator=g)
    sampled = torch.randint(0, V, (S,), generator=g)
    sampled[0] = labels[0]  # force an accidental hit
    corr_true = torch.log(R.log_uniform_probs(labels, V) * S)
    corr_samp = torch.log(R.log_uniform_probs(sampled, V) * S)
    return logits, labels, sampled, corr_true, corr_samp


def autograd_loss(logits, labels, sampled, corr_true, corr_samp):
    z0 = logits.diagonal() - corr_true
    zs = logits[:, B:] - corr_samp.reshape(1, -1)
    hit = sampled.reshape(1, -1) == labels.reshape(-1, 1)
    zs = torch.where(hit, torch.full_like(zs, -3.0e38), zs)
    z = torch.cat([z0.reshape(-1, 1), zs], dim=1)
    return (torch.logsumexp(z, dim=1) - z0).mean()


def test_sampled_ce_fwd_matches_autograd():
    logits, labels, sampled, ct, cs = make_inputs()
    loss_rows, lse = R.sampled_ce_fwd(logits, labels, sampled, ct, cs)
    ref = autograd_loss(logits, labels, sampled, ct, cs)
    assert torch.allclose(loss_rows.mean(), ref, atol=1e-5)


def test_sampled_ce_bwd_matches_autograd():
    logits, labels, sampled, ct, cs = make_inputs(1)
    leaf = logits.clone().requires_grad_(True)
    autograd_loss(leaf, labels, sampled, ct, cs).backward()
    _, lse = R.sampled_ce_fwd(logits, labels, sampled, ct, cs)
    d = R.sampled_ce_bwd(logits, labels, sampled, ct, cs, lse, 1.0 / B)
    assert torch.allclose(d, leaf.grad, atol=1e-5)
    # cols 0..B-1 off the diagonal must be exactly zero
    off_diag = d[:, :B] * (1 - torch.eye(B))
    assert torch.all(off_diag == 0)


def test_log_uniform_sampler_distribution():
    ids = R.sample_log_uniform(200000, 1000, 'cpu')
    assert ids.min() >= 0 and ids.max() < 1000
    # log-uniform: P(id < 31) = log(32)/log(1001) ≈ 0.50
    frac = (ids < 31).float().mean().item()
    assert 0.45 < frac < 0.55
    # probabilities sum to ~1
    q = R.log_uniform_probs(torch.arange(1000), 1000)
    assert abs(q.sum().item() - 1.0) < 1e-4


def test_sampled_training_path_learns_and_is_sparse():
    cfg = Config(set_defaults=True)
    cfg.TRAIN_DATA_PATH_PREFIX = 'unused'
    cfg.MAX_CONTEXTS = 5
    cfg.TOKEN_EMBEDDINGS_SIZE = 8
    cfg.PATH_EMBEDDINGS_SIZE = 8
    cfg.CODE_VECTOR_SIZE = 24
    cfg.TARGET_EMBEDDINGS_SIZE = 24
    cfg.DROPOUT_KEEP_RATE = 1.0
    cfg.COMPUTE_DTYPE = 'fp32'
    cfg.SAMPLED_SOFTMAX_SIZE = 16
    torch.manual_seed(3)
    net = Code2VecNetwork(cfg, 40, 30, V, device='cpu')
    p0 = net.target_table.clone()
    g = torch.Generator().manual_seed(4)
    src = torch.randint(0, 40, (B, 5), generator=g, dtype=torch.int32)
    pth = torch.randint(0, 30, (B, 5), generator=g, dtype=torch.int32)
    tgt = torch.randint(0, 40, (B, 5), generator=g, dtype=torch.int32)
    mask = torch.ones(B, 5)
    labels = torch.randint(1, V, (B,), generator=g)
    torch.manual_seed(100)
    losses = [float(net.train_step(src, pth, tgt, mask, labels))
              for _ in range(200)]
    assert losses[-1] < losses[0] * 0.5
    # after a single further step, only candidate rows may change
    p_before = net.target_table.clone()
    torch.manual_seed(200)
    net.train_step(src, pth, tgt, mask, labels)
    changed = (net.target_table != p_before).any(dim=1)
    # labels certainly changed; most of the vocab untouched in one step
    assert changed[labels.long()].all()
    assert int(changed.sum()) <= B + cfg.SAMPLED_SOFTMAX_SIZE
    # sanity: something moved from init overall
    assert not torch.equal(net.target_table, p0)
