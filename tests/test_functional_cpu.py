"""CPU fallback paths of the GEMM-family dispatchers in ops/functional.py:
on CPU every logits_bwd_* / grad_weight helper must reduce to the plain
matmul composition it replaces on GPU."""

import torch

from code2vec_amd.ops import functional as F
from code2vec_amd.ops import reference as R

B, V, D = 8, 50, 12


def setup():
    torch.manual_seed(3)
    logits = torch.randn(B, V)
    labels = torch.randint(0, V, (B,))
    shadow = torch.randn(V, D)
    code = torch.randn(B, D)
    loss, lse = R.ce_fwd(logits, labels)
    return logits, labels, shadow, code, lse


def test_logits_bwd_code_cpu():
    logits, labels, shadow, code, lse = setup()
    d_logits = F.ce_bwd(logits, lse, labels, 1.0 / B)
    out = F.logits_bwd_code(d_logits, shadow)
    assert torch.allclose(out, (d_logits @ shadow).float())


def test_logits_bwd_target_cpu():
    logits, labels, shadow, code, lse = setup()
    d_logits = F.ce_bwd(logits, lse, labels, 1.0 / B)
    out = F.logits_bwd_target(d_logits, code)
    assert torch.allclose(out, d_logits.t() @ code)


def test_grad_weight_gemm_cpu():
    torch.manual_seed(4)
    ctx = torch.randn(32, D)
    dz = torch.randn(32, D)
    out = F.grad_weight_gemm(ctx, dz)
    assert torch.allclose(out, (ctx.t() @ dz).float())


def test_ce_fused_dispatchers_fall_back_on_cpu():
    logits, labels, shadow, code, lse = setup()
    scale = 1.0 / B
    assert F.ce_bwd_mode(logits) == 0
    d_logits = F.ce_bwd(logits, lse, labels, scale)

    out1 = F.logits_bwd_code_ce(logits, shadow, lse, labels, scale)
    assert torch.allclose(out1, (d_logits @ shadow).float())

    out2 = F.logits_bwd_target_ce(logits, code, lse, labels, scale)
    assert torch.allclose(out2, d_logits.t() @ code)

    out3, dl3 = F.logits_bwd_code_ce_write(logits, shadow, lse, labels,
                                           scale)
    assert torch.allclose(out3, (d_logits @ shadow).float())
    assert torch.allclose(dl3, d_logits)
