"""hipGraph-captured inference parity vs the eager predict path (GPU)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_graph_capture_matches_eager():
    from code2vec_amd.config import Config
    from code2vec_amd.models.network import Code2VecNetwork
    from code2vec_amd.serving.graph_infer import GraphCapturedPredictor

    cfg = Config(set_defaults=True)
    cfg.MAX_CONTEXTS = 40
    torch.manual_seed(1)
    net = Code2VecNetwork(cfg, 5000, 4000, 3000, device='cuda:0')
    B, C = 64, cfg.MAX_CONTEXTS
    g = torch.Generator().manual_seed(2)
    src = torch.randint(0, 5000, (B, C), generator=g, dtype=torch.int32).cuda()
    pth = torch.randint(0, 4000, (B, C), generator=g, dtype=torch.int32).cuda()
    tgt = torch.randint(0, 5000, (B, C), generator=g, dtype=torch.int32).cuda()
    mask = (torch.rand(B, C, generator=g) > 0.2).float().cuda()
    mask[:, 0] = 1.0

    idx_e, sc_e, code_e, _ = net.predict_batch(src, pth, tgt, mask, top_k=10)
    predictor = GraphCapturedPredictor(net, B, 10)
    idx_g, sc_g, code_g = predictor.predict(src, pth, tgt, mask)
    assert torch.equal(idx_e, idx_g)
    assert torch.allclose(code_e, code_g, atol=1e-5)
    assert torch.allclose(sc_e.float(), sc_g.float(), atol=1e-3)

    # replay with different inputs gives different (and again matching) results
    src2 = torch.randint(0, 5000, (B, C), generator=g, dtype=torch.int32).cuda()
    idx_e2, _, code_e2, _ = net.predict_batch(src2, pth, tgt, mask, top_k=10)
    idx_g2, _, code_g2 = predictor.predict(src2, pth, tgt, mask)
    assert torch.equal(idx_e2, idx_g2)
    assert torch.allclose(code_e2, code_g2, atol=1e-5)
    assert not torch.equal(idx_e, idx_e2)
