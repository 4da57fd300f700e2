#!/usr/bin/env python3
"""Inference benchmark (BASELINE.json config 5): top-k predict + code-vector
export at batch 4096 on 1 MI355X, eager launches vs hipGraph replay."""

import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import argparse
import json
import time

import torch

from code2vec_amd.config import Config
from code2vec_amd.models.network import Code2VecNetwork
from code2vec_amd.serving.graph_infer import GraphCapturedPredictor


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--batch', type=int, default=4096)
    ap.add_argument('--steps', type=int, default=30)
    ap.add_argument('--warmup', type=int, default=10)
    ap.add_argument('--topk', type=int, default=10)
    args = ap.parse_args()

    assert torch.cuda.is_available()
    cfg = Config(set_defaults=True)
    net = Code2VecNetwork(cfg,
                          token_vocab_size=cfg.MAX_TOKEN_VOCAB_SIZE + 1,
                          path_vocab_size=cfg.MAX_PATH_VOCAB_SIZE + 1,
                          target_vocab_size=cfg.MAX_TARGET_VOCAB_SIZE + 1,
                          device='cuda:0')
    B, C = args.batch, cfg.MAX_CONTEXTS
    g = torch.Generator().manual_seed(0)
    src = torch.randint(1, cfg.MAX_TOKEN_VOCAB_SIZE, (B, C), generator=g,
                        dtype=torch.int32).cuda()
    pth = torch.randint(1, cfg.MAX_PATH_VOCAB_SIZE, (B, C), generator=g,
                        dtype=torch.int32).cuda()
    tgt = torch.randint(1, cfg.MAX_TOKEN_VOCAB_SIZE, (B, C), generator=g,
                        dtype=torch.int32).cuda()
    mask = torch.ones(B, C, device='cuda')

    def timed(fn):
        for _ in range(args.warmup):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / args.steps

    eager = timed(lambda: net.predict_batch(src, pth, tgt, mask, top_k=args.topk))

    predictor = GraphCapturedPredictor(net, B, args.topk)
    graphed = timed(lambda: predictor.predict(src, pth, tgt, mask))

    # correctness cross-check
    idx_e, sc_e, code_e, _ = net.predict_batch(src, pth, tgt, mask, top_k=args.topk)
    idx_g, sc_g, code_g = predictor.predict(src, pth, tgt, mask)
    assert torch.equal(idx_e, idx_g), 'graph vs eager top-k mismatch'
    assert torch.allclose(code_e, code_g, atol=1e-5)

    print(json.dumps({
        'metric': 'inference_examples_per_sec',
        'batch': B,
        'eager_ms': round(eager * 1e3, 3),
        'graph_ms': round(graphed * 1e3, 3),
        'eager_ex_per_sec': round(B / eager, 1),
        'graph_ex_per_sec': round(B / graphed, 1),
        'topk': args.topk,
        'dtype': 'bf16',
        'data': 'synthetic',
    }))


if __name__ == '__main__':
    main()
