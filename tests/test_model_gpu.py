"""Model-stack e2e on the GPU: train a tiny-but-bf16 model through the real
train loop (native reader + prefetcher + HIP kernels), then evaluate and
predict — the full reference workflow on device."""

import os
import pickle
import random

import pytest
import torch

pytestmark = pytest.mark.gpu


def write_dataset(tmp_path, n_train=256, n_test=48, max_contexts=8):
    rng = random.Random(3)
    tokens = ['tk%d' % i for i in range(64)]
    paths = ['ph%d' % i for i in range(48)]
    targets = ['alpha|one', 'beta|two', 'gamma|three', 'delta']

    def gen_line():
        t = rng.choice(targets)
        ti = targets.index(t)
        k = rng.randint(2, max_contexts)
        ctxs = ' '.join('%s,%s,%s' % (tokens[(ti * 7 + rng.randint(0, 3)) % 64],
                                      paths[(ti * 5 + rng.randint(0, 2)) % 48],
                                      tokens[(ti * 7 + rng.randint(0, 3)) % 64])
                        for _ in range(k))
        return t + ' ' + ctxs + ' ' * (max_contexts - k) + '\n'

    prefix = str(tmp_path / 'gds')
    with open(prefix + '.train.c2v', 'w') as f:
        f.writelines(gen_line() for _ in range(n_train))
    with open(prefix + '.val.c2v', 'w') as f:
        f.writelines(gen_line() for _ in range(n_test))
    with open(prefix + '.dict.c2v', 'wb') as f:
        pickle.dump({t: 9 for t in tokens}, f)
        pickle.dump({p: 9 for p in paths}, f)
        pickle.dump({t: 9 for t in targets}, f)
    return prefix


def test_gpu_train_evaluate_predict(tmp_path):
    from code2vec_amd.config import Config
    from code2vec_amd.models.torch_model import Code2VecModel

    prefix = write_dataset(tmp_path)
    cfg = Config(set_defaults=True)
    cfg.TRAIN_DATA_PATH_PREFIX = prefix
    cfg.TEST_DATA_PATH = prefix + '.val.c2v'
    cfg.MODEL_SAVE_PATH = str(tmp_path / 'm' / 'model')
    cfg.MAX_CONTEXTS = 8
    cfg.TOKEN_EMBEDDINGS_SIZE = 64
    cfg.PATH_EMBEDDINGS_SIZE = 64
    cfg.CODE_VECTOR_SIZE = 192
    cfg.TARGET_EMBEDDINGS_SIZE = 192
    cfg.TRAIN_BATCH_SIZE = cfg.TEST_BATCH_SIZE = 32
    cfg.NUM_TRAIN_EPOCHS = 40
    cfg.SAVE_EVERY_EPOCHS = 40
    cfg.COMPUTE_DTYPE = 'bf16'
    cfg.DEVICE = 'cuda:0'
    cfg.VERBOSE_MODE = 0
    cfg.SHUFFLE_BUFFER_SIZE = 128

    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        model = Code2VecModel(cfg)
        assert model.network.device.type == 'cuda'
        model.train()
        results = model.evaluate()
    finally:
        os.chdir(cwd)
    assert results.topk_acc[-1] > 0.7, results
    # predict path (bf16 + topk kernel + attention strings)
    line = 'zz tk0,ph0,tk1 tk2,ph1,tk3'
    preds = model.predict([line])
    assert preds[0].original_name == 'zz'
    assert len(preds[0].topk_predicted_words) == 5  # 4 targets + PAD_OR_OOV
    assert torch.isfinite(torch.tensor(preds[0].code_vector)).all()
