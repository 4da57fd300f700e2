"""End-to-end tiny-synthetic run on CPU (BASELINE.json config 1): dataset
build → train via the real train loop → evaluate → predict → checkpoint
save/load/release round-trip → w2v/code-vector export."""

import os
import pickle
import random

import numpy as np
import pytest

from code2vec_amd.config import Config
from code2vec_amd.models.torch_model import Code2VecModel
from code2vec_amd.vocabularies import VocabType

TOKENS = ['val%d' % i for i in range(20)]
PATHS = ['path%d' % i for i in range(15)]
TARGETS = ['get|x', 'set|x', 'make|thing', 'do|stuff', 'run']


def write_dataset(tmp_path, n_train=64, n_test=16, max_contexts=6):
    rng = random.Random(11)

    def gen_line():
        target = rng.choice(TARGETS)
        # correlate contexts with the target so the model can learn
        tid = TARGETS.index(target)
        n_ctx = rng.randint(1, max_contexts)
        ctxs = []
        for _ in range(n_ctx):
            s = TOKENS[(tid * 3 + rng.randint(0, 2)) % len(TOKENS)]
            p = PATHS[(tid * 2 + rng.randint(0, 1)) % len(PATHS)]
            t = TOKENS[(tid * 3 + rng.randint(0, 2)) % len(TOKENS)]
            ctxs.append('%s,%s,%s' % (s, p, t))
        pad = ' ' * (max_contexts - n_ctx)
        return target + ' ' + ' '.join(ctxs) + pad + '\n'

    prefix = str(tmp_path / 'tiny')
    with open(prefix + '.train.c2v', 'w') as f:
        f.writelines(gen_line() for _ in range(n_train))
    with open(prefix + '.val.c2v', 'w') as f:
        f.writelines(gen_line() for _ in range(n_test))
    tok_counts = {t: 10 for t in TOKENS}
    path_counts = {p: 10 for p in PATHS}
    tgt_counts = {t: 10 for t in TARGETS}
    with open(prefix + '.dict.c2v', 'wb') as f:
        pickle.dump(tok_counts, f)
        pickle.dump(path_counts, f)
        pickle.dump(tgt_counts, f)
        pickle.dump(n_train, f)
    return prefix


def tiny_train_config(tmp_path, prefix):
    cfg = Config(set_defaults=True)
    cfg.TRAIN_DATA_PATH_PREFIX = prefix
    cfg.TEST_DATA_PATH = prefix + '.val.c2v'
    cfg.MODEL_SAVE_PATH = str(tmp_path / 'models' / 'tiny_model')
    cfg.MAX_CONTEXTS = 6
    cfg.TOKEN_EMBEDDINGS_SIZE = 16
    cfg.PATH_EMBEDDINGS_SIZE = 16
    cfg.CODE_VECTOR_SIZE = 48
    cfg.TARGET_EMBEDDINGS_SIZE = 48
    cfg.TRAIN_BATCH_SIZE = cfg.TEST_BATCH_SIZE = 16
    cfg.NUM_TRAIN_EPOCHS = 12
    cfg.SAVE_EVERY_EPOCHS = 12   # one save+eval at the very end
    cfg.NUM_BATCHES_TO_LOG_PROGRESS = 8
    cfg.COMPUTE_DTYPE = 'fp32'
    cfg.DEVICE = 'cpu'
    cfg.VERBOSE_MODE = 0
    cfg.SHUFFLE_BUFFER_SIZE = 64
    return cfg


@pytest.fixture(scope='module')
def trained(tmp_path_factory):
    tmp_path = tmp_path_factory.mktemp('e2e')
    cwd = os.getcwd()
    os.chdir(tmp_path)  # log.txt lands here
    try:
        prefix = write_dataset(tmp_path)
        cfg = tiny_train_config(tmp_path, prefix)
        model = Code2VecModel(cfg)
        model.train()
    finally:
        os.chdir(cwd)
    return tmp_path, prefix, cfg, model


def test_training_learns(trained):
    tmp_path, prefix, cfg, model = trained
    results = model.evaluate()
    # the tiny dataset is nearly deterministic — top-10 accuracy should be high
    assert results.topk_acc[-1] > 0.8
    assert results.subtoken_f1 > 0.5
    # eval loss is reported (reference Keras backend behavior) and sane for
    # a model that has learned the 5-target task
    assert results.loss is not None and 0.0 < results.loss < 2.0


def test_checkpoint_roundtrip(trained, tmp_path):
    _, prefix, cfg, model = trained
    import torch
    save_base = cfg.MODEL_SAVE_PATH + '_iter12'
    assert os.path.isfile(cfg.get_entire_model_path(save_base))
    assert os.path.isfile(cfg.get_vocabularies_path_from_model_path(save_base))

    cfg2 = tiny_train_config(tmp_path, prefix)
    cfg2.TRAIN_DATA_PATH_PREFIX = None
    cfg2.MODEL_LOAD_PATH = save_base
    cfg2.TEST_DATA_PATH = prefix + '.val.c2v'
    model2 = Code2VecModel(cfg2)
    for n in model.network.param_names():
        assert torch.equal(model.network.get_param(n), model2.network.get_param(n)), n
    # adam state restored too
    assert model2.network.adam_step == model.network.adam_step


def test_release_strips_optimizer(trained, tmp_path):
    _, prefix, cfg, model = trained
    import torch
    save_base = cfg.MODEL_SAVE_PATH + '_iter12'
    cfg3 = tiny_train_config(tmp_path, prefix)
    cfg3.TRAIN_DATA_PATH_PREFIX = None
    cfg3.MODEL_LOAD_PATH = save_base
    cfg3.RELEASE = True
    model3 = Code2VecModel(cfg3)
    assert model3.evaluate() is None  # release flow returns None
    release_path = save_base + '.release'
    assert os.path.isfile(release_path)
    payload = torch.load(release_path, map_location='cpu', weights_only=False)
    assert 'adam_m.w' not in payload['model']
    full = torch.load(cfg.get_entire_model_path(save_base), map_location='cpu',
                      weights_only=False)
    assert 'adam_m.w' in full['model']
    # a released model loads fine
    cfg4 = tiny_train_config(tmp_path, prefix)
    cfg4.TRAIN_DATA_PATH_PREFIX = None
    cfg4.MODEL_LOAD_PATH = release_path[:-len('.release')]
    model4 = Code2VecModel(cfg4)
    assert model4.network.adam_step == 0 or model4.network.adam_step >= 0


def test_predict_interface(trained):
    _, prefix, cfg, model = trained
    line = 'whatever val0,path0,val1 val2,path1,val0'
    results = model.predict([line])
    assert len(results) == 1
    r = results[0]
    assert r.original_name == 'whatever'
    expected_k = min(cfg.TOP_K_WORDS_CONSIDERED_DURING_PREDICTION,
                     model.vocabs.target_vocab.size)
    assert len(r.topk_predicted_words) == expected_k
    # normalized scores sum to 1 (softmax over top-k)
    np.testing.assert_allclose(r.topk_predicted_words_scores.sum(), 1.0, atol=1e-4)
    assert isinstance(r.attention_per_context, dict)
    assert ('val0', 'path0', 'val1') in r.attention_per_context
    assert r.code_vector.shape == (cfg.CODE_VECTOR_SIZE,)


def test_w2v_export(trained, tmp_path):
    _, prefix, cfg, model = trained
    out = str(tmp_path / 'tokens.w2v')
    model.save_word2vec_format(out, VocabType.Token)
    with open(out) as f:
        header = f.readline().split()
        assert int(header[0]) == model.vocabs.token_vocab.size
        assert int(header[1]) == cfg.TOKEN_EMBEDDINGS_SIZE
        first = f.readline().split()
        assert len(first) == 1 + cfg.TOKEN_EMBEDDINGS_SIZE


def test_export_code_vectors(trained, tmp_path):
    tmp, prefix, cfg, model = trained
    cfg.EXPORT_CODE_VECTORS = True
    cwd = os.getcwd()
    os.chdir(tmp)
    try:
        model.evaluate()
    finally:
        cfg.EXPORT_CODE_VECTORS = False
        os.chdir(cwd)
    vec_file = cfg.TEST_DATA_PATH + '.vectors'
    assert os.path.isfile(vec_file)
    with open(vec_file) as f:
        row = f.readline().split()
        assert len(row) == cfg.CODE_VECTOR_SIZE


def test_resume_training_is_exact(tmp_path):
    """Optimizer-state continuity: train 6 steps straight vs train 3 +
    checkpoint + fresh-network load + 3 more — parameters must match
    exactly (full Adam state, dropout-stream position and step counter all
    round-trip, so the resumed run replays identical stochasticity)."""
    import torch

    from code2vec_amd.config import Config
    from code2vec_amd.models.network import Code2VecNetwork

    def make_cfg():
        cfg = Config(set_defaults=True)
        cfg.MAX_CONTEXTS = 6
        cfg.TOKEN_EMBEDDINGS_SIZE = 8
        cfg.PATH_EMBEDDINGS_SIZE = 8
        cfg.CODE_VECTOR_SIZE = 24
        cfg.TARGET_EMBEDDINGS_SIZE = 24
        cfg.DROPOUT_KEEP_RATE = 0.75   # exercised: the seed stream resumes
        cfg.COMPUTE_DTYPE = 'fp32'
        cfg.DEVICE = 'cpu'
        return cfg

    def batches():
        g = torch.Generator().manual_seed(11)
        out = []
        for _ in range(6):
            src = torch.randint(0, 30, (4, 6), generator=g, dtype=torch.int32)
            pth = torch.randint(0, 20, (4, 6), generator=g, dtype=torch.int32)
            tgt = torch.randint(0, 30, (4, 6), generator=g, dtype=torch.int32)
            mask = torch.ones(4, 6)
            lab = torch.randint(1, 15, (4,), generator=g)
            out.append((src, pth, tgt, mask, lab))
        return out

    bs = batches()
    net_a = Code2VecNetwork(make_cfg(), 30, 20, 15, device='cpu')
    for b in bs:
        net_a.train_step(*b)

    net_b = Code2VecNetwork(make_cfg(), 30, 20, 15, device='cpu')
    for b in bs[:3]:
        net_b.train_step(*b)
    ckpt = tmp_path / 'mid.pt'
    torch.save(net_b.state_dict(), ckpt)
    net_c = Code2VecNetwork(make_cfg(), 30, 20, 15, device='cpu')
    net_c.load_state_dict(torch.load(ckpt, weights_only=False))
    assert net_c.adam_step == 3
    for b in bs[3:]:
        net_c.train_step(*b)

    for n in net_a.param_names():
        assert torch.equal(net_a.get_param(n), net_c.get_param(n)), n
        assert torch.equal(net_a._adam_m[n], net_c._adam_m[n]), n
