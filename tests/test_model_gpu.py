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


@pytest.mark.timeout(1800)
def test_fullvocab_loss_trajectory_gpu_vs_cpu_oracle():
    """30 training steps of the FULL-vocab java14m architecture: the bf16
    GPU engine's loss trajectory must track the fp32 CPU oracle step by
    step (catches integration drift the per-op parity tests and the 5-step
    small-model test cannot — VERDICT r01 weak #6). The oracle runs the
    fp32 CPU engine at full vocab scale: a few seconds per step."""
    from code2vec_amd.config import Config
    from code2vec_amd.models.network import Code2VecNetwork

    def build(device, dtype):
        cfg = Config(set_defaults=True)
        cfg.TRAIN_DATA_PATH_PREFIX = 'unused'
        cfg.DROPOUT_KEEP_RATE = 1.0   # seed streams differ CPU vs GPU
        cfg.COMPUTE_DTYPE = dtype
        torch.manual_seed(1234)
        return Code2VecNetwork(cfg, 1301137, 911418, 261246, device=device)

    net_gpu = build('cuda:0', 'bf16')
    net_cpu = build('cpu', 'fp32')

    # learnable synthetic task: the label determines which id cluster the
    # contexts are drawn from, so the loss genuinely decreases
    g = torch.Generator().manual_seed(55)
    B, C, n_lab = 192, 200, 64
    batches = []
    for _ in range(4):
        labels = torch.randint(1, n_lab + 1, (B,), generator=g)
        base = labels.unsqueeze(1) * 37
        src = (base + torch.randint(0, 17, (B, C), generator=g)).to(torch.int32)
        pth = (labels.unsqueeze(1) * 23
               + torch.randint(0, 11, (B, C), generator=g)).to(torch.int32)
        tgt = (base + torch.randint(0, 17, (B, C), generator=g)).to(torch.int32)
        n_valid = torch.randint(C // 2, C + 1, (B,), generator=g)
        mask = (torch.arange(C).unsqueeze(0) < n_valid.unsqueeze(1)).float()
        src = torch.where(mask.bool(), src, torch.zeros_like(src))
        pth = torch.where(mask.bool(), pth, torch.zeros_like(pth))
        tgt = torch.where(mask.bool(), tgt, torch.zeros_like(tgt))
        batches.append((src, pth, tgt, mask, labels))

    gpu_losses, cpu_losses = [], []
    for step in range(30):
        b = batches[step % len(batches)]
        gpu_losses.append(float(net_gpu.train_step(*[t.cuda() for t in b])))
        cpu_losses.append(float(net_cpu.train_step(*b)))

    # trajectory tracking: every step within a few percent of the oracle
    for step, (lg, lc) in enumerate(zip(gpu_losses, cpu_losses)):
        assert abs(lg - lc) < 0.04 * max(1.0, abs(lc)), \
            (step, lg, lc, gpu_losses, cpu_losses)
    # and training actually learns, by the same margin, on both engines
    assert cpu_losses[-1] < cpu_losses[0] - 0.5, (cpu_losses[0], cpu_losses[-1])
    assert gpu_losses[-1] < gpu_losses[0] - 0.5, (gpu_losses[0], gpu_losses[-1])
    assert abs(gpu_losses[-1] - cpu_losses[-1]) < 0.04 * abs(cpu_losses[-1])


@pytest.mark.timeout(600)
@pytest.mark.gpu
@pytest.mark.parametrize("env", [
    {'C2V_SEED_PRE': '1'},
    {'C2V_SEED_PRE': '1', 'C2V_HASH_POS': 'postgather'},
    {'C2V_HASH_OVERLAP': '0'},
], ids=['seed-pre', 'postgather', 'no-overlap'])
def test_schedule_variants_train_identically(env, monkeypatch):
    """The env-gated step schedules (C2V_SEED_PRE, C2V_HASH_POS,
    C2V_HASH_OVERLAP — docs/TUNING.md) reorder kernel launches but must
    not change the math: a graph-captured training run under each variant
    produces the same losses as the default schedule up to the atomic
    fp32 accumulation reorder in the sparse-grad kernels. Dropout is ON
    so the seed stream (the state the variants move) is exercised."""
    from code2vec_amd.config import Config
    from code2vec_amd.models.network import Code2VecNetwork, GraphTrainStep

    def run(extra_env):
        for k in ('C2V_SEED_PRE', 'C2V_HASH_POS', 'C2V_HASH_OVERLAP'):
            monkeypatch.delenv(k, raising=False)
        for k, v in extra_env.items():
            monkeypatch.setenv(k, v)
        cfg = Config(set_defaults=True)
        cfg.TRAIN_DATA_PATH_PREFIX = 'unused'
        cfg.MAX_CONTEXTS = 20
        cfg.TOKEN_EMBEDDINGS_SIZE = 64
        cfg.PATH_EMBEDDINGS_SIZE = 64
        cfg.CODE_VECTOR_SIZE = 192
        cfg.TARGET_EMBEDDINGS_SIZE = 192
        cfg.DROPOUT_KEEP_RATE = 0.75
        cfg.COMPUTE_DTYPE = 'bf16'
        torch.manual_seed(7)
        net = Code2VecNetwork(cfg, 500, 300, 200, device='cuda:0')
        gs = GraphTrainStep(net, batch_size=16)
        g = torch.Generator().manual_seed(23)
        losses = []
        for _ in range(4):
            src = torch.randint(0, 500, (16, 20), generator=g,
                                dtype=torch.int32).cuda()
            pth = torch.randint(0, 300, (16, 20), generator=g,
                                dtype=torch.int32).cuda()
            tgt = torch.randint(0, 500, (16, 20), generator=g,
                                dtype=torch.int32).cuda()
            mask = torch.ones(16, 20).cuda()
            labels = torch.randint(1, 200, (16,), generator=g).cuda()
            losses.append(float(gs.step(src, pth, tgt, mask, labels)))
        return losses

    base = run({})
    var = run(env)
    for step, (lb, lv) in enumerate(zip(base, var)):
        assert abs(lb - lv) < 1e-3 * max(1.0, abs(lb)), (step, base, var)
