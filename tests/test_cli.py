"""CLI driver end-to-end: single-process training via code2vec.py flags, and
2-process data-parallel training via torchrun (gloo on CPU, 127.0.0.1
rendezvous) exercising the exact launch path the GPU scaling bench uses."""

import os
import pickle
import random
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def write_tiny_dataset(tmp_path, n=48, max_contexts=4):
    rng = random.Random(7)
    tokens = ['t%d' % i for i in range(12)]
    paths = ['p%d' % i for i in range(8)]
    targets = ['aa|bb', 'cc', 'dd|ee']
    prefix = str(tmp_path / 'tiny')

    def line():
        tgt = rng.choice(targets)
        k = rng.randint(1, max_contexts)
        ctxs = ' '.join('%s,%s,%s' % (rng.choice(tokens), rng.choice(paths),
                                      rng.choice(tokens)) for _ in range(k))
        return tgt + ' ' + ctxs + ' ' * (max_contexts - k) + '\n'

    for role, cnt in (('train', n), ('val', 12)):
        with open(prefix + '.%s.c2v' % role, 'w') as f:
            f.writelines(line() for _ in range(cnt))
    with open(prefix + '.dict.c2v', 'wb') as f:
        pickle.dump({t: 5 for t in tokens}, f)
        pickle.dump({p: 5 for p in paths}, f)
        pickle.dump({t: 5 for t in targets}, f)
    return prefix


def patch_config_env():
    # small hyperparams via a sitecustomize-free route: pass flags the CLI
    # supports; sizes come from the dataset vocab (tiny), epochs via env
    return {}


@pytest.mark.timeout(300)
def test_cli_train_and_eval(tmp_path):
    prefix = write_tiny_dataset(tmp_path)
    save = str(tmp_path / 'm' / 'model')
    env = dict(os.environ, PYTHONPATH=ROOT)
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, 'code2vec.py'),
         '--data', prefix, '--test', prefix + '.val.c2v', '--save', save,
         '--dtype', 'fp32', '--device', 'cpu', '-v', '0'],
        capture_output=True, text=True, env=env, cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-2000:]
    # an epoch checkpoint and dictionaries.bin were written
    files = os.listdir(tmp_path / 'm')
    assert any(f.endswith('__entire-model') for f in files), files
    assert 'dictionaries.bin' in files


@pytest.mark.timeout(420)
def test_cli_torchrun_dp2(tmp_path):
    prefix = write_tiny_dataset(tmp_path)
    save = str(tmp_path / 'm2' / 'model')
    env = dict(os.environ, PYTHONPATH=ROOT, MASTER_ADDR='127.0.0.1')
    r = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29531',
         os.path.join(ROOT, 'code2vec.py'),
         '--data', prefix, '--test', prefix + '.val.c2v', '--save', save,
         '--dtype', 'fp32', '--device', 'cpu', '-v', '0'],
        capture_output=True, text=True, env=env, cwd=str(tmp_path))
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
    files = os.listdir(tmp_path / 'm2')
    assert any(f.endswith('__entire-model') for f in files), files
