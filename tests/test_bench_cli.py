"""bench.py contract tests: single-process and torchrun-DP2 runs of the REAL
bench code path (tiny vocab via C2V_BENCH_TINY) — validates the exact wiring
the driver's round-end SCALE run uses (distributed init, equal-shard reducer,
barrier timing, one-JSON-line output)."""

import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(extra, env_extra=None, timeout=360):
    env = dict(os.environ, PYTHONPATH=ROOT, C2V_BENCH_TINY='1',
               MASTER_ADDR='127.0.0.1', **(env_extra or {}))
    r = subprocess.run(extra, capture_output=True, text=True, env=env,
                       cwd=ROOT, timeout=timeout)
    assert r.returncode == 0, (r.stdout[-800:], r.stderr[-1500:])
    lines = [l for l in r.stdout.splitlines() if l.strip().startswith('{')]
    assert len(lines) == 1, r.stdout
    return json.loads(lines[0])


@pytest.mark.timeout(420)
def test_bench_single_process_cpu():
    out = run_bench([sys.executable, 'bench.py', '--steps', '3', '--warmup',
                     '1', '--batch', '8'])
    for key in ('metric', 'value', 'n_gpus', 'ms_per_step', 'scaling',
                'vs_baseline', 'dtype', 'data', 'config'):
        assert key in out, key
    assert out['n_gpus'] == 1
    assert out['value'] > 0
    assert out['config']['global_batch'] == 8
    assert out['data'] == 'synthetic'


@pytest.mark.timeout(600)
def test_bench_torchrun_dp2_cpu():
    out = run_bench([sys.executable, '-m', 'torch.distributed.run',
                     '--nnodes=1', '--nproc-per-node', '2',
                     '--master-addr', '127.0.0.1', '--master-port', '29553',
                     'bench.py', '--steps', '3', '--warmup', '1',
                     '--batch', '8'])
    assert out['n_gpus'] == 2
    assert out['config']['parallelism'] == 'dp2'
    assert out['config']['global_batch'] == 16


@pytest.mark.timeout(420)
def test_bench_sampled_flag():
    out = run_bench([sys.executable, 'bench.py', '--steps', '3', '--warmup',
                     '1', '--batch', '8', '--sampled-softmax', '64'])
    assert out['config']['softmax'] == 'sampled-64'


@pytest.mark.timeout(600)
def test_bench_gpus_flag_self_launches_torchrun():
    """`--gpus 2` without a torchrun rendezvous must run 2 ranks (a single
    process would report inflated aggregate throughput)."""
    out = run_bench([sys.executable, 'bench.py', '--gpus', '2', '--steps',
                     '3', '--warmup', '1', '--batch', '8'])
    assert out['n_gpus'] == 2
    assert out['config']['parallelism'] == 'dp2'
    assert out['config']['global_batch'] == 16
