#!/usr/bin/env python3
"""Flagship training benchmark — the driver contract.

Measures training examples/sec on the java14m-shaped config (BASELINE.json):
1.3M-token / 911K-path / 261K-target vocabularies, 200 contexts, d=128,
per-GPU batch 1024, bf16 compute, full-softmax CE, TF-formulation Adam —
synthetic data (no network for datasets) with random-init weights.

`--gpus N` is informational for single-process runs; under torchrun each rank
reads RANK/LOCAL_RANK/WORLD_SIZE from the env (weak scaling: per-GPU batch is
fixed, global batch = 1024*N). Timing: W untimed warmup steps, then exactly K
steps bracketed by barrier + torch.cuda.synchronize on both sides; the
reported time is the MAX across ranks; rank 0 prints one JSON line.
"""

import argparse
import json
import os
import time

# Enable hipBLASLt/rocBLAS algorithm selection from the vendored tuning table
# (code2vec_amd/ops/tunableop_gfx950.csv, recorded on MI355X): +10% step time
# on the java14m GEMM shapes with zero startup cost (tuning itself is off;
# set PYTORCH_TUNABLEOP_TUNING=1 to re-tune). Must be set before torch import.
_here = os.path.dirname(os.path.abspath(__file__))
_tuned = os.path.join(_here, 'code2vec_amd', 'ops', 'tunableop_gfx950.csv')
if os.path.isfile(_tuned):
    # torch appends the device ordinal before ".csv" — provide a copy per
    # possible ordinal so every DP rank finds its table
    import shutil
    _base = '/tmp/c2v_tunableop.csv'
    for _i in range(8):
        _dst = '/tmp/c2v_tunableop%d.csv' % _i
        if not os.path.isfile(_dst):
            try:
                shutil.copy2(_tuned, _dst)
            except OSError:
                pass
    os.environ.setdefault('PYTORCH_TUNABLEOP_ENABLED', '1')
    os.environ.setdefault('PYTORCH_TUNABLEOP_TUNING', '0')
    os.environ.setdefault('PYTORCH_TUNABLEOP_FILENAME', _base)
    os.environ.setdefault('PYTORCH_TUNABLEOP_RECORD_UNTUNED', '0')

import torch

from code2vec_amd.config import Config
from code2vec_amd.models.network import Code2VecNetwork, NullReducer

BASELINE_V100_EX_PER_SEC = 4700.0  # BASELINE.md derived V100 throughput


def make_config(device):
    cfg = Config(set_defaults=True)
    cfg.DEVICE = device
    cfg.COMPUTE_DTYPE = 'bf16' if device.startswith('cuda') else 'fp32'
    if os.environ.get('C2V_BENCH_TINY') == '1':
        # test-only shrink (tests/test_bench_cli.py): exercises the exact
        # bench code path — distributed init, reducer, timing, JSON — on CPU
        cfg.MAX_TOKEN_VOCAB_SIZE = 1000
        cfg.MAX_PATH_VOCAB_SIZE = 800
        cfg.MAX_TARGET_VOCAB_SIZE = 500
        cfg.MAX_CONTEXTS = 16
    return cfg


ZIPF_S = 1.1   # id-frequency skew of the synthetic data (see synth_batches)

_zipf_cdf_cache = {}


def zipf_ids(vocab_size, shape, g, s=ZIPF_S):
    """Zipf(s)-distributed ids in [1, vocab_size): real java14m token/path
    frequencies are Zipf-like (a handful of identifiers and path shapes
    dominate), and the id distribution is performance-relevant on both sides:
    hot rows stress the sparse-Adam accumulate (replica path) and determine
    the post-dedup DP gather volume. Uniform ids would be an unrealistically
    EASY case for the accumulate and an unrealistically HARD one for DP."""
    key = (vocab_size, s)
    if key not in _zipf_cdf_cache:
        p = 1.0 / torch.arange(1, vocab_size, dtype=torch.float64) ** s
        _zipf_cdf_cache[key] = torch.cumsum(p / p.sum(), 0)
    cdf = _zipf_cdf_cache[key]
    u = torch.rand(shape, generator=g, dtype=torch.float64)
    return (torch.searchsorted(cdf, u) + 1).clamp_(1, vocab_size - 1) \
        .to(torch.int32)


def synth_batches(cfg, device, batch_size, n_batches=8, seed=0):
    """Pool of synthetic java14m-shaped batches, resident on device.
    C2V_BENCH_ID_DIST=uniform switches to round-1's uniform ids (for
    apples-to-apples comparisons against the r01 numbers)."""
    g = torch.Generator(device='cpu').manual_seed(seed)
    batches = []
    V_tok = cfg.MAX_TOKEN_VOCAB_SIZE + 1
    V_path = cfg.MAX_PATH_VOCAB_SIZE + 1
    V_tgt = cfg.MAX_TARGET_VOCAB_SIZE + 1
    C = cfg.MAX_CONTEXTS
    uniform = os.environ.get('C2V_BENCH_ID_DIST') == 'uniform'

    def draw(vocab, shape):
        if uniform:
            return torch.randint(1, vocab, shape, generator=g,
                                 dtype=torch.int32)
        return zipf_ids(vocab, shape, g)

    for _ in range(n_batches):
        src = draw(V_tok, (batch_size, C))
        pth = draw(V_path, (batch_size, C))
        tgt = draw(V_tok, (batch_size, C))
        # realistic context-count distribution: valid prefix of U[64, 200]
        lo = min(64, max(1, C // 2))
        n_valid = torch.randint(lo, C + 1, (batch_size,), generator=g)
        mask = (torch.arange(C).unsqueeze(0) < n_valid.unsqueeze(1)).float()
        src = torch.where(mask.bool(), src, torch.zeros_like(src))
        pth = torch.where(mask.bool(), pth, torch.zeros_like(pth))
        tgt = torch.where(mask.bool(), tgt, torch.zeros_like(tgt))
        labels = torch.randint(1, V_tgt, (batch_size,), generator=g)
        batches.append(tuple(t.to(device) for t in (src, pth, tgt, mask, labels)))
    return batches


def _tok_word(i):
    return 'tok%07d' % i


def _path_word(i):
    # java14m paths are Java String#hashCode ints on the wire
    h = (i * 2654435761) & 0xFFFFFFFF
    return str(h - (1 << 32) if h >= (1 << 31) else h)


def _target_word(i):
    return 'name|w%06d' % i


def generate_e2e_dataset(path, cfg, n_rows, seed=7):
    """Write a java14m-shaped synthetic `.c2v` to disk: Zipf-frequency
    token/path/target words (ids map rank->word), U[64,C] contexts per row,
    trailing-space padding exactly like preprocess.py output."""
    g = torch.Generator().manual_seed(seed)
    C = cfg.MAX_CONTEXTS
    with open(path, 'w') as f:
        for base in range(0, n_rows, 4096):
            nb = min(4096, n_rows - base)
            n_ctx = torch.randint(max(1, C // 3), C + 1, (nb,), generator=g)
            # ids are Zipf ranks; subtract 1 -> word index (0-based)
            src = zipf_ids(cfg.MAX_TOKEN_VOCAB_SIZE + 1, (nb, C), g) - 1
            pth = zipf_ids(cfg.MAX_PATH_VOCAB_SIZE + 1, (nb, C), g) - 1
            tgt = zipf_ids(cfg.MAX_TOKEN_VOCAB_SIZE + 1, (nb, C), g) - 1
            lab = zipf_ids(cfg.MAX_TARGET_VOCAB_SIZE + 1, (nb,), g) - 1
            src_l, pth_l, tgt_l = src.tolist(), pth.tolist(), tgt.tolist()
            lab_l, nc_l = lab.tolist(), n_ctx.tolist()
            out = []
            for r in range(nb):
                n = nc_l[r]
                ctxs = ['%s,%s,%s' % (_tok_word(src_l[r][c]),
                                      _path_word(pth_l[r][c]),
                                      _tok_word(tgt_l[r][c]))
                        for c in range(n)]
                out.append(_target_word(lab_l[r]) + ' ' + ' '.join(ctxs)
                           + ' ' * (C - n) + '\n')
            f.writelines(out)


def run_end_to_end(args, device, rank, world_size, reducer, distributed):
    """BASELINE end-to-end mode: reader -> H2D -> train step on on-disk
    java14m-shaped data with the real vocab-scale string->index maps
    (VERDICT r01 missing #3: nothing measured reader+H2D+step together)."""
    import json as _json
    from types import SimpleNamespace

    from code2vec_amd.data.prefetcher import BatchPrefetcher
    from code2vec_amd.data.reader import EstimatorAction, PathContextReader
    from code2vec_amd.vocabularies import (
        Vocab, VocabType, _SpecialVocabWords_JoinedOovPad)

    # small CPU ops (reader filters, pool shuffle slicing) thrash when
    # torch's intra-op pool spans a 256-core box; the GPU does the math here
    torch.set_num_threads(min(8, os.cpu_count() or 8))
    cfg = make_config(device)
    cfg.SAMPLED_SOFTMAX_SIZE = args.sampled_softmax
    # prefetch depth A/B on one box (r02_call10): 8 -> 244K ex/s,
    # 24 -> 228K; the deeper queue only added pinned-memory pressure
    cfg.READER_QUEUE_DEPTH = int(os.environ.get('C2V_E2E_QDEPTH', 8))
    n_gpus = world_size if distributed else 1

    data_path = os.path.join(os.environ.get('TMPDIR', '/tmp'),
                             'c2v_e2e.train.c2v')
    if rank == 0 and (not os.path.isfile(data_path)
                      or os.path.getsize(data_path) < 1000):
        t0 = time.perf_counter()
        generate_e2e_dataset(data_path, cfg, args.e2e_rows, seed=7)
        print('# generated %s (%.1f MB) in %.1fs'
              % (data_path, os.path.getsize(data_path) / 1e6,
                 time.perf_counter() - t0), flush=True)
    if distributed:
        import torch.distributed as dist
        dist.barrier()

    # full-vocab-scale string->index maps (rank->word is the id mapping the
    # generator used, so lookups resolve to the same Zipf-shaped ids)
    def make_vocab(vt, n, word_fn):
        # default regime: joined <PAD_OR_OOV> for every vocab
        return Vocab(vt, [word_fn(i) for i in range(n)],
                     _SpecialVocabWords_JoinedOovPad)

    t0 = time.perf_counter()
    vocabs = SimpleNamespace(
        token_vocab=make_vocab(VocabType.Token, cfg.MAX_TOKEN_VOCAB_SIZE,
                               _tok_word),
        path_vocab=make_vocab(VocabType.Path, cfg.MAX_PATH_VOCAB_SIZE,
                              _path_word),
        target_vocab=make_vocab(VocabType.Target, cfg.MAX_TARGET_VOCAB_SIZE,
                                _target_word))
    if rank == 0:
        print('# vocabs built in %.1fs' % (time.perf_counter() - t0),
              flush=True)

    net = Code2VecNetwork(cfg,
                          token_vocab_size=vocabs.token_vocab.size,
                          path_vocab_size=vocabs.path_vocab.size,
                          target_vocab_size=vocabs.target_vocab.size,
                          device=device)
    # capture the graph BEFORE the reader/prefetcher threads start: graph
    # capture and background pin_memory/H2D activity must not overlap
    use_graph = device.startswith('cuda') and not distributed \
        and not args.no_graph
    if use_graph:
        graph_step = net.make_graph_step(args.batch)

    reader = PathContextReader(vocabs=vocabs, config=cfg,
                               estimator_action=EstimatorAction.Train,
                               repeat_endlessly=True,
                               world_size=world_size, rank=rank)
    prefetcher = BatchPrefetcher(reader.iter_batches(data_path=data_path),
                                 device, depth=cfg.READER_QUEUE_DEPTH)
    batches = iter(prefetcher)

    def do_step(b):
        if b.source_token_indices.shape[0] != args.batch:
            return  # ragged shuffle-tail batch: skip for fixed-shape timing
        if use_graph:
            graph_step.step(b.source_token_indices, b.path_indices,
                            b.target_token_indices, b.context_valid_mask,
                            b.target_index)
        else:
            net.train_step(b.source_token_indices, b.path_indices,
                           b.target_token_indices, b.context_valid_mask,
                           b.target_index, reducer=reducer)

    def barrier_sync():
        if distributed:
            import torch.distributed as dist
            dist.barrier()
        if device.startswith('cuda'):
            torch.cuda.synchronize()

    if args.e2e_reader_only:
        # isolate the data pipeline: consume batches without stepping
        def do_step(b):  # noqa: F811
            return None

    for _ in range(args.warmup):
        do_step(next(batches))
    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        do_step(next(batches))
    barrier_sync()
    elapsed = time.perf_counter() - t0
    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.startswith('cuda') else 'cpu')
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        global_batch = args.batch * n_gpus
        ex_per_sec = global_batch * args.steps / elapsed
        print(_json.dumps({
            'metric': 'train_examples_per_sec',
            'value': round(ex_per_sec, 1),
            'unit': 'examples/s',
            'n_gpus': n_gpus,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(elapsed / args.steps * 1000.0, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': round(ex_per_sec / BASELINE_V100_EX_PER_SEC, 2),
            'dtype': cfg.COMPUTE_DTYPE,
            'data': 'synthetic-on-disk (end-to-end reader->H2D->step)',
            'config': {
                'model': 'code2vec-java14m',
                'global_batch': global_batch,
                'seq_len': cfg.MAX_CONTEXTS,
                'parallelism': 'dp%d' % n_gpus,
                'softmax': ('sampled-%d' % args.sampled_softmax)
                           if args.sampled_softmax else 'full',
                'stepping': 'hipgraph' if use_graph else 'eager',
                'token_vocab': cfg.MAX_TOKEN_VOCAB_SIZE,
                'path_vocab': cfg.MAX_PATH_VOCAB_SIZE,
                'target_vocab': cfg.MAX_TARGET_VOCAB_SIZE,
                'id_dist': 'zipf-%s' % ZIPF_S,
                'end_to_end': True,
                'reader_only': bool(args.e2e_reader_only),
            },
        }))
    # orderly shutdown: signal the reader threads, then join the prefetch
    # worker (a daemon thread still inside the C++ parser at interpreter
    # finalization aborts the process)
    reader.stop_streaming(join=False)
    prefetcher.stop()
    reader.stop_streaming(join=True)
    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument('--gpus', type=int, default=1)
    ap.add_argument('--steps', type=int, default=30)
    ap.add_argument('--warmup', type=int, default=10)
    ap.add_argument('--batch', type=int, default=1024, help='per-GPU batch size')
    ap.add_argument('--no-graph', action='store_true',
                    help='disable hipGraph-captured stepping (single-GPU only)')
    ap.add_argument('--sampled-softmax', type=int, default=0,
                    help='train with sampled softmax over N negatives '
                         '(BASELINE config 4); 0 = full softmax (default)')
    ap.add_argument('--burnin', type=int,
                    default=int(os.environ.get('C2V_BENCH_BURNIN', 400)),
                    help='untimed pre-warmup steps (device-busy telemetry)')
    ap.add_argument('--end-to-end', action='store_true',
                    help='time reader->H2D->step on generated on-disk data '
                         'instead of resident synthetic batches')
    ap.add_argument('--e2e-rows', type=int, default=250000,
                    help='rows of on-disk data to generate for --end-to-end')
    ap.add_argument('--e2e-reader-only', action='store_true',
                    help='with --end-to-end: consume batches without '
                         'stepping (isolates the data-pipeline ceiling)')
    args = ap.parse_args()

    # --gpus N without a torchrun rendezvous: self-launch one rank per GPU
    # (a single process would otherwise run 1 GPU while reporting N-GPU
    # aggregate throughput)
    if args.gpus > 1 and 'WORLD_SIZE' not in os.environ:
        import socket
        import subprocess
        import sys
        # pick a free rendezvous port so concurrent bench runs don't collide
        with socket.socket() as s:
            s.bind(('127.0.0.1', 0))
            port = s.getsockname()[1]
        cmd = [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
               '--nproc-per-node', str(args.gpus),
               '--master-addr', '127.0.0.1', '--master-port', str(port),
               os.path.abspath(__file__)] + sys.argv[1:]
        raise SystemExit(subprocess.call(cmd))

    world_size = int(os.environ.get('WORLD_SIZE', '1'))
    rank = int(os.environ.get('RANK', '0'))
    local_rank = int(os.environ.get('LOCAL_RANK', str(rank)))
    distributed = world_size > 1

    if torch.cuda.is_available():
        # modulo lets N ranks share fewer GPUs (RCCL-path validation on a
        # 1-GPU box); on a full node local_rank < device_count anyway
        dev_idx = local_rank % max(1, torch.cuda.device_count())
        torch.cuda.set_device(dev_idx)
        device = 'cuda:%d' % dev_idx
    else:
        device = 'cpu'

    reducer = NullReducer()
    if distributed:
        import torch.distributed as dist
        from code2vec_amd.parallel.ddp import Reducer
        backend = 'nccl' if device.startswith('cuda') else 'gloo'
        dist.init_process_group(backend=backend, rank=rank, world_size=world_size)
        reducer = Reducer(assume_equal_shards=True)  # fixed per-rank batch

    if args.end_to_end:
        run_end_to_end(args, device, rank, world_size, reducer, distributed)
        return

    cfg = make_config(device)
    cfg.SAMPLED_SOFTMAX_SIZE = args.sampled_softmax
    n_gpus = world_size if distributed else args.gpus
    net = Code2VecNetwork(cfg,
                          token_vocab_size=cfg.MAX_TOKEN_VOCAB_SIZE + 1,
                          path_vocab_size=cfg.MAX_PATH_VOCAB_SIZE + 1,
                          target_vocab_size=cfg.MAX_TARGET_VOCAB_SIZE + 1,
                          device=device)
    batches = synth_batches(cfg, device, args.batch, seed=1234 + rank)

    # hipGraph-captured stepping: the whole train step (fwd+bwd+Adam) replays
    # as one graph launch. Single-process path only; DP keeps eager launches
    # (they overlap the RCCL collectives).
    use_graph = (device.startswith('cuda') and not distributed
                 and not args.no_graph)
    if use_graph:
        graph_step = net.make_graph_step(args.batch)

        def do_step(b):
            graph_step.step(*b)
    else:
        def do_step(b):
            net.train_step(*b, reducer=reducer)

    def barrier_sync():
        if distributed:
            import torch.distributed as dist
            dist.barrier()
        if device.startswith('cuda'):
            torch.cuda.synchronize()

    # burn-in: a couple of seconds of untimed GPU work before the measured
    # window so out-of-band telemetry (SMI sampling by the bench driver)
    # observes a busy device even when the timed region is sub-second
    for i in range(args.burnin):
        do_step(batches[i % len(batches)])
    for i in range(args.warmup):
        do_step(batches[i % len(batches)])
    barrier_sync()

    t0 = time.perf_counter()
    for i in range(args.steps):
        do_step(batches[i % len(batches)])
    barrier_sync()
    elapsed = time.perf_counter() - t0

    if distributed:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.startswith('cuda') else 'cpu')
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        global_batch = args.batch * n_gpus
        ex_per_sec = global_batch * args.steps / elapsed
        out = {
            'metric': 'train_examples_per_sec',
            'value': round(ex_per_sec, 1),
            'unit': 'examples/s',
            'n_gpus': n_gpus,
            'steps': args.steps,
            'warmup': args.warmup,
            'ms_per_step': round(elapsed / args.steps * 1000.0, 3),
            'higher_is_better': True,
            'scaling': 'weak',
            'vs_baseline': round(ex_per_sec / BASELINE_V100_EX_PER_SEC, 2),
            'dtype': cfg.COMPUTE_DTYPE,
            'data': 'synthetic',
            'config': {
                'model': 'code2vec-java14m',
                'global_batch': global_batch,
                'seq_len': cfg.MAX_CONTEXTS,
                'parallelism': 'dp%d' % n_gpus,
                'softmax': ('sampled-%d' % args.sampled_softmax)
                           if args.sampled_softmax else 'full',
                'stepping': 'hipgraph' if use_graph else 'eager',
                'token_vocab': cfg.MAX_TOKEN_VOCAB_SIZE,
                'path_vocab': cfg.MAX_PATH_VOCAB_SIZE,
                'target_vocab': cfg.MAX_TARGET_VOCAB_SIZE,
                'id_dist': ('uniform'
                            if os.environ.get('C2V_BENCH_ID_DIST') == 'uniform'
                            else 'zipf-%s' % ZIPF_S),
            },
        }
        print(json.dumps(out))

    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == '__main__':
    main()
