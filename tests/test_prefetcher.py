"""BatchPrefetcher contract: full drain + sentinel delivery (a dropped
sentinel deadlocks the train loop — regression test for the r02 bug), and
mid-stream stop() joins the worker."""

import time

import torch

from code2vec_amd.data.prefetcher import BatchPrefetcher


def _batches(n, delay=0.0):
    for i in range(n):
        if delay:
            time.sleep(delay)
        yield torch.full((4,), i)


def test_yields_everything_then_terminates():
    pf = BatchPrefetcher(_batches(57), 'cpu', depth=4)
    got = [int(b[0]) for b in pf]
    assert got == list(range(57))


def test_sentinel_survives_full_queue():
    # producer finishes while the bounded queue is full: the consumer must
    # still see the end of stream (not hang)
    pf = BatchPrefetcher(_batches(9), 'cpu', depth=2)
    time.sleep(1.0)          # let the worker fill the queue and finish
    got = [int(b[0]) for b in pf]
    assert got == list(range(9))


def test_stop_midstream_joins_worker():
    pf = BatchPrefetcher(_batches(10_000, delay=0.001), 'cpu', depth=2)
    it = iter(pf)
    for _ in range(5):
        next(it)
    pf.stop()
    assert not pf._thread.is_alive()
