"""Background batch prefetcher: reader thread + bounded queue + async H2D.

The reference reader's analog knobs: parallel parse calls (READER_NUM_PARALLEL
_BATCHES feeds the C++ parser's thread pool) and prefetch depth
(READER_QUEUE_DEPTH ≙ tf.data prefetch, path_context_reader.py:150).
On GPU the copies run on a dedicated HIP stream with pinned sources so they
overlap the previous step's compute; `next()` returns device tensors plus an
event the compute stream waits on."""

import queue
import threading
from typing import Iterator, Optional

import torch

from .reader import ReaderBatch


class BatchPrefetcher:
    _SENTINEL = object()

    def __init__(self, batch_iter: Iterator[ReaderBatch], device,
                 depth: int = 8):
        self.device = torch.device(device)
        self.use_gpu = self.device.type == 'cuda'
        self.copy_stream = torch.cuda.Stream() if self.use_gpu else None
        self.q: "queue.Queue" = queue.Queue(maxsize=max(2, depth))
        self._err: Optional[BaseException] = None
        self._stopping = False
        self._thread = threading.Thread(target=self._worker,
                                        args=(batch_iter,), daemon=True)
        self._thread.start()

    def _worker(self, batch_iter):
        try:
            for batch in batch_iter:
                placed = False
                while not self._stopping and not placed:
                    try:
                        self.q.put(batch, timeout=0.5)
                        placed = True
                    except queue.Full:
                        pass
                if self._stopping:
                    break
        except BaseException as e:  # noqa: BLE001
            self._err = e
        finally:
            # blocking-with-stop-check: the sentinel MUST reach the consumer
            # on normal completion (a dropped sentinel deadlocks the train
            # loop); under stop() the queue is being drained anyway
            while not self._stopping:
                try:
                    self.q.put(self._SENTINEL, timeout=0.5)
                    break
                except queue.Full:
                    pass

    def stop(self, timeout: float = 5.0):
        """Terminate the worker before interpreter shutdown (a daemon thread
        still executing extension code during finalization aborts the
        process). The underlying reader should be signalled to stop first so
        the worker's batch_iter returns."""
        self._stopping = True
        try:
            while True:
                self.q.get_nowait()
        except queue.Empty:
            pass
        self._thread.join(timeout=timeout)

    def __iter__(self):
        while True:
            item = self.q.get()
            if item is self._SENTINEL:
                if self._err is not None:
                    raise self._err
                return
            if not self.use_gpu:
                yield item
                continue
            with torch.cuda.stream(self.copy_stream):
                moved = item.to(self.device, non_blocking=True)
                event = torch.cuda.Event()
                event.record(self.copy_stream)
            cur = torch.cuda.current_stream()
            cur.wait_event(event)
            # The device tensors were allocated under copy_stream; mark them
            # in use by the consumer stream so the caching allocator cannot
            # hand their blocks to a LATER H2D copy while this stream still
            # reads them (without this, a fast reader recycles a batch
            # mid-step: garbage ids -> embedding-gather memory fault).
            for t in moved:
                if torch.is_tensor(t) and t.is_cuda:
                    t.record_stream(cur)
            yield moved
