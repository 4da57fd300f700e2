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
        self._thread = threading.Thread(target=self._worker,
                                        args=(batch_iter,), daemon=True)
        self._thread.start()

    def _worker(self, batch_iter):
        try:
            for batch in batch_iter:
                self.q.put(batch)
        except BaseException as e:  # noqa: BLE001
            self._err = e
        finally:
            self.q.put(self._SENTINEL)

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
            torch.cuda.current_stream().wait_event(event)
            yield moved
