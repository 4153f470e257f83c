"""Data loader interfaces (reference: horovod/data/data_loader_base.py).

BaseDataLoader is the abstract loader contract; AsyncDataLoaderMixin adds a
background-thread prefetch queue so host-side batch preparation overlaps the
training step (on MI355X the HBM3E capacity makes deep prefetch cheap).
"""
import queue
from threading import Thread


class BaseDataLoader:
    def __len__(self):
        raise NotImplementedError()

    def _iterate(self):
        """Yield batches; subclasses implement the actual iteration."""
        raise NotImplementedError()

    def __iter__(self):
        for batch in self._iterate():
            yield self._process_batch(batch)

    def _process_batch(self, batch):
        return batch


class AsyncDataLoaderMixin:
    """Mix in BEFORE the loader class:

        class AsyncXLoader(AsyncDataLoaderMixin, XLoader): pass

    Prefetches `async_loader_queue_size` batches on a worker thread
    (reference: data_loader_base.py AsyncDataLoaderMixin).
    """

    def __init__(self, async_loader_queue_size=64, *args, **kwargs):
        self.async_loader_queue_size = async_loader_queue_size
        self.started = False
        self.finished_loading_event = None
        self.queue = None
        self.thread = None
        super().__init__(*args, **kwargs)

    def close_async_loader(self):
        if self.started and self.thread is not None:
            self._closing = True
            try:
                while True:
                    self.queue.get_nowait()
            except queue.Empty:
                pass
            self.thread.join()
            self.started = False

    def _async_worker(self):
        try:
            while not self._closing:
                for batch in super()._iterate():
                    if self._closing:
                        break
                    self.queue.put(batch)
                if not self._repeat:
                    break
        except Exception as e:  # propagate to consumer
            self.queue.put(e)
        finally:
            self.queue.put(None)

    def _iterate(self):
        if self.async_loader_queue_size <= 0:
            yield from super()._iterate()
            return
        if not self.started:
            self.started = True
            self._closing = False
            self._repeat = False
            self.queue = queue.Queue(self.async_loader_queue_size)
            self.thread = Thread(target=self._async_worker, daemon=True)
            self.thread.start()
        while True:
            batch = self.queue.get()
            if batch is None:
                self.started = False
                self.thread.join()
                break
            if isinstance(batch, Exception):
                self.started = False
                raise batch
            yield batch
