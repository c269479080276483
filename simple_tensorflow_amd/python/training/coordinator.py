"""Coordinator + QueueRunner (analogs of reference python/training/
coordinator.py:32 and queue_runner_impl.py:30): python threads drive enqueue
subgraphs; the coordinator fans stop requests and re-raises exceptions."""
import contextlib
import sys
import threading
import time

from simple_tensorflow_amd.python.framework import errors, ops


class Coordinator(object):
    def __init__(self, clean_stop_exception_types=None):
        self._stop_event = threading.Event()
        self._lock = threading.Lock()
        self._exc_info = None
        self._registered = []
        self._clean_types = tuple(clean_stop_exception_types or
                                  (errors.OutOfRangeError,))

    def should_stop(self):
        return self._stop_event.is_set()

    def request_stop(self, ex=None):
        with self._lock:
            if ex is not None and self._exc_info is None and \
                    not isinstance(ex, self._clean_types):
                if isinstance(ex, tuple):
                    self._exc_info = ex
                else:
                    self._exc_info = (type(ex), ex, None)
        self._stop_event.set()

    def wait_for_stop(self, timeout=None):
        return self._stop_event.wait(timeout)

    def register_thread(self, thread):
        with self._lock:
            self._registered.append(thread)

    def join(self, threads=None, stop_grace_period_secs=120):
        threads = list(threads or []) + self._registered
        if not threads:
            self._stop_event.wait()
        for t in threads:
            t.join(stop_grace_period_secs)
        with self._lock:
            if self._exc_info is not None:
                et, ev, tb = self._exc_info
                raise ev

    @contextlib.contextmanager
    def stop_on_exception(self):
        try:
            yield
        except Exception as ex:  # noqa: BLE001
            self.request_stop(ex)

    def clear_stop(self):
        self._stop_event.clear()
        with self._lock:
            self._exc_info = None


class QueueRunner(object):
    def __init__(self, queue=None, enqueue_ops=None, close_op=None,
                 cancel_op=None, queue_closed_exception_types=None):
        self._queue = queue
        self._enqueue_ops = list(enqueue_ops or [])
        self._close_op = close_op
        self._cancel_op = cancel_op
        self._closed_types = tuple(queue_closed_exception_types or
                                   (errors.OutOfRangeError,
                                    errors.CancelledError))
        self._threads = []

    @property
    def queue(self):
        return self._queue

    def create_threads(self, sess, coord=None, daemon=False, start=False):
        threads = []
        for op in self._enqueue_ops:
            t = threading.Thread(target=self._run, args=(sess, op, coord))
            t.daemon = daemon
            threads.append(t)
        if coord is not None:
            for t in threads:
                coord.register_thread(t)
        if start:
            for t in threads:
                t.start()
        self._threads = threads
        return threads

    def _run(self, sess, enqueue_op, coord):
        try:
            while True:
                if coord is not None and coord.should_stop():
                    break
                try:
                    sess.run(enqueue_op)
                except self._closed_types:
                    break
                except Exception as ex:  # noqa: BLE001
                    if coord is not None:
                        coord.request_stop(ex)
                        break
                    raise
        finally:
            if coord is not None and coord.should_stop() and \
                    self._close_op is not None:
                try:
                    sess.run(self._close_op)
                except Exception:  # noqa: BLE001
                    pass


def add_queue_runner(qr, collection=ops.GraphKeys.QUEUE_RUNNERS):
    ops.get_default_graph().add_to_collection(collection, qr)


def start_queue_runners(sess, coord=None, daemon=True, start=True,
                        collection=ops.GraphKeys.QUEUE_RUNNERS):
    threads = []
    for qr in ops.get_default_graph().get_collection(collection):
        threads.extend(qr.create_threads(sess, coord=coord, daemon=daemon,
                                         start=start))
    return threads
