"""Thread pool executing unit runs.

Parity target: reference ``veles/thread_pool.py`` (``ThreadPool`` with
failure callbacks, pause/resume, ordered shutdown).  The MI355X engine
runs the unit graph sequentially by default (GPU work is stream-async, so
host threads add nothing to the hot path), but the pool is kept for
callback execution (snapshot writing, plotting) and API parity.
"""

import threading
import queue


class ThreadPool(object):
    def __init__(self, minthreads=1, maxthreads=4, name="veles"):
        self.name = name
        self.maxthreads = max(1, int(maxthreads))
        self._queue = queue.Queue()
        self._threads = []
        self._shutdown = threading.Event()
        self._paused = threading.Event()
        self._paused.clear()
        self._resume = threading.Event()
        self._resume.set()
        self._failure_callbacks = []
        self._lock = threading.Lock()
        self._started = False

    # -- lifecycle ---------------------------------------------------------
    def start(self):
        with self._lock:
            if self._started:
                return
            self._started = True
            for i in range(self.maxthreads):
                t = threading.Thread(target=self._worker,
                                     name="%s-%d" % (self.name, i),
                                     daemon=True)
                t.start()
                self._threads.append(t)

    def shutdown(self, execute_remaining=True, timeout=5.0):
        if not self._started:
            return
        if execute_remaining:
            self._queue.join()
        self._shutdown.set()
        for _ in self._threads:
            self._queue.put(None)
        for t in self._threads:
            t.join(timeout)
        self._threads = []
        self._started = False
        self._shutdown.clear()

    def pause(self):
        self._resume.clear()

    def resume(self):
        self._resume.set()

    # -- work --------------------------------------------------------------
    def callInThread(self, func, *args, **kwargs):  # noqa: N802 (ref name)
        self.start()
        self._queue.put((func, args, kwargs))

    submit = callInThread

    def register_on_failure(self, cb):
        self._failure_callbacks.append(cb)

    def _worker(self):
        while not self._shutdown.is_set():
            item = self._queue.get()
            if item is None:
                self._queue.task_done()
                break
            self._resume.wait()
            func, args, kwargs = item
            try:
                func(*args, **kwargs)
            except Exception as e:  # silent-failure protection
                handled = False
                for cb in self._failure_callbacks:
                    try:
                        cb(e)
                        handled = True
                    except Exception:
                        pass
                if not handled:
                    import traceback
                    traceback.print_exc()
            finally:
                self._queue.task_done()

    def join(self):
        self._queue.join()
