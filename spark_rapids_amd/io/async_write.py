"""Async write throttle (reference analogue: the async-write holding /
throttling logic around GpuFileFormatWriter): writes run on a background
thread behind a bounded queue so compute overlaps file IO, with the
queue bound applying back-pressure instead of unbounded host buffering.
"""
from __future__ import annotations

import queue
import threading
from typing import Callable, Optional


class AsyncWriter:
    """Run write tasks on one background thread; submit() blocks when
    more than `max_pending` writes are in flight (the throttle)."""

    def __init__(self, max_pending: int = 2):
        self._q: "queue.Queue" = queue.Queue(maxsize=max_pending)
        self._err: Optional[BaseException] = None
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def _run(self):
        while True:
            task = self._q.get()
            if task is None:
                return
            try:
                task()
            except BaseException as e:  # noqa: BLE001 - surfaced on close
                self._err = e

    def submit(self, task: Callable[[], None]):
        if self._err:
            raise self._err
        self._q.put(task)

    def close(self):
        self._q.put(None)
        self._thread.join()
        if self._err:
            raise self._err
