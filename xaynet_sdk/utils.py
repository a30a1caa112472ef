"""Client-side concurrency helper — the reference's `ConcurrentFutures`
analog (rust/xaynet-sdk/src/utils/concurrent_futures.rs): drive many
blocking tasks with bounded concurrency and collect results as they finish.
"""
from __future__ import annotations

import threading
from concurrent.futures import FIRST_COMPLETED, Future, ThreadPoolExecutor, wait
from typing import Callable, Iterable, Iterator, Tuple


def run_concurrently(tasks: Iterable[Callable], max_concurrency: int = 8
                     ) -> Iterator[Tuple[int, object, BaseException | None]]:
    """Run callables with at most `max_concurrency` in flight; yields
    (index, result, exception) in completion order."""
    tasks = list(tasks)
    with ThreadPoolExecutor(max_workers=max_concurrency) as pool:
        pending: dict[Future, int] = {}
        it = iter(enumerate(tasks))
        exhausted = False
        while pending or not exhausted:
            while not exhausted and len(pending) < max_concurrency:
                try:
                    i, fn = next(it)
                except StopIteration:
                    exhausted = True
                    break
                pending[pool.submit(fn)] = i
            if not pending:
                break
            done, _ = wait(pending, return_when=FIRST_COMPLETED)
            for fut in done:
                i = pending.pop(fut)
                exc = fut.exception()
                yield (i, None if exc else fut.result(), exc)


class Notifier:
    """threading.Event with a counter — handy for new-global-model waits."""

    def __init__(self):
        self._event = threading.Event()
        self._count = 0
        self._lock = threading.Lock()

    def set(self):
        with self._lock:
            self._count += 1
        self._event.set()

    def wait(self, timeout=None) -> bool:
        ok = self._event.wait(timeout)
        if ok:
            self._event.clear()
        return ok

    @property
    def count(self) -> int:
        with self._lock:
            return self._count
