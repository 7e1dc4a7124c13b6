"""Concurrent asyncio event loop on a background thread (parity: reference
python/distributed/event_loop.py:39-102)."""
import asyncio
import threading
from typing import Coroutine, Optional


def wrap_torch_future(loop: asyncio.AbstractEventLoop, torch_future):
    """Bridge a torch.futures.Future into an awaitable asyncio future."""
    af = loop.create_future()

    def _done(f):
        try:
            v = f.value()
            loop.call_soon_threadsafe(af.set_result, v)
        except Exception as e:  # noqa: BLE001
            loop.call_soon_threadsafe(af.set_exception, e)

    torch_future.add_done_callback(_done)
    return af


class ConcurrentEventLoop:
    def __init__(self, concurrency: int = 4):
        self.concurrency = concurrency
        self._loop = asyncio.new_event_loop()
        self._sem = None
        self._thread: Optional[threading.Thread] = None

    def start_loop(self):
        if self._thread is not None:
            return

        def run():
            asyncio.set_event_loop(self._loop)
            self._sem = asyncio.Semaphore(self.concurrency)
            self._loop.run_forever()

        self._thread = threading.Thread(target=run, daemon=True,
                                        name="glt-sampler-loop")
        self._thread.start()
        while self._sem is None:
            pass  # tiny spin until loop thread is live

    def shutdown_loop(self):
        if self._thread is None:
            return
        self._loop.call_soon_threadsafe(self._loop.stop)
        self._thread.join(timeout=10)
        self._thread = None

    @property
    def loop(self):
        return self._loop

    def add_task(self, coro: Coroutine, callback=None):
        """Schedule a coroutine bounded by the concurrency semaphore;
        returns a concurrent.futures.Future."""

        async def guarded():
            async with self._sem:
                return await coro

        fut = asyncio.run_coroutine_threadsafe(guarded(), self._loop)
        if callback is not None:
            fut.add_done_callback(callback)
        return fut

    def run_task(self, coro: Coroutine):
        """Schedule and wait."""
        return self.add_task(coro).result()
