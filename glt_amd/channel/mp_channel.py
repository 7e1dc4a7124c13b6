"""torch.multiprocessing queue channel (parity: reference
python/channel/mp_channel.py)."""
import torch.multiprocessing as mp

from .base import ChannelBase, SampleMessage


class MpChannel(ChannelBase):
    def __init__(self, capacity: int = 128):
        ctx = mp.get_context("spawn")
        self._queue = ctx.Queue(maxsize=capacity)

    def send(self, msg: SampleMessage):
        self._queue.put(msg)

    def recv(self, timeout_ms: int = -1) -> SampleMessage:
        if timeout_ms is None or timeout_ms < 0:
            return self._queue.get()
        return self._queue.get(timeout=timeout_ms / 1000.0)

    def empty(self) -> bool:
        return self._queue.empty()
