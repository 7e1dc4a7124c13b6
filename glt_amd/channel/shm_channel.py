"""Shared-memory channel over the native SampleQueue ring
(parity: reference python/channel/shm_channel.py:24-66).

Zero-copy on receive; `pin_memory()` hipHostRegisters the whole ring so
received tensors H2D-copy at pinned bandwidth.
"""
from typing import Optional, Union

from ..utils.common import parse_size
from .base import ChannelBase, SampleMessage


class ShmChannel(ChannelBase):
    def __init__(self, capacity: int = 128,
                 shm_size: Union[str, int] = "256MB",
                 shmid: Optional[int] = None):
        from .. import _C

        self._C = _C
        if shmid is not None:
            self._queue = _C.SampleQueue(shmid)
        else:
            self._queue = _C.SampleQueue(capacity, parse_size(shm_size))
        self._pinned = False

    @property
    def shmid(self) -> int:
        return self._queue.shmid

    def pin_memory(self):
        if not self._pinned:
            try:
                self._queue.pin_memory()
                self._pinned = True
            except Exception:
                pass  # no GPU on this box

    def send(self, msg: SampleMessage):
        self._queue.send(list(msg.items()))

    def recv(self, timeout_ms: int = -1) -> SampleMessage:
        return dict(self._queue.receive(timeout_ms))

    def empty(self) -> bool:
        return self._queue.empty()

    def __reduce__(self):
        return (ShmChannel, (0, 0, self.shmid))
