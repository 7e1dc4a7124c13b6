"""Client-side receiving channel pulling sampled messages from remote
servers over RPC with prefetch (parity: reference
python/channel/remote_channel.py:24-131).

Protocol: `fetch_fn(server_rank, producer_id)` returns a SampleMessage or
an end-of-epoch marker dict {'#END': tensor(epoch)}; the channel keeps
`prefetch_size` requests in flight per server and raises StopIteration to
its consumer (via the '#END' message) only after EVERY server has signaled
end-of-epoch.
"""
import collections
import threading
from typing import Callable, Dict, List

import torch

from .base import ChannelBase, SampleMessage

END_KEY = "#END"


class RemoteReceivingChannel(ChannelBase):
    def __init__(self, server_ranks: List[int], producer_ids: Dict[int, int],
                 fetch_fn: Callable, prefetch_size: int = 4):
        self.server_ranks = server_ranks
        self.producer_ids = producer_ids  # server_rank -> producer id
        self.fetch_fn = fetch_fn
        self.prefetch_size = prefetch_size
        self._lock = threading.Lock()
        self._futures = collections.deque()
        self._ended = set()

    def reset(self):
        with self._lock:
            self._futures.clear()
            self._ended.clear()

    def _issue(self, server_rank):
        fut = self.fetch_fn(server_rank, self.producer_ids[server_rank])
        self._futures.append((server_rank, fut))

    def _prime(self):
        with self._lock:
            if not self._futures and len(self._ended) < len(
                    self.server_ranks):
                for s in self.server_ranks:
                    if s in self._ended:
                        continue
                    for _ in range(self.prefetch_size):
                        self._issue(s)

    def recv(self, timeout_ms: int = -1) -> SampleMessage:
        while True:
            self._prime()
            with self._lock:
                if not self._futures:
                    # all servers ended
                    return {END_KEY: torch.tensor([1])}
                server, fut = self._futures.popleft()
            msg = fut.wait() if hasattr(fut, "wait") else fut
            if isinstance(msg, dict) and END_KEY in msg:
                with self._lock:
                    self._ended.add(server)
                    # drop outstanding futures for that server
                    self._futures = collections.deque(
                        (s, f) for s, f in self._futures if s != server)
                if len(self._ended) >= len(self.server_ranks):
                    return {END_KEY: torch.tensor([1])}
                continue
            with self._lock:
                if server not in self._ended:
                    self._issue(server)
            return msg

    def send(self, msg):
        raise RuntimeError("RemoteReceivingChannel is receive-only")

    def empty(self) -> bool:
        return not self._futures
