"""Sample-message channel interface (parity: reference
python/channel/base.py:25-45).  A SampleMessage is a flat Dict[str, Tensor];
glt_amd.distributed.message defines the key conventions."""
from typing import Dict

import torch

SampleMessage = Dict[str, torch.Tensor]


class ChannelBase:
    def send(self, msg: SampleMessage):
        raise NotImplementedError

    def recv(self, timeout_ms: int = -1) -> SampleMessage:
        raise NotImplementedError

    def empty(self) -> bool:
        raise NotImplementedError
