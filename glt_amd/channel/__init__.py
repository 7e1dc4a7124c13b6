from .base import ChannelBase, SampleMessage
from .shm_channel import ShmChannel
from .mp_channel import MpChannel
from .remote_channel import RemoteReceivingChannel

__all__ = ["ChannelBase", "SampleMessage", "ShmChannel", "MpChannel",
           "RemoteReceivingChannel"]
