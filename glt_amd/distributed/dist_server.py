"""DistServer: server-side of the server-client disaggregated mode
(parity: reference python/distributed/dist_server.py:50-296).

A server process owns a DistDataset and a pool of sampling producers keyed
by producer id; clients drive them over RPC (create / start epoch / fetch /
destroy) and can also use the PyG-remote-backend style getters."""
import threading
import time
from typing import Dict, Optional

import torch

from ..channel import ShmChannel
from ..sampler import SamplingConfig
from .dist_context import _set_server_context, get_context
from .dist_dataset import DistDataset
from .dist_sampling_producer import DistMpSamplingProducer
from .dist_options import MpDistSamplingWorkerOptions
from .message import END_KEY
from .rpc import init_rpc, shutdown_rpc

SERVER_EXIT_STATUS_CHECK_INTERVAL = 2.0

_dist_server: Optional["DistServer"] = None


def get_server() -> Optional["DistServer"]:
    return _dist_server


class DistServer:
    def __init__(self, dataset: DistDataset):
        self.dataset = dataset
        self._producers: Dict[int, DistMpSamplingProducer] = {}
        self._channels: Dict[int, ShmChannel] = {}
        self._ends: Dict[int, int] = {}
        self._epoch_done: Dict[int, bool] = {}
        self._fetch_locks: Dict[int, threading.Lock] = {}
        self._next_id = 0
        self._lock = threading.Lock()
        self._exit = False

    # -- lifecycle ----------------------------------------------------------
    def exit(self):
        self._exit = True
        for pid in list(self._producers):
            self.destroy_sampling_producer(pid)
        return True

    def wait_for_exit(self):
        while not self._exit:
            time.sleep(SERVER_EXIT_STATUS_CHECK_INTERVAL)

    # -- dataset meta / PyG remote backend surface --------------------------
    def get_dataset_meta(self):
        ds = self.dataset
        return {
            "num_partitions": ds.num_partitions,
            "partition_idx": ds.partition_idx,
            "edge_dir": ds.edge_dir,
            "is_hetero": isinstance(ds.graph, dict),
        }

    def get_node_feature(self, ids: torch.Tensor, ntype=None):
        f = self.dataset.get_node_feature(ntype)
        return f.cpu_get(ids) if f is not None else None

    def get_node_label(self, ids: torch.Tensor, ntype=None):
        lab = self.dataset.get_node_label(ntype)
        return lab[ids.cpu()] if lab is not None else None

    def get_edge_index(self, etype=None, layout: str = "COO"):
        g = self.dataset.get_graph(etype)
        rows, cols, _ = g.topo.to_coo()
        return torch.stack([rows, cols])

    def get_tensor_size(self, ntype=None):
        f = self.dataset.get_node_feature(ntype)
        return tuple(f.shape) if f is not None else None

    def get_node_partition_id(self, ids: torch.Tensor, ntype=None):
        pb = self.dataset.node_pb
        if isinstance(pb, dict):
            pb = pb.get(ntype)
        return pb[ids] if pb is not None else None

    # -- sampling producers --------------------------------------------------
    def create_sampling_producer(self, seeds_input,
                                 sampling_config: SamplingConfig,
                                 num_workers: int, buffer_capacity: int,
                                 buffer_bytes: int, worker_key: str,
                                 worker_concurrency: int = 4) -> int:
        from ..sampler import RemoteSamplerInput

        if isinstance(seeds_input, RemoteSamplerInput):
            seeds_input = seeds_input.resolve(self.dataset)
        with self._lock:
            pid = self._next_id
            self._next_id += 1
        channel = ShmChannel(buffer_capacity, buffer_bytes)
        opts = MpDistSamplingWorkerOptions(
            num_workers=num_workers,
            worker_concurrency=worker_concurrency,
            master_addr="127.0.0.1",
            master_port=_free_port(),
            channel_capacity=buffer_capacity)
        producer = DistMpSamplingProducer(
            self.dataset, seeds_input, sampling_config, opts, channel)
        producer.init()
        self._producers[pid] = producer
        self._channels[pid] = channel
        self._ends[pid] = 0
        self._epoch_done[pid] = True  # nothing sampled until epoch start
        self._fetch_locks[pid] = threading.Lock()
        return pid

    def start_new_epoch_sampling(self, producer_id: int):
        with self._fetch_locks[producer_id]:
            self._ends[producer_id] = 0
            self._epoch_done[producer_id] = False
            self._producers[producer_id].produce_all()
        return True

    def fetch_one_sampled_message(self, producer_id: int):
        """Returns the next SampleMessage, or {'#END'} once every worker of
        this producer has finished the epoch.  Serialized per producer so
        concurrent client prefetches cannot race the END accounting, and
        never blocks across an epoch boundary (a drained epoch returns END
        immediately to every outstanding fetch)."""
        producer = self._producers[producer_id]
        channel = self._channels[producer_id]
        with self._fetch_locks[producer_id]:
            while True:
                if self._epoch_done[producer_id] and channel.empty():
                    return {END_KEY: torch.tensor([1])}
                msg = channel.recv(timeout_ms=300_000)
                if END_KEY in msg:
                    self._ends[producer_id] += 1
                    if self._ends[producer_id] >= \
                            producer.num_expected_ends:
                        self._epoch_done[producer_id] = True
                        return {END_KEY: torch.tensor([1])}
                    continue
                return msg

    def destroy_sampling_producer(self, producer_id: int):
        p = self._producers.pop(producer_id, None)
        if p is not None:
            p.shutdown()
        self._channels.pop(producer_id, None)
        self._ends.pop(producer_id, None)
        self._epoch_done.pop(producer_id, None)
        self._fetch_locks.pop(producer_id, None)
        return True


def _free_port():
    from ..utils.common import get_free_port

    return get_free_port()


# -- module-level dispatch target for client RPCs ---------------------------

def _call_func_on_server(func_name: str, args, kwargs):
    server = get_server()
    assert server is not None, "server not initialized"
    return getattr(server, func_name)(*args, **kwargs)


def init_server(num_servers: int, server_rank: int, dataset: DistDataset,
                master_addr: str, master_port: int, num_clients: int = 0,
                num_rpc_threads: int = 16, rpc_timeout: float = 240.0,
                server_group_name: str = "distributed_server",
                is_dynamic: bool = False):
    global _dist_server
    _set_server_context(num_servers, server_rank, num_clients,
                        server_group_name)
    _dist_server = DistServer(dataset)
    init_rpc(master_addr, master_port, num_rpc_threads, rpc_timeout,
             is_dynamic=is_dynamic)


def wait_and_shutdown_server():
    server = get_server()
    if server is not None:
        server.wait_for_exit()
    shutdown_rpc()
