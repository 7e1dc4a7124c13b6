"""Sampling worker options (parity: reference
python/distributed/dist_options.py:26-298)."""
import os
from dataclasses import dataclass, field
from typing import List, Optional, Union

import torch


class _BasicDistSamplingWorkerOptions:
    def __init__(self, num_workers: int = 1, worker_devices=None,
                 worker_concurrency: int = 4,
                 master_addr: Optional[str] = None,
                 master_port: Optional[Union[int, str]] = None,
                 num_rpc_threads: int = 16, rpc_timeout: float = 180.0,
                 use_all2all: bool = False):
        self.use_all2all = use_all2all
        self.num_workers = num_workers
        self.worker_devices = worker_devices
        self.worker_concurrency = min(max(worker_concurrency, 1), 32)
        self.master_addr = (master_addr or
                            os.environ.get("MASTER_ADDR", "127.0.0.1"))
        mp = master_port or os.environ.get("MASTER_PORT")
        self.master_port = int(mp) if mp is not None else None
        self.num_rpc_threads = num_rpc_threads
        self.rpc_timeout = rpc_timeout

    def _resolve_devices(self, base_device=None):
        if self.worker_devices is not None:
            devs = [torch.device(d) for d in self.worker_devices]
        elif base_device is not None and base_device.type == "cuda":
            devs = [base_device] * self.num_workers
        elif torch.cuda.is_available():
            n = torch.cuda.device_count()
            devs = [torch.device("cuda", i % n)
                    for i in range(self.num_workers)]
        else:
            devs = [torch.device("cpu")] * self.num_workers
        return devs


class CollocatedDistSamplingWorkerOptions(_BasicDistSamplingWorkerOptions):
    """Sample synchronously in the training process."""

    def __init__(self, master_addr=None, master_port=None,
                 num_rpc_threads: int = 16, rpc_timeout: float = 180.0,
                 use_all2all: bool = False):
        super().__init__(1, None, 1, master_addr, master_port,
                         num_rpc_threads, rpc_timeout, use_all2all)


class MpDistSamplingWorkerOptions(_BasicDistSamplingWorkerOptions):
    """Spawn `num_workers` sampling subprocesses streaming into a pinned shm
    channel (sampling <-> training pipeline parallelism)."""

    def __init__(self, num_workers: int = 1, worker_devices=None,
                 worker_concurrency: int = 4, master_addr=None,
                 master_port=None, num_rpc_threads: int = 16,
                 rpc_timeout: float = 180.0, channel_size="256MB",
                 channel_capacity: int = 128, pin_memory: bool = False,
                 use_all2all: bool = True,
                 a2a_port: Optional[int] = None):
        """use_all2all (default on): sampling workers exchange feature
        rows with torch.distributed all_to_all collectives instead of
        per-partition RPC pulls when the topology allows (one rank per
        partition, homo node sampling with feature collection).  Workers
        rendezvous their own process group on `a2a_port` (default
        master_port + 1) and enforce lockstep with a per-epoch
        batch-count handshake; the ragged tail and every unsupported
        shape fall back to the RPC path automatically."""
        super().__init__(num_workers, worker_devices, worker_concurrency,
                         master_addr, master_port, num_rpc_threads,
                         rpc_timeout, use_all2all)
        self.channel_size = channel_size
        self.channel_capacity = channel_capacity
        self.pin_memory = pin_memory
        self.a2a_port = a2a_port


class RemoteDistSamplingWorkerOptions(_BasicDistSamplingWorkerOptions):
    """Sampling runs on remote server(s); batches are pulled over RPC."""

    def __init__(self, server_rank=None, num_workers: int = 1,
                 worker_devices=None, worker_concurrency: int = 4,
                 master_addr=None, master_port=None,
                 num_rpc_threads: int = 16, rpc_timeout: float = 180.0,
                 buffer_size="256MB", buffer_capacity: int = 128,
                 prefetch_size: int = 4, glt_graph=None,
                 workload_type: Optional[str] = None,
                 worker_key: Optional[str] = None):
        super().__init__(num_workers, worker_devices, worker_concurrency,
                         master_addr, master_port, num_rpc_threads,
                         rpc_timeout)
        self.server_rank = server_rank
        self.buffer_size = buffer_size
        self.buffer_capacity = buffer_capacity
        self.prefetch_size = prefetch_size
        # accepted for reference signature compatibility; workload_type
        # namespaces the server-side producer key alongside worker_key
        self.glt_graph = glt_graph
        self.workload_type = workload_type
        self.worker_key = (f"{workload_type}:{worker_key}"
                           if workload_type and worker_key else
                           worker_key or workload_type or "default")


AllDistSamplingWorkerOptions = Union[CollocatedDistSamplingWorkerOptions,
                                     MpDistSamplingWorkerOptions,
                                     RemoteDistSamplingWorkerOptions]
