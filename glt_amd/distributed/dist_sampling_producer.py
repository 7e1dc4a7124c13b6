"""Sampling producers: spawned subprocess pool (mp) and in-process
(collocated) variants streaming SampleMessages into a channel.

Parity: reference python/distributed/dist_sampling_producer.py:54-365.
Each mp worker forms its own RPC group with the other trainers' sampling
workers, owns a DistNeighborSampler on its device, and pulls seed batches
from its pre-split shard; an '#END' marker is sent per worker per epoch.
"""
import queue
import threading
from enum import Enum
from typing import List, Optional, Union

import torch
import torch.multiprocessing as mp

from ..channel import ChannelBase, ShmChannel
from ..sampler import (EdgeSamplerInput, NodeSamplerInput, SamplingConfig,
                       SamplingType)
from .dist_context import get_context, init_worker_group
from .dist_dataset import DistDataset
from .dist_neighbor_sampler import DistNeighborSampler
from .dist_options import (CollocatedDistSamplingWorkerOptions,
                           MpDistSamplingWorkerOptions)
from .message import END_KEY
from .rpc import init_rpc, shutdown_rpc


class _Cmd(Enum):
    SAMPLE_ALL = 1
    STOP = 2


def _sampling_worker_loop(worker_idx: int, dataset: DistDataset,
                          seeds_input, sampling_config: SamplingConfig,
                          worker_options, device_str: str,
                          channel: ChannelBase, task_queue, done_counter,
                          ctx_info):
    """Body of one spawned sampling worker process."""
    try:
        (world_size, rank, group_name, num_workers) = ctx_info
        # each sampling worker joins the cross-trainer sampling RPC world
        init_worker_group(world_size * num_workers,
                          rank * num_workers + worker_idx,
                          group_name=f"{group_name}_sampling")
        device = torch.device(device_str)
        if device.type == "cuda":
            torch.cuda.set_device(device)
        init_rpc(worker_options.master_addr,
                 worker_options.master_port,
                 worker_options.num_rpc_threads,
                 worker_options.rpc_timeout)
        # optional all2all transport: sampling workers form their own
        # torch.distributed world (gloo — feature rows are served from
        # host memory); worker i of every trainer makes a subgroup, so
        # collectives pair partition-to-partition at equal worker index.
        a2a_group = None
        use_a2a = (getattr(worker_options, "use_all2all", False)
                   and world_size > 1
                   and worker_options.master_port is not None
                   and sampling_config.sampling_type == SamplingType.NODE
                   and sampling_config.collect_features
                   and not isinstance(dataset.graph, dict)
                   and dataset.num_partitions == world_size)
        if use_a2a:
            import datetime

            import torch.distributed as dist

            a2a_port = getattr(worker_options, "a2a_port", None) or \
                int(worker_options.master_port) + 1
            try:
                if not dist.is_initialized():
                    dist.init_process_group(
                        "gloo",
                        init_method="tcp://%s:%d" % (
                            worker_options.master_addr, a2a_port),
                        world_size=world_size * num_workers,
                        rank=rank * num_workers + worker_idx,
                        timeout=datetime.timedelta(seconds=120))
                groups = [dist.new_group([p * num_workers + i
                                          for p in range(world_size)])
                          for i in range(num_workers)]
                a2a_group = groups[worker_idx]
            except Exception:  # rendezvous failed: pure-RPC fallback
                a2a_group = None
                use_a2a = False
        sampler = DistNeighborSampler(
            dataset, sampling_config.num_neighbors,
            with_edge=sampling_config.with_edge,
            with_weight=sampling_config.with_weight,
            collect_features=sampling_config.collect_features,
            edge_dir=sampling_config.edge_dir, device=device,
            concurrency=worker_options.worker_concurrency, channel=channel,
            use_all2all=use_a2a, all2all_group=a2a_group)
        while True:
            cmd = task_queue.get()
            if cmd == _Cmd.STOP:
                break
            # sample the whole assigned shard for this epoch
            bs = sampling_config.batch_size
            n = len(seeds_input)
            if a2a_group is not None:
                # lockstep handshake: collective batches = the minimum
                # epoch batch count across the subgroup; the tail RPCs
                import torch.distributed as dist

                nb = (n // bs) if sampling_config.drop_last else \
                    (n + bs - 1) // bs
                cnt = torch.tensor([nb])
                outs = [torch.zeros_like(cnt) for _ in range(world_size)]
                dist.all_gather(outs, cnt, group=a2a_group)
                sampler.set_all2all_budget(
                    min(int(t.item()) for t in outs))
            order = torch.randperm(n) if sampling_config.shuffle else \
                torch.arange(n)
            futures = []
            for s in range(0, n, bs):
                if sampling_config.drop_last and s + bs > n:
                    break
                batch = seeds_input[order[s:s + bs]]
                if sampling_config.sampling_type == SamplingType.NODE:
                    fut = sampler.sample_from_nodes(batch)
                elif sampling_config.sampling_type == SamplingType.LINK:
                    fut = sampler.sample_from_edges(batch)
                elif sampling_config.sampling_type == SamplingType.SUBGRAPH:
                    fut = sampler.subgraph(batch)
                else:
                    raise ValueError(sampling_config.sampling_type)
                futures.append(fut)
            errors = []
            for f in futures:
                try:
                    f.result()
                except Exception:  # noqa: BLE001
                    import traceback

                    errors.append(traceback.format_exc())
            # END must flow even on failure or the trainer hangs forever
            channel.send({END_KEY: torch.tensor([1])})
            with done_counter.get_lock():
                done_counter.value += 1
            if errors:
                import sys

                print(f"[glt_amd sampling worker {worker_idx}] "
                      f"{len(errors)} batch(es) failed:\n{errors[0]}",
                      file=sys.stderr, flush=True)
        sampler.shutdown()
        if a2a_group is not None:
            import torch.distributed as dist

            try:
                dist.destroy_process_group()
            except Exception:
                pass
        shutdown_rpc()
    except KeyboardInterrupt:
        pass


class DistMpSamplingProducer:
    """Spawns N sampling subprocesses feeding one shm channel."""

    def __init__(self, data: DistDataset, seeds_input,
                 sampling_config: SamplingConfig,
                 worker_options: MpDistSamplingWorkerOptions,
                 channel: ShmChannel):
        self.data = data
        self.seeds_input = seeds_input
        self.sampling_config = sampling_config
        self.worker_options = worker_options
        self.channel = channel
        self.num_workers = worker_options.num_workers
        self._procs: List[mp.Process] = []
        self._task_queues = []
        self._done = None
        self.num_expected_ends = self.num_workers

    def init(self):
        ctx = get_context()
        mp_ctx = mp.get_context("spawn")
        self._done = mp_ctx.Value("i", 0)
        devices = self.worker_options._resolve_devices()
        # pre-split seeds across workers
        n = len(self.seeds_input)
        per = (n + self.num_workers - 1) // self.num_workers
        for w in range(self.num_workers):
            shard = self.seeds_input[
                torch.arange(w * per, min((w + 1) * per, n))]
            tq = mp_ctx.Queue()
            self._task_queues.append(tq)
            p = mp_ctx.Process(
                target=_sampling_worker_loop,
                args=(w, self.data, shard, self.sampling_config,
                      self.worker_options, str(devices[w]), self.channel,
                      tq, self._done,
                      (ctx.world_size, ctx.rank, ctx.group_name,
                       self.num_workers)),
                daemon=True)
            p.start()
            self._procs.append(p)

    def produce_all(self):
        """Kick one epoch of sampling on every worker."""
        with self._done.get_lock():
            self._done.value = 0
        for tq in self._task_queues:
            tq.put(_Cmd.SAMPLE_ALL)

    def is_all_sampling_completed(self) -> bool:
        with self._done.get_lock():
            return self._done.value >= self.num_workers

    def shutdown(self):
        for tq in self._task_queues:
            try:
                tq.put(_Cmd.STOP)
            except Exception:
                pass
        for p in self._procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()
        self._procs = []


class DistCollocatedSamplingProducer:
    """Synchronous in-process sampling (no channel round trip)."""

    def __init__(self, data: DistDataset, seeds_input,
                 sampling_config: SamplingConfig,
                 worker_options: CollocatedDistSamplingWorkerOptions,
                 device: Optional[torch.device] = None):
        self.data = data
        self.seeds_input = seeds_input
        self.config = sampling_config
        self.worker_options = worker_options
        self.device = device
        self.sampler: Optional[DistNeighborSampler] = None
        self._order = None
        self._pos = 0

    def init(self):
        self.sampler = DistNeighborSampler(
            self.data, self.config.num_neighbors,
            with_edge=self.config.with_edge,
            with_weight=self.config.with_weight,
            collect_features=self.config.collect_features,
            edge_dir=self.config.edge_dir, device=self.device,
            concurrency=self.worker_options.worker_concurrency,
            channel=None,
            use_all2all=getattr(self.worker_options, "use_all2all",
                                False))

    def reset(self):
        n = len(self.seeds_input)
        self._order = torch.randperm(n) if self.config.shuffle else \
            torch.arange(n)
        self._pos = 0

    def next_batch(self):
        n = len(self.seeds_input)
        if self._pos >= n:
            return None
        end = min(self._pos + self.config.batch_size, n)
        if self.config.drop_last and end - self._pos < \
                self.config.batch_size:
            return None
        batch = self.seeds_input[self._order[self._pos:end]]
        self._pos = end
        if self.config.sampling_type == SamplingType.NODE:
            fut = self.sampler.sample_from_nodes(batch)
        elif self.config.sampling_type == SamplingType.LINK:
            fut = self.sampler.sample_from_edges(batch)
        elif self.config.sampling_type == SamplingType.SUBGRAPH:
            fut = self.sampler.subgraph(batch)
        else:
            raise ValueError(self.config.sampling_type)
        return fut.result()

    def shutdown(self):
        if self.sampler is not None:
            self.sampler.shutdown()
