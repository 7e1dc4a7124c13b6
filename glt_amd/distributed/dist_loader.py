"""DistLoader family: collocated / mp-subprocess / remote-server sampling
modes behind one iterator (parity: reference
python/distributed/dist_loader.py:127-451)."""
from typing import List, Optional, Union

import torch

from ..channel import (ChannelBase, RemoteReceivingChannel, ShmChannel)
from ..loader.transform import to_data, to_hetero_data
from ..sampler import (EdgeSamplerInput, HeteroSamplerOutput,
                       NodeSamplerInput, SamplerOutput, SamplingConfig,
                       SamplingType)
from ..utils.common import parse_size
from .dist_context import get_context
from .dist_dataset import DistDataset
from .dist_options import (AllDistSamplingWorkerOptions,
                           CollocatedDistSamplingWorkerOptions,
                           MpDistSamplingWorkerOptions,
                           RemoteDistSamplingWorkerOptions)
from .dist_sampling_producer import (DistCollocatedSamplingProducer,
                                     DistMpSamplingProducer)
from .message import END_KEY, decode_sample_message
from .rpc import init_rpc


class DistLoader:
    def __init__(self, data: Optional[DistDataset], input_data,
                 sampling_config: SamplingConfig,
                 to_device: Optional[torch.device] = None,
                 worker_options: Optional[AllDistSamplingWorkerOptions]
                 = None):
        self.data = data
        self.input_data = input_data
        self.sampling_config = sampling_config
        self.to_device = to_device
        self.worker_options = worker_options or \
            CollocatedDistSamplingWorkerOptions()
        self.edge_dir = sampling_config.edge_dir
        self._channel: Optional[ChannelBase] = None
        self._producer = None
        self._ends_seen = 0
        self._epoch = 0
        # zero-copy safety: decoded messages alias the shm ring; their
        # blocks must stay referenced until the async H2D copies complete
        import collections as _c

        self._inflight = _c.deque()

        if isinstance(self.worker_options,
                      CollocatedDistSamplingWorkerOptions):
            self._mode = "collocated"
            self._init_collocated()
        elif isinstance(self.worker_options, MpDistSamplingWorkerOptions):
            self._mode = "mp"
            self._init_mp()
        elif isinstance(self.worker_options,
                        RemoteDistSamplingWorkerOptions):
            self._mode = "remote"
            self._init_remote()
        else:
            raise ValueError("unknown worker options")

    # -- init per mode ------------------------------------------------------
    def _init_collocated(self):
        opts = self.worker_options
        if opts.master_port is not None:
            init_rpc(opts.master_addr, opts.master_port,
                     opts.num_rpc_threads, opts.rpc_timeout)
        self._producer = DistCollocatedSamplingProducer(
            self.data, self.input_data, self.sampling_config, opts,
            device=self.to_device)
        self._producer.init()

    def _init_mp(self):
        opts = self.worker_options
        self._channel = ShmChannel(opts.channel_capacity,
                                   opts.channel_size)
        if opts.pin_memory:
            self._channel.pin_memory()
        self._producer = DistMpSamplingProducer(
            self.data, self.input_data, self.sampling_config, opts,
            self._channel)
        self._producer.init()

    def _init_remote(self):
        from . import dist_client

        opts = self.worker_options
        server_ranks = opts.server_rank
        if server_ranks is None:
            server_ranks = dist_client.get_assigned_servers()
        elif isinstance(server_ranks, int):
            server_ranks = [server_ranks]
        self._server_ranks = server_ranks
        producer_ids = {}
        for s in server_ranks:
            pid = dist_client.request_server(
                s, "create_sampling_producer",
                self.input_data, self.sampling_config,
                opts.num_workers, opts.buffer_capacity,
                parse_size(opts.buffer_size), opts.worker_key,
                opts.worker_concurrency)
            producer_ids[s] = pid
        self._producer_ids = producer_ids

        def fetch(server_rank, producer_id):
            return dist_client.async_request_server(
                server_rank, "fetch_one_sampled_message", producer_id)

        self._channel = RemoteReceivingChannel(
            server_ranks, producer_ids, fetch,
            prefetch_size=opts.prefetch_size)

    # -- iteration ----------------------------------------------------------
    def __iter__(self):
        self._ends_seen = 0
        self._epoch += 1
        if self._mode == "collocated":
            self._producer.reset()
        elif self._mode == "mp":
            # drain stale messages from an abandoned epoch, then start anew
            try:
                while not self._channel.empty():
                    self._channel.recv(timeout_ms=100)
            except Exception:
                pass
            self._producer.produce_all()
        else:
            from . import dist_client

            for s in self._server_ranks:
                dist_client.request_server(s, "start_new_epoch_sampling",
                                           self._producer_ids[s])
            self._channel.reset()
        return self

    def __next__(self):
        if self._mode == "collocated":
            out = self._producer.next_batch()
            if out is None:
                raise StopIteration
            msg = out  # already an encoded SampleMessage
            return self._collate(msg)
        # channel modes
        while True:
            msg = self._channel.recv(timeout_ms=300_000)
            if END_KEY in msg:
                if self._mode == "remote":
                    raise StopIteration
                self._ends_seen += 1
                if self._ends_seen >= self._producer.num_expected_ends:
                    raise StopIteration
                continue
            return self._collate(msg)

    def __len__(self):
        n = len(self.input_data)
        bs = self.sampling_config.batch_size
        if self.sampling_config.drop_last:
            return n // bs
        return (n + bs - 1) // bs

    def _collate(self, msg):
        out, x, y, ea = decode_sample_message(msg)
        if isinstance(out, SamplerOutput):
            data = to_data(out, batch_labels=y, node_feats=x, edge_feats=ea)
        else:
            data = to_hetero_data(out, batch_label_dict=y,
                                  node_feat_dict=x, edge_feat_dict=ea,
                                  edge_dir=self.edge_dir)
        if self.to_device is not None:
            on_gpu = (torch.device(self.to_device).type == "cuda")
            data = data.to(self.to_device, non_blocking=True)
            if on_gpu:
                ev = torch.cuda.Event()
                ev.record()
                self._inflight.append((msg, ev))
                while self._inflight and self._inflight[0][1].query():
                    self._inflight.popleft()
        return data

    def shutdown(self):
        if self._producer is not None:
            self._producer.shutdown()
        if self._mode == "remote":
            from . import dist_client

            for s in self._server_ranks:
                try:
                    dist_client.request_server(
                        s, "destroy_sampling_producer",
                        self._producer_ids[s])
                except Exception:
                    pass

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass
