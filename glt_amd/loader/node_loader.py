"""NodeLoader: iterate seed batches -> sampled PyG batches.

Parity: reference python/loader/node_loader.py:54-115 (seed DataLoader +
collate features/labels into Data/HeteroData).
"""
import threading
from typing import Optional, Union

import torch

from ..data import Dataset
from ..sampler import (BaseSampler, NodeSamplerInput, SamplerOutput,
                       HeteroSamplerOutput)
from ..utils.tracing import trace_region
from .transform import to_data, to_hetero_data


class _SeedIterator:
    def __init__(self, seeds: torch.Tensor, batch_size: int, shuffle: bool,
                 drop_last: bool, generator=None):
        self.batch_size = batch_size
        self.drop_last = drop_last
        n = seeds.numel()
        if shuffle:
            # materialize the permutation once: batches become
            # contiguous SLICES, which stay pinned when the seed tensor
            # is pinned (per-batch fancy indexing would return unpinned
            # copies and turn every seed upload into a blocking H2D)
            seeds = seeds[torch.randperm(n, generator=generator)
                          .to(seeds.device)]
            if not seeds.is_cuda and torch.cuda.is_available():
                try:
                    seeds = seeds.pin_memory()
                except RuntimeError:
                    pass
        self.seeds = seeds
        self.order = None
        self.pos = 0
        self.n = n

    def __iter__(self):
        return self

    def __next__(self):
        if self.pos >= self.n:
            raise StopIteration
        end = min(self.pos + self.batch_size, self.n)
        if self.drop_last and end - self.pos < self.batch_size:
            raise StopIteration
        idx = slice(self.pos, end)
        batch = (self.seeds[self.order[idx]] if self.order is not None
                 else self.seeds[idx])
        self.pos = end
        return batch

    def __len__(self):
        if self.drop_last:
            return self.n // self.batch_size
        return (self.n + self.batch_size - 1) // self.batch_size


class _Prefetcher:
    """Background sampling pipeline on a dedicated HIP stream.

    The producer thread samples/collates batches on `side_stream` and
    records an event per batch; the consumer makes its current stream wait
    on that event, so sampling kernels for batch i+1 overlap the training
    kernels of batch i on the GPU.
    """

    def __init__(self, loader, seed_iter, depth: int):
        import queue as _q

        self._queue: "_q.Queue" = _q.Queue(maxsize=depth)
        self._loader = loader
        self._seed_iter = seed_iter
        self._stop = False
        self._stream = torch.cuda.Stream()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="glt-prefetch")
        self._thread.start()

    def _run(self):
        try:
            with torch.cuda.stream(self._stream):
                for seeds in self._seed_iter:
                    if self._stop:
                        return
                    data = self._loader._produce(seeds)
                    ev = torch.cuda.Event()
                    ev.record(self._stream)
                    self._queue.put((data, ev, None))
            self._queue.put((None, None, None))
        except BaseException as e:  # noqa: BLE001
            self._queue.put((None, None, e))

    def next(self):
        data, ev, err = self._queue.get()
        if err is not None:
            raise err
        if data is None:
            raise StopIteration
        torch.cuda.current_stream().wait_event(ev)
        # allocations came from the side stream; register reuse on this one
        cur = torch.cuda.current_stream()

        def rec(t):
            if t.is_cuda:
                t.record_stream(cur)
            return t

        from ..pygcompat.data import apply_to_tensors

        apply_to_tensors(data, rec)
        return data

    def stop(self):
        """Stop and JOIN the producer thread.  A daemon thread killed
        mid-HIP-call at interpreter exit aborts the process ("terminate
        called without an active exception" in the HIP runtime
        teardown) — observed when the consumer drains batches faster
        than the producer makes them (sampler-only benchmarks)."""
        self._stop = True
        try:
            while True:
                self._queue.get_nowait()
        except Exception:
            pass
        self._thread.join(timeout=5)


class NodeLoader:
    """Args beyond the reference surface:
      prefetch: >0 enables pipelined sampling — a background thread samples
        and collates the next `prefetch` batches on a dedicated side HIP
        stream while the caller trains on the current batch (the
        single-process analogue of the reference's subprocess sampling
        pipeline)."""

    def __init__(self, data: Dataset, node_sampler: BaseSampler,
                 input_nodes, batch_size: int = 1, shuffle: bool = False,
                 drop_last: bool = False, with_edge: bool = False,
                 to_device: Optional[torch.device] = None,
                 prefetch: int = 0):
        self.data = data
        self.sampler = node_sampler
        self.input_nodes = NodeSamplerInput.cast(
            input_nodes if not isinstance(input_nodes, tuple)
            else NodeSamplerInput(node=input_nodes[1],
                                  input_type=input_nodes[0]))
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.drop_last = drop_last
        self.with_edge = with_edge
        self.to_device = to_device
        self.prefetch = prefetch
        self._prefetcher = None
        # pin CPU seed tensors once so per-batch seed uploads are true
        # async H2D (pageable copies stall the producer thread; the
        # hetero producer issues one per (hop, etype))
        node = self.input_nodes.node
        if (torch.cuda.is_available() and torch.is_tensor(node)
                and not node.is_cuda and not node.is_pinned()):
            try:
                self.input_nodes.node = node.pin_memory()
            except RuntimeError:
                pass

    def __iter__(self):
        self._it = _SeedIterator(self.input_nodes.node, self.batch_size,
                                 self.shuffle, self.drop_last)
        if self._prefetcher is not None:
            self._prefetcher.stop()
            self._prefetcher = None
        if self.prefetch > 0 and torch.cuda.is_available():
            self._prefetcher = _Prefetcher(self, self._it, self.prefetch)
        return self

    def __len__(self):
        return len(_SeedIterator(self.input_nodes.node, self.batch_size,
                                 False, self.drop_last))

    def _produce(self, seeds):
        inp = NodeSamplerInput(seeds, self.input_nodes.input_type)
        out = self.sampler.sample_from_nodes(inp)
        with trace_region("collate"):
            return self._collate_fn(out)

    def __next__(self):
        if self._prefetcher is not None:
            return self._prefetcher.next()
        return self._produce(next(self._it))

    def shutdown(self):
        """Stop the background prefetcher (called automatically on
        garbage collection; call explicitly before interpreter exit)."""
        if self._prefetcher is not None:
            self._prefetcher.stop()
            self._prefetcher = None

    def __del__(self):
        try:
            self.shutdown()
        except Exception:
            pass

    # -- feature/label collection ------------------------------------------
    def _collate_fn(self, out: Union[SamplerOutput, HeteroSamplerOutput]):
        if isinstance(out, HeteroSamplerOutput):
            return self._hetero_collate(out)
        x = None
        nf = self.data.get_node_feature()
        if nf is not None:
            x = nf[out.node]
        ef = None
        if self.data.get_edge_feature() is not None and out.edge is not None:
            ef = self.data.get_edge_feature()[out.edge]
        y = None
        labels = self.data.get_node_label()
        if labels is not None:
            seed_nodes = (out.batch if out.batch is not None else out.node)
            y = labels.to(seed_nodes.device)[seed_nodes] \
                if not labels.is_cuda else labels[seed_nodes]
        data = to_data(out, batch_labels=y, node_feats=x, edge_feats=ef)
        if self.to_device is not None:
            data = data.to(self.to_device, non_blocking=True)
        return data

    def _hetero_collate(self, out: HeteroSamplerOutput):
        node_feats = {}
        for nt, nodes in out.node.items():
            f = self.data.get_node_feature(nt)
            if f is not None and nodes.numel() > 0:
                node_feats[nt] = f[nodes]
        edge_feats = {}
        if out.edge is not None:
            for et, eids in out.edge.items():
                f = self.data.get_edge_feature(et)
                if f is not None and eids.numel() > 0:
                    edge_feats[et] = f[eids]
        labels = {}
        for nt, b in (out.batch or {}).items():
            lab = self.data.get_node_label(nt)
            if lab is not None:
                labels[nt] = lab.to(b.device)[b] if not lab.is_cuda \
                    else lab[b]
        data = to_hetero_data(out, batch_label_dict=labels or None,
                              node_feat_dict=node_feats or None,
                              edge_feat_dict=edge_feats or None,
                              edge_dir=getattr(self.sampler, "edge_dir",
                                               "out"))
        if self.to_device is not None:
            data = data.to(self.to_device, non_blocking=True)
        return data
