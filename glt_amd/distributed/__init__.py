from .dist_context import (DistContext, DistRole, assign_server_by_order,
                           get_context, init_worker_group)
from .dist_dataset import DistDataset
from .dist_feature import DistFeature
from .dist_graph import DistGraph
from .dist_loader import DistLoader
from .dist_neighbor_loader import (DistLinkNeighborLoader,
                                   DistNeighborLoader, DistSubGraphLoader)
from .dist_neighbor_sampler import DistNeighborSampler
from .dist_options import (CollocatedDistSamplingWorkerOptions,
                           MpDistSamplingWorkerOptions,
                           RemoteDistSamplingWorkerOptions)
from .dist_random_partitioner import (DistHeteroRandomPartitioner,
                                      DistRandomPartitioner)
from .dist_sampling_producer import (DistCollocatedSamplingProducer,
                                     DistMpSamplingProducer)
from .dist_server import (DistServer, get_server, init_server,
                          wait_and_shutdown_server)
from .dist_client import (async_request_server, init_client,
                          request_server, shutdown_client)
from .event_loop import ConcurrentEventLoop
from .pyg_remote_backend import RemoteFeatureStore, RemoteGraphStore
from .message import decode_sample_message, encode_sampler_output
from .rpc import (barrier, init_rpc, rpc_is_initialized, rpc_register,
                  rpc_request_async, shutdown_rpc)

__all__ = [
    "DistContext", "DistRole", "assign_server_by_order", "get_context",
    "init_worker_group", "DistDataset", "DistFeature", "DistGraph",
    "DistLoader", "DistLinkNeighborLoader", "DistNeighborLoader",
    "DistSubGraphLoader", "DistNeighborSampler", "DistRandomPartitioner",
    "DistHeteroRandomPartitioner",
    "CollocatedDistSamplingWorkerOptions", "MpDistSamplingWorkerOptions",
    "RemoteDistSamplingWorkerOptions", "DistCollocatedSamplingProducer",
    "DistMpSamplingProducer", "DistServer", "get_server", "init_server",
    "wait_and_shutdown_server", "async_request_server", "init_client",
    "request_server", "shutdown_client", "ConcurrentEventLoop",
    "RemoteFeatureStore", "RemoteGraphStore",
    "decode_sample_message", "encode_sampler_output", "barrier",
    "init_rpc", "rpc_is_initialized", "rpc_register", "rpc_request_async",
    "shutdown_rpc",
]
