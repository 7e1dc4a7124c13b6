from .common import (
    assign_device,
    ensure_dir,
    get_free_port,
    id2idx,
    index_select,
    load_ckpt,
    merge_hetero_sampler_output,
    format_hetero_sampler_output,
    parse_size,
    save_ckpt,
    seed_everything,
    share_memory,
    tensor_equal_with_device,
)
from .exit_status import python_exit_status
from .mixin import CastMixin
from .singleton import Singleton
from .topo import coo_to_csr, coo_to_csc, sort_csr_indices
from .tracing import range_pop, range_push, trace_region

__all__ = [
    "assign_device", "ensure_dir", "get_free_port", "id2idx", "index_select",
    "load_ckpt", "merge_hetero_sampler_output", "format_hetero_sampler_output",
    "parse_size", "save_ckpt", "seed_everything", "share_memory", "tensor_equal_with_device",
    "python_exit_status", "CastMixin", "Singleton", "coo_to_csr", "coo_to_csc", "sort_csr_indices",
    "range_pop", "range_push", "trace_region",
]
