"""Client side of the server-client mode (parity: reference
python/distributed/dist_client.py:24-101)."""
from typing import Any, List, Optional

import torch.distributed.rpc as torch_rpc

from .dist_context import (_set_client_context, assign_server_by_order,
                           get_context)
from .rpc import init_rpc, shutdown_rpc

_assigned_servers: List[int] = []


def init_client(num_servers: int, num_clients: int, client_rank: int,
                master_addr: str, master_port: int,
                num_rpc_threads: int = 4, rpc_timeout: float = 240.0,
                client_group_name: str = "distributed_client",
                is_dynamic: bool = False,
                server_group_name: str = "distributed_server"):
    """is_dynamic worlds resolve peer names through the registry living
    on `{server_group_name}_0` (the rendezvous anchor)."""
    global _assigned_servers
    _set_client_context(num_servers, num_clients, client_rank,
                        client_group_name)
    init_rpc(master_addr, master_port, num_rpc_threads, rpc_timeout,
             is_dynamic=is_dynamic,
             anchor_name=f"{server_group_name}_0" if is_dynamic else None)
    _assigned_servers = assign_server_by_order(client_rank, num_servers,
                                               num_clients)


def get_assigned_servers() -> List[int]:
    return list(_assigned_servers)


def _server_name(server_rank: int) -> str:
    from .dist_context import DistRole
    from .rpc import group_worker_name

    return group_worker_name(server_rank, DistRole.SERVER)


def request_server(server_rank: int, func_name: str, *args, **kwargs):
    from .dist_server import _call_func_on_server

    return torch_rpc.rpc_sync(_server_name(server_rank),
                              _call_func_on_server,
                              args=(func_name, args, kwargs))


def async_request_server(server_rank: int, func_name: str, *args, **kwargs):
    from .dist_server import _call_func_on_server

    return torch_rpc.rpc_async(_server_name(server_rank),
                               _call_func_on_server,
                               args=(func_name, args, kwargs))


def shutdown_client(exit_servers: bool = True):
    ctx = get_context()
    if exit_servers and ctx is not None and ctx.rank == 0:
        for s in range(_num_servers()):
            try:
                request_server(s, "exit")
            except Exception:
                pass
    shutdown_rpc()


def _num_servers() -> int:
    ctx = get_context()
    return ctx.global_world_size - ctx.world_size if ctx else 0
