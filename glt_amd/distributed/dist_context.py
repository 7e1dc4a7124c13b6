"""Distributed role/context management (parity: reference
python/distributed/dist_context.py:20-104, 140-213)."""
import os
from dataclasses import dataclass
from enum import Enum
from typing import List, Optional


class DistRole(Enum):
    WORKER = 1
    SERVER = 2
    CLIENT = 3


@dataclass
class DistContext:
    role: DistRole
    world_size: int          # size of this role's group
    rank: int                # rank within this role's group
    global_world_size: int   # all processes across roles
    global_rank: int
    group_name: str

    @property
    def is_worker(self):
        return self.role == DistRole.WORKER

    @property
    def is_server(self):
        return self.role == DistRole.SERVER

    @property
    def is_client(self):
        return self.role == DistRole.CLIENT

    @property
    def worker_name(self) -> str:
        return f"{self.group_name}_{self.rank}"


_dist_context: Optional[DistContext] = None


def get_context() -> Optional[DistContext]:
    return _dist_context


def _set_context(ctx: DistContext):
    global _dist_context
    _dist_context = ctx


def init_worker_group(world_size: int, rank: int,
                      group_name: str = "distributed_worker"):
    _set_context(DistContext(
        role=DistRole.WORKER, world_size=world_size, rank=rank,
        global_world_size=world_size, global_rank=rank,
        group_name=group_name))


def _set_server_context(num_servers: int, server_rank: int,
                        num_clients: int = 0,
                        group_name: str = "distributed_server"):
    _set_context(DistContext(
        role=DistRole.SERVER, world_size=num_servers, rank=server_rank,
        global_world_size=num_servers + num_clients,
        global_rank=server_rank, group_name=group_name))


def _set_client_context(num_servers: int, num_clients: int,
                        client_rank: int,
                        group_name: str = "distributed_client"):
    _set_context(DistContext(
        role=DistRole.CLIENT, world_size=num_clients, rank=client_rank,
        global_world_size=num_servers + num_clients,
        global_rank=num_servers + client_rank, group_name=group_name))


def assign_server_by_order(client_rank: int, num_servers: int,
                           num_clients: int) -> List[int]:
    """Round-robin client -> server assignment (parity: reference
    dist_context.py:174-196)."""
    if num_clients >= num_servers:
        return [client_rank % num_servers]
    # fewer clients than servers: each client gets a contiguous span
    per = num_servers // num_clients
    extra = num_servers % num_clients
    start = client_rank * per + min(client_rank, extra)
    count = per + (1 if client_rank < extra else 0)
    return list(range(start, start + count))
