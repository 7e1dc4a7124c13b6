"""RPC layer over torch.distributed.rpc (TensorPipe).

Parity: reference python/distributed/rpc.py — role-scoped gather/barrier,
dynamic-world init with server wait, callee registry, partition router.
Control traffic stays on TCP; bulk tensors move over RCCL collectives or
the shm channels, not through here.
"""
import atexit
import threading
import time
from typing import Any, Callable, Dict, List, Optional

import torch.distributed.rpc as torch_rpc

from .dist_context import DistContext, DistRole, get_context

_rpc_inited = False
_rpc_lock = threading.RLock()
# name -> DistContext of every peer (gathered at init)
_rpc_worker_names: Dict[str, "DistContext"] = {}
_rpc_current_group_names: List[str] = []

MAX_RETRY = 60
RETRY_INTERVAL = 2.0


def rpc_is_initialized() -> bool:
    return _rpc_inited


def _ctx_info():
    ctx = get_context()
    assert ctx is not None, "init_worker_group/init_server/init_client first"
    return ctx


def init_rpc(master_addr: str, master_port: int,
             num_rpc_threads: int = 16, rpc_timeout: float = 240.0,
             is_dynamic: bool = False):
    """Bring up torch rpc for all processes of the global world."""
    global _rpc_inited
    with _rpc_lock:
        if _rpc_inited:
            return
        ctx = _ctx_info()
        options = torch_rpc.TensorPipeRpcBackendOptions(
            _transports=["uv"],
            num_worker_threads=num_rpc_threads,
            rpc_timeout=rpc_timeout,
            init_method=f"tcp://{master_addr}:{master_port}")
        torch_rpc.init_rpc(
            name=ctx.worker_name,
            rank=ctx.global_rank,
            world_size=None if is_dynamic else ctx.global_world_size,
            rpc_backend_options=options)
        _rpc_inited = True
        atexit.register(shutdown_rpc)
        # gather every peer's context (for name resolution)
        if not is_dynamic:
            global _rpc_worker_names, _rpc_current_group_names
            infos = all_gather(ctx)
            _rpc_worker_names = {c.worker_name: c for c in infos.values()}
            _rpc_current_group_names = sorted(
                (c.worker_name for c in infos.values()
                 if c.role == ctx.role),
                key=lambda n: _rpc_worker_names[n].rank)


def shutdown_rpc(graceful: bool = True):
    global _rpc_inited
    with _rpc_lock:
        if _rpc_inited:
            try:
                torch_rpc.shutdown(graceful=graceful)
            except Exception:
                pass
            _rpc_inited = False


def global_world_size() -> int:
    return _ctx_info().global_world_size


def rpc_worker_names() -> Dict[str, "DistContext"]:
    return _rpc_worker_names


def group_worker_name(role_rank: int, role: Optional[DistRole] = None) -> str:
    """Name of the peer with `role_rank` inside its role group."""
    ctx = _ctx_info()
    role = role or ctx.role
    for name, c in _rpc_worker_names.items():
        if c.role == role and c.rank == role_rank:
            return name
    # fallback to canonical group names (dynamic worlds skip the gather)
    if role == ctx.role:
        return f"{ctx.group_name}_{role_rank}"
    canonical = {DistRole.WORKER: "distributed_worker",
                 DistRole.SERVER: "distributed_server",
                 DistRole.CLIENT: "distributed_client"}
    return f"{canonical[role]}_{role_rank}"


# ---------------------------------------------------------------------------
# leader-based role/global gather + barrier (reference rpc.py:136-233)
# ---------------------------------------------------------------------------

_gather_state: Dict[str, Dict[int, Any]] = {}
_gather_events: Dict[str, threading.Event] = {}
_gather_lock = threading.Lock()


def _gather_push(tag: str, rank: int, obj: Any, expected: int):
    with _gather_lock:
        st = _gather_state.setdefault(tag, {})
        st[rank] = obj
        if len(st) >= expected:
            _gather_events.setdefault(tag, threading.Event()).set()


def _gather_pull(tag: str):
    with _gather_lock:
        return dict(_gather_state.get(tag, {}))


_gather_seq = 0


def all_gather(obj: Any, timeout: float = 300.0) -> Dict[int, Any]:
    """Gather obj from every process of the GLOBAL world; returns
    {global_rank: obj}."""
    global _gather_seq
    ctx = _ctx_info()
    _gather_seq += 1
    tag = f"g{_gather_seq}"
    world = ctx.global_world_size
    leader = _leader_name()
    if ctx.global_rank == 0:
        _gather_push(tag, 0, obj, world)
    else:
        torch_rpc.rpc_sync(leader, _gather_push,
                           args=(tag, ctx.global_rank, obj, world),
                           timeout=timeout)
    if ctx.global_rank == 0:
        ev = None
        deadline = time.time() + timeout
        while time.time() < deadline:
            with _gather_lock:
                st = _gather_state.get(tag, {})
                if len(st) >= world:
                    break
            time.sleep(0.01)
        result = _gather_pull(tag)
        assert len(result) >= world, f"gather timeout: {len(result)}/{world}"
        return result
    # non-leader: poll leader for the full map
    deadline = time.time() + timeout
    while time.time() < deadline:
        result = torch_rpc.rpc_sync(leader, _gather_pull, args=(tag,),
                                    timeout=timeout)
        if len(result) >= world:
            return result
        time.sleep(0.05)
    raise TimeoutError("all_gather timeout")


def _leader_name() -> str:
    # global rank 0 is always a worker (worker mode) or server 0
    for name, c in _rpc_worker_names.items():
        if c.global_rank == 0:
            return name
    ctx = _ctx_info()
    if ctx.role == DistRole.WORKER:
        return f"{ctx.group_name}_0"
    return "distributed_server_0"


def barrier(timeout: float = 300.0):
    all_gather(None, timeout=timeout)


# ---------------------------------------------------------------------------
# callee registry (reference rpc.py:419-473)
# ---------------------------------------------------------------------------

class RpcCalleeBase:
    def call(self, *args, **kwargs):
        raise NotImplementedError


_callee_registry: Dict[int, RpcCalleeBase] = {}
_callee_counter = threading.Lock()
_callee_next_id = [0]


def rpc_register(callee: RpcCalleeBase) -> int:
    """Register a callee on THIS process; every process must register its
    callees in the same order so ids line up across the fleet."""
    with _callee_counter:
        cid = _callee_next_id[0]
        _callee_next_id[0] += 1
    _callee_registry[cid] = callee
    return cid


def _rpc_call(callee_id: int, *args, **kwargs):
    return _callee_registry[callee_id].call(*args, **kwargs)


def rpc_request_async(target_name_or_rank, callee_id: int, args=(),
                      kwargs=None):
    target = (group_worker_name(target_name_or_rank)
              if isinstance(target_name_or_rank, int) else
              target_name_or_rank)
    return torch_rpc.rpc_async(target, _rpc_call,
                               args=(callee_id, *args),
                               kwargs=kwargs or {})


def rpc_request_sync(target_name_or_rank, callee_id: int, args=(),
                     kwargs=None):
    return rpc_request_async(target_name_or_rank, callee_id, args,
                             kwargs).wait()


def rpc_global_request_async(global_rank: int, func: Callable, args=()):
    for name, c in _rpc_worker_names.items():
        if c.global_rank == global_rank:
            return torch_rpc.rpc_async(name, func, args=args)
    raise ValueError(f"no rpc peer with global rank {global_rank}")


# ---------------------------------------------------------------------------
# data-partition routing (reference rpc.py:364-414)
# ---------------------------------------------------------------------------

class RpcDataPartitionRouter:
    """Round-robins sampling requests over the workers serving each data
    partition."""

    def __init__(self, partition2workers: List[List[str]]):
        self.partition2workers = partition2workers
        self._next = [0] * len(partition2workers)

    def get_to_worker(self, partition_idx: int) -> str:
        workers = self.partition2workers[partition_idx]
        i = self._next[partition_idx]
        self._next[partition_idx] = (i + 1) % len(workers)
        return workers[i]


def rpc_sync_data_partitions(num_data_partitions: int,
                             current_partition_idx: int):
    """Gather (worker_name -> partition) across the role group and build
    partition -> [worker names] (reference rpc.py:385-414)."""
    ctx = _ctx_info()
    infos = all_gather((ctx.worker_name, ctx.role.value,
                        num_data_partitions, current_partition_idx))
    p2w: List[List[str]] = [[] for _ in range(num_data_partitions)]
    for _, (name, role, nparts, pidx) in sorted(infos.items()):
        if role == ctx.role.value and pidx >= 0:
            p2w[pidx].append(name)
    return RpcDataPartitionRouter(p2w)
