"""SHM sample-queue tests: cross-process FIFO, zero-copy release, timeout,
wrap-around (property-style)."""
import multiprocessing as mp
import time

import pytest
import torch

from glt_amd import _C


def test_fifo_and_zero_copy():
    q = _C.SampleQueue(4, 1 << 16)
    for i in range(4):
        q.send([("i", torch.tensor([i])), ("x", torch.full((8,), float(i)))])
    assert q.pending() == 4
    for i in range(4):
        msg = dict(q.receive(1000))
        assert msg["i"].item() == i
        assert (msg["x"] == float(i)).all()
    assert q.pending() == 0


def test_timeout():
    q = _C.SampleQueue(2, 1 << 12)
    with pytest.raises(_C.QueueTimeoutError):
        q.receive(100)


def test_ring_wraparound_many_messages():
    # ring much smaller than total traffic: forces wrap + tail-skip paths
    q = _C.SampleQueue(3, 4096)
    for i in range(200):
        q.send([("t", torch.full((100,), float(i)))])
        msg = dict(q.receive(1000))
        assert (msg["t"] == float(i)).all()


def _producer(shmid, n):
    import torch
    from glt_amd import _C as C

    q = C.SampleQueue(shmid)
    for i in range(n):
        q.send([("seq", torch.tensor([i])),
                ("payload", torch.arange(i % 50, dtype=torch.float32))])


def test_cross_process():
    ctx = mp.get_context("spawn")
    q = _C.SampleQueue(8, 1 << 16)
    n = 64
    p = ctx.Process(target=_producer, args=(q.shmid, n))
    p.start()
    got = []
    for _ in range(n):
        msg = dict(q.receive(20000))
        got.append(msg["seq"].item())
        assert msg["payload"].numel() == msg["seq"].item() % 50
    p.join(10)
    assert got == list(range(n))


def _producer_multi(shmid, worker, n):
    import torch
    from glt_amd import _C as C

    q = C.SampleQueue(shmid)
    for i in range(n):
        q.send([("w", torch.tensor([worker])), ("i", torch.tensor([i]))])


def test_multi_producer():
    ctx = mp.get_context("spawn")
    q = _C.SampleQueue(8, 1 << 16)
    workers, per = 3, 20
    procs = [ctx.Process(target=_producer_multi, args=(q.shmid, w, per))
             for w in range(workers)]
    for p in procs:
        p.start()
    seen = {w: [] for w in range(workers)}
    for _ in range(workers * per):
        msg = dict(q.receive(30000))
        seen[msg["w"].item()].append(msg["i"].item())
    for p in procs:
        p.join(10)
    for w in range(workers):
        assert seen[w] == list(range(per))  # per-producer order preserved
