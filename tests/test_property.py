"""Property-based tests (hypothesis) for serialization, topology and
relabeling invariants."""
import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings

from glt_amd import _C
from glt_amd.data import Topology
from glt_amd.utils.topo import coo_to_csr


@settings(max_examples=30, deadline=None)
@given(st.lists(st.tuples(st.integers(1, 3000),
                          st.sampled_from(["float32", "int64", "uint8"])),
                min_size=1, max_size=8),
       st.integers(1, 5))
def test_sample_queue_roundtrip_property(specs, batches):
    q = _C.SampleQueue(4, 1 << 20)
    for b in range(batches):
        msg = []
        for i, (numel, dt) in enumerate(specs):
            t = (torch.arange(numel, dtype=torch.float32) * (b + 1)).to(
                getattr(torch, dt))
            msg.append((f"k{i}", t))
        q.send(msg)
        got = dict(q.receive(2000))
        assert set(got) == {f"k{i}" for i in range(len(specs))}
        for i, (numel, dt) in enumerate(specs):
            ref = (torch.arange(numel, dtype=torch.float32) * (b + 1)).to(
                getattr(torch, dt))
            assert torch.equal(got[f"k{i}"], ref)


@settings(max_examples=30, deadline=None)
@given(st.integers(2, 60), st.integers(1, 300), st.randoms())
def test_topology_coo_roundtrip_property(n, e, rnd):
    src = torch.tensor([rnd.randrange(n) for _ in range(e)])
    dst = torch.tensor([rnd.randrange(n) for _ in range(e)])
    w = torch.rand(e)
    topo = Topology(torch.stack([src, dst]), edge_weights=w, num_nodes=n)
    # edge multiset preserved, weights follow their edges via edge_ids
    rows, cols, eids = topo.to_coo()
    orig = sorted(zip(src.tolist(), dst.tolist()))
    got = sorted(zip(rows.tolist(), cols.tolist()))
    assert orig == got
    assert torch.allclose(topo.edge_weights, w[eids])
    # indptr consistent with degrees
    deg = torch.bincount(src, minlength=n)
    assert torch.equal(topo.degrees, deg)
    # indices sorted per row
    for v in range(n):
        seg = topo.indices[topo.indptr[v]:topo.indptr[v + 1]].tolist()
        assert seg == sorted(seg)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 40), st.integers(0, 4), st.randoms())
def test_inducer_matches_python_reference(n_seeds, hops, rnd):
    ind = _C.CPUInducer(64)
    seeds = torch.tensor([rnd.randrange(100) for _ in range(n_seeds)])
    uniq = ind.init_node(seeds)
    ref_map = {}
    for v in seeds.tolist():
        if v not in ref_map:
            ref_map[v] = len(ref_map)
    assert uniq.tolist() == list(ref_map.keys())
    srcs = uniq
    for _ in range(hops):
        counts = torch.tensor([rnd.randrange(4) for _ in srcs.tolist()],
                              dtype=torch.long)
        nbrs = torch.tensor([rnd.randrange(100)
                             for _ in range(int(counts.sum()))],
                            dtype=torch.long)
        nodes, rows, cols = ind.induce_next(srcs, nbrs, counts)
        fresh = []
        for v in nbrs.tolist():
            if v not in ref_map:
                ref_map[v] = len(ref_map)
                fresh.append(v)
        assert nodes.tolist() == fresh
        # relabeled edges match the reference map
        e = 0
        for i, s in enumerate(srcs.tolist()):
            for _ in range(int(counts[i])):
                assert rows[e] == ref_map[s]
                assert cols[e] == ref_map[int(nbrs[e])]
                e += 1
        srcs = nodes
        if srcs.numel() == 0:
            break


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 20), st.integers(2, 4), st.randoms())
def test_stitch_matches_python_reference(n_ids, parts, rnd):
    # assign each seed position to one partition
    owner = [rnd.randrange(parts) for _ in range(n_ids)]
    idx_list, nbrs_list, num_list = [], [], []
    expect = [None] * n_ids
    for p in range(parts):
        pos = [i for i in range(n_ids) if owner[i] == p]
        counts = [rnd.randrange(4) for _ in pos]
        flat = []
        for i, c in zip(pos, counts):
            vals = [rnd.randrange(1000) for _ in range(c)]
            expect[i] = vals
            flat.extend(vals)
        idx_list.append(torch.tensor(pos, dtype=torch.long))
        nbrs_list.append(torch.tensor(flat, dtype=torch.long))
        num_list.append(torch.tensor(counts, dtype=torch.long))
    nbrs, num, _ = _C.stitch_sample_results(n_ids, idx_list, nbrs_list,
                                            num_list)
    flat_ref = [v for vals in expect for v in (vals or [])]
    assert nbrs.tolist() == flat_ref
    assert num.tolist() == [len(v or []) for v in expect]


@settings(max_examples=30, deadline=None)
@given(st.integers(2, 50), st.integers(0, 200), st.integers(-1, 8),
       st.randoms())
def test_cpu_sampler_invariants(n, e, k, rnd):
    """For ANY random CSR + seeds + fan-out: counts = min(k, deg) (or deg
    when k=-1), every sampled neighbor is a true neighbor of its row, and
    uniform sampling is without replacement."""
    if k == 0:
        k = 1
    rows = torch.tensor([rnd.randrange(n) for _ in range(e)],
                        dtype=torch.long)
    cols = torch.tensor([rnd.randrange(n) for _ in range(e)],
                        dtype=torch.long)
    topo = Topology(torch.stack([rows, cols]) if e else
                    torch.empty(2, 0, dtype=torch.long), num_nodes=n)
    seeds = torch.tensor([rnd.randrange(n) for _ in range(5)],
                         dtype=torch.long)
    nbrs, num, _ = _C.sample_neighbors(topo.indptr, topo.indices, seeds, k)
    adj = {}
    for r, c in zip(rows.tolist(), cols.tolist()):
        adj.setdefault(r, []).append(c)
    off = 0
    for s, cnt in zip(seeds.tolist(), num.tolist()):
        deg = len(adj.get(s, []))
        expect = deg if k == -1 else min(k, deg)
        assert cnt == expect, (s, cnt, expect)
        picked = nbrs[off:off + cnt].tolist()
        off += cnt
        # without replacement over EDGES: each value appears at most as
        # often as its multiplicity in the (multi)graph's adjacency
        import collections
        have = collections.Counter(adj.get(s, []))
        for v, c in collections.Counter(picked).items():
            assert c <= have[v], (s, v, c, have[v])
    assert off == nbrs.numel()


@settings(max_examples=20, deadline=None)
@given(st.integers(3, 40), st.integers(1, 120), st.randoms())
def test_negative_sampler_strict_property(n, e, rnd):
    """Strict negatives never collide with an existing edge."""
    rows = torch.tensor([rnd.randrange(n) for _ in range(e)])
    cols = torch.tensor([rnd.randrange(n) for _ in range(e)])
    topo = Topology(torch.stack([rows, cols]), num_nodes=n)
    neg = _C.sample_negative(topo.indptr, topo.indices, n, 32, 8, False)
    present = set(zip(rows.tolist(), cols.tolist()))
    for r, c in neg.t().tolist():
        assert (r, c) not in present
