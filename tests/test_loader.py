"""Loader end-to-end tests on the closed-form ring graph (CPU)."""
import pytest
import torch

import glt_amd
from glt_amd import Dataset, NeighborLoader, LinkNeighborLoader
from glt_amd.loader import SubGraphLoader
from glt_amd.sampler import NegativeSampling

from conftest import check_ring_edges


def make_dataset(ring_graph, with_efeat=True):
    ds = Dataset()
    ds.init_graph(edge_index=ring_graph["edge_index"], graph_mode="CPU",
                  num_nodes=ring_graph["num_nodes"])
    ds.init_node_features(ring_graph["feats"], with_gpu=False)
    if with_efeat:
        ds.init_edge_features(ring_graph["efeats"], with_gpu=False)
    ds.init_node_labels(ring_graph["labels"])
    return ds


def test_neighbor_loader_basic(ring_graph):
    glt_amd.seed_everything(11)
    ds = make_dataset(ring_graph)
    loader = NeighborLoader(ds, [2, 2], input_nodes=torch.arange(40),
                            batch_size=5, shuffle=True, with_edge=True)
    n_batches = 0
    for data in loader:
        n_batches += 1
        assert data.batch_size == 5
        # features/labels are closed-form
        assert (data.y == data.node[:data.batch_size]).all()
        assert (data.x == data.node.float().unsqueeze(1)).all()
        assert data.edge is not None
        assert (data.edge_attr == data.edge.float().unsqueeze(1)).all()
        check_ring_edges(data.node, data.edge_index)
        assert data.num_sampled_nodes[0] == 5
        assert len(data.num_sampled_nodes) == 3
        assert len(data.num_sampled_edges) == 2
        assert data.num_sampled_edges[0] == 10
    assert n_batches == 8


def test_neighbor_loader_pyg_v1(ring_graph):
    ds = make_dataset(ring_graph)
    loader = NeighborLoader(ds, [2, 2], input_nodes=torch.arange(10),
                            batch_size=10, as_pyg_v1=True, with_edge=True)
    bs, node, adjs = next(iter(loader))
    assert bs == 10
    assert len(adjs) == 2  # one per hop, deepest first
    # the LAST adj (hop 1) has target side = the seeds
    assert adjs[-1].size[1] == 10
    assert (adjs[-1].edge_index[0] < 10).all()
    # deepest adj covers more rows than the shallow one
    assert adjs[0].size[0] >= adjs[-1].size[0]
    assert adjs[0].e_id is not None


def test_link_neighbor_loader_binary(ring_graph):
    glt_amd.seed_everything(5)
    ds = make_dataset(ring_graph, with_efeat=False)
    eli = ring_graph["edge_index"]
    loader = LinkNeighborLoader(
        ds, [2], edge_label_index=eli,
        neg_sampling=NegativeSampling("binary"), batch_size=8)
    data = next(iter(loader))
    assert data.edge_label_index is not None
    assert data.edge_label is not None
    assert data.edge_label.numel() == data.edge_label_index.size(1)
    # positives come first: 8 positive edges
    pos = data.edge_label_index[:, :8]
    # note: to_data flips edge_label_index into message-flow direction
    src = data.node[pos[1]]
    dst = data.node[pos[0]]
    assert (((dst - src) % 40 == 1) | ((dst - src) % 40 == 2)).all()
    check_ring_edges(data.node, data.edge_index)


def test_link_neighbor_loader_triplet(ring_graph):
    ds = make_dataset(ring_graph, with_efeat=False)
    eli = ring_graph["edge_index"]
    loader = LinkNeighborLoader(
        ds, [2], edge_label_index=eli,
        neg_sampling=NegativeSampling("triplet", amount=2), batch_size=4)
    data = next(iter(loader))
    assert data.src_index.numel() == 4
    assert data.dst_pos_index.numel() == 4
    assert data.dst_neg_index.shape == (4, 2)
    src = data.node[data.src_index]
    dst = data.node[data.dst_pos_index]
    assert (((dst - src) % 40 == 1) | ((dst - src) % 40 == 2)).all()


def test_subgraph_loader(ring_graph):
    ds = make_dataset(ring_graph, with_efeat=False)
    loader = SubGraphLoader(ds, input_nodes=torch.arange(6), batch_size=6)
    data = next(iter(loader))
    # all edges among nodes 0..5
    got = set(zip(data.node[data.edge_index[0]].tolist(),
                  data.node[data.edge_index[1]].tolist()))
    expect = {(v, v + 1) for v in range(5)} | {(v, v + 2) for v in range(4)}
    assert got == expect


def test_graphsage_trains_cpu(ring_graph):
    glt_amd.seed_everything(0)
    from glt_amd.models import GraphSAGE

    ds = make_dataset(ring_graph, with_efeat=False)
    model = GraphSAGE(16, 32, 2, out_channels=40)
    opt = torch.optim.Adam(model.parameters(), lr=1e-2)
    loader = NeighborLoader(ds, [2, 2], input_nodes=torch.arange(40),
                            batch_size=10, shuffle=True)
    losses = []
    for epoch in range(30):
        total = 0.0
        for data in loader:
            opt.zero_grad()
            out = model(data.x, data.edge_index)[:data.batch_size]
            loss = torch.nn.functional.cross_entropy(
                out, data.y[:data.batch_size])
            loss.backward()
            opt.step()
            total += loss.item()
        losses.append(total)
    assert losses[-1] < losses[0] * 0.7, losses  # learning happens


def test_hetero_loader(ring_graph):
    # bipartite: user v -> item (v+1)%40 and (v+2)%40
    ei = ring_graph["edge_index"]
    ds = Dataset()
    ds.init_graph(edge_index={("user", "buys", "item"): ei},
                  graph_mode="CPU", num_nodes=40)
    ds.init_node_features({"user": ring_graph["feats"],
                           "item": ring_graph["feats"] * 2.0},
                          with_gpu=False)
    loader = NeighborLoader(ds, [2, 2],
                            input_nodes=("user", torch.arange(10)),
                            batch_size=5)
    data = next(iter(loader))
    et = ("user", "buys", "item")
    assert data["user"].batch_size == 5
    ei_s = data[et].edge_index
    assert ei_s is not None and ei_s.size(1) > 0
    # user u buys items u+1, u+2
    src = data["user"].node[ei_s[0]]
    dst = data["item"].node[ei_s[1]]
    assert (((dst - src) % 40 == 1) | ((dst - src) % 40 == 2)).all()
    assert (data["user"].x == data["user"].node.float().unsqueeze(1)).all()
    assert (data["item"].x ==
            data["item"].node.float().unsqueeze(1) * 2.0).all()


def test_trim_equivalence_gat_gcn(ring_graph):
    import glt_amd
    from glt_amd.models import GAT, GCN

    glt_amd.seed_everything(3)
    ds = make_dataset(ring_graph, with_efeat=False)
    loader = NeighborLoader(ds, [2, 2], input_nodes=torch.arange(20),
                            batch_size=10)
    data = next(iter(loader))
    # GAT is exactly trim-equivalent (dropped edges never target surviving
    # rows, so per-target softmax denominators are unchanged)
    model = GAT(16, 8, 2, out_channels=5, heads=2)
    model.eval()
    with torch.no_grad():
        a = model(data.x, data.edge_index)[:data.batch_size]
        b = model(data.x, data.edge_index, data.num_sampled_nodes,
                  data.num_sampled_edges)[:data.batch_size]
    assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()
    # GCN's symmetric normalization counts source-side degrees, which
    # change under trimming: only shape/finite checks here
    gcn = GCN(16, 8, 2, out_channels=5)
    gcn.eval()
    with torch.no_grad():
        out = gcn(data.x, data.edge_index, data.num_sampled_nodes,
                  data.num_sampled_edges)[:data.batch_size]
    assert out.shape == (data.batch_size, 5)
    assert torch.isfinite(out).all()


def test_edge_dir_in(ring_graph):
    """CSC sampling: walking in-edges of seeds."""
    import glt_amd

    ds = Dataset(edge_dir="in")
    ds.init_graph(edge_index=ring_graph["edge_index"], graph_mode="CPU",
                  num_nodes=40)
    ds.init_node_features(ring_graph["feats"], with_gpu=False)
    ds.init_node_labels(ring_graph["labels"])
    loader = NeighborLoader(ds, [2], input_nodes=torch.arange(10),
                            batch_size=5, edge_dir="in")
    data = next(iter(loader))
    # seed v's in-neighbors are v-1, v-2: sampled (row=seed, col=src)
    rows = data.node[data.edge_index[0]]
    cols = data.node[data.edge_index[1]]
    diff = (rows - cols) % 40
    assert ((diff == 1) | (diff == 2)).all()


def test_sort_by_in_degree(ring_graph):
    from glt_amd.data import Topology, sort_by_in_degree

    # star: node 0 has max in-degree
    ei = torch.tensor([[1, 2, 3, 1], [0, 0, 0, 2]])
    topo = Topology(ei, num_nodes=4)
    feats = torch.arange(4, dtype=torch.float32).unsqueeze(1)
    reordered, id2index = sort_by_in_degree(feats, 0.5, topo)
    assert id2index[0] == 0  # hottest first
    assert (reordered[id2index[torch.arange(4)]] == feats).all()


def test_feature_id2index_roundtrip():
    from glt_amd.data import Feature

    feats = torch.arange(10, dtype=torch.float32).unsqueeze(1)
    perm = torch.randperm(10)
    id2index = torch.empty(10, dtype=torch.long)
    id2index[perm] = torch.arange(10)
    f = Feature(feats[perm], with_gpu=False, id2index=id2index)
    ids = torch.tensor([3, 7, 1])
    assert (f[ids] == ids.float().unsqueeze(1)).all()
    assert (f.cpu_get(ids) == ids.float().unsqueeze(1)).all()


def test_hetero_edge_dir_in(ring_graph):
    """'in' sampling over a hetero graph: results keyed by the reversed
    edge type (reference neighbor_sampler.py:261-269 convention)."""
    ei = ring_graph["edge_index"]
    # graph stores (user, buys, item); with edge_dir='in' we walk item<-user
    ds = Dataset(edge_dir="in")
    ds.init_graph(edge_index={("user", "buys", "item"): ei},
                  graph_mode="CPU", num_nodes=40)
    ds.init_node_features({"user": ring_graph["feats"],
                           "item": ring_graph["feats"]}, with_gpu=False)
    loader = NeighborLoader(ds, [2], input_nodes=("item", torch.arange(10)),
                            batch_size=5, edge_dir="in")
    data = next(iter(loader))
    rev = ("item", "rev_buys", "user")
    assert rev in data.edge_types
    ei_s = data[rev].edge_index
    # item i's in-sources under 'buys' are users i-1, i-2
    items = data["item"].node[ei_s[0]]
    users = data["user"].node[ei_s[1]]
    diff = (items - users) % 40
    assert ((diff == 1) | (diff == 2)).all()


def test_rgnn_learns_cpu(ring_graph):
    """RGNN (rsage) trains on a closed-form hetero task."""
    import torch.nn.functional as F

    from glt_amd.models import RGNN

    glt_amd.seed_everything(1)
    ei = ring_graph["edge_index"]
    et = ("user", "buys", "item")
    ds = Dataset()
    ds.init_graph(edge_index={et: ei}, graph_mode="CPU", num_nodes=40)
    ds.init_node_features(
        {"user": torch.nn.functional.one_hot(
            torch.arange(40) % 4, 8).float(),
         "item": torch.randn(40, 8)}, with_gpu=False)
    ds.init_node_labels({"user": torch.arange(40) % 4})
    loader = NeighborLoader(ds, [2], input_nodes=("user", torch.arange(40)),
                            batch_size=10, shuffle=True)
    model = RGNN([et], 8, 16, 4, num_layers=1, model="rsage")
    opt = torch.optim.Adam(model.parameters(), lr=5e-2)
    first = last = None
    for epoch in range(20):
        total = 0.0
        for data in loader:
            opt.zero_grad()
            out = model(data.x_dict, data.edge_index_dict,
                        predict_type="user")
            bs = data["user"].batch_size
            loss = F.cross_entropy(out[:bs], data["user"].y[:bs])
            loss.backward()
            opt.step()
            total += float(loss)
        if first is None:
            first = total
        last = total
    assert last < first * 0.5, (first, last)


def test_sageconv_bipartite_matches_stacked():
    from glt_amd.models import SAGEConv
    from glt_amd.models.hetero import conv_bipartite

    torch.manual_seed(0)
    conv = SAGEConv(8, 6)
    x_tgt = torch.randn(10, 8)
    x_src = torch.randn(20, 8)
    tgt = torch.sort(torch.randint(0, 10, (30,))).values
    src = torch.randint(0, 20, (30,))
    ei = torch.stack([tgt, src])
    a = conv((x_tgt, x_src), ei)
    b = conv_bipartite(conv, x_tgt, x_src, ei)
    assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()


def test_hetero_dict_fanout(ring_graph):
    """Per-edge-type fan-out dicts (PyG-style num_neighbors={etype: [...]})."""
    ei = ring_graph["edge_index"]
    e1 = ("user", "buys", "item")
    e2 = ("user", "views", "item")
    ds = Dataset()
    ds.init_graph(edge_index={e1: ei, e2: ei.flip(1)}, graph_mode="CPU",
                  num_nodes=40)
    loader = NeighborLoader(ds, {e1: [2], e2: [0]},
                            input_nodes=("user", torch.arange(10)),
                            batch_size=5)
    data = next(iter(loader))
    assert data[e1].edge_index is not None
    assert data[e1].edge_index.size(1) > 0
    # e2 fanout 0: no edges sampled for that relation
    assert e2 not in data.edge_types or data[e2].edge_index is None \
        or data[e2].edge_index.size(1) == 0


def test_weighted_loader_e2e(ring_graph):
    glt_amd.seed_everything(9)
    ds = Dataset()
    w = torch.ones(80)
    # make the (v -> v+1) edges overwhelmingly likely
    ds.init_graph(edge_index=ring_graph["edge_index"],
                  edge_weights=torch.where(
                      torch.arange(80) % 2 == 0, 1000.0, 0.001),
                  graph_mode="CPU", num_nodes=40)
    ds.init_node_features(ring_graph["feats"], with_gpu=False)
    loader = NeighborLoader(ds, [1], input_nodes=torch.arange(40),
                            batch_size=40, with_weight=True)
    data = next(iter(loader))
    diff = (data.node[data.edge_index[1]] -
            data.node[data.edge_index[0]]) % 40
    # heavy edges dominate: nearly every draw is the +1 neighbor
    assert (diff == 1).float().mean() > 0.9


def test_hetero_batched_gat_matches_per_relation():
    """HeteroConv's batched per-type projection GEMM must match the
    per-relation bipartite computation exactly (same modules/weights)."""
    import torch
    from glt_amd.models import GATConv
    from glt_amd.models.hetero import HeteroConv

    torch.manual_seed(0)
    ets = [("u", "a", "v"), ("u", "b", "w"), ("v", "c", "u")]
    convs = {et: GATConv(8, 4, heads=2) for et in ets}
    hc = HeteroConv(convs)
    x = {"u": torch.randn(6, 8), "v": torch.randn(5, 8),
         "w": torch.randn(4, 8)}
    sizes = {"u": 6, "v": 5, "w": 4}
    ei = {}
    for et in ets:
        nt, ns = sizes[et[0]], sizes[et[2]]
        tgt = torch.sort(torch.randint(0, nt, (12,)))[0]
        src = torch.randint(0, ns, (12,))
        ei[et] = torch.stack([tgt, src])
    out = hc(x, ei)
    ref = {}
    for et in ets:
        h = convs[et]((x[et[0]], x[et[2]]), ei[et])
        ref.setdefault(et[0], []).append(h)
    ref = {t: torch.stack(v).sum(0) if len(v) > 1 else v[0]
           for t, v in ref.items()}
    assert set(out) == set(ref)
    for t in out:
        assert torch.allclose(out[t], ref[t], atol=1e-6), t
    # gradients flow through the batched path
    sum(o.sum() for o in out.values()).backward()
    for et in ets:
        assert convs[et].lin.weight.grad is not None
        assert convs[et].att_src.grad is not None


def test_hetero_conv_folded_self_projection():
    """self_lins riding in the batched GEMM must equal plain Linear."""
    import torch
    from glt_amd.models import GATConv
    from glt_amd.models.hetero import HeteroConv

    torch.manual_seed(1)
    ets = [("u", "a", "v"), ("v", "b", "u")]
    convs = {et: GATConv(8, 4, heads=2) for et in ets}
    hc = HeteroConv(convs)
    lins = torch.nn.ModuleDict(
        {"u": torch.nn.Linear(8, 8), "v": torch.nn.Linear(8, 8),
         "w": torch.nn.Linear(8, 8)})
    x = {"u": torch.randn(6, 8), "v": torch.randn(5, 8),
         "w": torch.randn(3, 8)}  # w: no relations, plain path
    ei = {et: torch.stack([torch.sort(torch.randint(0, 5, (9,)))[0],
                           torch.randint(0, 5, (9,))]) for et in ets}
    out, self_out = hc(x, ei, self_lins=lins)
    for t in ("u", "v", "w"):
        assert torch.allclose(self_out[t], lins[t](x[t]), atol=1e-6), t
    assert set(out) == {"u", "v"}
