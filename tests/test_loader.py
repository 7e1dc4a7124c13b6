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
    loader = NeighborLoader(ds, [2], input_nodes=torch.arange(10),
                            batch_size=10, as_pyg_v1=True, with_edge=True)
    bs, node, adjs = next(iter(loader))
    assert bs == 10
    assert len(adjs) == 1


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
