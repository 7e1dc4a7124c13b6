"""Utility-layer unit tests (ckpt, channels, hetero merge, misc)."""
import os

import pytest
import torch

import glt_amd
from glt_amd.utils import (id2idx, load_ckpt, merge_hetero_sampler_output,
                           parse_size, save_ckpt, tensor_equal_with_device)


def test_parse_size():
    assert parse_size("256MB") == 256 << 20
    assert parse_size("1GB") == 1 << 30
    assert parse_size("2k") == 2048
    assert parse_size(4096) == 4096
    assert parse_size(None) is None


def test_id2idx():
    ids = torch.tensor([7, 3, 9])
    m = id2idx(ids)
    assert m[7] == 0 and m[3] == 1 and m[9] == 2


def test_ckpt_roundtrip(tmp_path):
    model = torch.nn.Linear(4, 4)
    opt = torch.optim.Adam(model.parameters())
    model(torch.randn(2, 4)).sum().backward()
    opt.step()
    for seq in range(7):
        save_ckpt(seq, str(tmp_path), model, opt, epoch=seq, keep=3)
    # pruning keeps the last 3
    files = sorted(os.listdir(tmp_path))
    assert len(files) == 3
    model2 = torch.nn.Linear(4, 4)
    opt2 = torch.optim.Adam(model2.parameters())
    epoch = load_ckpt(str(tmp_path), model2, opt2)
    assert epoch == 6
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.equal(a, b)


def test_mp_channel():
    from glt_amd.channel import MpChannel

    ch = MpChannel(capacity=4)
    ch.send({"a": torch.arange(5)})
    msg = ch.recv(1000)
    assert torch.equal(msg["a"], torch.arange(5))


def test_merge_hetero_sampler_output():
    from glt_amd.sampler import HeteroSamplerOutput

    et = ("u", "e", "v")
    a = HeteroSamplerOutput(
        node={"u": torch.tensor([1, 2]), "v": torch.tensor([10])},
        row={et: torch.tensor([0, 1])}, col={et: torch.tensor([0, 0])})
    b = HeteroSamplerOutput(
        node={"u": torch.tensor([2, 3]), "v": torch.tensor([11])},
        row={et: torch.tensor([0])}, col={et: torch.tensor([0])})
    merged = merge_hetero_sampler_output(a, b)
    assert set(merged.node["u"].tolist()) == {1, 2, 3}
    assert set(merged.node["v"].tolist()) == {10, 11}
    # every merged edge maps back to an original (u, v) pair
    pairs = set(zip(merged.node["u"][merged.row[et]].tolist(),
                    merged.node["v"][merged.col[et]].tolist()))
    assert pairs == {(1, 10), (2, 10), (2, 11)}


def test_tensor_equal_with_device():
    a = torch.ones(3)
    assert tensor_equal_with_device(a, torch.ones(3))
    assert not tensor_equal_with_device(a, torch.zeros(3))
    assert not tensor_equal_with_device(a, torch.ones(4))


def test_random_node_split(ring_graph):
    from glt_amd import Dataset

    ds = Dataset()
    ds.init_graph(edge_index=ring_graph["edge_index"], graph_mode="CPU",
                  num_nodes=40)
    ds.random_node_split(num_val=0.1, num_test=0.2)
    assert ds.val_idx.numel() == 4
    assert ds.test_idx.numel() == 8
    assert ds.train_idx.numel() == 28
    all_ids = torch.cat([ds.train_idx, ds.val_idx, ds.test_idx])
    assert set(all_ids.tolist()) == set(range(40))


def test_index_select_nested():
    from glt_amd.utils.common import index_select

    assert index_select(None, torch.tensor([0])) is None
    d = {"a": torch.arange(5), "b": {"c": torch.arange(10, 15)}}
    out = index_select(d, torch.tensor([1, 3]))
    assert out["a"].tolist() == [1, 3]
    assert out["b"]["c"].tolist() == [11, 13]


def test_format_hetero_sampler_output():
    from glt_amd.sampler import HeteroSamplerOutput
    from glt_amd.utils import format_hetero_sampler_output

    et = ("u", "e", "w")  # 'w' never appears as a node dict key
    out = HeteroSamplerOutput(
        node={"u": torch.tensor([1])}, row={et: torch.tensor([0])},
        col={et: torch.tensor([0])})
    fixed = format_hetero_sampler_output(out)
    assert "w" in fixed.node and fixed.node["w"].numel() == 0


def test_assign_device():
    from glt_amd.utils.common import assign_device

    d = assign_device(0)
    assert d.type in ("cuda", "cpu")  # cpu fallback without a GPU
