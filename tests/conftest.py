import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X); skipped on CPU")


def pytest_collection_modifyitems(config, items):
    if torch.cuda.is_available():
        return
    skip = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip)


@pytest.fixture
def ring_graph():
    """Deterministic 40-node ring, degree 2 (v -> v+1, v+2), feature row
    v = [v]*16, edge feature e = [e]*8.  Mirrors the closed-form fixture the
    reference uses (reference test/python/dist_test_utils.py:41-120) so
    correctness checks are exact.
    """
    n = 40
    rows, cols = [], []
    for v in range(n):
        rows += [v, v]
        cols += [(v + 1) % n, (v + 2) % n]
    edge_index = torch.tensor([rows, cols], dtype=torch.long)
    feats = torch.arange(n, dtype=torch.float32).unsqueeze(1).repeat(1, 16)
    efeats = torch.arange(2 * n, dtype=torch.float32).unsqueeze(1).repeat(1, 8)
    labels = torch.arange(n, dtype=torch.long)
    return {
        "num_nodes": n,
        "edge_index": edge_index,
        "feats": feats,
        "efeats": efeats,
        "labels": labels,
    }


def check_ring_edges(node, edge_index, n=40):
    """Every sampled edge must satisfy row == col+1 or col+2 (mod n)."""
    rows = node[edge_index[0]]
    cols = node[edge_index[1]]
    diff = (cols - rows) % n
    assert ((diff == 1) | (diff == 2)).all(), (rows, cols)
