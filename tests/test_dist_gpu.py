"""Distributed sampling on GPU: 2 RPC worker ranks sharing cuda:0 on a
1-GPU box (per-hop fan-out, GPU stitch, feature collection)."""
import multiprocessing as mp

import pytest
import torch

pytestmark = pytest.mark.gpu

VNUM = 40


def _worker(rank, world, port, fail_q):
    try:
        import torch

        import glt_amd
        from glt_amd.data import Feature
        from glt_amd.distributed import (CollocatedDistSamplingWorkerOptions,
                                         DistDataset, DistNeighborLoader,
                                         barrier, init_worker_group)
        from glt_amd.partition import GLTPartitionBook

        glt_amd.seed_everything(42 + rank)
        torch.cuda.set_device(0)
        init_worker_group(world, rank)
        rows, cols = [], []
        for v in range(rank, VNUM, 2):
            rows += [v, v]
            cols += [(v + 1) % VNUM, (v + 2) % VNUM]
        ds = DistDataset(num_partitions=2, partition_idx=rank)
        ds.init_graph(edge_index=torch.tensor([rows, cols]),
                      graph_mode="CUDA", num_nodes=VNUM, device=0)
        node_pb = GLTPartitionBook(torch.arange(VNUM) % 2)
        ds.node_pb = node_pb
        feats = torch.arange(VNUM, dtype=torch.float32).unsqueeze(1) \
            .repeat(1, 16)
        local_ids = torch.arange(rank, VNUM, 2)
        id2index = torch.full((VNUM,), -1, dtype=torch.long)
        id2index[local_ids] = torch.arange(local_ids.numel())
        ds.node_features = Feature(feats[local_ids], split_ratio=1.0,
                                   device=0, with_gpu=True,
                                   id2index=id2index.cuda())
        ds._node_feat_pb = node_pb
        ds.node_labels = torch.arange(VNUM).cuda()

        opts = CollocatedDistSamplingWorkerOptions(
            master_addr="127.0.0.1", master_port=port)
        seeds = torch.arange(rank, VNUM, 2)
        loader = DistNeighborLoader(ds, [2, 2], input_nodes=seeds,
                                    batch_size=5, shuffle=True,
                                    to_device=torch.device("cuda", 0),
                                    worker_options=opts)
        for data in loader:
            assert data.batch_size == 5
            node = data.node.cpu()
            ei = data.edge_index.cpu()
            diff = (node[ei[1]] - node[ei[0]]) % VNUM
            assert ((diff == 1) | (diff == 2)).all()
            assert data.x.is_cuda
            assert (data.x.cpu() == node.float().unsqueeze(1)).all()
        barrier()
        fail_q.put((rank, None))
    except Exception:  # noqa: BLE001
        import traceback

        fail_q.put((rank, traceback.format_exc()))


@pytest.mark.timeout(240)
def test_dist_sampling_two_ranks_one_gpu():
    from glt_amd.utils import get_free_port

    ctx = mp.get_context("spawn")
    port = get_free_port()
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, 2, port, q))
          for r in range(2)]
    for p in ps:
        p.start()
    results = [q.get(timeout=220) for _ in range(2)]
    for p in ps:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    for rank, err in results:
        assert err is None, f"rank {rank}:\n{err}"
