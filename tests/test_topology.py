"""Node-topology computation over a real c10d Store (CPU, multi-process)."""

import os

import pytest
import torch.distributed as dist

from dist_utils import run_dist


def _topo_worker(rank, ws, hostmap):
    import torch_cgx_amd
    from torch_cgx_amd import _C
    from torch.distributed import PrefixStore
    store = dist.distributed_c10d._get_default_store()
    pstore = PrefixStore(f"topotest", store)
    host = hostmap[rank]
    node_id, local_rank, local_size, n_nodes, uniform = _C.compute_topology(
        pstore, rank, ws, host)
    # expectations computed in python
    order = []
    node_of = []
    for r in range(ws):
        h = hostmap[r]
        if h not in order:
            order.append(h)
        node_of.append(order.index(h))
    exp_nodes = len(order)
    exp_node = node_of[rank]
    exp_local_rank = sum(1 for r in range(rank) if node_of[r] == exp_node)
    exp_local_size = node_of.count(exp_node)
    counts = [node_of.count(i) for i in range(exp_nodes)]
    assert node_id == exp_node
    assert local_rank == exp_local_rank
    assert local_size == exp_local_size
    assert n_nodes == exp_nodes
    assert uniform == all(c == counts[0] for c in counts)


def test_topology_single_node():
    run_dist(_topo_worker, 2, args=({0: "nodeA", 1: "nodeA"},))


def test_topology_two_nodes():
    run_dist(_topo_worker, 4,
             args=({0: "nodeA", 1: "nodeA", 2: "nodeB", 3: "nodeB"},))


def test_topology_nonuniform():
    run_dist(_topo_worker, 3,
             args=({0: "nodeA", 1: "nodeB", 2: "nodeB"},))
