import networkx as nx
import numpy as np
import pytest
import torch

from nn_distributed_training_amd.utils import graph_generation as gg


def test_generate_families():
    for gtype, N in [("wheel", 6), ("cycle", 8), ("complete", 5)]:
        n, g = gg.generate_from_conf({"num_nodes": N, "type": gtype})
        assert n == N and g.number_of_nodes() == N
        assert nx.is_connected(g)


def test_generate_random_connected():
    n, g = gg.generate_from_conf(
        {"num_nodes": 10, "type": "random", "p": 0.4, "gen_attempts": 100,
         "seed": 3}
    )
    assert nx.is_connected(g)


def test_metropolis_properties():
    _, g = gg.generate_from_conf({"num_nodes": 8, "type": "cycle"})
    W = gg.get_metropolis(g)
    # doubly stochastic, symmetric, zero off-graph entries
    assert torch.allclose(W.sum(dim=1), torch.ones(8))
    assert torch.allclose(W, W.T)
    for i in range(8):
        for j in range(8):
            if i != j and not g.has_edge(i, j):
                assert W[i, j] == 0.0
    # cycle graph: every node degree 2 -> off-diag weight 1/3
    assert torch.isclose(W[0, 1], torch.tensor(1.0 / 3.0))


def test_metropolis_matches_reference_formula():
    """W_ij = 1/(1+max(deg_i,deg_j)) — checked on a wheel graph where
    degrees differ (hub degree N-1)."""
    _, g = gg.generate_from_conf({"num_nodes": 6, "type": "wheel"})
    W = gg.get_metropolis(g)
    degs = dict(g.degree())
    for i, j in g.edges():
        expect = 1.0 / (1.0 + max(degs[i], degs[j]))
        assert abs(W[i, j].item() - expect) < 1e-12


def test_euclidean_disk_graph():
    poses = np.array([[0.0, 0.0], [1.0, 0.0], [10.0, 0.0]])
    g, conn = gg.euclidean_disk_graph(poses, 1.5)
    assert g.has_edge(0, 1) and not g.has_edge(0, 2)
    assert not conn
    g2, conn2 = gg.euclidean_disk_graph(poses, 20.0)
    assert conn2 and g2.number_of_edges() == 3


def test_delaunay():
    np.random.seed(0)
    g = gg.gen_delaunay(12)
    assert g.number_of_nodes() == 12
    assert nx.is_connected(g)


@pytest.mark.slow
def test_disk_with_fied_unreachable_target_raises_cleanly():
    # N=2 only ever has Fiedler 0 (disconnected) or 2 (one edge): a
    # target of 1.0 is unreachable. The bisection must retry draws and
    # fail with its terminal error, not die on a degenerate bracket.
    import random

    random.seed(0)
    with pytest.raises(NameError, match="Never found"):
        gg.disk_with_fied(2, 1.0, num_restarts=5)


def test_disk_with_fied():
    g = gg.disk_with_fied(12, 1.0)
    fied = nx.linalg.algebraic_connectivity(g, tol=1e-3, method="lanczos")
    assert abs(fied - 1.0) < 0.15
