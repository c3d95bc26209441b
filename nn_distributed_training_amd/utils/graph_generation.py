"""Communication-graph construction and mixing weights.

Behavioral parity with the reference's ``utils/graph_generation.py``
(/root/reference/utils/graph_generation.py:69-168): same graph families
(wheel / cycle / complete / Erdős–Rényi with connectivity retry), Metropolis
mixing weights W_ij = 1/(1+max(deg_i,deg_j)), euclidean disk graphs,
Fiedler-targeted disk graphs (bisection on radius) and Delaunay graphs.

The implementation is vectorized (numpy adjacency) rather than a translation;
the adjacency produced here also drives the RCCL point-to-point schedule
(parallel/schedule.py).
"""

from __future__ import annotations

import random

import networkx as nx
import numpy as np
import scipy.spatial
import torch


def generate_from_conf(graph_conf: dict):
    """Build a graph from a config dict.

    Mirrors reference utils/graph_generation.py:69-104. Returns (N, graph).
    Supported ``type``: wheel | cycle | complete | random (Erdős–Rényi with
    ``p`` and ``gen_attempts`` retries until connected).
    """
    N = graph_conf["num_nodes"]
    gtype = graph_conf["type"]
    if gtype == "wheel":
        graph = nx.wheel_graph(N)
    elif gtype == "cycle":
        graph = nx.cycle_graph(N)
    elif gtype == "complete":
        graph = nx.complete_graph(N)
    elif gtype == "random":
        seed = graph_conf.get("seed")
        rng = random.Random(seed) if seed is not None else None
        graph = nx.erdos_renyi_graph(N, graph_conf["p"], seed=rng)
        for _ in range(graph_conf["gen_attempts"]):
            if nx.is_connected(graph):
                break
            graph = nx.erdos_renyi_graph(N, graph_conf["p"], seed=rng)
        if not nx.is_connected(graph):
            raise NameError(
                "A connected random graph could not be generated,"
                " increase p or gen_attempts."
            )
    else:
        raise NameError("Unknown communication graph type.")
    return N, graph


def adjacency(graph: nx.Graph) -> np.ndarray:
    """Dense boolean adjacency in node order 0..N-1."""
    N = graph.number_of_nodes()
    A = np.zeros((N, N), dtype=bool)
    for i, j in graph.edges():
        A[i, j] = True
        A[j, i] = True
    return A


def get_metropolis(graph: nx.Graph) -> torch.Tensor:
    """Metropolis-Hastings mixing matrix.

    W_ij = 1/(1+max(deg_i, deg_j)) for edges, W_ii = 1 - sum_j W_ij.
    Parity with reference utils/graph_generation.py:107-122, computed
    vectorized. Returns a torch tensor in the default dtype.
    """
    A = adjacency(graph)
    degs = A.sum(axis=1).astype(np.float64)
    pair_max = np.maximum.outer(degs, degs)
    W = np.where(A, 1.0 / (pair_max + 1.0), 0.0)
    np.fill_diagonal(W, 0.0)
    np.fill_diagonal(W, 1.0 - W.sum(axis=1))
    return torch.as_tensor(W).to(torch.get_default_dtype())


def euclidean_disk_graph(poses: np.ndarray, radius: float):
    """Disk communication graph from node positions [N, 2].

    Nodes within ``radius`` of each other are connected. Returns
    (graph, is_connected). Parity with reference
    utils/graph_generation.py:125-146.
    """
    d = scipy.spatial.distance.squareform(
        scipy.spatial.distance.pdist(np.asarray(poses, dtype=np.float64))
    )
    adj = d <= radius
    np.fill_diagonal(adj, False)
    graph = nx.from_numpy_array(adj)
    return graph, nx.is_connected(graph)


def _fiedler_of_disk(N, positions, radius):
    G = nx.random_geometric_graph(N, radius, pos=positions)
    return nx.linalg.algebraic_connectivity(G, tol=1e-3, method="lanczos")


def disk_with_fied(N: int, targ: float, num_restarts: int = 50) -> nx.Graph:
    """Random geometric graph whose algebraic connectivity (Fiedler value)
    hits ``targ`` within ±0.01, found by bisection on the disk radius.

    Parity with reference utils/graph_generation.py:14-66 (used by the
    scaling study to sweep node count at constant connectivity).
    """
    tol = 0.01
    for _ in range(num_restarts):
        pos = {i: (random.random(), random.random()) for i in range(N)}
        lbr, ubr = 0.05, 0.8
        lbf = _fiedler_of_disk(N, pos, lbr)
        ubf = _fiedler_of_disk(N, pos, ubr)
        if abs(lbf - targ) < tol:
            return nx.random_geometric_graph(N, lbr, pos=pos)
        if abs(ubf - targ) < tol:
            return nx.random_geometric_graph(N, ubr, pos=pos)
        if not ubf > lbf:
            # disconnected even at the outer radius (can happen for any
            # N on an unlucky position draw, and always for tiny N when
            # two points land far apart) — try a fresh draw rather than
            # aborting the whole sweep
            continue
        if targ > ubf or targ < lbf:
            # target not bracketed for this position draw; try a new draw
            continue
        for _ in range(100):
            midr = 0.5 * (ubr + lbr)
            midf = _fiedler_of_disk(N, pos, midr)
            if abs(midf - targ) < tol:
                return nx.random_geometric_graph(N, midr, pos=pos)
            if midf > targ:
                ubr = midr
            else:
                lbr = midr
    raise NameError("Never found a viable graph!")


def gen_delaunay(N: int) -> nx.Graph:
    """Delaunay-triangulation graph of N uniform points in the unit box.

    Parity with reference utils/graph_generation.py:149-168.
    """
    positions = np.random.rand(N, 2)
    tri = scipy.spatial.Delaunay(positions)
    edges = set()
    for s in tri.simplices:
        edges.update({(s[0], s[1]), (s[1], s[2]), (s[0], s[2])})
    return nx.Graph(sorted(edges))
