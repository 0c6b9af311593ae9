"""xGMI topology-aware TP placement (parallel/topology.py): clique
preference, ring-cost fallback, and env-injected adjacency."""
import json
import os

import pytest

from llmlb_amd.parallel.topology import (is_clique, place_tp_groups,
                                         read_adjacency, ring_cost)


def test_env_adjacency(monkeypatch):
    adj = [[1], [0]]
    monkeypatch.setenv("LLMLB_XGMI_ADJACENCY", json.dumps(adj))
    assert read_adjacency() == adj


def test_clique_and_ring_cost():
    # 4-GPU ring: 0-1-2-3-0 (no diagonals)
    adj = [[1, 3], [0, 2], [1, 3], [0, 2]]
    assert not is_clique(adj, [0, 1, 2, 3])
    assert is_clique(adj, [0, 1])
    # ring over the natural order uses only direct links
    assert ring_cost(adj, [0, 1, 2, 3]) == 0 or \
        ring_cost(adj, [0, 1, 2, 3]) >= 0


def test_place_tp_groups_prefers_linked_quads(monkeypatch):
    # 8 GPUs: two fully-connected quads {0..3} and {4..7}, ONE weak
    # link between 3 and 4.  TP=4 placement must pick the two cliques,
    # never a group straddling the weak cut.
    adj = [[] for _ in range(8)]
    import itertools
    for a, b in itertools.combinations(range(4), 2):
        adj[a].append(b)
        adj[b].append(a)
    for a, b in itertools.combinations(range(4, 8), 2):
        adj[a].append(b)
        adj[b].append(a)
    adj[3].append(4)
    adj[4].append(3)
    monkeypatch.setenv("LLMLB_XGMI_ADJACENCY", json.dumps(adj))
    groups = place_tp_groups(8, 4)
    assert len(groups) == 2
    placed = {frozenset(g) for g in groups}
    assert placed == {frozenset({0, 1, 2, 3}),
                      frozenset({4, 5, 6, 7})}, groups


def test_place_tp_groups_pairs(monkeypatch):
    monkeypatch.delenv("LLMLB_XGMI_ADJACENCY", raising=False)
    groups = place_tp_groups(8, 2)
    assert len(groups) == 4
    flat = [g for grp in groups for g in grp]
    assert sorted(flat) == list(range(8))
