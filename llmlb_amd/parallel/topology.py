"""xGMI topology reading and TP-group placement for an 8x MI355X node.

MI355X GPUs connect point-to-point over xGMI (7 links x ~153 GB/s per
GPU — no switch).  A TP replica group's RCCL ring all-reduce is bound by
the slowest hop between consecutive ranks, so TP groups must be pinned
to mutually-adjacent GPUs (BASELINE config 4: 2x Llama-3-70B TP=4
replicas on adjacent quads).  The gateway owns *placement*; RCCL inside
the backend group does the collectives.

Topology sources (in order):
  1. amdsmi (amdsmi_topo_get_link_type between processor pairs),
  2. LLMLB_XGMI_ADJACENCY env (JSON adjacency list — tests/CI),
  3. fallback: fully-connected assumption (a single-hop mesh, which is
     the MI355X OAM baseboard default where every GPU pairs with every
     other).
"""
from __future__ import annotations

import itertools
import json
import os


def read_adjacency(n_gpus: int | None = None) -> list[list[int]]:
    """Returns adjacency[i] = list of GPUs with a direct xGMI link."""
    env = os.environ.get("LLMLB_XGMI_ADJACENCY")
    if env:
        adj = json.loads(env)
        return [list(map(int, row)) for row in adj]
    try:
        import amdsmi
        amdsmi.amdsmi_init()
        try:
            devs = amdsmi.amdsmi_get_processor_handles()
            n = len(devs)
            adj: list[list[int]] = [[] for _ in range(n)]
            for i, j in itertools.combinations(range(n), 2):
                try:
                    res = amdsmi.amdsmi_topo_get_link_type(devs[i], devs[j])
                    # res: dict with 'hops' and 'type' (XGMI == 2)
                    link_type = res.get("type") if isinstance(res, dict) \
                        else None
                    is_xgmi = (str(link_type).lower().find("xgmi") >= 0
                               or link_type == 2)
                    if is_xgmi:
                        adj[i].append(j)
                        adj[j].append(i)
                except Exception:
                    continue
            if any(adj):
                return adj
        finally:
            amdsmi.amdsmi_shut_down()
    except Exception:
        pass
    n = n_gpus or 8
    return [[j for j in range(n) if j != i] for i in range(n)]


def is_clique(adj: list[list[int]], group: list[int]) -> bool:
    sets = {i: set(adj[i]) for i in group}
    return all(b in sets[a] for a, b in itertools.combinations(group, 2))


def ring_cost(adj: list[list[int]], group: list[int]) -> int:
    """Number of non-adjacent consecutive pairs on the RCCL ring order.
    0 = every ring hop rides a direct xGMI link."""
    cost = 0
    n = len(group)
    for idx in range(n):
        a, b = group[idx], group[(idx + 1) % n]
        if b not in adj[a]:
            cost += 1
    return cost


def place_tp_groups(n_gpus: int, tp: int,
                    adj: list[list[int]] | None = None,
                    n_groups: int | None = None) -> list[list[int]]:
    """Pick disjoint TP groups of size ``tp`` whose ring hops all ride
    direct xGMI links (prefer cliques, then minimal ring cost).

    Returns a list of GPU-index groups, e.g. [[0,1,2,3],[4,5,6,7]].
    """
    if adj is None:
        adj = read_adjacency(n_gpus)
    want = n_groups if n_groups is not None else n_gpus // tp
    chosen: list[list[int]] = []
    used: set[int] = set()
    # greedy: among unused GPUs prefer groups that are cliques
    candidates = [list(c) for c in
                  itertools.combinations(range(n_gpus), tp)]
    # cliques first, then lowest ring cost
    candidates.sort(key=lambda g: (0 if is_clique(adj, g) else 1,
                                   ring_cost(adj, g), g))
    for group in candidates:
        if len(chosen) >= want:
            break
        if used.intersection(group):
            continue
        chosen.append(group)
        used.update(group)
    if len(chosen) < want:
        raise ValueError(
            f"cannot place {want} disjoint TP={tp} groups on {n_gpus} "
            f"GPUs with the given adjacency")
    return chosen


def rccl_env_for_group(group: list[int]) -> dict:
    """Environment for a backend process group pinned to ``group``.

    HIP_VISIBLE_DEVICES pins the quad; keeping collectives on-mesh
    follows from the placement (xGMI is point-to-point; RCCL rings the
    visible devices).  HSA_ENABLE_IPC_MODE_LEGACY=0 is required on this
    pool (dmabuf IPC)."""
    return {
        "HIP_VISIBLE_DEVICES": ",".join(map(str, group)),
        "HSA_ENABLE_IPC_MODE_LEGACY":
            os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "0"),
    }
