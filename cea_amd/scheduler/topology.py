"""Topology model + placement algorithm for the gang scheduler.

Role parity: the placement core of
/root/reference/gke-topology-scheduler/schedule-daemon.py —
topology keys from node labels (:46-48, :175-195), pairwise distance
(1e6/1e4/1e2 per level mismatch, :153-172) and the pod->node assignment
search (:500-544).

Redesign: the reference enumerates ALL sorted node combinations
(itertools.combinations at :500-544) which is exponential; here nodes are
sorted by (block, subblock, host) and assignment picks the best contiguous
window by total pairwise distance — same optimum for hierarchical distances
(an optimal set under a tree metric is always contiguous in DFS order of the
tree), O(N·G) instead of O(C(N,G)).
"""
from __future__ import annotations

import dataclasses
from typing import Dict, List, Optional, Tuple

# node labels carrying physical topology (parity schedule-daemon.py:46-48;
# the cloud.google.com/gce-topology-* spellings are also accepted so the
# scheduler is drop-in on GKE-labeled nodes)
LABEL_BLOCK = "topology.cea-amd.io/block"
LABEL_SUBBLOCK = "topology.cea-amd.io/subblock"
LABEL_HOST = "topology.cea-amd.io/host"
COMPAT_LABELS = {
    "block": [LABEL_BLOCK, "cloud.google.com/gce-topology-block",
              "topology.gke.io/cluster"],
    "subblock": [LABEL_SUBBLOCK, "cloud.google.com/gce-topology-subblock",
                 "topology.gke.io/rack"],
    "host": [LABEL_HOST, "cloud.google.com/gce-topology-host",
             "topology.gke.io/host"],
}

# distance weights per level mismatch (parity schedule-daemon.py:153-172)
W_BLOCK = 1_000_000
W_SUBBLOCK = 10_000
W_HOST = 100


@dataclasses.dataclass(frozen=True)
class TopoKey:
    block: str = ""
    subblock: str = ""
    host: str = ""

    def sort_key(self) -> Tuple[str, str, str]:
        return (self.block, self.subblock, self.host)


def topo_key_from_labels(labels: Dict[str, str]) -> TopoKey:
    vals = {}
    for level, names in COMPAT_LABELS.items():
        vals[level] = next((labels[n] for n in names if n in labels), "")
    return TopoKey(vals["block"], vals["subblock"], vals["host"])


def distance(a: TopoKey, b: TopoKey) -> int:
    """Pairwise topology distance (parity: the 1e6/1e4/1e2 weighting)."""
    d = 0
    if a.block != b.block:
        d += W_BLOCK
    if a.subblock != b.subblock:
        d += W_SUBBLOCK
    if a.host != b.host:
        d += W_HOST
    return d


@dataclasses.dataclass
class CandidateNode:
    name: str
    topo: TopoKey
    # how many pods of THIS job the node can hold given its free resources
    capacity: int


def assign_pods(num_pods: int, nodes: List[CandidateNode]
                ) -> Optional[List[str]]:
    """Assign `num_pods` gang members to nodes minimizing summed pairwise
    topology distance.  Returns one node name per pod (nodes repeat up to
    their capacity), or None if the gang does not fit.

    Nodes are sorted by topology key; every contiguous window with enough
    capacity is scored by the summed pairwise distance of the slots it
    provides; best window wins.
    """
    nodes = [n for n in nodes if n.capacity > 0]
    nodes.sort(key=lambda n: (n.topo.sort_key(), n.name))
    total_cap = sum(n.capacity for n in nodes)
    if total_cap < num_pods or num_pods <= 0:
        return None

    best: Optional[Tuple[int, List[str]]] = None
    for start in range(len(nodes)):
        slots: List[str] = []
        topos: List[TopoKey] = []
        for n in nodes[start:]:
            take = min(n.capacity, num_pods - len(slots))
            slots.extend([n.name] * take)
            topos.extend([n.topo] * take)
            if len(slots) == num_pods:
                break
        if len(slots) < num_pods:
            break  # windows further right only get smaller
        score = _pairwise_score(topos)
        if best is None or score < best[0]:
            best = (score, slots)
        if best[0] == 0:
            break
    return best[1] if best else None


def _pairwise_score(topos: List[TopoKey]) -> int:
    """Summed pairwise distance in O(G) instead of O(G^2): distance terms
    are label-equality indicators, so sum over pairs = weight x (total
    pairs - same-label pairs), with same-label pairs counted per group."""
    g = len(topos)
    total_pairs = g * (g - 1) // 2

    def same_pairs(keyfn) -> int:
        counts: dict = {}
        for t in topos:
            k = keyfn(t)
            counts[k] = counts.get(k, 0) + 1
        return sum(c * (c - 1) // 2 for c in counts.values())

    # Labels compared exactly as distance() does (bare label equality).
    score = W_BLOCK * (total_pairs - same_pairs(lambda t: t.block))
    score += W_SUBBLOCK * (total_pairs - same_pairs(lambda t: t.subblock))
    score += W_HOST * (total_pairs - same_pairs(lambda t: t.host))
    return score
