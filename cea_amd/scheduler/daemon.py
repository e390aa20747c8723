"""Gang scheduler daemon: scheduling-gate driven, topology-aware.

Role parity: /root/reference/gke-topology-scheduler/schedule-daemon.py
(814 LoC).  Protocol kept identical so workloads are drop-in:
  * pods opt in with a scheduling gate named `gke.io/topology-aware-auto-*`
    (schedule-daemon.py:760) — the daemon finds Pending gated pods, groups
    them into jobs, and for each complete gang picks nodes minimizing
    topology distance, then binds by REMOVING the gate and pinning the pod
    with required nodeAffinity on kubernetes.io/hostname
    (schedule-daemon.py:447-497);
  * job grouping by job-name / jobset labels with owner-UID fallback
    (:594-647); pods sorted by completion index (:119-150);
  * node feasibility = Ready + taint-toleration compatible + free
    cpu/memory/amd.com/gpu computed from running pods (:245-332).
Resource key is amd.com/gpu (the reference hardcodes nvidia.com/gpu at
:221,311,389).
"""
from __future__ import annotations

import logging
import re
import time
from collections import defaultdict
from typing import Dict, List, Optional

from .topology import (
    CandidateNode,
    HeteroNode,
    assign_pods,
    assign_pods_hetero,
    selector_matches,
    topo_key_from_labels,
)

log = logging.getLogger(__name__)

GATE_PREFIX = "gke.io/topology-aware-auto-"   # parity schedule-daemon.py:760
COMPLETION_INDEX_LABEL = "batch.kubernetes.io/job-completion-index"
# kubeflow training-operator pods carry their gang index here instead
# (parity schedule-daemon.py:59,373-377)
KUBEFLOW_REPLICA_INDEX_LABEL = "training.kubeflow.org/replica-index"
JOB_NAME_LABELS = (
    "job-name",
    "jobset.sigs.k8s.io/jobset-name",
    "batch.kubernetes.io/job-name",
)

_QUANTITY_RE = re.compile(r"^(\d+(?:\.\d+)?)([a-zA-Z]*)$")
_SUFFIX = {
    "": 1, "k": 10**3, "M": 10**6, "G": 10**9, "T": 10**12, "P": 10**15,
    "Ki": 2**10, "Mi": 2**20, "Gi": 2**30, "Ti": 2**40, "Pi": 2**50,
}


def parse_quantity(q) -> float:
    """k8s resource quantity -> float (cpu in cores, memory in bytes).
    Total: malformed input parses as 0.0 — a bad quantity in one pod
    spec must never abort a whole scheduling pass (fuzz-pinned)."""
    if isinstance(q, (int, float)):
        return float(q)
    q = str(q).strip()
    if q.endswith("m"):
        try:
            return float(q[:-1]) / 1000.0
        except ValueError:
            return 0.0
    m = _QUANTITY_RE.match(q)
    if not m:
        return 0.0
    return float(m.group(1)) * _SUFFIX.get(m.group(2), 1)


def pod_requests(pod: dict) -> Dict[str, float]:
    """Summed container requests (parity :356-417)."""
    out: Dict[str, float] = defaultdict(float)
    for c in pod.get("spec", {}).get("containers", []):
        reqs = c.get("resources", {}).get("requests", {}) or {}
        for key, val in reqs.items():
            out[key] += parse_quantity(val)
    return dict(out)


def has_topology_gate(pod: dict) -> Optional[str]:
    for gate in pod.get("spec", {}).get("schedulingGates", []) or []:
        name = gate.get("name", "")
        if name.startswith(GATE_PREFIX):
            return name
    return None


def job_key(pod: dict) -> str:
    """Group key (parity :594-647): job labels, then owner UID, then the
    pod's own name."""
    meta = pod.get("metadata", {})
    labels = meta.get("labels", {}) or {}
    ns = meta.get("namespace", "default")
    for lab in JOB_NAME_LABELS:
        if lab in labels:
            return f"{ns}/{labels[lab]}"
    for owner in meta.get("ownerReferences", []) or []:
        if owner.get("uid"):
            return f"{ns}/{owner['uid']}"
    return f"{ns}/{meta.get('name', '')}"


def completion_index(pod: dict):
    """Sorting key, parity :119-150 — job completion-index label, then the
    kubeflow replica-index label, else (name-prefix, trailing ordinal) so
    "xxx-pod2" sorts before "xxx-pod10" (pod_sorting_key semantics).
    Always returns a (prefix, int) tuple so keys compare consistently."""
    labels = pod.get("metadata", {}).get("labels", {}) or {}
    for lab in (COMPLETION_INDEX_LABEL, KUBEFLOW_REPLICA_INDEX_LABEL):
        if lab in labels:
            try:
                return ("", int(labels[lab]))
            except ValueError:
                pass
    name = pod.get("metadata", {}).get("name", "")
    m = re.fullmatch(r"(.*?)(\d+)", name)
    if m:
        return (m.group(1), int(m.group(2)))
    return (name, 0)


def node_is_ready(node: dict) -> bool:
    for c in node.get("status", {}).get("conditions", []) or []:
        if c.get("type") == "Ready":
            return c.get("status") == "True"
    return False


def tolerates(pod: dict, node: dict) -> bool:
    """NoSchedule/NoExecute taint compatibility (parity :245-303)."""
    taints = node.get("spec", {}).get("taints", []) or []
    tolerations = pod.get("spec", {}).get("tolerations", []) or []
    for taint in taints:
        if taint.get("effect") not in ("NoSchedule", "NoExecute"):
            continue
        ok = False
        for tol in tolerations:
            op = tol.get("operator", "Equal")
            if tol.get("key") in (None, "", taint.get("key")) and op == "Exists":
                ok = True
            elif (tol.get("key") == taint.get("key")
                  and op == "Equal"
                  and tol.get("value", "") == taint.get("value", "")):
                ok = True
            if ok and tol.get("effect") in (None, "", taint.get("effect")):
                break
            ok = False
        if not ok:
            return False
    return True


def node_free_resources(node: dict, pods_on_node: List[dict]) -> Dict[str, float]:
    """allocatable - sum(requests of non-terminal pods) (parity :305-332)."""
    free: Dict[str, float] = {
        k: parse_quantity(v)
        for k, v in (node.get("status", {}).get("allocatable", {}) or {}).items()
    }
    for pod in pods_on_node:
        phase = pod.get("status", {}).get("phase", "")
        if phase in ("Succeeded", "Failed"):
            continue
        for k, v in pod_requests(pod).items():
            if k in free:
                free[k] -= v
    return free


def pods_fit_count(free: Dict[str, float], req: Dict[str, float]) -> int:
    """How many pods with request `req` fit into `free`."""
    count = float("inf")
    for k, v in req.items():
        if v <= 0:
            continue
        count = min(count, free.get(k, 0.0) // v)
    return int(count) if count != float("inf") else 2**30


def bind_pod(kube, pod: dict, node_name: str) -> None:
    """Drop the topology gate + inject required nodeAffinity on
    kubernetes.io/hostname (parity :447-497)."""
    ns = pod["metadata"].get("namespace", "default")
    name = pod["metadata"]["name"]
    spec = pod.setdefault("spec", {})
    spec["schedulingGates"] = [
        g for g in spec.get("schedulingGates", []) or []
        if not g.get("name", "").startswith(GATE_PREFIX)
    ]
    affinity = spec.setdefault("affinity", {})
    node_affinity = affinity.setdefault("nodeAffinity", {})
    required = node_affinity.setdefault(
        "requiredDuringSchedulingIgnoredDuringExecution", {"nodeSelectorTerms": []}
    )
    required["nodeSelectorTerms"] = [{
        "matchExpressions": [{
            "key": "kubernetes.io/hostname",
            "operator": "In",
            "values": [node_name],
        }]
    }]
    kube.replace_pod(ns, name, pod)
    log.info("bound %s/%s -> %s", ns, name, node_name)


class TopologyScheduler:
    def __init__(self, kube, interval_s: float = 5.0,
                 gate_cooloff_s: float = 60.0,
                 pod_informer=None, node_informer=None):
        """pod_informer/node_informer: optional started Informer caches
        (cea_amd/kube/informer.py).  When set, passes read memory instead
        of full-listing the cluster each loop — beyond the reference,
        which lists per pass (schedule-daemon.py:783)."""
        self.kube = kube
        self.interval_s = interval_s
        self.gate_cooloff_s = gate_cooloff_s   # parity :777-807
        self.pod_informer = pod_informer
        self.node_informer = node_informer
        self._last_attempt: Dict[str, float] = {}
        self._stop = False

    def _list_pods_and_pending(self):
        if self.pod_informer is not None:
            all_pods = self.pod_informer.items()
            pending = [p for p in all_pods
                       if p.get("status", {}).get("phase") == "Pending"]
            return all_pods, pending
        pending = self.kube.list_pods(field_selector="status.phase=Pending")
        return self.kube.list_pods(), pending

    def _list_nodes(self):
        if self.node_informer is not None:
            return self.node_informer.items()
        return self.kube.list_nodes()

    def schedule_once(self) -> int:
        """One pass; returns the number of pods bound."""
        all_pods, pods = self._list_pods_and_pending()
        gated = [p for p in pods if has_topology_gate(p)]
        if not gated:
            return 0
        jobs: Dict[str, List[dict]] = defaultdict(list)
        for p in gated:
            jobs[job_key(p)].append(p)

        nodes = [n for n in self._list_nodes() if node_is_ready(n)]
        pods_by_node: Dict[str, List[dict]] = defaultdict(list)
        for p in all_pods:
            node_name = p.get("spec", {}).get("nodeName")
            if node_name:
                pods_by_node[node_name].append(p)

        bound = 0
        now = time.monotonic()
        for key, job_pods in jobs.items():
            last = self._last_attempt.get(key, 0)
            if now - last < self.gate_cooloff_s and last > 0:
                continue
            self._last_attempt[key] = now
            job_pods.sort(key=completion_index)
            assignment = self._assign_job(job_pods, nodes, pods_by_node, key)
            if assignment is None:
                continue
            for pod, node_name in zip(job_pods, assignment):
                try:
                    bind_pod(self.kube, pod, node_name)
                    bound += 1
                    pods_by_node[node_name].append(pod)
                except Exception as e:  # noqa: BLE001
                    log.error("bind failed for %s: %s",
                              pod["metadata"].get("name"), e)
        return bound

    def _assign_job(self, job_pods: List[dict], nodes: List[dict],
                    pods_by_node: Dict[str, List[dict]], key: str
                    ) -> Optional[List[str]]:
        """Per-pod resource extraction (parity :356-417) with the fast
        provably-optimal window path for homogeneous gangs and the greedy
        hetero path otherwise.  Per-pod nodeSelector (parity :421-436) and
        taint toleration are checked in both paths."""
        reqs = [pod_requests(p) for p in job_pods]
        selectors = [
            p.get("spec", {}).get("nodeSelector") or {} for p in job_pods
        ]
        homogeneous = (
            all(r == reqs[0] for r in reqs[1:])
            and all(s == selectors[0] for s in selectors[1:])
            and all(
                (p.get("spec", {}).get("tolerations") or [])
                == (job_pods[0].get("spec", {}).get("tolerations") or [])
                for p in job_pods[1:]
            )
        )
        free_by_node = {
            n["metadata"]["name"]: node_free_resources(
                n, pods_by_node[n["metadata"]["name"]])
            for n in nodes
        }
        if homogeneous:
            candidates = []
            for n in nodes:
                labels = n["metadata"].get("labels", {}) or {}
                if not tolerates(job_pods[0], n):
                    continue
                if not selector_matches(selectors[0], labels):
                    continue
                cap = pods_fit_count(free_by_node[n["metadata"]["name"]],
                                     reqs[0])
                if cap > 0:
                    candidates.append(CandidateNode(
                        name=n["metadata"]["name"],
                        topo=topo_key_from_labels(labels),
                        capacity=cap,
                    ))
            assignment = assign_pods(len(job_pods), candidates)
            if assignment is None:
                log.info("job %s: %d pods do not fit on %d candidate nodes; "
                         "waiting", key, len(job_pods), len(candidates))
            return assignment
        # heterogeneous gang: per-pod requests/selectors differ
        hetero_pods = []
        for pod, req, sel in zip(job_pods, reqs, selectors):
            hetero_pods.append({"requests": req, "node_selector": sel,
                                "_pod": pod})
        hetero_nodes = []
        for n in nodes:
            # toleration must hold for every pod that could land here; check
            # per-pod by folding it into the selector path is not possible,
            # so conservatively require all gang pods tolerate the node
            if not all(tolerates(p, n) for p in job_pods):
                continue
            hetero_nodes.append(HeteroNode(
                name=n["metadata"]["name"],
                topo=topo_key_from_labels(
                    n["metadata"].get("labels", {}) or {}),
                free=free_by_node[n["metadata"]["name"]],
                labels=n["metadata"].get("labels", {}) or {},
            ))
        assignment = assign_pods_hetero(hetero_pods, hetero_nodes)
        if assignment is None:
            log.info("job %s (heterogeneous): %d pods do not fit on %d "
                     "candidate nodes; waiting", key, len(job_pods),
                     len(hetero_nodes))
        return assignment

    def run_forever(self, warmup_s: float = 90.0, settle_s: float = 5.0
                    ) -> None:
        """Parity with the reference loop cadence (:777-807): a 90 s warmup
        before the first pass (so pods scheduled by a previous daemon
        incarnation become visible on nodes and count against free
        resources), then a settle delay once gated pods are seen (so all
        members of a gang are visible before placement)."""
        log.info("topology scheduler loop starting (interval %.0fs, "
                 "warmup %.0fs)", self.interval_s, warmup_s)
        if warmup_s > 0:
            time.sleep(warmup_s)
        while not self._stop:
            try:
                if settle_s > 0 and self._has_gated_pending():
                    # gang members may still be appearing; let them settle
                    time.sleep(settle_s)
                self.schedule_once()
            except Exception as e:  # noqa: BLE001
                log.error("scheduling pass failed: %s", e)
            time.sleep(self.interval_s)

    def _has_gated_pending(self) -> bool:
        try:
            pods = self.kube.list_pods(field_selector="status.phase=Pending")
        except Exception:  # noqa: BLE001
            return False
        return any(has_topology_gate(p) for p in pods)
