"""GPU health checker: AMD-SMI events -> device health + Node conditions.

Role parity: /root/reference/pkg/gpu/nvidia/health_check/health_checker.go
(476 LoC), re-based on AMD-SMI:
  * NVML Xid events -> amdsmi event notifications (VM fault, thermal
    throttle, GPU pre/post reset, page fault) via the native shim's
    wait_events, PLUS a polling watchdog over uncorrectable-ECC counters
    and device reachability (AMD-SMI has no ECC *event*, so code 48/79
    parity comes from polling deltas).
  * two code sets, same policy split as the reference:
      - health_critical (ConfigMap EVENT_CONFIG, default {48}): marks
        devices Unhealthy on the kubelet channel (health_checker.go:97)
      - monitor_critical (hardcoded): only sets the Node condition
        (health_checker.go:91)
  * Node condition `GPUCriticalError` whose Reason carries the JSON set of
    seen codes and whose Message carries the BootID, enabling the
    auto-repair detection flow (health_checker.go:330-337); a 1-minute
    heartbeat and a reset-with-backoff on BootID change are kept
    behavior-for-behavior (health_checker.go:101-160, 348-358).
  * node name from the downward-API NODE_NAME env (the reference reads GCE
    metadata at health_checker.go:164; it already has the NODE_NAME
    precedent at nvidia_gpu.go:206 — cloud-agnostic here).
"""
from __future__ import annotations

import json
import logging
import os
import threading
import time
from typing import Dict, Optional, Set

from .. import amdsmi
from ..amdsmi.iface import (
    EVT_ECC_UNCORRECTABLE,
    EVT_GPU_POST_RESET,
    EVT_LOST,
    EVT_XGMI_ERROR,
    MONITOR_CRITICAL_EVENTS,
    Event,
)
from ..kube import protos as api

log = logging.getLogger(__name__)

CONDITION_TYPE = "GPUCriticalError"   # analog of XidCriticalError
EVENT_REASON = "GPUCriticalError"
BOOT_ID_PATH = "/proc/sys/kernel/random/boot_id"

EVENT_WAIT_MS = 5000            # parity: 5 s WaitForEvent (health_checker.go:461)
ECC_POLL_INTERVAL_S = 10.0
HEARTBEAT_INTERVAL_S = 60.0     # parity: setXIDheartbeat (health_checker.go:348)
RESET_BACKOFF_S = (5, 10, 20, 40, 80)


def read_boot_id(path: str = BOOT_ID_PATH) -> str:
    try:
        with open(path) as f:
            return f.read().strip()
    except OSError:
        return ""


class GPUHealthChecker:
    def __init__(
        self,
        manager,
        kube_client,
        node_name: Optional[str] = None,
        health_critical_events: Optional[Set[int]] = None,
        boot_id_path: str = BOOT_ID_PATH,
        ecc_poll_interval_s: float = ECC_POLL_INTERVAL_S,
        heartbeat_interval_s: float = HEARTBEAT_INTERVAL_S,
        recover_on_reset: bool = True,
    ):
        self.manager = manager
        self.kube = kube_client
        self.node_name = node_name or os.environ.get("NODE_NAME", "")
        self.health_critical = set(
            health_critical_events
            if health_critical_events is not None
            else manager.config.health_critical_events
        )
        self.monitor_critical = set(MONITOR_CRITICAL_EVENTS) | self.health_critical
        self.boot_id_path = boot_id_path
        self.ecc_poll_interval_s = ecc_poll_interval_s
        self.heartbeat_interval_s = heartbeat_interval_s
        self.recover_on_reset = recover_on_reset
        self._stop = threading.Event()
        self._threads = []
        self._ecc_baseline: Dict[int, int] = {}
        self._xgmi_state: Dict[int, int] = {}
        self._condition_lock = threading.Lock()
        # Checker-local record of devices this checker marked Unhealthy and
        # the fault codes that did it.  Recovery keys off THIS, not
        # manager.device_health: the manager map is only populated when a
        # ListAndWatch stream drains the health queue, so relying on it made
        # recovery ordering-dependent when kubelet was disconnected
        # (ADVICE r01 low).
        self._unhealthy: Dict[str, Set[int]] = {}

    # -- lifecycle -----------------------------------------------------------
    def start(self) -> None:
        """Parity: Start (health_checker.go:163-241)."""
        if not self.node_name:
            log.warning("health checker: no NODE_NAME; node conditions disabled")
        ops = amdsmi.get_ops()
        for i in range(ops.device_count()):
            try:
                self._ecc_baseline[i] = ops.ecc_uncorrectable_count(i)
            except Exception as e:  # noqa: BLE001 - parity: tolerate Not Supported
                log.warning("ecc baseline for device %d unavailable: %s", i, e)
        if self.kube and self.node_name:
            t = threading.Thread(
                target=self._reset_condition_with_backoff, daemon=True
            )
            t.start()
            self._threads.append(t)
            t = threading.Thread(target=self._heartbeat_loop, daemon=True)
            t.start()
            self._threads.append(t)
        for target in (self._event_loop, self._ecc_poll_loop):
            t = threading.Thread(target=target, daemon=True)
            t.start()
            self._threads.append(t)

    def stop(self) -> None:
        self._stop.set()

    # -- event sources -------------------------------------------------------
    def _event_loop(self) -> None:
        """Parity: listenToEvents (health_checker.go:452-468)."""
        ops = amdsmi.get_ops()
        while not self._stop.is_set():
            try:
                events = ops.wait_events(EVENT_WAIT_MS)
            except Exception as e:  # noqa: BLE001
                log.error("event wait failed: %s", e)
                time.sleep(1)
                continue
            for ev in events:
                self.catch_error(ev)

    def _ecc_poll_loop(self) -> None:
        """Synthesizes ECC/lost-device events from polling (no AMD-SMI event
        exists for these, unlike NVML Xid 48/79)."""
        ops = amdsmi.get_ops()
        while not self._stop.wait(self.ecc_poll_interval_s):
            for i in list(self._ecc_baseline):
                try:
                    count = ops.ecc_uncorrectable_count(i)
                except Exception as e:  # noqa: BLE001
                    log.error("device %d unreachable: %s", i, e)
                    try:
                        uuid = ops.device_info(i).uuid
                    except Exception:  # noqa: BLE001
                        uuid = ""
                    self.catch_error(Event(
                        device_uuid=uuid, code=EVT_LOST,
                        message="device unreachable",
                    ))
                    continue
                if count > self._ecc_baseline[i]:
                    self._ecc_baseline[i] = count
                    try:
                        uuid = ops.device_info(i).uuid
                    except Exception:  # noqa: BLE001
                        uuid = ""
                    self.catch_error(Event(
                        device_uuid=uuid, code=EVT_ECC_UNCORRECTABLE,
                        message=f"uncorrectable ECC count {count}",
                    ))
                # xGMI link errors: synthetic code 63 on 0->error transition
                try:
                    xgmi = ops.xgmi_error_status(i)
                except Exception:  # noqa: BLE001 - not supported on 1-GPU
                    xgmi = 0
                if xgmi != 0 and self._xgmi_state.get(i, 0) == 0:
                    try:
                        uuid = ops.device_info(i).uuid
                    except Exception:  # noqa: BLE001
                        uuid = ""
                    self.catch_error(Event(
                        device_uuid=uuid, code=EVT_XGMI_ERROR,
                        message=f"xGMI link error status {xgmi}",
                    ))
                self._xgmi_state[i] = xgmi

    # -- the policy core ------------------------------------------------------
    def catch_error(self, ev: Event) -> None:
        """Parity: catchError (health_checker.go:395-449), plus post-reset
        recovery the reference cannot do (NVML has no usable post-reset
        signal; a device stays Unhealthy until plugin restart)."""
        log.warning("GPU event code=%d uuid=%s msg=%s", ev.code, ev.device_uuid,
                    ev.message)
        if ev.code == EVT_GPU_POST_RESET and self.recover_on_reset:
            self._recover_after_reset(ev)
            return
        if ev.code not in self.monitor_critical:
            return
        self._record_event(ev)
        self._monitor_condition(ev)
        if ev.code not in self.health_critical:
            return
        affected = self._affected_device_ids(ev)
        for dev_id in affected:
            log.warning("marking device %s Unhealthy (event %d)", dev_id, ev.code)
            self._unhealthy.setdefault(dev_id, set()).add(ev.code)
            self.manager.health.put(api.Device(ID=dev_id, health=api.UNHEALTHY))

    def _recover_after_reset(self, ev: Event) -> None:
        """GPU_POST_RESET: re-mark the partitions of the *matched* die
        Healthy, but only after (a) the device answers a probe again and
        (b) every persistent fault class that made it Unhealthy re-verifies
        clean.  Unlike catch_error, an empty or unmatched event UUID
        recovers NOTHING: the all-devices fallback is right when marking
        Unhealthy (never drop a node-critical signal over an id-format
        mismatch) and wrong for recovery (never return a persistently-bad
        device to the kubelet pool on a vague signal)."""
        ops = amdsmi.get_ops()
        matched = self._matched_device_ids(ev)
        if not matched:
            log.warning(
                "post-reset event uuid %r matches no enumerated device; "
                "not recovering anything", ev.device_uuid)
            return
        for dev_id in matched:
            codes = set(self._unhealthy.get(dev_id, ()))
            if not codes and self.manager.device_health.get(dev_id) != api.UNHEALTHY:
                continue
            idx = self._device_index(dev_id)
            if idx is None:
                continue
            try:
                # probe through the same seam discovery uses
                ops.memory_info(idx)
            except Exception as e:  # noqa: BLE001 - still broken, stay Unhealthy
                log.warning("post-reset probe failed for %s: %s", dev_id, e)
                continue
            if not self._fault_classes_clean(dev_id, idx, codes, ops):
                continue
            log.warning("device %s recovered after GPU reset; marking Healthy",
                        dev_id)
            self._unhealthy.pop(dev_id, None)
            self.manager.health.put(api.Device(ID=dev_id, health=api.HEALTHY))

    def _fault_classes_clean(self, dev_id: str, idx: int, codes: Set[int],
                             ops) -> bool:
        """Re-verify the persistent fault classes (uncorrectable ECC, xGMI
        link error) before declaring a reset successful.  A reset is
        expected to clear the fault; if the counter is still above the
        recorded baseline or the link still reports errors, the fault
        survived the reset and the device stays Unhealthy."""
        if EVT_ECC_UNCORRECTABLE in codes:
            try:
                count = ops.ecc_uncorrectable_count(idx)
            except Exception as e:  # noqa: BLE001
                log.warning("post-reset ECC re-check failed for %s: %s",
                            dev_id, e)
                return False
            if count > self._ecc_baseline.get(idx, 0):
                log.warning(
                    "device %s still reports uncorrectable ECC after reset "
                    "(%d > baseline %d); staying Unhealthy",
                    dev_id, count, self._ecc_baseline.get(idx, 0))
                return False
            # counters may have been cleared by the reset; resync the poll
            # baseline so the watchdog does not re-fire on the old value
            self._ecc_baseline[idx] = count
        if EVT_XGMI_ERROR in codes:
            try:
                xgmi = ops.xgmi_error_status(idx)
            except Exception as e:  # noqa: BLE001
                log.warning("post-reset xGMI re-check failed for %s: %s",
                            dev_id, e)
                return False
            if xgmi != 0:
                log.warning("device %s xGMI link still in error state %d "
                            "after reset; staying Unhealthy", dev_id, xgmi)
                return False
            self._xgmi_state[idx] = 0
        return True

    def _device_index(self, dev_id: str):
        pm = self.manager.partition_manager
        infos = pm.devices if pm else self.manager.device_infos
        info = infos.get(dev_id)
        return info.index if info is not None else None

    def _matched_device_ids(self, ev: Event):
        """Device ids whose die UUID matches the event's, or [] when the
        UUID is empty or unmatched.  No all-devices fallback here — callers
        choose their own policy for the vague cases."""
        if not ev.device_uuid:
            return []
        pm = self.manager.partition_manager
        if pm:
            return pm.devices_for_die_uuid(ev.device_uuid)
        uuid = ev.device_uuid.strip().lower()
        return [
            dev_id
            for dev_id, info in self.manager.device_infos.items()
            if info.uuid.strip().lower() == uuid
        ]

    def _affected_device_ids(self, ev: Event):
        """No UUID => every device (parity health_checker.go:415-424); with a
        UUID => all partitions of the faulting die (the CPX analog of MIG
        UUID+GI/CI matching, health_checker.go:426-445).  A UUID that
        matches NO enumerated device also marks every device: a critical
        event on this node must never be silently dropped because the
        event source formats the id differently than enumeration."""
        pm = self.manager.partition_manager
        all_ids = (list(pm.devices.keys()) if pm
                   else list(self.manager.devices.keys()))
        if not ev.device_uuid:
            return all_ids
        matched = self._matched_device_ids(ev)
        if not matched:
            log.warning(
                "event uuid %s matches no enumerated device; marking all",
                ev.device_uuid,
            )
            return all_ids
        return matched

    # -- kube plumbing ---------------------------------------------------------
    def _record_event(self, ev: Event) -> None:
        """Parity: recordXIDEvent (health_checker.go:386-393)."""
        if not (self.kube and self.node_name):
            return
        try:
            self.kube.create_event("default", {
                "metadata": {"generateName": "amd-gpu-event-"},
                "involvedObject": {"kind": "Node", "name": self.node_name},
                "reason": EVENT_REASON,
                "message": f"GPU critical event {ev.code}: {ev.message}",
                "type": "Warning",
                "source": {"component": "amd-gpu-device-plugin",
                           "host": self.node_name},
            })
        except Exception as e:  # noqa: BLE001
            log.error("failed to record event: %s", e)

    def _get_condition(self, node: dict) -> Optional[dict]:
        for c in node.get("status", {}).get("conditions", []):
            if c.get("type") == CONDITION_TYPE:
                return c
        return None

    def _monitor_condition(self, ev: Event) -> None:
        """Merge the code into the JSON Reason of the node condition; Message
        carries the BootID (parity: monitorXidevent, health_checker.go:288-346).
        """
        if not (self.kube and self.node_name):
            return
        with self._condition_lock:
            try:
                node = self.kube.get_node(self.node_name)
                cond = self._get_condition(node)
                codes = set()
                if cond and cond.get("reason"):
                    try:
                        codes = set(json.loads(cond["reason"]))
                    except (ValueError, TypeError):
                        codes = set()
                codes.add(ev.code)
                now = _now_iso()
                new_cond = {
                    "type": CONDITION_TYPE,
                    "status": "True",
                    "reason": json.dumps(sorted(codes)),
                    "message": read_boot_id(self.boot_id_path),
                    "lastHeartbeatTime": now,
                    "lastTransitionTime": (
                        cond.get("lastTransitionTime", now) if cond else now
                    ),
                }
                self._put_condition(node, new_cond)
            except Exception as e:  # noqa: BLE001
                log.error("failed to update node condition: %s", e)

    def _put_condition(self, node: dict, cond: Optional[dict]) -> None:
        conditions = node.setdefault("status", {}).setdefault("conditions", [])
        conditions[:] = [c for c in conditions if c.get("type") != CONDITION_TYPE]
        if cond is not None:
            conditions.append(cond)
        self.kube.update_node_status(self.node_name, node)

    def _heartbeat_loop(self) -> None:
        """Parity: setXIDheartbeat (health_checker.go:348-358)."""
        while not self._stop.wait(self.heartbeat_interval_s):
            self.update_heartbeat()

    def update_heartbeat(self) -> None:
        with self._condition_lock:
            try:
                node = self.kube.get_node(self.node_name)
                cond = self._get_condition(node)
                if cond is None:
                    return
                cond["lastHeartbeatTime"] = _now_iso()
                self.kube.update_node_status(self.node_name, node)
            except Exception as e:  # noqa: BLE001
                log.error("heartbeat failed: %s", e)

    def _reset_condition_with_backoff(self) -> None:
        """If the node was rebooted/auto-repaired (BootID in the condition
        Message differs from the current BootID), clear the stale condition.
        Parity: resetXIDConditionWithBackoff (health_checker.go:101-160)."""
        for delay in RESET_BACKOFF_S:
            if self.try_reset_condition():
                return
            if self._stop.wait(delay):
                return
        log.error("giving up resetting stale %s condition", CONDITION_TYPE)

    def try_reset_condition(self) -> bool:
        with self._condition_lock:
            try:
                node = self.kube.get_node(self.node_name)
                cond = self._get_condition(node)
                if cond is None:
                    return True
                current = read_boot_id(self.boot_id_path)
                if cond.get("message") == current:
                    return True  # same boot: condition is live, keep it
                self._put_condition(node, None)
                log.info("cleared stale %s condition (node was repaired)",
                         CONDITION_TYPE)
                return True
            except Exception as e:  # noqa: BLE001
                log.error("condition reset failed: %s", e)
                return False


def _now_iso() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
