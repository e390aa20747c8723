"""AMD GPU manager — the device plugin's core state machine.

Role parity: /root/reference/pkg/gpu/nvidia/manager.go (560 LoC), redesigned
for MI355X:
  * device nodes are /dev/kfd (shared, always injected — the analog of
    nvidiactl+nvidia-uvm defaultDevices, manager.go:378-388) plus one
    /dev/dri/renderD<minor> per allocated GPU (vs /dev/nvidiaN),
  * discovery goes through the AmdSmiOperations seam (vs NVML),
  * partitioning is SPX/DPX/QPX/CPX render-node enumeration (vs MIG capability
    files),
  * the MPS strategy maps to CU-mask env fencing (cu-fencing) since ROCm has
    no MPS daemon.
The serve/restart state machine (socket poll / GPU hot-add poll / kubelet
restart watch, manager.go:442-549) is kept trigger-for-trigger.
"""
from __future__ import annotations

import dataclasses
import logging
import os
import queue
import re
import threading
import time
from concurrent import futures
from typing import Dict, List, Optional

import grpc

from .. import amdsmi
from ..amdsmi.iface import numa_node_for_bdf
from ..kube import protos as api
from . import RESOURCE_NAME, sharing
from .partition import PartitionDeviceManager
from .util import FileWatcher

log = logging.getLogger(__name__)

RENDERD_DEV_RE = re.compile(r"^renderD[0-9]+$")  # parity: gpuCheckRegexp manager.go:55
KFD_PATH = "kfd"

MI355X_TOTAL_CUS = 256

# watchdog cadences, parity manager.go:56-57
SOCKET_CHECK_INTERVAL_S = 1.0
GPU_CHECK_INTERVAL_S = 10.0


@dataclasses.dataclass
class GPUSharingConfig:
    # parity: GPUSharingConfig manager.go:80-90
    gpu_sharing_strategy: str = ""
    max_shared_clients_per_gpu: int = 0


@dataclasses.dataclass
class GPUConfig:
    # parity: GPUConfig manager.go:72-78
    compute_partition: str = ""  # "", "spx", "dpx", "cpx", "cpx-nps1", ...
    gpu_sharing_config: Optional[GPUSharingConfig] = None
    health_critical_events: set = dataclasses.field(
        default_factory=lambda: set(amdsmi.iface.DEFAULT_HEALTH_CRITICAL_EVENTS)
    )
    gpu_fraction_divisor: int = 1

    def add_defaults_and_validate(self) -> None:
        # parity: AddDefaultsAndValidate manager.go:92-117
        if self.gpu_fraction_divisor < 1:
            self.gpu_fraction_divisor = 1
        sc = self.gpu_sharing_config
        if sc and sc.gpu_sharing_strategy:
            if sc.gpu_sharing_strategy not in sharing.VALID_STRATEGIES:
                raise ValueError(
                    f"invalid GPU sharing strategy {sc.gpu_sharing_strategy!r}:"
                    f" want one of {sharing.VALID_STRATEGIES}"
                )
            if sc.max_shared_clients_per_gpu < 1:
                raise ValueError(
                    "maxSharedClientsPerGPU must be >= 1 when a sharing "
                    "strategy is set"
                )
            sharing.sharing_strategy = sc.gpu_sharing_strategy
        else:
            sharing.sharing_strategy = ""

    def add_health_critical_events(self, env_value: str) -> None:
        """Parse the EVENT_CONFIG env (comma-separated ints from a ConfigMap),
        parity: AddHealthCriticalXid / XID_CONFIG (manager.go:119-139)."""
        if not env_value:
            return
        events = set()
        for tok in env_value.split(","):
            tok = tok.strip()
            if not tok:
                continue
            if not tok.lstrip("-").isdigit():
                raise ValueError(f"invalid event code in EVENT_CONFIG: {tok!r}")
            events.add(int(tok))
        if events:
            self.health_critical_events = events

    @property
    def max_shared_clients(self) -> int:
        sc = self.gpu_sharing_config
        return sc.max_shared_clients_per_gpu if sc and sc.gpu_sharing_strategy else 0


class AmdGPUManager:
    """Parity: nvidiaGPUManager (manager.go:142-157)."""

    def __init__(
        self,
        config: GPUConfig,
        dev_directory: str = "/dev",
        host_path: str = "/home/kubernetes/bin/amd",
        container_path: str = "/usr/local/amd",
        plugin_directory: str = "/device-plugin",
        sysfs_root: str = "/sys",
        extra_mounts: Optional[List[dict]] = None,
    ):
        self.config = config
        self.dev_directory = dev_directory
        self.host_path = host_path
        self.container_path = container_path
        self.plugin_directory = plugin_directory
        self.sysfs_root = sysfs_root
        # optional additional read-only mounts (e.g. the OpenCL vendor ICD —
        # the MI355X-compute analog of the reference's Vulkan ICD mounts,
        # nvidia_gpu.go:50-61); only mounted when the host path exists
        self.extra_mounts = list(extra_mounts or [])

        self.devices: Dict[str, api.Device] = {}
        self.device_infos: Dict[str, "amdsmi.DeviceInfo"] = {}
        # health overlay applied in list_devices(); keyed by base device id
        self.device_health: Dict[str, str] = {}
        self.health: "queue.Queue[api.Device]" = queue.Queue()
        self.grpc_server: Optional[grpc.Server] = None
        self.socket_name = "amdgpu.sock"
        self.partition_manager: Optional[PartitionDeviceManager] = None
        self.total_mem_per_gpu = 0
        self._stop = threading.Event()
        self._num_devices_at_start = 0

    # -- discovery ----------------------------------------------------------
    def check_device_paths(self) -> None:
        """Block-worthy readiness probe: /dev/kfd plus >=1 render node.
        Parity: CheckDevicePaths waiting for nvidiactl/uvm (manager.go:366)."""
        kfd = os.path.join(self.dev_directory, KFD_PATH)
        if not os.path.exists(kfd):
            raise FileNotFoundError(f"{kfd} not found (amdgpu driver not ready)")
        if self.discover_num_gpus() == 0:
            raise FileNotFoundError(
                f"no {self.dev_directory}/dri/renderD* nodes found"
            )

    def discover_num_gpus(self) -> int:
        """Count render nodes by regexp, parity: discoverNumGPUs
        (manager.go:288-304)."""
        dri = os.path.join(self.dev_directory, "dri")
        if not os.path.isdir(dri):
            return 0
        return sum(1 for f in os.listdir(dri) if RENDERD_DEV_RE.match(f))

    def has_additional_gpus_installed(self) -> bool:
        # parity: hasAdditionalGPUsInstalled (manager.go:271-286)
        n = self.discover_num_gpus()
        if n > self._num_devices_at_start:
            log.warning(
                "found %d GPUs, started with %d: restarting plugin",
                n, self._num_devices_at_start,
            )
            return True
        return False

    def discover_gpus(self) -> None:
        """Enumerate physical GPUs via amdsmi, naming devices amdgpu<index>
        and attaching NUMA topology from sysfs.
        Parity: discoverGPUs (manager.go:237-269)."""
        ops = amdsmi.get_ops()
        n = ops.device_count()
        self.devices.clear()
        self.device_infos.clear()
        for i in range(n):
            info = ops.device_info(i)
            name = f"amdgpu{i}"
            d = api.Device(ID=name, health=api.HEALTHY)
            numa = numa_node_for_bdf(info.bdf, self.sysfs_root)
            if numa is not None:
                d.topology.nodes.add(ID=numa)
            self.devices[name] = d
            self.device_infos[name] = info
            log.info("found device %s (renderD%d, bdf %s)", name, info.render_minor, info.bdf)

    # -- device fan-out / specs ---------------------------------------------
    def _base_devices(self) -> Dict[str, api.Device]:
        if self.partition_manager:
            return self.partition_manager.list_devices(self.sysfs_root)
        return dict(self.devices)

    def set_device_health(self, device_id: str, health: str) -> None:
        """Record a health override for a base (physical or partition)
        device; virtual ids are collapsed to their base first."""
        if sharing.is_virtual_id(device_id):
            device_id = sharing.virtual_to_physical(device_id)
        self.device_health[device_id] = health
        if device_id in self.devices:
            self.devices[device_id].health = health

    def list_devices(self) -> List[api.Device]:
        """Physical (or partition) devices, multiplied into virtual clones
        when sharing or a fraction divisor is on.
        Parity: ListDevices (manager.go:187-204)."""
        base = self._base_devices()
        for dev_id, health in self.device_health.items():
            if dev_id in base:
                base[dev_id].health = health
        clients = self.config.max_shared_clients
        if clients == 0 and self.config.gpu_fraction_divisor > 1:
            clients = self.config.gpu_fraction_divisor
        if clients == 0:
            return list(base.values())
        out = []
        for dev_id, dev in base.items():
            for i in range(clients):
                v = api.Device()
                v.CopyFrom(dev)
                v.ID = sharing.virtual_id(dev_id, i)
                out.append(v)
        return out

    def device_spec(self, device_id: str) -> List[dict]:
        """Virtual -> physical, then the render-node DeviceSpec.
        Parity: DeviceSpec (manager.go:207-234)."""
        if sharing.is_virtual_id(device_id):
            device_id = sharing.virtual_to_physical(device_id)
        if self.partition_manager:
            specs = self.partition_manager.device_spec(device_id)
            if specs is None:
                raise KeyError(f"invalid allocation request with device {device_id}")
            return specs
        if device_id not in self.devices:
            raise KeyError(f"invalid allocation request with device {device_id}")
        info = self.device_infos[device_id]
        path = f"{self.dev_directory}/dri/renderD{info.render_minor}"
        return [{"host_path": path, "container_path": path, "permissions": "mrw"}]

    def default_devices(self) -> List[dict]:
        """Devices injected into every GPU container: /dev/kfd is the shared
        compute door on amdgpu (the analog of nvidiactl+nvidia-uvm,
        manager.go:378-388)."""
        kfd = os.path.join(self.dev_directory, KFD_PATH)
        return [{"host_path": kfd, "container_path": "/dev/kfd", "permissions": "mrw"}]

    def mounts(self) -> List[dict]:
        """Driver/userspace tree mount, parity with the reference mounting
        /home/kubernetes/bin/nvidia (nvidia_gpu.go:113-115), plus any
        existing extra mounts (OpenCL ICD analog of the Vulkan ICD paths)."""
        out = [
            {
                "host_path": self.host_path,
                "container_path": self.container_path,
                "read_only": True,
            }
        ]
        for m in self.extra_mounts:
            if os.path.exists(m["host_path"]):
                out.append({
                    "host_path": m["host_path"],
                    "container_path": m["container_path"],
                    "read_only": True,
                })
        return out

    def envs(self, num_virtual_requested: int) -> Dict[str, str]:
        """Per-container env fencing for the cu-fencing strategy: the
        MPS-thread-percentage analog (manager.go Envs() :335-348) realized
        with HSA_CU_MASK over the MI355X's 256 CUs.  VRAM limit is advisory
        (ROCm does not enforce one); hard memory isolation = CPX partitions.
        """
        if sharing.sharing_strategy != sharing.CU_FENCING:
            return {}
        clients = max(1, self.config.max_shared_clients)
        share = max(1, num_virtual_requested)
        ncus = max(1, (MI355X_TOTAL_CUS * share) // clients)
        ncus = min(ncus, MI355X_TOTAL_CUS)
        envs = {
            "HSA_CU_MASK": f"0:0-{ncus - 1}",
            "GPU_MAX_HW_QUEUES": "4",
        }
        if self.total_mem_per_gpu > 0:
            envs["CEA_AMD_VRAM_LIMIT_BYTES"] = str(
                self.total_mem_per_gpu * share // clients
            )
        return envs

    # -- lifecycle -----------------------------------------------------------
    def start(self) -> None:
        """Parity: Start (manager.go:378-420)."""
        self.discover_gpus()
        self._num_devices_at_start = self.discover_num_gpus()
        if self.config.compute_partition and self.config.compute_partition.upper().split("-")[0] != "SPX":
            pm = PartitionDeviceManager(
                self.config.compute_partition, dev_root=self.dev_directory
            )
            pm.start(self.sysfs_root)
            self.partition_manager = pm
        if sharing.sharing_strategy == sharing.CU_FENCING:
            try:
                self.total_mem_per_gpu = amdsmi.get_ops().memory_info(0).total_bytes
            except Exception as e:  # noqa: BLE001 - cache is best-effort
                log.warning("could not read total VRAM: %s", e)

    def serve(self, kubelet_socket: Optional[str] = None, max_restarts: Optional[int] = None) -> None:
        """gRPC serve + kubelet registration + watchdog restart loop.
        Parity: Serve (manager.go:442-549).  Restart triggers:
          (a) plugin socket vanished (1 s poll, manager.go:515-521)
          (b) new GPUs appeared (10 s poll, manager.go:523-532)
          (c) kubelet.sock recreated = kubelet restart (manager.go:534-539)
        `max_restarts` bounds the loop for tests (None = forever)."""
        from .plugin_service import PluginService  # local import, avoids cycle

        kubelet_socket = kubelet_socket or os.path.join(
            self.plugin_directory, api.KUBELET_SOCKET
        )
        restarts = 0
        first = True
        while not self._stop.is_set():
            if not first:
                # A restart trigger fired: re-run discovery so hot-added
                # GPUs (or recreated partitions) are advertised
                # (parity: the reference re-Starts the manager after its
                # watchdog fires, manager.go:511-544).
                try:
                    self.start()
                except Exception as e:  # noqa: BLE001
                    log.error("restart discovery failed (retrying): %s", e)
                    time.sleep(2)
                    continue
            first = False
            socket_path = os.path.join(self.plugin_directory, self.socket_name)
            try:
                os.unlink(socket_path)
            except OSError:
                pass
            service = PluginService(self)
            # Allocate() is the latency-critical RPC (pure in-memory map
            # lookups, parity manager.go Allocate path); bias grpc for
            # latency.  Measured on an MI355X box: p50 ≈ 220 µs end-to-end
            # with a python-grpc client — the python-grpc round-trip floor;
            # the handler itself is ≈7 µs (profiles/BENCH_n1_r01c.json).
            server = grpc.server(
                futures.ThreadPoolExecutor(max_workers=8),
                options=[("grpc.optimization_target", "latency")],
            )
            service.add_to_server(server)
            server.add_insecure_port(f"unix://{socket_path}")
            server.start()
            self.grpc_server = server
            log.info("device plugin listening on %s", socket_path)
            try:
                # Snapshot the kubelet-socket identity BEFORE the
                # registration attempt: a kubelet that comes up between a
                # failed registration and the watchdog's first stat would
                # otherwise be baselined as "already there" and never
                # trigger the re-register restart.
                kubelet_watch = FileWatcher(kubelet_socket)
                if os.path.exists(kubelet_socket):
                    self.register_with_kubelet(kubelet_socket)
                else:
                    log.warning("kubelet socket %s absent; serving unregistered", kubelet_socket)
                self._status_check_loop(socket_path, kubelet_watch)
            finally:
                server.stop(grace=0.5)
                self.grpc_server = None
            restarts += 1
            if max_restarts is not None and restarts > max_restarts:
                return

    def register_with_kubelet(self, kubelet_socket: str) -> None:
        """Parity: RegisterWithV1Beta1Kubelet (beta_plugin.go:110-131)."""
        with grpc.insecure_channel(f"unix://{kubelet_socket}") as channel:
            register = channel.unary_unary(
                api.REGISTRATION_REGISTER,
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=api.Empty.FromString,
            )
            req = api.RegisterRequest(
                version=api.DEVICE_PLUGIN_VERSION,
                endpoint=self.socket_name,
                resource_name=RESOURCE_NAME,
            )
            req.options.get_preferred_allocation_available = True
            register(req, timeout=10)
        log.info("registered %s with kubelet", RESOURCE_NAME)

    def preferred_allocation(self, available, must_include, size: int):
        """Die/NUMA-aware device selection for GetPreferredAllocation —
        the reference stubs this RPC (beta_plugin.go:95-103); on MI355X it
        is worth implementing: CPX partitions of one die share L3/HBM and
        the xGMI endpoint, so packing a request onto as few dies as
        possible keeps the workload local and leaves whole dies free for
        the next gang member.  Greedy: honor must_include, then repeatedly
        take candidates from the die with the most free partitions
        (preferring dies already used by this request)."""
        chosen: List[str] = []
        seen = set()
        for d in must_include:
            if d not in seen:
                chosen.append(d)
                seen.add(d)
        chosen = chosen[:size]

        def die_of(dev_id: str) -> str:
            base = (sharing.virtual_to_physical(dev_id)
                    if sharing.is_virtual_id(dev_id) else dev_id)
            return base.split("/")[0]

        remaining: Dict[str, List[str]] = {}
        for d in available:
            if d in seen:
                continue
            remaining.setdefault(die_of(d), []).append(d)
        for ids in remaining.values():
            ids.sort()
        used_dies = {die_of(d) for d in chosen}
        while len(chosen) < size and remaining:
            need = size - len(chosen)
            # dies this request already touches come first; then the
            # smallest die that still covers the need (exact-fit packing,
            # keeps big dies whole); if none covers it, the largest.
            fits = [k for k in remaining if len(remaining[k]) >= need]
            pool = fits or list(remaining)
            die = min(
                pool,
                key=lambda k: (
                    k not in used_dies,
                    len(remaining[k]) if fits else -len(remaining[k]),
                    k,
                ),
            )
            ids = remaining.pop(die)
            used_dies.add(die)
            for d in ids:
                if len(chosen) >= size:
                    break
                chosen.append(d)
                seen.add(d)
        return chosen

    def _status_check_loop(self, socket_path: str,
                           kubelet_watch: FileWatcher) -> None:
        last_gpu_check = time.monotonic()
        while not self._stop.is_set():
            time.sleep(SOCKET_CHECK_INTERVAL_S)
            if not os.path.lexists(socket_path):
                log.warning("device plugin socket %s removed; restarting", socket_path)
                return
            if time.monotonic() - last_gpu_check >= GPU_CHECK_INTERVAL_S:
                last_gpu_check = time.monotonic()
                if self.has_additional_gpus_installed():
                    return
            if kubelet_watch.changed():
                log.warning("kubelet socket changed (kubelet restart); restarting plugin")
                return

    def stop(self) -> None:
        self._stop.set()
