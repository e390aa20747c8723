"""kubelet device-plugin v1beta1 gRPC service.

Parity: /root/reference/pkg/gpu/nvidia/beta_plugin.go —
GetDevicePluginOptions (:35), ListAndWatch (:39-54, initial send +
resend-on-health-event), Allocate (:56-93, sharing validation -> DeviceSpecs
-> default devices -> mounts -> fencing envs), PreStartContainer /
GetPreferredAllocation error stubs (:95-103).

grpcio generic handlers are used because this image has no protoc; the wire
format is identical to the generated stubs (see cea_amd/kube/protos.py).
"""
from __future__ import annotations

import logging
import queue

import grpc

from ..kube import protos as api
from . import sharing

log = logging.getLogger(__name__)


class PluginService:
    def __init__(self, manager):
        self.manager = manager

    # -- RPC handlers --------------------------------------------------------
    def get_device_plugin_options(self, request, context):
        opts = api.DevicePluginOptions()
        # unlike the reference (error stub, beta_plugin.go:95-103) this
        # plugin implements die-aware preferred allocation
        opts.get_preferred_allocation_available = True
        return opts

    def list_and_watch(self, request, context):
        """Initial device list, then a resend whenever a health event drains
        from the manager's channel (parity beta_plugin.go:39-54)."""
        log.info("device-plugin: ListAndWatch start")
        yield api.ListAndWatchResponse(devices=self.manager.list_devices())
        while context.is_active():
            try:
                changed = self.manager.health.get(timeout=1.0)
            except queue.Empty:
                continue
            # Apply the health mutation (+ any further queued ones) before
            # resending the whole list.
            while True:
                self.manager.set_device_health(changed.ID, changed.health)
                try:
                    changed = self.manager.health.get_nowait()
                except queue.Empty:
                    break
            yield api.ListAndWatchResponse(devices=self.manager.list_devices())

    def allocate(self, request, context):
        """Parity: Allocate (beta_plugin.go:56-93)."""
        resp = api.AllocateResponse()
        num_physical = len(self.manager.devices) or 1
        for creq in request.container_requests:
            ids = list(creq.devices_ids)
            try:
                sharing.validate_request(ids, num_physical)
            except sharing.SharingError as e:
                context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(e))
            cresp = resp.container_responses.add()
            num_virtual = 0
            seen_paths = set()
            for dev_id in ids:
                if sharing.is_virtual_id(dev_id):
                    num_virtual += 1
                try:
                    specs = self.manager.device_spec(dev_id)
                except KeyError as e:
                    context.abort(grpc.StatusCode.INVALID_ARGUMENT, str(e))
                for s in specs:
                    if s["host_path"] in seen_paths:
                        continue
                    seen_paths.add(s["host_path"])
                    cresp.devices.add(
                        host_path=s["host_path"],
                        container_path=s["container_path"],
                        permissions=s["permissions"],
                    )
            for s in self.manager.default_devices():
                cresp.devices.add(
                    host_path=s["host_path"],
                    container_path=s["container_path"],
                    permissions=s["permissions"],
                )
            for m in self.manager.mounts():
                cresp.mounts.add(
                    host_path=m["host_path"],
                    container_path=m["container_path"],
                    read_only=m.get("read_only", False),
                )
            for k, v in self.manager.envs(num_virtual).items():
                cresp.envs[k] = v
        return resp

    def get_preferred_allocation(self, request, context):
        """Die/NUMA-aware selection (an MI355X improvement over the
        reference's error stub, beta_plugin.go:95-103): pack each container
        request onto as few physical dies as possible."""
        resp = api.PreferredAllocationResponse()
        for creq in request.container_requests:
            chosen = self.manager.preferred_allocation(
                list(creq.available_deviceIDs),
                list(creq.must_include_deviceIDs),
                int(creq.allocation_size),
            )
            cresp = resp.container_responses.add()
            cresp.device_ids.extend(chosen)
        return resp

    def pre_start_container(self, request, context):
        log.error("PreStartContainer should not be called")
        return api.PreStartContainerResponse()

    # -- wiring --------------------------------------------------------------
    def add_to_server(self, server: grpc.Server) -> None:
        rpcs = {
            "GetDevicePluginOptions": grpc.unary_unary_rpc_method_handler(
                self.get_device_plugin_options,
                request_deserializer=api.Empty.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
            "ListAndWatch": grpc.unary_stream_rpc_method_handler(
                self.list_and_watch,
                request_deserializer=api.Empty.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
            "Allocate": grpc.unary_unary_rpc_method_handler(
                self.allocate,
                request_deserializer=api.AllocateRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
            "GetPreferredAllocation": grpc.unary_unary_rpc_method_handler(
                self.get_preferred_allocation,
                request_deserializer=api.PreferredAllocationRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
            "PreStartContainer": grpc.unary_unary_rpc_method_handler(
                self.pre_start_container,
                request_deserializer=api.PreStartContainerRequest.FromString,
                response_serializer=lambda m: m.SerializeToString(),
            ),
        }
        server.add_generic_rpc_handlers(
            (grpc.method_handlers_generic_handler("v1beta1.DevicePlugin", rpcs),)
        )
