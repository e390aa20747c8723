"""Test fixtures shared across the suite.

KubeletStub: in-process gRPC Registration server on a temp unix socket,
parity with the reference's KubeletStub (beta_plugin_test.go:37-71).
"""
from __future__ import annotations

import os
import time
import threading
from concurrent import futures

import grpc

from cea_amd.kube import protos as api


class KubeletStub:
    def __init__(self, plugin_dir: str):
        self.plugin_dir = plugin_dir
        self.socket_path = os.path.join(plugin_dir, api.KUBELET_SOCKET)
        self.registered = threading.Event()
        self.requests = []
        self.server = None

    def _register(self, request, context):
        self.requests.append(request)
        self.registered.set()
        return api.Empty()

    def start(self):
        server = grpc.server(futures.ThreadPoolExecutor(max_workers=2))
        handler = grpc.method_handlers_generic_handler(
            "v1beta1.Registration",
            {
                "Register": grpc.unary_unary_rpc_method_handler(
                    self._register,
                    request_deserializer=api.RegisterRequest.FromString,
                    response_serializer=lambda m: m.SerializeToString(),
                )
            },
        )
        server.add_generic_rpc_handlers((handler,))
        server.add_insecure_port(f"unix://{self.socket_path}")
        server.start()
        self.server = server

    def stop(self):
        if self.server:
            self.server.stop(grace=0)

    def restart_kubelet_socket(self):
        """Simulate a kubelet restart: tear the registration server down,
        remove the socket, and bring a fresh server up — the plugin's
        fsnotify watcher must see the recreation and re-register (parity:
        manager.go:534-539)."""
        self.stop()
        if os.path.exists(self.socket_path):
            os.unlink(self.socket_path)
        time.sleep(0.3)
        self.start()


class PluginClient:
    """Minimal device-plugin client (what kubelet does after registration)."""

    def __init__(self, socket_path: str):
        self.channel = grpc.insecure_channel(f"unix://{socket_path}")

    def list_and_watch_once(self, timeout=5):
        stream = self.channel.unary_stream(
            api.DP_LIST_AND_WATCH,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.ListAndWatchResponse.FromString,
        )(api.Empty(), timeout=timeout)
        return stream

    def allocate(self, device_ids_per_container, timeout=5):
        req = api.AllocateRequest()
        for ids in device_ids_per_container:
            req.container_requests.add(devices_ids=ids)
        return self.channel.unary_unary(
            api.DP_ALLOCATE,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.AllocateResponse.FromString,
        )(req, timeout=timeout)

    def get_options(self, timeout=5):
        return self.channel.unary_unary(
            api.DP_GET_OPTIONS,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.DevicePluginOptions.FromString,
        )(api.Empty(), timeout=timeout)

    def close(self):
        self.channel.close()
