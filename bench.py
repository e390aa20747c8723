#!/usr/bin/env python3
"""cea_amd flagship benchmark: RCCL all_reduce bus bandwidth over xGMI.

BASELINE.json metric: "rccl-tests all_reduce bus-bw (GB/s) at 1/2/4/8 GPUs;
Allocate() p50 latency".  The reference ships nccl-tests harnesses with the
sweep 1 MiB -> 512 MiB, factor 2, 5 warmup, 100 iters, no data check
(/root/reference/gpudirect-tcpx/nccl-config.yaml:17,55-58); this bench runs
the same protocol through torch.distributed (backend "nccl" IS RCCL on
ROCm), one process per GPU over xGMI.

Step semantics (the driver's timed contract): one step = one all_reduce of
the headline 512 MiB fp32 buffer.  For world>=2 it is the in-place
collective — exactly what rccl-tests' in-place row times; for world==1
(where all_reduce is a no-op) it is the out-of-place d2d copy a world-1
out-of-place all_reduce performs.  The full message sweep runs outside the
timed region and is reported in config.sweep for the scaling curve.

bus-bw convention (nccl-tests): busbw = algbw * 2*(n-1)/n for all_reduce.
For n==1 that factor degenerates to 0, so the single-GPU value reported is
the out-of-place algbw (the d2d copy that a world-1 out-of-place all_reduce
performs); scaling judgments should use the n>=2 points.

Allocate() p50: measured in-process against the real device plugin gRPC
service over a unix socket with the mock amdsmi backend (pure control-plane
latency, no GPU needed), reported in config.allocate_p50_us.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import time
import sys


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=5)
    p.add_argument("--max-bytes", type=int, default=512 * 1024 * 1024)
    p.add_argument("--min-bytes", type=int, default=1024 * 1024)
    p.add_argument("--sweep-iters", type=int, default=100)
    p.add_argument("--sweep-warmup", type=int, default=5)
    p.add_argument("--no-sweep", action="store_true",
                   help="skip the per-size sweep (headline size only)")
    p.add_argument("--backend", default="", help="nccl|gloo (default: auto)")
    p.add_argument("--op", default="all_reduce",
                   choices=["all_reduce", "all_gather", "reduce_scatter",
                            "broadcast"],
                   help="collective to benchmark (nccl-tests -o parity; "
                        "the driver contract uses the default all_reduce)")
    return p.parse_args()


def allocate_p50_us(iters: int = 300) -> float:
    """p50 latency of the device-plugin Allocate RPC over a real unix-socket
    gRPC round-trip (the latency-critical kubelet path, beta_plugin.go:56)."""
    import tempfile
    import threading

    import cea_amd.amdsmi as amdsmi
    from cea_amd.amdsmi.mock import MockAmdSmi, make_fake_dev
    from cea_amd.deviceplugin.manager import AmdGPUManager, GPUConfig

    import grpc
    from cea_amd.kube import protos as api

    with tempfile.TemporaryDirectory() as tmp:
        dev = os.path.join(tmp, "dev")
        make_fake_dev(dev, 8)
        mock = MockAmdSmi(dev)
        mock.init()
        amdsmi.set_ops(mock)
        plugin_dir = os.path.join(tmp, "plugin")
        os.makedirs(plugin_dir)
        mgr = AmdGPUManager(GPUConfig(), dev_directory=dev,
                            plugin_directory=plugin_dir)
        mgr.config.add_defaults_and_validate()
        mgr.start()
        t = threading.Thread(target=mgr.serve, daemon=True)
        t.start()
        sock = os.path.join(plugin_dir, mgr.socket_name)
        deadline = time.time() + 5
        while not os.path.exists(sock) and time.time() < deadline:
            time.sleep(0.01)
        channel = grpc.insecure_channel(f"unix://{sock}")
        allocate = channel.unary_unary(
            api.DP_ALLOCATE,
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=api.AllocateResponse.FromString,
        )
        req = api.AllocateRequest()
        req.container_requests.add(devices_ids=["amdgpu0"])
        lat = []
        for _ in range(iters):
            t0 = time.perf_counter()
            allocate(req, timeout=5)
            lat.append((time.perf_counter() - t0) * 1e6)
        channel.close()
        mgr.stop()
        amdsmi.ops = None
        return statistics.median(lat)


def main():
    args = parse_args()
    import torch
    import torch.distributed as dist

    env_world = int(os.environ.get("WORLD_SIZE", "1"))
    world = max(env_world, 1)
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    has_gpu = torch.cuda.is_available()
    backend = args.backend or ("nccl" if has_gpu else "gloo")
    device = torch.device("cpu")
    if has_gpu:
        # modulo: lets an oversubscription experiment run world>deviceCount
        # (e.g. 2 ranks on a 1-GPU box); normal runs have world==deviceCount
        dev_idx = local_rank % torch.cuda.device_count()
        torch.cuda.set_device(dev_idx)
        device = torch.device("cuda", dev_idx)

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29571")
    # dmabuf IPC: required for RCCL cross-process CUDA-tensor sharing on
    # hosts whose driver only supports dmabuf IPC
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    kwargs = {"device_id": device} if has_gpu else {}
    dist.init_process_group(backend=backend, world_size=world, rank=rank, **kwargs)

    n = world
    elems = args.max_bytes // 4
    send = torch.empty(elems, dtype=torch.float32, device=device)
    send.uniform_()
    recv = torch.empty_like(send)

    # Step protocol (rccl-tests parity):
    #   n >= 2 — in-place all_reduce, exactly what rccl-tests' in-place row
    #     times (one ncclAllReduce, no extra traffic).  torch's all_reduce
    #     is in-place-only, so the out-of-place emulation (copy + reduce)
    #     would charge a send->recv copy rccl-tests does not time.
    #   n == 1 — all_reduce degenerates to a no-op, so time the d2d copy a
    #     world-1 out-of-place all_reduce performs, autotuned between the
    #     in-tree gfx950 kernel (cea_amd/ops) and torch copy_ (SDMA).
    copy_impl = None
    mode = "in-place"
    if n == 1:
        mode = "out-of-place"
        copy_impl = "torch"
        if has_gpu:
            from cea_amd.ops import native

            native.assert_native_available()

            def _time_copies(fn, reps=5):
                fn(recv, send)  # warm
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                for _ in range(reps):
                    fn(recv, send)
                torch.cuda.synchronize()
                return time.perf_counter() - t0

            t_native = _time_copies(lambda d, s: native.copy_(d, s))
            t_torch = _time_copies(lambda d, s: d.copy_(s))
            if t_native < t_torch:
                copy_impl = "hip-exact-nt"

                def step(buf_send, buf_recv):
                    native.copy_(buf_recv, buf_send)
                    dist.all_reduce(buf_recv)
            else:

                def step(buf_send, buf_recv):
                    buf_recv.copy_(buf_send)
                    dist.all_reduce(buf_recv)
        else:

            def step(buf_send, buf_recv):
                buf_recv.copy_(buf_send)
                dist.all_reduce(buf_recv)
    else:
        if has_gpu:
            from cea_amd.ops import native

            native.assert_native_available()

        if args.op == "all_reduce":

            def step(buf_send, buf_recv):
                dist.all_reduce(buf_recv)
        elif args.op == "all_gather":
            # rccl-tests semantics: per-rank input = size/n, output = size
            def step(buf_send, buf_recv):
                e = buf_recv.numel() // n * n
                dist.all_gather_into_tensor(buf_recv[:e],
                                            buf_send[: e // n])
        elif args.op == "reduce_scatter":

            def step(buf_send, buf_recv):
                e = buf_send.numel() // n * n
                dist.reduce_scatter_tensor(buf_recv[: e // n], buf_send[:e])
        else:  # broadcast

            def step(buf_send, buf_recv):
                dist.broadcast(buf_recv, src=0)

    def sync():
        if has_gpu:
            torch.cuda.synchronize()

    # hipGraph capture of the step (nccl-tests -G semantics: capture
    # `iters_per_graph` iterations per graph) — one replay enqueues the
    # whole captured loop, removing per-launch CPU overhead, which
    # dominates the small end of the sweep (a 1 MiB step is ~13 us eager
    # from python, most of it launch).  Default: n==1 only.  At n>=2 the
    # eager loop already pipelines (launch ~13 us/iter < comm time, no
    # per-iter sync), and a capture hang across ranks would kill the one
    # driver-run scaling measurement — robustness wins over the last few
    # percent at the 1 MiB sweep point.  CEA_BENCH_GRAPH=1 forces capture
    # on at any world size; =0 forces off.
    graph_env = os.environ.get("CEA_BENCH_GRAPH", "")
    if graph_env:
        use_graphs = has_gpu and graph_env == "1"
    else:
        use_graphs = has_gpu and n == 1

    def make_step(buf_send, buf_recv, iters_per_graph=1):
        """Returns (callable, is_graph); the callable executes
        `iters_per_graph` timed steps."""
        if not use_graphs:

            def eager():
                for _ in range(iters_per_graph):
                    step(buf_send, buf_recv)

            return eager, False
        try:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    step(buf_send, buf_recv)
            torch.cuda.current_stream().wait_stream(side)
            # Let the ProcessGroupNCCL watchdog dequeue the completed
            # warmup works before capture starts: its hipEventQuery on a
            # still-enqueued work during capture is
            # hipErrorStreamCaptureUnsupported and aborts the process
            # (a watchdog-thread abort cannot be caught by the except
            # below).  The watchdog sweeps its work list every ~100 ms,
            # so poll in slices — synchronize each slice so every work's
            # event is complete when the watchdog looks — instead of one
            # fixed sleep; bound is tunable for loaded nodes, and
            # CEA_BENCH_GRAPH=0 disables capture entirely.
            drain_ms = int(os.environ.get("CEA_BENCH_GRAPH_DRAIN_MS", "600"))
            deadline = time.monotonic() + drain_ms / 1000.0
            while True:
                torch.cuda.synchronize()
                if time.monotonic() >= deadline:
                    break
                time.sleep(0.1)
            g = torch.cuda.CUDAGraph()
            # thread_local: event queries from OTHER threads (the NCCL
            # watchdog) stay legal while this thread captures.
            with torch.cuda.graph(g, capture_error_mode="thread_local"):
                for _ in range(iters_per_graph):
                    step(buf_send, buf_recv)
            return g.replay, True
        except Exception as e:  # noqa: BLE001 - capture support is optional
            print(f"hipGraph capture failed ({e}); falling back to eager. "
                  "Set CEA_BENCH_GRAPH=0 to skip capture, or raise "
                  "CEA_BENCH_GRAPH_DRAIN_MS on loaded nodes.",
                  file=sys.stderr)
            torch.cuda.synchronize()

            def eager():
                for _ in range(iters_per_graph):
                    step(buf_send, buf_recv)

            return eager, False

    def busbw_factor(nranks: int) -> float:
        """nccl-tests conventions per collective."""
        if nranks <= 1:
            return 1.0
        if args.op == "all_reduce":
            return 2.0 * (nranks - 1) / nranks
        if args.op in ("all_gather", "reduce_scatter"):
            return (nranks - 1) / nranks
        return 1.0  # broadcast

    # ---- per-size sweep (reference protocol: -w 5 --iters 100 -c 0) ------
    sweep = []
    if not args.no_sweep:
        size = args.min_bytes
        while size <= args.max_bytes:
            e = size // 4
            s_send, s_recv = send[:e], recv[:e]
            # whole sweep loop in one graph (-G sweep_iters equivalent):
            # one replay = all iterations, so python launch cost is paid
            # once per size instead of once per iteration.
            run_iters, _ = make_step(s_send, s_recv,
                                     iters_per_graph=args.sweep_iters)
            run_warm, _ = make_step(s_send, s_recv,
                                    iters_per_graph=args.sweep_warmup)
            run_warm()
            sync()
            dist.barrier()
            t0 = time.perf_counter()
            run_iters()
            sync()
            el = time.perf_counter() - t0
            dist.barrier()
            t_max = torch.tensor([el], dtype=torch.float64, device=device)
            dist.all_reduce(t_max, op=dist.ReduceOp.MAX)
            t_iter = t_max.item() / args.sweep_iters
            algbw = size / t_iter / 1e9
            sweep.append(
                {
                    "bytes": size,
                    "time_us": round(t_iter * 1e6, 2),
                    "algbw_GBps": round(algbw, 2),
                    "busbw_GBps": round(algbw * busbw_factor(n), 2),
                }
            )
            size *= 2

    # ---- timed region: K steps at the headline size ----------------------
    run_step, is_graph = make_step(send, recv)
    for _ in range(args.warmup):
        run_step()
    sync()
    dist.barrier()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        run_step()
    sync()
    elapsed = time.perf_counter() - t0
    dist.barrier()
    t_max = torch.tensor([elapsed], dtype=torch.float64, device=device)
    dist.all_reduce(t_max, op=dist.ReduceOp.MAX)
    elapsed = t_max.item()

    ms_per_step = elapsed / args.steps * 1e3
    algbw = args.max_bytes / (elapsed / args.steps) / 1e9
    busbw = algbw * busbw_factor(n)

    if rank == 0:
        try:
            p50 = round(allocate_p50_us(), 1)
        except Exception as e:  # noqa: BLE001 - p50 is auxiliary
            p50 = None
        result = {
            "metric": f"rccl-tests {args.op} bus-bw (GB/s)",
            "value": round(busbw, 2),
            "unit": "GB/s",
            "n_gpus": n,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": f"{args.op}_perf",
                "collective": args.op,
                "message_bytes": args.max_bytes,
                "mode": mode,
                "global_batch": None,
                "seq_len": None,
                "parallelism": f"dp{n}",
                "backend": backend,
                "transport": "RCCL/xGMI" if has_gpu else "gloo(cpu-test)",
                "copy_impl": copy_impl,
                "hip_graph": bool(is_graph),
                "busbw_factor": round(busbw_factor(n), 4),
                "algbw_GBps": round(algbw, 2),
                "sweep": sweep,
                "allocate_p50_us": p50,
                # RCCL tuning in effect (empty = library defaults; the
                # shipped recipe is deploy/rccl/rccl-config.yaml)
                "nccl_env": {k: v for k, v in sorted(os.environ.items())
                             if k.startswith(("NCCL_", "RCCL_"))},
            },
        }
        print(json.dumps(result))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
