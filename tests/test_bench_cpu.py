"""CPU validation of the bench contract: 2-process gloo all_reduce, JSON
output schema, and the Allocate p50 microbenchmark."""
import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_bench_two_process_gloo():
    port = free_port()
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update(
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE="2",
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(port),
            PYTHONPATH=REPO,
        )
        procs.append(
            subprocess.Popen(
                [
                    sys.executable,
                    os.path.join(REPO, "bench.py"),
                    "--gpus", "2",
                    "--steps", "3",
                    "--warmup", "1",
                    "--max-bytes", str(1 << 20),
                    "--min-bytes", str(1 << 18),
                    "--sweep-iters", "2",
                    "--sweep-warmup", "1",
                    "--backend", "gloo",
                ],
                stdout=subprocess.PIPE,
                stderr=subprocess.PIPE,
                text=True,
                env=env,
            )
        )
    outs = [p.communicate(timeout=240) for p in procs]
    for p, (out, err) in zip(procs, outs):
        assert p.returncode == 0, err
    line = [l for l in outs[0][0].splitlines() if l.startswith("{")][-1]
    result = json.loads(line)
    assert result["metric"].startswith("rccl-tests all_reduce")
    assert result["n_gpus"] == 2
    assert result["steps"] == 3
    assert result["value"] > 0
    assert result["ms_per_step"] > 0
    # busbw factor for n=2 is 2*(2-1)/2 = 1.0
    assert abs(result["config"]["busbw_factor"] - 1.0) < 1e-6
    # n>=2 times the in-place collective (rccl-tests in-place row)
    assert result["config"]["mode"] == "in-place"
    assert len(result["config"]["sweep"]) == 3  # 256K, 512K, 1M
    assert result["config"]["allocate_p50_us"] is None or result["config"]["allocate_p50_us"] > 0
    # rank 1 must not print the JSON line
    assert not [l for l in outs[1][0].splitlines() if l.startswith("{")]


def test_allocate_p50_microbench():
    sys.path.insert(0, REPO)
    import bench

    p50 = bench.allocate_p50_us(iters=50)
    assert 0 < p50 < 100000  # a unix-socket RPC should be well under 100 ms


def test_bench_torchrun_launch_parity():
    """The driver launches bench via `python -m torch.distributed.run
    --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 ...` — exercise
    that exact mechanism (env propagation, rendezvous, rank-0-only JSON)
    at world 4 on gloo."""
    port = free_port()
    env = dict(os.environ, PYTHONPATH=REPO)
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node", "4",
         "--master-addr", "127.0.0.1", "--master-port", str(port),
         os.path.join(REPO, "bench.py"),
         "--gpus", "4", "--steps", "2", "--warmup", "1",
         "--max-bytes", str(1 << 20), "--no-sweep", "--backend", "gloo"],
        capture_output=True, text=True, timeout=300, env=env,
    )
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [l for l in proc.stdout.splitlines() if l.startswith('{"metric"')]
    assert len(lines) == 1, "exactly one JSON line (rank 0)"
    result = json.loads(lines[0])
    assert result["n_gpus"] == 4
    assert result["config"]["parallelism"] == "dp4"
    # busbw factor for n=4 all_reduce: 2*(4-1)/4 = 1.5
    assert abs(result["config"]["busbw_factor"] - 1.5) < 1e-6
