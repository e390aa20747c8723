"""GPU test for the native RCCL benchmark binary (all_reduce_perf analog)."""
import os
import re
import subprocess

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BIN = os.path.join(REPO, "cea_amd", "bin", "all_reduce_perf")

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("needs an MI355X", allow_module_level=True)


def test_binary_exists():
    assert os.path.exists(BIN), "make rcclbench not run — native bench missing"


def run_bench(extra):
    env = dict(os.environ, HSA_ENABLE_IPC_MODE_LEGACY="0")
    r = subprocess.run(
        [BIN, "-b", "1M", "-e", "16M", "-f", "2", "-g", "1", "-w", "2",
         "-n", "10"] + extra,
        capture_output=True, text=True, timeout=300, env=env,
    )
    assert r.returncode == 0, r.stdout + r.stderr
    return r.stdout


def test_all_reduce_sweep_with_check():
    out = run_bench(["-c", "1"])
    rows = [l for l in out.splitlines() if re.match(r"^\s+\d+", l)]
    assert len(rows) == 5  # 1M..16M factor 2
    # last column busbw must be positive and parse as float
    for row in rows:
        cols = row.split()
        assert float(cols[3]) > 0
    assert "# done" in out


@pytest.mark.parametrize("op", ["all_gather", "reduce_scatter", "broadcast"])
def test_other_collectives(op):
    out = run_bench(["-o", op])
    assert "# done" in out


def test_graph_mode_and_dtypes():
    """-G hipGraph replay and -d datatypes (bf16/fp8) must complete the
    sweep (nccl-tests CLI parity features)."""
    out = run_bench(["-G", "1", "-c", "1"])
    assert "# done" in out
    out = run_bench(["-d", "bf16"])
    assert "dtype=bf16" in out and "# done" in out
    out = run_bench(["-d", "fp8e4m3"])
    assert "# done" in out
