"""rccl-env smoke: every env script the RCCL ConfigMaps ship is
executable in this image and every knob it sets is one the image's
actual libraries read (parity model: the reference's
set_nccl_env.sh sourcing at gpudirect-rdma/nccl-test.yaml:88).

Validation levels:
  1. each `*.sh` ConfigMap entry sources cleanly under `bash -eu`;
  2. every NCCL_* exported is a string present in this image's
     librccl binary (i.e. RCCL actually parses that variable —
     misspelled knobs silently do nothing, the worst failure mode of
     env-recipe ConfigMaps);
  3. every HSA_* exported is read by the HSA runtime the same way;
  4. the benchmark script's binary path matches the image layout the
     Dockerfile builds (/opt/cea-amd/bin/all_reduce_perf).
"""
import glob
import os
import re
import subprocess

import pytest
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

RCCL_SO = next(
    (p for p in ("/opt/rocm/lib/librccl.so.1", "/opt/rocm/lib/librccl.so")
     if os.path.exists(p)), None)
HSA_SO = next(iter(glob.glob("/opt/rocm/lib/libhsa-runtime64.so*")), None)


def configmap_scripts():
    """(yaml_path, key, script) for every shell entry in deploy ConfigMaps."""
    out = []
    for path in glob.glob(os.path.join(REPO, "deploy", "**", "*.yaml"),
                          recursive=True):
        with open(path) as f:
            try:
                docs = list(yaml.safe_load_all(f))
            except yaml.YAMLError:
                continue
        for doc in docs:
            if not isinstance(doc, dict) or doc.get("kind") != "ConfigMap":
                continue
            for key, val in (doc.get("data") or {}).items():
                if key.endswith(".sh") and isinstance(val, str):
                    out.append((os.path.relpath(path, REPO), key, val))
    return out


ENV_SCRIPTS = [(p, k, v) for p, k, v in configmap_scripts() if "env" in k]
ALL_SCRIPTS = configmap_scripts()


def test_found_env_scripts():
    keys = {k for _, k, _ in ENV_SCRIPTS}
    assert "rccl-env.sh" in keys
    assert "rccl-rdma-env.sh" in keys


@pytest.mark.parametrize("path,key,script",
                         ENV_SCRIPTS, ids=[k for _, k, _ in ENV_SCRIPTS])
def test_env_script_sources_cleanly(path, key, script):
    proc = subprocess.run(["bash", "-euc", script + "\nenv"],
                          capture_output=True, text=True, timeout=30)
    assert proc.returncode == 0, (path, key, proc.stderr)


def lib_reads(var: str, so: str) -> bool:
    return subprocess.run(["grep", "-qc", var, so],
                          capture_output=True).returncode == 0


@pytest.mark.skipif(RCCL_SO is None, reason="no librccl in image")
@pytest.mark.parametrize("path,key,script",
                         ENV_SCRIPTS, ids=[k for _, k, _ in ENV_SCRIPTS])
def test_every_exported_knob_is_read_by_the_image(path, key, script):
    exported = re.findall(r"^\s*export\s+([A-Z0-9_]+)=", script, re.M)
    assert exported, (path, key)
    for var in exported:
        if var.startswith("NCCL_"):
            assert lib_reads(var, RCCL_SO), (
                f"{var} (from {path}:{key}) is not a string in {RCCL_SO}: "
                "RCCL would silently ignore it")
        elif var.startswith("HSA_"):
            assert HSA_SO and lib_reads(var, HSA_SO), (
                f"{var} (from {path}:{key}) not read by the HSA runtime")


def test_benchmark_script_paths_match_image_layout():
    """run-allreduce.sh invokes the binary where the image puts it."""
    script = next(v for _, k, v in ALL_SCRIPTS if k == "run-allreduce.sh")
    m = re.search(r"exec\s+(\S*all_reduce_perf)", script)
    assert m, script
    bin_path = m.group(1)
    dockerfile = open(os.path.join(REPO, "Dockerfile")).read()
    # Dockerfile: WORKDIR /opt/cea-amd + `make all` + ln -s cea_amd/bin bin
    assert bin_path == "/opt/cea-amd/bin/all_reduce_perf"
    assert "ln -s /opt/cea-amd/cea_amd/bin /opt/cea-amd/bin" in dockerfile
    # and the Makefile target actually builds that binary name
    mk = open(os.path.join(REPO, "Makefile")).read()
    assert "cea_amd/bin/all_reduce_perf" in mk
    # the sourced config path matches the volume mount used by the test pods
    assert "source /configs/rccl-env.sh" in script
    for p, k, v in ALL_SCRIPTS:
        if k == "run-allreduce.sh":
            continue
    pod_yaml = open(os.path.join(REPO, "deploy", "rccl",
                                 "rccl-test.yaml")).read()
    assert "/configs" in pod_yaml and "rccl-config" in pod_yaml
