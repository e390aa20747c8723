"""ctypes bindings for the in-tree HIP kernel library (csrc/gpu_ops.hip).

The library is built for gfx950 only and lives IN-TREE
(cea_amd/ops/libceaamd_gpu.so) so it travels to GPU nodes with the source
snapshot.  On a machine with a GPU, a missing library is a hard error —
GPU paths never silently fall back to eager/CPU.
"""
from __future__ import annotations

import ctypes
import os
from typing import Optional

_LIB_NAME = "libceaamd_gpu.so"
_lib: Optional[ctypes.CDLL] = None


def _gpu_present() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:  # noqa: BLE001
        return os.path.exists("/dev/kfd")


def load_library() -> ctypes.CDLL:
    global _lib
    if _lib is not None:
        return _lib
    path = os.path.join(os.path.dirname(os.path.abspath(__file__)), _LIB_NAME)
    if not os.path.exists(path):
        raise RuntimeError(
            f"HIP kernel library not built: {path} missing — run `make gpu` "
            "(hipcc --offload-arch=gfx950). GPU ops never fall back."
        )
    _lib = ctypes.CDLL(path)
    _lib.cea_gpu_last_error.restype = ctypes.c_char_p
    return _lib


def _check(lib, rc: int, what: str) -> None:
    if rc != 0:
        err = lib.cea_gpu_last_error().decode(errors="replace")
        raise RuntimeError(f"{what} failed (rc={rc}): {err}")


def assert_native_available() -> None:
    """Called by GPU entrypoints: on a GPU machine the native library must
    load and see the device."""
    lib = load_library()
    if _gpu_present():
        n = ctypes.c_int()
        _check(lib, lib.cea_gpu_device_count(ctypes.byref(n)), "device_count")
        if n.value < 1:
            raise RuntimeError("GPU present (/dev/kfd) but HIP sees 0 devices")


def device_probe(device: int = 0) -> dict:
    """VRAM + CU count + wavefront size (parity:
    reference example/cuda-mps/cuda_mem_and_sm_count.c)."""
    lib = load_library()
    free_b = ctypes.c_ulonglong()
    total_b = ctypes.c_ulonglong()
    cu = ctypes.c_int()
    wf = ctypes.c_int()
    name = ctypes.create_string_buffer(256)
    _check(
        lib,
        lib.cea_device_probe(
            device,
            ctypes.byref(free_b),
            ctypes.byref(total_b),
            ctypes.byref(cu),
            ctypes.byref(wf),
            name,
            256,
        ),
        "cea_device_probe",
    )
    return {
        "device": device,
        "name": name.value.decode(errors="replace"),
        "free_bytes": free_b.value,
        "total_bytes": total_b.value,
        "cu_count": cu.value,
        "wavefront_size": wf.value,
    }


def _tensor_ptr(t) -> int:
    assert t.is_cuda and t.dtype.is_floating_point
    return t.data_ptr()


def vector_add(a, b, c, inject_fault: bool = False, stream=None) -> None:
    """c = a + b on device float32 tensors.  inject_fault=True launches the
    deliberate out-of-bounds kernel (GPU page fault) used to exercise the
    health-check path — parity with the reference's
    demo/gpu-error/illegal-memory-access workload."""
    import torch

    lib = load_library()
    assert a.dtype == torch.float32 and a.is_contiguous()
    n = a.numel()
    sp = ctypes.c_void_p(
        torch.cuda.current_stream().cuda_stream if stream is None else stream
    )
    _check(
        lib,
        lib.cea_vector_add(
            ctypes.c_void_p(a.data_ptr()),
            ctypes.c_void_p(b.data_ptr()),
            ctypes.c_void_p(c.data_ptr()),
            ctypes.c_long(n),
            ctypes.c_int(1 if inject_fault else 0),
            sp,
        ),
        "cea_vector_add",
    )


def copy_(dst, src, stream=None) -> None:
    """Async d2d copy dst <- src (contiguous CUDA tensors, same nbytes,
    multiple of 16 bytes) with the tuned gfx950 kernel (exact-cover 16-B
    vector lanes; nontemporal past the 256 MiB Infinity Cache)."""
    import torch

    lib = load_library()
    assert dst.is_cuda and src.is_cuda
    assert dst.is_contiguous() and src.is_contiguous()
    nbytes = src.numel() * src.element_size()
    assert dst.numel() * dst.element_size() == nbytes
    sp = ctypes.c_void_p(
        torch.cuda.current_stream().cuda_stream if stream is None else stream
    )
    _check(
        lib,
        lib.cea_copy(
            ctypes.c_void_p(dst.data_ptr()),
            ctypes.c_void_p(src.data_ptr()),
            ctypes.c_long(nbytes),
            sp,
        ),
        "cea_copy",
    )


def copy_bandwidth_gbps(bytes_n: int = 1 << 30, iters: int = 20) -> float:
    """Device d2d copy bandwidth in GB/s (read+write)."""
    import torch

    lib = load_library()
    src = torch.empty(bytes_n // 4, dtype=torch.float32, device="cuda")
    dst = torch.empty_like(src)
    src.uniform_()
    out = ctypes.c_double()
    _check(
        lib,
        lib.cea_copy_bw(
            ctypes.c_void_p(dst.data_ptr()),
            ctypes.c_void_p(src.data_ptr()),
            ctypes.c_long(bytes_n),
            ctypes.c_int(iters),
            ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
            ctypes.byref(out),
        ),
        "cea_copy_bw",
    )
    return out.value


def reduce_sum(t) -> float:
    """Sum-reduce a float32 CUDA tensor with the wave64 HIP kernel."""
    import torch

    lib = load_library()
    assert t.dtype == torch.float32 and t.is_contiguous() and t.is_cuda
    out = ctypes.c_float()
    _check(
        lib,
        lib.cea_reduce_f32(
            ctypes.c_void_p(t.data_ptr()),
            ctypes.c_long(t.numel()),
            ctypes.c_void_p(torch.cuda.current_stream().cuda_stream),
            ctypes.byref(out),
        ),
        "cea_reduce_f32",
    )
    return out.value
