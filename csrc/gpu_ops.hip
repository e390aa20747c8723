// cea_amd HIP kernels for MI355X (gfx950, CDNA4).  Native HIP, no ports.
//
// Components and their reference parity:
//  * cea_vector_add(..., inject_fault=1): the GPU-fault injection workload
//    used to validate the health-check path on hardware — parity with
//    /root/reference/demo/gpu-error/illegal-memory-access/vectorAdd.cu:28-34
//    (deliberate out-of-bounds index to raise a GPU page fault / RAS event).
//    The healthy path is a proper CDNA4 kernel: float4 (16 B/lane) vector
//    loads, grid-stride, sized to fill 256 CUs.
//  * cea_device_probe: VRAM + CU-count query — parity with
//    /root/reference/example/cuda-mps/cuda_mem_and_sm_count.c:38-55 (used to
//    verify CU-mask fencing: a cu-fenced container sees fewer CUs).
//  * cea_copy_bw / cea_reduce_f32: measured-bandwidth + wave64 reduction
//    helpers used by tests and the bench harness.
//
// Build: hipcc --offload-arch=gfx950 -O3 (Makefile target `gpu`).

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstring>

namespace {

thread_local char g_err[512];

int fail(const char* what, hipError_t e) {
  snprintf(g_err, sizeof(g_err), "%s: %s", what, hipGetErrorString(e));
  return (int)e;
}

#define CHECK(expr)                                    \
  do {                                                 \
    hipError_t _e = (expr);                            \
    if (_e != hipSuccess) return fail(#expr, _e);      \
  } while (0)

// Healthy path: vectorized float4 grid-stride add (16 B/lane loads — scalar
// f32 loads leave ~2x bandwidth on the table on CDNA4).
__global__ void vector_add_vec4(const float4* __restrict__ a,
                                const float4* __restrict__ b,
                                float4* __restrict__ c, long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    float4 va = a[i], vb = b[i];
    c[i] = make_float4(va.x + vb.x, va.y + vb.y, va.z + vb.z, va.w + vb.w);
  }
}

// Fault path: same purpose as the reference demo kernel (vectorAdd.cu:28-34,
// +1e12 index) — deliberately write out of bounds so the KFD raises a VM
// fault the health checker must observe.  The offset is +1 GiB, NOT +1e12
// elements: a multi-TiB offset lands outside the GPU VA aperture on some
// ROCm builds and is classified HSA_STATUS_ERROR_MEMORY_APERTURE_VIOLATION
// instead of a page fault, which does NOT emit the amdsmi VMFAULT event.
// +1 GiB stays inside the aperture but beyond any backing -> true VM page
// fault (AMDSMI_EVT_NOTIF_VMFAULT / PAGE_FAULT_START) on every build.
__global__ void vector_add_oob(const float* a, const float* b, float* c,
                               long n) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    c[i + (1L << 28)] = a[i] + b[i];  // +2^28 floats = +1 GiB
  }
}

// Tail for n not divisible by 4.
__global__ void vector_add_tail(const float* a, const float* b, float* c,
                                long start, long n) {
  long i = start + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) c[i] = a[i] + b[i];
}

// Native 16-B vector type: HIP's float4 is a class, and the nontemporal
// builtins only take real vector types.
typedef float vf4 __attribute__((ext_vector_type(4)));

// Exact-cover copy: one 16-B vector per lane, no loop.  Tuned on MI355X
// (profiles/copy_tune_r01.csv): beats every grid-stride geometry.
__global__ void copy_exact(const vf4* __restrict__ src,
                           vf4* __restrict__ dst, long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n4) dst[i] = src[i];
}

// Nontemporal variant for streaming sizes (working set past the 256 MiB
// Infinity Cache): sc-bit loads/stores keep the one-shot traffic from
// displacing L2/L3 lines — 6.56 TB/s r+w at 512 MiB vs 4.9-5.5 for the
// grid-stride cached copy.
__global__ void copy_exact_nt(const vf4* __restrict__ src,
                              vf4* __restrict__ dst, long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n4)
    __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// Grid-stride nontemporal for giant buffers: the AMD dispatch packet's
// global work size per dimension is 32-bit, so an exact-cover launch is
// limited to 2^32-1 work items (= just under 64 GiB of float4 lanes).
// Past that the runtime either rejects the launch (exactly 2^32: invalid
// configuration) or — worse — the 32-bit size check wraps and the kernel
// SILENTLY covers only n4 mod 2^32 elements (observed at 120 GiB on
// ROCm 7.0; caught by tests/test_gpu_integration.py::
// test_large_vram_copy_288gb_sizing).  Same measured bandwidth as the
// exact cover (stride_nt 6564 vs exact_nt 6564 GB/s r+w,
// profiles/copy_tune_r01.csv).
__global__ void copy_stride_nt(const vf4* __restrict__ src,
                               vf4* __restrict__ dst, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride)
    __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// Wave64 shuffle reduction -> per-block LDS reduction -> one atomic/block.
__global__ void reduce_sum_f32(const float4* __restrict__ in, long n4,
                               const float* __restrict__ tail, long ntail,
                               float* out) {
  float acc = 0.f;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) {
    float4 v = in[i];
    acc += v.x + v.y + v.z + v.w;
  }
  if (blockIdx.x == 0 && threadIdx.x < ntail) acc += tail[threadIdx.x];
  // wave64 shuffle reduction
  for (int off = warpSize / 2; off > 0; off >>= 1)
    acc += __shfl_down(acc, off, warpSize);
  __shared__ float wave_sums[16];  // up to 1024 threads = 16 waves
  int wave = threadIdx.x / warpSize;
  int lane = threadIdx.x % warpSize;
  if (lane == 0) wave_sums[wave] = acc;
  __syncthreads();
  int nwaves = (blockDim.x + warpSize - 1) / warpSize;
  if (wave == 0) {
    float s = (lane < nwaves) ? wave_sums[lane] : 0.f;
    for (int off = warpSize / 2; off > 0; off >>= 1)
      s += __shfl_down(s, off, warpSize);
    if (lane == 0) atomicAdd(out, s);
  }
}

dim3 grid_for(long work_items, int block) {
  // >=2048 workgroups so the 256-CU / 8-XCD chip is filled (a 256-WG launch
  // is one workgroup per CU — the minimum, not a target).
  long blocks = (work_items + block - 1) / block;
  if (blocks > 8192) blocks = 8192;
  if (blocks < 1) blocks = 1;
  return dim3((unsigned)blocks);
}

}  // namespace

extern "C" {

const char* cea_gpu_last_error() { return g_err; }

int cea_gpu_device_count(int* n) {
  CHECK(hipGetDeviceCount(n));
  return 0;
}

// Parity: cuda_mem_and_sm_count.c — mem info + multiprocessor count.
int cea_device_probe(int device, unsigned long long* free_b,
                     unsigned long long* total_b, int* cu_count,
                     int* wavefront_size, char* name, int name_len) {
  CHECK(hipSetDevice(device));
  size_t f = 0, t = 0;
  CHECK(hipMemGetInfo(&f, &t));
  hipDeviceProp_t prop;
  CHECK(hipGetDeviceProperties(&prop, device));
  *free_b = f;
  *total_b = t;
  *cu_count = prop.multiProcessorCount;
  *wavefront_size = prop.warpSize;
  snprintf(name, name_len, "%s", prop.name);
  return 0;
}

// a,b,c are device float pointers of length n.  inject_fault=1 launches the
// deliberate OOB kernel (GPU page fault) instead of the healthy one.
int cea_vector_add(const float* a, const float* b, float* c, long n,
                   int inject_fault, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  const int block = 256;
  if (inject_fault) {
    long blocks = (n + block - 1) / block;
    hipLaunchKernelGGL(vector_add_oob, dim3((unsigned)blocks), dim3(block), 0,
                       s, a, b, c, n);
    CHECK(hipGetLastError());
    return 0;
  }
  long n4 = n / 4;
  if (n4 > 0) {
    hipLaunchKernelGGL(vector_add_vec4, grid_for(n4, block), dim3(block), 0, s,
                       (const float4*)a, (const float4*)b, (float4*)c, n4);
    CHECK(hipGetLastError());
  }
  if (n % 4) {
    hipLaunchKernelGGL(vector_add_tail, dim3(1), dim3(block), 0, s, a, b, c,
                       n4 * 4, n);
    CHECK(hipGetLastError());
  }
  return 0;
}

namespace {

// Tuned dispatch (profiles/copy_tune_r01.csv, MI355X): while src+dst still
// fit the 256 MiB Infinity Cache a plain cached copy wins (~6.9 TB/s r+w at
// 64 MiB); past it the nontemporal exact-cover kernel wins (~6.56 TB/s at
// 512 MiB vs ~4.9 for grid-stride cached).
void launch_copy(const void* src, void* dst, long n4, hipStream_t s) {
  long bytes = n4 * 16;
  if (2 * bytes <= (256L << 20)) {
    const int block = 512;
    long g = (n4 + block - 1) / block;
    hipLaunchKernelGGL(copy_exact, dim3((unsigned)(g ? g : 1)), dim3(block), 0,
                       s, (const vf4*)src, (vf4*)dst, n4);
  } else {
    const int block = 256;
    long g = (n4 + block - 1) / block;
    if (g * (long)block > 0xFFFFFFFFL) {
      // 32-bit dispatch global-size limit: bounded grid-stride cover.
      // Config-insensitive at these sizes: every grid/block/unroll combo
      // lands at ~6.28 TB/s r+w @65 GiB (profiles/stride_tune_65g.csv —
      // TLB/page-walk bound, ~5% below the 256 MiB streaming rate).
      hipLaunchKernelGGL(copy_stride_nt, dim3(32768), dim3(256), 0, s,
                         (const vf4*)src, (vf4*)dst, n4);
    } else {
      hipLaunchKernelGGL(copy_exact_nt, dim3((unsigned)g), dim3(block), 0, s,
                         (const vf4*)src, (vf4*)dst, n4);
    }
  }
}

}  // namespace

// Plain async d2d copy of `bytes` bytes (multiple of 16) on `stream` —
// used by bench.py's out-of-place all_reduce step.  Measured 6.9 TB/s r+w
// L3-resident / 6.56 TB/s streaming on MI355X vs ~5.1 for torch copy_.
int cea_copy(void* dst, const void* src, long bytes, void* stream) {
  if (bytes % 16) {
    snprintf(g_err, sizeof(g_err), "bytes must be a multiple of 16");
    return -2;
  }
  launch_copy(src, dst, bytes / 16, (hipStream_t)stream);
  CHECK(hipGetLastError());
  return 0;
}

// Device-to-device copy bandwidth: `iters` timed copies of `bytes` bytes
// (bytes must be a multiple of 16).  Returns achieved GB/s counting
// read+write traffic.
int cea_copy_bw(void* dst, const void* src, long bytes, int iters,
                void* stream, double* gbps) {
  if (bytes % 16) {
    snprintf(g_err, sizeof(g_err), "bytes must be a multiple of 16");
    return -2;
  }
  hipStream_t s = (hipStream_t)stream;
  long n4 = bytes / 16;
  hipEvent_t t0, t1;
  CHECK(hipEventCreate(&t0));
  CHECK(hipEventCreate(&t1));
  // warmup
  launch_copy(src, dst, n4, s);
  CHECK(hipGetLastError());
  CHECK(hipEventRecord(t0, s));
  for (int i = 0; i < iters; i++) launch_copy(src, dst, n4, s);
  CHECK(hipEventRecord(t1, s));
  CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  CHECK(hipEventElapsedTime(&ms, t0, t1));
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  *gbps = (2.0 * bytes * iters) / (ms * 1e6);
  return 0;
}

// Sum-reduce n device floats into *out_sum (host).  Used by GPU numerics
// tests against a plain PyTorch fp32 reference.
int cea_reduce_f32(const float* in, long n, void* stream, float* out_sum) {
  hipStream_t s = (hipStream_t)stream;
  float* d_out = nullptr;
  CHECK(hipMalloc(&d_out, sizeof(float)));
  CHECK(hipMemsetAsync(d_out, 0, sizeof(float), s));
  long n4 = n / 4;
  long ntail = n % 4;
  const int block = 1024;
  hipLaunchKernelGGL(reduce_sum_f32, grid_for(n4 ? n4 : 1, block), dim3(block),
                     0, s, (const float4*)in, n4, in + n4 * 4, ntail, d_out);
  CHECK(hipGetLastError());
  CHECK(hipMemcpyAsync(out_sum, d_out, sizeof(float), hipMemcpyDeviceToHost, s));
  CHECK(hipStreamSynchronize(s));
  CHECK(hipFree(d_out));
  return 0;
}

int cea_device_synchronize() {
  CHECK(hipDeviceSynchronize());
  return 0;
}

}  // extern "C"
