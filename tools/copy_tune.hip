// Copy-kernel tuner for MI355X (gfx950): sweeps implementation variants and
// launch geometries for the d2d streaming copy that backs bench.py's
// out-of-place all_reduce step.  Standalone; prints CSV to stdout.
//
//   hipcc --offload-arch=gfx950 -O3 tools/copy_tune.hip -o gpurun_out/copy_tune
//   ./gpurun_out/copy_tune [bytes...]
#include <hip/hip_runtime.h>

// Native 16-B vector type (HIP vf4 is a class; the nontemporal builtins
// need a real vector type).
typedef float vf4 __attribute__((ext_vector_type(4)));

#include <cstdio>
#include <cstdlib>
#include <vector>

#define CHECK(x)                                                      \
  do {                                                                \
    hipError_t e = (x);                                               \
    if (e != hipSuccess) {                                            \
      fprintf(stderr, "%s: %s\n", #x, hipGetErrorString(e));          \
      exit(1);                                                        \
    }                                                                 \
  } while (0)

// v0: grid-stride vf4 (current production kernel).
__global__ void k_stride(const vf4* __restrict__ src,
                         vf4* __restrict__ dst, long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n4; i += stride) dst[i] = src[i];
}

// v1: grid-stride vf4 with nontemporal load+store (bypass L2/L3 — pure
// streaming traffic should not displace cache lines).
__global__ void k_stride_nt(const vf4* __restrict__ src,
                            vf4* __restrict__ dst, long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n4; i += stride)
    __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// v2: grid-stride, 2x unrolled nt (two independent loads in flight/lane).
__global__ void k_stride_nt_u2(const vf4* __restrict__ src,
                               vf4* __restrict__ dst, long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + stride < n4; i += 2 * stride) {
    vf4 a = __builtin_nontemporal_load(&src[i]);
    vf4 b = __builtin_nontemporal_load(&src[i + stride]);
    __builtin_nontemporal_store(a, &dst[i]);
    __builtin_nontemporal_store(b, &dst[i + stride]);
  }
  if (i < n4)
    __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// v3: exact-cover, one vf4 per thread, no loop (launch as many WGs as
// needed; hardware dispatch is cheap and addressing is trivial).
__global__ void k_exact(const vf4* __restrict__ src,
                        vf4* __restrict__ dst, long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n4) dst[i] = src[i];
}

// v4: exact-cover nt.
__global__ void k_exact_nt(const vf4* __restrict__ src,
                           vf4* __restrict__ dst, long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n4)
    __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// v5: blocked — each workgroup owns a contiguous span of BLK*U vf4s,
// threads make U coalesced nt accesses 256 apart inside the span.  Keeps
// each XCD's traffic contiguous (better DRAM page locality).
template <int U>
__global__ void k_block_nt(const vf4* __restrict__ src,
                           vf4* __restrict__ dst, long n4) {
  long base = (long)blockIdx.x * blockDim.x * U + threadIdx.x;
  long lim = n4;
#pragma unroll
  for (int u = 0; u < U; u++) {
    long i = base + (long)u * blockDim.x;
    if (i < lim)
      __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]),
                                  &dst[i]);
  }
}

struct Variant {
  const char* name;
  void (*launch)(const vf4*, vf4*, long, int grid, int block,
                 hipStream_t);
  bool exact;  // grid derived from size
  int unroll;
};

template <void (*K)(const vf4*, vf4*, long)>
void launch_plain(const vf4* s, vf4* d, long n4, int grid, int block,
                  hipStream_t st) {
  hipLaunchKernelGGL(K, dim3(grid), dim3(block), 0, st, s, d, n4);
}

int main(int argc, char** argv) {
  std::vector<long> sizes;
  for (int i = 1; i < argc; i++) sizes.push_back(atol(argv[i]));
  if (sizes.empty()) sizes = {64L << 20, 256L << 20, 512L << 20};

  long max_b = 0;
  for (long b : sizes)
    if (b > max_b) max_b = b;
  vf4 *src, *dst;
  CHECK(hipMalloc(&src, max_b));
  CHECK(hipMalloc(&dst, max_b));
  CHECK(hipMemset(src, 1, max_b));
  hipStream_t st;
  CHECK(hipStreamCreate(&st));
  hipEvent_t t0, t1;
  CHECK(hipEventCreate(&t0));
  CHECK(hipEventCreate(&t1));

  const Variant variants[] = {
      {"stride", launch_plain<k_stride>, false, 1},
      {"stride_nt", launch_plain<k_stride_nt>, false, 1},
      {"stride_nt_u2", launch_plain<k_stride_nt_u2>, false, 1},
      {"exact", launch_plain<k_exact>, true, 1},
      {"exact_nt", launch_plain<k_exact_nt>, true, 1},
      {"block_nt_u4", launch_plain<k_block_nt<4>>, true, 4},
      {"block_nt_u8", launch_plain<k_block_nt<8>>, true, 8},
  };
  const int grids[] = {2048, 4096, 8192, 16384};
  const int blocks[] = {256, 512, 1024};
  const int iters = 20;

  printf("variant,grid,block,bytes,ms,gbps_rw\n");
  for (const auto& v : variants) {
    for (int block : blocks) {
      for (int grid : grids) {
        for (long bytes : sizes) {
          long n4 = bytes / 16;
          int g = grid;
          if (v.exact) {
            long need = (n4 + (long)block * v.unroll - 1) /
                        ((long)block * v.unroll);
            g = (int)need;
            if (grid != grids[0]) continue;  // geometry fixed; run once
          }
          v.launch(src, dst, n4, g, block, st);  // warmup
          CHECK(hipGetLastError());
          CHECK(hipEventRecord(t0, st));
          for (int i = 0; i < iters; i++) v.launch(src, dst, n4, g, block, st);
          CHECK(hipEventRecord(t1, st));
          CHECK(hipEventSynchronize(t1));
          float ms = 0;
          CHECK(hipEventElapsedTime(&ms, t0, t1));
          double gbps = 2.0 * bytes * iters / (ms * 1e6);
          printf("%s,%d,%d,%ld,%.4f,%.1f\n", v.name, g, block, bytes,
                 ms / iters, gbps);
          fflush(stdout);
        }
      }
    }
  }
  return 0;
}
