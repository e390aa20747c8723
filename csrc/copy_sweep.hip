// Standalone d2d copy-kernel sweep for gfx950 — round-2 follow-up to the
// round-1 sweep (profiles/copy_tune_r01.csv) exploring per-thread unroll
// (memory-level parallelism) on top of the winning nontemporal variant.
// Prints CSV: variant,unroll,block,grid,bytes,ms,gbps_rw.
// Build: hipcc --offload-arch=gfx950 -O3 csrc/copy_sweep.hip -o copy_sweep
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <vector>

#define CK(x)                                                      \
  do {                                                             \
    hipError_t e = (x);                                            \
    if (e != hipSuccess) {                                         \
      fprintf(stderr, "HIP error %s at %s:%d\n",                   \
              hipGetErrorString(e), __FILE__, __LINE__);           \
      exit(1);                                                     \
    }                                                              \
  } while (0)

typedef float vf4 __attribute__((ext_vector_type(4)));

// one element per thread (round-1 winner, the baseline)
__global__ void copy_nt_u1(const vf4* __restrict__ src, vf4* __restrict__ dst,
                           long n4) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n4)
    __builtin_nontemporal_store(__builtin_nontemporal_load(&src[i]), &dst[i]);
}

// strided unroll: U coalesced loads issued before U stores per thread
template <int U>
__global__ void copy_nt_uN(const vf4* __restrict__ src, vf4* __restrict__ dst,
                           long n4) {
  long base = ((long)blockIdx.x * U) * blockDim.x + threadIdx.x;
  vf4 v[U];
#pragma unroll
  for (int u = 0; u < U; ++u) {
    long i = base + (long)u * blockDim.x;
    if (i < n4) v[u] = __builtin_nontemporal_load(&src[i]);
  }
#pragma unroll
  for (int u = 0; u < U; ++u) {
    long i = base + (long)u * blockDim.x;
    if (i < n4) __builtin_nontemporal_store(v[u], &dst[i]);
  }
}

// grid-stride nontemporal with unroll (bounded grid, persistent waves)
template <int U>
__global__ void copy_nt_gs(const vf4* __restrict__ src, vf4* __restrict__ dst,
                           long n4) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long base = (long)blockIdx.x * blockDim.x + threadIdx.x;
       base < n4; base += stride * U) {
    vf4 v[U];
#pragma unroll
    for (int u = 0; u < U; ++u) {
      long i = base + (long)u * stride;
      if (i < n4) v[u] = __builtin_nontemporal_load(&src[i]);
    }
#pragma unroll
    for (int u = 0; u < U; ++u) {
      long i = base + (long)u * stride;
      if (i < n4) __builtin_nontemporal_store(v[u], &dst[i]);
    }
  }
}

struct Variant {
  const char* name;
  int unroll;
  void (*kernel)(const vf4*, vf4*, long);
  bool grid_stride;
};

int main(int argc, char** argv) {
  long max_bytes = argc > 1 ? atol(argv[1]) : (512L << 20);
  int iters = argc > 2 ? atoi(argv[2]) : 20;
  std::vector<long> sizes = {256L << 20, max_bytes};
  vf4 *src, *dst;
  CK(hipMalloc(&src, max_bytes));
  CK(hipMalloc(&dst, max_bytes));
  CK(hipMemset(src, 1, max_bytes));

  Variant variants[] = {
      {"exact_nt", 1, copy_nt_u1, false},
      {"exact_nt", 2, copy_nt_uN<2>, false},
      {"exact_nt", 4, copy_nt_uN<4>, false},
      {"exact_nt", 8, copy_nt_uN<8>, false},
      {"gs_nt", 1, copy_nt_gs<1>, true},
      {"gs_nt", 2, copy_nt_gs<2>, true},
      {"gs_nt", 4, copy_nt_gs<4>, true},
  };
  int blocks[] = {256, 512, 1024};
  int gs_grids[] = {16384, 32768, 65536};

  hipEvent_t t0, t1;
  CK(hipEventCreate(&t0));
  CK(hipEventCreate(&t1));
  printf("variant,unroll,block,grid,bytes,ms,gbps_rw\n");
  for (auto& v : variants) {
    for (int block : blocks) {
          for (long bytes : sizes) {
        long n4 = bytes / 16;
        if (v.grid_stride) {
          for (int g : gs_grids) {
            // warmup + timed
            for (int w = 0; w < 3; ++w)
              hipLaunchKernelGGL(v.kernel, dim3(g), dim3(block), 0, 0, src,
                                 dst, n4);
            CK(hipDeviceSynchronize());
            CK(hipEventRecord(t0));
            for (int it = 0; it < iters; ++it)
              hipLaunchKernelGGL(v.kernel, dim3(g), dim3(block), 0, 0, src,
                                 dst, n4);
            CK(hipEventRecord(t1));
            CK(hipEventSynchronize(t1));
            float ms = 0;
            CK(hipEventElapsedTime(&ms, t0, t1));
            ms /= iters;
            printf("%s,%d,%d,%d,%ld,%.4f,%.1f\n", v.name, v.unroll, block, g,
                   bytes, ms, 2.0 * bytes / (ms * 1e6));
          }
        } else {
          long g = (n4 + (long)block * v.unroll - 1) / ((long)block * v.unroll);
          for (int w = 0; w < 3; ++w)
            hipLaunchKernelGGL(v.kernel, dim3((unsigned)g), dim3(block), 0, 0,
                               src, dst, n4);
          CK(hipDeviceSynchronize());
          CK(hipEventRecord(t0));
          for (int it = 0; it < iters; ++it)
            hipLaunchKernelGGL(v.kernel, dim3((unsigned)g), dim3(block), 0, 0,
                               src, dst, n4);
          CK(hipEventRecord(t1));
          CK(hipEventSynchronize(t1));
          float ms = 0;
          CK(hipEventElapsedTime(&ms, t0, t1));
          ms /= iters;
          printf("%s,%d,%d,%ld,%ld,%.4f,%.1f\n", v.name, v.unroll, block, g,
                 bytes, ms, 2.0 * bytes / (ms * 1e6));
        }
      }
    }
  }
  // correctness spot-check on the largest size with the unrolled winner
  CK(hipMemset(dst, 0, max_bytes));
  hipLaunchKernelGGL(copy_nt_uN<4>,
                     dim3((unsigned)((max_bytes / 16 + 1023) / 1024)),
                     dim3(256), 0, 0, src, dst, max_bytes / 16);
  CK(hipDeviceSynchronize());
  std::vector<unsigned char> a(4096), b(4096);
  CK(hipMemcpy(a.data(), src, 4096, hipMemcpyDeviceToHost));
  CK(hipMemcpy(b.data(), (char*)dst, 4096, hipMemcpyDeviceToHost));
  CK(hipMemcpy(a.data(), (char*)src + max_bytes - 4096, 4096,
               hipMemcpyDeviceToHost));
  CK(hipMemcpy(b.data(), (char*)dst + max_bytes - 4096, 4096,
               hipMemcpyDeviceToHost));
  if (a != b) {
    fprintf(stderr, "MISMATCH in copy_nt_uN<4>\n");
    return 1;
  }
  fprintf(stderr, "correctness ok\n");
  return 0;
}
