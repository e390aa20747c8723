// Standalone HIP probe: per-device VRAM and CU count, as seen from inside a
// container.  Used to verify GPU-sharing fencing on MI355X the same way the
// reference verifies MPS fencing with
// /root/reference/example/cuda-mps/cuda_mem_and_sm_count.c:38-55
// (cudaMemGetInfo + cudaDeviceGetAttribute(MultiProcessorCount)).
//
// MI355X semantics this program makes visible:
//   - SPX mode: one device, 256 CUs, ~288 GB VRAM.
//   - CPX mode: up to 8 devices per physical GPU, 32 CUs (one XCD) and
//     1/8 of the HBM each — the partition IS the fence.
//   - HSA_CU_MASK / time-sharing: device count and VRAM unchanged; only
//     occupancy differs (ROCm does not enforce a memory limit, which is why
//     the framework maps the reference's "mps" strategy onto CPX — see
//     docs/gpu-sharing-and-partitioning.md).
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>

#define HIP_CHECK(expr)                                                  \
  do {                                                                   \
    hipError_t err_ = (expr);                                            \
    if (err_ != hipSuccess) {                                            \
      std::fprintf(stderr, "%s failed: %s\n", #expr,                     \
                   hipGetErrorString(err_));                             \
      std::exit(1);                                                      \
    }                                                                    \
  } while (0)

int main() {
  int ndev = 0;
  HIP_CHECK(hipGetDeviceCount(&ndev));
  std::printf("visible devices: %d\n", ndev);
  for (int i = 0; i < ndev; ++i) {
    HIP_CHECK(hipSetDevice(i));
    size_t free_b = 0, total_b = 0;
    HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
    int cus = 0, wave = 0;
    HIP_CHECK(hipDeviceGetAttribute(
        &cus, hipDeviceAttributeMultiprocessorCount, i));
    HIP_CHECK(hipDeviceGetAttribute(&wave, hipDeviceAttributeWarpSize, i));
    hipDeviceProp_t prop;
    HIP_CHECK(hipGetDeviceProperties(&prop, i));
    std::printf(
        "device %d: %s gcn=%s CUs=%d wave=%d vram_total=%.1f GiB "
        "vram_free=%.1f GiB\n",
        i, prop.name, prop.gcnArchName, cus, wave,
        double(total_b) / (1 << 30), double(free_b) / (1 << 30));
  }
  return 0;
}
