// Standalone GPU-ASan probe (SURVEY.md §5.2 sanitizer tier).
// Build: hipcc --offload-arch=gfx950:xnack+ -fsanitize=address -g -O1
//        tools/asan/asan_probe.hip -o gpurun_out/asan_probe
// Run:   HSA_XNACK=1 ./asan_probe        -> CLEAN (exit 0)
//        HSA_XNACK=1 ./asan_probe oob    -> ASan heap-buffer-overflow report
// A standalone binary links the ASan runtime itself (no LD_PRELOAD), so
// this exercises DEVICE address sanitizing end to end — the tier's
// "does it actually catch a GPU OOB on this pool" proof.
#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstring>

__global__ void saxpy_kernel(float* out, const float* in, int n, int stride) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = in[i * stride] * 2.0f + 1.0f;  // stride>1 walks OOB
}

#define CK(x)                                                       \
  do {                                                              \
    hipError_t e = (x);                                             \
    if (e != hipSuccess) {                                          \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); \
      return 2;                                                     \
    }                                                               \
  } while (0)

int main(int argc, char** argv) {
  const bool oob = argc > 1 && strcmp(argv[1], "oob") == 0;
  const int n = 4096;
  float *in, *out;
  CK(hipMalloc(&in, n * sizeof(float)));
  CK(hipMalloc(&out, n * sizeof(float)));
  CK(hipMemset(in, 0, n * sizeof(float)));
  saxpy_kernel<<<(n + 255) / 256, 256>>>(out, in, n, oob ? 2 : 1);
  CK(hipDeviceSynchronize());
  float host[4];
  CK(hipMemcpy(host, out, sizeof(host), hipMemcpyDeviceToHost));
  printf("ASAN_PROBE_%s out[0]=%f\n", oob ? "OOB_SURVIVED" : "CLEAN",
         host[0]);
  return 0;
}
