// probe_sched: price the pwg4x64-style MFMA stream on gfx950 silicon.
// Build: hipcc --offload-arch=gfx950 -O3 -o /tmp/probe_sched tools/probe_sched.hip
// Run (GPU box): /tmp/probe_sched   -> cyc/MFMA for each variant.
//
// M0: bare 32-MFMA phase, 4 rotating f32x16 accumulators (floor).
// M1: same + exactly 5 hand-placed independent v_fma_f32 fillers per gap
//     (the guide's budget: one wave/SIMD hides <= 5 single-issue
//     instructions per v_mfma_f32_32x32x16_bf16).
// M2: the same multiset (32 builtin MFMAs + 160 fma) in C, compiler-
//     scheduled — measures what hipcc does with the freedom.
// Launch: ONE block of 4 waves (one per SIMD of one CU), occupancy 1.
#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

template <int MODE>
__global__ __launch_bounds__(256, 1) void probe_kernel(
    float* out, int iters) {
  f32x16 c0 = {}, c1 = {}, c2 = {}, c3 = {};
  __shared__ __attribute__((aligned(16))) unsigned short lds[4096];
  for (int i = threadIdx.x; i < 4096; i += 256) lds[i] = (unsigned short)i;
  typedef __attribute__((ext_vector_type(2))) unsigned u32x2;
  u32x2 trd = {0u, 0u};
  const unsigned lds_addr =
      (unsigned)(uintptr_t)(__attribute__((address_space(3))) unsigned short*)
          &lds[(threadIdx.x & 63) * 4];
  bf16x8 a, b;
  float f0 = 1.0f, f1 = 1.0f;
  const float k = 1.0000001f;
#pragma unroll
  for (int i = 0; i < 8; ++i) { a[i] = (__bf16)(threadIdx.x + i); b[i] = (__bf16)(i); }
  asm volatile("s_waitcnt lgkmcnt(0) vmcnt(0)");
  __syncthreads();
  for (int it = 0; it < iters; ++it) {
    if (MODE == 0) {
      asm volatile("v_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3\n\tv_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3\n\tv_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3\n\tv_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3\n\tv_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3\n\tv_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3\n\tv_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3\n\tv_mfma_f32_32x32x16_bf16 %0, %4, %5, %0\n\tv_mfma_f32_32x32x16_bf16 %1, %4, %5, %1\n\tv_mfma_f32_32x32x16_bf16 %2, %4, %5, %2\n\tv_mfma_f32_32x32x16_bf16 %3, %4, %5, %3"
                   : "+v"(c0), "+v"(c1), "+v"(c2), "+v"(c3)
                   : "v"(a), "v"(b));
    } else if (MODE == 1) {
      asm volatile("v_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5\n\tv_fma_f32 %4, %4, %8, %4\n\tv_fma_f32 %5, %5, %8, %5"
                   : "+v"(c0), "+v"(c1), "+v"(c2), "+v"(c3),
                     "+v"(f0), "+v"(f1)
                   : "v"(a), "v"(b), "v"(k));
    } else if (MODE == 3) {
      asm volatile("v_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %0, %6, %7, %0\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %1, %6, %7, %1\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %2, %6, %7, %2\n\tv_exp_f32 %4, %4\n\tv_fma_f32 %4, %4, %8, %4\n\tv_max3_f32 %4, %4, %8, %8\n\tv_cvt_pk_bf16_f32 %4, %4, %8\n\tds_read_b64_tr_b16 %10, %9\n\tv_mfma_f32_32x32x16_bf16 %3, %6, %7, %3\n\tv_exp_f32 %5, %5\n\tv_fma_f32 %5, %5, %8, %5\n\tv_max3_f32 %5, %5, %8, %8\n\tv_cvt_pk_bf16_f32 %5, %5, %8\n\tds_read_b64_tr_b16 %10, %9\n\ts_waitcnt lgkmcnt(0)"
                   : "+v"(c0), "+v"(c1), "+v"(c2), "+v"(c3),
                     "+v"(f0), "+v"(f1)
                   : "v"(a), "v"(b), "v"(k), "v"(lds_addr), "v"(trd));
    } else {
#pragma unroll
      for (int i = 0; i < 32; ++i) {
        f32x16& c = (i % 4 == 0) ? c0 : (i % 4 == 1) ? c1 : (i % 4 == 2) ? c2 : c3;
        c = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
#pragma unroll
        for (int f = 0; f < 5; ++f) {
          float& d = ((i * 5 + f) % 2) ? f1 : f0;
          d = __builtin_fmaf(d, k, d);
        }
      }
    }
  }
  asm volatile("s_nop 11");
  // keep results alive
  out[threadIdx.x] = c0[0] + c1[0] + c2[0] + c3[0] + f0 + f1 + (float)trd[0];
}

template <int MODE>
double run(int iters) {
  float* out;
  (void)hipMalloc(&out, 256 * sizeof(float));
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  probe_kernel<MODE><<<1, 256>>>(out, iters);  // warm
  (void)hipDeviceSynchronize();
  (void)hipEventRecord(e0);
  probe_kernel<MODE><<<1, 256>>>(out, iters);
  (void)hipEventRecord(e1);
  (void)hipDeviceSynchronize();
  float ms = 0.f;
  (void)hipEventElapsedTime(&ms, e0, e1);
  (void)hipFree(out);
  // ns per MFMA; cyc assumes ~2.4 GHz shader clock
  return (double)ms * 1e6 / (iters * 32.0);
}

int main() {
  const int iters = 20000;
  const double ghz = 2.4;  // nominal; scale if rocm-smi shows otherwise
  double m0 = run<0>(iters), m1 = run<1>(iters), m2 = run<2>(iters);
  printf("M0 bare 32-MFMA phase:        %.2f ns/MFMA  (%.1f cyc @%.1fGHz)\n", m0, m0 * ghz, ghz);
  printf("M1 hand-placed 5 fillers/gap: %.2f ns/MFMA  (%.1f cyc @%.1fGHz)\n", m1, m1 * ghz, ghz);
  printf("M2 compiler-scheduled same:   %.2f ns/MFMA  (%.1f cyc @%.1fGHz)\n", m2, m2 * ghz, ghz);
  double m3 = run<3>(iters);
  printf("M3 real filler mix hand-placed (exp/fma/max3/cvt_pk/tr16): %.2f ns/MFMA  (%.1f cyc @%.1fGHz)\n", m3, m3 * ghz, ghz);
  return 0;
}
