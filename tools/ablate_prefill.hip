// Phase ablation for the prefill2 structure (ANALYSIS TOOL — a trimmed
// bf16 copy of fa_prefill2_kernel with template<int MODE> phase stubs).
// Guide §5.4 rules 17/24: stubs keep upstream values alive with empty asm
// (DCE otherwise deletes the producing phase too); all variants co-compiled
// and interleaved in ONE process so deltas are within-probe.
//
// MODE bits: 1 = stub softmax+pack, 2 = stub PV (tr reads + MFMAs),
//            4 = stub QK^T (K reads + MFMAs). 7 = staging skeleton only.
//
// Build:  hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/ablate_prefill.hip -o /tmp/ablate
// Run:    /tmp/ablate  (prints per-mode ms over interleaved rounds)

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;
#define LDS_SPACE __attribute__((address_space(3)))
#define GLOBAL_SPACE __attribute__((address_space(1)))

constexpr int kD = 128;
constexpr int kThreads = 512;
constexpr int kQBlk = 256;
constexpr int kKvBlk = 128;
constexpr int kTileBytes = kKvBlk * 256;
constexpr float kMinInit = -3.0e38f;

template <typename T>
__device__ __forceinline__ void keep(T& x) {
  asm volatile("" : "+v"(x));
}

template <int MODE>
__global__ __launch_bounds__(kThreads, 2) void ablate_kernel(
    const __bf16* __restrict__ q, const __bf16* __restrict__ k,
    const __bf16* __restrict__ v, float* __restrict__ o_out, const int Hq,
    const long Tq, const long Tkv, const float scale) {
  __shared__ __attribute__((aligned(16))) char smem[4 * kTileBytes];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row32 = lane & 31, h = lane >> 5, m16 = lane & 15, g16 = lane >> 4;
  const int nQB = (int)((Tq + kQBlk - 1) / kQBlk);
  const int hq = blockIdx.x % Hq;
  const int qb_i = blockIdx.x / Hq;
  const long row0 = (long)qb_i * kQBlk + wave * 32;
  const __bf16* k_head = k + (size_t)hq * Tkv * kD;
  const __bf16* v_head = v + (size_t)hq * Tkv * kD;
  const __bf16* q_head = q + (size_t)hq * Tq * kD;
  (void)nQB;

  bf16x8 q_frag[8];
#pragma unroll
  for (int s = 0; s < 8; ++s) {
    q_frag[s] = *reinterpret_cast<const bf16x8*>(
        q_head + (size_t)(row0 + row32) * kD + s * 16 + h * 8);
  }
  f32x16 o_acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) o_acc[i] = f32x16{};
  float m_run = kMinInit, l_run = 0.f;

  unsigned k_soff[4], v_soff[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int seg = c * 8 + wave, g = seg * 64 + lane;
    const int row = g >> 4, col = g & 15;
    k_soff[c] = (unsigned)(row * 256 + ((col ^ (row & 15)) * 16));
    v_soff[c] = (unsigned)(row * 256 + ((col ^ ((row & 7) << 1)) * 16));
  }
  auto stage = [&](long t, int buf) {
    char* k_buf = smem + buf * kTileBytes;
    char* v_buf = smem + (2 + buf) * kTileBytes;
    const char* kg = reinterpret_cast<const char*>(k_head) + (size_t)t * kKvBlk * 256;
    const char* vg = reinterpret_cast<const char*>(v_head) + (size_t)t * kKvBlk * 256;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int seg = c * 8 + wave;
      __builtin_amdgcn_global_load_lds((const GLOBAL_SPACE unsigned int*)(kg + k_soff[c]),
                                       (LDS_SPACE unsigned int*)(k_buf + seg * 1024), 16, 0, 2);
      __builtin_amdgcn_global_load_lds((const GLOBAL_SPACE unsigned int*)(vg + v_soff[c]),
                                       (LDS_SPACE unsigned int*)(v_buf + seg * 1024), 16, 0, 2);
    }
  };

  const long n_tiles = Tkv / kKvBlk;
  stage(0, 0);
  for (long t = 0; t < n_tiles; ++t) {
    const int buf = (int)(t & 1);
    const char* k_buf = smem + buf * kTileBytes;
    const char* v_buf = smem + (2 + buf) * kTileBytes;
    if (t + 1 < n_tiles) {
      stage(t + 1, buf ^ 1);
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

#pragma unroll
    for (int nb = 0; nb < 4; ++nb) {
      f32x16 s_acc = {};
      if constexpr (!(MODE & 4)) {  // QK^T
#pragma unroll
        for (int s = 0; s < 8; ++s) {
          const int key = nb * 32 + row32;
          const bf16x8 k_frag = *reinterpret_cast<const bf16x8*>(
              k_buf + key * 256 + ((s * 32 + h * 16) ^ ((key & 15) << 4)));
          s_acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(k_frag, q_frag[s], s_acc, 0, 0, 0);
        }
      } else {
        // synthetic scores, kept cheap but data-dependent-ish
        s_acc[0] = (float)(lane + nb);
        keep(s_acc);
      }

      unsigned c32[8];
      float alpha = 1.f;
      if constexpr (!(MODE & 1)) {  // softmax + pack
#pragma unroll
        for (int reg = 0; reg < 16; ++reg) s_acc[reg] *= scale;
        float mt;
        if constexpr (MODE & 8) {  // tree reductions variant
          float mx[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) mx[i] = fmaxf(s_acc[2 * i], s_acc[2 * i + 1]);
#pragma unroll
          for (int st = 4; st >= 1; st >>= 1)
#pragma unroll
            for (int i = 0; i < 8; ++i)
              if (i < st) mx[i] = fmaxf(mx[i], mx[i + st]);
          mt = mx[0];
        } else {
          mt = s_acc[0];
#pragma unroll
          for (int reg = 1; reg < 16; ++reg) mt = fmaxf(mt, s_acc[reg]);
        }
        mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
        const float m_new = fmaxf(m_run, mt);
        alpha = __expf(m_run - m_new);
        m_run = m_new;
        float rs = 0.f;
        float rsp[8];
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const float p0 = __expf(s_acc[2 * i] - m_new);
          const float p1 = __expf(s_acc[2 * i + 1] - m_new);
          if constexpr (MODE & 8) rsp[i] = p0 + p1; else rs += p0 + p1;
          const __bf16 lo8 = (__bf16)p0, hi8 = (__bf16)p1;
          c32[i] = (unsigned)__builtin_bit_cast(unsigned short, lo8) |
                   ((unsigned)__builtin_bit_cast(unsigned short, hi8) << 16);
        }
        if constexpr (MODE & 8) {
#pragma unroll
          for (int st = 4; st >= 1; st >>= 1)
#pragma unroll
            for (int i = 0; i < 8; ++i)
              if (i < st) rsp[i] += rsp[i + st];
          rs = rsp[0];
        }
        rs += __shfl_xor(rs, 32, 64);
        l_run = l_run * alpha + rs;
        if (!__all(alpha == 1.f)) {
#pragma unroll
          for (int nd = 0; nd < 4; ++nd)
#pragma unroll
            for (int reg = 0; reg < 16; ++reg) o_acc[nd][reg] *= alpha;
        }
#pragma unroll
        for (int i = 0; i < 8; i += 4) {
          auto r0 = __builtin_amdgcn_permlane32_swap(c32[i + 0], c32[i + 2], false, false);
          c32[i + 0] = r0[0];
          c32[i + 2] = r0[1];
          auto r1 = __builtin_amdgcn_permlane32_swap(c32[i + 1], c32[i + 3], false, false);
          c32[i + 1] = r1[0];
          c32[i + 3] = r1[1];
        }
      } else {
        keep(s_acc);  // keep QK^T alive
#pragma unroll
        for (int i = 0; i < 8; ++i) c32[i] = 0x3f803f80u + (unsigned)i;
        l_run += 1.f;
      }

      if constexpr (!(MODE & 2)) {  // PV: tr reads + MFMAs
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          bf16x8 a_frag;
#pragma unroll
          for (int i = 0; i < 4; ++i) {
            a_frag[2 * i] = __builtin_bit_cast(__bf16, (unsigned short)(c32[ks * 4 + i] & 0xffffu));
            a_frag[2 * i + 1] = __builtin_bit_cast(__bf16, (unsigned short)(c32[ks * 4 + i] >> 16));
          }
          const int vkey = nb * 32 + ks * 16 + h * 8 + (m16 >> 2);
          const unsigned vbase0 = (unsigned)(uintptr_t)(LDS_SPACE const char*)(
              v_buf + vkey * 256 + (m16 & 3) * 8);
          const unsigned vbase1 = (unsigned)(uintptr_t)(LDS_SPACE const char*)(
              v_buf + (vkey + 4) * 256 + (m16 & 3) * 8);
          const unsigned sw0 = (unsigned)((vkey & 7) << 5);
          const unsigned sw1 = (unsigned)(((vkey + 4) & 7) << 5);
          const unsigned dimoff = (unsigned)(((g16 & 1) << 5));
          u32x2 vr[4][2];
#pragma unroll
          for (int nd = 0; nd < 4; ++nd) {
            const unsigned a0 = vbase0 + (sw0 ^ (unsigned)(nd * 64 + dimoff));
            const unsigned a1 = vbase1 + (sw1 ^ (unsigned)(nd * 64 + dimoff));
            asm volatile("ds_read_b64_tr_b16 %0, %2\n\tds_read_b64_tr_b16 %1, %3"
                         : "=&v"(vr[nd][0]), "=&v"(vr[nd][1])
                         : "v"(a0), "v"(a1));
          }
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
#pragma unroll
          for (int nd = 0; nd < 4; ++nd) {
            bf16x8 v_frag;
#pragma unroll
            for (int p4 = 0; p4 < 2; ++p4)
#pragma unroll
              for (int j = 0; j < 4; ++j) {
                const unsigned half32 = vr[nd][p4][j >> 1];
                v_frag[p4 * 4 + j] = __builtin_bit_cast(
                    __bf16, (unsigned short)((j & 1) ? (half32 >> 16) : (half32 & 0xffffu)));
              }
            o_acc[nd] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(v_frag, a_frag, o_acc[nd], 0, 0, 0);
          }
        }
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) keep(c32[i]);
        o_acc[0][0] += (float)c32[0];
      }
    }
    __builtin_amdgcn_s_barrier();
  }

  const float inv = l_run > 0.f ? 1.f / l_run : 0.f;
  float* orow = o_out + ((size_t)hq * Tq + row0 + row32) * kD;
#pragma unroll
  for (int nd = 0; nd < 4; ++nd) {
#pragma unroll
    for (int q4 = 0; q4 < 4; ++q4) {
      f32x4 st;
#pragma unroll
      for (int j = 0; j < 4; ++j) st[j] = o_acc[nd][q4 * 4 + j] * inv;
      *reinterpret_cast<f32x4*>(orow + nd * 32 + q4 * 8 + 4 * h) = st;
    }
  }
}

#define HIP_CHECK(x) do { hipError_t e = (x); if (e) { printf("ERR %s\n", hipGetErrorString(e)); exit(1);} } while (0)

int main() {
  const int Hq = 32;
  const long Tq = 8192, Tkv = 8192;
  __bf16 *q, *k, *v;
  float* o;
  HIP_CHECK(hipMalloc(&q, (size_t)Hq * Tq * kD * 2));
  HIP_CHECK(hipMalloc(&k, (size_t)Hq * Tkv * kD * 2));
  HIP_CHECK(hipMalloc(&v, (size_t)Hq * Tkv * kD * 2));
  HIP_CHECK(hipMalloc(&o, (size_t)Hq * Tq * kD * 4));
  // random-ish fill (guide rule 25: never bench attention on zeros)
  std::vector<unsigned short> host((size_t)Hq * Tkv * kD);
  unsigned s = 12345;
  for (auto& x : host) { s = s * 1664525u + 1013904223u; x = (unsigned short)(0x3f80 + ((s >> 16) & 0x7ff)) ; }
  HIP_CHECK(hipMemcpy(k, host.data(), host.size() * 2, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(v, host.data(), host.size() * 2, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemcpy(q, host.data(), (size_t)Hq * Tq * kD * 2, hipMemcpyHostToDevice));

  const dim3 grid(Hq * (Tq / kQBlk));
  auto run = [&](int mode) {
    switch (mode) {
      case 0: ablate_kernel<0><<<grid, kThreads>>>(q, k, v, o, Hq, Tq, Tkv, 0.0883f); break;
      case 1: ablate_kernel<1><<<grid, kThreads>>>(q, k, v, o, Hq, Tq, Tkv, 0.0883f); break;
      case 2: ablate_kernel<2><<<grid, kThreads>>>(q, k, v, o, Hq, Tq, Tkv, 0.0883f); break;
      case 4: ablate_kernel<4><<<grid, kThreads>>>(q, k, v, o, Hq, Tq, Tkv, 0.0883f); break;
      case 7: ablate_kernel<7><<<grid, kThreads>>>(q, k, v, o, Hq, Tq, Tkv, 0.0883f); break;
      case 8: ablate_kernel<8><<<grid, kThreads>>>(q, k, v, o, Hq, Tq, Tkv, 0.0883f); break;
    }
  };
  const int modes[6] = {0, 8, 1, 2, 4, 7};
  double ms[6] = {0, 0, 0, 0, 0, 0};
  for (int m = 0; m < 6; ++m) { run(modes[m]); }  // warm
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t e0, e1;
  hipEventCreate(&e0);
  hipEventCreate(&e1);
  const int rounds = 20;
  for (int r = 0; r < rounds; ++r) {
    for (int m = 0; m < 6; ++m) {  // interleaved within-probe
      hipEventRecord(e0);
      run(modes[m]);
      hipEventRecord(e1);
      HIP_CHECK(hipEventSynchronize(e1));
      float dt;
      hipEventElapsedTime(&dt, e0, e1);
      ms[m] += dt;
    }
  }
  const char* names[6] = {"FULL", "FULL+tree", "noSM", "noPV", "noQKT", "stage-only"};
  for (int m = 0; m < 6; ++m) printf("%-10s %.3f ms\n", names[m], ms[m] / rounds);
  printf("tree delta: %.4f ms (neg = tree faster)\n", (ms[1] - ms[0]) / rounds);
  return 0;
}
