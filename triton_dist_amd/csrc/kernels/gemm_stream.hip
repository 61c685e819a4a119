// Weight-streaming decode GEMM: BM = 512 (the whole decode batch is ONE
// m-tile) x BN = 64, K accumulated in registers.
//
// Why: decode-shaped GEMMs (m = 512, K up to 27k) are WEIGHT-BANDWIDTH
// bound — the floor is "read W[N,K] once from HBM" (~35 us for a 283 MB
// down-proj). The 256^2 tile kernels re-read the B panel tiles_m times
// and walk full-K A panels (14-28 MB) that fall out of the 4 MB per-XCD
// L2, measuring ~6x off that floor (profiles/README.md, skinny-tier
// notes). This kernel makes the traffic optimal instead:
//   * BM = 512 covers all of m  ->  every B byte is read EXACTLY once;
//   * A (the small operand, 512 x K) is re-read by all n-tiles but only
//     through L2: the k-split chunk (<= 4 MB) stays resident per XCD
//     because the s-major grid layout + xcd_remap puts all n-tiles of
//     one k-chunk on the same XCD;
//   * K accumulates in registers (4x4 f32x4 accs / wave) — no split-K
//     workspace round trip unless sk > 1, and then the existing
//     two-stage k_sk2_reduce (gemm256.hip) handles the fp32 partials.
// Capability row: the reference's decode/persistent GEMM zoo
// (Triton-distributed python/triton_dist/kernels/amd/gemm.py:62-541,
// chunked split-K + XCD pid remap — behavior only; the BM=m streaming
// design is MI355X-native).
//
// Layout per WG (8 waves, 512 threads): wave w owns rows [64w, 64w+64)
// of the 512 x 64 output; per BK=32 slice each wave runs 16
// mfma_f32_16x16x32_bf16. LDS per buffer: A 512x32 + B 64x32 bf16 =
// 36 KB; 4-slice ring = 144 KB -> 1 block/CU (depth beats occupancy
// for latency hiding, same trade the 256^2 ring kernel makes). Staging
// uses global_load_lds (16 B/lane, wave-uniform LDS base) with the g256
// 2-bit XOR chunk swizzle on both sides; ledger: 5 loads/wave/slice
// (4 A + 1 B with lanes 0-31), 3 slices in flight -> steady
// s_waitcnt vmcnt(10), drain 10/5/0.
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// two-stage split-K reduce (defined in gemm256.hip)
__global__ void k_sk2_reduce(const float *__restrict__ ws,
                             bf16 *__restrict__ c,
                             const bf16 *__restrict__ bias, int rows, int n,
                             int sk);

namespace gs {

constexpr int BM = 512, BN = 64, BK = 32, NTH = 512;
constexpr int ABUF = BM * BK;  // elems
constexpr int BBUF = BN * BK;

TD_DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// 4 x 16B chunks per 32-elem row; physical = logical ^ ((row>>1)&3)
// (same involution as g256: applied to the global SOURCE address at
// stage time and to the read address — lds rows stay 64B but the 16
// rows of an MFMA frag read hit 4 distinct chunk columns)
TD_DEV int swz(int row, int j) { return j ^ ((row >> 1) & 3); }

}  // namespace gs

__global__ __launch_bounds__(gs::NTH, 1) void k_gemm_stream_bf16(
    GemmArgs g, float *__restrict__ ws, int sk) {
  using namespace gs;
  __shared__ bf16 lds_a[4 * ABUF];
  __shared__ bf16 lds_b[4 * BBUF];
  const int tiles_n = g.n / BN;
  // s-major layout: wgid = s * tiles_n + tn, so xcd_remap's contiguous
  // per-XCD ranges keep one k-chunk's n-tiles (one A chunk) per XCD
  int wgid = xcd_remap(blockIdx.x, tiles_n * sk);
  const int s = wgid / tiles_n;
  const int tn = wgid % tiles_n;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int kchunk = g.k / sk;
  const int ksl = kchunk / BK;
  const bf16 *ga = (const bf16 *)g.a + (size_t)s * kchunk;
  const bf16 *gb = (const bf16 *)g.b + (size_t)tn * BN * g.ldb +
                   (size_t)s * kchunk;
  const int mlim = g.m;  // rows >= m are staged clamped, masked at store

  // stage slice t into buffer buf: per wave 4 A loads + 1 B load
  auto stage = [&](int t, int buf) {
    const int kk = t * BK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int u = it * NTH + tid;          // 16B unit id, 4 per A row
      int row = u >> 2, jp = u & 3;
      int ar = row < mlim ? row : mlim - 1;
      int jg = swz(row, jp);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              ga + (size_t)ar * g.lda + kk + jg * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_a + buf * ABUF + (it * NTH + wave * 64) * 8),
          16, 0, 0);
    }
    {
      int u = wave * 32 + (lane & 31);  // 256 units, lanes 0-31 per wave
      int row = u >> 2, jp = u & 3;
      int jg = swz(row, jp);
      if (lane < 32)
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int *)(
                gb + (size_t)row * g.ldb + kk + jg * 8),
            (__attribute__((address_space(3))) unsigned int *)(
                lds_b + buf * BBUF + (wave * 32) * 8),
            16, 0, 0);
    }
  };

  // 4-slice ring pipeline (the proven g256 kloop cadence): slices t..t+2
  // in flight (15 loads/wave), ONE barrier per slice. The entry barrier
  // of slice t proves (a) every wave's slice-t loads landed (each waited
  // its own vmcnt before arriving) and (b) every wave consumed ring slot
  // (t-1)%4 in the previous slice (its MFMAs — which register-consume
  // the ds_reads — precede its barrier arrival), so staging slice t+3
  // into that slot after the barrier is safe.
  f32x4 acc[4][4] = {};
  stage(0, 0);
  if (ksl > 1) stage(1, 1);
  if (ksl > 2) stage(2, 2);
  for (int t = 0; t < ksl; ++t) {
    const int buf = t & 3;
    // retire slice t: 5 loads per newer in-flight slice may remain
    if (t + 2 < ksl) {
      asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
    } else if (t + 1 < ksl) {
      asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
    if (t + 3 < ksl) stage(t + 3, (t + 3) & 3);
    const int kq = (lane >> 4) * 8;  // k offset of my frag quarter
    bf16x8 af[4], bfr[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wave * 64 + i * 16 + (lane & 15);
      af[i] = *(const bf16x8 *)(lds_a + buf * ABUF + row * BK +
                                swz(row, kq >> 3) * 8);
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int row = j * 16 + (lane & 15);
      bfr[j] = *(const bf16x8 *)(lds_b + buf * BBUF + row * BK +
                                 swz(row, kq >> 3) * 8);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = mfma16(af[i], bfr[j], acc[i][j]);
    __builtin_amdgcn_s_setprio(0);
  }

  // epilogue: sk == 1 -> bf16 C (+bias); sk > 1 -> fp32 ws slice s
  const bf16 *bias = (const bf16 *)g.bias;
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = wave * 64 + i * 16 + (lane >> 4) * 4 + r;
        int col = tn * BN + j * 16 + (lane & 15);
        if (row >= mlim) continue;
        if (sk == 1) {
          float v = acc[i][j][r];
          if (bias) v += (float)bias[col];
          ((bf16 *)g.c)[(size_t)row * g.ldc + col] = (bf16)v;
        } else {
          ws[((size_t)s * g.m + row) * g.n + col] = acc[i][j][r];
        }
      }
}

void launch_gemm_stream_bf16(const GemmArgs &g, float *ws, int sk,
                             hipStream_t stream) {
  if (g.m > gs::BM)
    throw std::runtime_error("gemm_stream: m > 512");
  if (g.n % gs::BN || g.k % (gs::BK * sk))
    throw std::runtime_error("gemm_stream: n % 64 or k % (32*sk) != 0");
  if (sk > 1 && !ws)
    throw std::runtime_error("gemm_stream: sk > 1 needs ws");
  int grid = (g.n / gs::BN) * sk;
  hipLaunchKernelGGL(k_gemm_stream_bf16, dim3(grid), dim3(gs::NTH), 0,
                     stream, g, ws, sk);
  if (sk > 1) {
    int cgrid = g.m < 2048 ? g.m : 2048;
    hipLaunchKernelGGL(k_sk2_reduce, dim3(cgrid), dim3(256), 0, stream, ws,
                       (bf16 *)g.c, (const bf16 *)g.bias, g.m, g.n, sk);
  }
}

}  // namespace td
