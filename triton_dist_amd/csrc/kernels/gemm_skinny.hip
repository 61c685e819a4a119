// Skinny-M GEMM tier: BM=32 x BN=128 tiles, BK=64 3-buffer pipelined
// K-loop (the discipline proven by the MoE pq kernel, measured at decode
// loads) — the DIRECT kernel for M<=1024 decode shapes where the 256^2
// tile grid starves the 256 CUs and the fp32-atomic split-K tier doubles
// traffic (ws round trip measured: down-proj 172 us/layer vs a ~35 us
// weight-BW floor; hipBLASLt 57-60 us on qkv/o vs ~20 us floors).
//
//   * grid = tiles_m * tiles_n with COLUMN-MAJOR XCD mapping: consecutive
//     workgroup ids (which xcd_remap makes XCD-contiguous) share a B
//     column block, so each XCD streams its 1/8 slice of the weight
//     matrix through its own L2 once instead of all of B per row band.
//   * optional fused-SwiGLU epilogue (interleaved gate/up columns, same
//     adjacent-lane pairing as moe.hip) and bias add.
//
// Capability parity: the reference's small-M GEMM configs
// (kernels/amd/gemm.py:62-541 config space — behavior only).
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace gsk {
constexpr int BM = 32, BN = 128, BK = 64, NTH = 256;
constexpr int ABUF = BM * BK, BBUF = BN * BK;
}  // namespace gsk

__global__ __launch_bounds__(gsk::NTH) void k_gemm_skinny(
    GemmArgs g, int fuse_swiglu) {
  using namespace gsk;
  __shared__ bf16 lds_a[3 * ABUF];
  __shared__ bf16 lds_b[3 * BBUF];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int tiles_m = g.m / BM;
  const int tiles_n = g.n / BN;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  const int tn = wgid / tiles_m;   // column-major: B block reused per XCD
  const int tm = wgid % tiles_m;

  f32x4 acc[2][2] = {};
  const bf16 *ga = (const bf16 *)g.a + (size_t)tm * BM * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)tn * BN * g.ldb;
  const int ksteps = g.k / BK;

  auto stage = [&](int t, int buf) {
    const int k0 = t * BK;
    {
      int row = tid >> 3, kc = tid & 7;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              ga + (size_t)row * g.lda + k0 + kc * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_a + buf * ABUF + (wave * 64) * 8),
          16, 0, 0);
    }
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int qb = it * NTH + tid;
      int rowb = qb >> 3, kcb = qb & 7;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              gb + (size_t)rowb * g.ldb + k0 + kcb * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_b + buf * BBUF + (it * NTH + wave * 64) * 8),
          16, 0, 0);
    }
  };
  stage(0, 0);
  if (ksteps > 1) stage(1, 1);
  for (int t = 0; t < ksteps; ++t) {
    const int buf = t % 3;
    if (t + 1 < ksteps) {
      asm volatile("s_waitcnt vmcnt(5)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
    if (t + 2 < ksteps) stage(t + 2, (t + 2) % 3);
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      bf16x8 af[2], bfr[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int arow = i * 16 + (lane & 15);
        int brow = wave * 32 + i * 16 + (lane & 15);
        int kk = ks * 32 + (lane >> 4) * 8;
        af[i] = *(const bf16x8 *)(lds_a + buf * ABUF + arow * BK + kk);
        bfr[i] = *(const bf16x8 *)(lds_b + buf * BBUF + brow * BK + kk);
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bfr[j], acc[i][j], 0, 0, 0);
    }
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
  }
  bf16 *out = (bf16 *)g.c;
  const bf16 *bias = (const bf16 *)g.bias;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = tm * BM + i * 16 + (lane >> 4) * 4 + r;
        int col = wave * 32 + j * 16 + (lane & 15);
        if (fuse_swiglu) {
          float gv = acc[i][j][r];
          float partner = __shfl_xor(gv, 1);
          if ((lane & 1) == 0) {
            float silu = gv / (1.f + __expf(-gv));
            out[(size_t)row * (g.n / 2) + ((size_t)tn * BN + col) / 2] =
                (bf16)(silu * partner);
          }
        } else {
          float v = acc[i][j][r];
          if (bias) v += (float)bias[tn * BN + col];
          out[(size_t)row * g.ldc + tn * BN + col] = (bf16)v;
        }
      }
}

void launch_gemm_skinny(const GemmArgs &g, int fuse_swiglu,
                        hipStream_t stream) {
  using namespace gsk;
  if (g.m % BM || g.n % BN || g.k % BK)
    throw std::runtime_error("gemm_skinny: m%32, n%128, k%64 required");
  int grid = (g.m / BM) * (g.n / BN);
  hipLaunchKernelGGL(k_gemm_skinny, dim3(grid), dim3(NTH), 0, stream, g,
                     fuse_swiglu);
}

}  // namespace td
