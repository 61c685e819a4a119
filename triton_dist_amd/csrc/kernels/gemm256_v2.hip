// EXPERIMENTAL 256x256 / BK=64 quadrant-phase GEMM — the guide template's
// structure (8 phases per K-tile PAIR, one C-quadrant x K=64 per phase,
// double-buffered K-tiles, one half-tile stage per phase, lockstep
// barriers around a 16-MFMA cluster). The production kernel
// (gemm256.hip) is the K-slice ring at ~1000/1180 TF; the template
// measured 1563/1728 on the same hardware, and an earlier attempt to
// graft its fine interleave ONTO the ring regressed (docs/ROADMAP.md) —
// this file is the faithful-structure rebuild to iterate on next round.
//
// Ledger (sound variant; the guide's vmcnt(6)-at-phase-0 could not be
// proven correct for the halves consumed in that same phase):
//   pair P processes K-tiles (2P, 2P+1) from bufs (0, 1);
//   phases 0-3 stage tile 2P+1 -> buf1 (freed at the END of pair P-1;
//     consumed THIS pair at phases 4-7),
//   phases 4-7 stage tile 2P+2 -> buf0 (freed after phase 3; consumed
//     at pair P+1 phases 0-3);
//   with only 2 buffers the stage-to-consume gap is 4 phases on both
//   streams, so BOTH half-pair boundaries drain: vmcnt(0) at phase 0
//   and phase 4. That is one drain per 64 MFMA/wave — the knob round 2
//   iterates on (3 narrower buffers, or wave-aligned staging, to get a
//   counted non-zero wait; see ROADMAP).
//
// NOT wired into any dispatch path: reachable only via
// _C.gemm256_v2_bf16 (scripts/bench_gemm_v2.py; tests env-gated by
// TD_EXPERIMENTAL=1).
#include <stdexcept>

#include "td/api.hpp"

namespace td {

namespace g256v2 {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int NTH = 512;  // 8 waves, 2 (M) x 4 (N)
constexpr int TILE_ELEMS = BM * BK;         // per matrix per buffer
constexpr int HALF_ELEMS = TILE_ELEMS / 2;  // rows 0-127 / 128-255

TD_DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// st_16x32-style swizzle on the tile-relative byte offset: XOR bit 9
// (512B row-group) into bit 5 (32B chunk). Applied to the global SOURCE
// address at stage time and to the ds_read address (involution).
TD_DEV int swz_off(int byte_off) {
  return byte_off ^ (((byte_off >> 9) & 1) << 5);
}

// Stage one half-tile (rows h*128..h*128+127) of the K-tile starting at
// column k0 into buffer `buf`: 2 x global_load_lds per thread.
TD_DEV void stage_half(const bf16 *g, int ld, int k0, bf16 *lds, int buf,
                       int h) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    // 1024 chunks of 16B per half-tile; chunk q covers row q/8, cols
    // (q%8)*8 of the 64-wide K slice
    int q = it * NTH + tid;
    int dst_byte = (h * HALF_ELEMS + q * 8) * 2;
    int un = swz_off(dst_byte);          // logical position for this slot
    int row = un / (BK * 2);
    int col = (un % (BK * 2)) / 2;
    const bf16 *src = g + (size_t)row * ld + k0 + col;
    int wave_chunk0 = it * NTH + wave * 64;  // wave-uniform LDS base
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int *)src,
        (__attribute__((address_space(3))) unsigned int *)(
            lds + buf * TILE_ELEMS + h * HALF_ELEMS + wave_chunk0 * 8),
        16, 0, 0);
  }
}

// ds_read one A- or B-fragment (bf16x8, K=32 half ks) with the swizzle.
TD_DEV bf16x8 read_frag(const bf16 *lds, int buf, int row, int ks,
                        int lane) {
  int byte_off = (row * BK + ks * 32 + (lane >> 4) * 8) * 2;
  return *(const bf16x8 *)((const char *)(lds + buf * TILE_ELEMS) +
                           swz_off(byte_off));
}

__global__ __launch_bounds__(NTH, 2) void k_gemm256_v2(GemmArgs args) {
  __shared__ bf16 lds_a[2 * TILE_ELEMS];
  __shared__ bf16 lds_b[2 * TILE_ELEMS];
  const int tiles_n = args.n / BN;
  const int tiles_m = args.m / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  // GROUP_M supertile walk (same as gemm.hip tile_coords)
  constexpr int GM = 4;
  int group = wgid / (GM * tiles_n);
  int first_m = group * GM;
  int gsz = min(tiles_m - first_m, GM);
  int pid_m = first_m + (wgid % (GM * tiles_n)) % gsz;
  int pid_n = (wgid % (GM * tiles_n)) / gsz;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2;  // M half
  const int wc = wave & 3;   // N quarter

  const bf16 *ga = (const bf16 *)args.a + (size_t)pid_m * BM * args.lda;
  const bf16 *gb = (const bf16 *)args.b + (size_t)pid_n * BN * args.ldb;
  const int ktiles = args.k / BK;
  const int pairs = ktiles / 2;

  f32x4 acc[8][4] = {};

  // prologue: tile 0 only (tile 1 is staged during pair-0 phases 0-3)
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    stage_half(ga, args.lda, 0, lds_a, 0, h);
    stage_half(gb, args.ldb, 0, lds_b, 0, h);
  }

  for (int p = 0; p < pairs; ++p) {
#pragma unroll
    for (int ph = 0; ph < 8; ++ph) {
      const int tp = ph >> 2;          // tile parity within the pair
      const int buf = tp;
      const int qq = ph & 3;           // quadrant, gray-coded below
      // gray order (ih, jh): (0,0) (0,1) (1,1) (1,0)
      const int ih = (qq == 2 || qq == 3) ? 1 : 0;
      const int jh = (qq == 1 || qq == 2) ? 1 : 0;

      if (ph == 0 || ph == 4) {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_sched_barrier(0);

      // stage stream (see ledger above): phases 0-3 -> tile 2p+1 (buf1),
      // phases 4-7 -> tile 2p+2 (buf0, skipped on the last pair)
      const int stage_tile = (ph >= 4) ? 2 * p + 2 : 2 * p + 1;
      if (stage_tile < ktiles) {
        const int sph = ph & 3;        // 0: A h0, 1: A h1, 2: B h0, 3: B h1
        const int k0 = stage_tile * BK;
        const int sbuf = stage_tile & 1;
        if (sph == 0) stage_half(ga, args.lda, k0, lds_a, sbuf, 0);
        if (sph == 1) stage_half(ga, args.lda, k0, lds_a, sbuf, 1);
        if (sph == 2) stage_half(gb, args.ldb, k0, lds_b, sbuf, 0);
        if (sph == 3) stage_half(gb, args.ldb, k0, lds_b, sbuf, 1);
      }

      // ds_read this quadrant's fragments: A 4 rows x 2 ks, B 2 rows x 2
      bf16x8 af[4][2], bfrag[2][2];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wr * 128 + (ih * 4 + i) * 16 + (lane & 15);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          af[i][ks] = read_frag(lds_a, buf, row, ks, lane);
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int row = wc * 64 + (jh * 2 + j) * 16 + (lane & 15);
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          bfrag[j][ks] = read_frag(lds_b, buf, row, ks, lane);
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            acc[ih * 4 + i][jh * 2 + j] = mfma16(
                af[i][ks], bfrag[j][ks], acc[ih * 4 + i][jh * 2 + j]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_sched_barrier(0);
    }
  }

  // epilogue: direct 16B-per-lane stores (no LDS C-tile needed for the
  // plain kernel)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  bf16 *dst = (bf16 *)args.c + (size_t)pid_m * BM * args.ldc + pid_n * BN;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = wr * 128 + i * 16 + ((lane >> 4) * 4 + r);
        int col = wc * 64 + j * 16 + (lane & 15);
        dst[(size_t)row * args.ldc + col] = (bf16)acc[i][j][r];
      }
}

}  // namespace g256v2

void launch_gemm256_v2_bf16(const GemmArgs &args, hipStream_t stream) {
  using namespace g256v2;
  if (args.m % BM || args.n % BN || args.k % (2 * BK))
    throw std::runtime_error("gemm256_v2: m%256, n%256, k%128 required");
  int grid = (args.m / BM) * (args.n / BN);
  hipLaunchKernelGGL(k_gemm256_v2, dim3(grid), dim3(NTH), 0, stream, args);
}

}  // namespace td
