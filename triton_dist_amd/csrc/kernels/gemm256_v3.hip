// gemm256_v3 — 256^2 8-phase gray-quadrant GEMM (BK=64, double-buffered
// K-tile pair), rebuilt from the CDNA4 guide's verified template and then
// pipelined one step further: ds_reads for phase p+1 issue in phase p's
// TAIL (after the MFMA cluster), so the LDS-read latency hides under the
// MFMAs and the per-phase serial window shrinks to s_barrier +
// s_waitcnt lgkmcnt(0) (nearly satisfied). One barrier per phase.
//
// Schedule (all positions locked by the CPU ledger test
// tests/test_mappings_cpu.py::test_gemm256_v3_pipeline_ledger):
//   * Staged unit = QUADRANT UNION, matching what the gray walk reads:
//     A-half h = rows {(cr/64)*128 + h*64 + cr%64}, B-half h with band 32.
//   * Stage order per tile [A0, B1, A1, B0], 7-half-tile prologue; phase
//     g stages half-tile s = g+7 (variant (ph+3)&3), so each stage issues
//     >= 1 barrier-separated phase after the replaced slot's last read.
//   * Counted drains: vmcnt(6) at the tails of phases 3 and 7 (3
//     half-tiles = 6 global_load_lds in flight); vmcnt(0) only at the
//     last pair's phase 3 once staging is exhausted.
//   * Phase body: [boundary only: read A+B at top] -> lgkmcnt(0) ->
//     16 MFMA -> tail: stage one half-tile, pre-read next phase's
//     changed operand, [drain], s_barrier.
//   * All stage addresses are precomputed per thread and advanced by
//     constants (+BK elems per use; LDS base XOR-toggles the buffer), so
//     the phase window carries no 64-bit address VALU chains.
#include <stdexcept>

#include "td/api.hpp"

namespace td {

namespace g256v3 {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int NTH = 512;                    // 8 waves, 2 (M) x 4 (N)
constexpr int TILE_ELEMS = BM * BK;         // per matrix per buffer

TD_DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// st_16x32 swizzle: XOR bit 9 into bit 5 of the tile-relative byte
// offset (row-preserving involution; source-side at stage, read-side in
// read_frag).
TD_DEV int swz_off(int byte_off) {
  return byte_off ^ (((byte_off >> 9) & 1) << 5);
}

// Per-variant banded row map: compact row cr (0..127) -> tile row.
TD_DEV int quad_row(int cr, int band, int h) {
  return (cr / band) * 2 * band + h * band + (cr % band);
}

// Precomputed stage state. Variants: 0=A h0 (band 64), 1=B h1 (band 32),
// 2=A h1, 3=B h0. Each holds 2 per-thread global sources (it 0/1) that
// advance by BK elements per use, a wave-uniform LDS element base, and
// the buffer parity (XOR-toggled per use).
struct StageState {
  const bf16 *a0[2], *b1[2], *a1[2], *b0[2];
  int da0[2], db1[2], da1[2], db0[2];  // LDS elem offsets (buf 0)
  int pa0, pb1, pa1, pb0;              // buffer parity of next use
};

TD_DEV void init_variant(const bf16 *g, int ld, int band, int h,
                         const bf16 *src[2], int dst[2]) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int q = it * NTH + tid;
    int cr = q >> 3;
    int row = quad_row(cr, band, h);
    int p_byte = (row * BK + (q & 7) * 8) * 2;
    int un = swz_off(p_byte);  // same row, swizzled column chunk
    int col = (un % (BK * 2)) / 2;
    src[it] = g + (size_t)row * ld + col;
    int cr0 = it * 64 + wave * 8;
    dst[it] = quad_row(cr0, band, h) * BK;
  }
}

TD_DEV void init_stage(StageState &st, const bf16 *ga, const bf16 *gb,
                       int lda, int ldb) {
  init_variant(ga, lda, 64, 0, st.a0, st.da0);
  init_variant(gb, ldb, 32, 1, st.b1, st.db1);
  init_variant(ga, lda, 64, 1, st.a1, st.da1);
  init_variant(gb, ldb, 32, 0, st.b0, st.db0);
  st.pa0 = st.pb1 = st.pa1 = st.pb0 = 0;
}

template <int V>
TD_DEV void stage_step(StageState &st, bf16 *lds_a, bf16 *lds_b) {
  const bf16 **src;
  int *dst;
  int *par;
  bf16 *lds;
  if constexpr (V == 0) {
    src = st.a0; dst = st.da0; par = &st.pa0; lds = lds_a;
  } else if constexpr (V == 1) {
    src = st.b1; dst = st.db1; par = &st.pb1; lds = lds_b;
  } else if constexpr (V == 2) {
    src = st.a1; dst = st.da1; par = &st.pa1; lds = lds_a;
  } else {
    src = st.b0; dst = st.db0; par = &st.pb0; lds = lds_b;
  }
  const int boff = *par * TILE_ELEMS;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int *)src[it],
        (__attribute__((address_space(3))) unsigned int *)(
            lds + boff + dst[it]),
        16, 0, 0);
    src[it] += BK;
  }
  *par ^= 1;
}

TD_DEV bf16x8 read_frag(const bf16 *lds, int buf, int row, int ks,
                        int lane) {
  int byte_off = (row * BK + ks * 32 + (lane >> 4) * 8) * 2;
  return *(const bf16x8 *)((const char *)(lds + buf * TILE_ELEMS) +
                           swz_off(byte_off));
}

// dispatch helper: variant is compile-time under the unrolled ph loop
TD_DEV void constexpr_stage(StageState &st, bf16 *lds_a, bf16 *lds_b,
                            int ph) {
  switch ((ph + 3) & 3) {
    case 0: stage_step<0>(st, lds_a, lds_b); break;
    case 1: stage_step<1>(st, lds_a, lds_b); break;
    case 2: stage_step<2>(st, lds_a, lds_b); break;
    default: stage_step<3>(st, lds_a, lds_b); break;
  }
}

// The pipelined 8-phase K loop. ktiles must be even (k % 128 == 0).
TD_DEV void kloop(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                  int ktiles, bf16 *lds_a, bf16 *lds_b, int wr, int wc,
                  int lane, f32x4 acc[8][4]) {
  const int pairs = ktiles / 2;
  StageState st;
  init_stage(st, ga, gb, lda, ldb);
  // prologue: 7 half-tiles (tile 0 complete + tile 1 [A0,B1,A1])
  stage_step<0>(st, lds_a, lds_b);
  stage_step<1>(st, lds_a, lds_b);
  stage_step<2>(st, lds_a, lds_b);
  stage_step<3>(st, lds_a, lds_b);
  stage_step<0>(st, lds_a, lds_b);
  stage_step<1>(st, lds_a, lds_b);
  stage_step<2>(st, lds_a, lds_b);
  if (ktiles > 1) {
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  bf16x8 af[4][2], bfr[2][2];

  auto read_a = [&](int rbuf, int ih) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wr * 128 + (ih * 4 + i) * 16 + (lane & 15);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        af[i][ks] = read_frag(lds_a, rbuf, row, ks, lane);
    }
  };
  auto read_b = [&](int rbuf, int jh) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int row = wc * 64 + (jh * 2 + j) * 16 + (lane & 15);
#pragma unroll
      for (int ks = 0; ks < 2; ++ks)
        bfr[j][ks] = read_frag(lds_b, rbuf, row, ks, lane);
    }
  };

  for (int p = 0; p < pairs; ++p) {
    const bool last = (p == pairs - 1);
#pragma unroll
    for (int ph = 0; ph < 8; ++ph) {
      const int rbuf = ph >> 2;  // tile 2p is buf 0, 2p+1 is buf 1
      const int qq = ph & 3;
      const int ih = (qq >= 2) ? 1 : 0;
      const int jh = (qq == 1 || qq == 2) ? 1 : 0;

      if (qq == 0) {
        // boundary: fresh buffer — read both operands at the top (the
        // previous tail's drain + this barrier certify residency)
        read_a(rbuf, 0);
        read_b(rbuf, 0);
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
                af[i][ks], bfr[j][ks], acc[ih * 4 + i][jh * 2 + j]);
      __builtin_amdgcn_s_setprio(0);

      // tail: stage half-tile s = 8p+ph+7 (variant (ph+3)&3), pre-read
      // the NEXT phase's changed operand, drain at half-pair ends.
      if (!last || ph == 0) {
        constexpr_stage(st, lds_a, lds_b, ph);
      }
      if (qq == 0) {
        read_b(rbuf, 1);        // phase +1 consumes B1
      } else if (qq == 1) {
        read_a(rbuf, 1);        // phase +2 consumes A1
      } else if (qq == 2) {
        read_b(rbuf, 0);        // phase +3 re-consumes B0
      }
      if (qq == 3 && (ph == 3 || !last)) {
        if (last && ph == 3) {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        } else {
          asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        }
      }
      __builtin_amdgcn_s_barrier();
    }
  }
}

// GROUP_M supertile walk (same mapping as gemm256.hip tile_coords).
TD_DEV void tile_coords(int wgid, int tiles_m, int tiles_n, int &pid_m,
                        int &pid_n) {
  constexpr int GM = 4;
  int group = wgid / (GM * tiles_n);
  int first_m = group * GM;
  int gsz = min(tiles_m - first_m, GM);
  pid_m = first_m + (wgid % (GM * tiles_n)) % gsz;
  pid_n = (wgid % (GM * tiles_n)) / gsz;
}

__global__ __launch_bounds__(NTH, 2) void k_gemm256_v3(GemmArgs args) {
  __shared__ bf16 lds_a[2 * TILE_ELEMS];
  __shared__ bf16 lds_b[2 * TILE_ELEMS];
  const int tiles_n = args.n / BN;
  const int tiles_m = args.m / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m, pid_n;
  tile_coords(wgid, tiles_m, tiles_n, pid_m, pid_n);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2;  // M half
  const int wc = wave & 3;   // N quarter

  const bf16 *ga = (const bf16 *)args.a + (size_t)pid_m * BM * args.lda;
  const bf16 *gb = (const bf16 *)args.b + (size_t)pid_n * BN * args.ldb;

  f32x4 acc[8][4] = {};
  kloop(ga, gb, args.lda, args.ldb, args.k / BK, lds_a, lds_b, wr, wc,
        lane, acc);

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

}  // namespace g256v3

void launch_gemm256_v3_bf16(const GemmArgs &args, hipStream_t stream) {
  using namespace g256v3;
  if (args.m % BM || args.n % BN || args.k % (2 * BK))
    throw std::runtime_error("gemm256_v3: m%256, n%256, k%128 required");
  int grid = (args.m / BM) * (args.n / BN);
  hipLaunchKernelGGL(k_gemm256_v3, dim3(grid), dim3(NTH), 0, stream, args);
}

}  // namespace td
