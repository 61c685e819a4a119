// gemm256_v3 — faithful rebuild of the CDNA4 guide's verified 256^2
// 8-phase template (BK=64, double-buffered K-tile pair, one C-quadrant x
// K=64 per phase, ONE half-tile staged per phase, counted vmcnt drains).
//
// What v2 got wrong (measured 712/737 TF on MI355X vs the template's
// 1563/1728 and the production ring's 1035/1154):
//   1. re-read all 12 fragments per phase. The template's gray-coded
//      quadrant walk (0,0)(0,1)(1,1)(1,0) changes ONE operand half per
//      phase, so only 4 (B) or 8 (A) ds_read_b128 are issued per phase
//      and the other operand stays in registers (12 only at a buffer
//      switch).
//   2. vmcnt(0) full drains at phase 0/4. The sound counted schedule:
//      with per-tile stage order [A0, B1, A1, B0] and a 7-half-tile
//      prologue, the drain needed at the END of each half-pair (phases
//      3/7, before the barrier that opens the next buffer's reads) is
//      exactly vmcnt(6) = 3 half-tiles in flight. Ledger: at pair p
//      phase 3, issued = 8p+11 half-tiles, consumption of tile 2p+1
//      needs s <= 8p+7 landed -> 3 outstanding. Every slot overwrite
//      (stage of tile t's half h at global phase 4t+idx(h)-7) lands one
//      barrier-separated phase after the last ds_read issue of the slot
//      it replaces (tile t-2's half h at phase 4t-8+idx(h)).
//   3. sched_barrier(0) order-pinning (the guide measured that class of
//      pinning as a regression).
//
// Phase body (per the guide template):
//      ds_read the CHANGED operand half  (4/8/12 x ds_read_b128)
//      stage one half-tile               (2 x global_load_lds, 16B)
//      [phase 3/7] s_waitcnt vmcnt(6)    (vmcnt(0) on the last pair)
//      s_barrier
//      s_waitcnt lgkmcnt(0)
//      s_setprio(1); 16 x mfma_f32_16x16x32_bf16; s_setprio(0)
//      s_barrier
//
// The kloop is shared by the plain kernel and the fused AG-consumer /
// RS-producer / split-K variants (same roles as gemm256.hip's ring).
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
constexpr int HALF_ELEMS = TILE_ELEMS / 2;  // rows 0-127 / 128-255

TD_DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// st_16x32 swizzle on the tile-relative byte offset: XOR bit 9 into bit
// 5. Applied to the global SOURCE address at stage time and to the
// ds_read address (involution). Same mapping as v2 (CPU-simulated and
// GPU-numerics-proven there).
TD_DEV int swz_off(int byte_off) {
  return byte_off ^ (((byte_off >> 9) & 1) << 5);
}

// Stage one QUADRANT-union half-tile of the K-tile at column k0 into
// buffer `buf`. The consumed unit of the gray walk is not a contiguous
// 128-row block: wave wr reads A rows wr*128 + ih*64 + [0,64), so the
// A-"half" ih is the union {0-63, 128-191} (ih=0) / {64-127, 192-255}
// (ih=1) — band 64. B likewise with band 32 (wave wc reads rows
// wc*64 + jh*32 + [0,32)). Staging exactly these unions is what makes
// the [A0,B1,A1,B0] overwrite ledger sound (each stage lands one
// barrier-separated phase after the replaced slot's last read issue).
//
// compact row cr (0..127) -> tile row (cr/band)*2*band + h*band +
// cr%band. Each wave's 64 chunks cover 8 consecutive tile rows (cr
// stays inside one band), so the wave-uniform LDS base + lane*16B
// placement of global_load_lds lands every chunk at its natural
// full-tile linear position; the st_16x32 swizzle is applied on the
// SOURCE address (involution against the read side, row-preserving).
template <int BAND>
TD_DEV void stage_quad(const bf16 *g, int ld, int k0, bf16 *lds, int buf,
                       int h) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int q = it * NTH + tid;  // 1024 chunks of 16B per half-tile
    int cr = q >> 3;         // compact row 0..127
    int row = (cr / BAND) * 2 * BAND + h * BAND + (cr % BAND);
    int p_byte = (row * BK + (q & 7) * 8) * 2;  // physical LDS byte
    int un = swz_off(p_byte);                   // logical slot (same row)
    int col = (un % (BK * 2)) / 2;
    const bf16 *src = g + (size_t)row * ld + k0 + col;
    int cr0 = it * 64 + wave * 8;  // wave's first compact row
    int row0 = (cr0 / BAND) * 2 * BAND + h * BAND + (cr0 % BAND);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int *)src,
        (__attribute__((address_space(3))) unsigned int *)(
            lds + buf * TILE_ELEMS + row0 * BK),
        16, 0, 0);
  }
}

TD_DEV bf16x8 read_frag(const bf16 *lds, int buf, int row, int ks,
                        int lane) {
  int byte_off = (row * BK + ks * 32 + (lane >> 4) * 8) * 2;
  return *(const bf16x8 *)((const char *)(lds + buf * TILE_ELEMS) +
                           swz_off(byte_off));
}

// Per-tile stage order [A0, B1, A1, B0] (quadrant unions): read issues
// per window are A0@ph0, B1@ph1, A1@ph2, B0@ph0+ph3 — so each stage
// trails the replaced slot's last read issue by exactly one
// barrier-separated phase (A0 staged at +1, B1 at +2, A1 at +3, B0 at
// +4 relative to the replaced tile's window start).
TD_DEV void stage_seq(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                      bf16 *lds_a, bf16 *lds_b, int s) {
  const int t = s >> 2;
  const int buf = t & 1;
  const int k0 = t * BK;
  switch (s & 3) {
    case 0: stage_quad<64>(ga, lda, k0, lds_a, buf, 0); break;
    case 1: stage_quad<32>(gb, ldb, k0, lds_b, buf, 1); break;
    case 2: stage_quad<64>(ga, lda, k0, lds_a, buf, 1); break;
    default: stage_quad<32>(gb, ldb, k0, lds_b, buf, 0); break;
  }
}

// The 8-phase pipelined K loop. ktiles must be even (k % 128 == 0).
TD_DEV void kloop(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                  int ktiles, bf16 *lds_a, bf16 *lds_b, int wr, int wc,
                  int lane, f32x4 acc[8][4]) {
  const int total_halves = 4 * ktiles;
  const int pairs = ktiles / 2;

  // prologue: 7 half-tiles (tile 0 complete + tile 1 [A0,B1,A1]), then
  // drain so tile 0 is resident: 3 half-tiles (6 loads) may stay in
  // flight.
  for (int s = 0; s < 7 && s < total_halves; ++s)
    stage_seq(ga, gb, lda, ldb, lds_a, lds_b, s);
  if (total_halves > 3) {
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  bf16x8 af[4][2], bfr[2][2];
  int s = 7;  // next half-tile to stage

  for (int p = 0; p < pairs; ++p) {
#pragma unroll
    for (int ph = 0; ph < 8; ++ph) {
      const int rbuf = (2 * p + (ph >> 2)) & 1;
      const int qq = ph & 3;
      // gray order (ih, jh): (0,0) (0,1) (1,1) (1,0)
      const int ih = (qq >= 2) ? 1 : 0;
      const int jh = (qq == 1 || qq == 2) ? 1 : 0;

      // ds_read only the operand half that changed this phase:
      // A at qq 0 (buffer switch) and 2; B at qq 0 (switch), 1 and 3.
      const bool new_a = (qq == 0 || qq == 2);
      const bool new_b = (qq != 2);
      if (new_a) {
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          int row = wr * 128 + (ih * 4 + i) * 16 + (lane & 15);
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            af[i][ks] = read_frag(lds_a, rbuf, row, ks, lane);
        }
      }
      if (new_b) {
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          int row = wc * 64 + (jh * 2 + j) * 16 + (lane & 15);
#pragma unroll
          for (int ks = 0; ks < 2; ++ks)
            bfr[j][ks] = read_frag(lds_b, rbuf, row, ks, lane);
        }
      }

      // stage the next half-tile in sequence
      if (s < total_halves) {
        stage_seq(ga, gb, lda, ldb, lds_a, lds_b, s);
        ++s;
      }

      // counted drain at half-pair ends: certifies the buffer the NEXT
      // half-pair reads. vmcnt(6) steady state; vmcnt(0) once staging
      // is exhausted (last pair's phase 3).
      if (qq == 3 && (p < pairs - 1 || ph == 3)) {
        if (s >= total_halves) {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        } else {
          asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
        }
      }

      __builtin_amdgcn_s_barrier();
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
