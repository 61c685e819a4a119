// 256x256-tile, BK=128, 8-wave bf16 MFMA GEMM with a K-slice ring pipeline
// — the perf-tier kernel behind gemm / ag_gemm / gemm_rs (v1 128x128 in
// gemm.hip stays as the small-shape fallback).
//
// Design (derived from the CDNA4 guide's verified 256^2 8-phase template,
// re-architected as a K-slice ring so staging never overwrites live data):
//   * LDS: A[4 slices][256 rows][32 k] + B[same] bf16 = 128 KiB, single
//     buffer. Phase p (p = 0..3 per K-tile) computes MFMA K-slice p and —
//     after the barrier that retires all reads of slice p — stages slice p
//     of the NEXT K-tile into the same region. The ring gives a full
//     K-tile of prefetch with no double buffer.
//   * Counted vmcnt: 4 global_load_lds per thread per phase; steady state
//     16 in flight, `s_waitcnt vmcnt(12)` retires exactly the slice about
//     to be read (drain schedule 12/8/4/0 on the last tile).
//   * LDS swizzle: within a slice row (4 x 16B chunks), physical chunk =
//     logical ^ ((row>>1)&3) — 64 lanes of a ds_read_b128 hit 8 distinct
//     bank groups (2-way = free). Applied on the global SOURCE address at
//     stage time (global_load_lds writes linearly) and on the read address;
//     same involution both sides.
//   * s_setprio(1) around the 32-MFMA cluster (T5: pays off once phases
//     create wave role-split).
//   * 8 waves 2x4; per-wave output 128x64 = 8x4 fragments of 16x16x32.
//   * Epilogue reuses the full 128 KiB LDS as the C tile for 16B stores
//     (required for the RS variant's remote xGMI stores).
//
// Fused-comm variants carry the same semantics as gemm.hip's v1 kernels
// (reference behavior: Triton-distributed kernels/amd/allgather_gemm.py
// :552-660, gemm_reduce_scatter.py:128-227 — capability only).
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace g256 {

constexpr int BM = 256, BN = 256, BK = 128;
constexpr int NTH = 512;             // 8 waves
constexpr int SLICES = 4;            // K-slices of 32 per K-tile
constexpr int SLICE_K = 32;
constexpr int CH_ROW = SLICE_K / 8;  // 4 x 16B chunks per row per slice
// LDS elems per slice per matrix: 256*32
constexpr int SLICE_ELEMS = BM * SLICE_K;

TD_DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

TD_DEV int swz(int row, int j) { return j ^ ((row >> 1) & 3); }

// Stage one K-slice (A and B) of tile kt into the slice-p LDS region.
// 2 loads per thread per matrix. Global src is pre-swizzled.
TD_DEV void stage_slice(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                        int k0, bf16 *lds_a, bf16 *lds_b, int p) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int q = it * NTH + tid;          // chunk id 0..1023 within the slice
    int row = q >> 2;                // 4 chunks per row
    int jp = q & 3;                  // physical chunk in row
    int jg = swz(row, jp);           // source (logical) chunk
    const bf16 *sa = ga + (size_t)row * lda + k0 + jg * 8;
    const bf16 *sb = gb + (size_t)row * ldb + k0 + jg * 8;
    int wave_chunk0 = it * NTH + wave * 64;  // wave-uniform LDS base
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int *)sa,
        (__attribute__((address_space(3))) unsigned int *)(
            lds_a + p * SLICE_ELEMS + wave_chunk0 * 8),
        16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int *)sb,
        (__attribute__((address_space(3))) unsigned int *)(
            lds_b + p * SLICE_ELEMS + wave_chunk0 * 8),
        16, 0, 0);
  }
}

struct WaveCtx {
  int lane, wr, wc;
};

TD_DEV WaveCtx wave_ctx() {
  WaveCtx w;
  int tid = threadIdx.x;
  w.lane = tid & 63;
  int wave = tid >> 6;
  w.wr = wave >> 2;  // 0..1  (M halves of 128)
  w.wc = wave & 3;   // 0..3  (N quarters of 64)
  return w;
}

// The full pipelined K loop over `ktiles` tiles of BK.
//
// Staging is shifted one phase late relative to consumption: phase p of
// tile t stages INTO region p-1 (freed when phase p-1's MFMAs consumed it
// — each wave's lgkmcnt before its MFMA plus the barrier at phase-p entry
// prove all reads retired). Phase 0 stages region 3 with the CURRENT
// tile's last slice. One barrier per phase, no LDS-read drain before a
// second barrier; ds_read latency hides under the stage issue + barrier.
// Issue ledger: 4 loads/thread/phase, 12 in flight, vmcnt(8) steady
// (drain 8/8/4/0 on the last tile).
TD_DEV void kloop(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                  int ktiles, bf16 *lds_a, bf16 *lds_b, const WaveCtx &w,
                  f32x4 acc[8][4], int kt0 = 0) {
  // kt0: staggered K start (StaggerU, adopted from the hipBLASLt SK3
  // ISA study — profiles/hipblaslt_sk3_isa_study.md): the walk covers
  // tiles kt0, kt0+1, ... mod ktiles, so concurrent WGs hit different
  // HBM channels at any instant instead of all camping on k=0 together.
  // fp32 accumulation order changes; sum is reassociation-equivalent.
  // prologue: stage slices 0..2 of tile kt0 (slice 3 staged in phase 0)
#pragma unroll
  for (int p = 0; p < SLICES - 1; ++p)
    stage_slice(ga + (size_t)kt0 * BK, gb + (size_t)kt0 * BK, lda, ldb,
                p * SLICE_K, lds_a, lds_b, p);

  for (int t = 0; t < ktiles; ++t) {
    int tt = kt0 + t;
    if (tt >= ktiles) tt -= ktiles;
    const bool has_next = (t + 1) < ktiles;
    int tn = tt + 1 == ktiles ? 0 : tt + 1;
    const size_t knext = (size_t)tn * BK;
#pragma unroll
    for (int p = 0; p < SLICES; ++p) {
      // retire the loads for slice p of tile t
      if (has_next || p <= 1) {
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      } else {
        if (p == 2) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        if (p == 3) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();  // slice p landed; region p-1 free
      __builtin_amdgcn_sched_barrier(0);
      if (p == 0) {
        stage_slice(ga + (size_t)tt * BK, gb + (size_t)tt * BK, lda, ldb,
                    3 * SLICE_K, lds_a, lds_b, 3);
      } else if (has_next) {
        stage_slice(ga + knext, gb + knext, lda, ldb, (p - 1) * SLICE_K,
                    lds_a, lds_b, p - 1);
      }
      const int jn = w.lane >> 4;
      bf16x8 af[8], bf[4];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        int row = w.wr * 128 + i * 16 + (w.lane & 15);
        af[i] = *(const bf16x8 *)(lds_a + p * SLICE_ELEMS + row * SLICE_K +
                                  swz(row, jn) * 8);
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = w.wc * 64 + j * 16 + (w.lane & 15);
        bf[j] = *(const bf16x8 *)(lds_b + p * SLICE_ELEMS + row * SLICE_K +
                                  swz(row, jn) * 8);
      }
      // no explicit lgkmcnt: the ds_reads are compiler-visible loads, so
      // the backend inserts fine-grained lgkmcnt(N) per dependent MFMA
      // (guide: explicit lgkmcnt(0) serializes the whole read burst)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = mfma16(af[i], bf[j], acc[i][j]);
      __builtin_amdgcn_s_setprio(0);
    }
  }
}

// Epilogue: accumulators -> LDS C tile [256][256] bf16 -> 16B stores.
TD_DEV void epilogue(f32x4 acc[8][4], const WaveCtx &w, bf16 *lds_c,
                     bf16 *dst, int ldc) {
  __syncthreads();  // staging LDS is dead; reuse as C tile
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = w.wr * 128 + i * 16 + ((w.lane >> 4) * 4 + r);
        int col = w.wc * 64 + j * 16 + (w.lane & 15);
        lds_c[row * BN + col] = (bf16)acc[i][j][r];
      }
  __syncthreads();
  const int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 16; ++it) {
    int idx = it * NTH + tid;       // 8192 chunks of 16B
    int row = idx >> 5;             // 32 chunks per row
    int cc = idx & 31;
    *(ulonglong2 *)(dst + (size_t)row * ldc + cc * 8) =
        *(const ulonglong2 *)(lds_c + row * BN + cc * 8);
  }
}

}  // namespace g256

// ---------------------------------------------------------------------------
// Kernels
// ---------------------------------------------------------------------------
using namespace g256;

// GROUP_M supertile mapping: consecutive wgids (which the XCD remap makes
// contiguous per XCD) walk a GMx(tiles_n) band column-major, so an XCD's
// window reuses GM A-panels and a narrow set of B-panels from L2/LLC
// instead of streaming every B panel per tile row.
TD_DEV void tile_coords(int wgid, int tiles_m, int tiles_n, int &pid_m,
                        int &pid_n) {
  constexpr int GM = 4;
  int group = wgid / (GM * tiles_n);
  int first_m = group * GM;
  int gsz = min(tiles_m - first_m, GM);
  pid_m = first_m + (wgid % (GM * tiles_n)) % gsz;
  pid_n = (wgid % (GM * tiles_n)) / gsz;
}

__global__ __launch_bounds__(NTH, 2) void k_gemm256_bf16(GemmArgs args,
                                                          int stagger) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  const int tiles_n = args.n / BN;
  const int tiles_m = args.m / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m, pid_n;
  tile_coords(wgid, tiles_m, tiles_n, pid_m, pid_n);
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const bf16 *ga = (const bf16 *)args.a + (size_t)pid_m * BM * args.lda;
  const bf16 *gb = (const bf16 *)args.b + (size_t)pid_n * BN * args.ldb;
  const int ktiles = args.k / BK;
  const int kt0 = stagger ? wgid % ktiles : 0;
  kloop(ga, gb, args.lda, args.ldb, ktiles, lds_a, lds_b, w, acc, kt0);
  bf16 *dst = (bf16 *)args.c + (size_t)pid_m * BM * args.ldc + pid_n * BN;
  epilogue(acc, w, lds_a, dst, args.ldc);
}

__global__ __launch_bounds__(NTH, 2) void k_ag_gemm256_consumer_bf16(
    AgGemmArgs args) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  GemmArgs &g = args.g;
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  const int tiles_per_rank = args.m_per_rank / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m, pid_n;
  tile_coords(wgid, tiles_m, tiles_n, pid_m, pid_n);
  pid_m = (pid_m + args.rank * tiles_per_rank) % tiles_m;
  int rows_per_chunk = args.m_per_rank / args.chunks_per_rank;
  int c_lo = (pid_m * BM) / rows_per_chunk;
  int c_hi = (pid_m * BM + BM - 1) / rows_per_chunk;
  unsigned long long tw0 = wallclock();
  if (threadIdx.x < 64) {
    for (int c = c_lo + (int)threadIdx.x; c <= c_hi; c += 64)
      wait_ge_one<Scope::Sys>(args.flags + c, args.expect);
  }
  __syncthreads();
  unsigned long long tw1 = wallclock();
  kprof_record(args.prof, 0, tw0, tw1);
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const int seg = (pid_m * BM) / args.m_per_rank;
  const size_t arow0 =
      (size_t)seg * args.ws_stride + (pid_m * BM - seg * args.m_per_rank);
  const bf16 *ga = (const bf16 *)g.a + arow0 * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb;
  kloop(ga, gb, g.lda, g.ldb, g.k / BK, lds_a, lds_b, w, acc);
  bf16 *dst = (bf16 *)g.c + (size_t)pid_m * BM * g.ldc + pid_n * BN;
  epilogue(acc, w, lds_a, dst, g.ldc);
  kprof_record(args.prof, 1, tw1, wallclock());
}

__global__ __launch_bounds__(NTH, 2) void k_gemm256_rs_producer_bf16(
    GemmRsArgs args) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  GemmArgs &g = args.g;
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  const int tiles_per_rank = args.m_per_rank / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m, pid_n;
  tile_coords(wgid, tiles_m, tiles_n, pid_m, pid_n);
  pid_m = (pid_m + (args.rank + 1) * tiles_per_rank) % tiles_m;
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const bf16 *ga = (const bf16 *)g.a + (size_t)pid_m * BM * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb;
  kloop(ga, gb, g.lda, g.ldb, g.k / BK, lds_a, lds_b, w, acc);
  int owner = (pid_m * BM) / args.m_per_rank;
  int local_row0 = pid_m * BM - owner * args.m_per_rank;
  bf16 *seg = (bf16 *)((char *)args.pt.bases[owner] + args.scatter_off) +
              ((size_t)args.rank * args.ws_stride + local_row0) * g.n +
              pid_n * BN;
  epilogue(acc, w, lds_a, seg, g.n);
}

// ---------------------------------------------------------------------------
// Single-fused-kernel AG-GEMM (execution paradigm 2 of the reference,
// kernels/amd/allgather_gemm.py:662-870 — behavior only): the first
// `comm_wgs` workgroups push my shard's chunks to every peer's workspace
// segment (16B vector copies over xGMI) and release-signal per-chunk
// flags (sub-chunk WGs batch through a local arrive counter; the last
// one signals); the REMAINING workgroups run the standard flag-waiting
// persistent consumer GEMM. One launch, no comm streams, minimal launch
// overhead (decode-friendly, graph-capturable). Producers sit at LOW
// workgroup ids so the hardware dispatcher makes them resident before
// the consumers start spinning.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(NTH, 2) void k_ag_gemm256_fused(
    AgGemmArgs args, PeerTable pt, const bf16 *__restrict__ src,
    size_t ws_off, size_t flags_off, int *__restrict__ arrive,
    int comm_wgs, int subsplit) {
  const int world = pt.world;
  const int rank = pt.rank;
  const int chunks = args.chunks_per_rank;
  if ((int)blockIdx.x < comm_wgs) {
    // producer role: items = (peer order i, chunk c, subchunk u)
    const int items = world * chunks * subsplit;
    const size_t rows_per_chunk = args.m_per_rank / chunks;
    const size_t chunk_elems = rows_per_chunk * args.g.k;
    const size_t sub_elems = chunk_elems / subsplit;
    for (int it = blockIdx.x; it < items; it += comm_wgs) {
      const int u = it % subsplit;
      const int c = (it / subsplit) % chunks;
      const int i = it / (subsplit * chunks);
      const int peer = (rank + 1 + i) % world;  // i == world-1 -> self
      bf16 *seg = (bf16 *)((char *)pt.bases[peer] + ws_off) +
                  (size_t)rank * args.ws_stride * args.g.k;
      const bf16 *s0 = src + c * chunk_elems + u * sub_elems;
      bf16 *d0 = seg + c * chunk_elems + u * sub_elems;
      for (size_t e = (size_t)threadIdx.x * 8; e < sub_elems;
           e += (size_t)blockDim.x * 8)
        *(bf16x8 *)(d0 + e) = *(const bf16x8 *)(s0 + e);
      __syncthreads();
      if (threadIdx.x == 0) {
        int prev = atomic_add<Scope::Gpu>(arrive + i * chunks + c, 1);
        if (prev + 1 == subsplit) {
          fence_release_sys();
          int *fl = (int *)((char *)pt.bases[peer] + flags_off);
          st_release<Scope::Sys>(fl + rank * chunks + c, args.expect);
        }
      }
    }
    return;
  }
  // consumer role: standard per-tile flag-waiting persistent GEMM
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  GemmArgs &g = args.g;
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  const int tiles_per_rank = args.m_per_rank / BM;
  int wgid = xcd_remap(blockIdx.x - comm_wgs, tiles_m * tiles_n);
  int pid_m, pid_n;
  tile_coords(wgid, tiles_m, tiles_n, pid_m, pid_n);
  pid_m = (pid_m + args.rank * tiles_per_rank) % tiles_m;
  int rows_per_chunk = args.m_per_rank / args.chunks_per_rank;
  int c_lo = (pid_m * BM) / rows_per_chunk;
  int c_hi = (pid_m * BM + BM - 1) / rows_per_chunk;
  if (threadIdx.x < 64) {
    for (int c = c_lo + (int)threadIdx.x; c <= c_hi; c += 64)
      wait_ge_one<Scope::Sys>(args.flags + c, args.expect);
  }
  __syncthreads();
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const int seg = (pid_m * BM) / args.m_per_rank;
  const size_t arow0 =
      (size_t)seg * args.ws_stride + (pid_m * BM - seg * args.m_per_rank);
  const bf16 *ga = (const bf16 *)g.a + arow0 * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb;
  kloop(ga, gb, g.lda, g.ldb, g.k / BK, lds_a, lds_b, w, acc);
  bf16 *dst = (bf16 *)g.c + (size_t)pid_m * BM * g.ldc + pid_n * BN;
  epilogue(acc, w, lds_a, dst, g.ldc);
}

void launch_ag_gemm256_fused(const AgGemmArgs &args, const PeerTable &pt,
                             const void *src, size_t ws_off,
                             size_t flags_off, int *arrive, int comm_wgs,
                             int subsplit, hipStream_t stream) {
  int grid = comm_wgs + (args.g.m / BM) * (args.g.n / BN);
  hipLaunchKernelGGL(k_ag_gemm256_fused, dim3(grid), dim3(NTH), 0, stream,
                     args, pt, (const bf16 *)src, ws_off, flags_off,
                     arrive, comm_wgs, subsplit);
}

// ---------------------------------------------------------------------------
// Ulysses fused qkv-GEMM + head all-to-all (capability parity with the
// reference kernels/nvidia/sp_ulysess_qkv_gemm_all2all.py:64-545 —
// behavior only): the qkv projection's epilogue writes each 256^2 C tile
// straight into the OWNER rank's recv buffer over xGMI, where the owner
// is the rank whose head shard covers the tile's column block
// (peer_cols = qkv_dim / world, tile-aligned). Per-owner arrive counters
// (local, agent scope) release-signal the owner's per-src flag when my
// last tile for it lands — the attention consumer waits world flags.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(NTH, 2) void k_gemm256_colscatter(
    UlyssesQkvArgs args) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  GemmArgs &g = args.g;
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m, pid_n;
  tile_coords(wgid, tiles_m, tiles_n, pid_m, pid_n);
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const bf16 *ga = (const bf16 *)g.a + (size_t)pid_m * BM * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb;
  kloop(ga, gb, g.lda, g.ldb, g.k / BK, lds_a, lds_b, w, acc);
  const int owner = (pid_n * BN) / args.peer_cols;
  const int lcol = pid_n * BN - owner * args.peer_cols;
  bf16 *dst = (bf16 *)((char *)args.pt.bases[owner] + args.recv_off) +
              ((size_t)args.pt.rank * args.slot_rows + pid_m * BM) *
                  args.peer_cols +
              lcol;
  epilogue(acc, w, lds_a, dst, args.peer_cols);
  __syncthreads();
  if (threadIdx.x == 0) {
    int prev = atomic_add<Scope::Gpu>(args.arrive + owner, 1);
    if (prev + 1 == args.tiles_per_peer) {
      fence_release_sys();
      int *fl = (int *)((char *)args.pt.bases[owner] + args.flags_off);
      st_release<Scope::Sys>(fl + args.pt.rank, args.expect);
    }
  }
}

void launch_gemm256_colscatter(const UlyssesQkvArgs &args,
                               hipStream_t stream) {
  if (args.g.m % BM || args.g.n % BN || args.g.k % BK ||
      args.peer_cols % BN)
    throw std::runtime_error("colscatter: 256/128 alignment required");
  int grid = (args.g.m / BM) * (args.g.n / BN);
  hipLaunchKernelGGL(k_gemm256_colscatter, dim3(grid), dim3(NTH), 0,
                     stream, args);
}

// Accumulating GEMM: fp32 atomic adds into ws[M,N] (no memset, no
// convert) — the Ulysses o-GEMM sums per-source K-block partials into a
// shared ws as each source's a2a segment arrives.
__global__ __launch_bounds__(NTH, 2) void k_gemm256_acc(GemmArgs g,
                                                        float *ws) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m, pid_n;
  tile_coords(wgid, tiles_m, tiles_n, pid_m, pid_n);
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const bf16 *ga = (const bf16 *)g.a + (size_t)pid_m * BM * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb;
  kloop(ga, gb, g.lda, g.ldb, g.k / BK, lds_a, lds_b, w, acc);
  float *wsb = ws + (size_t)pid_m * BM * g.ldc + pid_n * BN;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = w.wr * 128 + i * 16 + ((w.lane >> 4) * 4 + r);
        int col = w.wc * 64 + j * 16 + (w.lane & 15);
        __hip_atomic_fetch_add(&wsb[(size_t)row * g.ldc + col],
                               acc[i][j][r], __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
      }
}

void launch_gemm256_acc_bf16(const GemmArgs &g, float *ws,
                             hipStream_t stream) {
  if (g.m % BM || g.n % BN || g.k % BK)
    throw std::runtime_error("gemm256_acc: 256/128 alignment required");
  int grid = (g.m / BM) * (g.n / BN);
  hipLaunchKernelGGL(k_gemm256_acc, dim3(grid), dim3(NTH), 0, stream, g,
                     ws);
}

__global__ void k_f32_to_bf16(const float *__restrict__ ws,
                              bf16 *__restrict__ c,
                              const bf16 *__restrict__ bias, int rows,
                              int n);

void launch_f32_to_bf16(const void *ws, void *c, const void *bias, int rows,
                        int n, hipStream_t stream) {
  int cgrid = rows < 2048 ? rows : 2048;
  hipLaunchKernelGGL(k_f32_to_bf16, dim3(cgrid), dim3(256), 0, stream,
                     (const float *)ws, (bf16 *)c, (const bf16 *)bias, rows,
                     n);
}

// Split-K decode tier: each workgroup runs the same K-slice-ring pipeline
// over a contiguous K/sk range and atomically accumulates its fp32 tile into
// ws[M,N] (LLC-resident for decode shapes: 512x5120 fp32 = 10.5 MB). A
// convert kernel then narrows ws -> bf16 C. Exists because M=512 decode
// shapes give the direct kernel a 40-80 wg grid (256 CUs idle); splitting K
// restores occupancy while keeping the proven per-CU schedule.
__global__ __launch_bounds__(NTH, 2) void k_gemm256_sk_bf16(GemmArgs g,
                                                            float *ws,
                                                            int sk,
                                                            int stagger) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  const int total = tiles_m * tiles_n * sk;
  int wgid = xcd_remap(blockIdx.x, total);
  const int tile = wgid / sk;
  const int sid = wgid % sk;
  int pid_m, pid_n;
  tile_coords(tile, tiles_m, tiles_n, pid_m, pid_n);
  const int ktiles_per = g.k / BK / sk;
  const int k0 = sid * ktiles_per * BK;
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const bf16 *ga = (const bf16 *)g.a + (size_t)pid_m * BM * g.lda + k0;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb + k0;
  const int kt0 = stagger ? wgid % ktiles_per : 0;
  kloop(ga, gb, g.lda, g.ldb, ktiles_per, lds_a, lds_b, w, acc, kt0);
  float *wsb = ws + (size_t)pid_m * BM * g.ldc + pid_n * BN;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = w.wr * 128 + i * 16 + ((w.lane >> 4) * 4 + r);
        int col = w.wc * 64 + j * 16 + (w.lane & 15);
        __hip_atomic_fetch_add(&wsb[(size_t)row * g.ldc + col], acc[i][j][r],
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      }
}

__global__ void k_f32_to_bf16(const float *__restrict__ ws,
                              bf16 *__restrict__ c,
                              const bf16 *__restrict__ bias, int rows,
                              int n) {
  typedef __attribute__((ext_vector_type(4))) float f4;
  typedef __attribute__((ext_vector_type(4))) bf16 b4;
  // row-block mapping (no per-element div/mod)
  for (int r = blockIdx.x; r < rows; r += gridDim.x) {
    const float *wr = ws + (size_t)r * n;
    bf16 *cr = c + (size_t)r * n;
    for (int col = threadIdx.x * 4; col < n; col += blockDim.x * 4) {
      f4 v = *(const f4 *)(wr + col);
      b4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float f = v[j];
        if (bias) f += (float)bias[col + j];
        o[j] = (bf16)f;
      }
      *(b4 *)(cr + col) = o;
    }
  }
}

// Two-stage split-K variant: each split block STORES its fp32 partial
// tile into a private ws slice [sk, M, N] (coalesced 64B-per-16-lane
// stores), and one fused reduce kernel sums the slices + bias +
// converts to bf16. Replaces the ~M*N*sk atomic RMWs of the atomic tier
// (down-proj: 42M atomics/call) with plain streaming stores + one read
// pass.
__global__ __launch_bounds__(NTH, 2) void k_gemm256_sk2_bf16(GemmArgs g,
                                                             float *ws,
                                                             int sk,
                                                             int stagger) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  const int total = tiles_m * tiles_n * sk;
  int wgid = xcd_remap(blockIdx.x, total);
  const int tile = wgid / sk;
  const int sid = wgid % sk;
  int pid_m, pid_n;
  tile_coords(tile, tiles_m, tiles_n, pid_m, pid_n);
  const int ktiles_per = g.k / BK / sk;
  const int k0 = sid * ktiles_per * BK;
  WaveCtx w = wave_ctx();
  f32x4 acc[8][4] = {};
  const bf16 *ga = (const bf16 *)g.a + (size_t)pid_m * BM * g.lda + k0;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb + k0;
  const int kt0 = stagger ? wgid % ktiles_per : 0;
  kloop(ga, gb, g.lda, g.ldb, ktiles_per, lds_a, lds_b, w, acc, kt0);
  float *wsb = ws + ((size_t)sid * g.m + (size_t)pid_m * BM) * g.n +
               (size_t)pid_n * BN;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = w.wr * 128 + i * 16 + ((w.lane >> 4) * 4 + r);
        int col = w.wc * 64 + j * 16 + (w.lane & 15);
        wsb[(size_t)row * g.n + col] = acc[i][j][r];
      }
}

__global__ void k_sk2_reduce(const float *__restrict__ ws,
                             bf16 *__restrict__ c,
                             const bf16 *__restrict__ bias, int rows,
                             int n, int sk) {
  typedef __attribute__((ext_vector_type(4))) float f4;
  typedef __attribute__((ext_vector_type(4))) bf16 b4;
  const size_t slice = (size_t)rows * n;
  for (int r = blockIdx.x; r < rows; r += gridDim.x) {
    const float *wr = ws + (size_t)r * n;
    bf16 *cr = c + (size_t)r * n;
    for (int col = threadIdx.x * 4; col < n; col += blockDim.x * 4) {
      f4 acc = *(const f4 *)(wr + col);
      for (int s2 = 1; s2 < sk; ++s2) {
        f4 v = *(const f4 *)(wr + s2 * slice + col);
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[j] += v[j];
      }
      b4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float f = acc[j];
        if (bias) f += (float)bias[col + j];
        o[j] = (bf16)f;
      }
      *(b4 *)(cr + col) = o;
    }
  }
}

void launch_gemm256_sk2_bf16(const GemmArgs &g, float *ws, int sk,
                             hipStream_t stream) {
  if (g.k % (BK * sk))
    throw std::runtime_error("gemm256_sk2: k % (128*sk) != 0");
  int grid = (g.m / BM) * (g.n / BN) * sk;
  // StaggerU measured NEGATIVE here (profiles/README.md r02 addendum):
  // per-WG K offsets destroy the GROUP_M L2 panel sharing. Default off;
  // TD_GEMM_STAGGER=1 reproduces the experiment.
  static const int stagger = [] {
    const char *e = getenv("TD_GEMM_STAGGER");
    return (e && e[0] == '1') ? 1 : 0;
  }();
  hipLaunchKernelGGL(k_gemm256_sk2_bf16, dim3(grid), dim3(NTH), 0, stream,
                     g, ws, sk, stagger);
  int cgrid = g.m < 2048 ? g.m : 2048;
  hipLaunchKernelGGL(k_sk2_reduce, dim3(cgrid), dim3(256), 0, stream, ws,
                     (bf16 *)g.c, (const bf16 *)g.bias, g.m, g.n, sk);
}

void launch_gemm256_sk_bf16(const GemmArgs &g, float *ws, int sk,
                            hipStream_t stream) {
  if (g.k % (BK * sk))
    throw std::runtime_error("gemm256_sk: k % (128*sk) != 0");
  size_t elems = (size_t)g.m * g.n;
  TD_CHECK_HIP(hipMemsetAsync(ws, 0, elems * sizeof(float), stream));
  int grid = (g.m / BM) * (g.n / BN) * sk;
  // StaggerU measured NEGATIVE here (profiles/README.md r02 addendum):
  // per-WG K offsets destroy the GROUP_M L2 panel sharing. Default off;
  // TD_GEMM_STAGGER=1 reproduces the experiment.
  static const int stagger = [] {
    const char *e = getenv("TD_GEMM_STAGGER");
    return (e && e[0] == '1') ? 1 : 0;
  }();
  hipLaunchKernelGGL(k_gemm256_sk_bf16, dim3(grid), dim3(NTH), 0, stream, g,
                     ws, sk, stagger);
  int cgrid = g.m < 2048 ? g.m : 2048;
  hipLaunchKernelGGL(k_f32_to_bf16, dim3(cgrid), dim3(256), 0, stream, ws,
                     (bf16 *)g.c, (const bf16 *)g.bias, g.m, g.n);
}

// ---------------------------------------------------------------------------
// Dispatch helpers (called from gemm.hip launchers)
// ---------------------------------------------------------------------------
bool gemm256_ok(int m, int n, int k) {
  return m % BM == 0 && n % BN == 0 && k % BK == 0;
}

void launch_gemm256_bf16(const GemmArgs &args, hipStream_t stream) {
  int grid = (args.m / BM) * (args.n / BN);
  // TD_GEMM_STAGGER=0 disables the StaggerU K walk (A/B)
  // StaggerU measured NEGATIVE here (profiles/README.md r02 addendum):
  // per-WG K offsets destroy the GROUP_M L2 panel sharing. Default off;
  // TD_GEMM_STAGGER=1 reproduces the experiment.
  static const int stagger = [] {
    const char *e = getenv("TD_GEMM_STAGGER");
    return (e && e[0] == '1') ? 1 : 0;
  }();
  hipLaunchKernelGGL(k_gemm256_bf16, dim3(grid), dim3(NTH), 0, stream, args,
                     stagger);
}

void launch_ag_gemm256_consumer_bf16(const AgGemmArgs &args,
                                     hipStream_t stream) {
  int grid = (args.g.m / BM) * (args.g.n / BN);
  hipLaunchKernelGGL(k_ag_gemm256_consumer_bf16, dim3(grid), dim3(NTH), 0,
                     stream, args);
}

void launch_gemm256_rs_producer_bf16(const GemmRsArgs &args,
                                     hipStream_t stream) {
  int grid = (args.g.m / BM) * (args.g.n / BN);
  hipLaunchKernelGGL(k_gemm256_rs_producer_bf16, dim3(grid), dim3(NTH), 0,
                     stream, args);
}

}  // namespace td
