// Tile-granular fused GEMM + AllReduce — the gemm_ar TP mode's core op.
//
// Capability parity (behavior only) with the reference's overlapped
// GEMM+AR pair (Triton-distributed kernels/amd/gemm_allreduce.py:104-200:
// persistent GEMM release-signals per (tile, rank); a consumer AR kernel
// on the comm stream reduces tiles as they land and broadcasts).
//
// MI355X-first redesign:
//   * Tile ownership is round-robin BY TILE (owner = linear_tile % world),
//     not by contiguous row block — any M works (no m_per_rank
//     divisibility), and because adjacent pid_n cycle through owners, the
//     producer's visit order spreads xGMI traffic over all 7 links from
//     the first tile on.
//   * Producer = the proven gemm256 K-slice-ring kernel whose epilogue
//     stores the C tile straight into the OWNER's symmetric scatter slot
//     (16B stores over xGMI, staged through the LDS C tile) and then
//     bumps the owner's per-tile arrive counter (system-scope release).
//     A split-K variant covers decode shapes: fp32 atomics into a local
//     ws, per-tile done counters, and the LAST split block converts +
//     pushes + arrives, so per-tile signalling survives the K split.
//   * Consumer (launched on the comm stream): one workgroup per OWNED
//     tile; waits arrive[slot] == world, reduces the world contributions
//     (bf16 -> fp32 -> bf16), writes the reduced tile into EVERY rank's
//     symmetric out buffer (xGMI broadcast), release-signals the per-tile
//     out flag on every rank. Tiles reduce while the producer is still
//     computing later tiles — the AR cost hides under the GEMM tail.
//   * Compute stream finally waits all out flags (wait_eq) — including
//     tiles owned by peers, whose consumers signal us remotely.
//
// Traffic per rank: C*(W-1)/W contribution push + C*(W-1)/W broadcast —
// the two-shot (bandwidth-optimal) schedule, at tile granularity.
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// The producer replicates gemm256.hip's K-slice-ring kloop locally
// (device inline internals are TU-local; the shared contract is locked
// by the CPU mirror tests in tests/test_mappings_cpu.py).
namespace gar {

constexpr int BM = 256, BN = 256, BK = 128;
constexpr int NTH = 512;
constexpr int SLICES = 4;
constexpr int SLICE_ELEMS = BM * (BK / SLICES);

TD_DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

TD_DEV int swz(int row, int j) { return j ^ ((row >> 1) & 3); }

constexpr int SLICE_K = 32;
constexpr int CH_ROW = SLICE_K / 8;

TD_DEV void stage_slice(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                        int k0, bf16 *lds_a, bf16 *lds_b, int p) {
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int q = it * NTH + tid;
    int row = q >> 2;
    int jp = q & 3;
    int jg = swz(row, jp);
    const bf16 *sa = ga + (size_t)row * lda + k0 + jg * 8;
    const bf16 *sb = gb + (size_t)row * ldb + k0 + jg * 8;
    int wave_chunk0 = it * NTH + wave * 64;
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
  w.wr = wave >> 2;
  w.wc = wave & 3;
  return w;
}

TD_DEV void kloop(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                  int ktiles, bf16 *lds_a, bf16 *lds_b, const WaveCtx &w,
                  f32x4 acc[8][4]) {
#pragma unroll
  for (int p = 0; p < SLICES - 1; ++p)
    stage_slice(ga, gb, lda, ldb, p * SLICE_K, lds_a, lds_b, p);

  for (int t = 0; t < ktiles; ++t) {
    const bool has_next = (t + 1) < ktiles;
    const int knext = (t + 1) * BK;
#pragma unroll
    for (int p = 0; p < SLICES; ++p) {
      if (has_next || p <= 1) {
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
      } else {
        if (p == 2) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        if (p == 3) asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_sched_barrier(0);
      if (p == 0) {
        stage_slice(ga + t * BK, gb + t * BK, lda, ldb, 3 * SLICE_K, lds_a,
                    lds_b, 3);
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

TD_DEV void epilogue_to(f32x4 acc[8][4], const WaveCtx &w, bf16 *lds_c,
                        bf16 *dst, int ldc) {
  __syncthreads();
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
    int idx = it * NTH + tid;
    int row = idx >> 5;
    int cc = idx & 31;
    *(ulonglong2 *)(dst + (size_t)row * ldc + cc * 8) =
        *(const ulonglong2 *)(lds_c + row * BN + cc * 8);
  }
}

TD_DEV void tile_coords(int wgid, int tiles_m, int tiles_n, int &pid_m,
                        int &pid_n) {
  constexpr int GM = 4;
  int group = wgid / (GM * tiles_n);
  int first_m = group * GM;
  int gsz = min(tiles_m - first_m, GM);
  pid_m = first_m + (wgid % (GM * tiles_n)) % gsz;
  pid_n = (wgid % (GM * tiles_n)) / gsz;
}

}  // namespace gar

using namespace gar;

// ---------------------------------------------------------------------------
// Producer: gemm256 whose epilogue pushes the tile to its owner + arrives.
// scatter layout on each rank: [world_src][slots][BM*BN] bf16.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(gar::NTH, 2) void k_gemm256_ar_producer(
    GemmArArgs args) {
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
  const int lt = pid_m * tiles_n + pid_n;
  const int owner = lt % args.pt.world;
  const int slot = lt / args.pt.world;
  bf16 *dst = (bf16 *)((char *)args.pt.bases[owner] + args.scatter_off) +
              ((size_t)args.pt.rank * args.slots + slot) * (BM * BN);
  epilogue_to(acc, w, lds_a, dst, BN);
  __syncthreads();
  if (threadIdx.x == 0) {
    fence_release_sys();
    int *arrive = (int *)((char *)args.pt.bases[owner] + args.arrive_off);
    atomic_add<Scope::Sys>(arrive + slot, 1);
  }
}

// Split-K producer for occupancy-starved (decode) shapes: grid =
// tiles * sk; fp32 atomic accumulation into ws[M,N]; per-tile done
// counters; the LAST split block of each tile converts the fp32 tile to
// bf16 into the owner's scatter slot and arrives.
__global__ __launch_bounds__(gar::NTH, 2) void k_gemm256_sk_ar_producer(
    GemmArArgs args, float *ws, int *done, int sk) {
  __shared__ bf16 lds_a[SLICES * SLICE_ELEMS];
  __shared__ bf16 lds_b[SLICES * SLICE_ELEMS];
  GemmArgs &g = args.g;
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
  kloop(ga, gb, g.lda, g.ldb, ktiles_per, lds_a, lds_b, w, acc);
  float *wsb = ws + (size_t)pid_m * BM * g.n + (size_t)pid_n * BN;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = w.wr * 128 + i * 16 + ((w.lane >> 4) * 4 + r);
        int col = w.wc * 64 + j * 16 + (w.lane & 15);
        __hip_atomic_fetch_add(&wsb[(size_t)row * g.n + col], acc[i][j][r],
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
      }
  // last split block of this tile converts + pushes + arrives
  __syncthreads();
  __shared__ int is_last;
  if (threadIdx.x == 0) {
    __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
    int prev = __hip_atomic_fetch_add(done + tile, 1, __ATOMIC_ACQ_REL,
                                      __HIP_MEMORY_SCOPE_AGENT);
    is_last = (prev == sk - 1);
  }
  __syncthreads();
  if (!is_last) return;
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
  const int lt = pid_m * tiles_n + pid_n;
  const int owner = lt % args.pt.world;
  const int slot = lt / args.pt.world;
  bf16 *dst = (bf16 *)((char *)args.pt.bases[owner] + args.scatter_off) +
              ((size_t)args.pt.rank * args.slots + slot) * (BM * BN);
  const int tid = threadIdx.x;
  typedef __attribute__((ext_vector_type(4))) float f4;
  typedef __attribute__((ext_vector_type(4))) bf16 b4;
  for (int idx = tid * 4; idx < BM * BN; idx += NTH * 4) {
    int row = idx / BN, col = idx % BN;
    f4 v = *(const f4 *)(wsb + (size_t)row * g.n + col);
    b4 o;
#pragma unroll
    for (int e = 0; e < 4; ++e) o[e] = (bf16)v[e];
    *(b4 *)(dst + (size_t)row * BN + col) = o;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    fence_release_sys();
    int *arrive = (int *)((char *)args.pt.bases[owner] + args.arrive_off);
    atomic_add<Scope::Sys>(arrive + slot, 1);
  }
}

// ---------------------------------------------------------------------------
// Consumer (comm stream): one WG per owned tile — wait world arrivals,
// reduce, broadcast to every rank's out, signal per-tile flags everywhere.
// ---------------------------------------------------------------------------
__global__ void k_ar_tile_consumer(GemmArArgs args) {
  GemmArgs &g = args.g;
  const int tiles_n = g.n / BN;
  const int world = args.pt.world;
  const int rank = args.pt.rank;
  const int lt = rank + (int)blockIdx.x * world;  // my owned global tile
  const int slot = (int)blockIdx.x;
  const int pid_m = lt / tiles_n, pid_n = lt % tiles_n;
  int *arrive = (int *)((char *)args.pt.bases[rank] + args.arrive_off);
  if (threadIdx.x == 0) wait_ge_one<Scope::Sys>(arrive + slot, world);
  __syncthreads();
  fence_acquire_sys();
  const bf16 *scat =
      (const bf16 *)((char *)args.pt.bases[rank] + args.scatter_off);
  const size_t out_tile_off =
      (size_t)pid_m * BM * g.n + (size_t)pid_n * BN;
  for (int idx = threadIdx.x * 8; idx < BM * BN; idx += blockDim.x * 8) {
    int row = idx / BN, col = idx % BN;
    float acc8[8];
    bf16x8 v = *(const bf16x8 *)(scat + (size_t)slot * (BM * BN) + idx);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc8[e] = (float)v[e];
    for (int s = 1; s < world; ++s) {
      bf16x8 u = *(const bf16x8 *)(
          scat + ((size_t)s * args.slots + slot) * (BM * BN) + idx);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc8[e] += (float)u[e];
    }
    bf16x8 o;
#pragma unroll
    for (int e = 0; e < 8; ++e) o[e] = (bf16)acc8[e];
    for (int p = 0; p < world; ++p) {
      bf16 *ob = (bf16 *)((char *)args.pt.bases[p] + args.out_off);
      *(bf16x8 *)(ob + out_tile_off + (size_t)row * g.n + col) = o;
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    fence_release_sys();
    for (int p = 0; p < world; ++p) {
      int *fl = (int *)((char *)args.pt.bases[p] + args.oflags_off);
      st_release<Scope::Sys>(fl + lt, 1);
    }
  }
}

// NOTE: the consumer reads scatter contributions placed by SRC rank s at
// [s][slot]; contribution s==rank came from the local producer. The
// first slot read (s = rank-local index 0) starts at the local rank's own
// contribution only when rank == 0; the loop above reads s=0 first via
// `v` — it is simply src 0, then 1..world-1 skipping nothing. (Order is
// irrelevant for a sum.)

void launch_gemm256_ar_producer(const GemmArArgs &args, hipStream_t stream) {
  int grid = (args.g.m / BM) * (args.g.n / BN);
  hipLaunchKernelGGL(k_gemm256_ar_producer, dim3(grid), dim3(gar::NTH), 0,
                     stream, args);
}

void launch_gemm256_sk_ar_producer(const GemmArArgs &args, float *ws,
                                   int *done, int sk, hipStream_t stream) {
  if (args.g.k % (BK * sk))
    throw std::runtime_error("gemm_ar_sk: k % (128*sk) != 0");
  size_t elems = (size_t)args.g.m * args.g.n;
  TD_CHECK_HIP(hipMemsetAsync(ws, 0, elems * sizeof(float), stream));
  int tiles = (args.g.m / BM) * (args.g.n / BN);
  TD_CHECK_HIP(hipMemsetAsync(done, 0, tiles * sizeof(int), stream));
  hipLaunchKernelGGL(k_gemm256_sk_ar_producer, dim3(tiles * sk),
                     dim3(gar::NTH), 0, stream, args, ws, done, sk);
}

void launch_ar_tile_consumer(const GemmArArgs &args, int n_owned,
                             hipStream_t stream) {
  if (n_owned <= 0) return;
  hipLaunchKernelGGL(k_ar_tile_consumer, dim3(n_owned), dim3(512), 0,
                     stream, args);
}

}  // namespace td
