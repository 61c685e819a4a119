// Persistent MFMA bf16 GEMM family for gfx950 + the fused-communication
// variants (AG-GEMM consumer with per-chunk flag waits, GEMM-RS producer
// with remote-scatter epilogue).
//
// Capability parity targets (behavior only, no code taken):
//   Triton-distributed python/triton_dist/kernels/amd/allgather_gemm.py
//     :552-660 (kernel_consumer_gemm_persistent — per-tile dl.wait on chunk
//     flags, XCD-aware pid remap, rank-staggered swizzle)
//   python/triton_dist/kernels/amd/gemm_reduce_scatter.py:128-284
//     (kernel_gemm_rs_producer_fuse_scatter + kernel_consumer_reduce)
//
// Design (MI355X-first, per the CDNA4 guide §5):
//   - 128x128 tile, BK=64, 4 waves (2x2), 16x16x32 bf16 MFMA, 4x4 fragments
//     per wave; A and B^T both [dim,K] row-major so fragment loads are
//     symmetric.
//   - global_load_lds dwordx4 staging (async HBM->LDS, no VGPR round trip).
//   - XCD-aware tile remap (8 XCDs, bijective).
//   - Epilogue stages C through LDS for 16B coalesced stores — required for
//     the RS variant (2B remote stores over xGMI would be unusable).
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// MFMA operand layout for v_mfma_f32_16x16x32_bf16.
// A (16x32): lane l, vreg j  ->  (m = l&15, k = klane(l>>4, j))
// B (32x16): lane l, vreg j  ->  (k = klane(l>>4, j), n = l&15)
// C/D:       lane l, vreg r  ->  (m = (l>>4)*4 + r, n = l&15)   [verified
//            mapping per MI355X_MICROARCH notes]
// Two klane candidates exist for the gfx950 2xK instructions; the probe
// kernel below selects at runtime; the GEMM uses TD_MFMA_KLANE.
//   0: k = (l>>4)*8 + j                (contiguous 8)
//   1: k = (l>>4)*4 + (j&3) + (j>>2)*16  (two stacked K halves)
#ifndef TD_MFMA_KLANE
#define TD_MFMA_KLANE 0
#endif

TD_DEV f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// ---------------------------------------------------------------------------
// Layout probe: one wave computes C[16][16] = A[16][32] @ B[32][16] with the
// given klane candidate; host compares with reference to pick TD_MFMA_KLANE.
// ---------------------------------------------------------------------------
__global__ void k_probe_mfma(const bf16 *A, const bf16 *B, float *C,
                             int klane_layout) {
  int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    int k = (klane_layout == 0) ? (l >> 4) * 8 + j
                                : (l >> 4) * 4 + (j & 3) + ((j >> 2) * 16);
    a[j] = A[(l & 15) * 32 + k];   // A row-major [16][32]
    b[j] = B[k * 16 + (l & 15)];   // B row-major [32][16]
  }
  f32x4 c = {0.f, 0.f, 0.f, 0.f};
  c = mfma16(a, b, c);
  for (int r = 0; r < 4; ++r) {
    C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
  }
}

void launch_probe_mfma(const void *a, const void *b, void *c, int layout,
                       hipStream_t stream) {
  hipLaunchKernelGGL(k_probe_mfma, dim3(1), dim3(64), 0, stream,
                     (const bf16 *)a, (const bf16 *)b, (float *)c, layout);
}

// ---------------------------------------------------------------------------
// Shared tile machinery
// ---------------------------------------------------------------------------
constexpr int BM = 128, BN = 128, BK = 64;
constexpr int NTHREADS = 256;  // 4 waves, 2x2
constexpr int LDS_A_ELEMS = BM * BK;          // bf16
constexpr int LDS_B_ELEMS = BN * BK;

struct TileCtx {
  int wave;    // 0..3
  int lane;    // 0..63
  int wr, wc;  // wave row/col in 2x2
};

TD_DEV TileCtx tile_ctx() {
  TileCtx t;
  int tid = threadIdx.x;
  t.wave = tid >> 6;
  t.lane = tid & 63;
  t.wr = t.wave >> 1;
  t.wc = t.wave & 1;
  return t;
}

// Stage one BMxBK A-tile and BNxBK B-tile into LDS with global_load_lds
// dwordx4. Both tiles are [dim][BK] row-major, 8 bf16 per 16B chunk,
// BK/8 = 8 chunks per row. 1024 chunks per tile, 4 per thread.
// Caller guarantees full tiles (M,N multiples of 128, K multiple of 64).
TD_DEV void stage_tile(const bf16 *ga, const bf16 *gb, int lda, int ldb,
                       bf16 *lds_a, bf16 *lds_b) {
  int tid = threadIdx.x;
  int wave = tid >> 6;
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int idx = it * NTHREADS + tid;            // chunk index 0..1023
    int row = idx >> 3;                       // BK/8 = 8 chunks/row
    int kc = idx & 7;
    const bf16 *src_a = ga + (size_t)row * lda + kc * 8;
    const bf16 *src_b = gb + (size_t)row * ldb + kc * 8;
    // wave-uniform LDS base: chunks idx&~63 .. for this wave
    int wave_chunk0 = it * NTHREADS + wave * 64;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int *)src_a,
        (__attribute__((address_space(3))) unsigned int *)(lds_a + wave_chunk0 * 8),
        16, 0, 0);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int *)src_b,
        (__attribute__((address_space(3))) unsigned int *)(lds_b + wave_chunk0 * 8),
        16, 0, 0);
  }
}

// Per-wave compute of one K-step from LDS: 2 k-subtiles of 32, 4x4 fragment
// grid. acc[4][4] of f32x4.
TD_DEV void compute_tile(const bf16 *lds_a, const bf16 *lds_b,
                         const TileCtx &t, f32x4 acc[4][4]) {
#pragma unroll
  for (int ks = 0; ks < BK / 32; ++ks) {
    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int arow = t.wr * 64 + i * 16 + (t.lane & 15);
      int brow = t.wc * 64 + i * 16 + (t.lane & 15);
#if TD_MFMA_KLANE == 0
      int k0 = ks * 32 + (t.lane >> 4) * 8;
      afrag[i] = *(const bf16x8 *)(lds_a + arow * BK + k0);
      bfrag[i] = *(const bf16x8 *)(lds_b + brow * BK + k0);
#else
      int k0 = ks * 32 + (t.lane >> 4) * 4;
      typedef __attribute__((ext_vector_type(4))) bf16 bf16x4;
      bf16x4 alo = *(const bf16x4 *)(lds_a + arow * BK + k0);
      bf16x4 ahi = *(const bf16x4 *)(lds_a + arow * BK + k0 + 16);
      bf16x4 blo = *(const bf16x4 *)(lds_b + brow * BK + k0);
      bf16x4 bhi = *(const bf16x4 *)(lds_b + brow * BK + k0 + 16);
      for (int j = 0; j < 4; ++j) {
        afrag[i][j] = alo[j]; afrag[i][j + 4] = ahi[j];
        bfrag[i][j] = blo[j]; bfrag[i][j + 4] = bhi[j];
      }
#endif
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = mfma16(afrag[i], bfrag[j], acc[i][j]);
  }
}

// Epilogue: acc -> LDS bf16 [BM][BN] -> vectorized 16B stores to `dst`
// (local or xGMI-remote), optional bias add.
TD_DEV void epilogue_store(f32x4 acc[4][4], const TileCtx &t, bf16 *lds_c,
                           bf16 *dst, int ldc, const bf16 *bias, int bias_col0) {
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = t.wr * 64 + i * 16 + (t.lane >> 4) * 4 + r;
        int col = t.wc * 64 + j * 16 + (t.lane & 15);
        float v = acc[i][j][r];
        if (bias) v += (float)bias[bias_col0 + col];
        lds_c[row * BN + col] = (bf16)v;
      }
  __syncthreads();
  // 128x128 bf16 = 32 KiB = 2048 x 16B chunks; 8 per thread.
  int tid = threadIdx.x;
#pragma unroll
  for (int it = 0; it < 8; ++it) {
    int idx = it * NTHREADS + tid;
    int row = idx >> 4;  // BN*2/16 = 16 chunks per row
    int cc = idx & 15;
    *(ulonglong2 *)(dst + (size_t)row * ldc + cc * 8) =
        *(const ulonglong2 *)(lds_c + row * BN + cc * 8);
  }
}

// ---------------------------------------------------------------------------
// Plain GEMM: C[M,N] = A[M,K] @ B[N,K]^T (+bias)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(NTHREADS) void k_gemm_bf16(GemmArgs args) {
  __shared__ bf16 lds_a[LDS_A_ELEMS];
  __shared__ bf16 lds_b[LDS_B_ELEMS];
  const int tiles_n = args.n / BN;
  const int tiles_m = args.m / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m = wgid / tiles_n, pid_n = wgid % tiles_n;
  TileCtx t = tile_ctx();
  f32x4 acc[4][4] = {};
  const bf16 *ga = (const bf16 *)args.a + (size_t)pid_m * BM * args.lda;
  const bf16 *gb = (const bf16 *)args.b + (size_t)pid_n * BN * args.ldb;
  for (int k0 = 0; k0 < args.k; k0 += BK) {
    stage_tile(ga + k0, gb + k0, args.lda, args.ldb, lds_a, lds_b);
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    compute_tile(lds_a, lds_b, t, acc);
    __syncthreads();
  }
  bf16 *lds_c = lds_a;  // reuse staging LDS for the epilogue
  bf16 *dst = (bf16 *)args.c + (size_t)pid_m * BM * args.ldc + pid_n * BN;
  epilogue_store(acc, t, lds_c, dst, args.ldc, (const bf16 *)args.bias,
                 pid_n * BN);
}

// gemm256.hip perf tier (256^2/BK128 K-slice ring); these fall back to the
// 128^2 kernel for shapes that don't tile by 256.
bool gemm256_ok(int m, int n, int k);
// the 256-tier runs 1 block/CU (128 KiB LDS): it needs >= ~224 workgroups
// to fill the chip; below that the 128-tier's 4x denser grid wins
static inline bool gemm256_fills(int m, int n) {
  return (m / 256) * (n / 256) >= 224;
}
void launch_gemm256_bf16(const GemmArgs &args, hipStream_t stream);
void launch_ag_gemm256_consumer_bf16(const AgGemmArgs &args,
                                     hipStream_t stream);
void launch_gemm256_rs_producer_bf16(const GemmRsArgs &args,
                                     hipStream_t stream);

void launch_gemm_bf16(const GemmArgs &args, hipStream_t stream) {
  if (gemm256_ok(args.m, args.n, args.k) && !args.bias &&
      gemm256_fills(args.m, args.n)) {
    launch_gemm256_bf16(args, stream);
    return;
  }
  if (args.m % BM || args.n % BN || args.k % BK)
    throw std::runtime_error("gemm_bf16: M%128/N%128/K%64 must be 0");
  int grid = (args.m / BM) * (args.n / BN);
  hipLaunchKernelGGL(k_gemm_bf16, dim3(grid), dim3(NTHREADS), 0, stream, args);
}

// ---------------------------------------------------------------------------
// AG-GEMM consumer: A is the gathered [world*m_per_rank, K] symmetric
// workspace; chunk c of rank r is ready when flags[r*chunks_per_rank+c] >=
// expect. Tiles are visited rank-staggered (this rank's rows first).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(NTHREADS) void k_ag_gemm_consumer_bf16(
    AgGemmArgs args) {
  __shared__ bf16 lds_a[LDS_A_ELEMS];
  __shared__ bf16 lds_b[LDS_B_ELEMS];
  GemmArgs &g = args.g;
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  const int tiles_per_rank = args.m_per_rank / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m = wgid / tiles_n, pid_n = wgid % tiles_n;
  // rank-stagger: start at own shard so first tiles need only local data
  pid_m = (pid_m + args.rank * tiles_per_rank) % tiles_m;

  // wait for the chunks covering rows [pid_m*BM, pid_m*BM+BM)
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
  kprof_record(args.prof, 0 /*tile-wait*/, tw0, tw1);

  TileCtx t = tile_ctx();
  f32x4 acc[4][4] = {};
  // segment-strided workspace: segment s of the gathered A lives at row
  // s * ws_stride (allocation) but logical row s * m_per_rank
  const int seg = (pid_m * BM) / args.m_per_rank;
  const size_t arow0 =
      (size_t)seg * args.ws_stride + (pid_m * BM - seg * args.m_per_rank);
  const bf16 *ga = (const bf16 *)g.a + arow0 * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb;
  for (int k0 = 0; k0 < g.k; k0 += BK) {
    stage_tile(ga + k0, gb + k0, g.lda, g.ldb, lds_a, lds_b);
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    compute_tile(lds_a, lds_b, t, acc);
    __syncthreads();
  }
  bf16 *dst = (bf16 *)g.c + (size_t)pid_m * BM * g.ldc + pid_n * BN;
  epilogue_store(acc, t, lds_a, dst, g.ldc, nullptr, 0);
  kprof_record(args.prof, 1 /*tile-compute*/, tw1, wallclock());
}

void launch_ag_gemm_consumer_bf16(const AgGemmArgs &args, hipStream_t stream) {
  const GemmArgs &g = args.g;
  if (g.m % BM || g.n % BN || g.k % BK || args.m_per_rank % BM)
    throw std::runtime_error("ag_gemm: shape must tile by 128/128/64");
  if (args.m_per_rank % args.chunks_per_rank)
    throw std::runtime_error("ag_gemm: chunks_per_rank must divide m_per_rank");
  if (gemm256_ok(g.m, g.n, g.k) && args.m_per_rank % 256 == 0 &&
      gemm256_fills(g.m, g.n)) {
    launch_ag_gemm256_consumer_bf16(args, stream);
    return;
  }
  int grid = (g.m / BM) * (g.n / BN);
  hipLaunchKernelGGL(k_ag_gemm_consumer_bf16, dim3(grid), dim3(NTHREADS), 0,
                     stream, args);
}

// ---------------------------------------------------------------------------
// GEMM-RS producer: computes partial C[M,N] (M = world*m_per_rank) and
// scatters each 128-row tile into the owner rank's symmetric scatter buffer
// segment for this rank: owner_buf[src=rank][local_row][N]. Rank-staggered
// pid_m so ranks hit different owners first (spreads xGMI load).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(NTHREADS) void k_gemm_rs_producer_bf16(
    GemmRsArgs args) {
  __shared__ bf16 lds_a[LDS_A_ELEMS];
  __shared__ bf16 lds_b[LDS_B_ELEMS];
  GemmArgs &g = args.g;
  const int tiles_n = g.n / BN;
  const int tiles_m = g.m / BM;
  const int tiles_per_rank = args.m_per_rank / BM;
  int wgid = xcd_remap(blockIdx.x, tiles_m * tiles_n);
  int pid_m = wgid / tiles_n, pid_n = wgid % tiles_n;
  // start at the shard owned by rank+1 (own shard last: it needs no wire)
  pid_m = (pid_m + (args.rank + 1) * tiles_per_rank) % tiles_m;

  TileCtx t = tile_ctx();
  f32x4 acc[4][4] = {};
  const bf16 *ga = (const bf16 *)g.a + (size_t)pid_m * BM * g.lda;
  const bf16 *gb = (const bf16 *)g.b + (size_t)pid_n * BN * g.ldb;
  for (int k0 = 0; k0 < g.k; k0 += BK) {
    stage_tile(ga + k0, gb + k0, g.lda, g.ldb, lds_a, lds_b);
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    compute_tile(lds_a, lds_b, t, acc);
    __syncthreads();
  }
  int owner = (pid_m * BM) / args.m_per_rank;
  int local_row0 = pid_m * BM - owner * args.m_per_rank;
  bf16 *seg = (bf16 *)((char *)args.pt.bases[owner] + args.scatter_off) +
              ((size_t)args.rank * args.ws_stride + local_row0) * g.n + pid_n * BN;
  epilogue_store(acc, t, lds_a, seg, g.n, nullptr, 0);
}

void launch_gemm_rs_producer_bf16(const GemmRsArgs &args, hipStream_t stream) {
  const GemmArgs &g = args.g;
  if (g.m % BM || g.n % BN || g.k % BK || args.m_per_rank % BM)
    throw std::runtime_error("gemm_rs: shape must tile by 128/128/64");
  if (gemm256_ok(g.m, g.n, g.k) && args.m_per_rank % 256 == 0 &&
      gemm256_fills(g.m, g.n)) {
    launch_gemm256_rs_producer_bf16(args, stream);
    return;
  }
  int grid = (g.m / BM) * (g.n / BN);
  hipLaunchKernelGGL(k_gemm_rs_producer_bf16, dim3(grid), dim3(NTHREADS), 0,
                     stream, args);
}

// ---------------------------------------------------------------------------
// RS reduce: out[m_per_rank, n] = sum over world segments, ring order
// starting at rank+1 (cf. gemm_reduce_scatter.py:230-284 semantics).
// ---------------------------------------------------------------------------
__global__ void k_rs_reduce_bf16(const bf16 *segments, bf16 *out, int world,
                                 int rank, size_t elems, size_t seg_stride) {
  size_t i = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  size_t stride = (size_t)gridDim.x * blockDim.x * 8;
  for (; i < elems; i += stride) {
    float acc[8] = {};
    for (int s = 0; s < world; ++s) {
      int r = (rank + 1 + s) % world;
      bf16x8 v = *(const bf16x8 *)(segments + (size_t)r * seg_stride + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += (float)v[j];
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (bf16)acc[j];
    *(bf16x8 *)(out + i) = o;
  }
}

void launch_rs_reduce_bf16(const void *segments, void *out, int world,
                           int rank, int m_per_rank, int ws_stride, int n,
                           hipStream_t stream) {
  size_t elems = (size_t)m_per_rank * n;
  if (elems % 8) throw std::runtime_error("rs_reduce: elems % 8 != 0");
  size_t work = elems / 8;
  int blocks = (int)((work + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(k_rs_reduce_bf16, dim3(blocks), dim3(256), 0, stream,
                     (const bf16 *)segments, (bf16 *)out, world, rank, elems,
                     (size_t)ws_stride * n);
}

}  // namespace td
