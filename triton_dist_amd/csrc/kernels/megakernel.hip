// Megakernel: one persistent kernel executes a whole decode step as a task
// graph — per-workgroup task queues, op-level scoreboard dependencies, and
// a dispatch switch over a fixed CDNA4 task vocabulary.
//
// Capability parity with the reference megakernel subsystem
// (Triton-distributed mega_triton_kernel/core/{builder.py:34,
// scheduler.py:31-157, code_generator.py:67-280} — behavior only). The
// reference CODEGENS a Triton kernel per model; here the task vocabulary
// is a fixed set of hand-written HIP device functions (RMSNorm,
// add+RMSNorm, GEMM tile, SwiGLU, qkv-prologue, flash-decode, embed,
// copy), so no JIT is needed — the model builder only emits descriptors.
//
// Scheduling contract (deadlock freedom): tasks are assigned to queues in
// TOPOLOGICAL-LEVEL order, round-robin across workgroups. Every queue is
// executed in order, so when a workgroup blocks on a level-L dependency,
// every level<L task is either done or ahead of all blocked tasks in some
// queue — progress is guaranteed. Scoreboard slots are per-op arrive
// counters (system-scope not needed: single GPU, device scope).
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

namespace mk {

enum TaskType : int {
  T_RMSNORM = 0,      // a0=x, a1=w, a2=out, a3=rows, a4=cols, a5=row0
  T_ADD_RMSNORM = 1,  // a0=delta, a1=resid_in, a2=resid_out, a3=w, a4=out,
                      // a5=rows, a6=cols, a7=row0
  T_GEMM_TILE = 2,    // a0=A, a1=B(NxK), a2=C, a3=m(valid rows), a4=n,
                      // a5=k, a6=pid_m, a7=pid_n
  T_SWIGLU = 3,       // a0=h, a1=out, a2=rows, a3=inter, a4=chunk, a5=nchunks
  T_QKV_PROLOGUE = 4, // a0=qkv, a1=qout, a2=kcache, a3=vcache, a4=cos,
                      // a5=sin, a6=qnw, a7=knw, a8=offset(i64 cell),
                      // a9=b, a10=qh, a11=kvh, a12=maxlen
  T_FLASH_DECODE = 5, // a0=q, a1=kc, a2=vc, a3=out, a4=offset, a5=b,
                      // a6=kh, a7=qh, a8=kvh, a9=maxlen
  T_EMBED = 6,        // a0=tokens(i64), a1=table, a2=out, a3=rows, a4=cols,
                      // a5=row0
  T_KV_ADVANCE = 7,   // a0=offset cell (i64): += 1
  T_GEMM_TILE_PART = 8,  // a0=A, a1=B, a2=ws(f32), a3=m, a4=n, a5=k,
                         // a6=pid_m, a7=pid_n, a8=k0, a9=klen, a10=sk
  T_TILE_REDUCE = 9,     // a0=ws, a1=C, a2=m, a3=n, a4=pid_m, a5=pid_n,
                         // a6=ksplit
  T_GEMV = 10,           // a0=A[m,K], a1=W[N,K], a2=C[m,N], a3=m, a4=n,
                         // a5=k, a6=col0, a7=ncols — bsz<=4 decode GEMV:
                         // one task = a 512-col chunk x full K (weight-
                         // bandwidth-bound; ~25x fewer tasks than the
                         // 32x128 tile scheme at bsz 1, which is what the
                         // 0.25 us/task dispatch overhead demands)
  T_PRO_FLASH_DECODE = 11,  // hop fusion (TD_MK_FUSE): qkv prologue for
                            // ONE (b, kh)'s heads + flash decode in one
                            // task — removes the 1-task-per-row prologue
                            // serialization hop. a0=qkv, a1=q, a2=kc,
                            // a3=vc, a4=cos, a5=sin, a6=qnw, a7=knw,
                            // a8=offset, a9=attn_out, a10=pack(b,kh),
                            // a11=pack(qh,kvh), a12=maxlen
  T_GEMM_TILE_PART_NR = 12,  // hop fusion: K-range GEMM partial whose A
                             // operand is rmsnorm(x [+ res]) computed on
                             // the fly — removes the [add+]rmsnorm hop
                             // before each projection. Args of
                             // T_GEMM_TILE_PART with a0=x, a11=ln_w,
                             // a12=res (0 = none)
};

struct Task {
  int type;
  int score_slot;     // arrive counter this task bumps when done
  int dep0, dep0_n;   // wait scoreboard[dep0] >= dep0_n   (-1 = none)
  int dep1, dep1_n;
  long long a[13];    // pointers / ints (pointers pre-resolved on host)
};

constexpr int NTH = 256;

// ---------------------------------------------------------------------------
// task bodies (256 threads each)
// ---------------------------------------------------------------------------
TD_DEV void t_rmsnorm(const Task &t, bool add) {
  // one task = one row (grid of row tasks emitted by the builder)
  const bf16 *x = (const bf16 *)t.a[0];
  const int cols = (int)t.a[add ? 6 : 4];
  const int row = (int)t.a[add ? 7 : 5];
  const bf16 *xr = x + (size_t)row * cols;
  const bf16 *rr = nullptr;
  bf16 *ro = nullptr, *orow;
  const bf16 *w;
  if (add) {
    rr = (const bf16 *)t.a[1] + (size_t)row * cols;
    ro = (bf16 *)t.a[2] + (size_t)row * cols;
    w = (const bf16 *)t.a[3];
    orow = (bf16 *)t.a[4] + (size_t)row * cols;
  } else {
    w = (const bf16 *)t.a[1];
    orow = (bf16 *)t.a[2] + (size_t)row * cols;
  }
  __shared__ float red[16];
  float ss = 0.f;
  const int nv = cols / 8;
  for (int i = threadIdx.x; i < nv; i += NTH) {
    bf16x8 v = *(const bf16x8 *)(xr + i * 8);
    if (add) {
      bf16x8 r = *(const bf16x8 *)(rr + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = (bf16)((float)v[j] + (float)r[j]);
      *(bf16x8 *)(ro + i * 8) = v;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += (float)v[j] * (float)v[j];
  }
  for (int off = 32; off > 0; off >>= 1) ss += __shfl_down(ss, off);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = ss;
  __syncthreads();
  if (threadIdx.x == 0) {
    float tt = 0.f;
    for (int i = 0; i < NTH / 64; ++i) tt += red[i];
    red[0] = rsqrtf(tt / cols + 1e-6f);
  }
  __syncthreads();
  const float scale = red[0];
  const bf16 *src = add ? ro : xr;
  for (int i = threadIdx.x; i < nv; i += NTH) {
    bf16x8 v = *(const bf16x8 *)(src + i * 8);
    bf16x8 wv = *(const bf16x8 *)(w + i * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (bf16)((float)v[j] * scale * (float)wv[j]);
    *(bf16x8 *)(orow + i * 8) = o;
  }
}

TD_DEV void t_gemm_tile(const Task &t, bf16 *lds) {
  // 32x128 C tile (decode-friendly: small-M waste bounded at 32 rows),
  // 4 waves side-by-side on N, 3-buffer BK=64 pipelined K loop (counted
  // vmcnt, one barrier pair per step — the measured-fast structure from
  // the grouped-GEMM tier). Masked rows for M < 32-multiples.
  constexpr int BM = 32, BN = 128, BK = 64;
  constexpr int ABUF = BM * BK, BBUF = BN * BK;
  bf16 *lds_a = lds;                 // 2 x 4 KB
  bf16 *lds_b = lds + 2 * ABUF;      // 2 x 16 KB (2-buffer: the union
                                     // drops to 40 KB -> 3 blocks/CU)
  const bf16 *A = (const bf16 *)t.a[0];
  const bf16 *B = (const bf16 *)t.a[1];
  bf16 *C = (bf16 *)t.a[2];
  const int m = (int)t.a[3], n = (int)t.a[4], k = (int)t.a[5];
  const int pid_m = (int)t.a[6], pid_n = (int)t.a[7];
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  f32x4 acc[2][2] = {};
  const bf16 *ga = A + (size_t)pid_m * BM * k;
  const bf16 *gb = B + (size_t)pid_n * BN * k;
  const int ksteps = k / BK;
  auto stage = [&](int ti, int buf) {
    const int k0 = ti * BK;
    {
      int row = tid >> 3, kc = tid & 7;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              ga + (size_t)row * k + k0 + kc * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_a + buf * ABUF + (wave * 64) * 8),
          16, 0, 0);
    }
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int qb = it * 256 + tid;
      int rowb = qb >> 3, kcb = qb & 7;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              gb + (size_t)rowb * k + k0 + kcb * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_b + buf * BBUF + (it * 256 + wave * 64) * 8),
          16, 0, 0);
    }
  };
  stage(0, 0);
  for (int ti = 0; ti < ksteps; ++ti) {
    const int buf = ti & 1;
    // 2-buffer, 1-step lookahead: stage ti is the newest in flight at
    // this wait -> full drain; the overlap is stage ti+1 issuing under
    // compute ti
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
    if (ti + 1 < ksteps) stage(ti + 1, (ti + 1) & 1);
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
  const int row_lim = m - pid_m * BM;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i * 16 + (lane >> 4) * 4 + r;
        int col = wave * 32 + j * 16 + (lane & 15);
        if (row < row_lim)
          C[((size_t)pid_m * BM + row) * n + (size_t)pid_n * BN + col] =
              (bf16)acc[i][j][r];
      }
}

// K-split GEMM: one K-range partial of a 32x128 tile -> fp32 ws slice
// [sk][m_pad][n]. Same pipelined structure as t_gemm_tile but over
// [k0, k0+klen). Opt-in (TD_MK_KSPLIT): shortens the megakernel's per-op
// critical path from a full-K tile to K/ksplit.
TD_DEV void t_gemm_tile_part(const Task &t, bf16 *lds) {
  constexpr int BM = 32, BN = 128, BK = 64;
  constexpr int ABUF = BM * BK, BBUF = BN * BK;
  bf16 *lds_a = lds;
  bf16 *lds_b = lds + 2 * ABUF;
  const bf16 *A = (const bf16 *)t.a[0];
  const bf16 *B = (const bf16 *)t.a[1];
  float *ws = (float *)t.a[2];
  const int m = (int)t.a[3], n = (int)t.a[4], k = (int)t.a[5];
  const int pid_m = (int)t.a[6], pid_n = (int)t.a[7];
  const int k0 = (int)t.a[8], klen = (int)t.a[9], sk = (int)t.a[10];
  const int m_pad = (m + BM - 1) / BM * BM;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  f32x4 acc[2][2] = {};
  const bf16 *ga = A + (size_t)pid_m * BM * k + k0;
  const bf16 *gb = B + (size_t)pid_n * BN * k + k0;
  const int ksteps = klen / BK;
  auto stage = [&](int ti, int buf) {
    const int kk0 = ti * BK;
    {
      int row = tid >> 3, kc = tid & 7;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              ga + (size_t)row * k + kk0 + kc * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_a + buf * ABUF + (wave * 64) * 8),
          16, 0, 0);
    }
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int qb = it * 256 + tid;
      int rowb = qb >> 3, kcb = qb & 7;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              gb + (size_t)rowb * k + kk0 + kcb * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_b + buf * BBUF + (it * 256 + wave * 64) * 8),
          16, 0, 0);
    }
  };
  stage(0, 0);
  for (int ti = 0; ti < ksteps; ++ti) {
    const int buf = ti & 1;
    // 2-buffer, 1-step lookahead: stage ti is the newest in flight at
    // this wait -> full drain; the overlap is stage ti+1 issuing under
    // compute ti
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
    if (ti + 1 < ksteps) stage(ti + 1, (ti + 1) & 1);
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
  const int row_lim = m - pid_m * BM;
  float *wsl = ws + (size_t)sk * m_pad * n;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i * 16 + (lane >> 4) * 4 + r;
        int col = wave * 32 + j * 16 + (lane & 15);
        if (row < row_lim)
          wsl[((size_t)pid_m * BM + row) * n + (size_t)pid_n * BN + col] =
              acc[i][j][r];
      }
}

// Hop-fused K-split partial: A = rmsnorm(x [+ res]) computed in
// registers at stage time (per-row inv-rms over the FULL row first, then
// VALU-normalized A staging; B keeps the async global_load_lds path).
// Removes the one-task-per-row [add+]rmsnorm serialization hop that
// gated every projection. The residual UPDATE (x' = x + res) runs as a
// parallel off-critical-path task writing a ping-pong buffer, so no
// reader ever sees a half-updated x.
TD_DEV void t_gemm_tile_part_nr(const Task &t, bf16 *lds, float *rs) {
  constexpr int BM = 32, BN = 128, BK = 64;
  constexpr int ABUF = BM * BK, BBUF = BN * BK;
  bf16 *lds_a = lds;
  bf16 *lds_b = lds + 2 * ABUF;
  const bf16 *X = (const bf16 *)t.a[0];
  const bf16 *B = (const bf16 *)t.a[1];
  float *ws = (float *)t.a[2];
  const int m = (int)t.a[3], n = (int)t.a[4], k = (int)t.a[5];
  const int pid_m = (int)t.a[6], pid_n = (int)t.a[7];
  const int k0 = (int)t.a[8], klen = (int)t.a[9], sk = (int)t.a[10];
  const bf16 *lnw = (const bf16 *)t.a[11];
  const bf16 *res = (const bf16 *)t.a[12];
  const int m_pad = (m + BM - 1) / BM * BM;
  const int tid = threadIdx.x;
  const int wave = tid >> 6, lane = tid & 63;
  const int row_g0 = pid_m * BM;
  // per-row inv-rms of (x [+ res]) over the full row: 8 threads per row
  {
    const int r = tid >> 3, tsub = tid & 7;
    int rr = row_g0 + r;
    if (rr >= m) rr = m > 0 ? m - 1 : 0;
    const bf16 *xr = X + (size_t)rr * k;
    const bf16 *dr = res ? res + (size_t)rr * k : nullptr;
    float ss = 0.f;
    for (int c = tsub * 8; c < k; c += 64) {
      bf16x8 v = *(const bf16x8 *)(xr + c);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j];
        if (dr) f += (float)*(dr + c + j);
        ss += f * f;
      }
    }
#pragma unroll
    for (int off = 1; off < 8; off <<= 1) ss += __shfl_xor(ss, off);
    if (tsub == 0) rs[r] = rsqrtf(ss / k + 1e-6f);
  }
  __syncthreads();
  f32x4 acc[2][2] = {};
  const bf16 *gb = B + (size_t)pid_n * BN * k + k0;
  const int ksteps = klen / BK;
  auto stage_b = [&](int ti, int buf) {
    const int kk0 = ti * BK;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int qb = it * 256 + tid;
      int rowb = qb >> 3, kcb = qb & 7;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(
              gb + (size_t)rowb * k + kk0 + kcb * 8),
          (__attribute__((address_space(3))) unsigned int *)(
              lds_b + buf * BBUF + (it * 256 + wave * 64) * 8),
          16, 0, 0);
    }
  };
  auto stage_a = [&](int ti, int buf) {
    // same 32x64 LDS layout the glds path produced (elem off = tid*8)
    const int row = tid >> 3;
    const int col = k0 + ti * BK + (tid & 7) * 8;
    int rr = row_g0 + row;
    if (rr >= m) rr = m > 0 ? m - 1 : 0;
    bf16x8 v = *(const bf16x8 *)(X + (size_t)rr * k + col);
    bf16x8 wv = *(const bf16x8 *)(lnw + col);
    const float sc = rs[row];
    bf16x8 o;
    if (res) {
      bf16x8 d = *(const bf16x8 *)(res + (size_t)rr * k + col);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = (bf16)(((float)v[j] + (float)d[j]) * sc * (float)wv[j]);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = (bf16)((float)v[j] * sc * (float)wv[j]);
    }
    *(bf16x8 *)(lds_a + buf * ABUF + tid * 8) = o;
  };
  stage_b(0, 0);
  stage_a(0, 0);
  for (int ti = 0; ti < ksteps; ++ti) {
    const int buf = ti & 1;
    // B: 2-buffer full drain (stage ti is newest in flight); A: my
    // ds_writes for ti must land before peers read them post-barrier
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    __builtin_amdgcn_sched_barrier(0);
    if (ti + 1 < ksteps) {
      stage_b(ti + 1, (ti + 1) & 1);
      stage_a(ti + 1, (ti + 1) & 1);
    }
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
  const int row_lim = m - pid_m * BM;
  float *wsl = ws + (size_t)sk * m_pad * n;
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i * 16 + (lane >> 4) * 4 + r;
        int col = wave * 32 + j * 16 + (lane & 15);
        if (row < row_lim)
          wsl[((size_t)pid_m * BM + row) * n + (size_t)pid_n * BN + col] =
              acc[i][j][r];
      }
}

// Sum the ksplit fp32 slices of one 32x128 tile -> bf16 C.
TD_DEV void t_tile_reduce(const Task &t) {
  constexpr int BM = 32, BN = 128;
  const float *ws = (const float *)t.a[0];
  bf16 *C = (bf16 *)t.a[1];
  const int m = (int)t.a[2], n = (int)t.a[3];
  const int pid_m = (int)t.a[4], pid_n = (int)t.a[5];
  const int ksplit = (int)t.a[6];
  const int m_pad = (m + BM - 1) / BM * BM;
  const int row_lim = m - pid_m * BM;
  for (int i = threadIdx.x; i < BM * BN; i += blockDim.x) {
    int row = i / BN, col = i % BN;
    if (row >= row_lim) continue;
    size_t off = ((size_t)pid_m * BM + row) * n + (size_t)pid_n * BN + col;
    float acc = 0.f;
    for (int sk = 0; sk < ksplit; ++sk)
      acc += ws[(size_t)sk * m_pad * n + off];
    C[off] = (bf16)acc;
  }
}

TD_DEV void t_swiglu(const Task &t) {
  const bf16 *h = (const bf16 *)t.a[0];
  bf16 *out = (bf16 *)t.a[1];
  const size_t rows = (size_t)t.a[2];
  const int inter = (int)t.a[3];
  const int chunk = (int)t.a[4], nchunks = (int)t.a[5];
  size_t total = rows * (size_t)inter / 8;
  size_t per = (total + nchunks - 1) / nchunks;
  size_t lo = chunk * per, hi = min(lo + per, total);
  for (size_t v = lo + threadIdx.x; v < hi; v += NTH) {
    size_t i = v * 8;
    size_t r = i / inter, c = i % inter;
    bf16x8 g = *(const bf16x8 *)(h + r * 2 * inter + c);
    bf16x8 u = *(const bf16x8 *)(h + r * 2 * inter + inter + c);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = (float)g[j];
      o[j] = (bf16)(gf / (1.f + __expf(-gf)) * (float)u[j]);
    }
    *(bf16x8 *)(out + i) = o;
  }
}

TD_DEV void t_qkv_prologue(const Task &t) {
  // one task = one batch row, all (qh + 2*kvh) heads; 4 heads at a time
  constexpr int D = 128;
  const bf16 *qkv = (const bf16 *)t.a[0];
  bf16 *q_out = (bf16 *)t.a[1];
  bf16 *kcache = (bf16 *)t.a[2];
  bf16 *vcache = (bf16 *)t.a[3];
  const float *cos_t = (const float *)t.a[4];
  const float *sin_t = (const float *)t.a[5];
  const bf16 *qnw = (const bf16 *)t.a[6];
  const bf16 *knw = (const bf16 *)t.a[7];
  const long pos = *(const long *)t.a[8];
  const int b = (int)t.a[9];
  const int qh = (int)t.a[10], kvh = (int)t.a[11], maxlen = (int)t.a[12];
  const int nh = qh + 2 * kvh;
  const int lane = threadIdx.x & 63;
  for (int h = threadIdx.x >> 6; h < nh; h += NTH / 64) {
    const bf16 *src = qkv + ((size_t)b * nh + h) * D;
    float v0 = (float)src[lane * 2];
    float v1 = (float)src[lane * 2 + 1];
    const bool is_q = h < qh;
    const bool is_k = h >= qh && h < qh + kvh;
    if (is_q || is_k) {
      float ss = v0 * v0 + v1 * v1;
      for (int off = 32; off > 0; off >>= 1) ss += __shfl_down(ss, off);
      float scale = rsqrtf(__shfl(ss, 0) / D + 1e-6f);
      const bf16 *nw = is_q ? qnw : knw;
      v0 *= scale * (float)nw[lane * 2];
      v1 *= scale * (float)nw[lane * 2 + 1];
      float p0 = __shfl_xor(v0, 32);
      float p1 = __shfl_xor(v1, 32);
      int d2 = (lane & 31) * 2;
      float c0 = cos_t[pos * (D / 2) + d2];
      float s0 = sin_t[pos * (D / 2) + d2];
      float c1 = cos_t[pos * (D / 2) + d2 + 1];
      float s1 = sin_t[pos * (D / 2) + d2 + 1];
      if (lane < 32) {
        v0 = v0 * c0 - p0 * s0;
        v1 = v1 * c1 - p1 * s1;
      } else {
        v0 = v0 * c0 + p0 * s0;
        v1 = v1 * c1 + p1 * s1;
      }
    }
    if (is_q) {
      bf16 *dst = q_out + ((size_t)b * qh + h) * D;
      dst[lane * 2] = (bf16)v0;
      dst[lane * 2 + 1] = (bf16)v1;
    } else {
      const int kh = h - qh;
      const bool k_side = kh < kvh;
      const int hh = k_side ? kh : kh - kvh;
      bf16 *cache = k_side ? kcache : vcache;
      bf16 *dst = cache + (((size_t)b * maxlen + pos) * kvh + hh) * D;
      dst[lane * 2] = (bf16)v0;
      dst[lane * 2 + 1] = (bf16)v1;
    }
  }
}

TD_DEV void t_embed(const Task &t) {
  const long *tokens = (const long *)t.a[0];
  const bf16 *table = (const bf16 *)t.a[1];
  bf16 *out = (bf16 *)t.a[2];
  const int cols = (int)t.a[4];
  const int row = (int)t.a[5];
  const bf16 *src = table + (size_t)tokens[row] * cols;
  bf16 *dst = out + (size_t)row * cols;
  for (int i = threadIdx.x; i < cols / 8; i += NTH)
    *(bf16x8 *)(dst + i * 8) = *(const bf16x8 *)(src + i * 8);
}

}  // namespace mk

// flash-decode task shares the standalone kernel's body via a device
// function (mirrors k_flash_decode in attention.hip, LDS passed in).
TD_DEV void mk_flash_decode_body(const bf16 *q, const bf16 *kcache,
                                 const bf16 *vcache, bf16 *out,
                                 const long *offset, int b, int kh, int qh,
                                 int kvh, int max_len, char *lds_raw) {
  constexpr int kD = 128, kTile = 32;
  const int G = qh / kvh;
  const int tid = threadIdx.x;
  const int g = tid >> 5;
  const int t = tid & 31;
  const long seqlen = *offset + 1;
  bf16(*k_lds)[kD] = (bf16(*)[kD])lds_raw;
  bf16(*v_lds)[kD] = (bf16(*)[kD])(lds_raw + kTile * kD * 2);
  float(*p_lds)[kTile] = (float(*)[kTile])(lds_raw + 2 * kTile * kD * 2);
  float *m_lds = (float *)(lds_raw + 2 * kTile * kD * 2 + 8 * kTile * 4);
  float *r_lds = m_lds + 8;
  float *l_lds = r_lds + 8;
  bf16(*q_lds)[kD] = (bf16(*)[kD])(l_lds + 8);

  for (int i = tid; i < 8 * kD / 8; i += 256) {
    int hh = i / (kD / 8);
    int c = (i % (kD / 8)) * 8;
    bf16x8 v{};
    if (hh < G)
      v = *(const bf16x8 *)(q + (((size_t)b * qh) + kh * G + hh) * kD + c);
    *(bf16x8 *)(&q_lds[hh][c]) = v;
  }
  if (tid < 8) {
    m_lds[tid] = -1e30f;
    l_lds[tid] = 0.f;
  }
  __syncthreads();
  float acc[4] = {};
  const int my_d0 = t * 4;
  const long ntiles = (seqlen + kTile - 1) / kTile;
  for (long tile = 0; tile < ntiles; ++tile) {
    const long pos0 = tile * kTile;
    __syncthreads();
    for (int i = tid; i < kTile * kD / 8; i += 256) {
      int r = i / (kD / 8);
      int c = (i % (kD / 8)) * 8;
      long pos = pos0 + r;
      bf16x8 kv{}, vv{};
      if (pos < seqlen) {
        size_t base = (((size_t)b * max_len + pos) * kvh + kh) * kD + c;
        kv = *(const bf16x8 *)(kcache + base);
        vv = *(const bf16x8 *)(vcache + base);
      }
      *(bf16x8 *)(&k_lds[r][c]) = kv;
      *(bf16x8 *)(&v_lds[r][c]) = vv;
    }
    __syncthreads();
    float s = -1e30f;
    if (g < G && pos0 + t < seqlen) {
      float d = 0.f;
#pragma unroll
      for (int c = 0; c < kD / 8; ++c) {
        bf16x8 qv = *(const bf16x8 *)(&q_lds[g][c * 8]);
        bf16x8 kv = *(const bf16x8 *)(&k_lds[t][c * 8]);
#pragma unroll
        for (int j = 0; j < 8; ++j) d += (float)qv[j] * (float)kv[j];
      }
      s = d * 0.08838834764831845f;  // 1/sqrt(128)
    }
    float mx = s;
    for (int off = 16; off > 0; off >>= 1)
      mx = fmaxf(mx, __shfl_xor(mx, off));
    float m_old = m_lds[g];
    float m_new = fmaxf(m_old, mx);
    float pp = (s > -1e29f) ? __expf(s - m_new) : 0.f;
    p_lds[g][t] = pp;
    float psum = pp;
    for (int off = 16; off > 0; off >>= 1) psum += __shfl_xor(psum, off);
    if (t == 0) {
      float r = __expf(m_old - m_new);
      r_lds[g] = r;
      l_lds[g] = l_lds[g] * r + psum;
      m_lds[g] = m_new;
    }
    __syncthreads();
    const float r = r_lds[g];
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[j] *= r;
    for (int tt = 0; tt < kTile; ++tt) {
      float pp2 = p_lds[g][tt];
      if (pp2 != 0.f) {
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[j] += pp2 * (float)v_lds[tt][my_d0 + j];
      }
    }
  }
  __syncthreads();
  if (g < G) {
    float inv_l = 1.f / l_lds[g];
    bf16 *dst = out + (((size_t)b * qh) + kh * G + g) * kD + my_d0;
#pragma unroll
    for (int j = 0; j < 4; ++j) dst[j] = (bf16)(acc[j] * inv_l);
  }
}

// Hop-fused prologue + flash decode for ONE (b, kh): qk-norm + RoPE +
// cache append for exactly this task's heads (q slots kh*G..kh*G+G-1,
// k/v head kh — the (b, kh) partition of the prologue is exact, so no
// cross-task dependency exists), then the standard flash-decode body
// reading the q / cache rows this task just wrote (vmcnt-drained +
// block-synced before the read).
TD_DEV void mk_pro_flash_decode(const mk::Task &t, char *lds_raw) {
  constexpr int D = 128;
  const bf16 *qkv = (const bf16 *)t.a[0];
  bf16 *q_out = (bf16 *)t.a[1];
  bf16 *kcache = (bf16 *)t.a[2];
  bf16 *vcache = (bf16 *)t.a[3];
  const float *cos_t = (const float *)t.a[4];
  const float *sin_t = (const float *)t.a[5];
  const bf16 *qnw = (const bf16 *)t.a[6];
  const bf16 *knw = (const bf16 *)t.a[7];
  const long pos = *(const long *)t.a[8];
  bf16 *attn_out = (bf16 *)t.a[9];
  const int b = (int)(t.a[10] & 0xFFFFFFFF);
  const int kh = (int)(t.a[10] >> 32);
  const int qh = (int)(t.a[11] & 0xFFFFFFFF);
  const int kvh = (int)(t.a[11] >> 32);
  const int maxlen = (int)t.a[12];
  const int G = qh / kvh;
  const int lane = threadIdx.x & 63;
  for (int hl = threadIdx.x >> 6; hl < G + 2; hl += mk::NTH / 64) {
    // local head -> global qkv head: q slots first, then k, then v
    const bool is_q = hl < G;
    const bool is_k = hl == G;
    const int h = is_q ? kh * G + hl : (is_k ? qh + kh : qh + kvh + kh);
    const bf16 *src = qkv + ((size_t)b * (qh + 2 * kvh) + h) * D;
    float v0 = (float)src[lane * 2];
    float v1 = (float)src[lane * 2 + 1];
    if (is_q || is_k) {
      float ss = v0 * v0 + v1 * v1;
      for (int off = 32; off > 0; off >>= 1) ss += __shfl_down(ss, off);
      float scale = rsqrtf(__shfl(ss, 0) / D + 1e-6f);
      const bf16 *nw = is_q ? qnw : knw;
      v0 *= scale * (float)nw[lane * 2];
      v1 *= scale * (float)nw[lane * 2 + 1];
      float p0 = __shfl_xor(v0, 32);
      float p1 = __shfl_xor(v1, 32);
      int d2 = (lane & 31) * 2;
      float c0 = cos_t[pos * (D / 2) + d2];
      float s0 = sin_t[pos * (D / 2) + d2];
      float c1 = cos_t[pos * (D / 2) + d2 + 1];
      float s1 = sin_t[pos * (D / 2) + d2 + 1];
      if (lane < 32) {
        v0 = v0 * c0 - p0 * s0;
        v1 = v1 * c1 - p1 * s1;
      } else {
        v0 = v0 * c0 + p0 * s0;
        v1 = v1 * c1 + p1 * s1;
      }
    }
    bf16 *dst;
    if (is_q) {
      dst = q_out + ((size_t)b * qh + h) * D;
    } else {
      bf16 *cache = is_k ? kcache : vcache;
      dst = cache + (((size_t)b * maxlen + pos) * kvh + kh) * D;
    }
    dst[lane * 2] = (bf16)v0;
    dst[lane * 2 + 1] = (bf16)v1;
  }
  // my global stores must be readable below (same block, other threads)
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  mk_flash_decode_body(q_out, kcache, vcache, attn_out,
                       (const long *)t.a[8], b, kh, qh, kvh, maxlen,
                       lds_raw);
}

namespace mk {

// Decode GEMV: x rows staged in LDS (m <= 4), each thread owns
// ncols/blockDim output columns and streams the weight rows with bf16x8
// loads — HBM-bound by construction, the right regime for bsz<=4.
TD_DEV void t_gemv(const Task &t, char *lds) {
  const bf16 *A = (const bf16 *)t.a[0];
  const bf16 *W = (const bf16 *)t.a[1];
  bf16 *C = (bf16 *)t.a[2];
  const int m = (int)t.a[3];
  const int n = (int)t.a[4];
  const int k = (int)t.a[5];
  const int col0 = (int)t.a[6];
  const int ncols = (int)t.a[7];
  bf16 *x_lds = (bf16 *)lds;  // [m][k] (m*k*2 <= 40960 guaranteed by host)
  for (int i = threadIdx.x * 8; i < m * k; i += blockDim.x * 8)
    *(bf16x8 *)(x_lds + i) = *(const bf16x8 *)(A + i);
  __syncthreads();
  // wave-per-column: 64 lanes stream one weight row's K contiguously
  // (coalesced 1 KiB per load instruction), shuffle-tree reduce, lane 0
  // stores — the column-per-thread layout was fully uncoalesced.
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int nwaves = blockDim.x >> 6;
  for (int cb = wave; cb < ncols; cb += nwaves) {
    const int c = col0 + cb;
    const bf16 *wr = W + (size_t)c * k;
    float acc[4] = {0.f, 0.f, 0.f, 0.f};
    for (int kk = lane * 8; kk < k; kk += 64 * 8) {
      bf16x8 w8 = *(const bf16x8 *)(wr + kk);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        if (r < m) {
          bf16x8 x8 = *(const bf16x8 *)(x_lds + r * k + kk);
#pragma unroll
          for (int e = 0; e < 8; ++e)
            acc[r] += (float)x8[e] * (float)w8[e];
        }
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      if (r < m) {
#pragma unroll
        for (int off = 32; off > 0; off >>= 1)
          acc[r] += __shfl_xor(acc[r], off);
        if (lane == 0) C[(size_t)r * n + c] = (bf16)acc[r];
      }
    }
  }
}

// relaxed spin (no per-iteration L2 invalidate); caller issues ONE
// acquire fence after all deps are satisfied
TD_DEV void spin_ge_relaxed(const int *flag, int bound) {
  uint64_t t0 = wallclock();
  while (ld_relaxed<Scope::Gpu>(flag) < bound) {
    __builtin_amdgcn_s_sleep(1);
    if (wallclock() - t0 > TD_SPIN_TIMEOUT_TICKS) __builtin_trap();
  }
}

}  // namespace mk

// ---------------------------------------------------------------------------
// the persistent megakernel
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(mk::NTH) void k_megakernel(
    const mk::Task *__restrict__ tasks, const int *__restrict__ queue,
    const int *__restrict__ queue_off, int *__restrict__ scoreboard,
    int fence_mode, unsigned long long *__restrict__ prof) {
  // prof (optional): [16 types x 2] wallclock accumulators —
  // [type][0] += dependency-wait ticks, [type][1] += body ticks
  __shared__ char lds[41088];  // union: gemm 2-buf A/B | flash-decode
                               // state, + 128 B row-scale scratch for
                               // the norm-fused GEMM parts (40.1 KB ->
                               // still 3 blocks/CU; the 60 KB 3-buf
                               // union capped residency at 2)
  const int wg = blockIdx.x;
  const int q_lo = queue_off[wg], q_hi = queue_off[wg + 1];
  for (int qi = q_lo; qi < q_hi; ++qi) {
    const mk::Task t = tasks[queue[qi]];
    const unsigned long long t0 = prof ? wallclock() : 0;
    // dependency waits: relaxed spins + ONE acquire fence (a per-
    // iteration agent-scope acquire invalidates the XCD L2 every spin —
    // measured poison at 10k+ tasks/step)
    if (threadIdx.x == 0) {
      if (t.dep0 >= 0) mk::spin_ge_relaxed(scoreboard + t.dep0, t.dep0_n);
      if (t.dep1 >= 0) mk::spin_ge_relaxed(scoreboard + t.dep1, t.dep1_n);
      if (t.dep0 >= 0 || t.dep1 >= 0)
        __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
    }
    __syncthreads();
    const unsigned long long t1 = prof ? wallclock() : 0;
    switch (t.type) {
      case mk::T_RMSNORM:
        mk::t_rmsnorm(t, false);
        break;
      case mk::T_ADD_RMSNORM:
        mk::t_rmsnorm(t, true);
        break;
      case mk::T_GEMM_TILE:
        mk::t_gemm_tile(t, (bf16 *)lds);
        break;
      case mk::T_SWIGLU:
        mk::t_swiglu(t);
        break;
      case mk::T_QKV_PROLOGUE:
        mk::t_qkv_prologue(t);
        break;
      case mk::T_FLASH_DECODE:
        mk_flash_decode_body((const bf16 *)t.a[0], (const bf16 *)t.a[1],
                             (const bf16 *)t.a[2], (bf16 *)t.a[3],
                             (const long *)t.a[4], (int)t.a[5], (int)t.a[6],
                             (int)t.a[7], (int)t.a[8], (int)t.a[9], lds);
        break;
      case mk::T_EMBED:
        mk::t_embed(t);
        break;
      case mk::T_GEMM_TILE_PART:
        mk::t_gemm_tile_part(t, (bf16 *)lds);
        break;
      case mk::T_TILE_REDUCE:
        mk::t_tile_reduce(t);
        break;
      case mk::T_KV_ADVANCE:
        if (threadIdx.x == 0) *(long *)t.a[0] += 1;
        break;
      case mk::T_GEMV:
        mk::t_gemv(t, lds);
        break;
      case mk::T_PRO_FLASH_DECODE:
        mk_pro_flash_decode(t, lds);
        break;
      case mk::T_GEMM_TILE_PART_NR:
        mk::t_gemm_tile_part_nr(t, (bf16 *)lds, (float *)(lds + 40960));
        break;
      default:
        break;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      // agent scope: single GPU — the writeback makes this XCD's lines
      // visible to consumer tasks on other XCDs. fence_mode 1 elides it
      // (TD_MK_FENCE=1: A/B probe of the per-task wbl2 cost — NOT a
      // correct configuration, diagnosis only).
      if (fence_mode == 0)
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
      atomic_add<Scope::Gpu>(scoreboard + t.score_slot, 1);
      if (prof) {
        const int ty = t.type & 15;
        atomic_add<Scope::Gpu>(prof + ty * 2, t1 - t0);
        atomic_add<Scope::Gpu>(prof + ty * 2 + 1, wallclock() - t1);
      }
    }
    __syncthreads();
  }
}

void launch_megakernel(const void *tasks, const void *queue,
                       const void *queue_off, void *scoreboard, int n_wg,
                       hipStream_t stream, int fence_mode, void *prof) {
  hipLaunchKernelGGL(k_megakernel, dim3(n_wg), dim3(mk::NTH), 0, stream,
                     (const mk::Task *)tasks, (const int *)queue,
                     (const int *)queue_off, (int *)scoreboard,
                     fence_mode, (unsigned long long *)prof);
}

}  // namespace td
