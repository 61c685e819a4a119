// GQA flash-decode kernel (single new token per sequence).
//
// Capability parity with the reference's distributed flash-decode family
// (Triton-distributed python/triton_dist/kernels/nvidia/flash_decode.py:
// 130-1132 — split-KV online-softmax decode; the inter-rank LSE-merge
// combine comes with the SP layer). MI355X-native design:
//
//   grid = (batch, kv_heads); block = 256 threads (4 waves).
//   Each block serves ALL G = qh/kvh query heads of one (seq, kv head) —
//   K/V rows are read once and reused G times (decode is KV-bandwidth
//   bound; this is the whole game on a 8 TB/s HBM part).
//   Per 32-position tile: threads (g = tid>>5, t = tid&31) compute one
//   dot(q[g], k[pos]) each; per-g online-softmax (running max m, sum l)
//   via 32-lane half-wave shuffles; P tile staged in LDS; every thread
//   then updates its private (g, d) slice of the fp32 accumulator.
//   Sequence length = *offset + 1 read from DEVICE memory (hipGraph-safe).
//
// Layouts: q [B, qh, 128] bf16 (post-RoPE), cache [B, max_len, kvh, 128].
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

constexpr int kD = 128;
constexpr int kTile = 32;

__global__ __launch_bounds__(256) void k_flash_decode(
    const bf16 *__restrict__ q, const bf16 *__restrict__ kcache,
    const bf16 *__restrict__ vcache, bf16 *__restrict__ out,
    const long *__restrict__ offset, int qh, int kvh, int max_len,
    float scale) {
  // QK^T on MFMA: S[16 q-slots][32 pos] = Q[16,128] x K^T[128,32] — each
  // of the 4 waves contributes one 32-dim k-chunk (2 mfma_16x16x32) and
  // partials reduce through LDS. PV also on MFMA: P (bf16, standard
  // flash-attention practice) x V[32,128], V staged row-major in LDS and
  // consumed column-major via ds_read_b64_tr_b16 (the VALU PV loop this
  // replaced was 63% of kernel time — see scripts/probe/probe_attn.hip).
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = qh / kvh;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = tid >> 5;   // softmax role: query slot
  const int t = tid & 31;   // softmax role: position
  const long seqlen = *offset + 1;

  // KV tiles prefetched through REGISTERS (4 x bf16x8 per thread): tile
  // t+1's global loads issue before tile t's compute, so HBM latency
  // hides under the QK/softmax/PV work; LDS buffers stay SINGLE (the
  // double-LDS variant cost 5 -> 3 blocks/CU of occupancy). Stores land
  // between the two loop-top barriers — same barrier count as the r01
  // serial load -> compute structure.
  __shared__ bf16 k_lds[kTile][kD];
  // V rows padded +8 elems: tr_read's 4-rows-x-4-colchunks per 16-lane
  // group land on rotated banks (row stride 272 B = 4-bank rotate)
  __shared__ bf16 v_lds[kTile][kD + 8];
  __shared__ bf16 q_lds[16][kD];          // rows >= G zero-padded
  __shared__ bf16 p_bf[16][kTile + 8];    // P tile as MFMA A operand
  __shared__ float s_part[4][16][kTile];  // per-wave QK partials
  __shared__ float m_lds[16], r_lds[16], l_lds[16];

  for (int i = tid; i < 16 * kD / 8; i += 256) {
    int hh = i / (kD / 8);
    int c = (i % (kD / 8)) * 8;
    bf16x8 v{};
    if (hh < G)
      v = *(const bf16x8 *)(q + (((size_t)b * qh) + kh * G + hh) * kD + c);
    *(bf16x8 *)(&q_lds[hh][c]) = v;
  }
  if (tid < 16) {
    m_lds[tid] = -1e30f;
    l_lds[tid] = 0.f;
    r_lds[tid] = 1.f;
    for (int tt = 0; tt < kTile; ++tt) p_bf[tid][tt] = (bf16)0.f;
  }
  __syncthreads();

  // hoist the Q fragment (constant across tiles): wave w covers k-chunk w
  bf16x8 qfrag = *(const bf16x8 *)(
      &q_lds[lane & 15][wave * 32 + (lane >> 4) * 8]);

  f32x4 accPV[2] = {};
  const long ntiles = (seqlen + kTile - 1) / kTile;
  // per-thread slice: 2 chunks of each matrix per tile
  bf16x8 nk[2], nv[2];
  auto load_tile = [&](long tile, bf16x8 *rk, bf16x8 *rv) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int i = it * 256 + tid;
      int r = i / (kD / 8);
      int c = (i % (kD / 8)) * 8;
      long pos = tile * kTile + r;
      bf16x8 kv{}, vv{};
      if (pos < seqlen) {
        size_t base = (((size_t)b * max_len + pos) * kvh + kh) * kD + c;
        kv = *(const bf16x8 *)(kcache + base);
        vv = *(const bf16x8 *)(vcache + base);
      }
      rk[it] = kv;
      rv[it] = vv;
    }
  };
  auto store_tile = [&](const bf16x8 *rk, const bf16x8 *rv) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int i = it * 256 + tid;
      int r = i / (kD / 8);
      int c = (i % (kD / 8)) * 8;
      *(bf16x8 *)(&k_lds[r][c]) = rk[it];
      *(bf16x8 *)(&v_lds[r][c]) = rv[it];
    }
  };
  load_tile(0, nk, nv);
  for (long tile = 0; tile < ntiles; ++tile) {
    const long pos0 = tile * kTile;
    __syncthreads();  // previous tile's LDS reads retired
    store_tile(nk, nv);
    const bool has_next = tile + 1 < ntiles;
    if (has_next) load_tile(tile + 1, nk, nv);  // regs free after store
    __syncthreads();  // this tile's stores visible

    // MFMA QK^T: wave w, pos-half h: B[k][n=pos] = K[pos][k]
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      bf16x8 kfrag = *(const bf16x8 *)(
          &k_lds[(lane & 15) + 16 * h][wave * 32 + (lane >> 4) * 8]);
      f32x4 c4 = {0.f, 0.f, 0.f, 0.f};
      c4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag, kfrag, c4, 0, 0,
                                                   0);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        s_part[wave][(lane >> 4) * 4 + r][(lane & 15) + 16 * h] = c4[r];
    }
    __syncthreads();

    // softmax: 8 thread-groups of 32 sweep up to 16 q-slots (2 passes —
    // seed-oss-36b geometry has G = qh/kvh = 10)
#pragma unroll
    for (int gg = g; gg < 16; gg += 8) {
      float s = -1e30f;
      if (gg < G && pos0 + t < seqlen) {
        s = (s_part[0][gg][t] + s_part[1][gg][t] + s_part[2][gg][t] +
             s_part[3][gg][t]) * scale;
      }
      float mx = s;
      for (int off = 16; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
      float m_old = m_lds[gg];
      float m_new = fmaxf(m_old, mx);
      float p = (s > -1e29f) ? __expf(s - m_new) : 0.f;
      p_bf[gg][t] = (bf16)p;
      float psum = p;
      for (int off = 16; off > 0; off >>= 1) psum += __shfl_xor(psum, off);
      if (t == 0) {
        float r = __expf(m_old - m_new);
        r_lds[gg] = r;
        l_lds[gg] = l_lds[gg] * r + psum;
        m_lds[gg] = m_new;
      }
    }
    __syncthreads();

    // MFMA PV: wave w owns output cols w*32..w*32+31 (2 col-groups of 16).
    // tr_read semantics (measured, scripts/probe/probe_attn.hip): within a
    // 16-lane group, lane k supplies an 8B-word address; lane l receives
    // elem (l&3) of the words fetched by lanes ((l&15)>>2)+4j, j=0..3. So
    // lane l pointing at V row k0+((l&15)>>2), col-chunk 4*(l&3) makes the
    // instruction deliver V[k0+j][col l&15]: a column-major B-fragment
    // from PLAIN row-major V.
    bf16x8 afrag = *(const bf16x8 *)(&p_bf[lane & 15][(lane >> 4) * 8]);
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int cg = wave * 2 + h;
      const int k0 = (lane >> 4) * 8;
      unsigned addr = (unsigned)(uintptr_t)(
          &v_lds[k0 + ((lane & 15) >> 2)][cg * 16 + 4 * (lane & 3)]);
      unsigned long long lo, hi;
      asm volatile(
          "ds_read_b64_tr_b16 %0, %2\n"
          "ds_read_b64_tr_b16 %1, %2 offset:%3\n"
          "s_waitcnt lgkmcnt(0)"
          : "=v"(lo), "=v"(hi)
          : "v"(addr), "i"(4 * (kD + 8) * 2));
      bf16x8 bfrag;
      *(unsigned long long *)&bfrag = lo;
      *((unsigned long long *)&bfrag + 1) = hi;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        accPV[h][r] *= r_lds[(lane >> 4) * 4 + r];
      accPV[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         accPV[h], 0, 0, 0);
    }
  }
  __syncthreads();
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = (lane >> 4) * 4 + r;
      if (row < G) {
        float inv_l = 1.f / l_lds[row];
        out[(((size_t)b * qh) + kh * G + row) * kD + wave * 32 + h * 16 +
            (lane & 15)] = (bf16)(accPV[h][r] * inv_l);
      }
    }
}

void launch_flash_decode(const void *q, const void *kcache,
                         const void *vcache, void *out, const void *offset,
                         int batch, int qh, int kvh, int max_len,
                         hipStream_t stream) {
  if (qh / kvh > 16 || qh % kvh)
    throw std::runtime_error("flash_decode: qh/kvh must divide and be <= 16");
  float scale = 1.f / sqrtf((float)kD);
  hipLaunchKernelGGL(k_flash_decode, dim3(batch, kvh), dim3(256), 0, stream,
                     (const bf16 *)q, (const bf16 *)kcache,
                     (const bf16 *)vcache, (bf16 *)out,
                     (const long *)offset, qh, kvh, max_len, scale);
}

// ---------------------------------------------------------------------------
// EXPERIMENTAL paged flash-decode: identical math to k_flash_decode but K/V
// come from a block pool [n_blocks, block, kvh, D] via a per-sequence block
// table [B, max_blocks] (models/kv_cache.py PagedKVCache layout). Gated
// behind TD_EXPERIMENTAL in tests; round 2 validates + wires an engine
// path. Block size is a power of two (shift passed in).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_flash_decode_paged(
    const bf16 *__restrict__ q, const bf16 *__restrict__ k_pool,
    const bf16 *__restrict__ v_pool, const long *__restrict__ block_table,
    int max_blocks, int blk_shift, bf16 *__restrict__ out,
    const long *__restrict__ offset, int qh, int kvh, float scale) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = qh / kvh;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = tid >> 5;
  const int t = tid & 31;
  const long seqlen = *offset + 1;
  const long *bt = block_table + (size_t)b * max_blocks;
  const int blk_mask = (1 << blk_shift) - 1;

  __shared__ bf16 k_lds[kTile][kD];
  __shared__ bf16 v_lds[kTile][kD + 8];
  __shared__ bf16 q_lds[16][kD];
  __shared__ bf16 p_bf[16][kTile + 8];
  __shared__ float s_part[4][16][kTile];
  __shared__ float m_lds[16], r_lds[16], l_lds[16];

  for (int i = tid; i < 16 * kD / 8; i += 256) {
    int hh = i / (kD / 8);
    int c = (i % (kD / 8)) * 8;
    bf16x8 v{};
    if (hh < G)
      v = *(const bf16x8 *)(q + (((size_t)b * qh) + kh * G + hh) * kD + c);
    *(bf16x8 *)(&q_lds[hh][c]) = v;
  }
  if (tid < 16) {
    m_lds[tid] = -1e30f;
    l_lds[tid] = 0.f;
    r_lds[tid] = 1.f;
    for (int tt = 0; tt < kTile; ++tt) p_bf[tid][tt] = (bf16)0.f;
  }
  __syncthreads();
  bf16x8 qfrag = *(const bf16x8 *)(
      &q_lds[lane & 15][wave * 32 + (lane >> 4) * 8]);

  f32x4 accPV[2] = {};
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
        // pool addressing: block id from the table, row = pos % block
        long pb = bt[pos >> blk_shift];
        size_t base =
            (((size_t)pb << blk_shift | (pos & blk_mask)) * kvh + kh) * kD +
            c;
        kv = *(const bf16x8 *)(k_pool + base);
        vv = *(const bf16x8 *)(v_pool + base);
      }
      *(bf16x8 *)(&k_lds[r][c]) = kv;
      *(bf16x8 *)(&v_lds[r][c]) = vv;
    }
    __syncthreads();

#pragma unroll
    for (int h = 0; h < 2; ++h) {
      bf16x8 kfrag = *(const bf16x8 *)(
          &k_lds[(lane & 15) + 16 * h][wave * 32 + (lane >> 4) * 8]);
      f32x4 c4 = {0.f, 0.f, 0.f, 0.f};
      c4 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag, kfrag, c4, 0, 0,
                                                   0);
#pragma unroll
      for (int r = 0; r < 4; ++r)
        s_part[wave][(lane >> 4) * 4 + r][(lane & 15) + 16 * h] = c4[r];
    }
    __syncthreads();

#pragma unroll
    for (int gg = g; gg < 16; gg += 8) {
      float s = -1e30f;
      if (gg < G && pos0 + t < seqlen) {
        s = (s_part[0][gg][t] + s_part[1][gg][t] + s_part[2][gg][t] +
             s_part[3][gg][t]) * scale;
      }
      float mx = s;
      for (int off = 16; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
      float m_old = m_lds[gg];
      float m_new = fmaxf(m_old, mx);
      float p = (s > -1e29f) ? __expf(s - m_new) : 0.f;
      p_bf[gg][t] = (bf16)p;
      float psum = p;
      for (int off = 16; off > 0; off >>= 1) psum += __shfl_xor(psum, off);
      if (t == 0) {
        float r = __expf(m_old - m_new);
        r_lds[gg] = r;
        l_lds[gg] = l_lds[gg] * r + psum;
        m_lds[gg] = m_new;
      }
    }
    __syncthreads();

    bf16x8 afrag = *(const bf16x8 *)(&p_bf[lane & 15][(lane >> 4) * 8]);
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int cg = wave * 2 + h;
      const int k0 = (lane >> 4) * 8;
      unsigned addr = (unsigned)(uintptr_t)(
          &v_lds[k0 + ((lane & 15) >> 2)][cg * 16 + 4 * (lane & 3)]);
      unsigned long long lo, hi;
      asm volatile(
          "ds_read_b64_tr_b16 %0, %2\n"
          "ds_read_b64_tr_b16 %1, %2 offset:%3\n"
          "s_waitcnt lgkmcnt(0)"
          : "=v"(lo), "=v"(hi)
          : "v"(addr), "i"(4 * (kD + 8) * 2));
      bf16x8 bfrag;
      *(unsigned long long *)&bfrag = lo;
      *((unsigned long long *)&bfrag + 1) = hi;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        accPV[h][r] *= r_lds[(lane >> 4) * 4 + r];
      accPV[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         accPV[h], 0, 0, 0);
    }
  }
  __syncthreads();
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = (lane >> 4) * 4 + r;
      if (row < G) {
        float inv_l = 1.f / l_lds[row];
        out[(((size_t)b * qh) + kh * G + row) * kD + wave * 32 + h * 16 +
            (lane & 15)] = (bf16)(accPV[h][r] * inv_l);
      }
    }
}

void launch_flash_decode_paged(const void *q, const void *k_pool,
                               const void *v_pool, const void *block_table,
                               int max_blocks, int block, void *out,
                               const void *offset, int batch, int qh,
                               int kvh, hipStream_t stream) {
  if (qh / kvh > 16 || qh % kvh)
    throw std::runtime_error("flash_decode: qh/kvh must divide and be <= 16");
  if (block & (block - 1))
    throw std::runtime_error("flash_decode_paged: block must be a power of 2");
  int shift = 0;
  while ((1 << shift) < block) ++shift;
  float scale = 1.f / sqrtf((float)kD);
  hipLaunchKernelGGL(k_flash_decode_paged, dim3(batch, kvh), dim3(256), 0,
                     stream, (const bf16 *)q, (const bf16 *)k_pool,
                     (const bf16 *)v_pool, (const long *)block_table,
                     max_blocks, shift, (bf16 *)out, (const long *)offset,
                     qh, kvh, scale);
}

// ---------------------------------------------------------------------------
// SP split-KV decode: same kernel body, but the KV chunk length comes from
// `chunk_len` (device int64: number of valid positions in THIS rank's KV
// shard) and the outputs are the UNNORMALIZED partial accumulator plus the
// log-sum-exp, for the cross-rank combine (capability: reference
// flash_decode.py:482-532 inter-rank LSE-merge).
// out_part: [B, qh, 128] fp32 (acc / l), lse: [B, qh] fp32 (m + log l).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_flash_decode_partial(
    const bf16 *__restrict__ q, const bf16 *__restrict__ kcache,
    const bf16 *__restrict__ vcache, float *__restrict__ out_part,
    float *__restrict__ lse, const long *__restrict__ chunk_len, int qh,
    int kvh, int max_len, float scale) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = qh / kvh;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = tid >> 5;
  const int t = tid & 31;
  const long seqlen = *chunk_len;

  __shared__ bf16 k_lds[kTile][kD];
  __shared__ bf16 v_lds[kTile][kD + 8];  // padded for tr_read (see above)
  __shared__ bf16 p_bf[16][kTile + 8];
  __shared__ float m_lds[16], r_lds[16], l_lds[16];
  __shared__ bf16 q_lds[16][kD];

  for (int i = tid; i < 16 * kD / 8; i += 256) {
    int hh = i / (kD / 8);
    int c = (i % (kD / 8)) * 8;
    bf16x8 v{};
    if (hh < G)
      v = *(const bf16x8 *)(q + (((size_t)b * qh) + kh * G + hh) * kD + c);
    *(bf16x8 *)(&q_lds[hh][c]) = v;
  }
  if (tid < 16) {
    m_lds[tid] = -1e30f;
    l_lds[tid] = 0.f;
    r_lds[tid] = 1.f;
    for (int tt = 0; tt < kTile; ++tt) p_bf[tid][tt] = (bf16)0.f;
  }
  __syncthreads();

  f32x4 accPV[2] = {};
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
#pragma unroll
    for (int gg = g; gg < 16; gg += 8) {
      float s = -1e30f;
      if (gg < G && pos0 + t < seqlen) {
        float d = 0.f;
#pragma unroll
        for (int c = 0; c < kD / 8; ++c) {
          bf16x8 qv = *(const bf16x8 *)(&q_lds[gg][c * 8]);
          bf16x8 kv = *(const bf16x8 *)(&k_lds[t][c * 8]);
#pragma unroll
          for (int j = 0; j < 8; ++j) d += (float)qv[j] * (float)kv[j];
        }
        s = d * scale;
      }
      float mx = s;
      for (int off = 16; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
      float m_old = m_lds[gg];
      float m_new = fmaxf(m_old, mx);
      float p = (s > -1e29f) ? __expf(s - m_new) : 0.f;
      p_bf[gg][t] = (bf16)p;
      float psum = p;
      for (int off = 16; off > 0; off >>= 1) psum += __shfl_xor(psum, off);
      if (t == 0) {
        float r = __expf(m_old - m_new);
        r_lds[gg] = r;
        l_lds[gg] = l_lds[gg] * r + psum;
        m_lds[gg] = m_new;
      }
    }
    __syncthreads();
    // MFMA PV, same structure as k_flash_decode above
    bf16x8 afrag = *(const bf16x8 *)(&p_bf[lane & 15][(lane >> 4) * 8]);
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int cg = wave * 2 + h;
      const int k0 = (lane >> 4) * 8;
      unsigned addr = (unsigned)(uintptr_t)(
          &v_lds[k0 + ((lane & 15) >> 2)][cg * 16 + 4 * (lane & 3)]);
      unsigned long long lo, hi;
      asm volatile(
          "ds_read_b64_tr_b16 %0, %2\n"
          "ds_read_b64_tr_b16 %1, %2 offset:%3\n"
          "s_waitcnt lgkmcnt(0)"
          : "=v"(lo), "=v"(hi)
          : "v"(addr), "i"(4 * (kD + 8) * 2));
      bf16x8 bfrag;
      *(unsigned long long *)&bfrag = lo;
      *((unsigned long long *)&bfrag + 1) = hi;
#pragma unroll
      for (int r = 0; r < 4; ++r)
        accPV[h][r] *= r_lds[(lane >> 4) * 4 + r];
      accPV[h] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                         accPV[h], 0, 0, 0);
    }
  }
  __syncthreads();
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = (lane >> 4) * 4 + r;
      if (row < G) {
        float l = l_lds[row];
        float inv_l = l > 0.f ? 1.f / l : 0.f;
        out_part[(((size_t)b * qh) + kh * G + row) * kD + wave * 32 +
                 h * 16 + (lane & 15)] = accPV[h][r] * inv_l;
        if (h == 0 && wave == 0 && (lane & 15) == 0) {
          float m = m_lds[row];
          lse[(size_t)b * qh + kh * G + row] =
              (l > 0.f) ? m + __logf(l) : -1e30f;
        }
      }
    }
}

void launch_flash_decode_partial(const void *q, const void *kcache,
                                 const void *vcache, void *out_part,
                                 void *lse, const void *chunk_len, int batch,
                                 int qh, int kvh, int max_len,
                                 hipStream_t stream) {
  if (qh / kvh > 16 || qh % kvh)
    throw std::runtime_error("flash_decode: qh/kvh must divide and be <= 16");
  float scale = 1.f / sqrtf((float)kD);
  hipLaunchKernelGGL(k_flash_decode_partial, dim3(batch, kvh), dim3(256), 0,
                     stream, (const bf16 *)q, (const bf16 *)kcache,
                     (const bf16 *)vcache, (float *)out_part, (float *)lse,
                     (const long *)chunk_len, qh, kvh, max_len, scale);
}

// ---------------------------------------------------------------------------
// Cross-rank LSE combine: partials [world, B, qh, 128] fp32 + lses
// [world, B, qh] fp32 -> out [B, qh, 128] bf16, after waiting the per-rank
// ready flags. Block per (b, h); 32 lanes per 128 dims x 4.
// ---------------------------------------------------------------------------
__global__ void k_lse_combine(const float *__restrict__ parts,
                              const float *__restrict__ lses,
                              bf16 *__restrict__ out, const int *flags,
                              int world, int batch, int qh,
                              int slot_batch) {
  // slot_batch = allocated rows per rank slot (>= batch): peers write
  // into [r, 0..batch) of a max_batch-strided slot, so b < max_batch
  // calls index with the SLOT stride, not the call's batch
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  if (threadIdx.x < (unsigned)world)
    wait_ge_one<Scope::Sys>(flags + threadIdx.x, 1);
  __syncthreads();
  __shared__ float w_sh[kMaxRanks];
  if (threadIdx.x == 0) {
    float mx = -1e30f;
    for (int r = 0; r < world; ++r)
      mx = fmaxf(mx, lses[((size_t)r * slot_batch + b) * qh + h]);
    float denom = 0.f;
    for (int r = 0; r < world; ++r) {
      float w = __expf(lses[((size_t)r * slot_batch + b) * qh + h] - mx);
      w_sh[r] = w;
      denom += w;
    }
    for (int r = 0; r < world; ++r) w_sh[r] /= denom;
  }
  __syncthreads();
  for (int d = threadIdx.x; d < kD; d += blockDim.x) {
    float acc = 0.f;
    for (int r = 0; r < world; ++r)
      acc += w_sh[r] *
             parts[(((size_t)r * slot_batch + b) * qh + h) * kD + d];
    out[((size_t)b * qh + h) * kD + d] = (bf16)acc;
  }
}

void launch_lse_combine(const void *parts, const void *lses, void *out,
                        const void *flags, int world, int batch, int qh,
                        int slot_batch, hipStream_t stream) {
  hipLaunchKernelGGL(k_lse_combine, dim3(batch, qh), dim3(128), 0, stream,
                     (const float *)parts, (const float *)lses, (bf16 *)out,
                     (const int *)flags, world, batch, qh, slot_batch);
}

}  // namespace td

namespace td {

// ---------------------------------------------------------------------------
// Prefill flash-attention (FA2 forward, causal, GQA) — replaces torch sdpa
// on the prefill hot path (capability parity with the reference's own
// FA2-style consumer, kernels/nvidia/sp_ag_attention_intra_node.py:257-428
// — behavior only).
//
//   grid = (ceil(s/128), qh, b); block = 256 threads (4 waves).
//   Q tile 128 rows (wave w owns rows w*32..w*32+31), KV tiles of 32.
//   Per kv tile: S = Q K^T on MFMA (Q fragments hoisted from global into
//   registers once), online softmax with per-lane row state (each lane
//   tracks its 8 C-layout rows), P staged to LDS as the MFMA A operand,
//   PV via ds_read_b64_tr_b16 column-major V fragments from row-major
//   LDS (same measured semantics as the decode kernel above).
//   Causal mask: kv_pos > q_pos -> -inf (tiles fully above the diagonal
//   are never visited).
//
// Layouts: q [b, s, qh, 128], k/v [b, s, kvh, 128], out [b, s, qh, 128]
// — the layer's natural post-RoPE layout; no host-side transposes.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_flash_prefill(
    const bf16 *__restrict__ q, const bf16 *__restrict__ k,
    const bf16 *__restrict__ v, bf16 *__restrict__ out,
    float *__restrict__ lse, int s, int qh, int kvh, float scale,
    int causal, long kb_stride) {
  const int q0 = blockIdx.x * 128;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int G = qh / kvh;
  const int kh = h / G;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  __shared__ bf16 k_lds[kTile][kD + 8];
  __shared__ bf16 v_lds[kTile][kD + 8];
  __shared__ bf16 p_lds[128][kTile + 8];

  // hoist Q fragments: A operand rows = lane&15, 2 row-frags x 4 k-chunks
  bf16x8 qf[2][4];
  const int wrow0 = q0 + wave * 32;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int row = wrow0 + i * 16 + (lane & 15);
    const bf16 *qr =
        q + (((size_t)b * s + min(row, s - 1)) * qh + h) * kD;
    bool ok = row < s;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8 z{};
      qf[i][ks] = ok ? *(const bf16x8 *)(qr + ks * 32 + (lane >> 4) * 8)
                     : z;
    }
  }

  float m_st[2][4], l_st[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      m_st[i][r] = -1e30f;
      l_st[i][r] = 0.f;
    }
  f32x4 accO[2][8] = {};  // [row frag][d col group of 16]

  // causal: this block's rows reach q0+127 -> kv tiles up to that row.
  // KV tiles are prefetched through registers (2 chunks per matrix per
  // thread): tile t+1's HBM loads issue before tile t's compute — the
  // same single-LDS register-prefetch structure as the decode kernel.
  const int kv_max = causal ? min(s, q0 + 128) : s;
  const int ntiles = (kv_max + kTile - 1) / kTile;
  bf16x8 nk[2], nv[2];
  auto load_kv = [&](int tile, bf16x8 *rk, bf16x8 *rv) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int i = it * 256 + tid;
      int r = i / (kD / 8);
      int c = (i % (kD / 8)) * 8;
      int pos = tile * kTile + r;
      bf16x8 kv8{}, vv8{};
      if (pos < s) {
        // kb_stride lets k/v live in the KV cache ([b, max_len, kvh, D])
        size_t base =
            (size_t)b * kb_stride + ((size_t)pos * kvh + kh) * kD + c;
        kv8 = *(const bf16x8 *)(k + base);
        vv8 = *(const bf16x8 *)(v + base);
      }
      rk[it] = kv8;
      rv[it] = vv8;
    }
  };
  auto store_kv = [&](const bf16x8 *rk, const bf16x8 *rv) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      int i = it * 256 + tid;
      int r = i / (kD / 8);
      int c = (i % (kD / 8)) * 8;
      *(bf16x8 *)(&k_lds[r][c]) = rk[it];
      *(bf16x8 *)(&v_lds[r][c]) = rv[it];
    }
  };
  load_kv(0, nk, nv);
  for (int tile = 0; tile < ntiles; ++tile) {
    const int pos0 = tile * kTile;
    __syncthreads();  // previous tile's LDS reads retired
    store_kv(nk, nv);
    if (tile + 1 < ntiles) load_kv(tile + 1, nk, nv);
    __syncthreads();  // stores visible

    // S = Q K^T: B operand cols = kv pos (lane&15 + j*16)
    f32x4 accS[2][2] = {};
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 kf = *(const bf16x8 *)(
            &k_lds[(lane & 15) + 16 * j][ks * 32 + (lane >> 4) * 8]);
#pragma unroll
        for (int i = 0; i < 2; ++i)
          accS[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              qf[i][ks], kf, accS[i][j], 0, 0, 0);
      }
    }

    // online softmax per C-layout row (i, r); cols j*16 + lane&15
    float alpha[2][4];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = wrow0 + i * 16 + (lane >> 4) * 4 + r;
        float s0 = accS[i][0][r] * scale;
        float s1 = accS[i][1][r] * scale;
        int p0 = pos0 + (lane & 15);
        bool ok0 = (p0 < s) && (!causal || p0 <= row);
        bool ok1 = (p0 + 16 < s) && (!causal || p0 + 16 <= row);
        s0 = ok0 ? s0 : -1e30f;
        s1 = ok1 ? s1 : -1e30f;
        float mx = fmaxf(s0, s1);
#pragma unroll
        for (int off = 8; off > 0; off >>= 1)
          mx = fmaxf(mx, __shfl_xor(mx, off));
        float m_new = fmaxf(m_st[i][r], mx);
        float p0f = ok0 ? __expf(s0 - m_new) : 0.f;
        float p1f = ok1 ? __expf(s1 - m_new) : 0.f;
        float ps = p0f + p1f;
#pragma unroll
        for (int off = 8; off > 0; off >>= 1) ps += __shfl_xor(ps, off);
        alpha[i][r] = __expf(m_st[i][r] - m_new);
        l_st[i][r] = l_st[i][r] * alpha[i][r] + ps;
        m_st[i][r] = m_new;
        // store P to LDS (A-operand source)
        int lrow = wave * 32 + i * 16 + (lane >> 4) * 4 + r;
        p_lds[lrow][lane & 15] = (bf16)p0f;
        p_lds[lrow][(lane & 15) + 16] = (bf16)p1f;
      }
    __syncthreads();

    // PV: A = P[32 rows, 32 kv], B = V^T via tr_read; 8 d col-groups
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      bf16x8 af = *(const bf16x8 *)(
          &p_lds[wave * 32 + i * 16 + (lane & 15)][(lane >> 4) * 8]);
#pragma unroll
      for (int cg = 0; cg < 8; ++cg) {
        const int k0 = (lane >> 4) * 8;
        unsigned addr = (unsigned)(uintptr_t)(
            &v_lds[k0 + ((lane & 15) >> 2)][cg * 16 + 4 * (lane & 3)]);
        unsigned long long lo, hi;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %2\n"
            "ds_read_b64_tr_b16 %1, %2 offset:%3\n"
            "s_waitcnt lgkmcnt(0)"
            : "=v"(lo), "=v"(hi)
            : "v"(addr), "i"(4 * (kD + 8) * 2));
        bf16x8 bf;
        *(unsigned long long *)&bf = lo;
        *((unsigned long long *)&bf + 1) = hi;
#pragma unroll
        for (int r = 0; r < 4; ++r) accO[i][cg][r] *= alpha[i][r];
        accO[i][cg] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bf, accO[i][cg], 0, 0, 0);
      }
    }
    // NOTE: accO rescale uses alpha from THIS tile exactly once per
    // cg (applied inside the cg loop before each MFMA).
  }

#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = wrow0 + i * 16 + (lane >> 4) * 4 + r;
      if (row >= s) continue;
      float inv_l = l_st[i][r] > 0.f ? 1.f / l_st[i][r] : 0.f;
      bf16 *orow = out + (((size_t)b * s + row) * qh + h) * kD;
#pragma unroll
      for (int cg = 0; cg < 8; ++cg)
        orow[cg * 16 + (lane & 15)] = (bf16)(accO[i][cg][r] * inv_l);
      if (lse && (lane & 15) == 0)
        lse[((size_t)b * s + row) * qh + h] =
            m_st[i][r] + __logf(fmaxf(l_st[i][r], 1e-30f));
    }
}

void launch_flash_prefill(const void *q, const void *k, const void *v,
                          void *out, void *lse, int b, int s, int qh,
                          int kvh, float scale, bool causal,
                          hipStream_t stream, long kb_stride) {
  if (qh % kvh) throw std::runtime_error("flash_prefill: qh % kvh != 0");
  if (kb_stride == 0) kb_stride = (long)s * kvh * kD;
  dim3 grid((s + 127) / 128, qh, b);
  hipLaunchKernelGGL(k_flash_prefill, grid, dim3(256), 0, stream,
                     (const bf16 *)q, (const bf16 *)k, (const bf16 *)v,
                     (bf16 *)out, (float *)lse, s, qh, kvh, scale,
                     causal ? 1 : 0, kb_stride);
}

}  // namespace td
