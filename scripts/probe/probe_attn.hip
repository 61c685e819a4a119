// Decomposition probe for k_flash_decode: which phase costs what.
// Variants: 0=full (mirror of production), 1=no PV loop, 2=no QK/softmax
// (p=1/32), 3=no K-stage (reads garbage), 4=PV only w/ LDS-staged V.
#include <hip/hip_runtime.h>
#include <cstdio>

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(4))) bf16 bf16x4;
constexpr int kD = 128;
constexpr int kTile = 32;

template <int VAR>
__global__ __launch_bounds__(256) void k_fd(
    const bf16 *__restrict__ q, const bf16 *__restrict__ kcache,
    const bf16 *__restrict__ vcache, bf16 *__restrict__ out,
    const long *__restrict__ offset, int qh, int kvh, int max_len,
    float scale) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = qh / kvh;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = tid >> 5;
  const int t = tid & 31;
  const long seqlen = *offset + 1;

  __shared__ bf16 k_lds[kTile][kD];
  __shared__ bf16 q_lds[16][kD];
  __shared__ float s_part[4][16][kTile];
  __shared__ float p_lds[8][kTile];
  __shared__ float m_lds[8], r_lds[8], l_lds[8];

  for (int i = tid; i < 16 * kD / 8; i += 256) {
    int hh = i / (kD / 8);
    int c = (i % (kD / 8)) * 8;
    bf16x8 v{};
    if (hh < G)
      v = *(const bf16x8 *)(q + (((size_t)b * qh) + kh * G + hh) * kD + c);
    *(bf16x8 *)(&q_lds[hh][c]) = v;
  }
  if (tid < 8) { m_lds[tid] = -1e30f; l_lds[tid] = 0.f; }
  __syncthreads();
  bf16x8 qfrag = *(const bf16x8 *)(
      &q_lds[lane & 15][wave * 32 + (lane >> 4) * 8]);

  float acc[4] = {};
  const int my_d0 = t * 4;
  const long ntiles = (seqlen + kTile - 1) / kTile;
  for (long tile = 0; tile < ntiles; ++tile) {
    const long pos0 = tile * kTile;
    __syncthreads();
    if (VAR != 3) {
      for (int i = tid; i < kTile * kD / 8; i += 256) {
        int r = i / (kD / 8);
        int c = (i % (kD / 8)) * 8;
        long pos = pos0 + r;
        bf16x8 kv{};
        if (pos < seqlen) {
          size_t base = (((size_t)b * max_len + pos) * kvh + kh) * kD + c;
          kv = *(const bf16x8 *)(kcache + base);
        }
        *(bf16x8 *)(&k_lds[r][c]) = kv;
      }
    }
    __syncthreads();

    if (VAR != 2) {
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

      float s = -1e30f;
      if (g < G && pos0 + t < seqlen) {
        s = (s_part[0][g][t] + s_part[1][g][t] + s_part[2][g][t] +
             s_part[3][g][t]) * scale;
      }
      float mx = s;
      for (int off = 16; off > 0; off >>= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off));
      float m_old = m_lds[g];
      float m_new = fmaxf(m_old, mx);
      float p = (s > -1e29f) ? __expf(s - m_new) : 0.f;
      p_lds[g][t] = p;
      float psum = p;
      for (int off = 16; off > 0; off >>= 1) psum += __shfl_xor(psum, off);
      if (t == 0) {
        float r = __expf(m_old - m_new);
        r_lds[g] = r;
        l_lds[g] = l_lds[g] * r + psum;
        m_lds[g] = m_new;
      }
      __syncthreads();
    } else {
      p_lds[g][t] = 1.f / 32.f;
      if (t == 0) { r_lds[g] = 1.f; l_lds[g] = 1.f; }
      __syncthreads();
    }
    if (VAR != 1) {
      const float r = r_lds[g];
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[j] *= r;
      const long lim = min((long)kTile, seqlen - pos0);
      const bf16 *vbase = vcache +
          (((size_t)b * max_len + pos0) * kvh + kh) * kD + my_d0;
      for (int tt = 0; tt < (int)lim; ++tt) {
        float p2 = p_lds[g][tt];
        bf16x4 vv = *(const bf16x4 *)(vbase + (size_t)tt * kvh * kD);
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[j] += p2 * (float)vv[j];
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


// -------- tr_read semantics probe: LDS[i] = i, read via ds_read_b64_tr_b16
__global__ void k_trprobe(float *out) {
  __shared__ bf16 lds[512];
  const int tid = threadIdx.x;
  for (int i = tid; i < 512; i += 64) lds[i] = (bf16)(float)i;
  __syncthreads();
  if (tid < 64) {
    int l = tid;
    // pattern selected by blockIdx.y: 0 = all-lanes addr 0;
    // 1 = addr (l&3)*8B (quad lanes point at consecutive 8B words);
    // 2 = addr (l>>4)*128B (uniform per 16-lane group);
    // 3 = addr l*8B (linear b64)
    unsigned e;
    switch (blockIdx.y) {
      case 0: e = 0; break;
      case 1: e = (l & 3) * 4; break;
      case 2: e = (l >> 4) * 64; break;
      default: e = l * 4; break;
    }
    unsigned addr = (unsigned)(uintptr_t)(&lds[e]);
    unsigned long long lo;
    asm volatile(
        "ds_read_b64_tr_b16 %0, %1\n"
        "s_waitcnt lgkmcnt(0)"
        : "=v"(lo)
        : "v"(addr));
    bf16x4 a = *(bf16x4 *)&lo;
    for (int j = 0; j < 4; ++j)
      out[(blockIdx.y * 64 + l) * 4 + j] = (float)a[j];
  }
}

// -------- VAR4: PV on MFMA with subtiled LDS V + tr_read B-fragments
__global__ __launch_bounds__(256) void k_fd_mfma(
    const bf16 *__restrict__ q, const bf16 *__restrict__ kcache,
    const bf16 *__restrict__ vcache, bf16 *__restrict__ out,
    const long *__restrict__ offset, int qh, int kvh, int max_len,
    float scale) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = qh / kvh;
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int g = tid >> 5;
  const int t = tid & 31;
  const long seqlen = *offset + 1;

  __shared__ bf16 k_lds[kTile][kD];
  // V row-major, rows padded +8 elems so tr_read's 4-rows-x-4-colchunks
  // per 16-lane group hit rotated banks (row stride 272B = 4-bank rotate)
  __shared__ bf16 v_lds[kTile][kD + 8];
  __shared__ bf16 q_lds[16][kD];
  __shared__ bf16 p_bf[16][kTile + 8];  // padded: A-frag rows conflict-free
  __shared__ float s_part[4][16][kTile];
  __shared__ float m_lds[8], r_lds[16], l_lds[8];

  for (int i = tid; i < 16 * kD / 8; i += 256) {
    int hh = i / (kD / 8);
    int c = (i % (kD / 8)) * 8;
    bf16x8 v{};
    if (hh < G)
      v = *(const bf16x8 *)(q + (((size_t)b * qh) + kh * G + hh) * kD + c);
    *(bf16x8 *)(&q_lds[hh][c]) = v;
  }
  if (tid < 8) { m_lds[tid] = -1e30f; l_lds[tid] = 0.f; }
  if (tid < 16) {
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
        size_t base = (((size_t)b * max_len + pos) * kvh + kh) * kD + c;
        kv = *(const bf16x8 *)(kcache + base);
        vv = *(const bf16x8 *)(vcache + base);
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

    float s = -1e30f;
    if (g < G && pos0 + t < seqlen) {
      s = (s_part[0][g][t] + s_part[1][g][t] + s_part[2][g][t] +
           s_part[3][g][t]) * scale;
    }
    float mx = s;
    for (int off = 16; off > 0; off >>= 1)
      mx = fmaxf(mx, __shfl_xor(mx, off));
    float m_old = m_lds[g];
    float m_new = fmaxf(m_old, mx);
    float p = (s > -1e29f) ? __expf(s - m_new) : 0.f;
    p_bf[g][t] = (bf16)p;
    float psum = p;
    for (int off = 16; off > 0; off >>= 1) psum += __shfl_xor(psum, off);
    if (t == 0) {
      float r = __expf(m_old - m_new);
      r_lds[g] = r;
      l_lds[g] = l_lds[g] * r + psum;
      m_lds[g] = m_new;
    }
    __syncthreads();

    // MFMA PV: wave w covers output cols w*32..w*32+31 (2 col-groups)
    bf16x8 afrag = *(const bf16x8 *)(&p_bf[lane & 15][(lane >> 4) * 8]);
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      const int cg = wave * 2 + h;
      // tr_read semantics (measured, scripts/probe): within a 16-lane
      // group, lane k supplies an 8B-word address; lane l receives elem
      // (l&3) of the words fetched by lanes ((l&15)>>2)+4j, j=0..3. So
      // lane l pointing at V row k0+((l&15)>>2), col-chunk 4*(l&3) makes
      // the instruction deliver V[k0+j][col l&15] — B-frag col-major for
      // MFMA with V staged PLAIN row-major. Second read: rows k0+4..7;
      // cols 16..31 of the wave's 32-col slice come from cg=1 (col+16).
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

extern "C" void run_trprobe(void *out, void *stream) {
  hipLaunchKernelGGL(k_trprobe, dim3(1, 4), dim3(64), 0, (hipStream_t)stream,
                     (float *)out);
}

extern "C" void run_fd(int var, const void *q, const void *k, const void *v,
                       void *out, const void *off, int batch, int qh,
                       int kvh, int max_len, void *stream) {
  float scale = 1.f / sqrtf((float)kD);
  dim3 grid(batch, kvh), blk(256);
  hipStream_t s = (hipStream_t)stream;
#define L(V) hipLaunchKernelGGL(k_fd<V>, grid, blk, 0, s, (const bf16 *)q, \
    (const bf16 *)k, (const bf16 *)v, (bf16 *)out, (const long *)off, qh, \
    kvh, max_len, scale)
  switch (var) { case 0: L(0); break; case 1: L(1); break;
                 case 2: L(2); break; case 3: L(3); break;
                 case 4: hipLaunchKernelGGL(k_fd_mfma, grid, blk, 0, s,
                     (const bf16 *)q, (const bf16 *)k, (const bf16 *)v,
                     (bf16 *)out, (const long *)off, qh, kvh, max_len,
                     scale); break; }
#undef L
}
