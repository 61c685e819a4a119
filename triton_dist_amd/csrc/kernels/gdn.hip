// GDN (gated delta net) decode-step kernel — Qwen3-Next geometry.
//
// One block per (batch, head); the fp32 recurrent state S[K][V] lives in
// HBM and is touched exactly twice (read for the S^T k contraction, then
// fused decay+rank-1-update+output pass), so the step is state-bandwidth
// bound: 3*K*V*4 bytes per head. Thread t owns column v = t % V and row
// half t / V; column accesses are coalesced (consecutive threads read
// consecutive v of the same row).
//
// Recurrence (matches ops/gdn.py gated_delta_rule_recurrent_ref):
//   S    = e^g * S
//   vhat = beta * (v - S^T k)
//   S   += k vhat^T
//   o    = scale * S^T q
//
// Capability parity (behavior only): Triton-distributed
// python/triton_dist/kernels/nvidia/gdn.py (chunked fwd; the decode step
// is its T=1 specialization).
#include <stdexcept>

#include "td/api.hpp"

namespace td {

using bf16 = __bf16;

__global__ void k_gdn_decode(const bf16 *__restrict__ q,
                             const bf16 *__restrict__ k,
                             const bf16 *__restrict__ v,
                             const float *__restrict__ g,
                             const float *__restrict__ beta,
                             float *__restrict__ state,
                             bf16 *__restrict__ o, int H, int K, int V,
                             float scale) {
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int tid = threadIdx.x;
  const int vcol = tid % V;
  const int half = tid / V;          // 0 or 1: row range [half*K/2, ...)
  const int rows = K / 2;
  const int r0 = half * rows;

  extern __shared__ float sm[];
  float *kk = sm;                    // [K]
  float *qq = sm + K;                // [K]
  float *red = sm + 2 * K;           // [2][V] partial sums
  float *vhat = sm + 2 * K + 2 * V;  // [V]

  const size_t base_in = ((size_t)b * H + h);
  for (int i = tid; i < K; i += blockDim.x) {
    kk[i] = (float)k[base_in * K + i];
    qq[i] = (float)q[base_in * K + i];
  }
  __syncthreads();

  float *S = state + base_in * (size_t)K * V;
  const float eg = __expf(g[base_in]);
  const float bt = beta[base_in];

  // pass 1: sk[v] = e^g * sum_k S[k][v] * k[k]
  float acc = 0.f;
  for (int r = 0; r < rows; ++r)
    acc += S[(size_t)(r0 + r) * V + vcol] * kk[r0 + r];
  red[half * V + vcol] = acc;
  __syncthreads();
  if (half == 0) {
    float sk = eg * (red[vcol] + red[V + vcol]);
    vhat[vcol] = bt * ((float)v[base_in * V + vcol] - sk);
  }
  __syncthreads();

  // pass 2: S[k][v] = e^g*S + k[k]*vhat[v]; o[v] = scale * sum_k S*q
  const float vh = vhat[vcol];
  acc = 0.f;
  for (int r = 0; r < rows; ++r) {
    size_t idx = (size_t)(r0 + r) * V + vcol;
    float s = eg * S[idx] + kk[r0 + r] * vh;
    S[idx] = s;
    acc += s * qq[r0 + r];
  }
  red[half * V + vcol] = acc;
  __syncthreads();
  if (half == 0)
    o[base_in * V + vcol] = (bf16)(scale * (red[vcol] + red[V + vcol]));
}

void launch_gdn_decode(const void *q, const void *k, const void *v,
                       const void *g, const void *beta, void *state,
                       void *o, int B, int H, int K, int V, float scale,
                       hipStream_t stream) {
  if (K % 2 || K > 1024 || V > 512)
    throw std::runtime_error("gdn_decode: K%2==0, K<=1024, V<=512");
  size_t smem = (2 * K + 3 * V) * sizeof(float);
  hipLaunchKernelGGL(k_gdn_decode, dim3(B, H), dim3(2 * V), smem, stream,
                     (const bf16 *)q, (const bf16 *)k, (const bf16 *)v,
                     (const float *)g, (const float *)beta, (float *)state,
                     (bf16 *)o, H, K, V, scale);
}

}  // namespace td
