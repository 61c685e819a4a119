// Standalone collectives over the symmetric heap:
//
//   * reduce_scatter — push-your-segment + chunked flags + local reduce
//     (the first half of the two-shot AR, exposed as its own op;
//     capability parity with the reference's standalone ReduceScatter,
//     Triton-distributed kernels/nvidia/reduce_scatter.py:710-831 —
//     behavior only).
//   * LL (low-latency) allgather — flag-in-payload protocol: every 4-byte
//     data word travels as an 8-byte (data, tag) pair stored with ONE
//     dwordx2 store, and the receiver polls the tag word of the payload
//     itself, so a single xGMI crossing carries both data and signal (no
//     separate flag write, no barrier). Tag is a monotonic device cell ->
//     hipGraph-replayable with no reset. Parity: the reference's
//     _pack_ll_block/_recv_ll_block small-message path
//     (kernels/nvidia/low_latency_allgather.py:531-589 — behavior only).
#include "td/api.hpp"

namespace td {

using bf16 = __bf16;

// --------------------------------------------------------------------------
// reduce_scatter
// --------------------------------------------------------------------------

// Each rank pushes its PEER-OWNED segments of x into the owner's inbox row
// [rank], chunked, releasing a per-(src,chunk) flag with the call tag.
// grid: (chunks, world-1).
__global__ void k_rs_push(PeerTable pt, const bf16 *__restrict__ x,
                          size_t inbox_off, size_t flags_off,
                          size_t seg_elems, int chunks,
                          const int *__restrict__ tag_cell) {
  const int pr = (int)blockIdx.y;
  const int peer = pr >= pt.rank ? pr + 1 : pr;
  const int chunk = (int)blockIdx.x;
  // chunk boundaries rounded to 8 elems so the 16B vector path stays
  // aligned (the op requires seg_elems % 8 == 0)
  const size_t per =
      (((seg_elems + chunks - 1) / chunks) + 7) & ~(size_t)7;
  const size_t lo = (size_t)chunk * per;
  const size_t hi = lo + per < seg_elems ? lo + per : seg_elems;
  bf16 *inbox = (bf16 *)((char *)pt.bases[peer] + inbox_off) +
                (size_t)pt.rank * seg_elems;
  const bf16 *src = x + (size_t)peer * seg_elems;
  typedef __attribute__((ext_vector_type(8))) bf16 v8;
  for (size_t i = lo + threadIdx.x * 8; i + 8 <= hi; i += blockDim.x * 8)
    *(v8 *)(inbox + i) = *(const v8 *)(src + i);
  __syncthreads();
  if (threadIdx.x == 0) {
    fence_release_sys();
    int *fl = (int *)((char *)pt.bases[peer] + flags_off);
    st_release<Scope::Sys>(fl + pt.rank * chunks + chunk, *tag_cell);
  }
}

// Local reduce: out = x[own segment] + sum over peers' inbox rows, chunked
// waits on the pushers' flags. grid: (chunks).
__global__ void k_rs_reduce(PeerTable pt, const bf16 *__restrict__ x,
                            const bf16 *__restrict__ inbox,
                            const int *__restrict__ flags,
                            bf16 *__restrict__ out, size_t seg_elems,
                            int chunks, const int *__restrict__ tag_cell) {
  const int chunk = (int)blockIdx.x;
  const int tag = *tag_cell;
  if (threadIdx.x < (unsigned)pt.world && (int)threadIdx.x != pt.rank)
    wait_ge_one<Scope::Sys>(flags + threadIdx.x * chunks + chunk, tag);
  __syncthreads();
  const size_t per =
      (((seg_elems + chunks - 1) / chunks) + 7) & ~(size_t)7;
  const size_t lo = (size_t)chunk * per;
  const size_t hi = lo + per < seg_elems ? lo + per : seg_elems;
  for (size_t i = lo + threadIdx.x; i < hi; i += blockDim.x) {
    float acc = (float)x[(size_t)pt.rank * seg_elems + i];
    for (int r = 0; r < pt.world; ++r)
      if (r != pt.rank) acc += (float)inbox[(size_t)r * seg_elems + i];
    out[i] = (bf16)acc;
  }
}

void launch_reduce_scatter(const PeerTable &pt, const void *x,
                           size_t inbox_off, size_t flags_off,
                           const void *local_inbox, const void *local_flags,
                           void *out, size_t seg_elems, int chunks,
                           const void *tag_cell, hipStream_t stream) {
  if (pt.world > 1) {
    hipLaunchKernelGGL(k_rs_push, dim3(chunks, pt.world - 1), dim3(256), 0,
                       stream, pt, (const bf16 *)x, inbox_off, flags_off,
                       seg_elems, chunks, (const int *)tag_cell);
  }
  hipLaunchKernelGGL(k_rs_reduce, dim3(chunks), dim3(256), 0, stream, pt,
                     (const bf16 *)x, (const bf16 *)local_inbox,
                     (const int *)local_flags, (bf16 *)out, seg_elems,
                     chunks, (const int *)tag_cell);
}

// --------------------------------------------------------------------------
// all_to_all_single: the push half IS k_rs_push (segment p -> rank p's
// inbox row [my rank]); this kernel is the receive half — flag-gated copy
// of the inbox rows (+ the local segment straight from x) into out.
// Parity: reference all_to_all_single_2d (kernels/nvidia a2a family —
// behavior only).
// --------------------------------------------------------------------------
__global__ void k_a2a_recv(PeerTable pt, const bf16 *__restrict__ x,
                           const bf16 *__restrict__ inbox,
                           const int *__restrict__ flags,
                           bf16 *__restrict__ out, size_t seg_elems,
                           int chunks, const int *__restrict__ tag_cell) {
  const int chunk = (int)blockIdx.x;
  const int src = (int)blockIdx.y;
  const int tag = *tag_cell;
  if (src != pt.rank && threadIdx.x == 0)
    wait_ge_one<Scope::Sys>(flags + src * chunks + chunk, tag);
  __syncthreads();
  const size_t per =
      (((seg_elems + chunks - 1) / chunks) + 7) & ~(size_t)7;
  const size_t lo = (size_t)chunk * per;
  const size_t hi = lo + per < seg_elems ? lo + per : seg_elems;
  const bf16 *srcp = src == pt.rank
                         ? x + (size_t)pt.rank * seg_elems
                         : inbox + (size_t)src * seg_elems;
  bf16 *dst = out + (size_t)src * seg_elems;
  typedef __attribute__((ext_vector_type(8))) bf16 v8;
  for (size_t i = lo + threadIdx.x * 8; i + 8 <= hi; i += blockDim.x * 8)
    *(v8 *)(dst + i) = *(const v8 *)(srcp + i);
}

void launch_all_to_all(const PeerTable &pt, const void *x, size_t inbox_off,
                       size_t flags_off, const void *local_inbox,
                       const void *local_flags, void *out, size_t seg_elems,
                       int chunks, const void *tag_cell,
                       hipStream_t stream) {
  if (pt.world > 1) {
    hipLaunchKernelGGL(k_rs_push, dim3(chunks, pt.world - 1), dim3(256), 0,
                       stream, pt, (const bf16 *)x, inbox_off, flags_off,
                       seg_elems, chunks, (const int *)tag_cell);
  }
  hipLaunchKernelGGL(k_a2a_recv, dim3(chunks, pt.world), dim3(256), 0,
                     stream, pt, (const bf16 *)x, (const bf16 *)local_inbox,
                     (const int *)local_flags, (bf16 *)out, seg_elems,
                     chunks, (const int *)tag_cell);
}

// --------------------------------------------------------------------------
// LL allgather
// --------------------------------------------------------------------------

// Pack + push: word i of my payload -> (data, tag) int2 in EVERY peer's
// inbox row [my rank]. grid: (blocks, world-1).
__global__ void k_ll_ag_push(PeerTable pt, const int *__restrict__ x,
                             size_t inbox_off, int words,
                             const int *__restrict__ tag_cell) {
  const int pr = (int)blockIdx.y;
  const int peer = pr >= pt.rank ? pr + 1 : pr;
  const int tag = *tag_cell;
  int2 *inbox = (int2 *)((char *)pt.bases[peer] + inbox_off) +
                (size_t)pt.rank * words;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < words;
       i += gridDim.x * blockDim.x) {
    int2 v;
    v.x = x[i];
    v.y = tag;
    // single 8B store: data and tag become visible together (LL invariant)
    __hip_atomic_store((unsigned long long *)&inbox[i],
                       *(unsigned long long *)&v, __ATOMIC_RELAXED,
                       __HIP_MEMORY_SCOPE_SYSTEM);
  }
}

// Poll + unpack: out[r][i] from inbox rows (peers) / from x (self).
// No fence and no barrier: the tag IS the readiness signal per word.
__global__ void k_ll_ag_recv(PeerTable pt, const int *__restrict__ x,
                             const int2 *__restrict__ inbox,
                             int *__restrict__ out, int words,
                             const int *__restrict__ tag_cell) {
  const int tag = *tag_cell;
  for (int j = blockIdx.x * blockDim.x + threadIdx.x;
       j < words * pt.world; j += gridDim.x * blockDim.x) {
    const int r = j / words;
    const int i = j - r * words;
    if (r == pt.rank) {
      out[j] = x[i];
      continue;
    }
    unsigned long long raw;
    unsigned long long t0 = wallclock();
    for (;;) {
      raw = __hip_atomic_load((const unsigned long long *)&inbox
                                  [(size_t)r * words + i],
                              __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
      int2 v = *(int2 *)&raw;
      if (v.y == tag) {
        out[j] = v.x;
        break;
      }
      if (wallclock() - t0 > TD_SPIN_TIMEOUT_TICKS) __builtin_trap();
    }
  }
}

void launch_ll_allgather(const PeerTable &pt, const void *x,
                         size_t inbox_off, const void *local_inbox,
                         void *out, int words, const void *tag_cell,
                         hipStream_t stream) {
  int blocks = (words + 255) / 256;
  if (blocks > 64) blocks = 64;
  if (pt.world > 1) {
    hipLaunchKernelGGL(k_ll_ag_push, dim3(blocks, pt.world - 1), dim3(256),
                       0, stream, pt, (const int *)x, inbox_off, words,
                       (const int *)tag_cell);
  }
  int rblocks = (words * pt.world + 255) / 256;
  if (rblocks > 256) rblocks = 256;
  hipLaunchKernelGGL(k_ll_ag_recv, dim3(rblocks), dim3(256), 0, stream, pt,
                     (const int *)x, (const int2 *)local_inbox, (int *)out,
                     words, (const int *)tag_cell);
}

}  // namespace td
