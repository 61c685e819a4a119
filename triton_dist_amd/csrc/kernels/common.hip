// Core communication kernels: cross-GPU barrier, signal set/wait, vectorized
// copy, fused put+signal. MI355X-native replacements for the reference's
// barrier/copy kernels (Triton-distributed
// python/triton_dist/kernels/amd/common_ops.py:58-150 — semantics only;
// implementation is hand-written HIP for gfx950).
#include <stdexcept>
#include <string>

#include "td/api.hpp"

namespace td {

// Device scratch for put_signal arrive counters (defined in module.hip;
// lives at a fixed slot in the local heap head).
unsigned *g_arrive_counter();

// ---------------------------------------------------------------------------
// barrier_all: every rank r release-stores `epoch` into peer p's
// flags[r] for all p, then spin-waits until its own flags[0..world) == epoch.
// Epoch-counting avoids a reset pass (cf. common_ops.py:122-150's
// atomic_cas all-to-all barrier).
// `local_flags` must point into the symmetric heap so peers' slots resolve
// via symm_at.
// ---------------------------------------------------------------------------
// The epoch counter lives in DEVICE memory (heap scratch) and is bumped by
// the kernel itself, so the whole op sequence is hipGraph-replayable (no
// host-side state baked into captured kernel arguments). Peers may already
// be one barrier ahead, hence wait_ge.
__global__ void k_barrier_all(PeerTable pt, int *local_flags, int *epoch_cell) {
  __shared__ int e_sh;
  if (threadIdx.x == 0) {
    int e = *epoch_cell + 1;
    *epoch_cell = e;
    e_sh = e;
  }
  __syncthreads();
  int e = e_sh;
  int t = threadIdx.x;
  if (t < pt.world) {
    int *peer_flags = symm_at(pt, local_flags, t);
    st_release<Scope::Sys>(peer_flags + pt.rank, e);
    wait_ge_one<Scope::Sys>(local_flags + t, e);
  }
  __syncthreads();
}

void launch_barrier_all(const PeerTable &pt, int *local_flags, int *epoch_cell,
                        hipStream_t stream) {
  hipLaunchKernelGGL(k_barrier_all, dim3(1), dim3(kWave), 0, stream, pt,
                     local_flags, epoch_cell);
}

__global__ void k_signal_set(int *flag, int val) {
  st_release<Scope::Sys>(flag, val);
}
void launch_signal_set(int *flag, int val, hipStream_t stream) {
  hipLaunchKernelGGL(k_signal_set, dim3(1), dim3(1), 0, stream, flag, val);
}

__global__ void k_wait_eq(const int *flags, int n, int expect) {
  wait_eq<Scope::Sys>(flags, n, expect);
}
void launch_wait_eq(const int *flags, int n, int expect, hipStream_t stream) {
  hipLaunchKernelGGL(k_wait_eq, dim3(1), dim3(256), 0, stream, flags, n,
                     expect);
}

__global__ void k_reset_flags(int *flags, int n, int val) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) st_relaxed<Scope::Sys>(flags + i, val);
}
void launch_reset_flags(int *flags, int n, int val, hipStream_t stream) {
  int blocks = (n + 255) / 256;
  hipLaunchKernelGGL(k_reset_flags, dim3(blocks), dim3(256), 0, stream, flags,
                     n, val);
}

// ---------------------------------------------------------------------------
// Vectorized grid-stride copy (SM path; the CP-engine path is
// hipMemcpyAsync from the host module). 16 B per lane per iteration.
// ---------------------------------------------------------------------------
__global__ void k_copy16(ulonglong2 *dst, const ulonglong2 *src, size_t n16) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += stride) dst[i] = src[i];
}
__global__ void k_copy1(char *dst, const char *src, size_t n) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = src[i];
}

static inline int copy_grid(size_t work_items) {
  size_t blocks = (work_items + 255) / 256;
  if (blocks > 2048) blocks = 2048;  // 256 CUs × 8 blocks
  if (blocks == 0) blocks = 1;
  return (int)blocks;
}

void launch_copy(void *dst, const void *src, size_t nbytes, hipStream_t stream) {
  if (nbytes % 16 == 0 && ((uintptr_t)dst % 16 == 0) &&
      ((uintptr_t)src % 16 == 0)) {
    size_t n16 = nbytes / 16;
    hipLaunchKernelGGL(k_copy16, dim3(copy_grid(n16)), dim3(256), 0, stream,
                       (ulonglong2 *)dst, (const ulonglong2 *)src, n16);
  } else {
    hipLaunchKernelGGL(k_copy1, dim3(copy_grid(nbytes)), dim3(256), 0, stream,
                       (char *)dst, (const char *)src, nbytes);
  }
}

// ---------------------------------------------------------------------------
// Pull-mode allgather: block (peer, chunk) spin-waits the SOURCE rank's
// published chunk flag over xGMI, then pulls the chunk into the local
// gathered workspace (cf. reference kernels/amd/allgather.py pull variants
// :173-313 — behavior only). Pull reads ride the consumer GPU's own xGMI
// links, which balances differently than push under contention.
// ---------------------------------------------------------------------------
__global__ void k_ag_pull(PeerTable pt, size_t ws_off, size_t flags_off,
                          size_t seg_bytes, int chunks, int chunk_stride) {
  const int pi = blockIdx.y;  // peer index (skip self)
  const int peer = (pt.rank + 1 + pi) % pt.world;
  const int c = blockIdx.x;
  // 16B-aligned chunk boundaries: the copy is ulonglong2-vectorized and
  // an unaligned tail would overrun the segment into the next one
  const size_t per = (((seg_bytes + chunks - 1) / chunks) + 15) &
                     ~(size_t)15;
  const size_t lo = (size_t)c * per;
  const size_t hi = min(lo + per, seg_bytes);
  const int *pf = (const int *)((char *)pt.bases[peer] + flags_off);
  if (threadIdx.x == 0)
    wait_ge_one<Scope::Sys>(pf + peer * chunk_stride + c, 1);
  __syncthreads();
  const ulonglong2 *src =
      (const ulonglong2 *)((char *)pt.bases[peer] + ws_off +
                           (size_t)peer * seg_bytes + lo);
  ulonglong2 *dst = (ulonglong2 *)((char *)pt.bases[pt.rank] + ws_off +
                                   (size_t)peer * seg_bytes + lo);
  for (size_t i = threadIdx.x; i * 16 < hi - lo; i += blockDim.x)
    dst[i] = src[i];
}

void launch_ag_pull(const PeerTable &pt, size_t ws_off, size_t flags_off,
                    size_t seg_bytes, int chunks, int chunk_stride,
                    hipStream_t stream) {
  if (pt.world <= 1) return;
  if (seg_bytes % 16)
    throw std::runtime_error("ag_pull: seg_bytes % 16 != 0");
  hipLaunchKernelGGL(k_ag_pull, dim3(chunks, pt.world - 1), dim3(256), 0,
                     stream, pt, ws_off, flags_off, seg_bytes, chunks,
                     chunk_stride);
}

// ---------------------------------------------------------------------------
// put_signal: multi-block copy; last block to arrive does a system release
// fence and sets the flag. Self-contained (no stream-order dependence), the
// analog of rocshmem_putmem_signal (shmem/rocshmem_bind/runtime/
// rocshmem_wrapper.cc:29-201 — behavior only).
// ---------------------------------------------------------------------------
__global__ void k_put_signal16(ulonglong2 *dst, const ulonglong2 *src,
                               size_t n16, int *flag, int val, int add,
                               unsigned *arrive, unsigned nblocks) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n16; i += stride) dst[i] = src[i];
  __syncthreads();
  if (threadIdx.x == 0) {
    fence_release_sys();
    unsigned prev = atomic_add<Scope::Gpu>(arrive, 1u);
    if (prev == nblocks - 1) {
      st_relaxed<Scope::Gpu>(arrive, 0u);
      if (add)
        atomic_add<Scope::Sys>(flag, val);
      else
        st_release<Scope::Sys>(flag, val);
    }
  }
}

void launch_put_signal(const PeerTable &pt, void *dst, const void *src,
                       size_t nbytes, int *flag, int val, int add,
                       hipStream_t stream) {
  if (nbytes % 16 != 0 || ((uintptr_t)dst % 16) || ((uintptr_t)src % 16))
    throw std::runtime_error("put_signal requires 16B-aligned size/ptrs");
  size_t n16 = nbytes / 16;
  int blocks = copy_grid(n16);
  hipLaunchKernelGGL(k_put_signal16, dim3(blocks), dim3(256), 0, stream,
                     (ulonglong2 *)dst, (const ulonglong2 *)src, n16, flag, val,
                     add, g_arrive_counter(), (unsigned)blocks);
}

}  // namespace td
