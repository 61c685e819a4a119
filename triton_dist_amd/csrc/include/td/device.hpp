// td/device.hpp — MI355X (gfx950/CDNA4) device-side primitives for
// compute-communication overlap.
//
// This header is the MI355X-native replacement for the reference's
// Distributed MLIR dialect (Triton-distributed
// include/TritonDistributed/Dialect/Distributed/IR/DistributedOps.td:45-168
// and its AMD lowering lib/Conversion/TritonDistributedToLLVM/AMD/
// DistributedOpToLLVM.cpp:201-332): wait = per-thread strided spin on a
// scoped atomic load followed by a workgroup barrier; notify = scoped
// release store / atomic add through a peer-translated pointer;
// consume_token is a pure compiler dataflow edge and needs nothing here.
//
// Memory-model notes (CDNA4):
//  * Per-XCD L2s are NOT coherent. An acquire at agent ("gpu") or system
//    scope emits the L2 invalidate needed before reading data written by
//    another XCD or another GPU. A release at system scope writes back.
//  * xGMI supports native cross-GPU atomics; flags live in the symmetric
//    heap and are operated on with __hip_atomic_* at SYSTEM scope.
#pragma once

#include <hip/hip_runtime.h>
#include <stdint.h>

#define TD_DEV __device__ __forceinline__

namespace td {

constexpr int kMaxRanks = 8;       // one MI355X node = 8 GPUs over xGMI
constexpr int kWave = 64;          // CDNA wavefront

// ---------------------------------------------------------------------------
// Scoped atomics on int32/int64 flags.
// ---------------------------------------------------------------------------
enum class Scope : int { Cta = 0, Gpu = 1, Sys = 2 };

template <Scope S> TD_DEV constexpr int hip_scope() {
  return S == Scope::Cta   ? __HIP_MEMORY_SCOPE_WORKGROUP
         : S == Scope::Gpu ? __HIP_MEMORY_SCOPE_AGENT
                           : __HIP_MEMORY_SCOPE_SYSTEM;
}

template <Scope S = Scope::Sys, typename T>
TD_DEV T ld_acquire(const T *p) {
  return __hip_atomic_load(p, __ATOMIC_ACQUIRE, hip_scope<S>());
}
template <Scope S = Scope::Sys, typename T>
TD_DEV T ld_relaxed(const T *p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, hip_scope<S>());
}
template <Scope S = Scope::Sys, typename T>
TD_DEV void st_release(T *p, T v) {
  __hip_atomic_store(p, v, __ATOMIC_RELEASE, hip_scope<S>());
}
template <Scope S = Scope::Sys, typename T>
TD_DEV void st_relaxed(T *p, T v) {
  __hip_atomic_store(p, v, __ATOMIC_RELAXED, hip_scope<S>());
}
template <Scope S = Scope::Sys, typename T>
TD_DEV T atomic_add(T *p, T v) {
  return __hip_atomic_fetch_add(p, v, __ATOMIC_ACQ_REL, hip_scope<S>());
}
template <Scope S = Scope::Sys, typename T>
TD_DEV T atomic_cas(T *p, T cmp, T val) {
  __hip_atomic_compare_exchange_strong(p, &cmp, val, __ATOMIC_ACQ_REL,
                                       __ATOMIC_ACQUIRE, hip_scope<S>());
  return cmp;
}

// Release fence covering all prior global writes (system scope: visible to
// peer GPUs after a subsequent release store / signal).
TD_DEV void fence_release_sys() {
  __atomic_thread_fence(__ATOMIC_RELEASE);  // compiler fence
  __builtin_amdgcn_fence(__ATOMIC_RELEASE, "");  // system-scope HW fence
}
TD_DEV void fence_acquire_sys() {
  __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "");
}

// ---------------------------------------------------------------------------
// wait — spec from the reference AMD lowering (DistributedOpToLLVM.cpp:
// 201-319): every thread of the workgroup strides over the flag array,
// spinning with a scoped atomic load until each flag reaches `expect`;
// a workgroup barrier closes the op. Acquire semantics make subsequent
// data reads safe.
// ---------------------------------------------------------------------------
// Spin-timeout: a wedged flag traps (kernel aborts with an error the host
// sees) instead of hanging the GPU forever. ~100 MHz wall clock on CDNA;
// 3e9 ticks ≈ 30 s.
#ifndef TD_SPIN_TIMEOUT_TICKS
#define TD_SPIN_TIMEOUT_TICKS 3000000000ull
#endif

TD_DEV uint64_t wallclock() { return __builtin_amdgcn_s_memrealtime(); }

template <Scope S = Scope::Sys>
TD_DEV void wait_eq(const int *flags, int n, int expect) {
  const int tid = threadIdx.x + threadIdx.y * blockDim.x;
  const int nthreads = blockDim.x * blockDim.y * blockDim.z;
  for (int i = tid; i < n; i += nthreads) {
    uint64_t t0 = wallclock();
    while (ld_acquire<S>(flags + i) != expect) {
      __builtin_amdgcn_s_sleep(2);
      if (wallclock() - t0 > TD_SPIN_TIMEOUT_TICKS) __builtin_trap();
    }
  }
  __syncthreads();
}

// Single-thread variant (caller handles divergence / barriers).
template <Scope S = Scope::Sys>
TD_DEV void wait_eq_one(const int *flag, int expect) {
  uint64_t t0 = wallclock();
  while (ld_acquire<S>(flag) != expect) {
    __builtin_amdgcn_s_sleep(2);
    if (wallclock() - t0 > TD_SPIN_TIMEOUT_TICKS) __builtin_trap();
  }
}

template <Scope S = Scope::Sys>
TD_DEV void wait_ge_one(const int *flag, int bound) {
  uint64_t t0 = wallclock();
  while (ld_acquire<S>(flag) < bound) {
    __builtin_amdgcn_s_sleep(2);
    if (wallclock() - t0 > TD_SPIN_TIMEOUT_TICKS) __builtin_trap();
  }
}

// notify — release-store (SET) or atomic add (ADD) on a peer flag.
enum class SignalOp : int { Set = 0, Add = 1 };

template <Scope S = Scope::Sys>
TD_DEV void notify(int *flag, int val, SignalOp op = SignalOp::Set) {
  if (op == SignalOp::Set) {
    st_release<S>(flag, val);
  } else {
    atomic_add<S>(flag, val);
  }
}

// ---------------------------------------------------------------------------
// Symmetric-heap peer translation. The host passes a table of per-rank heap
// base pointers (mapped through hipIpc); symm_at translates a pointer in
// the local heap to the same offset in `rank`'s heap — the analog of the
// reference `distributed.symm_at` op / rocshmem_ptr.
// ---------------------------------------------------------------------------
struct PeerTable {
  void *bases[kMaxRanks];
  int rank;
  int world;
};

template <typename T>
TD_DEV T *symm_at(const PeerTable &pt, T *local_p, int rank) {
  intptr_t off = reinterpret_cast<intptr_t>(local_p) -
                 reinterpret_cast<intptr_t>(pt.bases[pt.rank]);
  return reinterpret_cast<T *>(reinterpret_cast<char *>(pt.bases[rank]) + off);
}

// ---------------------------------------------------------------------------
// XCD-aware workgroup remap (bijective for any nwg; MI355X has 8 XCDs and
// the command processor round-robins blockIdx across them — remapping gives
// each XCD a contiguous chunk of the tile space for L2 locality).
// ---------------------------------------------------------------------------
TD_DEV int xcd_remap(int wgid, int nwg, int nxcd = 8) {
  if (nwg < nxcd) return wgid;
  int xcd = wgid % nxcd;
  int idx = wgid / nxcd;
  int q = nwg / nxcd, r = nwg % nxcd;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

}  // namespace td
