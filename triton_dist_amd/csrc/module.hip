// triton_dist_amd._C — pybind11 module: hipIpc symmetric heap, DLPack tensor
// export, stream ops, and kernel launchers for gfx950.
//
// MI355X-native equivalent of the reference's pyrocshmem host binding
// (Triton-distributed shmem/rocshmem_bind/pyrocshmem/src/pyrocshmem.cc:87-137
// and python/pyrocshmem/__init__.py:47-151): same capabilities —
// symmetric-heap init by exchanged unique handles, peer pointer translation,
// per-peer tensor views, stream barrier — implemented directly on
// hipIpcMemHandle + HIP kernels instead of rocSHMEM.
//
// Tensors cross the Python boundary as DLPack capsules (torch.from_dlpack),
// so this module links only against amdhip64 — no torch C++ ABI coupling.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include "td/api.hpp"

namespace py = pybind11;
using namespace td;

// ---------------------------------------------------------------------------
// Minimal DLPack (v0.8 layout) — enough for torch.from_dlpack.
// ---------------------------------------------------------------------------
extern "C" {
typedef struct {
  int32_t device_type;
  int32_t device_id;
} DLDevice;
typedef struct {
  uint8_t code;
  uint8_t bits;
  uint16_t lanes;
} DLDataType;
typedef struct {
  void *data;
  DLDevice device;
  int32_t ndim;
  DLDataType dtype;
  int64_t *shape;
  int64_t *strides;
  uint64_t byte_offset;
} DLTensor;
typedef struct DLManagedTensor {
  DLTensor dl_tensor;
  void *manager_ctx;
  void (*deleter)(struct DLManagedTensor *);
} DLManagedTensor;
}
static constexpr int kDLROCM = 10;

// ---------------------------------------------------------------------------
// Symmetric heap state
// ---------------------------------------------------------------------------
namespace {

struct HeapState {
  void *bases[kMaxRanks] = {nullptr};
  bool opened[kMaxRanks] = {false};
  size_t size = 0;
  int rank = -1;
  int world = 0;
  int device = -1;
  bool fine_grained = false;
  PeerTable pt{};
  bool active = false;
};
HeapState g_heap;

// Ring of arrive counters for put_signal (slot 0.. in the heap head scratch).
constexpr size_t kScratchBytes = 4096;  // heap head reserved for internals
constexpr int kArriveSlots = 256;
std::atomic<uint32_t> g_arrive_next{0};

void check_active() {
  if (!g_heap.active) throw std::runtime_error("symmetric heap not initialized");
}

hipStream_t as_stream(uintptr_t s) { return reinterpret_cast<hipStream_t>(s); }

}  // namespace

namespace td {
unsigned *g_arrive_counter() {
  uint32_t slot = g_arrive_next.fetch_add(1) % kArriveSlots;
  return reinterpret_cast<unsigned *>(static_cast<char *>(g_heap.bases[g_heap.rank])) + slot;
}
}  // namespace td

static py::bytes heap_init(int rank, int world, int device, size_t size,
                           bool fine_grained) {
  if (g_heap.active) throw std::runtime_error("heap already initialized");
  if (world > kMaxRanks) throw std::runtime_error("world > kMaxRanks");
  TD_CHECK_HIP(hipSetDevice(device));
  void *base = nullptr;
  if (fine_grained) {
    hipError_t e = hipExtMallocWithFlags(&base, size, hipDeviceMallocFinegrained);
    if (e != hipSuccess) {
      fine_grained = false;
      base = nullptr;
    }
  }
  if (!base) TD_CHECK_HIP(hipMalloc(&base, size));
  TD_CHECK_HIP(hipMemset(base, 0, size));
  TD_CHECK_HIP(hipDeviceSynchronize());
  g_heap.bases[rank] = base;
  g_heap.size = size;
  g_heap.rank = rank;
  g_heap.world = world;
  g_heap.device = device;
  g_heap.fine_grained = fine_grained;
  hipIpcMemHandle_t handle;
  TD_CHECK_HIP(hipIpcGetMemHandle(&handle, base));
  return py::bytes(reinterpret_cast<const char *>(&handle), sizeof(handle));
}

static void heap_open(const std::vector<py::bytes> &handles) {
  if ((int)handles.size() != g_heap.world)
    throw std::runtime_error("handles size != world");
  for (int r = 0; r < g_heap.world; ++r) {
    if (r == g_heap.rank) continue;
    std::string h = handles[r];
    if (h.size() != sizeof(hipIpcMemHandle_t))
      throw std::runtime_error("bad ipc handle size");
    hipIpcMemHandle_t handle;
    std::memcpy(&handle, h.data(), sizeof(handle));
    void *p = nullptr;
    TD_CHECK_HIP(hipIpcOpenMemHandle(&p, handle, hipIpcMemLazyEnablePeerAccess));
    g_heap.bases[r] = p;
    g_heap.opened[r] = true;
  }
  g_heap.pt.rank = g_heap.rank;
  g_heap.pt.world = g_heap.world;
  for (int r = 0; r < g_heap.world; ++r) g_heap.pt.bases[r] = g_heap.bases[r];
  g_heap.active = true;
}

static void heap_close() {
  if (g_heap.rank < 0) return;
  for (int r = 0; r < g_heap.world; ++r) {
    if (g_heap.opened[r] && g_heap.bases[r]) {
      (void)hipIpcCloseMemHandle(g_heap.bases[r]);
    }
  }
  if (g_heap.bases[g_heap.rank]) (void)hipFree(g_heap.bases[g_heap.rank]);
  g_heap = HeapState{};
}

static uintptr_t heap_base(int rank) {
  if (rank < 0 || rank >= g_heap.world || !g_heap.bases[rank])
    throw std::runtime_error("heap_base: bad rank / not mapped");
  return reinterpret_cast<uintptr_t>(g_heap.bases[rank]);
}

static size_t heap_scratch_bytes() { return kScratchBytes; }

// ---------------------------------------------------------------------------
// DLPack export of a heap (or arbitrary device-pointer) region.
// ---------------------------------------------------------------------------
struct DLCtx {
  std::vector<int64_t> shape;
  DLManagedTensor mt;
};

static void dl_deleter(DLManagedTensor *mt) {
  delete static_cast<DLCtx *>(mt->manager_ctx);
}

static void capsule_destructor(PyObject *cap) {
  if (PyCapsule_IsValid(cap, "dltensor")) {
    auto *mt = static_cast<DLManagedTensor *>(PyCapsule_GetPointer(cap, "dltensor"));
    if (mt && mt->deleter) mt->deleter(mt);
  }
}

static py::object dlpack_from_ptr(uintptr_t ptr, std::vector<int64_t> shape,
                                  int code, int bits, int device_id) {
  auto *ctx = new DLCtx();
  ctx->shape = std::move(shape);
  DLTensor &t = ctx->mt.dl_tensor;
  t.data = reinterpret_cast<void *>(ptr);
  t.device = {kDLROCM, device_id};
  t.ndim = (int32_t)ctx->shape.size();
  t.dtype = {(uint8_t)code, (uint8_t)bits, 1};
  t.shape = ctx->shape.data();
  t.strides = nullptr;  // contiguous
  t.byte_offset = 0;
  ctx->mt.manager_ctx = ctx;
  ctx->mt.deleter = dl_deleter;
  PyObject *cap = PyCapsule_New(&ctx->mt, "dltensor", capsule_destructor);
  return py::reinterpret_steal<py::object>(cap);
}

// ---------------------------------------------------------------------------
// Stream ops
// ---------------------------------------------------------------------------
static void memcpy_async(uintptr_t dst, uintptr_t src, size_t nbytes,
                         uintptr_t stream) {
  TD_CHECK_HIP(hipMemcpyAsync(reinterpret_cast<void *>(dst),
                              reinterpret_cast<void *>(src), nbytes,
                              hipMemcpyDeviceToDevice, as_stream(stream)));
}

static void memset32_async(uintptr_t ptr, int value, size_t count,
                           uintptr_t stream) {
  TD_CHECK_HIP(hipMemsetD32Async(reinterpret_cast<hipDeviceptr_t>(ptr), value,
                                 count, as_stream(stream)));
}

static void device_sync() { TD_CHECK_HIP(hipDeviceSynchronize()); }

static void stream_sync(uintptr_t stream) {
  TD_CHECK_HIP(hipStreamSynchronize(as_stream(stream)));
}

// ---------------------------------------------------------------------------
// Kernel wrappers
// ---------------------------------------------------------------------------
static void barrier_all(uintptr_t flags_ptr, uintptr_t epoch_cell,
                        uintptr_t stream) {
  check_active();
  launch_barrier_all(g_heap.pt, reinterpret_cast<int *>(flags_ptr),
                     reinterpret_cast<int *>(epoch_cell), as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void signal_set(uintptr_t flag, int val, uintptr_t stream) {
  launch_signal_set(reinterpret_cast<int *>(flag), val, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void wait_eq_host(uintptr_t flags, int n, int expect, uintptr_t stream) {
  launch_wait_eq(reinterpret_cast<const int *>(flags), n, expect,
                 as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void reset_flags(uintptr_t flags, int n, int val, uintptr_t stream) {
  launch_reset_flags(reinterpret_cast<int *>(flags), n, val, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void ag_pull(size_t ws_off, size_t flags_off, size_t seg_bytes,
                    int chunks, int chunk_stride, uintptr_t stream) {
  check_active();
  launch_ag_pull(g_heap.pt, ws_off, flags_off, seg_bytes, chunks,
                 chunk_stride, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void copy_kernel(uintptr_t dst, uintptr_t src, size_t nbytes,
                        uintptr_t stream) {
  launch_copy(reinterpret_cast<void *>(dst), reinterpret_cast<void *>(src),
              nbytes, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void put_signal(uintptr_t dst, uintptr_t src, size_t nbytes,
                       uintptr_t flag, int val, bool add, uintptr_t stream) {
  check_active();
  launch_put_signal(g_heap.pt, reinterpret_cast<void *>(dst),
                    reinterpret_cast<void *>(src), nbytes,
                    reinterpret_cast<int *>(flag), val, add ? 1 : 0,
                    as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void probe_mfma(uintptr_t a, uintptr_t b, uintptr_t c, int layout,
                       uintptr_t stream) {
  launch_probe_mfma(reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                    reinterpret_cast<void *>(c), layout, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm_bf16(uintptr_t a, uintptr_t b, uintptr_t c, uintptr_t bias,
                      int m, int n, int k, uintptr_t stream) {
  GemmArgs args{reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                reinterpret_cast<void *>(c), reinterpret_cast<void *>(bias),
                m,  n,  k, k, k, n};
  launch_gemm_bf16(args, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void ag_gemm_consumer_bf16(uintptr_t a, uintptr_t b, uintptr_t c,
                                  int m, int n, int k, uintptr_t flags,
                                  int chunks_per_rank, int m_per_rank,
                                  int ws_stride, int world, int rank,
                                  int expect,
                                  uintptr_t stream, uintptr_t prof_buf = 0,
                                  uintptr_t prof_cursor = 0,
                                  unsigned prof_cap = 0) {
  AgGemmArgs args;
  args.g = GemmArgs{reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                    reinterpret_cast<void *>(c), nullptr, m, n, k, k, k, n};
  args.flags = reinterpret_cast<const int *>(flags);
  args.chunks_per_rank = chunks_per_rank;
  args.m_per_rank = m_per_rank;
  args.ws_stride = ws_stride;
  args.world = world;
  args.rank = rank;
  args.expect = expect;
  args.prof = KProf{(unsigned long long *)prof_buf, (unsigned *)prof_cursor,
                    prof_cap};
  launch_ag_gemm_consumer_bf16(args, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void ag_gemm_fused_bf16(uintptr_t a, uintptr_t b, uintptr_t c,
                               int m_total, int n, int k, size_t ws_off,
                               size_t flags_off, int chunks,
                               int m_per_rank, int ws_stride, int world,
                               int rank, int expect, uintptr_t src,
                               uintptr_t arrive, int comm_wgs,
                               int subsplit, uintptr_t stream) {
  check_active();
  AgGemmArgs args;
  args.g = GemmArgs{reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                    reinterpret_cast<void *>(c), nullptr,
                    m_total, n, k, k, k, n};
  args.flags = reinterpret_cast<const int *>(
      (char *)g_heap.bases[g_heap.rank] + flags_off);
  args.chunks_per_rank = chunks;
  args.m_per_rank = m_per_rank;
  args.ws_stride = ws_stride;
  args.world = world;
  args.rank = rank;
  args.expect = expect;
  launch_ag_gemm256_fused(args, g_heap.pt, reinterpret_cast<void *>(src),
                          ws_off, flags_off,
                          reinterpret_cast<int *>(arrive), comm_wgs,
                          subsplit, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm256_colscatter(uintptr_t a, uintptr_t b, int m, int n,
                               int k, size_t recv_off, size_t flags_off,
                               int peer_cols, int slot_rows,
                               uintptr_t arrive, int tiles_per_peer,
                               int expect, uintptr_t stream) {
  check_active();
  UlyssesQkvArgs args;
  args.g = GemmArgs{reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                    nullptr, nullptr, m, n, k, k, k, n};
  args.pt = g_heap.pt;
  args.recv_off = recv_off;
  args.flags_off = flags_off;
  args.peer_cols = peer_cols;
  args.slot_rows = slot_rows;
  args.arrive = reinterpret_cast<int *>(arrive);
  args.tiles_per_peer = tiles_per_peer;
  args.expect = expect;
  launch_gemm256_colscatter(args, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm256_acc_bf16(uintptr_t a, uintptr_t b, uintptr_t ws, int m,
                             int n, int k, uintptr_t stream) {
  GemmArgs args{reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                nullptr, nullptr, m, n, k, k, k, n};
  launch_gemm256_acc_bf16(args, reinterpret_cast<float *>(ws),
                          as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void f32_to_bf16(uintptr_t ws, uintptr_t c, uintptr_t bias, int rows,
                        int n, uintptr_t stream) {
  launch_f32_to_bf16(reinterpret_cast<void *>(ws),
                     reinterpret_cast<void *>(c),
                     reinterpret_cast<void *>(bias), rows, n,
                     as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm_rs_producer_bf16(uintptr_t a, uintptr_t b, int m, int n,
                                  int k, size_t scatter_off, int m_per_rank,
                                  int ws_stride, int world, int rank,
                                  uintptr_t stream) {
  check_active();
  GemmRsArgs args;
  args.g = GemmArgs{reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                    nullptr, nullptr, m, n, k, k, k, n};
  args.pt = g_heap.pt;
  args.scatter_off = scatter_off;
  args.m_per_rank = m_per_rank;
  args.ws_stride = ws_stride;
  args.world = world;
  args.rank = rank;
  launch_gemm_rs_producer_bf16(args, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm_ar_producer_bf16(uintptr_t a, uintptr_t b, int m, int n,
                                  int k, size_t scatter_off,
                                  size_t arrive_off, size_t out_off,
                                  size_t oflags_off, int slots, int sk,
                                  uintptr_t ws, uintptr_t done,
                                  uintptr_t stream) {
  check_active();
  GemmArArgs args;
  args.g = GemmArgs{reinterpret_cast<void *>(a), reinterpret_cast<void *>(b),
                    nullptr, nullptr, m, n, k, k, k, n};
  args.pt = g_heap.pt;
  args.scatter_off = scatter_off;
  args.arrive_off = arrive_off;
  args.out_off = out_off;
  args.oflags_off = oflags_off;
  args.slots = slots;
  if (sk > 1) {
    launch_gemm256_sk_ar_producer(args, reinterpret_cast<float *>(ws),
                                  reinterpret_cast<int *>(done), sk,
                                  as_stream(stream));
  } else {
    launch_gemm256_ar_producer(args, as_stream(stream));
  }
  TD_CHECK_HIP(hipGetLastError());
}

static void ar_tile_consumer(int m, int n, size_t scatter_off,
                             size_t arrive_off, size_t out_off,
                             size_t oflags_off, int slots, int n_owned,
                             uintptr_t stream) {
  check_active();
  GemmArArgs args;
  args.g = GemmArgs{nullptr, nullptr, nullptr, nullptr, m, n, 0, 0, 0, n};
  args.pt = g_heap.pt;
  args.scatter_off = scatter_off;
  args.arrive_off = arrive_off;
  args.out_off = out_off;
  args.oflags_off = oflags_off;
  args.slots = slots;
  launch_ar_tile_consumer(args, n_owned, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void rs_reduce_bf16(uintptr_t segments, uintptr_t out, int world,
                           int rank, int m_per_rank, int ws_stride, int n,
                           uintptr_t stream) {
  launch_rs_reduce_bf16(reinterpret_cast<void *>(segments),
                        reinterpret_cast<void *>(out), world, rank, m_per_rank,
                        ws_stride, n, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm_skinny_bf16(uintptr_t a, uintptr_t b, uintptr_t c,
                             uintptr_t bias, int m, int n, int k,
                             int fuse_swiglu, uintptr_t stream) {
  GemmArgs args{(void *)a, (void *)b, (void *)c, (void *)bias,
                m, n, k, k, k, n};
  launch_gemm_skinny(args, fuse_swiglu, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm_splitk_bf16(uintptr_t a, uintptr_t b, uintptr_t c,
                             uintptr_t bias, uintptr_t ws, int m, int n,
                             int k, int splits, uintptr_t stream) {
  GemmArgs args{(void *)a, (void *)b, (void *)c, (void *)bias,
                m, n, k, k, k, n};
  launch_gemm_splitk_bf16(args, (float *)ws, splits, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void flash_decode_paged(uintptr_t q, uintptr_t kp, uintptr_t vp,
                               uintptr_t bt, int max_blocks, int block,
                               uintptr_t out, uintptr_t off, int batch,
                               int qh, int kvh, uintptr_t stream) {
  launch_flash_decode_paged((const void *)q, (const void *)kp,
                            (const void *)vp, (const void *)bt, max_blocks,
                            block, (void *)out, (const void *)off, batch,
                            qh, kvh, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gdn_decode(uintptr_t q, uintptr_t k, uintptr_t v, uintptr_t g,
                       uintptr_t beta, uintptr_t state, uintptr_t o, int B,
                       int H, int K, int V, float scale, uintptr_t stream) {
  launch_gdn_decode((const void *)q, (const void *)k, (const void *)v,
                    (const void *)g, (const void *)beta, (void *)state,
                    (void *)o, B, H, K, V, scale, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void reduce_scatter_op(uintptr_t x, size_t inbox_off,
                              size_t flags_off, uintptr_t local_inbox,
                              uintptr_t local_flags, uintptr_t out,
                              size_t seg_elems, int chunks,
                              uintptr_t tag_cell, uintptr_t stream) {
  launch_reduce_scatter(g_heap.pt, (const void *)x, inbox_off, flags_off,
                        (const void *)local_inbox, (const void *)local_flags,
                        (void *)out, seg_elems, chunks,
                        (const void *)tag_cell, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void all_to_all_op(uintptr_t x, size_t inbox_off, size_t flags_off,
                          uintptr_t local_inbox, uintptr_t local_flags,
                          uintptr_t out, size_t seg_elems, int chunks,
                          uintptr_t tag_cell, uintptr_t stream) {
  launch_all_to_all(g_heap.pt, (const void *)x, inbox_off, flags_off,
                    (const void *)local_inbox, (const void *)local_flags,
                    (void *)out, seg_elems, chunks, (const void *)tag_cell,
                    as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void ll_allgather_op(uintptr_t x, size_t inbox_off,
                            uintptr_t local_inbox, uintptr_t out, int words,
                            uintptr_t tag_cell, uintptr_t stream) {
  launch_ll_allgather(g_heap.pt, (const void *)x, inbox_off,
                      (const void *)local_inbox, (void *)out, words,
                      (const void *)tag_cell, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_router(uintptr_t logits, uintptr_t ids, uintptr_t tw,
                       int T, int E, int K, bool norm, uintptr_t stream) {
  launch_moe_router((const void *)logits, (void *)ids, (void *)tw, T, E, K,
                    norm, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static pybind11::dict p2p_attributes(int dev, int peer) {
  // Device-query parity: the reference checks P2P native-atomic support
  // (utils.py:539-567) before picking cross-GPU atomic protocols.
  pybind11::dict d;
  int v = 0;
  TD_CHECK_HIP(hipDeviceGetP2PAttribute(&v, hipDevP2PAttrAccessSupported,
                                        dev, peer));
  d["access"] = v;
  TD_CHECK_HIP(hipDeviceGetP2PAttribute(
      &v, hipDevP2PAttrNativeAtomicSupported, dev, peer));
  d["native_atomics"] = v;
  TD_CHECK_HIP(hipDeviceGetP2PAttribute(
      &v, hipDevP2PAttrPerformanceRank, dev, peer));
  d["performance_rank"] = v;
  return d;
}

static void gemm256_v2_bf16(uintptr_t a, uintptr_t b, uintptr_t c, int mm,
                            int n, int k, uintptr_t stream) {
  GemmArgs args{(void *)a, (void *)b, (void *)c, nullptr,
                mm, n, k, k, k, n};
  launch_gemm256_v2_bf16(args, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm256_v3_bf16(uintptr_t a, uintptr_t b, uintptr_t c, int mm,
                            int n, int k, uintptr_t stream) {
  GemmArgs args{(void *)a, (void *)b, (void *)c, nullptr,
                mm, n, k, k, k, n};
  launch_gemm256_v3_bf16(args, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm256_sk_bf16(uintptr_t a, uintptr_t b, uintptr_t c,
                            uintptr_t bias, uintptr_t ws, int m, int n,
                            int k, int sk, uintptr_t stream) {
  GemmArgs args{(void *)a, (void *)b, (void *)c, (void *)bias,
                m, n, k, k, k, n};
  launch_gemm256_sk_bf16(args, (float *)ws, sk, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm256_sk2_bf16(uintptr_t a, uintptr_t b, uintptr_t c,
                             uintptr_t bias, uintptr_t ws, int m, int n,
                             int k, int sk, uintptr_t stream) {
  GemmArgs args{(void *)a, (void *)b, (void *)c, (void *)bias,
                m, n, k, k, k, n};
  launch_gemm256_sk2_bf16(args, (float *)ws, sk, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm_stream_bf16(uintptr_t a, uintptr_t b, uintptr_t c,
                             uintptr_t bias, uintptr_t ws, int m, int n,
                             int k, int sk, uintptr_t stream) {
  GemmArgs args{(void *)a, (void *)b, (void *)c, (void *)bias,
                m, n, k, k, k, n};
  launch_gemm_stream_bf16(args, (float *)ws, sk, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void ag_gemm_consumer_splitk_bf16(uintptr_t a, uintptr_t b,
                                         uintptr_t c, uintptr_t ws, int m,
                                         int n, int k, uintptr_t flags,
                                         int chunks_per_rank, int m_per_rank,
                                         int ws_stride, int world, int rank,
                                         int expect,
                                         int splits, uintptr_t stream) {
  AgGemmArgs args;
  args.g = GemmArgs{(void *)a, (void *)b, (void *)c, nullptr,
                    m, n, k, k, k, n};
  args.flags = (const int *)flags;
  args.chunks_per_rank = chunks_per_rank;
  args.m_per_rank = m_per_rank;
  args.ws_stride = ws_stride;
  args.world = world;
  args.rank = rank;
  args.expect = expect;
  launch_ag_gemm_consumer_splitk_bf16(args, (float *)ws, splits,
                                      as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void gemm_rs_producer_splitk_bf16(uintptr_t a, uintptr_t b,
                                         uintptr_t ws, int m, int n, int k,
                                         size_t scatter_off, int m_per_rank,
                                         int ws_stride, int world, int rank,
                                         int splits,
                                         uintptr_t stream) {
  check_active();
  GemmRsArgs args;
  args.g = GemmArgs{(void *)a, (void *)b, nullptr, nullptr,
                    m, n, k, k, k, n};
  args.pt = g_heap.pt;
  args.scatter_off = scatter_off;
  args.m_per_rank = m_per_rank;
  args.ws_stride = ws_stride;
  args.world = world;
  args.rank = rank;
  launch_gemm_rs_producer_splitk_bf16(args, (float *)ws, splits,
                                      as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void allreduce_oneshot(uintptr_t x, uintptr_t out, size_t inbox_off,
                              size_t flags_off, size_t elems, int chunks,
                              int straggler_rank, unsigned straggler_cycles,
                              uintptr_t stream) {
  check_active();
  launch_allreduce_oneshot(g_heap.pt, (void *)x, (void *)out, inbox_off,
                           flags_off, elems, chunks, straggler_rank,
                           straggler_cycles, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void allreduce_twoshot(uintptr_t x, uintptr_t out, size_t inbox_off,
                              size_t outbox_off, size_t flags_in_off,
                              size_t flags_out_off, size_t elems, int chunks,
                              uintptr_t stream) {
  check_active();
  launch_allreduce_twoshot(g_heap.pt, (void *)x, (void *)out, inbox_off,
                           outbox_off, flags_in_off, flags_out_off, elems,
                           chunks, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_count(uintptr_t topk_ids, uintptr_t counts,
                      uintptr_t send_pos, uintptr_t send_to_dst, int total,
                      int e_num, int e_loc, int world, uintptr_t stream) {
  launch_moe_count((void *)topk_ids, (void *)counts, (void *)send_pos,
                   (void *)send_to_dst, total, e_num, e_loc, world,
                   as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_layout(uintptr_t all_splits, int rank, int world, int e_num,
                       int e_loc, uintptr_t send_base, uintptr_t expert_base,
                       uintptr_t expert_rows, uintptr_t recv_from_src,
                       uintptr_t recv_total, uintptr_t stream,
                       uintptr_t work_items = 0, uintptr_t work_count = 0,
                       int bm = 128) {
  launch_moe_layout((void *)all_splits, rank, world, e_num, e_loc,
                    (void *)send_base, (void *)expert_base,
                    (void *)expert_rows, (void *)recv_from_src,
                    (void *)recv_total, (void *)work_items,
                    (void *)work_count, bm, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_grouped_gemm_pq(uintptr_t xin, uintptr_t weights,
                                uintptr_t out, uintptr_t expert_base,
                                uintptr_t expert_rows, uintptr_t work_items,
                                uintptr_t work_count, int n, int k,
                                uintptr_t stream, uintptr_t eflags = 0,
                                uintptr_t val_cell = 0, int world = 0,
                                int e_loc = 0, int fuse_swiglu = 0) {
  launch_moe_grouped_gemm_pq((void *)xin, (void *)weights, (void *)out,
                             (void *)expert_base, (void *)expert_rows,
                             (void *)work_items, (void *)work_count, n, k,
                             as_stream(stream), (void *)eflags,
                             (void *)val_cell, world, e_loc, fuse_swiglu);
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_dispatch(uintptr_t x, uintptr_t topk_ids, uintptr_t send_pos,
                         uintptr_t send_base, uintptr_t counts,
                         size_t recv_x_off, size_t meta_off,
                         size_t eflags_off, uintptr_t arrive_e, int T,
                         int K, int H, int e_loc, int e_num,
                         uintptr_t stream, uintptr_t val_cell = 0) {
  check_active();
  launch_moe_dispatch(g_heap.pt, (void *)x, (void *)topk_ids,
                      (void *)send_pos, (void *)send_base,
                      (void *)counts, recv_x_off, meta_off, eflags_off,
                      (unsigned *)arrive_e, (void *)val_cell, T, K, H,
                      e_loc, e_num, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_fused_dispatch_gemm(
    uintptr_t x, uintptr_t topk_ids, uintptr_t send_pos,
    uintptr_t send_base, uintptr_t counts, size_t recv_x_off,
    size_t meta_off, size_t eflags_off, uintptr_t arrive_e, int T, int K,
    int H, int e_loc, int e_num, uintptr_t weights, uintptr_t out,
    uintptr_t expert_base, uintptr_t expert_rows, uintptr_t work_items,
    uintptr_t work_count, int n, int k, int fuse_swiglu, uintptr_t stream,
    uintptr_t val_cell = 0) {
  check_active();
  launch_moe_fused_dispatch_gemm(
      g_heap.pt, (void *)x, (void *)topk_ids, (void *)send_pos,
      (void *)send_base, (void *)counts, recv_x_off, meta_off, eflags_off,
      (unsigned *)arrive_e, (void *)val_cell, T, K, H, e_loc, e_num,
      (void *)weights, (void *)out, (void *)expert_base,
      (void *)expert_rows, (void *)work_items, (void *)work_count, n, k,
      fuse_swiglu, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_wait_flags(uintptr_t flags, int world, uintptr_t stream,
                           uintptr_t cell = 0) {
  launch_moe_wait_flags((void *)flags, world, (void *)cell,
                        as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_dispatch_fp8(uintptr_t x, uintptr_t topk_ids,
                             uintptr_t send_pos, uintptr_t send_base,
                             uintptr_t send_to_dst, size_t recv_q_off,
                             size_t recv_s_off, size_t meta_off,
                             size_t flags_off, uintptr_t arrive, int T,
                             int K, int H, int e_loc, uintptr_t stream,
                             uintptr_t val_cell = 0) {
  check_active();
  launch_moe_dispatch_fp8(g_heap.pt, (void *)x, (void *)topk_ids,
                          (void *)send_pos, (void *)send_base,
                          (void *)send_to_dst, recv_q_off, recv_s_off,
                          meta_off, flags_off, (unsigned *)arrive,
                          (void *)val_cell, T, K, H, e_loc,
                          as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_grouped_gemm_pq_fp8(uintptr_t rq, uintptr_t rs,
                                    uintptr_t weights, uintptr_t out,
                                    uintptr_t expert_base,
                                    uintptr_t expert_rows,
                                    uintptr_t work_items,
                                    uintptr_t work_count, int n, int k,
                                    int fuse_swiglu, uintptr_t stream) {
  launch_moe_grouped_gemm_pq_fp8(
      (void *)rq, (void *)rs, (void *)weights, (void *)out,
      (void *)expert_base, (void *)expert_rows, (void *)work_items,
      (void *)work_count, n, k, fuse_swiglu, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_dequant(uintptr_t rq, uintptr_t rs, uintptr_t out,
                        uintptr_t recv_total, int cap, int H,
                        uintptr_t stream) {
  launch_moe_dequant((void *)rq, (void *)rs, (void *)out, (void *)recv_total,
                     cap, H, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void bump_cell(uintptr_t cell, uintptr_t stream) {
  launch_bump_cell((void *)cell, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void wait_flags_ge_cell(uintptr_t flags, int n, uintptr_t cell,
                               int delta, uintptr_t stream) {
  launch_wait_flags_ge_cell((void *)flags, n, (void *)cell, delta,
                            as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void signal_credit(uintptr_t credit_off, uintptr_t cell,
                          uintptr_t stream) {
  check_active();
  launch_signal_credit(g_heap.pt, (size_t)credit_off, (void *)cell,
                       as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_grouped_gemm(uintptr_t xin, uintptr_t weights, uintptr_t out,
                             uintptr_t expert_base, uintptr_t expert_rows,
                             int e_loc, int cap_tiles_m, int n, int k,
                             int cap_rows, uintptr_t stream,
                             bool small_m = false, uintptr_t eflags = 0,
                             uintptr_t val_cell = 0, int world = 0,
                             int fuse_swiglu = 0) {
  launch_moe_grouped_gemm((void *)xin, (void *)weights, (void *)out,
                          (void *)expert_base, (void *)expert_rows, e_loc,
                          cap_tiles_m, n, k, cap_rows, as_stream(stream),
                          small_m, (void *)eflags, (void *)val_cell, world,
                          fuse_swiglu);
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_combine_send(uintptr_t expert_out, uintptr_t meta,
                             uintptr_t recv_total, uintptr_t recv_from_src,
                             size_t combine_off, size_t cflags_off,
                             uintptr_t arrive, int cap, int H,
                             uintptr_t stream, uintptr_t val_cell = 0) {
  check_active();
  launch_moe_combine_send(g_heap.pt, (void *)expert_out, (void *)meta,
                          (void *)recv_total, (void *)recv_from_src,
                          combine_off, cflags_off, (unsigned *)arrive,
                          (void *)val_cell, cap, H, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void moe_combine_reduce(uintptr_t combine_buf, uintptr_t topk_w,
                               uintptr_t topk_ids, uintptr_t out,
                               uintptr_t cflags, int world, int T, int K,
                               int H, int e_num, uintptr_t stream,
                               uintptr_t val_cell = 0) {
  launch_moe_combine_reduce((void *)combine_buf, (void *)topk_w,
                            (void *)topk_ids, (void *)out, (void *)cflags,
                            (void *)val_cell, world, T, K, H, e_num,
                            as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void megakernel(uintptr_t tasks, uintptr_t queue, uintptr_t queue_off,
                       uintptr_t scoreboard, int n_wg, uintptr_t stream,
                       int fence_mode = 0, uintptr_t prof = 0) {
  launch_megakernel((const void *)tasks, (const void *)queue,
                    (const void *)queue_off, (void *)scoreboard, n_wg,
                    as_stream(stream), fence_mode, (void *)prof);
  TD_CHECK_HIP(hipGetLastError());
}

static void rmsnorm(uintptr_t x, uintptr_t w, uintptr_t out, int rows,
                    int cols, float eps, uintptr_t stream) {
  launch_rmsnorm((void *)x, (void *)w, (void *)out, rows, cols, eps,
                 as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void add_rmsnorm(uintptr_t x, uintptr_t resid_in, uintptr_t resid_out,
                        uintptr_t w, uintptr_t out, int rows, int cols,
                        float eps, uintptr_t stream) {
  launch_add_rmsnorm((void *)x, (void *)resid_in, (void *)resid_out,
                     (void *)w, (void *)out, rows, cols, eps,
                     as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void swiglu(uintptr_t h, uintptr_t out, int rows, int inter,
                   uintptr_t stream) {
  launch_swiglu((void *)h, (void *)out, rows, inter, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void qkv_prologue_decode(uintptr_t qkv, uintptr_t q_out,
                                uintptr_t kcache, uintptr_t vcache,
                                uintptr_t cos_t, uintptr_t sin_t,
                                uintptr_t qnw, uintptr_t knw,
                                uintptr_t offset, int batch, int qh, int kvh,
                                int max_len, float eps, bool use_qk_norm,
                                uintptr_t stream) {
  launch_qkv_prologue_decode((void *)qkv, (void *)q_out, (void *)kcache,
                             (void *)vcache, (void *)cos_t, (void *)sin_t,
                             (void *)qnw, (void *)knw, (void *)offset, batch,
                             qh, kvh, max_len, eps, use_qk_norm,
                             as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void flash_prefill(uintptr_t q, uintptr_t k, uintptr_t v,
                          uintptr_t out, int b, int s, int qh, int kvh,
                          float scale, bool causal, uintptr_t stream,
                          uintptr_t lse = 0, long kb_stride = 0) {
  launch_flash_prefill((const void *)q, (const void *)k, (const void *)v,
                       (void *)out, (void *)lse, b, s, qh, kvh, scale,
                       causal, as_stream(stream), kb_stride);
  TD_CHECK_HIP(hipGetLastError());
}

static void qkv_prologue_prefill(uintptr_t qkv, uintptr_t q_out,
                                 uintptr_t kcache, uintptr_t vcache,
                                 uintptr_t cos_t, uintptr_t sin_t,
                                 uintptr_t qnw, uintptr_t knw, int batch,
                                 int s, int qh, int kvh, int max_len,
                                 float eps, bool use_qk_norm,
                                 uintptr_t stream) {
  launch_qkv_prologue_prefill(
      (const void *)qkv, (void *)q_out, (void *)kcache, (void *)vcache,
      (const void *)cos_t, (const void *)sin_t, (const void *)qnw,
      (const void *)knw, batch, s, qh, kvh, max_len, eps, use_qk_norm,
      as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void flash_decode(uintptr_t q, uintptr_t kcache, uintptr_t vcache,
                         uintptr_t out, uintptr_t offset, int batch, int qh,
                         int kvh, int max_len, uintptr_t stream) {
  launch_flash_decode((void *)q, (void *)kcache, (void *)vcache, (void *)out,
                      (void *)offset, batch, qh, kvh, max_len,
                      as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void flash_decode_partial(uintptr_t q, uintptr_t kcache,
                                 uintptr_t vcache, uintptr_t out_part,
                                 uintptr_t lse, uintptr_t chunk_len,
                                 int batch, int qh, int kvh, int max_len,
                                 uintptr_t stream) {
  launch_flash_decode_partial((void *)q, (void *)kcache, (void *)vcache,
                              (void *)out_part, (void *)lse,
                              (void *)chunk_len, batch, qh, kvh, max_len,
                              as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

static void lse_combine(uintptr_t parts, uintptr_t lses, uintptr_t out,
                        uintptr_t flags, int world, int batch, int qh,
                        int slot_batch, uintptr_t stream) {
  launch_lse_combine((void *)parts, (void *)lses, (void *)out, (void *)flags,
                     world, batch, qh, slot_batch, as_stream(stream));
  TD_CHECK_HIP(hipGetLastError());
}

PYBIND11_MODULE(_C, m) {
  m.doc() = "triton_dist_amd native core: hipIpc symmetric heap + gfx950 kernels";
  m.def("heap_init", &heap_init, py::arg("rank"), py::arg("world"),
        py::arg("device"), py::arg("size"), py::arg("fine_grained") = true);
  m.def("heap_open", &heap_open);
  m.def("heap_close", &heap_close);
  m.def("heap_base", &heap_base);
  m.def("heap_scratch_bytes", &heap_scratch_bytes);
  m.def("heap_is_fine_grained", [] { return g_heap.fine_grained; });
  m.def("dlpack_from_ptr", &dlpack_from_ptr, py::arg("ptr"), py::arg("shape"),
        py::arg("code"), py::arg("bits"), py::arg("device_id"));
  m.def("memcpy_async", &memcpy_async);
  m.def("memset32_async", &memset32_async);
  m.def("device_sync", &device_sync);
  m.def("stream_sync", &stream_sync);
  m.def("barrier_all", &barrier_all);
  m.def("signal_set", &signal_set);
  m.def("wait_eq", &wait_eq_host);
  m.def("reset_flags", &reset_flags);
  m.def("ag_pull", &ag_pull);
  m.def("copy_kernel", &copy_kernel);
  m.def("put_signal", &put_signal);
  m.def("probe_mfma", &probe_mfma);
  m.def("gemm_bf16", &gemm_bf16);
  m.def("ag_gemm_consumer_bf16", &ag_gemm_consumer_bf16, py::arg("a"),
        py::arg("b"), py::arg("c"), py::arg("m"), py::arg("n"), py::arg("k"),
        py::arg("flags"), py::arg("chunks_per_rank"), py::arg("m_per_rank"),
        py::arg("ws_stride"), py::arg("world"), py::arg("rank"),
        py::arg("expect"),
        py::arg("stream"), py::arg("prof_buf") = 0,
        py::arg("prof_cursor") = 0, py::arg("prof_cap") = 0);
  m.def("gemm_rs_producer_bf16", &gemm_rs_producer_bf16);
  m.def("ag_gemm_fused_bf16", &ag_gemm_fused_bf16);
  m.def("gemm256_colscatter", &gemm256_colscatter);
  m.def("gemm256_acc_bf16", &gemm256_acc_bf16);
  m.def("f32_to_bf16", &f32_to_bf16);
  m.def("gemm_ar_producer_bf16", &gemm_ar_producer_bf16);
  m.def("ar_tile_consumer", &ar_tile_consumer);
  m.def("rs_reduce_bf16", &rs_reduce_bf16);
  m.def("gemm_splitk_bf16", &gemm_splitk_bf16);
  m.def("gemm_skinny_bf16", &gemm_skinny_bf16);
  m.def("gemm256_sk_bf16", &gemm256_sk_bf16);
  m.def("gemm256_sk2_bf16", &gemm256_sk2_bf16);
  m.def("gemm_stream_bf16", &gemm_stream_bf16);
  m.def("gemm256_v2_bf16", &gemm256_v2_bf16);
  m.def("gemm256_v3_bf16", &gemm256_v3_bf16);
  m.def("p2p_attributes", &p2p_attributes);
  m.def("moe_router", &moe_router);
  m.def("reduce_scatter", &reduce_scatter_op);
  m.def("gdn_decode", &gdn_decode);
  m.def("flash_decode_paged", &flash_decode_paged);
  m.def("flash_prefill", &flash_prefill, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("out"), py::arg("b"), py::arg("s"),
        py::arg("qh"), py::arg("kvh"), py::arg("scale"), py::arg("causal"),
        py::arg("stream"), py::arg("lse") = 0, py::arg("kb_stride") = 0);
  m.def("qkv_prologue_prefill", &qkv_prologue_prefill);
  m.def("ll_allgather", &ll_allgather_op);
  m.def("all_to_all", &all_to_all_op);
  m.def("ag_gemm_consumer_splitk_bf16", &ag_gemm_consumer_splitk_bf16);
  m.def("gemm_rs_producer_splitk_bf16", &gemm_rs_producer_splitk_bf16);
  m.def("allreduce_oneshot", &allreduce_oneshot);
  m.def("allreduce_twoshot", &allreduce_twoshot);
  m.def("moe_count", &moe_count);
  m.def("moe_layout", &moe_layout, py::arg("all_splits"), py::arg("rank"),
        py::arg("world"), py::arg("e_num"), py::arg("e_loc"),
        py::arg("send_base"), py::arg("expert_base"), py::arg("expert_rows"),
        py::arg("recv_from_src"), py::arg("recv_total"), py::arg("stream"),
        py::arg("work_items") = 0, py::arg("work_count") = 0,
        py::arg("bm") = 128);
  m.def("moe_fused_dispatch_gemm", &moe_fused_dispatch_gemm,
        py::arg("x"), py::arg("topk_ids"), py::arg("send_pos"),
        py::arg("send_base"), py::arg("counts"), py::arg("recv_x_off"),
        py::arg("meta_off"), py::arg("eflags_off"), py::arg("arrive_e"),
        py::arg("T"), py::arg("K"), py::arg("H"), py::arg("e_loc"),
        py::arg("e_num"), py::arg("weights"), py::arg("out"),
        py::arg("expert_base"), py::arg("expert_rows"),
        py::arg("work_items"), py::arg("work_count"), py::arg("n"),
        py::arg("k"), py::arg("fuse_swiglu"), py::arg("stream"),
        py::arg("val_cell") = 0);
  m.def("moe_grouped_gemm_pq", &moe_grouped_gemm_pq, py::arg("xin"),
        py::arg("weights"), py::arg("out"), py::arg("expert_base"),
        py::arg("expert_rows"), py::arg("work_items"),
        py::arg("work_count"), py::arg("n"), py::arg("k"),
        py::arg("stream"), py::arg("eflags") = 0, py::arg("val_cell") = 0,
        py::arg("world") = 0, py::arg("e_loc") = 0,
        py::arg("fuse_swiglu") = 0);
  m.def("moe_dispatch", &moe_dispatch, py::arg("x"), py::arg("topk_ids"),
        py::arg("send_pos"), py::arg("send_base"), py::arg("counts"),
        py::arg("recv_x_off"), py::arg("meta_off"), py::arg("eflags_off"),
        py::arg("arrive_e"), py::arg("T"), py::arg("K"), py::arg("H"),
        py::arg("e_loc"), py::arg("e_num"), py::arg("stream"),
        py::arg("val_cell") = 0);
  m.def("moe_dispatch_fp8", &moe_dispatch_fp8, py::arg("x"),
        py::arg("topk_ids"), py::arg("send_pos"), py::arg("send_base"),
        py::arg("send_to_dst"), py::arg("recv_q_off"), py::arg("recv_s_off"),
        py::arg("meta_off"), py::arg("flags_off"), py::arg("arrive"),
        py::arg("T"), py::arg("K"), py::arg("H"), py::arg("e_loc"),
        py::arg("stream"), py::arg("val_cell") = 0);
  m.def("moe_dequant", &moe_dequant);
  m.def("moe_grouped_gemm_pq_fp8", &moe_grouped_gemm_pq_fp8);
  m.def("bump_cell", &bump_cell);
  m.def("wait_flags_ge_cell", &wait_flags_ge_cell);
  m.def("signal_credit", &signal_credit);
  m.def("moe_wait_flags", &moe_wait_flags, py::arg("flags"),
        py::arg("world"), py::arg("stream"), py::arg("cell") = 0);
  m.def("moe_grouped_gemm", &moe_grouped_gemm, py::arg("xin"),
        py::arg("weights"), py::arg("out"), py::arg("expert_base"),
        py::arg("expert_rows"), py::arg("e_loc"), py::arg("cap_tiles_m"),
        py::arg("n"), py::arg("k"), py::arg("cap_rows"), py::arg("stream"),
        py::arg("small_m") = false, py::arg("eflags") = 0,
        py::arg("val_cell") = 0, py::arg("world") = 0,
        py::arg("fuse_swiglu") = 0);
  m.def("moe_combine_send", &moe_combine_send, py::arg("expert_out"),
        py::arg("meta"), py::arg("recv_total"), py::arg("recv_from_src"),
        py::arg("combine_off"), py::arg("cflags_off"), py::arg("arrive"),
        py::arg("cap"), py::arg("H"), py::arg("stream"),
        py::arg("val_cell") = 0);
  m.def("moe_combine_reduce", &moe_combine_reduce, py::arg("combine_buf"),
        py::arg("topk_w"), py::arg("topk_ids"), py::arg("out"),
        py::arg("cflags"), py::arg("world"), py::arg("T"), py::arg("K"),
        py::arg("H"), py::arg("e_num"), py::arg("stream"),
        py::arg("val_cell") = 0);
  m.def("megakernel", &megakernel, py::arg("tasks"), py::arg("queue"),
        py::arg("queue_off"), py::arg("scoreboard"), py::arg("n_wg"),
        py::arg("stream"), py::arg("fence_mode") = 0,
        py::arg("prof") = 0);
  m.def("rmsnorm", &rmsnorm);
  m.def("add_rmsnorm", &add_rmsnorm);
  m.def("swiglu", &swiglu);
  m.def("qkv_prologue_decode", &qkv_prologue_decode);
  m.def("flash_decode", &flash_decode);
  m.def("flash_decode_partial", &flash_decode_partial);
  m.def("lse_combine", &lse_combine);
}
