// Host-side API shared between the pybind module and kernel TUs.
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>

#include "td/device.hpp"
#include "td/profiler.hpp"

namespace td {

#define TD_CHECK_HIP(expr)                                                   \
  do {                                                                       \
    hipError_t _e = (expr);                                                  \
    if (_e != hipSuccess) {                                                  \
      throw std::runtime_error(std::string("HIP error at " __FILE__ ":") +   \
                               std::to_string(__LINE__) + ": " +             \
                               hipGetErrorString(_e));                       \
    }                                                                        \
  } while (0)

// kernels/common.hip ---------------------------------------------------------
void launch_barrier_all(const PeerTable &pt, int *local_flags, int *epoch_cell,
                        hipStream_t stream);
void launch_signal_set(int *flag, int val, hipStream_t stream);
void launch_wait_eq(const int *flags, int n, int expect, hipStream_t stream);
void launch_copy(void *dst, const void *src, size_t nbytes, hipStream_t stream);
void launch_put_signal(const PeerTable &pt, void *dst, const void *src,
                       size_t nbytes, int *flag, int val, int add,
                       hipStream_t stream);
void launch_reset_flags(int *flags, int n, int val, hipStream_t stream);
void launch_ag_pull(const PeerTable &pt, size_t ws_off, size_t flags_off,
                    size_t seg_bytes, int chunks, int chunk_stride,
                    hipStream_t stream);

// kernels/gemm.hip -----------------------------------------------------------
struct GemmArgs {
  const void *a;   // [M,K] bf16 row-major
  const void *b;   // [N,K] bf16 row-major (B^T layout: GEMM computes A @ B^T)
  void *c;         // [M,N] bf16 row-major
  const void *bias;  // [N] bf16 or nullptr
  int m, n, k;
  int lda, ldb, ldc;  // in elements
};

void launch_gemm_bf16(const GemmArgs &args, hipStream_t stream);

// MFMA operand-layout probe (see gemm.hip): C[16x16] f32 = A[16x32] @ B[32x16].
void launch_probe_mfma(const void *a, const void *b, void *c, int layout,
                       hipStream_t stream);

// AG-GEMM consumer: persistent GEMM over the gathered A [world*m_per_rank, K]
// stored in the local symmetric workspace; each tile spin-waits on the
// per-chunk flags before consuming rows of A.
struct AgGemmArgs {
  GemmArgs g;           // a = symm workspace [M_total, K]
  const int *flags;     // per-chunk ready flags (local heap)
  int chunks_per_rank;  // flag granularity along M
  int m_per_rank;       // ACTUAL rows per rank this call (m % BM == 0)
  int ws_stride;        // allocated rows per segment in the workspace
  int world;
  int rank;
  int expect;           // flag value that means "ready"
  KProf prof{};         // optional intra-kernel profiler (buf == nullptr: off)
};
void launch_ag_gemm_consumer_bf16(const AgGemmArgs &args, hipStream_t stream);

// Single-fused-kernel AG-GEMM (gemm256.hip): first comm_wgs workgroups
// push my shard (incl. the self-copy) chunk-by-chunk into every peer's
// workspace segment and signal per-chunk flags; the rest run the
// flag-waiting 256^2 consumer. `arrive` = local [world*chunks] int32
// sub-chunk counters (reset per call).
void launch_ag_gemm256_fused(const AgGemmArgs &args, const PeerTable &pt,
                             const void *src, size_t ws_off,
                             size_t flags_off, int *arrive, int comm_wgs,
                             int subsplit, hipStream_t stream);

// GEMM-RS producer: GEMM whose epilogue scatters each output tile directly
// into the owner rank's symmetric scatter buffer (remote store over xGMI).
struct GemmRsArgs {
  GemmArgs g;       // c unused; m = world * m_per_rank
  PeerTable pt;
  size_t scatter_off;  // offset of scatter buffer [world segments, ws_stride, N]
  int m_per_rank;      // actual rows per rank this call
  int ws_stride;       // allocated rows per segment
  int world;
  int rank;
};
void launch_gemm_rs_producer_bf16(const GemmRsArgs &args, hipStream_t stream);

// Reduce over world segments [world, m_per_rank, N] -> [m_per_rank, N].
void launch_rs_reduce_bf16(const void *segments, void *out, int world,
                           int rank, int m_per_rank, int ws_stride, int n,
                           hipStream_t stream);

// 256^2-tile split-K decode tier (gemm256.hip): fp32 atomic accumulation
// into ws[M,N] + convert. Requires m%256==0, n%256==0, k%(128*sk)==0,
// ldc==n.
void launch_gemm256_sk_bf16(const GemmArgs &g, float *ws, int sk,
                            hipStream_t stream);
// two-stage variant: private fp32 slices ws[sk, M, N] + fused reduce
// (no atomics; ws must hold sk*M*N floats)
void launch_gemm256_sk2_bf16(const GemmArgs &g, float *ws, int sk,
                             hipStream_t stream);

// weight-streaming decode GEMM (gemm_stream.hip): BM=512, B read once
void launch_gemm_stream_bf16(const GemmArgs &g, float *ws, int sk,
                             hipStream_t stream);

// EXPERIMENTAL BK=64 quadrant-phase template (gemm256_v2.hip) — not in
// any dispatch path; see the file header.
void launch_gemm256_v2_bf16(const GemmArgs &args, hipStream_t stream);

// gemm256_v3.hip — faithful 8-phase template rebuild (register-reuse gray
// quadrant walk + counted vmcnt drains); candidate production tier.
void launch_gemm256_v3_bf16(const GemmArgs &args, hipStream_t stream);

// kernels/gemm_skinny.hip — BM=32 x BN=128 direct tier for decode shapes
// (no split-K ws round trip); optional fused-SwiGLU epilogue.
void launch_gemm_skinny(const GemmArgs &g, int fuse_swiglu,
                        hipStream_t stream);

// kernels/gemm_splitk.hip ---------------------------------------------------
void launch_gemm_splitk_bf16(const GemmArgs &g, float *ws, int splits,
                             hipStream_t stream);
void launch_ag_gemm_consumer_splitk_bf16(const AgGemmArgs &a, float *ws,
                                         int splits, hipStream_t stream);
void launch_gemm_rs_producer_splitk_bf16(const GemmRsArgs &a, float *ws,
                                         int splits, hipStream_t stream);

void launch_moe_router(const void *logits, void *topk_ids, void *topk_w,
                       int T, int E, int K, bool norm, hipStream_t stream);

// Ulysses fused qkv-GEMM + head a2a (gemm256.hip): column-scatter
// epilogue into the owning rank's recv buffer + per-src completion flags.
struct UlyssesQkvArgs {
  GemmArgs g;        // a [T_loc, H]; b = w_qkv [qkv_dim, H]; c unused
  PeerTable pt;
  size_t recv_off;   // [world_src][slot_rows][peer_cols] bf16 on each rank
  size_t flags_off;  // [world] int32
  int peer_cols;     // qkv_dim / world (multiple of 256)
  int slot_rows;     // allocated rows per src slot (max_T)
  int *arrive;       // local [world] int32 (reset per call)
  int tiles_per_peer;
  int expect;
};
void launch_gemm256_colscatter(const UlyssesQkvArgs &args,
                               hipStream_t stream);
// Accumulating GEMM (fp32 atomic ws, no memset/convert) + the converter.
void launch_gemm256_acc_bf16(const GemmArgs &g, float *ws,
                             hipStream_t stream);
void launch_f32_to_bf16(const void *ws, void *c, const void *bias, int rows,
                        int n, hipStream_t stream);

// kernels/gemm_ar.hip — tile-granular fused GEMM + AllReduce -----------------
// Round-robin tile ownership (owner = linear_tile % world). Producer GEMM
// pushes each C tile to the owner's scatter slot + bumps the owner's
// arrive counter; the consumer (comm stream) reduces owned tiles as their
// world arrivals land and broadcasts into every rank's symmetric out
// buffer with per-tile flags. See the file header for the protocol.
struct GemmArArgs {
  GemmArgs g;          // a/b as usual; c unused (out lives in the heap)
  PeerTable pt;
  size_t scatter_off;  // [world_src][slots][256*256] bf16
  size_t arrive_off;   // [slots] int32 per rank
  size_t out_off;      // [M, N] bf16 symmetric out
  size_t oflags_off;   // [tiles_m * tiles_n] int32 per rank
  int slots;           // allocated slots per src segment
};
void launch_gemm256_ar_producer(const GemmArArgs &args, hipStream_t stream);
void launch_gemm256_sk_ar_producer(const GemmArArgs &args, float *ws,
                                   int *done, int sk, hipStream_t stream);
void launch_ar_tile_consumer(const GemmArArgs &args, int n_owned,
                             hipStream_t stream);

// kernels/allreduce.hip -----------------------------------------------------
void launch_allreduce_oneshot(const PeerTable &pt, const void *x, void *out,
                              size_t inbox_off, size_t flags_off,
                              size_t elems, int chunks, int straggler_rank,
                              unsigned straggler_cycles, hipStream_t stream);
void launch_allreduce_twoshot(const PeerTable &pt, const void *x, void *out,
                              size_t inbox_off, size_t outbox_off,
                              size_t flags_in_off, size_t flags_out_off,
                              size_t elems, int chunks, hipStream_t stream);

// kernels/gdn.hip ------------------------------------------------------------
void launch_gdn_decode(const void *q, const void *k, const void *v,
                       const void *g, const void *beta, void *state,
                       void *o, int B, int H, int K, int V, float scale,
                       hipStream_t stream);

// kernels/collectives.hip ---------------------------------------------------
void launch_reduce_scatter(const PeerTable &pt, const void *x,
                           size_t inbox_off, size_t flags_off,
                           const void *local_inbox, const void *local_flags,
                           void *out, size_t seg_elems, int chunks,
                           const void *tag_cell, hipStream_t stream);
void launch_all_to_all(const PeerTable &pt, const void *x, size_t inbox_off,
                       size_t flags_off, const void *local_inbox,
                       const void *local_flags, void *out, size_t seg_elems,
                       int chunks, const void *tag_cell, hipStream_t stream);
void launch_ll_allgather(const PeerTable &pt, const void *x,
                         size_t inbox_off, const void *local_inbox,
                         void *out, int words, const void *tag_cell,
                         hipStream_t stream);

// kernels/moe.hip ------------------------------------------------------------
void launch_moe_count(const void *topk_ids, void *counts, void *send_pos,
                      void *send_to_dst, int total, int e_num, int e_loc,
                      int world, hipStream_t stream);
void launch_moe_layout(const void *all_splits, int rank, int world,
                       int e_num, int e_loc, void *send_base,
                       void *expert_base, void *expert_rows,
                       void *recv_from_src, void *recv_total,
                       void *work_items, void *work_count, int bm,
                       hipStream_t stream);
void launch_moe_grouped_gemm_pq(const void *xin, const void *weights,
                                void *out, const void *expert_base,
                                const void *expert_rows,
                                const void *work_items,
                                const void *work_count, int n, int k,
                                hipStream_t stream,
                                const void *eflags = nullptr,
                                const void *val_cell = nullptr,
                                int world = 0, int e_loc = 0,
                                int fuse_swiglu = 0);
void launch_moe_dispatch(const PeerTable &pt, const void *x,
                         const void *topk_ids, const void *send_pos,
                         const void *send_base, const void *counts,
                         size_t recv_x_off, size_t meta_off,
                         size_t eflags_off, unsigned *arrive_e,
                         const void *val_cell, int T, int K, int H,
                         int e_loc, int e_num, hipStream_t stream);
// Fused single-launch EP dispatch + per-expert-gated grouped GEMM
// (reference ep_all2all_fused.py:316 mega-kernel — behavior only).
void launch_moe_fused_dispatch_gemm(
    const PeerTable &pt, const void *x, const void *topk_ids,
    const void *send_pos, const void *send_base, const void *counts,
    size_t recv_x_off, size_t meta_off, size_t eflags_off,
    unsigned *arrive_e, const void *val_cell, int T, int K, int H,
    int e_loc, int e_num, const void *weights, void *out,
    const void *expert_base, const void *expert_rows,
    const void *work_items, const void *work_count, int n, int k,
    int fuse_swiglu, hipStream_t stream);
void launch_moe_wait_flags(const void *flags, int world, const void *cell,
                           hipStream_t stream);
void launch_bump_cell(void *cell, hipStream_t stream);
void launch_moe_dispatch_fp8(const PeerTable &pt, const void *x,
                             const void *topk_ids, const void *send_pos,
                             const void *send_base, const void *send_to_dst,
                             size_t recv_q_off, size_t recv_s_off,
                             size_t meta_off, size_t flags_off,
                             unsigned *arrive, const void *val_cell, int T,
                             int K, int H, int e_loc, hipStream_t stream);
void launch_moe_grouped_gemm_pq_fp8(
    const void *rq, const void *rs, const void *weights, void *out,
    const void *expert_base, const void *expert_rows,
    const void *work_items, const void *work_count, int n, int k,
    int fuse_swiglu, hipStream_t stream);
void launch_moe_dequant(const void *rq, const void *rs, void *out,
                        const void *recv_total, int cap, int H,
                        hipStream_t stream);
void launch_wait_flags_ge_cell(const void *flags, int n, const void *cell,
                               int delta, hipStream_t stream);
void launch_signal_credit(const PeerTable &pt, size_t credit_off,
                          const void *cell, hipStream_t stream);
void launch_moe_grouped_gemm(const void *xin, const void *weights, void *out,
                             const void *expert_base, const void *expert_rows,
                             int e_loc, int cap_tiles_m, int n, int k,
                             int cap_rows, hipStream_t stream,
                             bool small_m = false,
                             const void *eflags = nullptr,
                             const void *val_cell = nullptr, int world = 0,
                             int fuse_swiglu = 0);
void launch_moe_combine_send(const PeerTable &pt, const void *expert_out,
                             const void *meta, const void *recv_total,
                             const void *recv_from_src, size_t combine_off,
                             size_t cflags_off, unsigned *arrive,
                             const void *val_cell, int cap, int H,
                             hipStream_t stream);
void launch_moe_combine_reduce(const void *combine_buf, const void *topk_w,
                               const void *topk_ids, void *out,
                               const void *cflags, const void *val_cell,
                               int world, int T, int K,
                               int H, int e_num, hipStream_t stream);

// kernels/megakernel.hip -----------------------------------------------------
void launch_megakernel(const void *tasks, const void *queue,
                       const void *queue_off, void *scoreboard, int n_wg,
                       hipStream_t stream, int fence_mode = 0,
                       void *prof = nullptr);

// kernels/elementwise.hip ----------------------------------------------------
void launch_rmsnorm(const void *x, const void *w, void *out, int rows,
                    int cols, float eps, hipStream_t stream);
void launch_add_rmsnorm(const void *x, const void *resid_in, void *resid_out,
                        const void *w, void *out, int rows, int cols,
                        float eps, hipStream_t stream);
void launch_swiglu(const void *h, void *out, int rows, int inter,
                   hipStream_t stream);
void launch_qkv_prologue_decode(const void *qkv, void *q_out, void *kcache,
                                void *vcache, const void *cos_t,
                                const void *sin_t, const void *qnw,
                                const void *knw, const void *offset,
                                int batch, int qh, int kvh, int max_len,
                                float eps, bool use_qk_norm,
                                hipStream_t stream);

// kernels/attention.hip ------------------------------------------------------
void launch_flash_prefill(const void *q, const void *k, const void *v,
                          void *out, void *lse, int b, int s, int qh,
                          int kvh, float scale, bool causal,
                          hipStream_t stream, long kb_stride = 0);
void launch_qkv_prologue_prefill(const void *qkv, void *q_out, void *kcache,
                                 void *vcache, const void *cos_t,
                                 const void *sin_t, const void *qnw,
                                 const void *knw, int batch, int s, int qh,
                                 int kvh, int max_len, float eps,
                                 bool use_qk_norm, hipStream_t stream);
void launch_flash_decode(const void *q, const void *kcache,
                         const void *vcache, void *out, const void *offset,
                         int batch, int qh, int kvh, int max_len,
                         hipStream_t stream);
void launch_flash_decode_paged(const void *q, const void *k_pool,
                               const void *v_pool, const void *block_table,
                               int max_blocks, int block, void *out,
                               const void *offset, int batch, int qh,
                               int kvh, hipStream_t stream);
void launch_flash_decode_partial(const void *q, const void *kcache,
                                 const void *vcache, void *out_part,
                                 void *lse, const void *chunk_len, int batch,
                                 int qh, int kvh, int max_len,
                                 hipStream_t stream);
void launch_lse_combine(const void *parts, const void *lses, void *out,
                        const void *flags, int world, int batch, int qh,
                        int slot_batch, hipStream_t stream);

}  // namespace td
