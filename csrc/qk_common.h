// Common declarations shared by the HIP kernel TUs and the torch-binding TU.
//
// Design: kernel TUs (sample_kernels.hip, reindex_kernels.hip,
// gather_kernels.hip) include only <hip/hip_runtime.h> and expose C-ABI
// launchers taking raw pointers + a hipStream_t.  module.cpp owns all
// torch::Tensor plumbing and memory allocation (torch caching allocator).
//
// Capability parity map (reference: quiver-team/torch-quiver):
//   sample_kernels.hip  ~ srcs/cpp/include/quiver/cuda_random.cu.hpp +
//                         quiver.cu.hpp new_sample path (re-designed wave64)
//   reindex_kernels.hip ~ srcs/cpp/include/quiver/reindex.cu.hpp
//   gather_kernels.hip  ~ srcs/cpp/include/quiver/shard_tensor.cu.hpp
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>
#include <stdexcept>
#include <string>

#define QK_CHECK_HIP(expr)                                                     \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        if (_e != hipSuccess) {                                                \
            throw std::runtime_error(std::string("HIP error at " __FILE__ ":") \
                                     + std::to_string(__LINE__) + ": "         \
                                     + hipGetErrorString(_e));                 \
        }                                                                      \
    } while (0)

namespace qk {

constexpr int kWaveSize = 64;  // CDNA4 wavefront

// ---------- sampling (sample_kernels.hip) ----------

// counts[i] = degree(seeds[i]); capped[i] = min(counts[i], k). k<0 => no cap.
//
// Dynamic-count convention (here and below): `n`/`n_nbrs` are HOST upper
// bounds used for grid sizing and slack zeroing; when `n_dev`/`m_dev` is
// non-null the exact count lives on the DEVICE (written by a prior kernel
// in the same stream) and the kernel reads it — this is what lets the
// fused multi-hop sampling loop run an entire batch without a single
// host-device synchronization (slack elements of count/flag buffers are
// zeroed so downstream scans stay exact).
void launch_capped_degree(hipStream_t s, const int64_t* indptr,
                          const int64_t* seeds, int64_t n, int k,
                          int64_t* capped, const int64_t* n_dev = nullptr);

// Exclusive scan over n int64 values; also writes total = sum to d_total.
// temp_bytes(n) tells the caller how much scratch to allocate.
size_t scan_temp_bytes(int64_t n);
void launch_exclusive_scan(hipStream_t s, void* temp, size_t temp_bytes,
                           const int64_t* in, int64_t* out, int64_t n,
                           int64_t* d_total);

// Wavefront-subgroup neighbor sampling (reservoir, without replacement).
// out_nbrs[prefix[i] .. prefix[i]+capped[i]) = sampled neighbor ids of seeds[i].
// If out_eids != nullptr also emits the matching edge ids (eid_base==nullptr
// means eid == CSR position).
void launch_sample(hipStream_t s, const int64_t* indptr, const int64_t* indices,
                   const int64_t* eid_base, const int64_t* seeds, int64_t n,
                   int k, const int64_t* prefix, int64_t* out_nbrs,
                   int64_t* out_eids, uint64_t rng_seed,
                   const uint64_t* rng_dev = nullptr,
                   const int64_t* n_dev = nullptr);

// *rng += golden-ratio step; the per-batch seed advance as a graph node
// (a captured chain must advance the seed ON DEVICE each replay).
void launch_rng_bump(hipStream_t s, uint64_t* rng);

// Access-probability propagation (one hop):
// cur[v] = 1 - (1 - last[v]) * prod_{u in N(v)} (1 - last[u]*min(1, k/deg(u)))
void launch_cal_next(hipStream_t s, const int64_t* indptr,
                     const int64_t* indices, const float* last, float* cur,
                     int64_t node_count, int k);

// ---------- reindex (reindex_kernels.hip) ----------

// Open-addressing hash table working set; all buffers caller-allocated.
//   capacity: power of two >= 2*(n_seeds+n_nbrs)
//   keys:   int64[capacity]  (pre-filled with -1 by launch_reindex_init)
//   pos:    int32[capacity]
//   local:  int32[capacity]
//   flags/scan: int64[n_seeds+n_nbrs]
void launch_reindex_init(hipStream_t s, int64_t* keys, int32_t* pos,
                         int64_t capacity);
void launch_hash_insert(hipStream_t s, int64_t* keys, int32_t* pos,
                        int64_t capacity, const int64_t* seeds, int64_t n_seeds,
                        const int64_t* nbrs, int64_t n_nbrs,
                        const int64_t* n_seeds_dev = nullptr,
                        const int64_t* n_nbrs_dev = nullptr);
void launch_mark_first(hipStream_t s, const int64_t* keys, const int32_t* pos,
                       int64_t capacity, const int64_t* seeds, int64_t n_seeds,
                       const int64_t* nbrs, int64_t n_nbrs, int64_t* flags,
                       const int64_t* n_seeds_dev = nullptr,
                       const int64_t* n_nbrs_dev = nullptr);
void launch_compact_unique(hipStream_t s, const int64_t* keys, int32_t* local,
                           const int32_t* pos, int64_t capacity,
                           const int64_t* seeds, int64_t n_seeds,
                           const int64_t* nbrs, int64_t n_nbrs,
                           const int64_t* scanned_flags, const int64_t* flags,
                           int64_t* out_nodes,
                           const int64_t* n_seeds_dev = nullptr,
                           const int64_t* n_nbrs_dev = nullptr);
void launch_lookup_local(hipStream_t s, const int64_t* keys,
                         const int32_t* local, int64_t capacity,
                         const int64_t* nbrs, int64_t n_nbrs, int64_t* col_idx,
                         const int64_t* n_nbrs_dev = nullptr);
// row_idx[prefix[i]+j] = i  for j < counts[i]
void launch_expand_rows(hipStream_t s, const int64_t* prefix,
                        const int64_t* counts, int64_t n_seeds,
                        int64_t* row_idx, const int64_t* n_dev = nullptr);

// ---------- feature gather (gather_kernels.hip) ----------

constexpr int kMaxShards = 16;

// One virtual row-major tensor made of up to kMaxShards row ranges living in
// local HBM, peer-GPU HBM (xGMI) or pinned host memory (zero-copy).
struct GatherSpec {
    const char* ptrs[kMaxShards];  // base pointer of shard s (device-visible)
    int64_t ends[kMaxShards];      // exclusive prefix of row counts
    uint32_t access_mask;          // bit s set => shard s readable from here
    int nshards;
    int64_t row_bytes;
    bool has_host_shard;           // any zero-copy pinned-host tier present
};

// out[i] = row indices[i] of the virtual tensor; rows whose shard is not
// accessible are left untouched (python layer fills them via a peer pass).
// n is a host upper bound; when n_dev != nullptr the exact row count is
// read from the device (async sample+gather chains, no host sync).
void launch_gather(hipStream_t s, const GatherSpec& spec,
                   const int64_t* indices, int64_t n, char* out,
                   const int64_t* n_dev = nullptr);

// Scatter-style update used by the python layer for cache fill / tests:
// shard-resident rows only.  dst row indices[i] <- src[i].
void launch_scatter(hipStream_t s, const GatherSpec& spec,
                    const int64_t* indices, int64_t n, const char* src);

// Fused GAT attention dots: asrc[n,h] = <h[n,h,:], att_src[h,:]>, adst
// over the first n_dst rows (prefix convention); backward also emits
// g_h and the (tiny) att gradients via per-block LDS accumulation.
// h (and g_h) may be fp32 or bf16 (h_bf16); logits/att stay fp32.
void launch_gat_dots_fwd(hipStream_t s, const void* h, bool h_bf16,
                         const float* att_src, const float* att_dst,
                         int64_t n, int64_t n_dst, int heads, int chead,
                         float* asrc, float* adst);
void launch_gat_dots_bwd(hipStream_t s, const void* h, bool h_bf16,
                         const float* att_src, const float* att_dst,
                         const float* g_asrc, const float* g_adst,
                         int64_t n, int64_t n_dst, int heads, int chead,
                         void* g_h, float* g_att_src, float* g_att_dst);

// ---------- tall-M GEMM (gemm_kernels.hip) -------------------------------

// C[M×N] = A[M×K] @ B[K×N] (+ bias[N]) with M huge, K/N <= ~1024 and B
// K-major (pass W^T for a forward linear, W itself for its data-grad).
void launch_tall_gemm(hipStream_t s, const float* A, const float* B,
                      const float* bias, float* C, int64_t M, int K, int N);

// ---------- tall-skinny weight-grad GEMM (wgrad_kernels.hip) -------------

// C[M×N] = A^T @ B with A [K×M], B [K×N] row-major, K huge.  MFMA
// split-K: per-chunk partials land in a caller-provided workspace and a
// fixed-order reduce writes C — deterministic, no pre-zeroing needed.
struct WgradPlan {
    int tm, tn;         // C tile grid
    int64_t nchunks;    // split-K factor
    int64_t k_chunk;    // K rows per chunk
    int64_t ws_floats;  // workspace floats: nchunks * padded C (+ bias col)
};
WgradPlan wgrad_plan(int64_t K, int M, int N);
void launch_wgrad(hipStream_t s, const float* A, const float* B, float* C,
                  float* bias_grad, int64_t K, int M, int N,
                  const WgradPlan& plan, float* ws);

// ---------- fused message-passing aggregation (segment_kernels.hip) ------

// out[d] = mean over edges e in [dst_ptr[d], dst_ptr[d+1]) of x[src[e]]
void launch_segment_mean_fwd(hipStream_t s, const float* x,
                             const int64_t* src, const int64_t* dst_ptr,
                             int64_t n_dst, int64_t dim, float* out);
// bf16 variants: bf16 rows, fp32 accumulation, packed-bf16 atomics (bwd)
void launch_segment_mean_fwd_bf16(hipStream_t s, const void* x,
                                  const int64_t* src, const int64_t* dst_ptr,
                                  int64_t n_dst, int64_t dim, void* out);
void launch_segment_mean_bwd_bf16(hipStream_t s, const void* grad_out,
                                  const int64_t* src, const int64_t* dst_ptr,
                                  int64_t n_dst, int64_t dim, void* grad_x);
// grad_x[src[e]] += grad_out[d] / deg(d)   (grad_x pre-zeroed)
void launch_segment_mean_bwd(hipStream_t s, const float* grad_out,
                             const int64_t* src, const int64_t* dst_ptr,
                             int64_t n_dst, int64_t dim, float* grad_x);

// Weighted segment sum (GAT attention aggregation), x [n_src, H*C],
// w [n_edges, H] (per-edge per-head attention):
//   out[d, h*C+c] = sum_{e in segment(d)} w[e,h] * x[src[e], h*C+c]
void launch_segment_wsum_fwd_bf16(hipStream_t s, const void* x,
                                  const float* w, const int64_t* src,
                                  const int64_t* dst_ptr, int64_t n_dst,
                                  int heads, int chead, void* out);
void launch_segment_wsum_bwd_x_bf16(hipStream_t s, const void* grad_out,
                                    const float* w, const int64_t* src,
                                    const int64_t* dst_ptr, int64_t n_dst,
                                    int heads, int chead, void* grad_x);
void launch_segment_wsum_bwd_w_bf16(hipStream_t s, const void* grad_out,
                                    const void* x, const int64_t* src,
                                    const int64_t* dst_ptr, int64_t n_dst,
                                    int heads, int chead, float* grad_w);
void launch_segment_wsum_fwd(hipStream_t s, const float* x, const float* w,
                             const int64_t* src, const int64_t* dst_ptr,
                             int64_t n_dst, int heads, int chead, float* out);
// grad_x[src[e], h*C+c] += w[e,h] * grad_out[d, h*C+c]  (pre-zeroed)
void launch_segment_wsum_bwd_x(hipStream_t s, const float* grad_out,
                               const float* w, const int64_t* src,
                               const int64_t* dst_ptr, int64_t n_dst,
                               int heads, int chead, float* grad_x);
// grad_w[e,h] = sum_c grad_out[d, h*C+c] * x[src[e], h*C+c]
void launch_segment_wsum_bwd_w(hipStream_t s, const float* grad_out,
                               const float* x, const int64_t* src,
                               const int64_t* dst_ptr, int64_t n_dst,
                               int heads, int chead, float* grad_w);

// Segment softmax over [n_edges, H] grouped by dst_ptr segments:
//   out[e,h] = exp(a[e,h] - max_seg) / sum_seg exp(...)
// (numerically-stable; empty segments produce nothing)
void launch_segment_softmax_fwd(hipStream_t s, const float* a,
                                const int64_t* dst_ptr, int64_t n_dst,
                                int heads, float* out);
// g_a[e,h] = out[e,h] * (g[e,h] - sum_seg g*out)
void launch_segment_softmax_bwd(hipStream_t s, const float* grad_out,
                                const float* out, const int64_t* dst_ptr,
                                int64_t n_dst, int heads, float* grad_a);

// Fully fused GAT attention coefficients: from per-node logits to
// normalized per-edge attention in one kernel,
//   alpha[e,h] = softmax_seg(leaky_relu(asrc[src[e],h] + adst[d,h]))
void launch_gat_alpha_fwd(hipStream_t s, const float* asrc,
                          const float* adst, const int64_t* src,
                          const int64_t* dst_ptr, int64_t n_dst, int heads,
                          float slope, float* alpha);
// backward through softmax + leaky_relu + the two logit gathers:
// g_asrc accumulated with atomics (pre-zeroed), g_adst written directly.
void launch_gat_alpha_bwd(hipStream_t s, const float* grad_alpha,
                          const float* alpha, const float* asrc,
                          const float* adst, const int64_t* src,
                          const int64_t* dst_ptr, int64_t n_dst, int heads,
                          float slope, float* g_asrc, float* g_adst);

}  // namespace qk
