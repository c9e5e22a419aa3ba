// Fused segment mean-aggregation for GNN message passing (gfx950).
//
// The sampler emits each hop's edges with dst ids sorted ascending
// (row-expansion order), so SAGE mean aggregation is a segment reduction:
//   out[d] = mean_{e in [dst_ptr[d], dst_ptr[d+1])} x[src[e]]
// One fused kernel replaces the torch chain
//   x[src] (materialize [E,D]) -> zeros -> index_add -> deg -> div
// reading each x row once per edge and writing out once per dst.
//
// Backward: grad_x[src[e]] += grad_out[d] / deg(d)  (device-scope f32
// atomics; layer-1 input features skip it entirely since they carry no
// grad).
#include <hip/hip_bf16.h>

#include "qk_common.h"

namespace qk {

namespace {

constexpr int BLOCK = 256;
constexpr int SUB = 16;                     // lanes per dst segment
constexpr int ROWS_PER_BLOCK = BLOCK / SUB;
constexpr int VPL = 4;                      // floats per lane per chunk

// 4-element load/store in fp32 working precision for fp32 and bf16
// element types (bf16 goes through 4-byte bf16x2 pairs: aligned for any
// even dim; the value math stays fp32 and only stores round).
template <typename T> struct vec4io;
template <> struct vec4io<float> {
    static __device__ __forceinline__ void load(const float* p, float v[4]) {
        const float4 t = *reinterpret_cast<const float4*>(p);
        v[0] = t.x; v[1] = t.y; v[2] = t.z; v[3] = t.w;
    }
    static __device__ __forceinline__ void store(float* p, const float v[4]) {
        *reinterpret_cast<float4*>(p) = float4{v[0], v[1], v[2], v[3]};
    }
};
template <> struct vec4io<__hip_bfloat16> {
    static __device__ __forceinline__ void load(const __hip_bfloat16* p,
                                                float v[4]) {
#pragma unroll
        for (int i = 0; i < 2; ++i) {
            const float2 f = __bfloat1622float2(
                *reinterpret_cast<const __hip_bfloat162*>(p + 2 * i));
            v[2 * i] = f.x;
            v[2 * i + 1] = f.y;
        }
    }
    static __device__ __forceinline__ void store(__hip_bfloat16* p,
                                                 const float v[4]) {
#pragma unroll
        for (int i = 0; i < 2; ++i)
            *reinterpret_cast<__hip_bfloat162*>(p + 2 * i) =
                __float22bfloat162_rn(float2{v[2 * i], v[2 * i + 1]});
    }
};

__global__ void __launch_bounds__(BLOCK)
segment_mean_fwd_kernel(const float* __restrict__ x,
                        const int64_t* __restrict__ src,
                        const int64_t* __restrict__ dst_ptr, int64_t n_dst,
                        int64_t dim, float* __restrict__ out) {
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        const float inv = (end > beg) ? 1.0f / (float)(end - beg) : 0.0f;
        // lanes cover the feature dim in chunks of VPL floats
        for (int64_t c = (int64_t)lane * VPL; c < dim; c += SUB * VPL) {
            float acc[VPL] = {0.f, 0.f, 0.f, 0.f};
            const int w = (int)min((int64_t)VPL, dim - c);
            for (int64_t e = beg; e < end; ++e) {
                const float* row = x + src[e] * dim + c;
                if (w == VPL) {
                    const float4 v = *reinterpret_cast<const float4*>(row);
                    acc[0] += v.x; acc[1] += v.y; acc[2] += v.z; acc[3] += v.w;
                } else {
                    for (int q = 0; q < w; ++q) acc[q] += row[q];
                }
            }
            float* orow = out + d * dim + c;
            if (w == VPL) {
                float4 v{acc[0] * inv, acc[1] * inv, acc[2] * inv,
                         acc[3] * inv};
                *reinterpret_cast<float4*>(orow) = v;
            } else {
                for (int q = 0; q < w; ++q) orow[q] = acc[q] * inv;
            }
        }
    }
}

__global__ void __launch_bounds__(BLOCK)
segment_mean_bwd_kernel(const float* __restrict__ grad_out,
                        const int64_t* __restrict__ src,
                        const int64_t* __restrict__ dst_ptr, int64_t n_dst,
                        int64_t dim, float* __restrict__ grad_x) {
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        if (end <= beg) continue;
        const float inv = 1.0f / (float)(end - beg);
        for (int64_t e = beg; e < end; ++e) {
            float* grow = grad_x + src[e] * dim;
            const float* orow = grad_out + d * dim;
            for (int64_t c = lane; c < dim; c += SUB)
                atomicAdd(&grow[c], orow[c] * inv);
        }
    }
}

// ---- bf16 variants of the segment mean (fp32 accumulate) ----------------
// Feature rows load as 16-byte bf16x8 chunks; accumulation stays fp32 and
// only the store rounds.  Backward uses the packed-bf16 global atomic
// (flat_atomic_fadd_v2bf16) via HIP's unsafeAtomicAdd — fine here: the
// gradient buffer lives in ordinary HBM (no system-scope requirement).

constexpr int VPL_BF = 8;  // channels per lane chunk (4x bf16x2 loads:
                           // 4B-aligned for any even dim; a 16B vector
                           // would need dim % 8 == 0)

__global__ void __launch_bounds__(BLOCK)
segment_mean_fwd_bf16_kernel(const __hip_bfloat16* __restrict__ x,
                             const int64_t* __restrict__ src,
                             const int64_t* __restrict__ dst_ptr,
                             int64_t n_dst, int64_t dim,
                             __hip_bfloat16* __restrict__ out) {
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        const float inv = (end > beg) ? 1.0f / (float)(end - beg) : 0.0f;
        for (int64_t c = (int64_t)lane * VPL_BF; c < dim; c += SUB * VPL_BF) {
            float acc[VPL_BF] = {};
            const int w = (int)min((int64_t)VPL_BF, dim - c);
            // dim % 4 == 0 makes every row base 8-byte aligned: use two
            // bf16x4 loads per 8-channel chunk instead of four bf16x2
            const bool v8 = (dim & 3) == 0;
            for (int64_t e = beg; e < end; ++e) {
                const __hip_bfloat16* row = x + src[e] * dim + c;
                if (w == VPL_BF && v8) {
#pragma unroll
                    for (int i = 0; i < 2; ++i) {
                        struct alignas(8) bf16x4v {
                            __hip_bfloat162 lo, hi;
                        };
                        const bf16x4v v4 =
                            *reinterpret_cast<const bf16x4v*>(row + 4 * i);
                        const float2 f0 = __bfloat1622float2(v4.lo);
                        const float2 f1 = __bfloat1622float2(v4.hi);
                        acc[4 * i] += f0.x;
                        acc[4 * i + 1] += f0.y;
                        acc[4 * i + 2] += f1.x;
                        acc[4 * i + 3] += f1.y;
                    }
                } else if (w == VPL_BF) {
#pragma unroll
                    for (int i = 0; i < 4; ++i) {
                        const float2 f = __bfloat1622float2(
                            *reinterpret_cast<const __hip_bfloat162*>(
                                row + 2 * i));
                        acc[2 * i] += f.x;
                        acc[2 * i + 1] += f.y;
                    }
                } else {
                    for (int q = 0; q < w; ++q) acc[q] += (float)row[q];
                }
            }
            __hip_bfloat16* orow = out + d * dim + c;
            if (w == VPL_BF) {
#pragma unroll
                for (int i = 0; i < 4; ++i)
                    *reinterpret_cast<__hip_bfloat162*>(orow + 2 * i) =
                        __float22bfloat162_rn(float2{acc[2 * i] * inv,
                                                     acc[2 * i + 1] * inv});
            } else {
                for (int q = 0; q < w; ++q)
                    orow[q] = (__hip_bfloat16)(acc[q] * inv);
            }
        }
    }
}

__global__ void __launch_bounds__(BLOCK)
segment_mean_bwd_bf16_kernel(const __hip_bfloat16* __restrict__ grad_out,
                             const int64_t* __restrict__ src,
                             const int64_t* __restrict__ dst_ptr,
                             int64_t n_dst, int64_t dim,
                             __hip_bfloat16* __restrict__ grad_x) {
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        if (end <= beg) continue;
        const float inv = 1.0f / (float)(end - beg);
        for (int64_t e = beg; e < end; ++e) {
            __hip_bfloat16* grow = grad_x + src[e] * dim;
            const __hip_bfloat16* orow = grad_out + d * dim;
            // packed atomic: lanes own even channel pairs
            for (int64_t c = 2 * lane; c + 1 < dim; c += 2 * SUB) {
                const float2 g = __bfloat1622float2(
                    *reinterpret_cast<const __hip_bfloat162*>(&orow[c]));
                const __hip_bfloat162 add = __float22bfloat162_rn(
                    float2{g.x * inv, g.y * inv});
                unsafeAtomicAdd(reinterpret_cast<__hip_bfloat162*>(&grow[c]),
                                add);
            }
            if (lane == 0 && (dim & 1)) {
                // odd tail channel
                const float g = (float)orow[dim - 1] * inv;
                unsafeAtomicAdd(&grow[dim - 1], (__hip_bfloat16)g);
            }
        }
    }
}

// ---- weighted segment sum (GAT attention aggregation) -------------------
// dim = heads*chead; weights indexed per (edge, head).  Lanes stride the
// feature dim; head of a channel c is c / chead (chead is a multiple of
// VPL in the models here, so a VPL chunk never straddles heads — guarded
// by the launcher).

template <typename XT>
__global__ void __launch_bounds__(BLOCK)
segment_wsum_fwd_kernel(const XT* __restrict__ x,
                        const float* __restrict__ w,
                        const int64_t* __restrict__ src,
                        const int64_t* __restrict__ dst_ptr, int64_t n_dst,
                        int heads, int chead, XT* __restrict__ out) {
    const int64_t dim = (int64_t)heads * chead;
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        for (int64_t c = (int64_t)lane * VPL; c < dim; c += SUB * VPL) {
            const int h = (int)(c / chead);
            float acc[VPL] = {0.f, 0.f, 0.f, 0.f};
            const int wd = (int)min((int64_t)VPL, dim - c);
            for (int64_t e = beg; e < end; ++e) {
                const float we = w[e * heads + h];
                const XT* row = x + src[e] * dim + c;
                if (wd == VPL) {
                    float v[VPL];
                    vec4io<XT>::load(row, v);
#pragma unroll
                    for (int q = 0; q < VPL; ++q) acc[q] += we * v[q];
                } else {
                    for (int q = 0; q < wd; ++q)
                        acc[q] += we * (float)row[q];
                }
            }
            XT* orow = out + d * dim + c;
            if (wd == VPL) {
                vec4io<XT>::store(orow, acc);
            } else {
                for (int q = 0; q < wd; ++q) orow[q] = (XT)acc[q];
            }
        }
    }
}

__global__ void __launch_bounds__(BLOCK)
segment_wsum_bwd_x_kernel(const float* __restrict__ grad_out,
                          const float* __restrict__ w,
                          const int64_t* __restrict__ src,
                          const int64_t* __restrict__ dst_ptr, int64_t n_dst,
                          int heads, int chead, float* __restrict__ grad_x) {
    const int64_t dim = (int64_t)heads * chead;
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        const float* orow = grad_out + d * dim;
        for (int64_t e = beg; e < end; ++e) {
            float* grow = grad_x + src[e] * dim;
            const float* wrow = w + e * heads;
            for (int64_t c = lane; c < dim; c += SUB)
                atomicAdd(&grow[c], wrow[c / chead] * orow[c]);
        }
    }
}

// bf16 variant: grad_out and grad_x are bf16; weights fp32.  Lanes own
// CHANNEL PAIRS so the scatter uses the packed-bf16 global atomic
// (dim % 2 == 0 guarded by the launcher; chead % 2 == 0 keeps a pair
// within one head when heads > 1 -- implied by the chead % 4 guard).
__global__ void __launch_bounds__(BLOCK)
segment_wsum_bwd_x_bf16_kernel(const __hip_bfloat16* __restrict__ grad_out,
                               const float* __restrict__ w,
                               const int64_t* __restrict__ src,
                               const int64_t* __restrict__ dst_ptr,
                               int64_t n_dst, int heads, int chead,
                               __hip_bfloat16* __restrict__ grad_x) {
    const int64_t dim = (int64_t)heads * chead;
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        const __hip_bfloat16* orow = grad_out + d * dim;
        for (int64_t e = beg; e < end; ++e) {
            __hip_bfloat16* grow = grad_x + src[e] * dim;
            const float* wrow = w + e * heads;
            for (int64_t c = 2 * lane; c + 1 < dim; c += 2 * SUB) {
                const float we = wrow[c / chead];
                const float2 g = __bfloat1622float2(
                    *reinterpret_cast<const __hip_bfloat162*>(&orow[c]));
                unsafeAtomicAdd(
                    reinterpret_cast<__hip_bfloat162*>(&grow[c]),
                    __float22bfloat162_rn(float2{we * g.x, we * g.y}));
            }
        }
    }
}

template <typename XT>
__global__ void __launch_bounds__(BLOCK)
segment_wsum_bwd_w_kernel(const XT* __restrict__ grad_out,
                          const XT* __restrict__ x,
                          const int64_t* __restrict__ src,
                          const int64_t* __restrict__ dst_ptr, int64_t n_dst,
                          int heads, int chead, float* __restrict__ grad_w) {
    // one subgroup per dst; lanes cooperate per edge-head dot over chead
    const int64_t dim = (int64_t)heads * chead;
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t d = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; d < n_dst; d += stride) {
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        const XT* orow = grad_out + d * dim;
        for (int64_t e = beg; e < end; ++e) {
            const XT* xrow = x + src[e] * dim;
            for (int h = 0; h < heads; ++h) {
                float acc = 0.f;
                const int64_t base = (int64_t)h * chead;
                for (int c = lane; c < chead; c += SUB)
                    acc += (float)orow[base + c] * (float)xrow[base + c];
                for (int off = SUB / 2; off > 0; off >>= 1)
                    acc += __shfl_down(acc, off, SUB);
                if (lane == 0) grad_w[e * heads + h] = acc;
            }
        }
    }
}

// ---- segment softmax ([E,H] grouped by dst segments) --------------------
// One thread per (dst, head): segments are fanout-sized (<= ~25), so a
// scalar two-pass (max, exp-sum) loop is cheap and avoids the sort-based
// torch scatter_reduce lowering (~120 rocprim kernels per step).

__global__ void __launch_bounds__(BLOCK)
segment_softmax_fwd_kernel(const float* __restrict__ a,
                           const int64_t* __restrict__ dst_ptr,
                           int64_t n_dst, int heads, float* __restrict__ out) {
    int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t total = n_dst * heads;
    for (; t < total; t += stride) {
        const int64_t d = t / heads;
        const int h = (int)(t % heads);
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        if (end <= beg) continue;
        float m = -INFINITY;
        for (int64_t e = beg; e < end; ++e)
            m = fmaxf(m, a[e * heads + h]);
        float s = 0.f;
        for (int64_t e = beg; e < end; ++e)
            s += __expf(a[e * heads + h] - m);
        const float inv = 1.0f / fmaxf(s, 1e-16f);
        for (int64_t e = beg; e < end; ++e)
            out[e * heads + h] = __expf(a[e * heads + h] - m) * inv;
    }
}

__global__ void __launch_bounds__(BLOCK)
segment_softmax_bwd_kernel(const float* __restrict__ grad_out,
                           const float* __restrict__ out,
                           const int64_t* __restrict__ dst_ptr,
                           int64_t n_dst, int heads,
                           float* __restrict__ grad_a) {
    int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t total = n_dst * heads;
    for (; t < total; t += stride) {
        const int64_t d = t / heads;
        const int h = (int)(t % heads);
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        float s = 0.f;
        for (int64_t e = beg; e < end; ++e)
            s += grad_out[e * heads + h] * out[e * heads + h];
        for (int64_t e = beg; e < end; ++e)
            grad_a[e * heads + h] =
                out[e * heads + h] * (grad_out[e * heads + h] - s);
    }
}

// ---- fully fused GAT attention coefficients -----------------------------
// One thread per (dst, head).  Segments are fanout-sized; the
// pre-activation is recomputed per pass (asrc/adst stay L2-hot) instead
// of materializing any [E,H] intermediate.

__device__ __forceinline__ float lrelu(float v, float slope) {
    return v > 0.f ? v : v * slope;
}

__global__ void __launch_bounds__(BLOCK)
gat_alpha_fwd_kernel(const float* __restrict__ asrc,
                     const float* __restrict__ adst,
                     const int64_t* __restrict__ src,
                     const int64_t* __restrict__ dst_ptr, int64_t n_dst,
                     int heads, float slope, float* __restrict__ alpha) {
    int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t total = n_dst * heads;
    for (; t < total; t += stride) {
        const int64_t d = t / heads;
        const int h = (int)(t % heads);
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        if (end <= beg) continue;
        const float ad = adst[d * heads + h];
        float m = -INFINITY;
        for (int64_t e = beg; e < end; ++e)
            m = fmaxf(m, lrelu(asrc[src[e] * heads + h] + ad, slope));
        float s = 0.f;
        for (int64_t e = beg; e < end; ++e)
            s += __expf(lrelu(asrc[src[e] * heads + h] + ad, slope) - m);
        const float inv = 1.0f / fmaxf(s, 1e-16f);
        for (int64_t e = beg; e < end; ++e)
            alpha[e * heads + h] =
                __expf(lrelu(asrc[src[e] * heads + h] + ad, slope) - m) * inv;
    }
}

__global__ void __launch_bounds__(BLOCK)
gat_alpha_bwd_kernel(const float* __restrict__ grad_alpha,
                     const float* __restrict__ alpha,
                     const float* __restrict__ asrc,
                     const float* __restrict__ adst,
                     const int64_t* __restrict__ src,
                     const int64_t* __restrict__ dst_ptr, int64_t n_dst,
                     int heads, float slope, float* __restrict__ g_asrc,
                     float* __restrict__ g_adst) {
    int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t total = n_dst * heads;
    for (; t < total; t += stride) {
        const int64_t d = t / heads;
        const int h = (int)(t % heads);
        const int64_t beg = dst_ptr[d], end = dst_ptr[d + 1];
        const float ad = adst[d * heads + h];
        float S = 0.f;
        for (int64_t e = beg; e < end; ++e)
            S += grad_alpha[e * heads + h] * alpha[e * heads + h];
        float acc = 0.f;
        for (int64_t e = beg; e < end; ++e) {
            const float gsm = alpha[e * heads + h] *
                              (grad_alpha[e * heads + h] - S);
            const float pre = asrc[src[e] * heads + h] + ad;
            const float gpre = pre > 0.f ? gsm : gsm * slope;
            atomicAdd(&g_asrc[src[e] * heads + h], gpre);
            acc += gpre;
        }
        g_adst[d * heads + h] = acc;
    }
}

// ---- fused GAT attention dots --------------------------------------------
// asrc[n,h] = <h[n,h,:], att_src[h,:]>; adst likewise over the first
// n_dst rows (bipartite prefix convention: h_dst == h_src[:n_dst]).
// Replaces the torch chain (h*att).sum(-1) x2 (a 1 GB broadcast-mul
// write + 1 GB reduce read per term) with ONE read of h.
constexpr int DSUB = 16;  // lanes per (row, head) pair

template <typename HT>
__global__ void __launch_bounds__(BLOCK)
gat_dots_fwd_kernel(const HT* __restrict__ h,
                    const float* __restrict__ att_src,
                    const float* __restrict__ att_dst, int64_t n,
                    int64_t n_dst, int heads, int chead,
                    float* __restrict__ asrc, float* __restrict__ adst) {
    const int sub_id = threadIdx.x / DSUB;
    const int lane = threadIdx.x % DSUB;
    const int per_block = BLOCK / DSUB;
    int64_t p = (int64_t)blockIdx.x * per_block + sub_id;
    const int64_t stride = (int64_t)gridDim.x * per_block;
    const int64_t total = n * heads;
    for (; p < total; p += stride) {
        const int64_t row = p / heads;
        const int hd = (int)(p % heads);
        const HT* hrow = h + (row * heads + hd) * (int64_t)chead;
        const float* as = att_src + hd * chead;
        const float* ad = att_dst + hd * chead;
        float sv = 0.f, dv = 0.f;
        const bool do_dst = row < n_dst;
        for (int c = lane * 4; c < chead; c += DSUB * 4) {
            float v[4];
            vec4io<HT>::load(hrow + c, v);
            const float4 a = *reinterpret_cast<const float4*>(as + c);
            sv += v[0] * a.x + v[1] * a.y + v[2] * a.z + v[3] * a.w;
            if (do_dst) {
                const float4 b = *reinterpret_cast<const float4*>(ad + c);
                dv += v[0] * b.x + v[1] * b.y + v[2] * b.z + v[3] * b.w;
            }
        }
        for (int off = DSUB / 2; off; off >>= 1) {
            sv += __shfl_down(sv, off, DSUB);
            dv += __shfl_down(dv, off, DSUB);
        }
        if (lane == 0) {
            asrc[row * heads + hd] = sv;
            if (do_dst) adst[row * heads + hd] = dv;
        }
    }
}

// g_h[row,h,:] = g_asrc[row,h]*att_src[h,:] (+ g_adst part for prefix
// rows); g_att_* accumulate per-block in LDS, one global atomic per
// element per block (the [H*C] output is tiny next to the 1M-row input).
template <typename HT>
__global__ void __launch_bounds__(BLOCK)
gat_dots_bwd_kernel(const HT* __restrict__ h,
                    const float* __restrict__ att_src,
                    const float* __restrict__ att_dst,
                    const float* __restrict__ g_asrc,
                    const float* __restrict__ g_adst, int64_t n,
                    int64_t n_dst, int heads, int chead,
                    HT* __restrict__ g_h, float* __restrict__ g_att_src,
                    float* __restrict__ g_att_dst) {
    extern __shared__ float lacc[];  // [2][heads*chead]
    const int D = heads * chead;
    float* lsrc = lacc;
    float* ldst = lacc + D;
    for (int i = threadIdx.x; i < 2 * D; i += BLOCK) lacc[i] = 0.f;
    __syncthreads();
    const int sub_id = threadIdx.x / DSUB;
    const int lane = threadIdx.x % DSUB;
    const int per_block = BLOCK / DSUB;
    int64_t p = (int64_t)blockIdx.x * per_block + sub_id;
    const int64_t stride = (int64_t)gridDim.x * per_block;
    const int64_t total = n * heads;
    for (; p < total; p += stride) {
        const int64_t row = p / heads;
        const int hd = (int)(p % heads);
        const HT* hrow = h + (row * heads + hd) * (int64_t)chead;
        HT* grow = g_h + (row * heads + hd) * (int64_t)chead;
        const float* as = att_src + hd * chead;
        const float* ad = att_dst + hd * chead;
        const bool do_dst = row < n_dst;
        const float gs = g_asrc[row * heads + hd];
        const float gd = do_dst ? g_adst[row * heads + hd] : 0.f;
        for (int c = lane * 4; c < chead; c += DSUB * 4) {
            float v[4];
            vec4io<HT>::load(hrow + c, v);
            const float4 a = *reinterpret_cast<const float4*>(as + c);
            float g[4] = {gs * a.x, gs * a.y, gs * a.z, gs * a.w};
            if (do_dst) {
                const float4 b = *reinterpret_cast<const float4*>(ad + c);
                g[0] += gd * b.x; g[1] += gd * b.y;
                g[2] += gd * b.z; g[3] += gd * b.w;
            }
            vec4io<HT>::store(grow + c, g);
            const int base = hd * chead + c;
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                atomicAdd(&lsrc[base + q], gs * v[q]);
                if (do_dst) atomicAdd(&ldst[base + q], gd * v[q]);
            }
        }
    }
    __syncthreads();
    for (int i = threadIdx.x; i < D; i += BLOCK) {
        atomicAdd(&g_att_src[i], lsrc[i]);
        atomicAdd(&g_att_dst[i], ldst[i]);
    }
}

inline int grid_for(int64_t work, int per_block) {
    int64_t blocks = (work + per_block - 1) / per_block;
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

}  // namespace

void launch_segment_mean_fwd(hipStream_t s, const float* x,
                             const int64_t* src, const int64_t* dst_ptr,
                             int64_t n_dst, int64_t dim, float* out) {
    if (n_dst == 0) return;
    segment_mean_fwd_kernel<<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
        x, src, dst_ptr, n_dst, dim, out);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_mean_bwd(hipStream_t s, const float* grad_out,
                             const int64_t* src, const int64_t* dst_ptr,
                             int64_t n_dst, int64_t dim, float* grad_x) {
    if (n_dst == 0) return;
    segment_mean_bwd_kernel<<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
        grad_out, src, dst_ptr, n_dst, dim, grad_x);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_mean_fwd_bf16(hipStream_t s, const void* x,
                                  const int64_t* src, const int64_t* dst_ptr,
                                  int64_t n_dst, int64_t dim, void* out) {
    if (n_dst == 0) return;
    segment_mean_fwd_bf16_kernel<<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0,
                                   s>>>((const __hip_bfloat16*)x, src,
                                        dst_ptr, n_dst, dim,
                                        (__hip_bfloat16*)out);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_mean_bwd_bf16(hipStream_t s, const void* grad_out,
                                  const int64_t* src, const int64_t* dst_ptr,
                                  int64_t n_dst, int64_t dim, void* grad_x) {
    if (n_dst == 0) return;
    segment_mean_bwd_bf16_kernel<<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0,
                                   s>>>((const __hip_bfloat16*)grad_out, src,
                                        dst_ptr, n_dst, dim,
                                        (__hip_bfloat16*)grad_x);
    QK_CHECK_HIP(hipGetLastError());
}

static void check_chead(int heads, int chead) {
    // heads == 1: the whole dim is one head, a vector chunk cannot
    // straddle a head boundary regardless of chead
    if (heads > 1 && chead % VPL != 0)
        throw std::runtime_error(
            "segment_wsum: per-head channels must be a multiple of 4 "
            "so vector chunks stay within one head");
}

void launch_segment_wsum_fwd(hipStream_t s, const float* x, const float* w,
                             const int64_t* src, const int64_t* dst_ptr,
                             int64_t n_dst, int heads, int chead, float* out) {
    if (n_dst == 0) return;
    check_chead(heads, chead);
    segment_wsum_fwd_kernel<float>
        <<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
            x, w, src, dst_ptr, n_dst, heads, chead, out);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_wsum_fwd_bf16(hipStream_t s, const void* x,
                                  const float* w, const int64_t* src,
                                  const int64_t* dst_ptr, int64_t n_dst,
                                  int heads, int chead, void* out) {
    if (n_dst == 0) return;
    check_chead(heads, chead);
    segment_wsum_fwd_kernel<__hip_bfloat16>
        <<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
            (const __hip_bfloat16*)x, w, src, dst_ptr, n_dst, heads, chead,
            (__hip_bfloat16*)out);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_wsum_bwd_x(hipStream_t s, const float* grad_out,
                               const float* w, const int64_t* src,
                               const int64_t* dst_ptr, int64_t n_dst,
                               int heads, int chead, float* grad_x) {
    if (n_dst == 0) return;
    check_chead(heads, chead);
    segment_wsum_bwd_x_kernel<<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0,
                                s>>>(grad_out, w, src, dst_ptr, n_dst, heads,
                                     chead, grad_x);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_wsum_bwd_x_bf16(hipStream_t s, const void* grad_out,
                                    const float* w, const int64_t* src,
                                    const int64_t* dst_ptr, int64_t n_dst,
                                    int heads, int chead, void* grad_x) {
    if (n_dst == 0) return;
    check_chead(heads, chead);
    segment_wsum_bwd_x_bf16_kernel
        <<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
            (const __hip_bfloat16*)grad_out, w, src, dst_ptr, n_dst, heads,
            chead, (__hip_bfloat16*)grad_x);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_wsum_bwd_w(hipStream_t s, const float* grad_out,
                               const float* x, const int64_t* src,
                               const int64_t* dst_ptr, int64_t n_dst,
                               int heads, int chead, float* grad_w) {
    if (n_dst == 0) return;
    check_chead(heads, chead);
    segment_wsum_bwd_w_kernel<float>
        <<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
            grad_out, x, src, dst_ptr, n_dst, heads, chead, grad_w);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_wsum_bwd_w_bf16(hipStream_t s, const void* grad_out,
                                    const void* x, const int64_t* src,
                                    const int64_t* dst_ptr, int64_t n_dst,
                                    int heads, int chead, float* grad_w) {
    if (n_dst == 0) return;
    check_chead(heads, chead);
    segment_wsum_bwd_w_kernel<__hip_bfloat16>
        <<<grid_for(n_dst, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
            (const __hip_bfloat16*)grad_out, (const __hip_bfloat16*)x, src,
            dst_ptr, n_dst, heads, chead, grad_w);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_softmax_fwd(hipStream_t s, const float* a,
                                const int64_t* dst_ptr, int64_t n_dst,
                                int heads, float* out) {
    if (n_dst == 0) return;
    segment_softmax_fwd_kernel<<<grid_for(n_dst * heads, BLOCK), BLOCK, 0,
                                 s>>>(a, dst_ptr, n_dst, heads, out);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_segment_softmax_bwd(hipStream_t s, const float* grad_out,
                                const float* out, const int64_t* dst_ptr,
                                int64_t n_dst, int heads, float* grad_a) {
    if (n_dst == 0) return;
    segment_softmax_bwd_kernel<<<grid_for(n_dst * heads, BLOCK), BLOCK, 0,
                                 s>>>(grad_out, out, dst_ptr, n_dst, heads,
                                      grad_a);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_gat_alpha_fwd(hipStream_t s, const float* asrc,
                          const float* adst, const int64_t* src,
                          const int64_t* dst_ptr, int64_t n_dst, int heads,
                          float slope, float* alpha) {
    if (n_dst == 0) return;
    gat_alpha_fwd_kernel<<<grid_for(n_dst * heads, BLOCK), BLOCK, 0, s>>>(
        asrc, adst, src, dst_ptr, n_dst, heads, slope, alpha);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_gat_alpha_bwd(hipStream_t s, const float* grad_alpha,
                          const float* alpha, const float* asrc,
                          const float* adst, const int64_t* src,
                          const int64_t* dst_ptr, int64_t n_dst, int heads,
                          float slope, float* g_asrc, float* g_adst) {
    if (n_dst == 0) return;
    gat_alpha_bwd_kernel<<<grid_for(n_dst * heads, BLOCK), BLOCK, 0, s>>>(
        grad_alpha, alpha, asrc, adst, src, dst_ptr, n_dst, heads, slope,
        g_asrc, g_adst);
    QK_CHECK_HIP(hipGetLastError());
}


void launch_gat_dots_fwd(hipStream_t s, const void* h, bool h_bf16,
                         const float* att_src, const float* att_dst,
                         int64_t n, int64_t n_dst, int heads, int chead,
                         float* asrc, float* adst) {
    if (n == 0) return;
    if (chead % 4 != 0)
        throw std::runtime_error("gat_dots: chead must be a multiple of 4");
    const int grid = grid_for(n * heads, BLOCK / DSUB);
    if (h_bf16)
        gat_dots_fwd_kernel<__hip_bfloat16><<<grid, BLOCK, 0, s>>>(
            (const __hip_bfloat16*)h, att_src, att_dst, n, n_dst, heads,
            chead, asrc, adst);
    else
        gat_dots_fwd_kernel<float><<<grid, BLOCK, 0, s>>>(
            (const float*)h, att_src, att_dst, n, n_dst, heads, chead,
            asrc, adst);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_gat_dots_bwd(hipStream_t s, const void* h, bool h_bf16,
                         const float* att_src, const float* att_dst,
                         const float* g_asrc, const float* g_adst,
                         int64_t n, int64_t n_dst, int heads, int chead,
                         void* g_h, float* g_att_src, float* g_att_dst) {
    if (n == 0) return;
    if (chead % 4 != 0)
        throw std::runtime_error("gat_dots: chead must be a multiple of 4");
    const int D = heads * chead;
    if (D > 4096)
        throw std::runtime_error("gat_dots: heads*chead too large for LDS");
    const int grid = grid_for(n * heads, BLOCK / DSUB);
    if (h_bf16)
        gat_dots_bwd_kernel<__hip_bfloat16>
            <<<grid, BLOCK, 2 * D * sizeof(float), s>>>(
                (const __hip_bfloat16*)h, att_src, att_dst, g_asrc, g_adst,
                n, n_dst, heads, chead, (__hip_bfloat16*)g_h, g_att_src,
                g_att_dst);
    else
        gat_dots_bwd_kernel<float>
            <<<grid, BLOCK, 2 * D * sizeof(float), s>>>(
                (const float*)h, att_src, att_dst, g_asrc, g_adst, n,
                n_dst, heads, chead, (float*)g_h, g_att_src, g_att_dst);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
