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
#include "qk_common.h"

namespace qk {

namespace {

constexpr int BLOCK = 256;
constexpr int SUB = 16;                     // lanes per dst segment
constexpr int ROWS_PER_BLOCK = BLOCK / SUB;
constexpr int VPL = 4;                      // floats per lane per chunk

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

}  // namespace qk
