// Tall-skinny weight-gradient GEMM for gfx950: C[M×N] = A^T @ B where
// A is [K×M], B is [K×N], K is huge (a GNN frontier, 10^4..10^6 rows) and
// M,N are layer widths (<= a few hundred).
//
// rocBLAS/Tensile handles this shape with stream-K 32x32 macro-tiles and
// reaches ~130 GB/s on the bench's layer-1 wgrad (1.1 ms for a 140 MB
// reduction) — 60x off the HBM3E roof.  This kernel is a plain split-K
// SGEMM: each block owns a K-chunk and a 64x64 C tile, stages 16-row
// slices of A and B through LDS, accumulates 4x4 per thread in VGPRs and
// atomically adds its partial into C.  fp32 global atomics make the
// reduction order nondeterministic (like Tensile's GSU path); tests
// compare with a K-scaled tolerance.
//
// The same pass optionally folds the bias gradient (column sums of A) in:
// blocks in the first N-tile column accumulate their LDS A-slices, saving
// the separate 100 GB+/s torch reduce_kernel pass over grad_out.
//
// Reference parity note: torch-quiver has no custom GEMMs (its models are
// PyG's); this exists because the MI355X rebuild owns its model layer
// (quiver/nn.py) end to end.
#include "qk_common.h"

namespace qk {

namespace {

constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 16;
constexpr int TDIM = 16;  // 16x16 threads, 4x4 microtile each

__global__ void __launch_bounds__(TDIM * TDIM)
wgrad_kernel(const float* __restrict__ A, const float* __restrict__ B,
             float* __restrict__ C, float* __restrict__ bias_grad,
             int64_t K, int M, int N, int64_t k_chunk) {
    __shared__ float As[BK][BM];
    __shared__ float Bs[BK][BN];

    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int64_t k_beg = (int64_t)blockIdx.z * k_chunk;
    const int64_t k_end = min(K, k_beg + k_chunk);

    const int tid = threadIdx.y * TDIM + threadIdx.x;
    const int tx = threadIdx.x;  // -> M direction (4 cols of A^T)
    const int ty = threadIdx.y;  // -> N direction

    float acc[4][4] = {};
    float bsum[4] = {};
    const bool do_bias = (bias_grad != nullptr) && (blockIdx.y == 0);

    for (int64_t k0 = k_beg; k0 < k_end; k0 += BK) {
        // stage A[k0..k0+BK) x [m0..m0+BM) and the matching B slice.
        // 1024 elements each, 4 per thread, coalesced over the row dim.
        for (int l = tid * 4; l < BK * BM; l += TDIM * TDIM * 4) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                int kk = (l + i) / BM, mm = (l + i) % BM;
                int64_t k = k0 + kk;
                As[kk][mm] = (k < k_end && m0 + mm < M)
                                 ? A[k * M + m0 + mm] : 0.f;
            }
        }
        for (int l = tid * 4; l < BK * BN; l += TDIM * TDIM * 4) {
#pragma unroll
            for (int i = 0; i < 4; ++i) {
                int kk = (l + i) / BN, nn = (l + i) % BN;
                int64_t k = k0 + kk;
                Bs[kk][nn] = (k < k_end && n0 + nn < N)
                                 ? B[k * N + n0 + nn] : 0.f;
            }
        }
        __syncthreads();
#pragma unroll
        for (int kk = 0; kk < BK; ++kk) {
            float a[4], b[4];
#pragma unroll
            for (int i = 0; i < 4; ++i) a[i] = As[kk][tx * 4 + i];
#pragma unroll
            for (int j = 0; j < 4; ++j) b[j] = Bs[kk][ty * 4 + j];
#pragma unroll
            for (int i = 0; i < 4; ++i)
#pragma unroll
                for (int j = 0; j < 4; ++j) acc[i][j] += a[i] * b[j];
            if (do_bias && ty == 0)
#pragma unroll
                for (int i = 0; i < 4; ++i) bsum[i] += a[i];
        }
        __syncthreads();
    }

#pragma unroll
    for (int i = 0; i < 4; ++i) {
        int m = m0 + tx * 4 + i;
        if (m >= M) continue;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            int n = n0 + ty * 4 + j;
            if (n < N) atomicAdd(&C[(int64_t)m * N + n], acc[i][j]);
        }
    }
    if (do_bias && ty == 0)
#pragma unroll
        for (int i = 0; i < 4; ++i) {
            int m = m0 + tx * 4 + i;
            if (m < M) atomicAdd(&bias_grad[m], bsum[i]);
        }
}

}  // namespace

void launch_wgrad(hipStream_t s, const float* A, const float* B, float* C,
                  float* bias_grad, int64_t K, int M, int N) {
    if (K == 0 || M == 0 || N == 0) return;
    int tm = (M + BM - 1) / BM, tn = (N + BN - 1) / BN;
    // size the K split so the grid comfortably covers 256 CUs x 8 XCDs
    int64_t want_blocks = 2048;
    int64_t nchunks = want_blocks / (tm * tn);
    int64_t min_chunk = 4 * BK;
    int64_t max_chunks = (K + min_chunk - 1) / min_chunk;
    if (nchunks > max_chunks) nchunks = max_chunks;
    if (nchunks < 1) nchunks = 1;
    int64_t k_chunk = ((K + nchunks - 1) / nchunks + BK - 1) / BK * BK;
    nchunks = (K + k_chunk - 1) / k_chunk;
    dim3 grid(tm, tn, (unsigned)nchunks);
    dim3 block(TDIM, TDIM);
    wgrad_kernel<<<grid, block, 0, s>>>(A, B, C, bias_grad, K, M, N,
                                        k_chunk);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
