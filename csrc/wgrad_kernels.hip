// Tall-skinny weight-gradient GEMM for gfx950: C[M×N] = A^T @ B where
// A is [K×M], B is [K×N], K is huge (a GNN frontier, 10^4..10^6 rows) and
// M,N are layer widths (<= a few hundred).
//
// rocBLAS/Tensile handles this shape with stream-K 32x32 macro-tiles and
// reaches ~130 GB/s on the bench's layer-1 wgrad — 60x off the HBM3E
// roof.  This kernel is a split-K GEMM on the f32 matrix cores
// (v_mfma_f32_16x16x4_f32, exact f32 numerics at 157 TF chip peak — the
// VALU path it replaced ran the same shape ~2.5-3x slower):
//
//  - The A^T shape is a gift on CDNA4: the MFMA A-fragment for a 16x16x4
//    tile wants lane l to hold A_mfma[m = l%16][k = l/16], which for our
//    K-major A is A[(k0 + l/16)*M + m0 + l%16] — 16 consecutive lanes
//    read 16 consecutive floats.  Both operands stage through LDS
//    (row-padded +4 floats: the 4 k-group lanes of a fragment read rows
//    64 floats apart, which would 2-way-conflict unpadded) purely for
//    cross-wave reuse; no transpose anywhere.
//  - Block = 4 waves = a 64x64 C macro-tile; each wave owns a 2x2 grid of
//    16x16 MFMA tiles (4 independent f32x4 accumulators -> the 40-cycle
//    dependent-accumulator latency is hidden at the 32-cycle issue rate).
//  - Split-K: each block reduces its K-chunk and atomically adds the
//    partial into C (fp32 atomics: reduction order nondeterministic, like
//    Tensile's GSU path; tests compare with a K-scaled tolerance).
//  - The bias gradient (column sums of A) folds into the same pass:
//    blocks in the first N-tile column accumulate their LDS A-slices,
//    saving the separate 100 GB+/s torch reduce pass over grad_out.
//
// Reference parity note: torch-quiver has no custom GEMMs (its models are
// PyG's); this exists because the MI355X rebuild owns its model layer
// (quiver/nn.py) end to end.
#include "qk_common.h"

namespace qk {

namespace {

typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 64;        // C tile rows per block
constexpr int BN = 64;        // C tile cols per block
constexpr int BK = 16;        // K rows staged per iteration
constexpr int LDP = BM + 4;   // LDS row pitch (pad: kill k-group conflicts)
constexpr int WAVES = 4;      // 2x2 wave grid over the 64x64 tile

__global__ void __launch_bounds__(WAVES * 64)
wgrad_mfma_kernel(const float* __restrict__ A, const float* __restrict__ B,
                  float* __restrict__ C, float* __restrict__ bias_grad,
                  int64_t K, int M, int N, int64_t k_chunk) {
    __shared__ float As[BK][LDP];
    __shared__ float Bs[BK][LDP];

    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int64_t k_beg = (int64_t)blockIdx.z * k_chunk;
    const int64_t k_end = min(K, k_beg + k_chunk);

    const int tid = threadIdx.x;
    const int wave = tid / 64;
    const int lane = tid % 64;
    // wave (0..3) -> 2x2 position in the 64x64 macro-tile; each wave's
    // quadrant is a 2x2 grid of 16x16 MFMA tiles
    const int wm = (wave % 2) * 32;  // wave's m offset in the block tile
    const int wn = (wave / 2) * 32;  // wave's n offset
    const int fcol = lane % 16;      // fragment column (m or n)
    const int fk = lane / 16;        // fragment k row (0..3)

    f32x4 acc00 = {0, 0, 0, 0}, acc01 = {0, 0, 0, 0};
    f32x4 acc10 = {0, 0, 0, 0}, acc11 = {0, 0, 0, 0};

    // bias fold: thread t accumulates column (t%BM) over its k-subrows
    const bool do_bias = (bias_grad != nullptr) && (blockIdx.y == 0);
    const int bm = tid % BM;       // bias column
    const int bk0 = tid / BM;      // first k-subrow (stride WAVES*64/BM)
    float bsum = 0.f;

    for (int64_t k0 = k_beg; k0 < k_end; k0 += BK) {
        // stage A[k0..k0+BK) x [m0..m0+BM) and the B slice; 1024 floats
        // each, 4 per thread, coalesced over the feature dim
        for (int l = tid; l < BK * BM; l += WAVES * 64) {
            const int kk = l / BM, mm = l % BM;
            const int64_t k = k0 + kk;
            As[kk][mm] = (k < k_end && m0 + mm < M) ? A[k * M + m0 + mm]
                                                    : 0.f;
            Bs[kk][mm] = (k < k_end && n0 + mm < N) ? B[k * N + n0 + mm]
                                                    : 0.f;
        }
        __syncthreads();
#pragma unroll
        for (int kk = 0; kk < BK; kk += 4) {
            // fragments: lane l reads (m|n = base + l%16, k = kk + l/16)
            const float a0 = As[kk + fk][wm + fcol];
            const float a1 = As[kk + fk][wm + 16 + fcol];
            const float b0 = Bs[kk + fk][wn + fcol];
            const float b1 = Bs[kk + fk][wn + 16 + fcol];
            acc00 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b0, acc00,
                                                         0, 0, 0);
            acc01 = __builtin_amdgcn_mfma_f32_16x16x4f32(a0, b1, acc01,
                                                         0, 0, 0);
            acc10 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b0, acc10,
                                                         0, 0, 0);
            acc11 = __builtin_amdgcn_mfma_f32_16x16x4f32(a1, b1, acc11,
                                                         0, 0, 0);
        }
        if (do_bias) {
#pragma unroll
            for (int kk = bk0; kk < BK; kk += WAVES * 64 / BM)
                bsum += As[kk][bm];
        }
        __syncthreads();
    }

    // D mapping: lane l, reg r -> row (l/16)*4 + r, col l%16
    const int n_out0 = n0 + wn + fcol;
    const int n_out1 = n0 + wn + 16 + fcol;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int mr0 = m0 + wm + fk * 4 + r;
        const int mr1 = mr0 + 16;
        if (mr0 < M) {
            if (n_out0 < N)
                atomicAdd(&C[(int64_t)mr0 * N + n_out0], acc00[r]);
            if (n_out1 < N)
                atomicAdd(&C[(int64_t)mr0 * N + n_out1], acc01[r]);
        }
        if (mr1 < M) {
            if (n_out0 < N)
                atomicAdd(&C[(int64_t)mr1 * N + n_out0], acc10[r]);
            if (n_out1 < N)
                atomicAdd(&C[(int64_t)mr1 * N + n_out1], acc11[r]);
        }
    }
    if (do_bias && m0 + bm < M) atomicAdd(&bias_grad[m0 + bm], bsum);
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
    dim3 block(WAVES * 64);
    wgrad_mfma_kernel<<<grid, block, 0, s>>>(A, B, C, bias_grad, K, M, N,
                                             k_chunk);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
