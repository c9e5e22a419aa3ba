// Tall-skinny weight-gradient GEMM for gfx950: C[M×N] = A^T @ B where
// A is [K×M], B is [K×N], K is huge (a GNN frontier, 10^4..10^6 rows) and
// M,N are layer widths (<= a few hundred).
//
// rocBLAS/Tensile handles this shape with stream-K 32x32 macro-tiles ~60x
// off the HBM3E roof; a VALU split-K ran it at ~14 TF.  This is a split-K
// GEMM on the f32 matrix cores (v_mfma_f32_16x16x4_f32, exact f32
// numerics, 157 TF chip peak), shaped by three observations:
//
//  - The A^T shape is a gift on CDNA4: the MFMA A-fragment for a 16x16x4
//    tile wants lane l to hold A_mfma[m = l%16][k = l/16], which for our
//    K-major A is A[(k0 + l/16)*M + m0 + l%16] — 16 consecutive lanes
//    read 16 consecutive floats.  Both operands stage through LDS
//    (row-padded +4 floats: the 4 k-group lanes of a fragment read rows
//    64 floats apart, which would 2-way-conflict unpadded) purely for
//    cross-wave reuse; no transpose anywhere.  Staging is float4 and
//    double-buffered through registers: the next K-slice's global loads
//    issue before the MFMA loop of the current slice.
//  - C is tiny (a few hundred KB) while the split-K grid is huge, so
//    atomic partials would pile ~300 ops on every C word.  Instead each
//    block stores its 64x64 partial to a per-chunk workspace slice
//    (plain coalesced stores) and a fixed-order reduce kernel writes C —
//    which also makes wgrad DETERMINISTIC, unlike Tensile's GSU path.
//  - Block = 4 waves = a 64x64 C macro-tile; each wave owns a 2x2 grid
//    of 16x16 MFMA tiles (4 independent f32x4 accumulators hide the
//    40-cycle dependent-accumulator latency at the 32-cycle issue rate).
//
// The bias gradient (column sums of A) folds into the same pass: blocks
// in the first N-tile column accumulate their LDS A-slices into an extra
// workspace column, reduced with C.
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
constexpr int BK = 32;        // K rows staged per iteration
constexpr int LDP = BM + 4;   // LDS row pitch (pad: kill k-group conflicts)
constexpr int WAVES = 4;      // 2x2 wave grid over the 64x64 tile
constexpr int NTHREADS = WAVES * 64;
constexpr int F4_PER_SLICE = BK * BM / 4 / NTHREADS;  // float4s per thread

// One K-slice of one operand, registers first (double buffering).
// Thread t, part j: float4 index f = t + j*NTHREADS covers
// (kk = f/16, col = base + (f%16)*4).
__device__ __forceinline__ void load_slice(const float* __restrict__ P,
                                           int64_t k0, int64_t k_end,
                                           int width, int base, int tid,
                                           float4 r[F4_PER_SLICE]) {
#pragma unroll
    for (int j = 0; j < F4_PER_SLICE; ++j) {
        const int f = tid + j * NTHREADS;
        const int kk = f / 16;
        const int col = base + (f % 16) * 4;
        const int64_t k = k0 + kk;
        if (k < k_end && col + 3 < width && (width % 4 == 0) &&
            (base % 4 == 0)) {
            r[j] = *reinterpret_cast<const float4*>(&P[k * width + col]);
        } else {
            float v[4];
#pragma unroll
            for (int q = 0; q < 4; ++q)
                v[q] = (k < k_end && col + q < width) ? P[k * width + col + q]
                                                      : 0.f;
            r[j] = make_float4(v[0], v[1], v[2], v[3]);
        }
    }
}

__device__ __forceinline__ void store_slice(float (*S)[LDP], int tid,
                                            const float4 r[F4_PER_SLICE]) {
#pragma unroll
    for (int j = 0; j < F4_PER_SLICE; ++j) {
        const int f = tid + j * NTHREADS;
        *reinterpret_cast<float4*>(&S[f / 16][(f % 16) * 4]) = r[j];
    }
}

__global__ void __launch_bounds__(NTHREADS)
wgrad_mfma_kernel(const float* __restrict__ A, const float* __restrict__ B,
                  float* __restrict__ ws, int64_t K, int M, int N,
                  int64_t k_chunk, int do_bias) {
    __shared__ float As[BK][LDP];
    __shared__ float Bs[BK][LDP];

    const int m0 = blockIdx.x * BM;
    const int n0 = blockIdx.y * BN;
    const int64_t k_beg = (int64_t)blockIdx.z * k_chunk;
    const int64_t k_end = min(K, k_beg + k_chunk);

    const int tid = threadIdx.x;
    const int wave = tid / 64;
    const int lane = tid % 64;
    const int wm = (wave % 2) * 32;  // wave's m offset in the block tile
    const int wn = (wave / 2) * 32;  // wave's n offset
    const int fcol = lane % 16;      // fragment column (m or n)
    const int fk = lane / 16;        // fragment k row (0..3)

    f32x4 acc00 = {0, 0, 0, 0}, acc01 = {0, 0, 0, 0};
    f32x4 acc10 = {0, 0, 0, 0}, acc11 = {0, 0, 0, 0};

    // bias fold: thread t accumulates column (t%BM) over its k-subrows
    const bool bias_block = do_bias && (blockIdx.y == 0);
    const int bm = tid % BM;       // bias column
    const int bk0 = tid / BM;      // first k-subrow (stride NTHREADS/BM)
    float bsum = 0.f;

    float4 ra[F4_PER_SLICE], rb[F4_PER_SLICE];
    load_slice(A, k_beg, k_end, M, m0, tid, ra);
    load_slice(B, k_beg, k_end, N, n0, tid, rb);

    for (int64_t k0 = k_beg; k0 < k_end; k0 += BK) {
        store_slice(As, tid, ra);
        store_slice(Bs, tid, rb);
        __syncthreads();
        // issue the next slice's global loads before the MFMA loop
        if (k0 + BK < k_end) {
            load_slice(A, k0 + BK, k_end, M, m0, tid, ra);
            load_slice(B, k0 + BK, k_end, N, n0, tid, rb);
        }
#pragma unroll
        for (int kk = 0; kk < BK; kk += 4) {
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
        if (bias_block) {
#pragma unroll
            for (int kk = bk0; kk < BK; kk += NTHREADS / BM)
                bsum += As[kk][bm];
        }
        __syncthreads();
    }

    // workspace slice for this K-chunk: padded C tile grid + bias column.
    // Plain coalesced stores; a fixed-order reduce kernel folds chunks.
    const int Np = (int)gridDim.y * BN;
    const int64_t row_pitch = Np + 1;  // +1: bias column at Np
    float* slice = ws + (int64_t)blockIdx.z * ((int64_t)gridDim.x * BM) *
                            row_pitch;
    const int n_out0 = n0 + wn + fcol;
    const int n_out1 = n0 + wn + 16 + fcol;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int mr0 = m0 + wm + fk * 4 + r;
        const int mr1 = mr0 + 16;
        slice[(int64_t)mr0 * row_pitch + n_out0] = acc00[r];
        slice[(int64_t)mr0 * row_pitch + n_out1] = acc01[r];
        slice[(int64_t)mr1 * row_pitch + n_out0] = acc10[r];
        slice[(int64_t)mr1 * row_pitch + n_out1] = acc11[r];
    }
    if (bias_block) {
        // 4 threads share a bias column: reduce through LDS (As is free)
        (&As[0][0])[tid] = bsum;
        __syncthreads();
        if (tid < BM) {
            float s = 0.f;
#pragma unroll
            for (int g = 0; g < NTHREADS / BM; ++g)
                s += (&As[0][0])[tid + g * BM];
            slice[(int64_t)(m0 + tid) * row_pitch + Np] = s;
        }
    }
}

// C[m][n] = sum_z ws[z][m][n]; bias_grad[m] = sum_z ws[z][m][Np].
// Fixed summation order -> deterministic result.
__global__ void __launch_bounds__(256)
wgrad_reduce_kernel(const float* __restrict__ ws, float* __restrict__ C,
                    float* __restrict__ bias_grad, int M, int N, int Mp,
                    int Np, int64_t nchunks) {
    const int64_t row_pitch = Np + 1;
    const int64_t slice_sz = (int64_t)Mp * row_pitch;
    const int64_t total = (int64_t)M * N + (bias_grad ? M : 0);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < total; i += (int64_t)gridDim.x * blockDim.x) {
        int64_t off;
        if (i < (int64_t)M * N) {
            off = (i / N) * row_pitch + (i % N);
        } else {
            off = (i - (int64_t)M * N) * row_pitch + Np;
        }
        float s = 0.f;
        for (int64_t z = 0; z < nchunks; ++z) s += ws[z * slice_sz + off];
        if (i < (int64_t)M * N) {
            C[i] = s;
        } else {
            bias_grad[i - (int64_t)M * N] = s;
        }
    }
}

}  // namespace

WgradPlan wgrad_plan(int64_t K, int M, int N) {
    WgradPlan p{};
    p.tm = (M + BM - 1) / BM;
    p.tn = (N + BN - 1) / BN;
    // size the K split so the grid comfortably covers 256 CUs x 8 XCDs
    int64_t want_blocks = 2048;
    int64_t nchunks = want_blocks / (p.tm * p.tn);
    int64_t min_chunk = 2 * BK;
    int64_t max_chunks = (K + min_chunk - 1) / min_chunk;
    if (nchunks > max_chunks) nchunks = max_chunks;
    if (nchunks < 1) nchunks = 1;
    p.k_chunk = ((K + nchunks - 1) / nchunks + BK - 1) / BK * BK;
    p.nchunks = (K + p.k_chunk - 1) / p.k_chunk;
    p.ws_floats =
        p.nchunks * (int64_t)(p.tm * BM) * ((int64_t)p.tn * BN + 1);
    return p;
}

void launch_wgrad(hipStream_t s, const float* A, const float* B, float* C,
                  float* bias_grad, int64_t K, int M, int N,
                  const WgradPlan& plan, float* ws) {
    if (K == 0 || M == 0 || N == 0) return;
    dim3 grid(plan.tm, plan.tn, (unsigned)plan.nchunks);
    wgrad_mfma_kernel<<<grid, NTHREADS, 0, s>>>(A, B, ws, K, M, N,
                                                plan.k_chunk,
                                                bias_grad != nullptr);
    QK_CHECK_HIP(hipGetLastError());
    int64_t total = (int64_t)M * N + (bias_grad ? M : 0);
    int rblocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
    wgrad_reduce_kernel<<<rblocks, 256, 0, s>>>(
        ws, C, bias_grad, M, N, plan.tm * BM, plan.tn * BN, plan.nchunks);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
