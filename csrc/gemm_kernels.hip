// Tall-M GEMM for gfx950: C[M×N] = A[M×K] @ B[K×N] (+ bias[N]) where M is
// a GNN frontier (10^5..10^6 rows) and K, N are layer widths (<= 1024).
//
// This is the OTHER hot GEMM family of the model layer (wgrad_kernels.hip
// covers A^T@B): big-frontier linear forwards and data-grads.  Measured
// status (benchmarks/bench_gemm.py): WINS on narrow-output shapes
// (N <= 128: 1.1-1.25x rocBLAS — QLinear routes those here) and loses
// ~15-45% on wide ones (rocBLAS's 256x128 macro-tiles amortize the
// barrier convoy better at these shallow K; QLinear keeps rocBLAS
// there).  PMC: 56% wave-parked / 33% issue-stall — the 64x64 tile's
// 4-wave barrier per 16-MFMA stage is the pacing structure.
// MI355X-first design:
//  - v_mfma_f32_32x32x2_f32 tiles (exact f32), one 32x32 tile per wave,
//    2x2 waves = a 64x64 C macro-tile per block.  No split-K: M/64 tiles
//    give tens of thousands of blocks by themselves.
//  - B is staged K-major (the caller passes W^T for forward — a 100 KB
//    host-side transpose — and W itself for data-grad, so both read
//    coalesced), A staged through LDS with a +1 row pad (fragment lanes
//    read a k-column of 32 consecutive rows: odd pitch -> conflict-free).
//  - Double-buffered through registers AND ping-pong LDS, same skeleton
//    as wgrad.
#include "qk_common.h"

namespace qk {

namespace {

typedef float f32x16 __attribute__((ext_vector_type(16)));

constexpr int BM = 64;
constexpr int BN = 64;
constexpr int BK = 32;
constexpr int AP = BK + 1;   // As row pitch (odd -> fragment reads hit
                             // distinct banks across the 32-row group)
constexpr int BP = BN + 4;   // Bs row pitch
constexpr int NT = 256;      // 4 waves

// A slice: rows [m0, m0+ROWS) x k [k0, k0+BK); thread t loads NF4 float4
// (ROWS*BK/4/NT) at flat float4 index f = t + j*NT -> (row = f/8, kk4 =
// f%8) covering k = kk4*4..+3.
template <int NF4>
__device__ __forceinline__ void load_a(const float* __restrict__ A,
                                       int64_t M, int K, int64_t m0, int k0,
                                       int tid, float4 r[NF4]) {
#pragma unroll
    for (int j = 0; j < NF4; ++j) {
        const int f = tid + j * NT;
        const int64_t row = m0 + f / 8;
        const int k = k0 + (f % 8) * 4;
        if (row < M && k + 3 < K && (K % 4 == 0)) {
            r[j] = *reinterpret_cast<const float4*>(&A[row * K + k]);
        } else {
            float v[4];
#pragma unroll
            for (int q = 0; q < 4; ++q)
                v[q] = (row < M && k + q < K) ? A[row * K + k + q] : 0.f;
            r[j] = make_float4(v[0], v[1], v[2], v[3]);
        }
    }
}

template <int NF4>
__device__ __forceinline__ void store_a(float (*As)[AP], int tid,
                                        const float4 r[NF4]) {
#pragma unroll
    for (int j = 0; j < NF4; ++j) {
        const int f = tid + j * NT;
        const int row = f / 8, k = (f % 8) * 4;
        As[row][k + 0] = r[j].x;
        As[row][k + 1] = r[j].y;
        As[row][k + 2] = r[j].z;
        As[row][k + 3] = r[j].w;
    }
}

// B slice: k [k0, k0+BK) x n [n0, n0+COLS); thread t loads NF4 float4 at
// flat index f = t + j*NT -> (kk = f/(COLS/4), n4 = f%(COLS/4)).
template <int NF4, int COLS4>
__device__ __forceinline__ void load_b(const float* __restrict__ B, int K,
                                       int N, int k0, int n0, int tid,
                                       float4 r[NF4]) {
#pragma unroll
    for (int j = 0; j < NF4; ++j) {
        const int f = tid + j * NT;
        const int kk = k0 + f / COLS4;
        const int n = n0 + (f % COLS4) * 4;
        if (kk < K && n + 3 < N && (N % 4 == 0)) {
            r[j] = *reinterpret_cast<const float4*>(&B[(int64_t)kk * N + n]);
        } else {
            float v[4];
#pragma unroll
            for (int q = 0; q < 4; ++q)
                v[q] = (kk < K && n + q < N) ? B[(int64_t)kk * N + n + q]
                                             : 0.f;
            r[j] = make_float4(v[0], v[1], v[2], v[3]);
        }
    }
}

template <int NF4, int COLS4, int PITCH>
__device__ __forceinline__ void store_b(float (*Bs)[PITCH], int tid,
                                        const float4 r[NF4]) {
#pragma unroll
    for (int j = 0; j < NF4; ++j) {
        const int f = tid + j * NT;
        *reinterpret_cast<float4*>(&Bs[f / COLS4][(f % COLS4) * 4]) = r[j];
    }
}

__global__ void __launch_bounds__(NT)
tall_gemm_kernel(const float* __restrict__ A, const float* __restrict__ B,
                 const float* __restrict__ bias, float* __restrict__ C,
                 int64_t M, int K, int N, int tn_) {
    __shared__ float As[2][BM][AP];
    __shared__ float Bs[2][BK][BP];

    // 1-D XCD-aware grid (hardware dispatches block i to XCD i%8): all
    // tn n-tiles of one m-chunk land on the SAME XCD, so the A slice is
    // pulled from HBM once and re-read from that XCD's L2 (PMC showed A
    // re-read x tn was the dominant traffic).  Each block walks several
    // m-tiles with one CONTINUOUS ping-pong pipeline: the next tile's
    // first slices prefetch during the current tile's last k-stage.
    const int n_xcd = 8;
    const int xcd = blockIdx.x % n_xcd;
    const int slot = blockIdx.x / n_xcd;
    const int nt = slot % tn_;
    const int walker = slot / tn_;          // m-walker within this xcd
    const int walkers = (int)(gridDim.x / (n_xcd * tn_));
    const int64_t m_stride = (int64_t)n_xcd * walkers * BM;
    const int n0 = nt * BN;
    const int tid = threadIdx.x;
    const int wave = tid / 64;
    const int lane = tid % 64;
    const int wm = (wave % 2) * 32;
    const int wn = (wave / 2) * 32;
    const int fcol = lane % 32;
    const int fk = lane / 32;

    const int n_out = n0 + wn + fcol;
    const float badd = (bias && n_out < N) ? bias[n_out] : 0.f;

    float4 ra[2], rb[2];
    int64_t m0 = (int64_t)(xcd + n_xcd * walker) * BM;
    if (m0 >= M) return;
    load_a<2>(A, M, K, m0, 0, tid, ra);
    load_b<2, 16>(B, K, N, 0, n0, tid, rb);
    store_a<2>(As[0], tid, ra);
    store_b<2, 16, BP>(Bs[0], tid, rb);
    __syncthreads();

    int buf = 0;
    for (; m0 < M; m0 += m_stride) {
        f32x16 acc = {};
        for (int k0 = 0; k0 < K; k0 += BK) {
            const bool more_k = k0 + BK < K;
            const bool more_m = m0 + m_stride < M;
            if (more_k) {
                load_a<2>(A, M, K, m0, k0 + BK, tid, ra);
                load_b<2, 16>(B, K, N, k0 + BK, n0, tid, rb);
            } else if (more_m) {
                load_a<2>(A, M, K, m0 + m_stride, 0, tid, ra);
                load_b<2, 16>(B, K, N, 0, n0, tid, rb);
            }
#pragma unroll
            for (int kk = 0; kk < BK; kk += 2) {
                // A fragment: lane (row = fcol, k = fk); the odd AP
                // pitch spreads the 32-row column read across banks
                const float a = As[buf][wm + fcol][kk + fk];
                const float b = Bs[buf][kk + fk][wn + fcol];
                acc = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc,
                                                           0, 0, 0);
            }
            if (more_k || more_m) {
                store_a<2>(As[buf ^ 1], tid, ra);
                store_b<2, 16, BP>(Bs[buf ^ 1], tid, rb);
            }
            __syncthreads();
            buf ^= 1;
        }
        // D mapping (same orientation as the wgrad kernel: the first
        // operand holds the LEFT matrix's rows): row m_out =
        // (r&3)+8*(r>>2)+4*fk, col n_out = fcol — coalesced stores.
        if (n_out < N) {
#pragma unroll
            for (int r = 0; r < 16; ++r) {
                const int64_t m_out =
                    m0 + wm + (r & 3) + 8 * (r >> 2) + 4 * fk;
                if (m_out < M) C[m_out * N + n_out] = acc[r] + badd;
            }
        }
    }
}


// 128x128 macro-tile variant for huge M: each wave owns a 64x64 quadrant
// as 2x2 MFMA tiles (4 accumulators), 64 MFMAs per barrier — 4x the
// compute per load round of the 64x64 kernel, which was convoy-bound at
// these shallow K (PMC: 56% parked / 33% issue-stall).
constexpr int BM2 = 128;
constexpr int BN2 = 128;
constexpr int BP2 = BN2 + 4;

__global__ void __launch_bounds__(NT)
tall_gemm128_kernel(const float* __restrict__ A, const float* __restrict__ B,
                    const float* __restrict__ bias, float* __restrict__ C,
                    int64_t M, int K, int N, int tn_) {
    __shared__ float As[2][BM2][AP];
    __shared__ float Bs[2][BK][BP2];

    const int n_xcd = 8;
    const int xcd = blockIdx.x % n_xcd;
    const int slot = blockIdx.x / n_xcd;
    const int nt = slot % tn_;
    const int walker = slot / tn_;
    const int walkers = (int)(gridDim.x / (n_xcd * tn_));
    const int64_t m_stride = (int64_t)n_xcd * walkers * BM2;
    const int n0 = nt * BN2;
    const int tid = threadIdx.x;
    const int wave = tid / 64;
    const int lane = tid % 64;
    const int wm = (wave % 2) * 64;
    const int wn = (wave / 2) * 64;
    const int fcol = lane % 32;
    const int fk = lane / 32;

    const int n_out0 = n0 + wn + fcol;
    const int n_out1 = n_out0 + 32;
    const float badd0 = (bias && n_out0 < N) ? bias[n_out0] : 0.f;
    const float badd1 = (bias && n_out1 < N) ? bias[n_out1] : 0.f;

    float4 ra[4], rb[4];
    int64_t m0 = (int64_t)(xcd + n_xcd * walker) * BM2;
    if (m0 >= M) return;
    load_a<4>(A, M, K, m0, 0, tid, ra);
    load_b<4, 32>(B, K, N, 0, n0, tid, rb);
    store_a<4>(As[0], tid, ra);
    store_b<4, 32, BP2>(Bs[0], tid, rb);
    __syncthreads();

    int buf = 0;
    for (; m0 < M; m0 += m_stride) {
        f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};
        for (int k0 = 0; k0 < K; k0 += BK) {
            const bool more_k = k0 + BK < K;
            const bool more_m = m0 + m_stride < M;
            if (more_k) {
                load_a<4>(A, M, K, m0, k0 + BK, tid, ra);
                load_b<4, 32>(B, K, N, k0 + BK, n0, tid, rb);
            } else if (more_m) {
                load_a<4>(A, M, K, m0 + m_stride, 0, tid, ra);
                load_b<4, 32>(B, K, N, 0, n0, tid, rb);
            }
#pragma unroll
            for (int kk = 0; kk < BK; kk += 2) {
                const float a0 = As[buf][wm + fcol][kk + fk];
                const float a1 = As[buf][wm + 32 + fcol][kk + fk];
                const float b0 = Bs[buf][kk + fk][wn + fcol];
                const float b1 = Bs[buf][kk + fk][wn + 32 + fcol];
                acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00,
                                                             0, 0, 0);
                acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01,
                                                             0, 0, 0);
                acc10 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc10,
                                                             0, 0, 0);
                acc11 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc11,
                                                             0, 0, 0);
            }
            if (more_k || more_m) {
                store_a<4>(As[buf ^ 1], tid, ra);
                store_b<4, 32, BP2>(Bs[buf ^ 1], tid, rb);
            }
            __syncthreads();
            buf ^= 1;
        }
#pragma unroll
        for (int r = 0; r < 16; ++r) {
            const int dm = (r & 3) + 8 * (r >> 2) + 4 * fk;
            const int64_t mr0 = m0 + wm + dm;
            const int64_t mr1 = mr0 + 32;
            if (mr0 < M) {
                if (n_out0 < N) C[mr0 * N + n_out0] = acc00[r] + badd0;
                if (n_out1 < N) C[mr0 * N + n_out1] = acc01[r] + badd1;
            }
            if (mr1 < M) {
                if (n_out0 < N) C[mr1 * N + n_out0] = acc10[r] + badd0;
                if (n_out1 < N) C[mr1 * N + n_out1] = acc11[r] + badd1;
            }
        }
    }
}

}  // namespace

void launch_tall_gemm(hipStream_t s, const float* A, const float* B,
                      const float* bias, float* C, int64_t M, int K, int N) {
    if (M == 0 || K == 0 || N == 0) return;
    // 128x128 tiles measured SLOWER on the huge-frontier shapes (1461
    // vs 1111 us at 1.06Mx100x256): LDS doubles -> 2 blocks/CU and the
    // lost concurrency outweighs 4x compute per barrier.  Kernel kept
    // for reference/perf archaeology; routing stays on 64x64.
    const bool big = false;
    const int bm = big ? BM2 : BM, bn = big ? BN2 : BN;
    int64_t tm = (M + bm - 1) / bm;
    int tn = (N + bn - 1) / bn;
    // walkers per (xcd, n-tile): enough blocks to fill the chip several
    // times over, few enough that each walks multiple m-tiles
    int64_t walkers = 4096 / (8 * tn);
    int64_t need = (tm + 7) / 8;  // m-tiles per xcd
    if (walkers > need) walkers = need;
    if (walkers < 1) walkers = 1;
    unsigned gx = (unsigned)(8 * tn * walkers);
    if (big)
        tall_gemm128_kernel<<<gx, NT, 0, s>>>(A, B, bias, C, M, K, N, tn);
    else
        tall_gemm_kernel<<<gx, NT, 0, s>>>(A, B, bias, C, M, K, N, tn);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
