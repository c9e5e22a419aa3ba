// Tall-skinny weight-gradient GEMM for gfx950: C[M×N] = A^T @ B where
// A is [K×M], B is [K×N], K is huge (a GNN frontier, 10^4..10^6 rows) and
// M,N are layer widths (<= a few hundred).
//
// rocBLAS/Tensile handles this shape with stream-K 32x32 macro-tiles ~60x
// off the HBM3E roof; a VALU split-K ran it at ~14 TF.  This is a split-K
// GEMM on the f32 matrix cores (v_mfma_f32_32x32x2_f32, exact f32
// numerics, 157 TF chip peak; measured 41-64 TF on the bench shapes =
// 2.8-5.7x rocBLAS), shaped by four observations:
//
//  - The A^T shape is a gift on CDNA4: the MFMA A-fragment wants lane l
//    to hold A_mfma[col = l%32][k = l/32], which for our K-major A is a
//    fully-coalesced read — both operands stage through LDS with no
//    transpose anywhere, purely for cross-wave reuse.  The 32x32x2 form
//    is used over 16x16x4 because its 32-lane fragment groups read ONE
//    LDS row each -> inherently bank-conflict-free (PMC showed 2
//    conflicts/MFMA on the 16x16x4 4-k-group layout), and it halves the
//    instruction-issue load.  Staging is float4 and double-buffered both
//    through registers (next K-slice's global loads issue before the
//    MFMA loop) and through ping-pong LDS (one barrier per stage).
//  - C is tiny (a few hundred KB) while the split-K grid is huge, so
//    atomic partials would pile ~300 ops on every C word.  Instead each
//    block stores its 64x64 partial to a per-chunk workspace slice
//    (plain coalesced stores) and a fixed-order reduce kernel writes C —
//    which also makes wgrad DETERMINISTIC, unlike Tensile's GSU path.
//  - The grid is 1-D and XCD-aware: blocks that re-read the same A/B
//    slice are mapped to the same XCD (hardware dispatches block i to
//    XCD i%8), so the tn x A + tm x B read amplification is absorbed by
//    that XCD's L2 instead of HBM (~9% measured).
//  - Block = 4 waves = a 64x64 C macro-tile; each wave owns one 32x32
//    MFMA tile (a single f32x16 accumulator chain reaches the 64-cycle
//    issue rate per the ISA's dependent-latency table).
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

typedef float f32x16 __attribute__((ext_vector_type(16)));

__global__ void __launch_bounds__(NTHREADS)
wgrad_mfma_kernel(const float* __restrict__ A, const float* __restrict__ B,
                  float* __restrict__ ws, int64_t K, int M, int N,
                  int64_t k_chunk, int do_bias, int tm_, int tn_,
                  int64_t nchunks_) {
    // ping-pong LDS: one barrier per stage (write next buffer while the
    // current one is being read)
    __shared__ float As[2][BK][LDP];
    __shared__ float Bs[2][BK][LDP];

    // XCD-aware tile mapping (1-D grid): the hardware dispatches block i
    // to XCD i%8, so make each XCD own whole (mi, ni) groups of one K
    // chunk — the tn blocks re-reading an A slice and the tm blocks
    // re-reading a B slice then share that XCD's L2 instead of pulling
    // from HBM again.  nchunks is padded to a multiple of 8 by the plan.
    const int n_xcd = 8;
    const int xcd = blockIdx.x % n_xcd;
    const int slot = blockIdx.x / n_xcd;
    const int tiles = tm_ * tn_;
    const int rem = slot % tiles;
    const int64_t z = xcd + (int64_t)n_xcd * (slot / tiles);
    if (z >= nchunks_) return;  // padding block
    const int m0 = (rem / tn_) * BM;
    const int n0 = (rem % tn_) * BN;
    const int64_t k_beg = z * k_chunk;
    const int64_t k_end = min(K, k_beg + k_chunk);

    const int tid = threadIdx.x;
    const int wave = tid / 64;
    const int lane = tid % 64;
    const int wm = (wave % 2) * 32;  // wave's m offset in the block tile
    const int wn = (wave / 2) * 32;  // wave's n offset
    // 32x32x2 fragment mapping: lane l holds op[col = l%32][k = l/32].
    // Within a 32-lane bank-conflict group all lanes read ONE LDS row at
    // consecutive addresses -> inherently conflict-free (the 16x16x4
    // 4-k-group layout 2-way-conflicted: PMC showed 2 conflicts/MFMA).
    const int fcol = lane % 32;
    const int fk = lane / 32;

    f32x16 acc = {};

    // bias fold: thread t accumulates column (t%BM) over its k-subrows
    const bool bias_block = do_bias && (n0 == 0);
    const int bm = tid % BM;       // bias column
    const int bk0 = tid / BM;      // first k-subrow (stride NTHREADS/BM)
    float bsum = 0.f;

    float4 ra[F4_PER_SLICE], rb[F4_PER_SLICE];
    load_slice(A, k_beg, k_end, M, m0, tid, ra);
    load_slice(B, k_beg, k_end, N, n0, tid, rb);
    store_slice(As[0], tid, ra);
    store_slice(Bs[0], tid, rb);
    __syncthreads();

    int buf = 0;
    for (int64_t k0 = k_beg; k0 < k_end; k0 += BK) {
        const bool more = k0 + BK < k_end;
        // issue the next slice's global loads before the MFMA loop
        if (more) {
            load_slice(A, k0 + BK, k_end, M, m0, tid, ra);
            load_slice(B, k0 + BK, k_end, N, n0, tid, rb);
        }
#pragma unroll
        for (int kk = 0; kk < BK; kk += 2) {
            const float a = As[buf][kk + fk][wm + fcol];
            const float b = Bs[buf][kk + fk][wn + fcol];
            acc = __builtin_amdgcn_mfma_f32_32x32x2f32(a, b, acc, 0, 0, 0);
        }
        if (bias_block) {
#pragma unroll
            for (int kk = bk0; kk < BK; kk += NTHREADS / BM)
                bsum += As[buf][kk][bm];
        }
        if (more) {
            store_slice(As[buf ^ 1], tid, ra);
            store_slice(Bs[buf ^ 1], tid, rb);
        }
        __syncthreads();
        buf ^= 1;
    }

    // workspace slice for this K-chunk: padded C tile grid + bias column.
    // Plain coalesced stores; a fixed-order reduce kernel folds chunks.
    // 32x32 D mapping: col = lane%32, row = (r&3) + 8*(r>>2) + 4*(lane/32).
    const int Np = tn_ * BN;
    const int64_t row_pitch = Np + 1;  // +1: bias column at Np
    float* slice = ws + z * ((int64_t)tm_ * BM) * row_pitch;
    const int n_out = n0 + wn + fcol;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
        const int m_out = m0 + wm + (r & 3) + 8 * (r >> 2) + 4 * fk;
        slice[(int64_t)m_out * row_pitch + n_out] = acc[r];
    }
    if (bias_block) {
        // 4 threads share a bias column: reduce through LDS (As is free)
        (&As[0][0][0])[tid] = bsum;
        __syncthreads();
        if (tid < BM) {
            float s = 0.f;
#pragma unroll
            for (int g = 0; g < NTHREADS / BM; ++g)
                s += (&As[0][0][0])[tid + g * BM];
            slice[(int64_t)(m0 + tid) * row_pitch + Np] = s;
        }
    }
}

// C[m][n] = sum_z ws[z][m][n]; bias_grad[m] = sum_z ws[z][m][Np].
// Fixed z-to-group assignment and fixed group order -> deterministic.
// Parallelized over (output, z-group): a one-thread-per-output reduce
// chains nchunks ~600ns strided loads and ran as long as the GEMM
// itself (83 us measured); 16 z-groups cut the chain 16x and an LDS
// combine folds the groups.
constexpr int ROUT = 64;  // outputs per block
constexpr int ZG = 16;    // parallel z groups

__global__ void __launch_bounds__(ROUT * ZG)
wgrad_reduce_kernel(const float* __restrict__ ws, float* __restrict__ C,
                    float* __restrict__ bias_grad, int M, int N, int Mp,
                    int Np, int64_t nchunks) {
    __shared__ float part[ZG][ROUT];
    const int64_t row_pitch = Np + 1;
    const int64_t slice_sz = (int64_t)Mp * row_pitch;
    const int64_t total = (int64_t)M * N + (bias_grad ? M : 0);
    const int j = threadIdx.x % ROUT;
    const int g = threadIdx.x / ROUT;
    const int64_t i = (int64_t)blockIdx.x * ROUT + j;
    float s = 0.f;
    int64_t off = 0;
    if (i < total) {
        off = i < (int64_t)M * N
                  ? (i / N) * row_pitch + (i % N)
                  : (i - (int64_t)M * N) * row_pitch + Np;
        for (int64_t z = g; z < nchunks; z += ZG)
            s += ws[z * slice_sz + off];
    }
    part[g][j] = s;
    __syncthreads();
    if (g == 0 && i < total) {
        float acc = 0.f;
#pragma unroll
        for (int gg = 0; gg < ZG; ++gg) acc += part[gg][j];
        if (i < (int64_t)M * N) {
            C[i] = acc;
        } else {
            bias_grad[i - (int64_t)M * N] = acc;
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
    // 1-D grid over XCD-padded chunks x tiles (see kernel mapping)
    int64_t z_pad = (plan.nchunks + 7) / 8 * 8;
    int64_t nblocks = z_pad * plan.tm * plan.tn;
    wgrad_mfma_kernel<<<(unsigned)nblocks, NTHREADS, 0, s>>>(
        A, B, ws, K, M, N, plan.k_chunk, bias_grad != nullptr, plan.tm,
        plan.tn, plan.nchunks);
    QK_CHECK_HIP(hipGetLastError());
    int64_t total = (int64_t)M * N + (bias_grad ? M : 0);
    int rblocks = (int)((total + ROUT - 1) / ROUT);
    wgrad_reduce_kernel<<<rblocks, ROUT * ZG, 0, s>>>(
        ws, C, bias_grad, M, N, plan.tm * BM, plan.tn * BN, plan.nchunks);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
