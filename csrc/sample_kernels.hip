// Wave64 CSR neighbor-sampling kernels for gfx950 (MI355X).
//
// Re-designed from the behavior of the reference's CSRRowWiseSampleKernel /
// cal_next (torch-quiver srcs/cpp/include/quiver/cuda_random.cu.hpp:7-104):
//  - CDNA4 wavefronts are 64 lanes; we assign each seed row to a 16-lane
//    subgroup (4 rows per wave) so small fanouts (k=5..25) don't waste lanes.
//    Subgroup lanes are wavefront-synchronous: no barriers needed.
//  - Reservoir slots live in LDS (per-row k ints) instead of global memory;
//    the sequential-replacement order is emulated with LDS atomicMax, the
//    same algorithm family as the reference/DGL, but on-chip.
//  - RNG is a stateless counter-based hash (SplitMix64) keyed by
//    (rng_seed, row, j) — no curand/hiprand state arrays, no init kernel.
//  - Scans use hipCUB (rocPRIM) device scan.
#include <hipcub/hipcub.hpp>

#include "qk_common.h"

namespace qk {

namespace {

constexpr int BLOCK = 256;
constexpr int SUB = 16;                    // lanes per seed row
constexpr int ROWS_PER_BLOCK = BLOCK / SUB;

__device__ __forceinline__ uint64_t splitmix64(uint64_t x) {
    x += 0x9e3779b97f4a7c15ULL;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
    return x ^ (x >> 31);
}

// Unbiased-enough bounded random for bound up to 2^31: take high bits.
__device__ __forceinline__ int64_t bounded_rand(uint64_t h, int64_t bound) {
    return (int64_t)(((h >> 11) * (uint64_t)bound) >> 53);
}

__global__ void capped_degree_kernel(const int64_t* __restrict__ indptr,
                                     const int64_t* __restrict__ seeds,
                                     int64_t n, int k,
                                     int64_t* __restrict__ capped,
                                     const int64_t* __restrict__ n_dev) {
    const int64_t nn = n_dev ? *n_dev : n;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {
        if (i >= nn) {          // slack: keep the downstream scan exact
            capped[i] = 0;
            continue;
        }
        int64_t v = seeds[i];
        int64_t deg = indptr[v + 1] - indptr[v];
        capped[i] = (k >= 0 && deg > k) ? k : deg;
    }
}

template <bool WITH_EID>
__global__ void __launch_bounds__(BLOCK)
sample_kernel(const int64_t* __restrict__ indptr,
              const int64_t* __restrict__ indices,
              const int64_t* __restrict__ eid_base,
              const int64_t* __restrict__ seeds, int64_t n, int k,
              const int64_t* __restrict__ prefix,
              int64_t* __restrict__ out_nbrs, int64_t* __restrict__ out_eids,
              uint64_t rng_seed, const uint64_t* __restrict__ rng_dev,
              const int64_t* __restrict__ n_dev) {
    // device-resident RNG state (hipGraph-capturable: a replayed graph
    // must not freeze a host-read seed); rng_seed is the per-hop salt
    if (rng_dev) rng_seed += *rng_dev;
    extern __shared__ __attribute__((aligned(16))) char smem[];
    int* slots = reinterpret_cast<int*>(smem);  // ROWS_PER_BLOCK * k

    const int64_t nn = n_dev ? *n_dev : n;
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t row = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t row_stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;

    for (; row < nn; row += row_stride) {
        const int64_t v = seeds[row];
        const int64_t beg = indptr[v];
        const int64_t deg = indptr[v + 1] - beg;
        const int64_t off = prefix[row];
        // RNG stream is keyed by ROW (not node id): duplicate seeds in a
        // batch must sample independently.
        const uint64_t base = rng_seed + (uint64_t)row * 0x9e3779b97f4a7c15ULL;
        if (deg <= (int64_t)k) {
            // copy all neighbors
            for (int64_t j = lane; j < deg; j += SUB) {
                out_nbrs[off + j] = indices[beg + j];
                if (WITH_EID)
                    out_eids[off + j] = eid_base ? eid_base[beg + j] : beg + j;
            }
        } else if (k <= 128 && deg > 2 * (int64_t)k) {
            // Floyd's uniform k-subset: O(k^2) independent of degree — on a
            // power-law graph hub seeds would otherwise serialize the whole
            // wave for O(deg) iterations.  One lane draws (sequential
            // dependency on the chosen set), all lanes gather.
            int* slot = slots + sub_id * k;
            if (lane == 0) {
                int c = 0;
                for (int64_t j = deg - k; j < deg; ++j) {
                    uint64_t h = splitmix64(
                        base + (uint64_t)j * 0x632be59bd9b4e019ULL);
                    int t = (int)bounded_rand(h, j + 1);
                    bool found = false;
                    for (int i = 0; i < c; ++i) {
                        if (slot[i] == t) { found = true; break; }
                    }
                    slot[c++] = found ? (int)j : t;
                }
            }
            // lane 0's LDS writes are visible to its wave without a barrier
            for (int i = lane; i < k; i += SUB) {
                int64_t p = beg + slot[i];
                out_nbrs[off + i] = indices[p];
                if (WITH_EID) out_eids[off + i] = eid_base ? eid_base[p] : p;
            }
        } else {
            // cooperative reservoir with LDS atomicMax replacement order
            int* slot = slots + sub_id * k;
            for (int i = lane; i < k; i += SUB) slot[i] = i;
            // lanes of one subgroup are in one wavefront: LDS writes above are
            // visible to the atomics below without a barrier (lock-step exec).
            for (int64_t j = k + lane; j < deg; j += SUB) {
                uint64_t h = splitmix64(base + (uint64_t)j * 0x632be59bd9b4e019ULL);
                int64_t r = bounded_rand(h, j + 1);
                if (r < k) atomicMax(&slot[r], (int)j);
            }
            for (int i = lane; i < k; i += SUB) {
                int64_t p = beg + slot[i];
                out_nbrs[off + i] = indices[p];
                if (WITH_EID) out_eids[off + i] = eid_base ? eid_base[p] : p;
            }
        }
    }
}

// One 16-lane subgroup per node; product-reduce over the subgroup with shfl.
__global__ void cal_next_kernel(const int64_t* __restrict__ indptr,
                                const int64_t* __restrict__ indices,
                                const float* __restrict__ last,
                                float* __restrict__ cur, int64_t node_count,
                                int k) {
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    int64_t v = (int64_t)blockIdx.x * ROWS_PER_BLOCK + sub_id;
    const int64_t stride = (int64_t)gridDim.x * ROWS_PER_BLOCK;
    for (; v < node_count; v += stride) {
        const int64_t beg = indptr[v];
        const int64_t deg = indptr[v + 1] - beg;
        float prod = 1.0f;
        for (int64_t j = lane; j < deg; j += SUB) {
            int64_t u = indices[beg + j];
            int64_t udeg = indptr[u + 1] - indptr[u];
            float w = (udeg > 0)
                          ? last[u] * fminf(1.0f, (float)k / (float)udeg)
                          : 0.0f;
            prod *= 1.0f - w;
        }
        // subgroup product reduction (within one wavefront)
        for (int off = SUB / 2; off > 0; off >>= 1)
            prod *= __shfl_down(prod, off, SUB);
        if (lane == 0) cur[v] = 1.0f - (1.0f - last[v]) * prod;
    }
}

inline int grid_for(int64_t work_items, int per_block) {
    int64_t blocks = (work_items + per_block - 1) / per_block;
    // 256 CUs x 8 blocks: cap and grid-stride the rest.
    // QUIVER_SAMPLE_BLOCKS: experimental cap for UVA-mode tuning (the
    // zero-copy CSR reads share the PCIe-contention physics of the
    // feature gather).
    static int env_cap = [] {
        const char* e = getenv("QUIVER_SAMPLE_BLOCKS");
        return e ? atoi(e) : 0;
    }();
    int64_t cap = env_cap > 0 ? env_cap : 2048;
    if (blocks > cap) blocks = cap;
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

}  // namespace

void launch_capped_degree(hipStream_t s, const int64_t* indptr,
                          const int64_t* seeds, int64_t n, int k,
                          int64_t* capped, const int64_t* n_dev) {
    if (n == 0) return;
    capped_degree_kernel<<<grid_for(n, BLOCK), BLOCK, 0, s>>>(
        indptr, seeds, n, k, capped, n_dev);
    QK_CHECK_HIP(hipGetLastError());
}

size_t scan_temp_bytes(int64_t n) {
    size_t bytes = 0;
    QK_CHECK_HIP((hipcub::DeviceScan::ExclusiveSum<const int64_t*, int64_t*>(
        nullptr, bytes, nullptr, nullptr, n)));
    return bytes + 64;
}

namespace {
__global__ void scan_total_kernel(const int64_t* in, const int64_t* scanned,
                                  int64_t n, int64_t* total) {
    *total = (n > 0) ? scanned[n - 1] + in[n - 1] : 0;
}
}  // namespace

void launch_exclusive_scan(hipStream_t s, void* temp, size_t temp_bytes,
                           const int64_t* in, int64_t* out, int64_t n,
                           int64_t* d_total) {
    QK_CHECK_HIP(
        hipcub::DeviceScan::ExclusiveSum(temp, temp_bytes, in, out, n, s));
    if (d_total) {
        scan_total_kernel<<<1, 1, 0, s>>>(in, out, n, d_total);
        QK_CHECK_HIP(hipGetLastError());
    }
}

void launch_sample(hipStream_t s, const int64_t* indptr, const int64_t* indices,
                   const int64_t* eid_base, const int64_t* seeds, int64_t n,
                   int k, const int64_t* prefix, int64_t* out_nbrs,
                   int64_t* out_eids, uint64_t rng_seed,
                   const uint64_t* rng_dev, const int64_t* n_dev) {
    if (n == 0) return;
    if (k < 1) throw std::runtime_error("sample: fanout k must be >= 1");
    size_t lds = (size_t)ROWS_PER_BLOCK * k * sizeof(int);
    if (lds > 160 * 1024)
        throw std::runtime_error("sample: fanout too large for LDS reservoir");
    int grid = grid_for(n, ROWS_PER_BLOCK);
    if (out_eids)
        sample_kernel<true><<<grid, BLOCK, lds, s>>>(indptr, indices, eid_base,
                                                     seeds, n, k, prefix,
                                                     out_nbrs, out_eids,
                                                     rng_seed, rng_dev, n_dev);
    else
        sample_kernel<false><<<grid, BLOCK, lds, s>>>(indptr, indices, eid_base,
                                                      seeds, n, k, prefix,
                                                      out_nbrs, nullptr,
                                                      rng_seed, rng_dev, n_dev);
    QK_CHECK_HIP(hipGetLastError());
}

__global__ void rng_bump_kernel(uint64_t* rng) {
    *rng += 0x9e3779b97f4a7c15ULL;
}

void launch_rng_bump(hipStream_t s, uint64_t* rng) {
    rng_bump_kernel<<<1, 1, 0, s>>>(rng);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_cal_next(hipStream_t s, const int64_t* indptr,
                     const int64_t* indices, const float* last, float* cur,
                     int64_t node_count, int k) {
    if (node_count == 0) return;
    cal_next_kernel<<<grid_for(node_count, ROWS_PER_BLOCK), BLOCK, 0, s>>>(
        indptr, indices, last, cur, node_count, k);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
