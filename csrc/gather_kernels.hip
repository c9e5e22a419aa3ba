// Multi-source feature gather/scatter kernels for gfx950 (MI355X).
//
// Capability parity with the reference's quiver_tensor_gather
// (torch-quiver srcs/cpp/include/quiver/shard_tensor.cu.hpp:19-61),
// re-designed MI355X-first:
//  - a virtual row-major tensor is a list of <=16 shards living in local
//    HBM3E, peer-GPU HBM reached over xGMI one-sided loads, or pinned host
//    DRAM (zero-copy).  Shard table is passed BY VALUE in kernel args, so the
//    offset search runs out of SGPRs (reference scans a global array).
//  - rows are copied with the widest aligned vector type (16B dwordx4 when
//    row_bytes % 16 == 0) by a subgroup of lanes sized to the row, so a
//    wave64 moves up to 1 KiB per instruction; reference copies byte-wise.
#include <cstdlib>

#include "qk_common.h"

namespace qk {

namespace {

constexpr int BLOCK = 256;

struct alignas(16) vec16 { uint64_t a, b; };

template <typename VecT, int SUB, bool SCATTER>
__global__ void __launch_bounds__(BLOCK)
copy_rows_kernel(GatherSpec spec, const int64_t* __restrict__ indices,
                 int64_t n, char* __restrict__ other,
                 const int64_t* __restrict__ n_dev) {
    if (n_dev) n = min(n, *n_dev);
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    const int rows_per_block = BLOCK / SUB;
    int64_t row = (int64_t)blockIdx.x * rows_per_block + sub_id;
    const int64_t stride = (int64_t)gridDim.x * rows_per_block;
    const int64_t nvec = spec.row_bytes / (int64_t)sizeof(VecT);

    for (; row < n; row += stride) {
        const int64_t idx = indices[row];
        // shard search: spec lives in kernel args (scalar regs), <=16 entries
        int s = 0;
        while (s < spec.nshards && idx >= spec.ends[s]) ++s;
        if (s >= spec.nshards) continue;                  // out of range
        if (!((spec.access_mask >> s) & 1u)) continue;    // peer pass fills it
        const int64_t start = (s == 0) ? 0 : spec.ends[s - 1];
        const char* shard_row =
            spec.ptrs[s] + (idx - start) * spec.row_bytes;
        char* io_row = other + row * spec.row_bytes;
        if (SCATTER) {
            VecT* dst = (VecT*)shard_row;
            const VecT* src = (const VecT*)io_row;
            for (int64_t j = lane; j < nvec; j += SUB) dst[j] = src[j];
        } else {
            const VecT* src = (const VecT*)shard_row;
            VecT* dst = (VecT*)io_row;
            for (int64_t j = lane; j < nvec; j += SUB) dst[j] = src[j];
        }
    }
}

inline int grid_for(int64_t work, int per_block, int max_blocks) {
    int64_t blocks = (work + per_block - 1) / per_block;
    if (blocks > max_blocks) blocks = max_blocks;
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

inline int gather_max_blocks(const GatherSpec& spec, bool overlapped) {
    // A gather touching a pinned-host (zero-copy) shard is PCIe-latency
    // bound and its uncached reads poison co-resident kernels, so the
    // grid is capped to leave CUs free.  Two profiles, both measured on
    // the products bench (blocks -> ms/step: 24:10.8, 56:5.0, 96:3.64,
    // 192:3.9, 640:4.3, 1280:4.4):
    //  - overlapped (the async sample->gather chain, which by
    //    construction runs concurrently with model compute): 96 blocks
    //    — the knee where the PCIe link still saturates but CU-slot
    //    contention stops hurting the compute stream;
    //  - standalone (direct Feature[...] indexing): 640 blocks for full
    //    link occupancy when nothing competes.
    // Pure-HBM/xGMI gathers want the full chip either way.
    static int env_cap = [] {
        const char* e = getenv("QUIVER_GATHER_BLOCKS");
        return e ? atoi(e) : 0;
    }();
    if (env_cap > 0) return env_cap;
    if (!spec.has_host_shard) return 2048;
    // knee band measured per row size (ms/step sweeps on the bench):
    // 400 B fp32 products rows: best 96-112; 200-256 B bf16 rows: best
    // ~160 (smaller rows need more rows in flight for the same PCIe
    // occupancy); papers 512 B rows within noise of 112-128.
    if (overlapped) return spec.row_bytes <= 256 ? 160 : 112;
    return 640;
}

template <typename VecT, bool SCATTER>
void dispatch_sub(hipStream_t s, const GatherSpec& spec,
                  const int64_t* indices, int64_t n, char* other,
                  const int64_t* n_dev) {
    int64_t nvec = spec.row_bytes / (int64_t)sizeof(VecT);
    int sub = 4;
    while (sub < 64 && sub < nvec) sub *= 2;  // cover the row in ~1 pass
    // n_dev-driven gathers are the async chain: always compute-overlapped
    int grid = grid_for(n, BLOCK / sub,
                        gather_max_blocks(spec, n_dev != nullptr));
    switch (sub) {
#define QK_CASE(S)                                                          \
    case S:                                                                 \
        copy_rows_kernel<VecT, S, SCATTER>                                  \
            <<<grid, BLOCK, 0, s>>>(spec, indices, n, other, n_dev);        \
        break;
        QK_CASE(4) QK_CASE(8) QK_CASE(16) QK_CASE(32) QK_CASE(64)
#undef QK_CASE
    }
    QK_CHECK_HIP(hipGetLastError());
}

template <bool SCATTER>
void copy_rows(hipStream_t s, const GatherSpec& spec, const int64_t* indices,
               int64_t n, char* other, const int64_t* n_dev) {
    if (n == 0 || spec.nshards == 0) return;
    if (spec.row_bytes % 16 == 0)
        dispatch_sub<vec16, SCATTER>(s, spec, indices, n, other, n_dev);
    else if (spec.row_bytes % 8 == 0)
        dispatch_sub<uint64_t, SCATTER>(s, spec, indices, n, other, n_dev);
    else if (spec.row_bytes % 4 == 0)
        dispatch_sub<uint32_t, SCATTER>(s, spec, indices, n, other, n_dev);
    else
        dispatch_sub<uint8_t, SCATTER>(s, spec, indices, n, other, n_dev);
}

}  // namespace

void launch_gather(hipStream_t s, const GatherSpec& spec,
                   const int64_t* indices, int64_t n, char* out,
                   const int64_t* n_dev) {
    copy_rows<false>(s, spec, indices, n, out, n_dev);
}

void launch_scatter(hipStream_t s, const GatherSpec& spec,
                    const int64_t* indices, int64_t n, const char* src) {
    copy_rows<true>(s, spec, indices, n, const_cast<char*>(src), nullptr);
}

}  // namespace qk
