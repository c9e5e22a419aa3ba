// GPU hash-table reindex for frontier deduplication (gfx950).
//
// Capability parity with the reference's DeviceOrderedHashTable + reindex
// pipeline (torch-quiver srcs/cpp/include/quiver/reindex.cu.hpp:20-491,
// quiver_sample.cu:202-357), re-designed:
//  - single open-addressing table (linear probe) over [seeds ++ neighbors],
//    first-occurrence order via atomicMin on a position field; local ids
//    assigned by an exclusive scan over first-occurrence flags, so seeds
//    always get ids [0, n_seeds) in order.
//  - plain grid-stride wave64 kernels; device-scope atomics (cross-XCD safe).
#include "qk_common.h"

namespace qk {

namespace {

constexpr int BLOCK = 256;
constexpr int32_t POS_EMPTY = INT32_MAX;

__device__ __forceinline__ uint64_t hash64(uint64_t x) {
    x ^= x >> 33;
    x *= 0xff51afd7ed558ccdULL;
    x ^= x >> 33;
    x *= 0xc4ceb9fe1a85ec53ULL;
    x ^= x >> 33;
    return x;
}

__device__ __forceinline__ int64_t probe_find(const int64_t* keys,
                                              int64_t capacity, int64_t key) {
    int64_t mask = capacity - 1;
    int64_t slot = (int64_t)(hash64((uint64_t)key) & (uint64_t)mask);
    while (true) {
        int64_t k = keys[slot];
        if (k == key) return slot;
        // key must be present when called after insert; -1 means corrupt use
        if (k == -1) return -1;
        slot = (slot + 1) & mask;
    }
}

__device__ __forceinline__ const int64_t* combined_at(
    const int64_t* seeds, int64_t n_seeds, const int64_t* nbrs, int64_t i,
    int64_t* out) {
    *out = (i < n_seeds) ? seeds[i] : nbrs[i - n_seeds];
    return out;
}

__global__ void init_kernel(int64_t* keys, int32_t* pos, int64_t capacity) {
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < capacity; i += stride) {
        keys[i] = -1;
        pos[i] = POS_EMPTY;
    }
}

__global__ void insert_kernel(int64_t* keys, int32_t* pos, int64_t capacity,
                              const int64_t* seeds, int64_t n_seeds,
                              const int64_t* nbrs, int64_t total,
                              const int64_t* n_seeds_dev,
                              const int64_t* n_nbrs_dev) {
    const int64_t nn = n_seeds_dev ? *n_seeds_dev : n_seeds;
    const int64_t tt = n_nbrs_dev ? nn + *n_nbrs_dev : total;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t mask = capacity - 1;
    for (; i < tt; i += stride) {
        int64_t key;
        combined_at(seeds, nn, nbrs, i, &key);
        int64_t slot = (int64_t)(hash64((uint64_t)key) & (uint64_t)mask);
        while (true) {
            int64_t prev = atomicCAS(
                reinterpret_cast<unsigned long long*>(&keys[slot]),
                (unsigned long long)(-1LL), (unsigned long long)key);
            if (prev == -1 || prev == key) {
                atomicMin(&pos[slot], (int32_t)i);
                break;
            }
            slot = (slot + 1) & mask;
        }
    }
}

__global__ void mark_first_kernel(const int64_t* keys, const int32_t* pos,
                                  int64_t capacity, const int64_t* seeds,
                                  int64_t n_seeds, const int64_t* nbrs,
                                  int64_t total, int64_t* flags,
                                  const int64_t* n_seeds_dev,
                                  const int64_t* n_nbrs_dev) {
    const int64_t nn = n_seeds_dev ? *n_seeds_dev : n_seeds;
    const int64_t tt = n_nbrs_dev ? nn + *n_nbrs_dev : total;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < total; i += stride) {
        if (i >= tt) {          // slack: keep the downstream scan exact
            flags[i] = 0;
            continue;
        }
        int64_t key;
        combined_at(seeds, nn, nbrs, i, &key);
        int64_t slot = probe_find(keys, capacity, key);
        flags[i] = (slot >= 0 && pos[slot] == (int32_t)i) ? 1 : 0;
    }
}

__global__ void compact_kernel(const int64_t* keys, int32_t* local,
                               const int32_t* pos, int64_t capacity,
                               const int64_t* seeds, int64_t n_seeds,
                               const int64_t* nbrs, int64_t total,
                               const int64_t* scanned, const int64_t* flags,
                               int64_t* out_nodes,
                               const int64_t* n_seeds_dev,
                               const int64_t* n_nbrs_dev) {
    const int64_t nn = n_seeds_dev ? *n_seeds_dev : n_seeds;
    const int64_t tt = n_nbrs_dev ? nn + *n_nbrs_dev : total;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < tt; i += stride) {
        if (!flags[i]) continue;
        int64_t key;
        combined_at(seeds, nn, nbrs, i, &key);
        int64_t slot = probe_find(keys, capacity, key);
        int64_t lid = scanned[i];
        local[slot] = (int32_t)lid;
        out_nodes[lid] = key;
    }
}

__global__ void lookup_kernel(const int64_t* keys, const int32_t* local,
                              int64_t capacity, const int64_t* nbrs,
                              int64_t n_nbrs, int64_t* col_idx,
                              const int64_t* n_nbrs_dev) {
    const int64_t mm = n_nbrs_dev ? *n_nbrs_dev : n_nbrs;
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (; i < mm; i += stride) {
        int64_t slot = probe_find(keys, capacity, nbrs[i]);
        col_idx[i] = (int64_t)local[slot];
    }
}

// 16-lane subgroup per seed row writes its run of row ids.
__global__ void expand_rows_kernel(const int64_t* prefix, const int64_t* counts,
                                   int64_t n_seeds, int64_t* row_idx,
                                   const int64_t* n_dev) {
    const int64_t nn = n_dev ? *n_dev : n_seeds;
    constexpr int SUB = 16;
    const int sub_id = threadIdx.x / SUB;
    const int lane = threadIdx.x % SUB;
    const int rows_per_block = BLOCK / SUB;
    int64_t row = (int64_t)blockIdx.x * rows_per_block + sub_id;
    const int64_t stride = (int64_t)gridDim.x * rows_per_block;
    for (; row < nn; row += stride) {
        int64_t off = prefix[row];
        int64_t cnt = counts[row];
        for (int64_t j = lane; j < cnt; j += SUB) row_idx[off + j] = row;
    }
}

inline int grid_for(int64_t work, int per_block) {
    int64_t blocks = (work + per_block - 1) / per_block;
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    return (int)blocks;
}

}  // namespace

void launch_reindex_init(hipStream_t s, int64_t* keys, int32_t* pos,
                         int64_t capacity) {
    init_kernel<<<grid_for(capacity, BLOCK), BLOCK, 0, s>>>(keys, pos,
                                                            capacity);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_hash_insert(hipStream_t s, int64_t* keys, int32_t* pos,
                        int64_t capacity, const int64_t* seeds, int64_t n_seeds,
                        const int64_t* nbrs, int64_t n_nbrs,
                        const int64_t* n_seeds_dev,
                        const int64_t* n_nbrs_dev) {
    int64_t total = n_seeds + n_nbrs;
    if (total == 0) return;
    insert_kernel<<<grid_for(total, BLOCK), BLOCK, 0, s>>>(
        keys, pos, capacity, seeds, n_seeds, nbrs, total, n_seeds_dev,
        n_nbrs_dev);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_mark_first(hipStream_t s, const int64_t* keys, const int32_t* pos,
                       int64_t capacity, const int64_t* seeds, int64_t n_seeds,
                       const int64_t* nbrs, int64_t n_nbrs, int64_t* flags,
                       const int64_t* n_seeds_dev,
                       const int64_t* n_nbrs_dev) {
    int64_t total = n_seeds + n_nbrs;
    if (total == 0) return;
    mark_first_kernel<<<grid_for(total, BLOCK), BLOCK, 0, s>>>(
        keys, pos, capacity, seeds, n_seeds, nbrs, total, flags, n_seeds_dev,
        n_nbrs_dev);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_compact_unique(hipStream_t s, const int64_t* keys, int32_t* local,
                           const int32_t* pos, int64_t capacity,
                           const int64_t* seeds, int64_t n_seeds,
                           const int64_t* nbrs, int64_t n_nbrs,
                           const int64_t* scanned_flags, const int64_t* flags,
                           int64_t* out_nodes,
                           const int64_t* n_seeds_dev,
                           const int64_t* n_nbrs_dev) {
    int64_t total = n_seeds + n_nbrs;
    if (total == 0) return;
    compact_kernel<<<grid_for(total, BLOCK), BLOCK, 0, s>>>(
        keys, local, pos, capacity, seeds, n_seeds, nbrs, total, scanned_flags,
        flags, out_nodes, n_seeds_dev, n_nbrs_dev);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_lookup_local(hipStream_t s, const int64_t* keys,
                         const int32_t* local, int64_t capacity,
                         const int64_t* nbrs, int64_t n_nbrs,
                         int64_t* col_idx, const int64_t* n_nbrs_dev) {
    if (n_nbrs == 0) return;
    lookup_kernel<<<grid_for(n_nbrs, BLOCK), BLOCK, 0, s>>>(
        keys, local, capacity, nbrs, n_nbrs, col_idx, n_nbrs_dev);
    QK_CHECK_HIP(hipGetLastError());
}

void launch_expand_rows(hipStream_t s, const int64_t* prefix,
                        const int64_t* counts, int64_t n_seeds,
                        int64_t* row_idx, const int64_t* n_dev) {
    if (n_seeds == 0) return;
    expand_rows_kernel<<<grid_for(n_seeds, BLOCK / 16), BLOCK, 0, s>>>(
        prefix, counts, n_seeds, row_idx, n_dev);
    QK_CHECK_HIP(hipGetLastError());
}

}  // namespace qk
