// torch_quiver — MI355X-native pybind module.
//
// Engine classes re-implementing the reference's native surface
// (torch-quiver srcs/cpp/src/quiver/{cuda/quiver_sample.cu,
// cuda/quiver_feature.cu, cuda/quiver_comm.cu, quiver.cpp,
// torch/module.cpp}) as an idiomatic ROCm design: torch caching allocator
// for scratch, c10::hip streams, RCCL collectives, hipIpc sharing, and the
// wave64 kernels in *_kernels.hip.
#include <torch/extension.h>

#include <ATen/Parallel.h>
#include <c10/hip/HIPCachingAllocator.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <algorithm>
#include <atomic>
#include <condition_variable>
#include <functional>
#include <mutex>
#include <thread>
#include <random>
#include <unordered_map>
#include <vector>

#include "qk_common.h"

namespace {

#define QK_CHECK_RCCL(expr)                                                   \
    do {                                                                      \
        ncclResult_t _r = (expr);                                             \
        if (_r != ncclSuccess) {                                              \
            throw std::runtime_error(std::string("RCCL error: ") +            \
                                     ncclGetErrorString(_r));                 \
        }                                                                     \
    } while (0)

inline hipStream_t current_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}

inline int64_t next_pow2(int64_t x) {
    int64_t p = 64;
    while (p < x) p <<= 1;
    return p;
}

struct DeviceScope {
    int prev = -1;
    explicit DeviceScope(int dev) {
        QK_CHECK_HIP(hipGetDevice(&prev));
        if (dev != prev) QK_CHECK_HIP(hipSetDevice(dev));
    }
    ~DeviceScope() { (void)hipSetDevice(prev); }
};

// Chunked hipHostRegister for multi-GB UVA buffers (papers100M indices are
// 12.5 GB): registering in 1 GB slices bounds the per-call pinning peak —
// the reference's hard-won lesson (quiverRegister, quiver.cu.hpp:16-26).
// Kernels need ONE contiguous device view, so the per-chunk device
// pointers are verified contiguous; if the runtime ever maps them apart,
// everything is unregistered and one whole-buffer registration is used
// instead.  Returns the device-visible base pointer; appends every
// registered host pointer to `registered` (for hipHostUnregister later).
// Already-registered memory (e.g. a shared tensor registered by a prior
// ShardTensor) is detected on the first chunk and reused.
void* register_host_chunked(void* p, size_t bytes,
                            std::vector<void*>& registered) {
    constexpr size_t kChunk = 1ull << 30;
    constexpr unsigned kFlags =
        hipHostRegisterMapped | hipHostRegisterPortable;
    auto devptr = [](void* hp) {
        void* dp = nullptr;
        QK_CHECK_HIP(hipHostGetDevicePointer(&dp, hp, 0));
        return dp;
    };
    char* base = (char*)p;
    std::vector<void*> regs;
    void* dp0 = nullptr;
    bool contiguous = true;
    for (size_t off = 0; off < bytes; off += kChunk) {
        size_t len = std::min(kChunk, bytes - off);
        hipError_t err = hipHostRegister(base + off, len, kFlags);
        if (err == hipErrorHostMemoryAlreadyRegistered && off == 0) {
            (void)hipGetLastError();
            return devptr(base);  // whole buffer registered by someone else
        }
        if (err != hipSuccess) {
            for (void* r : regs) (void)hipHostUnregister(r);
            (void)hipGetLastError();
            QK_CHECK_HIP(err);  // throws with the error string
        }
        regs.push_back(base + off);
        void* dp = devptr(base + off);
        if (off == 0) dp0 = dp;
        else if ((char*)dp != (char*)dp0 + off) contiguous = false;
    }
    if (!contiguous) {
        for (void* r : regs) (void)hipHostUnregister(r);
        QK_CHECK_HIP(hipHostRegister(p, bytes, kFlags));
        registered.push_back(p);
        return devptr(p);
    }
    registered.insert(registered.end(), regs.begin(), regs.end());
    return dp0;
}

// Persistent worker pool for the CPU-staged gather.  Spawning threads
// per call costs ~2 ms (64 x ~30 us) — more than the copy itself — and
// at::parallel_for from a non-main thread degrades to serial; a resident
// pool with an atomic chunk cursor avoids both.
class WorkerPool {
  public:
    static WorkerPool& inst() {
        // leaked singleton: worker threads must not race static
        // destruction at process exit
        static WorkerPool* p = new WorkerPool();
        return *p;
    }

    template <typename F>
    void run(int64_t nchunk, F&& fn) {
        if (nchunk <= 1) {
            if (nchunk == 1) fn((int64_t)0);
            return;
        }
        // one job at a time: concurrent callers would interleave epochs
        std::lock_guard<std::mutex> serial(run_mu_);
        {
            std::unique_lock<std::mutex> lk(mu_);
            job_ = [&fn](int64_t c) { fn(c); };
            nchunk_ = nchunk;
            cursor_.store(0);
            pending_.store((int)workers_.size());
            ++epoch_;
        }
        cv_.notify_all();
        // the caller helps
        drain();
        std::unique_lock<std::mutex> lk(mu_);
        done_cv_.wait(lk, [this] { return pending_.load() == 0; });
        job_ = nullptr;
    }

  private:
    WorkerPool() {
        unsigned hw = std::thread::hardware_concurrency();
        int n = std::min(63u, hw > 2 ? hw - 2 : 1u);
        for (int i = 0; i < n; ++i)
            workers_.emplace_back([this] { worker(); });
    }
    void drain() {
        int64_t c;
        while ((c = cursor_.fetch_add(1)) < nchunk_) job_(c);
    }
    void worker() {
        uint64_t seen = 0;
        while (true) {
            std::unique_lock<std::mutex> lk(mu_);
            cv_.wait(lk, [&] { return epoch_ != seen; });
            seen = epoch_;
            auto job = job_;
            int64_t total = nchunk_;
            lk.unlock();
            if (job) {
                int64_t c;
                while ((c = cursor_.fetch_add(1)) < total) job(c);
            }
            if (pending_.fetch_sub(1) == 1) {
                std::lock_guard<std::mutex> g(mu_);
                done_cv_.notify_all();
            }
        }
    }
    std::vector<std::thread> workers_;
    std::mutex run_mu_;
    std::mutex mu_;
    std::condition_variable cv_, done_cv_;
    std::function<void(int64_t)> job_;
    std::atomic<int64_t> cursor_{0};
    int64_t nchunk_ = 0;
    std::atomic<int> pending_{0};
    uint64_t epoch_ = 0;
};

template <typename F>
void parallel_chunks(int64_t nchunk, int /*nthreads*/, F&& fn) {
    WorkerPool::inst().run(nchunk, [&fn](int64_t c) { fn(c, c + 1); });
}

// Exclusive scan of an int64 device tensor; returns (prefix, total-on-host).
std::pair<torch::Tensor, int64_t> exclusive_scan_total(
    const torch::Tensor& vals, hipStream_t stream) {
    int64_t n = vals.numel();
    auto prefix = torch::empty_like(vals);
    auto temp = torch::empty(
        {(int64_t)qk::scan_temp_bytes(n)},
        torch::TensorOptions().dtype(torch::kUInt8).device(vals.device()));
    auto d_total = torch::empty(
        {1}, torch::TensorOptions().dtype(torch::kInt64).device(vals.device()));
    qk::launch_exclusive_scan(stream, temp.data_ptr(), temp.numel(),
                              vals.data_ptr<int64_t>(),
                              prefix.data_ptr<int64_t>(), n,
                              d_total.data_ptr<int64_t>());
    int64_t h_total = 0;
    QK_CHECK_HIP(hipMemcpyAsync(&h_total, d_total.data_ptr(), sizeof(int64_t),
                                hipMemcpyDeviceToHost, stream));
    QK_CHECK_HIP(hipStreamSynchronize(stream));
    return {prefix, h_total};
}

// ---------------------------------------------------------------------------
// GPU sampler engine (parity: reference TorchQuiver, quiver_sample.cu:77-358)
// ---------------------------------------------------------------------------
class GpuSampler {
  public:
    GpuSampler(torch::Tensor indptr, torch::Tensor indices, torch::Tensor eid,
               int device, bool dma_mode)
        : device_(device), dma_(dma_mode) {
        TORCH_CHECK(indptr.dtype() == torch::kInt64, "indptr must be int64");
        TORCH_CHECK(indices.dtype() == torch::kInt64, "indices must be int64");
        DeviceScope g(device_);
        auto dev = torch::Device(torch::kCUDA, device_);
        // indptr is always device-resident: it is read twice per seed and is
        // small (node_count+1).
        indptr_ = indptr.contiguous().to(dev);
        node_count_ = indptr_.numel() - 1;
        has_eid_ = eid.defined() && eid.numel() == indices.numel();
        if (dma_) {
            indices_ = indices.contiguous().to(dev);
            indices_dptr_ = indices_.data_ptr<int64_t>();
            if (has_eid_) {
                eid_ = eid.contiguous().to(dev);
                eid_dptr_ = eid_.data_ptr<int64_t>();
            }
        } else {
            // ZERO_COPY: pin the host CSR columns and read them from kernels
            // over PCIe (UVA).  Keeps graphs larger than HBM samplable.
            indices_ = indices.contiguous();
            TORCH_CHECK(indices_.device().is_cpu(),
                        "UVA mode expects CPU indices");
            indices_dptr_ = (int64_t*)register_host(
                indices_.data_ptr(), indices_.numel() * sizeof(int64_t));
            if (has_eid_) {
                eid_ = eid.contiguous();
                eid_dptr_ = (int64_t*)register_host(
                    eid_.data_ptr(), eid_.numel() * sizeof(int64_t));
            }
        }
    }

    ~GpuSampler() {
        for (void* p : registered_) (void)hipHostUnregister(p);
    }

    void set_seed(uint64_t seed) {
        rng_counter_.store(seed);
        write_rng(seed);
    }

    std::tuple<torch::Tensor, torch::Tensor> sample_neighbor(
        int /*stream_id*/, torch::Tensor seeds, int k) {
        DeviceScope g(device_);
        auto stream = current_stream();
        seeds = seeds.contiguous();
        TORCH_CHECK(seeds.device().is_cuda(), "seeds must be on GPU");
        int64_t n = seeds.numel();
        auto opts =
            torch::TensorOptions().dtype(torch::kInt64).device(seeds.device());
        auto counts = torch::empty({n}, opts);
        qk::launch_capped_degree(stream, indptr_.data_ptr<int64_t>(),
                                 seeds.data_ptr<int64_t>(), n, k,
                                 counts.data_ptr<int64_t>());
        auto [prefix, total] = exclusive_scan_total(counts, stream);
        auto out = torch::empty({total}, opts);
        qk::launch_rng_bump(stream, rng_dev(stream));
        qk::launch_sample(stream, indptr_.data_ptr<int64_t>(), indices_dptr_,
                          nullptr, seeds.data_ptr<int64_t>(), n, k,
                          prefix.data_ptr<int64_t>(), out.data_ptr<int64_t>(),
                          nullptr, 0, rng_dev(stream));
        return {out, counts};
    }

    // Returns (frontier, row_idx, col_idx): frontier[0:n_seeds] == seeds.
    std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> reindex_single(
        torch::Tensor seeds, torch::Tensor nbrs, torch::Tensor counts) {
        DeviceScope g(device_);
        auto stream = current_stream();
        seeds = seeds.contiguous();
        nbrs = nbrs.contiguous();
        counts = counts.contiguous();
        int64_t n = seeds.numel(), m = nbrs.numel();
        auto opts =
            torch::TensorOptions().dtype(torch::kInt64).device(seeds.device());
        auto i32 = opts.dtype(torch::kInt32);

        int64_t capacity = next_pow2(2 * (n + m) + 64);
        auto keys = torch::empty({capacity}, opts);
        auto pos = torch::empty({capacity}, i32);
        auto local = torch::empty({capacity}, i32);
        qk::launch_reindex_init(stream, keys.data_ptr<int64_t>(),
                                pos.data_ptr<int32_t>(), capacity);
        qk::launch_hash_insert(stream, keys.data_ptr<int64_t>(),
                               pos.data_ptr<int32_t>(), capacity,
                               seeds.data_ptr<int64_t>(), n,
                               nbrs.data_ptr<int64_t>(), m);
        auto flags = torch::empty({n + m}, opts);
        qk::launch_mark_first(stream, keys.data_ptr<int64_t>(),
                              pos.data_ptr<int32_t>(), capacity,
                              seeds.data_ptr<int64_t>(), n,
                              nbrs.data_ptr<int64_t>(), m,
                              flags.data_ptr<int64_t>());
        auto [scanned, unique_total] = exclusive_scan_total(flags, stream);
        auto frontier = torch::empty({unique_total}, opts);
        qk::launch_compact_unique(
            stream, keys.data_ptr<int64_t>(), local.data_ptr<int32_t>(),
            pos.data_ptr<int32_t>(), capacity, seeds.data_ptr<int64_t>(), n,
            nbrs.data_ptr<int64_t>(), m, scanned.data_ptr<int64_t>(),
            flags.data_ptr<int64_t>(), frontier.data_ptr<int64_t>());
        auto col_idx = torch::empty({m}, opts);
        qk::launch_lookup_local(stream, keys.data_ptr<int64_t>(),
                                local.data_ptr<int32_t>(), capacity,
                                nbrs.data_ptr<int64_t>(), m,
                                col_idx.data_ptr<int64_t>());
        auto [prefix, total] = exclusive_scan_total(counts, stream);
        TORCH_CHECK(total == m, "reindex: counts do not sum to neighbors");
        auto row_idx = torch::empty({m}, opts);
        qk::launch_expand_rows(stream, prefix.data_ptr<int64_t>(),
                               counts.data_ptr<int64_t>(), n,
                               row_idx.data_ptr<int64_t>());
        return {frontier, row_idx, col_idx};
    }

    // Fused multi-hop sample+reindex: the whole batch runs with ZERO
    // host-device synchronizations (one D2H of all hop sizes at the end).
    // Per-hop buffers are allocated at upper bounds (n_{h+1} <= n_h*(k+1));
    // kernels read exact counts from device scalars written by the scans.
    // Removes ~12 stream syncs and ~40 python/torch ops per batch vs the
    // per-hop path — the training step was launch/GIL-bound, not
    // kernel-bound (profiles/: both streams <50% busy).
    // Returns per hop: (frontier, edges) where edges is ONE [2, m] int64
    // tensor with edges[0] = col_idx (src local ids) and edges[1] =
    // row_idx (dst local ids) — the finished PyG edge_index with no
    // torch.stack copy on the consumer's critical path (the stack of two
    // ~1M-row hop buffers cost ~0.63 ms/step of main-stream time).
    // frontier[0:n_prev] == previous hop's frontier (seeds first), sizes
    // already exact.
    // Raw variant: returns UPPER-BOUND-sized per-hop tensors plus the
    // device-resident sizes vector [m_0, u_0, m_1, u_1, ...] WITHOUT any
    // host synchronization — callers chain the feature gather behind it
    // (gather kernel reads the exact frontier count from sizes_dev) and
    // read the sizes only when the whole chain has been consumed.
    // Frontier slack is zeroed so downstream indexing (feature_order
    // remap, gather) stays in-range.
    std::tuple<std::vector<std::tuple<torch::Tensor, torch::Tensor>>,
               torch::Tensor>
    sample_hops_raw(torch::Tensor seeds, const std::vector<int>& ks) {
        DeviceScope g(device_);
        auto stream = current_stream();
        seeds = seeds.contiguous();
        TORCH_CHECK(seeds.device().is_cuda(), "seeds must be on GPU");
        const int H = (int)ks.size();
        for (int k : ks) TORCH_CHECK(k >= 1, "sample_hops: fanouts must be >=1");
        auto opts =
            torch::TensorOptions().dtype(torch::kInt64).device(seeds.device());
        auto i32 = opts.dtype(torch::kInt32);

        auto sizes_dev = torch::empty({2 * H}, opts);
        int64_t* sd = sizes_dev.data_ptr<int64_t>();
        // one seed advance per batch, as a device-side graph node (the
        // chain stays hipGraph-capturable; per-hop streams get a salt)
        qk::launch_rng_bump(stream, rng_dev(stream));

        torch::Tensor cur = seeds;
        const int64_t* n_dev = nullptr;     // exact frontier count (device)
        int64_t n_ub = seeds.numel();       // upper bound (host)
        std::vector<torch::Tensor> frontiers, hop_edges;
        std::vector<int64_t> n_ubs;

        for (int h = 0; h < H; ++h) {
            const int k = ks[h];
            const int64_t m_ub = n_ub * k;
            auto counts = torch::empty({n_ub}, opts);
            qk::launch_capped_degree(stream, indptr_.data_ptr<int64_t>(),
                                     cur.data_ptr<int64_t>(), n_ub, k,
                                     counts.data_ptr<int64_t>(), n_dev);
            auto prefix = torch::empty({n_ub}, opts);
            auto temp = torch::empty({(int64_t)qk::scan_temp_bytes(n_ub)},
                                     opts.dtype(torch::kUInt8));
            qk::launch_exclusive_scan(stream, temp.data_ptr(), temp.numel(),
                                      counts.data_ptr<int64_t>(),
                                      prefix.data_ptr<int64_t>(), n_ub,
                                      sd + 2 * h);  // m_h
            auto out = torch::empty({m_ub}, opts);
            const uint64_t salt = (uint64_t)(h + 1) * 0xD1B54A32D192ED03ULL;
            qk::launch_sample(stream, indptr_.data_ptr<int64_t>(),
                              indices_dptr_, nullptr,
                              cur.data_ptr<int64_t>(), n_ub, k,
                              prefix.data_ptr<int64_t>(),
                              out.data_ptr<int64_t>(), nullptr, salt,
                              rng_dev(stream), n_dev);

            // reindex [cur ++ out] -> frontier + local col ids
            const int64_t total_ub = n_ub + m_ub;
            const int64_t capacity = next_pow2(2 * total_ub + 64);
            auto keys = torch::empty({capacity}, opts);
            auto pos = torch::empty({capacity}, i32);
            auto local = torch::empty({capacity}, i32);
            qk::launch_reindex_init(stream, keys.data_ptr<int64_t>(),
                                    pos.data_ptr<int32_t>(), capacity);
            qk::launch_hash_insert(stream, keys.data_ptr<int64_t>(),
                                   pos.data_ptr<int32_t>(), capacity,
                                   cur.data_ptr<int64_t>(), n_ub,
                                   out.data_ptr<int64_t>(), m_ub, n_dev,
                                   sd + 2 * h);
            auto flags = torch::empty({total_ub}, opts);
            qk::launch_mark_first(stream, keys.data_ptr<int64_t>(),
                                  pos.data_ptr<int32_t>(), capacity,
                                  cur.data_ptr<int64_t>(), n_ub,
                                  out.data_ptr<int64_t>(), m_ub,
                                  flags.data_ptr<int64_t>(), n_dev,
                                  sd + 2 * h);
            auto scanned = torch::empty({total_ub}, opts);
            auto temp2 = torch::empty(
                {(int64_t)qk::scan_temp_bytes(total_ub)},
                opts.dtype(torch::kUInt8));
            qk::launch_exclusive_scan(stream, temp2.data_ptr(), temp2.numel(),
                                      flags.data_ptr<int64_t>(),
                                      scanned.data_ptr<int64_t>(), total_ub,
                                      sd + 2 * h + 1);  // unique_h
            auto frontier = torch::zeros({total_ub}, opts);
            qk::launch_compact_unique(
                stream, keys.data_ptr<int64_t>(), local.data_ptr<int32_t>(),
                pos.data_ptr<int32_t>(), capacity, cur.data_ptr<int64_t>(),
                n_ub, out.data_ptr<int64_t>(), m_ub,
                scanned.data_ptr<int64_t>(), flags.data_ptr<int64_t>(),
                frontier.data_ptr<int64_t>(), n_dev, sd + 2 * h);
            // edges[0] = col (src local), edges[1] = row (dst local):
            // the finished edge_index, written in place by the kernels
            auto edges = torch::empty({2, m_ub}, opts);
            int64_t* colp = edges.data_ptr<int64_t>();
            int64_t* rowp = colp + m_ub;
            qk::launch_lookup_local(stream, keys.data_ptr<int64_t>(),
                                    local.data_ptr<int32_t>(), capacity,
                                    out.data_ptr<int64_t>(), m_ub,
                                    colp, sd + 2 * h);
            qk::launch_expand_rows(stream, prefix.data_ptr<int64_t>(),
                                   counts.data_ptr<int64_t>(), n_ub,
                                   rowp, n_dev);

            frontiers.push_back(frontier);
            hop_edges.push_back(edges);
            n_ubs.push_back(n_ub);
            cur = frontier;
            n_dev = sd + 2 * h + 1;
            n_ub = total_ub;
        }

        std::vector<std::tuple<torch::Tensor, torch::Tensor>> raw;
        for (int h = 0; h < H; ++h)
            raw.emplace_back(frontiers[h], hop_edges[h]);
        return {raw, sizes_dev};
    }

    std::vector<std::tuple<torch::Tensor, torch::Tensor>>
    sample_hops(torch::Tensor seeds, const std::vector<int>& ks) {
        auto [raw, sizes_dev] = sample_hops_raw(seeds, ks);
        // the ONE sync of the whole batch
        auto sizes_host = sizes_dev.cpu();
        const int64_t* sh = sizes_host.data_ptr<int64_t>();
        std::vector<std::tuple<torch::Tensor, torch::Tensor>> res;
        for (size_t h = 0; h < raw.size(); ++h) {
            int64_t m = sh[2 * h], u = sh[2 * h + 1];
            res.emplace_back(std::get<0>(raw[h]).narrow(0, 0, u),
                             std::get<1>(raw[h]).narrow(1, 0, m));
        }
        return res;
    }

    void cal_neighbor_prob(int /*stream_id*/, torch::Tensor last_prob,
                           torch::Tensor cur_prob, int k) {
        DeviceScope g(device_);
        TORCH_CHECK(dma_ || indices_dptr_, "sampler not initialised");
        qk::launch_cal_next(current_stream(), indptr_.data_ptr<int64_t>(),
                            indices_dptr_, last_prob.data_ptr<float>(),
                            cur_prob.data_ptr<float>(), node_count_, k);
    }

    int64_t node_count() const { return node_count_; }
    int64_t edge_count() const { return indices_.numel(); }
    int device() const { return device_; }

  private:
    void* register_host(void* p, size_t bytes) {
        return register_host_chunked(p, bytes, registered_);
    }

    // lazily-allocated device RNG word (int64 tensor reinterpreted)
    uint64_t* rng_dev(hipStream_t /*stream*/) {
        if (!rng_dev_.defined()) {
            rng_dev_ = torch::empty(
                {1}, torch::TensorOptions().dtype(torch::kInt64).device(
                         torch::Device(torch::kCUDA, device_)));
            write_rng(rng_counter_.load());
        }
        return (uint64_t*)rng_dev_.data_ptr();
    }

    void write_rng(uint64_t v) {
        if (!rng_dev_.defined()) return;  // picked up at first rng_dev()
        DeviceScope g(device_);
        QK_CHECK_HIP(hipMemcpy(rng_dev_.data_ptr(), &v, sizeof(v),
                               hipMemcpyHostToDevice));
    }

    int device_;
    bool dma_;
    bool has_eid_ = false;
    int64_t node_count_ = 0;
    torch::Tensor indptr_, indices_, eid_;
    int64_t* indices_dptr_ = nullptr;
    int64_t* eid_dptr_ = nullptr;
    std::vector<void*> registered_;
    std::atomic<uint64_t> rng_counter_{0x853c49e6748fea9bULL};
    torch::Tensor rng_dev_;
};

// ---------------------------------------------------------------------------
// CPU sampler engine (parity: reference CPUQuiver, quiver.cpp:11-85)
// ---------------------------------------------------------------------------
class CpuSampler {
  public:
    CpuSampler(torch::Tensor indptr, torch::Tensor indices)
        : indptr_(indptr.contiguous()), indices_(indices.contiguous()) {
        TORCH_CHECK(indptr_.device().is_cpu() && indices_.device().is_cpu(),
                    "CpuSampler expects CPU tensors");
        TORCH_CHECK(indptr_.dtype() == torch::kInt64 &&
                    indices_.dtype() == torch::kInt64);
        node_count_ = indptr_.numel() - 1;
    }

    std::tuple<torch::Tensor, torch::Tensor> sample_neighbor(
        torch::Tensor seeds, int k) {
        seeds = seeds.contiguous().cpu();
        int64_t n = seeds.numel();
        const int64_t* sp = seeds.data_ptr<int64_t>();
        const int64_t* indptr = indptr_.data_ptr<int64_t>();
        const int64_t* indices = indices_.data_ptr<int64_t>();
        auto counts = torch::empty({n}, seeds.options());
        int64_t* cp = counts.data_ptr<int64_t>();
        at::parallel_for(0, n, 512, [&](int64_t b, int64_t e) {
            for (int64_t i = b; i < e; ++i) {
                int64_t deg = indptr[sp[i] + 1] - indptr[sp[i]];
                cp[i] = (k >= 0 && deg > k) ? k : deg;
            }
        });
        std::vector<int64_t> prefix(n + 1, 0);
        for (int64_t i = 0; i < n; ++i) prefix[i + 1] = prefix[i] + cp[i];
        auto out = torch::empty({prefix[n]}, seeds.options());
        int64_t* op = out.data_ptr<int64_t>();
        at::parallel_for(0, n, 64, [&](int64_t b, int64_t e) {
            thread_local std::mt19937_64 gen(std::random_device{}());
            std::vector<int64_t> slot;
            for (int64_t i = b; i < e; ++i) {
                int64_t v = sp[i], beg = indptr[v];
                int64_t deg = indptr[v + 1] - beg;
                int64_t* dst = op + prefix[i];
                if (deg <= k) {
                    for (int64_t j = 0; j < deg; ++j) dst[j] = indices[beg + j];
                } else if (k <= 256 && deg > 2 * (int64_t)k) {
                    // Floyd's uniform k-subset: O(k^2), degree-independent
                    slot.clear();
                    for (int64_t j = deg - k; j < deg; ++j) {
                        int64_t t = std::uniform_int_distribution<int64_t>(
                            0, j)(gen);
                        bool found = std::find(slot.begin(), slot.end(), t) !=
                                     slot.end();
                        slot.push_back(found ? j : t);
                    }
                    for (int64_t j = 0; j < k; ++j)
                        dst[j] = indices[beg + slot[j]];
                } else {
                    slot.resize(k);
                    for (int64_t j = 0; j < k; ++j) slot[j] = j;
                    for (int64_t j = k; j < deg; ++j) {
                        int64_t r = std::uniform_int_distribution<int64_t>(
                            0, j)(gen);
                        if (r < k) slot[r] = j;
                    }
                    for (int64_t j = 0; j < k; ++j)
                        dst[j] = indices[beg + slot[j]];
                }
            }
        });
        return {out, counts};
    }

    std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> reindex_single(
        torch::Tensor seeds, torch::Tensor nbrs, torch::Tensor counts) {
        seeds = seeds.contiguous().cpu();
        nbrs = nbrs.contiguous().cpu();
        counts = counts.contiguous().cpu();
        int64_t n = seeds.numel(), m = nbrs.numel();
        const int64_t* sp = seeds.data_ptr<int64_t>();
        const int64_t* np = nbrs.data_ptr<int64_t>();
        const int64_t* cp = counts.data_ptr<int64_t>();
        std::unordered_map<int64_t, int64_t> table;
        table.reserve(2 * (n + m));
        std::vector<int64_t> frontier;
        frontier.reserve(n + m);
        for (int64_t i = 0; i < n; ++i) {
            auto [it, fresh] = table.emplace(sp[i], (int64_t)frontier.size());
            if (fresh) frontier.push_back(sp[i]);
        }
        auto col = torch::empty({m}, seeds.options());
        auto row = torch::empty({m}, seeds.options());
        int64_t* colp = col.data_ptr<int64_t>();
        int64_t* rowp = row.data_ptr<int64_t>();
        for (int64_t j = 0; j < m; ++j) {
            auto [it, fresh] = table.emplace(np[j], (int64_t)frontier.size());
            if (fresh) frontier.push_back(np[j]);
            colp[j] = it->second;
        }
        int64_t off = 0;
        for (int64_t i = 0; i < n; ++i)
            for (int64_t j = 0; j < cp[i]; ++j) rowp[off++] = i;
        TORCH_CHECK(off == m, "reindex: counts do not sum to neighbors");
        auto fr = torch::from_blob(frontier.data(),
                                   {(int64_t)frontier.size()},
                                   seeds.options())
                      .clone();
        return {fr, row, col};
    }

    int64_t node_count() const { return node_count_; }
    int64_t edge_count() const { return indices_.numel(); }

  private:
    torch::Tensor indptr_, indices_;
    int64_t node_count_;
};

// ---------------------------------------------------------------------------
// ShardTensor (parity: reference quiver_feature.cu:20-473), hipIpc + xGMI P2P
// ---------------------------------------------------------------------------
struct ShardItem {
    void* dptr = nullptr;     // device-visible base pointer
    int64_t rows = 0;
    int device = -1;          // -1 == pinned host
    bool owned_hip = false;   // free with hipFree
    bool from_ipc = false;    // close with hipIpcCloseMemHandle
    torch::Tensor keeper;     // keeps host/tensor storage alive
    std::vector<void*> host_regs;  // hipHostUnregister on destruction
};

class ShardTensorItem {
  public:
    int device = 0;
    int64_t rows = 0;
    std::vector<int64_t> row_shape;
    torch::Dtype dtype = torch::kFloat32;
    std::string handle;  // hipIpcMemHandle_t bytes

    py::tuple share_ipc() {
        return py::make_tuple(device, py::bytes(handle), rows,
                              row_shape, (int)dtype);
    }
    static ShardTensorItem from_ipc(py::tuple t) {
        ShardTensorItem it;
        it.device = t[0].cast<int>();
        it.handle = t[1].cast<std::string>();
        it.rows = t[2].cast<int64_t>();
        it.row_shape = t[3].cast<std::vector<int64_t>>();
        it.dtype = (torch::Dtype)t[4].cast<int>();
        return it;
    }
};

class ShardTensor {
  public:
    explicit ShardTensor(int current_device) : device_(current_device) {}

    ~ShardTensor() {
        for (auto& s : shards_) {
            if (s.owned_hip) {
                DeviceScope g(s.device);
                (void)hipFree(s.dptr);
            } else if (s.from_ipc) {
                DeviceScope g(device_);
                (void)hipIpcCloseMemHandle(s.dptr);
            }
            for (void* r : s.host_regs) (void)hipHostUnregister(r);
        }
    }

    void append(torch::Tensor tensor, int target_device) {
        tensor = tensor.contiguous();
        TORCH_CHECK(tensor.dim() >= 1, "need at least 1-d tensor");
        init_row_meta(tensor);
        ShardItem item;
        item.rows = tensor.size(0);
        item.device = target_device;
        size_t bytes = (size_t)tensor.numel() * tensor.element_size();
        if (target_device >= 0) {
            DeviceScope g(target_device);
            // raw hipMalloc (not the torch pool) so hipIpcGetMemHandle works
            // on the allocation base for cross-process sharing.
            QK_CHECK_HIP(hipMalloc(&item.dptr, bytes));
            if (tensor.device().is_cpu()) {
                QK_CHECK_HIP(hipMemcpy(item.dptr, tensor.data_ptr(), bytes,
                                       hipMemcpyHostToDevice));
            } else {
                QK_CHECK_HIP(hipMemcpy(item.dptr, tensor.data_ptr(), bytes,
                                       hipMemcpyDeviceToDevice));
            }
            item.owned_hip = true;
        } else {
            TORCH_CHECK(tensor.device().is_cpu(),
                        "host shard expects a CPU tensor");
            // zero-copy pinned host tier: register the tensor's own memory
            // (shared-memory friendly, 1 GB chunks) and read it from
            // gather kernels.
            item.dptr = register_host_chunked(tensor.data_ptr(), bytes,
                                              item.host_regs);
            item.keeper = tensor;
        }
        shards_.push_back(std::move(item));
    }

    // Adopt a shard allocated by another ShardTensor of THIS process
    // (hipIpcOpenMemHandle on a same-process handle is an error, so the
    // per-rank distributed build shares its own allocation by aliasing).
    // Ownership stays with `other`; the Python layer keeps it alive.
    void append_from(const ShardTensor& other, int idx) {
        TORCH_CHECK(idx >= 0 && idx < (int)other.shards_.size(),
                    "append_from: no such shard");
        const ShardItem& s = other.shards_[idx];
        init_row_meta_from(other.row_shape_, other.dtype_);
        ShardItem item;
        item.dptr = s.dptr;
        item.rows = s.rows;
        item.device = s.device;
        item.keeper = s.keeper;  // host-tier tensors stay alive either way
        shards_.push_back(std::move(item));
    }

    void append_item(const ShardTensorItem& it) {
        // open a peer shard exported from another process
        init_row_meta_from(it.row_shape, it.dtype);
        ShardItem item;
        item.rows = it.rows;
        item.device = it.device;
        item.from_ipc = true;
        DeviceScope g(device_);
        hipIpcMemHandle_t h;
        TORCH_CHECK(it.handle.size() == sizeof(h), "bad ipc handle");
        memcpy(&h, it.handle.data(), sizeof(h));
        QK_CHECK_HIP(hipIpcOpenMemHandle(&item.dptr, h,
                                         hipIpcMemLazyEnablePeerAccess));
        shards_.push_back(std::move(item));
    }

    std::vector<ShardTensorItem> share_ipc() {
        std::vector<ShardTensorItem> out;
        for (auto& s : shards_) {
            if (s.device < 0) continue;  // host tier travels via torch shm
            TORCH_CHECK(s.owned_hip,
                        "can only export shards this process allocated");
            ShardTensorItem it;
            it.device = s.device;
            it.rows = s.rows;
            it.row_shape = row_shape_;
            it.dtype = dtype_;
            hipIpcMemHandle_t h;
            DeviceScope g(s.device);
            QK_CHECK_HIP(hipIpcGetMemHandle(&h, s.dptr));
            it.handle.assign((char*)&h, sizeof(h));
            out.push_back(std::move(it));
        }
        return out;
    }

    torch::Tensor gather(torch::Tensor indices) {
        return gather_on(device_, indices);
    }

    // Run the gather with `dev` as the executing GPU (used by the
    // cross-clique fallback: remote device reads its own HBM, result is
    // copied back by the python layer).
    torch::Tensor gather_on(int dev, torch::Tensor indices) {
        return gather_impl(dev, indices, torch::Tensor());
    }

    // n_dev: 1-element int64 CUDA tensor holding the exact row count
    // (indices is upper-bound sized); enables fully async chains.
    torch::Tensor gather_n(torch::Tensor indices, torch::Tensor n_dev) {
        return gather_impl(device_, indices, n_dev);
    }

    torch::Tensor gather_impl(int dev, torch::Tensor indices,
                              torch::Tensor n_dev_t) {
        DeviceScope g(dev);
        indices = indices.contiguous();
        TORCH_CHECK(indices.device().is_cuda(), "indices must be on GPU");
        int64_t n = indices.numel();
        std::vector<int64_t> shape = {n};
        shape.insert(shape.end(), row_shape_.begin(), row_shape_.end());
        auto out = torch::empty(
            shape, torch::TensorOptions().dtype(dtype_).device(
                       torch::Device(torch::kCUDA, dev)));
        auto spec = build_spec(dev);
        auto cur = c10::hip::getCurrentHIPStream(dev);
        const int64_t* n_dev = n_dev_t.defined()
                                   ? n_dev_t.data_ptr<int64_t>()
                                   : nullptr;

        // Tier-split: a mixed gather would run at the capped host grid for
        // ALL rows (the cap keeps the PCIe-latency-bound zero-copy pass
        // from hogging CUs).  Split instead: pinned-host rows on the
        // current stream (long pole), HBM/xGMI rows concurrently on a side
        // stream at full grid.
        uint32_t host_mask = 0;
        for (int i = 0; i < spec.nshards; ++i)
            if (shards_[i].device < 0) host_mask |= (1u << i);
        host_mask &= spec.access_mask;
        uint32_t dev_mask = spec.access_mask & ~host_mask;

        if (host_mask != 0 && dev_mask != 0 && staged_wanted(n)) {
            // CPU-staged host tier: zero-copy kernel reads of scattered
            // 400 B rows top out near ~11 GB/s (64 B uncached PCIe reads,
            // latency-bound).  Instead: HBM/xGMI rows via the full-grid
            // kernel (async), indices D2H, CPU threads gather host rows
            // into pinned staging, ONE large H2D burst, then an
            // HBM-to-HBM scatter places them.  Large DMA bursts run at
            // full PCIe write bandwidth.
            qk::GatherSpec ds = spec;
            ds.access_mask = dev_mask;
            ds.has_host_shard = false;
            qk::launch_gather(cur.stream(), ds, indices.data_ptr<int64_t>(),
                              n, (char*)out.data_ptr(), n_dev);
            gather_host_staged(dev, cur.stream(), indices, n, spec,
                               host_mask, out, n_dev);
        } else if (host_mask != 0 && dev_mask != 0) {
            qk::GatherSpec hs = spec;
            hs.access_mask = host_mask;
            hs.has_host_shard = true;
            qk::GatherSpec ds = spec;
            ds.access_mask = dev_mask;
            ds.has_host_shard = false;
            auto side = c10::hip::getStreamFromPool(false, dev);
            auto& ev = split_events_[dev];  // events live on `dev`
            if (!ev.first) {
                QK_CHECK_HIP(hipEventCreateWithFlags(
                    &ev.first, hipEventDisableTiming));
                QK_CHECK_HIP(hipEventCreateWithFlags(
                    &ev.second, hipEventDisableTiming));
            }
            QK_CHECK_HIP(hipEventRecord(ev.first, cur.stream()));
            QK_CHECK_HIP(hipStreamWaitEvent(side.stream(), ev.first, 0));
            qk::launch_gather(cur.stream(), hs, indices.data_ptr<int64_t>(),
                              n, (char*)out.data_ptr(), n_dev);
            qk::launch_gather(side.stream(), ds, indices.data_ptr<int64_t>(),
                              n, (char*)out.data_ptr(), n_dev);
            QK_CHECK_HIP(hipEventRecord(ev.second, side.stream()));
            QK_CHECK_HIP(hipStreamWaitEvent(cur.stream(), ev.second, 0));
            // caching-allocator hazard: out/indices are used on `side`.
            // (Tensor::record_stream wants a masquerading-as-CUDA stream;
            // go to the HIP allocator directly.)  Under hipGraph capture
            // the tensors live in the graph pool with static lifetime —
            // skip the bookkeeping (recordStream during capture throws).
            hipStreamCaptureStatus cs = hipStreamCaptureStatusNone;
            (void)hipStreamIsCapturing(cur.stream(), &cs);
            if (cs == hipStreamCaptureStatusNone) {
                c10::hip::HIPCachingAllocator::recordStream(
                    out.storage().data_ptr(), side);
                c10::hip::HIPCachingAllocator::recordStream(
                    indices.storage().data_ptr(), side);
            }
        } else {
            qk::launch_gather(cur.stream(), spec,
                              indices.data_ptr<int64_t>(), n,
                              (char*)out.data_ptr(), n_dev);
        }
        return out;
    }

    void scatter_update(torch::Tensor indices, torch::Tensor src) {
        DeviceScope g(device_);
        indices = indices.contiguous();
        src = src.contiguous();
        auto spec = build_spec(device_);
        qk::launch_scatter(current_stream(), spec,
                           indices.data_ptr<int64_t>(), indices.numel(),
                           (const char*)src.data_ptr());
    }

    static bool staged_wanted(int64_t n) {
        static int mode = [] {
            const char* e = getenv("QUIVER_STAGED_GATHER");
            return e ? atoi(e) : -1;  // -1 = auto
        }();
        if (mode == 0) return false;
        if (mode == 1) return true;
        // default OFF.  Measured trade (profiles/timeline_analysis.md):
        // the zero-copy kernel's uncached PCIe reads slow concurrent
        // kernels 5-10x, but the staged path's CPU block + extra hops
        // cost more end-to-end (products step 5.33 ms staged vs 4.45
        // zero-copy; feature microbench 38 vs 71 GB/s even with the
        // persistent pool).  Opt in with QUIVER_STAGED_GATHER=1.
        (void)n;
        return false;
    }

    // CPU-staged gather of the pinned-host tier (see gather_on).
    void gather_host_staged(int dev, hipStream_t stream, torch::Tensor indices,
                            int64_t n, const qk::GatherSpec& spec,
                            uint32_t host_mask, torch::Tensor& out,
                            const int64_t* n_dev = nullptr) {
        const int64_t rb = row_bytes_;
        // host base pointer + global row range of every host shard
        struct HostShard { int64_t beg, end; const char* base; };
        std::vector<HostShard> hosts;
        for (int s = 0; s < spec.nshards; ++s) {
            if (!((host_mask >> s) & 1u)) continue;
            int64_t beg = s == 0 ? 0 : spec.ends[s - 1];
            hosts.push_back({beg, spec.ends[s],
                             (const char*)shards_[s].keeper.data_ptr()});
        }
        if (hosts.empty()) return;

        auto pin_opts = torch::TensorOptions()
                            .dtype(torch::kUInt8)
                            .device(torch::kCPU)
                            .pinned_memory(true);
        if (!idx_pin_.defined() || idx_pin_.numel() < n * 8 + 8)
            idx_pin_ = torch::empty({n * 8 + 8}, pin_opts);
        QK_CHECK_HIP(hipMemcpyAsync(idx_pin_.data_ptr(),
                                    indices.data_ptr<int64_t>(), n * 8,
                                    hipMemcpyDeviceToHost, stream));
        int64_t* pinned_n =
            (int64_t*)((char*)idx_pin_.data_ptr() + idx_pin_.numel() - 8);
        if (n_dev)
            QK_CHECK_HIP(hipMemcpyAsync(pinned_n, n_dev, 8,
                                        hipMemcpyDeviceToHost, stream));
        // waits for the D2H AND the already-launched device-tier kernel
        QK_CHECK_HIP(hipStreamSynchronize(stream));
        if (n_dev) n = std::min(n, *pinned_n);
        const int64_t* idx = (const int64_t*)idx_pin_.data_ptr();

        // parallel count -> chunk prefix -> parallel fill+copy
        const int64_t grain = 4096;
        const int64_t nchunk = (n + grain - 1) / grain;
        std::vector<int64_t> chunk_cnt(nchunk, 0);
        auto host_of = [&hosts](int64_t v) -> const HostShard* {
            for (const auto& h : hosts)
                if (v >= h.beg && v < h.end) return &h;
            return nullptr;
        };
        parallel_chunks(nchunk, 32, [&](int64_t c0, int64_t c1) {
            for (int64_t c = c0; c < c1; ++c) {
                int64_t cnt = 0;
                int64_t e = std::min(n, (c + 1) * grain);
                for (int64_t i = c * grain; i < e; ++i)
                    if (host_of(idx[i])) ++cnt;
                chunk_cnt[c] = cnt;
            }
        });
        std::vector<int64_t> chunk_off(nchunk + 1, 0);
        for (int64_t c = 0; c < nchunk; ++c)
            chunk_off[c + 1] = chunk_off[c] + chunk_cnt[c];
        const int64_t m = chunk_off[nchunk];
        if (m == 0) return;
        if (!pos_pin_.defined() || pos_pin_.numel() < m * 8)
            pos_pin_ = torch::empty({std::max(m, n) * 8}, pin_opts);
        if (!stage_pin_.defined() || stage_pin_.numel() < m * rb)
            stage_pin_ = torch::empty({std::max(m * rb, n * rb)}, pin_opts);
        int64_t* pos = (int64_t*)pos_pin_.data_ptr();
        char* stage = (char*)stage_pin_.data_ptr();
        parallel_chunks(nchunk, 64, [&](int64_t c0, int64_t c1) {
            for (int64_t c = c0; c < c1; ++c) {
                int64_t j = chunk_off[c];
                int64_t e = std::min(n, (c + 1) * grain);
                for (int64_t i = c * grain; i < e; ++i) {
                    const HostShard* h = host_of(idx[i]);
                    if (!h) continue;
                    memcpy(stage + j * rb, h->base + (idx[i] - h->beg) * rb,
                           rb);
                    pos[j++] = i;
                }
            }
        });

        // one big burst up + positions, then HBM->HBM scatter into `out`
        auto dev_opts = torch::TensorOptions()
                            .dtype(torch::kUInt8)
                            .device(torch::Device(torch::kCUDA, dev));
        auto stage_dev = torch::empty({m * rb}, dev_opts);
        auto pos_dev = torch::empty({m * 8}, dev_opts);
        QK_CHECK_HIP(hipMemcpyAsync(stage_dev.data_ptr(), stage, m * rb,
                                    hipMemcpyHostToDevice, stream));
        QK_CHECK_HIP(hipMemcpyAsync(pos_dev.data_ptr(), pos, m * 8,
                                    hipMemcpyHostToDevice, stream));
        qk::GatherSpec outspec{};
        outspec.ptrs[0] = (const char*)out.data_ptr();
        outspec.ends[0] = n;
        outspec.access_mask = 1u;
        outspec.nshards = 1;
        outspec.row_bytes = rb;
        outspec.has_host_shard = false;
        qk::launch_scatter(stream, outspec,
                           (const int64_t*)pos_dev.data_ptr(), m,
                           (const char*)stage_dev.data_ptr());
    }

    // Which shards are directly readable from `dev` (bit per shard).
    uint32_t access_mask_on(int dev) const {
        return build_spec(dev).access_mask;
    }
    std::vector<int64_t> shard_ends() const {
        std::vector<int64_t> ends;
        int64_t acc = 0;
        for (auto& s : shards_) {
            acc += s.rows;
            ends.push_back(acc);
        }
        return ends;
    }

    std::vector<int64_t> shape() const {
        std::vector<int64_t> s = {total_rows()};
        s.insert(s.end(), row_shape_.begin(), row_shape_.end());
        return s;
    }
    int64_t total_rows() const {
        int64_t t = 0;
        for (auto& s : shards_) t += s.rows;
        return t;
    }
    int64_t size(int dim) const { return shape().at(dim); }
    int device() const { return device_; }
    int shard_count() const { return (int)shards_.size(); }
    std::vector<int64_t> shard_rows() const {
        std::vector<int64_t> r;
        for (auto& s : shards_) r.push_back(s.rows);
        return r;
    }
    std::vector<int> shard_devices() const {
        std::vector<int> r;
        for (auto& s : shards_) r.push_back(s.device);
        return r;
    }

  private:
    void init_row_meta(const torch::Tensor& t) {
        std::vector<int64_t> rs(t.sizes().begin() + 1, t.sizes().end());
        init_row_meta_from(rs, t.scalar_type());
    }
    void init_row_meta_from(const std::vector<int64_t>& rs, torch::Dtype dt) {
        if (row_shape_init_) {
            TORCH_CHECK(rs == row_shape_ && dt == dtype_,
                        "all shards must share row shape and dtype");
            return;
        }
        row_shape_ = rs;
        dtype_ = dt;
        row_bytes_ = torch::elementSize(dt);
        for (auto d : rs) row_bytes_ *= d;
        row_shape_init_ = true;
    }

    qk::GatherSpec build_spec(int dev) const {
        TORCH_CHECK((int)shards_.size() <= qk::kMaxShards, "too many shards");
        qk::GatherSpec spec{};
        spec.nshards = (int)shards_.size();
        spec.row_bytes = row_bytes_;
        spec.access_mask = 0;
        spec.has_host_shard = false;
        int64_t acc = 0;
        DeviceScope g(dev);
        for (int i = 0; i < spec.nshards; ++i) {
            acc += shards_[i].rows;
            spec.ends[i] = acc;
            spec.ptrs[i] = (const char*)shards_[i].dptr;
            if (shards_[i].device < 0) spec.has_host_shard = true;
            bool ok = true;
            int sd = shards_[i].device;
            if (sd >= 0 && sd != dev) {
                int can = 0;
                QK_CHECK_HIP(hipDeviceCanAccessPeer(&can, dev, sd));
                if (can) {
                    // lazy xGMI peer enable (idempotent)
                    hipError_t err = hipDeviceEnablePeerAccess(sd, 0);
                    if (err != hipSuccess &&
                        err != hipErrorPeerAccessAlreadyEnabled)
                        QK_CHECK_HIP(err);
                    (void)hipGetLastError();
                }
                ok = can || shards_[i].from_ipc;
            }
            if (ok) spec.access_mask |= (1u << i);
        }
        return spec;
    }

    int device_;
    std::vector<ShardItem> shards_;
    std::vector<int64_t> row_shape_;
    torch::Dtype dtype_ = torch::kFloat32;
    int64_t row_bytes_ = 0;
    bool row_shape_init_ = false;
    // per-executing-device split/join events for the tier-split gather
    // (events must live on the device whose streams record them)
    mutable std::unordered_map<int, std::pair<hipEvent_t, hipEvent_t>>
        split_events_;
    // pinned buffers for the CPU-staged host-tier gather
    torch::Tensor idx_pin_, pos_pin_, stage_pin_;
};

// ---------------------------------------------------------------------------
// Fused SAGE-mean aggregation over dst-sorted edges (segment_kernels.hip)
// ---------------------------------------------------------------------------
torch::Tensor segment_mean_gather(torch::Tensor x, torch::Tensor src,
                                  torch::Tensor dst_ptr) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 2 &&
                (x.dtype() == torch::kFloat32 ||
                 x.dtype() == torch::kBFloat16));
    x = x.contiguous();
    src = src.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    auto out = torch::empty({n_dst, x.size(1)}, x.options());
    if (x.dtype() == torch::kBFloat16) {
        TORCH_CHECK(x.size(1) % 2 == 0,
                    "bf16 segment mean needs an even feature dim");
        qk::launch_segment_mean_fwd_bf16(
            current_stream(), x.data_ptr(), src.data_ptr<int64_t>(),
            dst_ptr.data_ptr<int64_t>(), n_dst, x.size(1), out.data_ptr());
        return out;
    }
    qk::launch_segment_mean_fwd(current_stream(), x.data_ptr<float>(),
                                src.data_ptr<int64_t>(),
                                dst_ptr.data_ptr<int64_t>(), n_dst, x.size(1),
                                out.data_ptr<float>());
    return out;
}

torch::Tensor segment_mean_gather_backward(torch::Tensor grad_out,
                                           torch::Tensor src,
                                           torch::Tensor dst_ptr,
                                           int64_t n_src) {
    grad_out = grad_out.contiguous();
    src = src.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    auto grad_x = torch::zeros({n_src, grad_out.size(1)}, grad_out.options());
    if (grad_out.dtype() == torch::kBFloat16) {
        TORCH_CHECK(grad_out.size(1) % 2 == 0,
                    "bf16 segment mean needs an even feature dim");
        qk::launch_segment_mean_bwd_bf16(
            current_stream(), grad_out.data_ptr(), src.data_ptr<int64_t>(),
            dst_ptr.data_ptr<int64_t>(), n_dst, grad_out.size(1),
            grad_x.data_ptr());
        return grad_x;
    }
    qk::launch_segment_mean_bwd(current_stream(),
                                grad_out.data_ptr<float>(),
                                src.data_ptr<int64_t>(),
                                dst_ptr.data_ptr<int64_t>(), n_dst,
                                grad_out.size(1), grad_x.data_ptr<float>());
    return grad_x;
}

// Tall-skinny weight gradient: (A^T @ B, optional column-sums of A).
// A [K×M] (grad_out), B [K×N] (layer input), K = frontier size.
// Deterministic: split-K partials land in a workspace (torch pool
// scratch) and are reduced in fixed order.
std::tuple<torch::Tensor, torch::Tensor> wgrad(torch::Tensor a,
                                               torch::Tensor b,
                                               bool want_bias) {
    TORCH_CHECK(a.is_cuda() && b.is_cuda() && a.dim() == 2 && b.dim() == 2 &&
                a.size(0) == b.size(0) &&
                a.dtype() == torch::kFloat32 &&
                b.dtype() == torch::kFloat32,
                "wgrad: fp32 CUDA 2-D tensors with matching K expected");
    a = a.contiguous();
    b = b.contiguous();
    int64_t k = a.size(0);
    int m = (int)a.size(1), n = (int)b.size(1);
    auto c = torch::empty({m, n}, a.options());
    auto bias = want_bias ? torch::empty({m}, a.options()) : torch::Tensor();
    auto plan = qk::wgrad_plan(k, m, n);
    auto ws = torch::empty({plan.ws_floats}, a.options());
    qk::launch_wgrad(current_stream(), a.data_ptr<float>(),
                     b.data_ptr<float>(), c.data_ptr<float>(),
                     want_bias ? bias.data_ptr<float>() : nullptr, k, m, n,
                     plan, ws.data_ptr<float>());
    return {c, bias};
}

// Weighted segment sum (GAT): out[d,h*C+c] = sum_e w[e,h]*x[src[e],h*C+c]
torch::Tensor segment_wsum(torch::Tensor x, torch::Tensor w,
                           torch::Tensor src, torch::Tensor dst_ptr,
                           int64_t heads) {
    TORCH_CHECK(x.is_cuda() && x.dim() == 2 &&
                (x.dtype() == torch::kFloat32 ||
                 x.dtype() == torch::kBFloat16));
    TORCH_CHECK(w.is_cuda() && w.dtype() == torch::kFloat32 && w.dim() == 2 &&
                w.size(1) == heads);
    x = x.contiguous();
    w = w.contiguous();
    src = src.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    int chead = (int)(x.size(1) / heads);
    TORCH_CHECK((int64_t)chead * heads == x.size(1), "dim % heads != 0");
    auto out = torch::empty({n_dst, x.size(1)}, x.options());
    if (x.dtype() == torch::kBFloat16) {
        qk::launch_segment_wsum_fwd_bf16(
            current_stream(), x.data_ptr(), w.data_ptr<float>(),
            src.data_ptr<int64_t>(), dst_ptr.data_ptr<int64_t>(), n_dst,
            (int)heads, chead, out.data_ptr());
        return out;
    }
    qk::launch_segment_wsum_fwd(current_stream(), x.data_ptr<float>(),
                                w.data_ptr<float>(), src.data_ptr<int64_t>(),
                                dst_ptr.data_ptr<int64_t>(), n_dst,
                                (int)heads, chead, out.data_ptr<float>());
    return out;
}

std::tuple<torch::Tensor, torch::Tensor> segment_wsum_backward(
    torch::Tensor grad_out, torch::Tensor x, torch::Tensor w,
    torch::Tensor src, torch::Tensor dst_ptr, int64_t heads,
    bool need_gx, bool need_gw) {
    grad_out = grad_out.contiguous();
    x = x.contiguous();
    w = w.contiguous();
    src = src.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    int chead = (int)(x.size(1) / heads);
    const bool bf16 = x.dtype() == torch::kBFloat16;
    torch::Tensor gx, gw;
    if (need_gx) {
        gx = torch::zeros_like(x);
        if (bf16) {
            TORCH_CHECK(((int64_t)chead * heads) % 2 == 0,
                        "bf16 wsum needs an even dim");
            qk::launch_segment_wsum_bwd_x_bf16(
                current_stream(), grad_out.data_ptr(), w.data_ptr<float>(),
                src.data_ptr<int64_t>(), dst_ptr.data_ptr<int64_t>(), n_dst,
                (int)heads, chead, gx.data_ptr());
        } else {
            qk::launch_segment_wsum_bwd_x(
                current_stream(), grad_out.data_ptr<float>(),
                w.data_ptr<float>(), src.data_ptr<int64_t>(),
                dst_ptr.data_ptr<int64_t>(), n_dst, (int)heads, chead,
                gx.data_ptr<float>());
        }
    }
    if (need_gw) {
        gw = torch::empty_like(w);
        if (bf16) {
            qk::launch_segment_wsum_bwd_w_bf16(
                current_stream(), grad_out.data_ptr(), x.data_ptr(),
                src.data_ptr<int64_t>(), dst_ptr.data_ptr<int64_t>(), n_dst,
                (int)heads, chead, gw.data_ptr<float>());
        } else {
            qk::launch_segment_wsum_bwd_w(
                current_stream(), grad_out.data_ptr<float>(),
                x.data_ptr<float>(), src.data_ptr<int64_t>(),
                dst_ptr.data_ptr<int64_t>(), n_dst, (int)heads, chead,
                gw.data_ptr<float>());
        }
    }
    return {gx, gw};
}

// Numerically-stable segment softmax over [E, H] (dst_ptr segments).
torch::Tensor segment_softmax(torch::Tensor a, torch::Tensor dst_ptr,
                              int64_t heads) {
    TORCH_CHECK(a.is_cuda() && a.dtype() == torch::kFloat32 && a.dim() == 2 &&
                a.size(1) == heads);
    a = a.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    auto out = torch::empty_like(a);
    qk::launch_segment_softmax_fwd(current_stream(), a.data_ptr<float>(),
                                   dst_ptr.data_ptr<int64_t>(), n_dst,
                                   (int)heads, out.data_ptr<float>());
    return out;
}

torch::Tensor segment_softmax_backward(torch::Tensor grad_out,
                                       torch::Tensor out,
                                       torch::Tensor dst_ptr, int64_t heads) {
    grad_out = grad_out.contiguous();
    out = out.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    auto ga = torch::empty_like(out);
    qk::launch_segment_softmax_bwd(current_stream(),
                                   grad_out.data_ptr<float>(),
                                   out.data_ptr<float>(),
                                   dst_ptr.data_ptr<int64_t>(), n_dst,
                                   (int)heads, ga.data_ptr<float>());
    return ga;
}

// Fused GAT attention coefficients (gather + add + leaky_relu + segment
// softmax in one pass; no [E,H] intermediates materialized by torch).
torch::Tensor gat_alpha(torch::Tensor asrc, torch::Tensor adst,
                        torch::Tensor src, torch::Tensor dst_ptr,
                        int64_t heads, double slope, int64_t n_edges) {
    TORCH_CHECK(asrc.is_cuda() && asrc.dtype() == torch::kFloat32 &&
                adst.dtype() == torch::kFloat32 &&
                asrc.size(1) == heads && adst.size(1) == heads);
    asrc = asrc.contiguous();
    adst = adst.contiguous();
    src = src.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    auto alpha = torch::empty({n_edges, heads}, asrc.options());
    qk::launch_gat_alpha_fwd(current_stream(), asrc.data_ptr<float>(),
                             adst.data_ptr<float>(),
                             src.data_ptr<int64_t>(),
                             dst_ptr.data_ptr<int64_t>(), n_dst, (int)heads,
                             (float)slope, alpha.data_ptr<float>());
    return alpha;
}

std::tuple<torch::Tensor, torch::Tensor> gat_alpha_backward(
    torch::Tensor grad_alpha, torch::Tensor alpha, torch::Tensor asrc,
    torch::Tensor adst, torch::Tensor src, torch::Tensor dst_ptr,
    int64_t heads, double slope) {
    grad_alpha = grad_alpha.contiguous();
    alpha = alpha.contiguous();
    asrc = asrc.contiguous();
    adst = adst.contiguous();
    src = src.contiguous();
    dst_ptr = dst_ptr.contiguous();
    int64_t n_dst = dst_ptr.numel() - 1;
    auto g_asrc = torch::zeros_like(asrc);
    auto g_adst = torch::zeros_like(adst);
    qk::launch_gat_alpha_bwd(
        current_stream(), grad_alpha.data_ptr<float>(),
        alpha.data_ptr<float>(), asrc.data_ptr<float>(),
        adst.data_ptr<float>(), src.data_ptr<int64_t>(),
        dst_ptr.data_ptr<int64_t>(), n_dst, (int)heads, (float)slope,
        g_asrc.data_ptr<float>(), g_adst.data_ptr<float>());
    return {g_asrc, g_adst};
}

// Tall-M GEMM: out[M×N] = a[M×K] @ b_kmajor[K×N] (+bias).  The caller
// passes W^T (tiny one-off transpose) for a forward linear and W itself
// for the data-grad, so both read coalesced.
torch::Tensor tall_gemm(torch::Tensor a, torch::Tensor b_kmajor,
                        c10::optional<torch::Tensor> bias_opt) {
    torch::Tensor bias =
        bias_opt.has_value() ? *bias_opt : torch::Tensor();
    TORCH_CHECK(a.is_cuda() && a.dim() == 2 &&
                a.dtype() == torch::kFloat32 &&
                b_kmajor.dtype() == torch::kFloat32 &&
                a.size(1) == b_kmajor.size(0),
                "tall_gemm: fp32 [M,K] @ [K,N] expected");
    a = a.contiguous();
    b_kmajor = b_kmajor.contiguous();
    const int64_t m = a.size(0);
    const int k = (int)a.size(1), n = (int)b_kmajor.size(1);
    auto c = torch::empty({m, (int64_t)n}, a.options());
    const float* bp = nullptr;
    if (bias.defined() && bias.numel() > 0) {
        bias = bias.contiguous();
        TORCH_CHECK(bias.numel() == n, "bias size mismatch");
        bp = bias.data_ptr<float>();
    }
    qk::launch_tall_gemm(current_stream(), a.data_ptr<float>(),
                         b_kmajor.data_ptr<float>(), bp,
                         c.data_ptr<float>(), m, k, n);
    return c;
}

// Fused GAT attention dots (see segment_kernels.hip).
std::tuple<torch::Tensor, torch::Tensor> gat_dots(torch::Tensor h,
                                                  torch::Tensor att_src,
                                                  torch::Tensor att_dst,
                                                  int64_t n_dst,
                                                  int64_t heads) {
    TORCH_CHECK(h.is_cuda() && h.dim() == 2 &&
                (h.dtype() == torch::kFloat32 ||
                 h.dtype() == torch::kBFloat16));
    TORCH_CHECK(att_src.dtype() == torch::kFloat32 &&
                att_dst.dtype() == torch::kFloat32,
                "attention vectors stay fp32 (logits are fp32)");
    h = h.contiguous();
    att_src = att_src.contiguous();
    att_dst = att_dst.contiguous();
    const int64_t n = h.size(0);
    const int chead = (int)(h.size(1) / heads);
    TORCH_CHECK((int64_t)chead * heads == h.size(1), "dim % heads != 0");
    TORCH_CHECK(att_src.numel() == h.size(1) &&
                att_dst.numel() == h.size(1));
    auto fopts = h.options().dtype(torch::kFloat32);
    auto asrc = torch::empty({n, heads}, fopts);
    auto adst = torch::empty({n_dst, heads}, fopts);
    qk::launch_gat_dots_fwd(current_stream(), h.data_ptr(),
                            h.dtype() == torch::kBFloat16,
                            att_src.data_ptr<float>(),
                            att_dst.data_ptr<float>(), n, n_dst, (int)heads,
                            chead, asrc.data_ptr<float>(),
                            adst.data_ptr<float>());
    return {asrc, adst};
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> gat_dots_backward(
    torch::Tensor h, torch::Tensor att_src, torch::Tensor att_dst,
    torch::Tensor g_asrc, torch::Tensor g_adst, int64_t heads) {
    h = h.contiguous();
    att_src = att_src.contiguous();
    att_dst = att_dst.contiguous();
    g_asrc = g_asrc.contiguous();
    g_adst = g_adst.contiguous();
    const int64_t n = h.size(0), n_dst = g_adst.size(0);
    const int chead = (int)(h.size(1) / heads);
    auto g_h = torch::empty_like(h);
    auto g_as = torch::zeros_like(att_src);
    auto g_ad = torch::zeros_like(att_dst);
    qk::launch_gat_dots_bwd(
        current_stream(), h.data_ptr(), h.dtype() == torch::kBFloat16,
        att_src.data_ptr<float>(), att_dst.data_ptr<float>(),
        g_asrc.data_ptr<float>(), g_adst.data_ptr<float>(), n, n_dst,
        (int)heads, chead, g_h.data_ptr(), g_as.data_ptr<float>(),
        g_ad.data_ptr<float>());
    return {g_h, g_as, g_ad};
}

void init_p2p(const std::vector<int>& devices) {
    // On an 8x MI355X node every pair is xGMI-connected: enable the full
    // clique (reference init_p2p, quiver_feature.cu:378-421; no NVLink-style
    // two-clique hardcode here).
    for (int a : devices) {
        DeviceScope g(a);
        for (int b : devices) {
            if (a == b) continue;
            int can = 0;
            QK_CHECK_HIP(hipDeviceCanAccessPeer(&can, a, b));
            if (can) {
                hipError_t err = hipDeviceEnablePeerAccess(b, 0);
                if (err != hipSuccess &&
                    err != hipErrorPeerAccessAlreadyEnabled)
                    QK_CHECK_HIP(err);
                (void)hipGetLastError();
            }
        }
    }
}

bool can_device_access_peer(int a, int b) {
    if (a == b) return true;
    int can = 0;
    QK_CHECK_HIP(hipDeviceCanAccessPeer(&can, a, b));
    return can != 0;
}

// ---------------------------------------------------------------------------
// RCCL communicator (parity: reference NcclComm, quiver_comm.cu:9-100)
// ---------------------------------------------------------------------------
py::bytes create_nccl_id() {
    ncclUniqueId id;
    QK_CHECK_RCCL(ncclGetUniqueId(&id));
    return py::bytes((const char*)&id, sizeof(id));
}

class RcclComm {
  public:
    RcclComm(int rank, int ws, py::bytes id_bytes) : rank_(rank), ws_(ws) {
        std::string s = id_bytes;
        TORCH_CHECK(s.size() == sizeof(ncclUniqueId), "bad rccl id");
        ncclUniqueId id;
        memcpy(&id, s.data(), sizeof(id));
        QK_CHECK_RCCL(ncclCommInitRank(&comm_, ws_, id, rank_));
    }
    ~RcclComm() {
        if (comm_) (void)ncclCommDestroy(comm_);
    }

    int rank() const { return rank_; }
    int size() const { return ws_; }

    void send(torch::Tensor t, int dst) {
        QK_CHECK_RCCL(ncclSend(t.data_ptr(), t.numel(), dtype_of(t), dst,
                               comm_, current_stream()));
    }
    void recv(torch::Tensor t, int src) {
        QK_CHECK_RCCL(ncclRecv(t.data_ptr(), t.numel(), dtype_of(t), src,
                               comm_, current_stream()));
    }
    void allreduce(torch::Tensor t) {
        QK_CHECK_RCCL(ncclAllReduce(t.data_ptr(), t.data_ptr(), t.numel(),
                                    dtype_of(t), ncclSum, comm_,
                                    current_stream()));
    }
    void allgather(torch::Tensor src, torch::Tensor dst) {
        QK_CHECK_RCCL(ncclAllGather(src.data_ptr(), dst.data_ptr(),
                                    src.numel(), dtype_of(src), comm_,
                                    current_stream()));
    }
    void alltoall(torch::Tensor src, torch::Tensor dst) {
        int64_t per = src.numel() / ws_;
        size_t esz = src.element_size();
        QK_CHECK_RCCL(ncclGroupStart());
        for (int r = 0; r < ws_; ++r) {
            QK_CHECK_RCCL(ncclSend((char*)src.data_ptr() + r * per * esz, per,
                                   dtype_of(src), r, comm_, current_stream()));
            QK_CHECK_RCCL(ncclRecv((char*)dst.data_ptr() + r * per * esz, per,
                                   dtype_of(dst), r, comm_, current_stream()));
        }
        QK_CHECK_RCCL(ncclGroupEnd());
    }
    void group_start() { QK_CHECK_RCCL(ncclGroupStart()); }
    void group_end() { QK_CHECK_RCCL(ncclGroupEnd()); }

  private:
    static ncclDataType_t dtype_of(const torch::Tensor& t) {
        switch (t.scalar_type()) {
            case torch::kFloat32: return ncclFloat32;
            case torch::kFloat16: return ncclFloat16;
            case torch::kBFloat16: return ncclBfloat16;
            case torch::kInt64: return ncclInt64;
            case torch::kInt32: return ncclInt32;
            case torch::kUInt8: return ncclUint8;
            default: TORCH_CHECK(false, "unsupported dtype for rccl");
        }
    }
    int rank_, ws_;
    ncclComm_t comm_ = nullptr;
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "MI355X-native GNN sampling / feature-collection engine";

    py::class_<GpuSampler>(m, "Quiver")
        .def("sample_neighbor", &GpuSampler::sample_neighbor,
             py::call_guard<py::gil_scoped_release>())
        .def("sample_hops", &GpuSampler::sample_hops,
             py::call_guard<py::gil_scoped_release>(),
             "fused multi-hop sample+reindex, one sync per batch")
        .def("sample_hops_raw", &GpuSampler::sample_hops_raw,
             py::call_guard<py::gil_scoped_release>(),
             "zero-sync variant: ub-sized tensors + device sizes vector")
        .def("reindex_single", &GpuSampler::reindex_single,
             py::call_guard<py::gil_scoped_release>())
        .def("cal_neighbor_prob", &GpuSampler::cal_neighbor_prob,
             py::call_guard<py::gil_scoped_release>())
        .def("set_seed", &GpuSampler::set_seed)
        .def("node_count", &GpuSampler::node_count)
        .def("edge_count", &GpuSampler::edge_count)
        .def("device", &GpuSampler::device);

    m.def("device_quiver_from_csr_array",
          [](torch::Tensor indptr, torch::Tensor indices, torch::Tensor eid,
             int device, bool dma) {
              return new GpuSampler(indptr, indices, eid, device, dma);
          },
          py::return_value_policy::take_ownership);

    py::class_<CpuSampler>(m, "CPUQuiver")
        .def("sample_neighbor", &CpuSampler::sample_neighbor,
             py::call_guard<py::gil_scoped_release>())
        .def("reindex_single", &CpuSampler::reindex_single,
             py::call_guard<py::gil_scoped_release>())
        .def("node_count", &CpuSampler::node_count)
        .def("edge_count", &CpuSampler::edge_count);

    m.def("cpu_quiver_from_csr_array",
          [](torch::Tensor indptr, torch::Tensor indices) {
              return new CpuSampler(indptr, indices);
          },
          py::return_value_policy::take_ownership);

    py::class_<ShardTensorItem>(m, "ShardTensorItem")
        .def(py::init<>())
        .def("share_ipc", &ShardTensorItem::share_ipc)
        .def_static("from_ipc", &ShardTensorItem::from_ipc)
        .def_readwrite("device", &ShardTensorItem::device);

    py::class_<ShardTensor>(m, "ShardTensor")
        .def(py::init<int>())
        .def("append", &ShardTensor::append)
        .def("append_item", &ShardTensor::append_item)
        .def("append_from", &ShardTensor::append_from,
             "alias a shard owned by another same-process ShardTensor")
        .def("share_ipc", &ShardTensor::share_ipc)
        .def("__getitem__", &ShardTensor::gather,
             py::call_guard<py::gil_scoped_release>())
        .def("gather", &ShardTensor::gather,
             py::call_guard<py::gil_scoped_release>())
        .def("gather_on", &ShardTensor::gather_on,
             py::call_guard<py::gil_scoped_release>())
        .def("gather_n", &ShardTensor::gather_n,
             py::call_guard<py::gil_scoped_release>())
        .def("access_mask_on", &ShardTensor::access_mask_on)
        .def("shard_ends", &ShardTensor::shard_ends)
        .def("scatter_update", &ShardTensor::scatter_update,
             py::call_guard<py::gil_scoped_release>())
        .def("shape", &ShardTensor::shape)
        .def("size", &ShardTensor::size)
        .def("device", &ShardTensor::device)
        .def("shard_count", &ShardTensor::shard_count)
        .def("shard_rows", &ShardTensor::shard_rows)
        .def("shard_devices", &ShardTensor::shard_devices);

    m.def("init_p2p", &init_p2p);
    m.def("can_device_access_peer", &can_device_access_peer);

    m.def("segment_mean_gather", &segment_mean_gather,
          py::call_guard<py::gil_scoped_release>());
    m.def("segment_mean_gather_backward", &segment_mean_gather_backward,
          py::call_guard<py::gil_scoped_release>());

    m.def("tall_gemm", &tall_gemm,
          "C[M,N] = A[M,K] @ B[K,N] (+bias), M huge, MFMA 32x32 tiles");
    m.def("gat_dots", &gat_dots,
          "fused per-head attention dots over a projected frontier");
    m.def("gat_dots_backward", &gat_dots_backward);
    m.def("gat_alpha", &gat_alpha,
          "fused GAT attention coefficients over dst-sorted edges");
    m.def("gat_alpha_backward", &gat_alpha_backward);
    m.def("segment_softmax", &segment_softmax,
          "numerically-stable softmax over dst-sorted edge segments");
    m.def("segment_softmax_backward", &segment_softmax_backward);
    m.def("segment_wsum", &segment_wsum,
          "weighted segment sum over dst-sorted edges (GAT aggregation)");
    m.def("segment_wsum_backward", &segment_wsum_backward);
    m.def("wgrad", &wgrad,
          "tall-skinny A^T@B (+ optional A column sums) via split-K atomics");
    m.def("create_nccl_id", &create_nccl_id);
    py::class_<RcclComm>(m, "NcclComm")
        .def(py::init<int, int, py::bytes>())
        .def("rank", &RcclComm::rank)
        .def("size", &RcclComm::size)
        .def("send", &RcclComm::send)
        .def("recv", &RcclComm::recv)
        .def("allreduce", &RcclComm::allreduce)
        .def("allgather", &RcclComm::allgather)
        .def("alltoall", &RcclComm::alltoall)
        .def("group_start", &RcclComm::group_start)
        .def("group_end", &RcclComm::group_end);
}
