// ddstore_amd native store classes -- public C++ API.
//
// The reference exposes `class DDStore` as a C++ library (include/ddstore.hpp,
// used by test/demo.cxx); this header is the MI355X-native equivalent:
// `ddstore::DeviceStore` (HBM shards + hipIpc peers + CDNA4 gather kernels)
// and `ddstore::HostStore` (POSIX-shm CPU path), both usable from C++
// directly (see tools/demo_native.cpp) and bound to Python in
// ddstore_core.hip. Requires torch headers (tensors are the argument type).
#pragma once

#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>
#include <ATen/Parallel.h>

#include <fcntl.h>
#include <sys/mman.h>
#include <unistd.h>

#if defined(__x86_64__)
#include <emmintrin.h>  // SSE2 non-temporal stores (x86-64 baseline)
#endif

#include <atomic>
#include <condition_variable>
#include <cstdlib>
#include <cstring>
#include <exception>
#include <functional>
#include <map>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "ddstore_kernels.h"

// roctx ranges so rocprofv3 --marker-trace attributes store phases
// (the reference has no tracing at all)
#include <roctracer/roctx.h>

namespace ddstore {

struct RoctxRange {
    explicit RoctxRange(const char* name) { roctxRangePushA(name); }
    ~RoctxRange() { roctxRangePop(); }
};

#define HIP_CHECK(expr)                                                        \
    do {                                                                       \
        hipError_t _e = (expr);                                                \
        TORCH_CHECK(_e == hipSuccess, "ddstore HIP error: ",                   \
                    hipGetErrorString(_e), " at " #expr);                      \
    } while (0)

inline int dds_type_of(const at::Tensor& t) {
    switch (t.scalar_type()) {
        case at::kByte: return DDS_U8;
        case at::kBool: return DDS_U8;  // stored/moved as u8 (0/1), NumPy layout
        case at::kInt: return DDS_I32;
        case at::kLong: return DDS_I64;
        case at::kFloat: return DDS_F32;
        case at::kDouble: return DDS_F64;
        case at::kHalf: return DDS_F16;
        case at::kBFloat16: return DDS_BF16;
        case at::kFloat8_e4m3fn: return DDS_F8E4M3;  // OCP fn, matches gfx950
        case at::kFloat8_e5m2: return DDS_F8E5M2;
        default:
            TORCH_CHECK(false, "ddstore: unsupported dtype ", t.scalar_type());
    }
}

inline int64_t dds_itemsize(int t) {
    switch (t) {
        case DDS_U8: case DDS_F8E4M3: case DDS_F8E5M2: return 1;
        case DDS_F16: case DDS_BF16: return 2;
        case DDS_I32: case DDS_F32: return 4;
        default: return 8;
    }
}

inline std::vector<int64_t> make_prefix(const std::vector<int64_t>& counts) {
    std::vector<int64_t> prefix(counts.size() + 1, 0);
    for (size_t i = 0; i < counts.size(); ++i) prefix[i + 1] = prefix[i] + counts[i];
    return prefix;
}

// Parallel-for for the HostStore hot loops. at::parallel_for from an
// extension compiled without the wheel's OpenMP backend runs SEQUENTIALLY
// (measured: 9 GB/s vs torch index_select's 57 on identical shm data, r2).
// A spawn-per-call std::thread version fixed the 8-core container but
// collapsed on the 128-thread GPU-box host (127 thread spawns ~ the whole
// gather); this PERSISTENT pool parks at::get_num_threads()-1 workers on a
// condition variable and hands out ranges via an atomic cursor. Worker
// exceptions are rethrown in the caller (TORCH_CHECK-safe); the singleton
// is re-created after fork (pid check) and intentionally leaked.
class HostPool {
public:
    static HostPool& inst() {
        static std::mutex m;
        static HostPool* p = nullptr;
        static pid_t pid = 0;
        std::lock_guard<std::mutex> l(m);
        if (p == nullptr || pid != getpid()) {
            p = new HostPool();  // leaked: no teardown-order races at exit
            pid = getpid();
        }
        return *p;
    }

    void run(int64_t begin, int64_t end, int64_t step,
             const std::function<void(int64_t, int64_t)>& f) {
        // one task at a time: a second caller (e.g. another Python thread
        // gathering concurrently) must not overwrite the live task state
        std::lock_guard<std::mutex> excl(run_mx_);
        std::unique_lock<std::mutex> lk(m_);
        next_.store(begin, std::memory_order_relaxed);
        end_ = end;
        step_ = step;
        fn_ = &f;
        err_ = nullptr;
        active_ = (int)workers_.size();
        ++gen_;
        cv_work_.notify_all();
        lk.unlock();
        work();  // the caller participates
        lk.lock();
        cv_done_.wait(lk, [&] { return active_ == 0; });
        fn_ = nullptr;
        if (err_) std::rethrow_exception(err_);
    }

private:
    HostPool() {
        const int n = std::max(1, (int)at::get_num_threads()) - 1;
        workers_.reserve((size_t)n);
        for (int i = 0; i < n; ++i)
            workers_.emplace_back([this] { worker_loop(); });
    }

    void worker_loop() {
        uint64_t seen = 0;
        for (;;) {
            std::unique_lock<std::mutex> lk(m_);
            cv_work_.wait(lk, [&] { return gen_ != seen; });
            seen = gen_;
            lk.unlock();
            work();
            lk.lock();
            if (--active_ == 0) cv_done_.notify_all();
        }
    }

    void work() {
        try {
            for (;;) {
                const int64_t b = next_.fetch_add(step_);
                if (b >= end_) break;
                (*fn_)(b, std::min(b + step_, end_));
            }
        } catch (...) {
            std::lock_guard<std::mutex> l(em_);
            if (!err_) err_ = std::current_exception();
        }
    }

    std::mutex run_mx_, m_, em_;
    std::condition_variable cv_work_, cv_done_;
    std::vector<std::thread> workers_;
    std::atomic<int64_t> next_{0};
    int64_t end_ = 0, step_ = 1;
    const std::function<void(int64_t, int64_t)>* fn_ = nullptr;
    std::exception_ptr err_;
    uint64_t gen_ = 0;
    int active_ = 0;
};

template <typename F>
inline void host_parallel_for(int64_t begin, int64_t end, int64_t grain,
                              const F& f) {
    const int64_t n = end - begin;
    if (n <= 0) return;
    if (grain < 1) grain = 1;
    const int maxt = std::max(1, (int)at::get_num_threads());
    if (n <= grain || maxt <= 1) {
        f(begin, end);
        return;
    }
    const int64_t step = std::max<int64_t>(grain, n / (8 * maxt));
    std::function<void(int64_t, int64_t)> fn(std::cref(f));
    HostPool::inst().run(begin, end, step, fn);
}

// Row copy for the host gather path: the destination is written once and
// never re-read, so bypass the cache with SSE2 non-temporal stores where
// alignment allows (saves the read-for-ownership write traffic; measured
// against plain memcpy on the 8-core GPU-box host). Caller issues one
// _mm_sfence after the batch.
inline bool host_nt_enabled() {
    static const bool s = [] {
        const char* e = std::getenv("DDSTORE_HOST_NT");
        return e == nullptr || e[0] != '0';  // default on
    }();
    return s;
}

inline int host_prefetch_dist() {
    static const int s = [] {
        const char* e = std::getenv("DDSTORE_HOST_PF");
        return e ? atoi(e) : 8;
    }();
    return s;
}

inline void copy_row_nt(char* dst, const char* src, size_t nb) {
#if defined(__x86_64__)
    if (host_nt_enabled() && nb % 16 == 0 && ((uintptr_t)dst & 15) == 0) {
        for (size_t o = 0; o < nb; o += 16) {
            __m128i v = _mm_loadu_si128(reinterpret_cast<const __m128i*>(src + o));
            _mm_stream_si128(reinterpret_cast<__m128i*>(dst + o), v);
        }
        return;
    }
#endif
    std::memcpy(dst, src, nb);
}

inline void copy_rows_fence() {
#if defined(__x86_64__)
    _mm_sfence();
#endif
}

#ifndef MADV_COLLAPSE
#define MADV_COLLAPSE 25  // kernel >= 6.1; harmless EINVAL on older kernels
#endif

// Synchronously collapse a populated shm mapping to transparent huge pages
// (MADV_COLLAPSE works independent of the shmem THP policy, which boxes
// commonly leave at "never"): the random host gather is 4-KB-page TLB-bound
// on many-core hosts -- measured 251 GB/s vs 1.78 TB/s for the same gather
// from THP-backed malloc memory. Best-effort (errors ignored), one-time at
// registration; DDSTORE_HOST_THP=0 disables.
inline void try_collapse_hugepages(void* p, size_t bytes) {
    static const bool on = [] {
        const char* e = std::getenv("DDSTORE_HOST_THP");
        return e == nullptr || e[0] != '0';
    }();
    if (on && bytes >= (size_t)2 << 20) (void)madvise(p, bytes, MADV_COLLAPSE);
}

// Host-side owner lookup over the prefix directory (binary search; the
// reference's linear `sortedsearch` is src/ddstore.cxx:5-17).
inline int owner_of_host(const std::vector<int64_t>& prefix, int64_t row) {
    int lo = 0, hi = (int)prefix.size() - 2;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (prefix[mid] <= row) lo = mid; else hi = mid - 1;
    }
    return lo;
}

// DDSTORE_STRICT=1: out-of-range / over-capacity samples raise at the next
// host sync point (epoch_end) instead of being silently skipped+counted --
// the reference THROWS on a bad get (reference ddstore.hpp:210-214); the
// default here keeps the async hot path and counts (query()["oob_skipped"]).
inline bool strict_mode() {
    static const bool s = [] {
        const char* e = std::getenv("DDSTORE_STRICT");
        return e != nullptr && e[0] == '1';
    }();
    return s;
}

struct EpochFSM {
    bool fence_active = false;
    void begin() {
        TORCH_CHECK(!fence_active, "ddstore: epoch already began");  // ref ddstore.cxx:57
        fence_active = true;
    }
    void end() {
        TORCH_CHECK(fence_active, "ddstore: epoch has not begun");   // ref ddstore.cxx:71
        fence_active = false;
    }
};

// ===========================================================================
// DeviceStore
// ===========================================================================

struct DeviceVar {
    bool active = false;
    bool is_csr = false;
    int dds_t = 0;
    at::ScalarType st = at::kFloat;
    int64_t row_elems = 0;    // "disp" in reference terms
    int64_t itemsize = 0;
    int64_t nrows_local = 0;  // rows for fixed-stride; samples for CSR
    std::vector<int64_t> prefix;
    void* base = nullptr;
    size_t base_bytes = 0;
    std::vector<void*> peers;
    std::vector<char> opened;    // 1 where hipIpcOpenMemHandle was used
    void** d_peers = nullptr;
    int64_t* d_prefix = nullptr;
    unsigned long long* d_oob = nullptr;  // in-kernel OOB-index counter
    // CSR extras
    std::vector<int64_t> elem_prefix;
    int64_t* d_elem_prefix = nullptr;
    int64_t* d_goff = nullptr;   // replicated global element offsets [ntotal+1]
    int64_t nelems_local = 0;
    // stats
    int64_t n_gather = 0, rows_gathered = 0, bytes_gathered = 0;
};

class DeviceStore {
public:
    DeviceStore(int device, int rank, int nparts)
        : device_(device), rank_(rank), nparts_(nparts) {
        TORCH_CHECK(nparts >= 1 && nparts <= DDS_MAX_PARTS,
                    "ddstore: world size must be in [1, ", DDS_MAX_PARTS, "]");
        TORCH_CHECK(rank >= 0 && rank < nparts, "ddstore: bad rank");
    }
    ~DeviceStore() { free_all_noexcept(); }

    hipStream_t stream() const {
        return c10::hip::getCurrentHIPStream(device_).stream();
    }

    void add(const std::string& name, const at::Tensor& src, int64_t nrows,
             int64_t row_elems, std::vector<int64_t> nrows_all) {
        check_new(name);
        TORCH_CHECK(src.is_contiguous(), "ddstore add: array must be C-contiguous");
        TORCH_CHECK(src.numel() == nrows * row_elems, "ddstore add: shape mismatch");
        DeviceVar v;
        v.st = src.scalar_type();
        v.dds_t = dds_type_of(src);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;
        v.nrows_local = nrows;
        v.prefix = make_prefix(nrows_all);
        alloc_base(v, (size_t)(nrows * row_elems * v.itemsize));
        try {
            ingest(v.base, src, 0);
        } catch (...) {
            release(v);
            throw;
        }
        v.active = true;
        vars_[name] = std::move(v);
    }

    void init(const std::string& name, int64_t nrows, int64_t row_elems,
              at::ScalarType st, std::vector<int64_t> nrows_all) {
        // reference: pre-allocate zeroed shard, fill later via update
        // (ddstore.hpp:110-179; README.md:107 -- no epoch required)
        check_new(name);
        at::Tensor proto = at::empty({0}, at::TensorOptions().dtype(st));
        DeviceVar v;
        v.st = st;
        v.dds_t = dds_type_of(proto);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;
        v.nrows_local = nrows;
        v.prefix = make_prefix(nrows_all);
        alloc_base(v, (size_t)(nrows * row_elems * v.itemsize));
        try {
            HIP_CHECK(hipMemsetAsync(v.base, 0, v.base_bytes, stream()));
            HIP_CHECK(hipStreamSynchronize(stream()));
        } catch (...) {
            release(v);
            throw;
        }
        v.active = true;
        vars_[name] = std::move(v);
    }

    void add_csr(const std::string& name, const at::Tensor& values,
                 int64_t nsamples, int64_t nelems, int64_t row_elems,
                 std::vector<int64_t> nsamples_all, std::vector<int64_t> nelems_all,
                 const at::Tensor& goff_cpu) {
        // CSR first-class layout: the reference layers variable-length records
        // on an element-addressed store with disp=1 (HydraGNN
        // pattern); here offsets are replicated per GPU and the gather kernel
        // packs whole samples.
        check_new(name);
        TORCH_CHECK(values.is_contiguous(), "ddstore add_csr: values must be contiguous");
        TORCH_CHECK(values.numel() == nelems * row_elems, "ddstore add_csr: shape mismatch");
        DeviceVar v;
        v.is_csr = true;
        v.st = values.scalar_type();
        v.dds_t = dds_type_of(values);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;     // feature width per element
        v.nrows_local = nsamples;
        v.nelems_local = nelems;
        v.prefix = make_prefix(nsamples_all);
        v.elem_prefix = make_prefix(nelems_all);
        TORCH_CHECK(goff_cpu.scalar_type() == at::kLong && goff_cpu.is_contiguous() &&
                        goff_cpu.device().is_cpu() &&
                        goff_cpu.numel() == v.prefix[nparts_] + 1,
                    "ddstore add_csr: bad global offsets");
        alloc_base(v, (size_t)(nelems * row_elems * v.itemsize));
        try {
            ingest(v.base, values, 0);
            size_t gbytes = (size_t)goff_cpu.numel() * 8;
            HIP_CHECK(hipMalloc((void**)&v.d_goff, gbytes));
            HIP_CHECK(hipMemcpy(v.d_goff, goff_cpu.data_ptr<int64_t>(), gbytes,
                                hipMemcpyHostToDevice));
        } catch (...) {
            release(v);
            throw;
        }
        v.active = true;
        vars_[name] = std::move(v);
    }

    void init_csr(const std::string& name, int64_t nsamples, int64_t nelems,
                  int64_t row_elems, at::ScalarType st,
                  std::vector<int64_t> nsamples_all,
                  std::vector<int64_t> nelems_all, const at::Tensor& goff_cpu) {
        // CSR analog of init: lengths are fixed at registration (they define
        // the global offset directory), VALUES are zeroed and filled later
        // with update_elems -- the incremental-fill pattern the reference
        // supports for fixed-stride variables (ddstore.hpp:110-195,
        // README.md:107), extended to the first-class CSR layout.
        check_new(name);
        at::Tensor proto = at::empty({0}, at::TensorOptions().dtype(st));
        DeviceVar v;
        v.is_csr = true;
        v.st = st;
        v.dds_t = dds_type_of(proto);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;
        v.nrows_local = nsamples;
        v.nelems_local = nelems;
        v.prefix = make_prefix(nsamples_all);
        v.elem_prefix = make_prefix(nelems_all);
        TORCH_CHECK(goff_cpu.scalar_type() == at::kLong && goff_cpu.is_contiguous() &&
                        goff_cpu.device().is_cpu() &&
                        goff_cpu.numel() == v.prefix[nparts_] + 1,
                    "ddstore init_csr: bad global offsets");
        alloc_base(v, (size_t)(nelems * row_elems * v.itemsize));
        try {
            HIP_CHECK(hipMemsetAsync(v.base, 0, v.base_bytes, stream()));
            HIP_CHECK(hipStreamSynchronize(stream()));
            size_t gbytes = (size_t)goff_cpu.numel() * 8;
            HIP_CHECK(hipMalloc((void**)&v.d_goff, gbytes));
            HIP_CHECK(hipMemcpy(v.d_goff, goff_cpu.data_ptr<int64_t>(), gbytes,
                                hipMemcpyHostToDevice));
        } catch (...) {
            release(v);
            throw;
        }
        v.active = true;
        vars_[name] = std::move(v);
    }

    void update_elems(const std::string& name, const at::Tensor& src,
                      int64_t elem_offset) {
        // purely local CSR fill: elements [elem_offset, elem_offset+k) of
        // this rank's shard (element offsets are LOCAL; the Python layer
        // translates a sample offset via the goff directory)
        DeviceVar& v = var(name);
        TORCH_CHECK(v.is_csr, "ddstore update_elems: CSR variables only");
        TORCH_CHECK(src.is_contiguous(), "ddstore update: array must be C-contiguous");
        TORCH_CHECK(dds_itemsize(dds_type_of(src)) == v.itemsize,
                    "ddstore update: itemsize mismatch");
        int64_t nelems = src.numel() / std::max<int64_t>(v.row_elems, 1);
        TORCH_CHECK(src.numel() == nelems * v.row_elems,
                    "ddstore update: shape mismatch");
        TORCH_CHECK(elem_offset >= 0 && elem_offset + nelems <= v.nelems_local,
                    "ddstore update: out of range");
        ingest(v.base, src, elem_offset * v.row_elems * v.itemsize);
    }

    py::bytes ipc_handle(const std::string& name) {
        DeviceVar& v = var(name);
        hipIpcMemHandle_t h;
        HIP_CHECK(hipIpcGetMemHandle(&h, v.base));
        return py::bytes(reinterpret_cast<const char*>(&h), sizeof(h));
    }

    // Finalize a variable: map peer shards (empty bytes => use own base) and
    // mirror the directory + pointer table to device memory.
    // (reference analog: the MPI_Win_create collective, ddstore.hpp:56-62, or
    // the libfabric handshake's 3x Allgather, common.cxx:285-302)
    void open_peers(const std::string& name, const std::vector<std::string>& handles) {
        DeviceVar& v = var(name);
        TORCH_CHECK((int)handles.size() == nparts_, "ddstore open_peers: size mismatch");
        v.peers.assign(nparts_, nullptr);
        v.opened.assign(nparts_, 0);
        for (int r = 0; r < nparts_; ++r) {
            if (r == rank_) {
                v.peers[r] = v.base;
            } else {
                TORCH_CHECK(handles[r].size() == sizeof(hipIpcMemHandle_t),
                            "ddstore open_peers: bad handle bytes for rank ", r);
                hipIpcMemHandle_t h;
                std::memcpy(&h, handles[r].data(), sizeof(h));
                void* p = nullptr;
                hipError_t e = hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess);
                TORCH_CHECK(e == hipSuccess,
                            "ddstore open_peers: hipIpcOpenMemHandle failed for peer rank ",
                            r, " (", hipGetErrorString(e),
                            ") -- peers must share one node/xGMI domain and "
                            "HSA_ENABLE_IPC_MODE_LEGACY=0 must be set");
                v.peers[r] = p;
                v.opened[r] = 1;
            }
        }
        // one device block: [peer ptrs | prefix | elem_prefix? | counters]
        size_t nb = nparts_ * sizeof(void*) + (nparts_ + 1) * 8 +
                    (v.is_csr ? (nparts_ + 1) * 8 : 0) + DDS_NCTR * 8;
        HIP_CHECK(hipMalloc((void**)&v.d_peers, nb));
        v.d_prefix = reinterpret_cast<int64_t*>(v.d_peers + nparts_);
        HIP_CHECK(hipMemcpy(v.d_peers, v.peers.data(), nparts_ * sizeof(void*),
                            hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(v.d_prefix, v.prefix.data(), (nparts_ + 1) * 8,
                            hipMemcpyHostToDevice));
        int64_t* tail = v.d_prefix + nparts_ + 1;
        if (v.is_csr) {
            v.d_elem_prefix = tail;
            HIP_CHECK(hipMemcpy(v.d_elem_prefix, v.elem_prefix.data(),
                                (nparts_ + 1) * 8, hipMemcpyHostToDevice));
            tail += nparts_ + 1;
        }
        v.d_oob = reinterpret_cast<unsigned long long*>(tail);
        HIP_CHECK(hipMemset(v.d_oob, 0, DDS_NCTR * 8));
    }

    void update(const std::string& name, const at::Tensor& src, int64_t offset) {
        // purely local fill at row offset (reference ddstore.hpp:181-195)
        DeviceVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore update: not supported for CSR variables");
        TORCH_CHECK(src.is_contiguous(), "ddstore update: array must be C-contiguous");
        TORCH_CHECK(dds_itemsize(dds_type_of(src)) == v.itemsize,
                    "ddstore update: itemsize mismatch");
        int64_t nrows = src.numel() / v.row_elems;
        TORCH_CHECK(src.numel() == nrows * v.row_elems, "ddstore update: shape mismatch");
        TORCH_CHECK(offset >= 0 && offset + nrows <= v.nrows_local,
                    "ddstore update: out of range");
        ingest(v.base, src, offset * v.row_elems * v.itemsize);
    }

    void get_range(const std::string& name, int64_t start, int64_t count,
                   at::Tensor out) {
        // reference get semantics: dense [start, start+count) rows from ONE
        // owner; throws if the range crosses a shard boundary
        // (ddstore.hpp:197-248)
        DeviceVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore get: use get_csr for CSR variables");
        TORCH_CHECK(out.is_contiguous(), "ddstore get: output must be C-contiguous");
        TORCH_CHECK(dds_itemsize(dds_type_of(out)) == v.itemsize,
                    "ddstore get: itemsize mismatch");
        TORCH_CHECK(out.numel() == count * v.row_elems, "ddstore get: shape mismatch");
        check_peers(v);
        int64_t ntotal = v.prefix[nparts_];
        TORCH_CHECK(start >= 0 && start < ntotal, "Invalid start on target");
        int target = owner_of_host(v.prefix, start);
        TORCH_CHECK(start + count <= v.prefix[target + 1], "Invalid count on target");
        const char* src = (const char*)v.peers[target] +
                          (start - v.prefix[target]) * v.row_elems * v.itemsize;
        size_t nb = (size_t)count * v.row_elems * v.itemsize;
        if (out.device().is_cpu()) {
            HIP_CHECK(hipStreamSynchronize(stream()));
            HIP_CHECK(hipMemcpy(out.data_ptr(), src, nb, hipMemcpyDeviceToHost));
        } else {
            TORCH_CHECK(out.device().index() == device_, "ddstore get: wrong device");
            HIP_CHECK(hipMemcpyAsync(out.data_ptr(), src, nb,
                                     hipMemcpyDeviceToDevice, stream()));
        }
        v.n_gather += 1;
        v.rows_gathered += count;
        v.bytes_gathered += nb;
    }

    void gather(const std::string& name, const at::Tensor& idx, at::Tensor out) {
        // the batched hot path: one launch per minibatch
        RoctxRange rr_("ddstore::gather");
        DeviceVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore gather: use gather_csr for CSR variables");
        check_peers(v);
        check_idx(idx);
        int64_t nidx = idx.numel();
        TORCH_CHECK(out.is_contiguous() && out.device().is_cuda() &&
                        out.device().index() == device_,
                    "ddstore gather: output must be a contiguous tensor on the store device");
        TORCH_CHECK(out.numel() == nidx * v.row_elems, "ddstore gather: shape mismatch");
        // byte move only for the SAME dds type (bool aliases u8); any other
        // dtype pair converts numerically -- a same-size pair like f16->bf16
        // must NOT silently reinterpret bits (the CPU path converts too)
        int out_t = dds_type_of(out);
        ddstore::gather_rows(stream(), (const void* const*)v.d_peers, v.d_prefix,
                             nparts_, idx.data_ptr<int64_t>(), nidx, v.row_elems,
                             v.dds_t, out_t, out.data_ptr(), v.d_oob);
        v.n_gather += 1;
        v.rows_gathered += nidx;
        v.bytes_gathered += nidx * v.row_elems * v.itemsize;
    }

    void gather_affine(const std::string& name, const at::Tensor& idx,
                       at::Tensor out, double scale, double shift) {
        RoctxRange rr_("ddstore::gather_affine");
        DeviceVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore gather: use gather_csr for CSR variables");
        check_peers(v);
        check_idx(idx);
        int64_t nidx = idx.numel();
        TORCH_CHECK(out.is_contiguous() && out.device().is_cuda() &&
                        out.device().index() == device_,
                    "ddstore gather: output must be a contiguous tensor on the store device");
        TORCH_CHECK(out.numel() == nidx * v.row_elems, "ddstore gather: shape mismatch");
        int out_t = dds_type_of(out);
        TORCH_CHECK(out_t == DDS_F32 || out_t == DDS_F16 || out_t == DDS_BF16,
                    "ddstore gather: affine output must be f32/f16/bf16");
        ddstore::gather_rows_affine(stream(), (const void* const*)v.d_peers,
                                    v.d_prefix, nparts_, idx.data_ptr<int64_t>(),
                                    nidx, v.row_elems, v.dds_t, out_t,
                                    (float)scale, (float)shift, out.data_ptr(),
                                    v.d_oob);
        v.n_gather += 1;
        v.rows_gathered += nidx;
        v.bytes_gathered += nidx * v.row_elems * v.itemsize;
    }

    void gather_csr(const std::string& name, const at::Tensor& idx,
                    const at::Tensor& out_off, at::Tensor out, int64_t total_elems) {
        RoctxRange rr_("ddstore::gather_csr");
        DeviceVar& v = var(name);
        TORCH_CHECK(v.is_csr, "ddstore gather_csr: not a CSR variable");
        check_peers(v);
        check_idx(idx);
        check_idx(out_off);
        int64_t nidx = idx.numel();
        TORCH_CHECK(out_off.numel() == nidx + 1, "ddstore gather_csr: bad out_off");
        TORCH_CHECK(out.is_contiguous() && out.device().is_cuda() &&
                        dds_type_of(out) == v.dds_t,
                    "ddstore gather_csr: bad output tensor");
        TORCH_CHECK(out.numel() >= total_elems * v.row_elems,
                    "ddstore gather_csr: output too small");
        // capacity passed to the kernel is the REAL buffer capacity: a sample
        // whose slice would end past it is skipped + counted, never written
        // out of bounds
        const int64_t cap = out.numel() / std::max<int64_t>(v.row_elems, 1);
        ddstore::gather_csr(stream(), (const void* const*)v.d_peers, v.d_prefix,
                            v.d_elem_prefix, nparts_, v.d_goff,
                            idx.data_ptr<int64_t>(), nidx,
                            out_off.data_ptr<int64_t>(),
                            v.row_elems * v.itemsize, cap,
                            out.data_ptr(), v.d_oob);
        v.n_gather += 1;
        v.rows_gathered += nidx;
        v.bytes_gathered += total_elems * v.row_elems * v.itemsize;
    }

    // One-call CSR fetch: ONE fused kernel (per-sample lens + decoupled-
    // lookback exclusive scan + payload gather), no host round trips and no
    // intermediate lens/zeros launches. `out` is a capacity buffer
    // (>= worst-case batch elements); returns the [n+1] element-offset
    // tensor (device). True gathered-element stats accumulate in the
    // device counter block (the capacity would over-report).
    at::Tensor gather_csr_fast(const std::string& name, const at::Tensor& idx,
                               at::Tensor out) {
        RoctxRange rr_("ddstore::gather_csr_fast");
        DeviceVar& v = var(name);
        TORCH_CHECK(v.is_csr, "ddstore gather_csr: not a CSR variable");
        check_peers(v);
        check_idx(idx);
        const int64_t nidx = idx.numel();
        TORCH_CHECK(out.is_contiguous() && out.device().is_cuda() &&
                        dds_type_of(out) == v.dds_t,
                    "ddstore gather_csr: bad output tensor");
        auto opts = at::TensorOptions().dtype(at::kLong).device(idx.device());
        at::Tensor off = at::empty({nidx + 1}, opts);
        if (nidx == 0) {
            off.zero_();
            return off;
        }
        const int64_t cap = out.numel() / std::max<int64_t>(v.row_elems, 1);
        const int64_t elem_bytes = v.row_elems * v.itemsize;
        auto bopts = at::TensorOptions().dtype(at::kByte).device(idx.device());
        at::Tensor scratch = at::empty(
            {(int64_t)ddstore::csr_plan_scratch_bytes(nidx)}, bopts);
        const int64_t desc_cap =
            cap / ddstore::csr_item_elems(elem_bytes) + nidx;
        at::Tensor desc = at::empty({2 * desc_cap}, opts);  // 2 i64 per item
        ddstore::gather_csr_balanced(
            stream(), (const void* const*)v.d_peers, v.d_prefix,
            v.d_elem_prefix, nparts_, v.d_goff, v.prefix[nparts_],
            idx.data_ptr<int64_t>(), nidx, elem_bytes, cap,
            off.data_ptr<int64_t>(), out.data_ptr(), v.d_oob,
            scratch.data_ptr(), desc.data_ptr<int64_t>(), desc_cap);
        v.n_gather += 1;
        v.rows_gathered += nidx;
        // bytes accounted via the device DDS_CTR_ELEMS counter (see query)
        return off;
    }

    void csr_lens(const std::string& name, const at::Tensor& idx, at::Tensor lens) {
        DeviceVar& v = var(name);
        TORCH_CHECK(v.is_csr, "ddstore csr_lens: not a CSR variable");
        check_peers(v);
        check_idx(idx);
        TORCH_CHECK(lens.scalar_type() == at::kLong && lens.is_contiguous() &&
                        lens.device().is_cuda() && lens.numel() == idx.numel(),
                    "ddstore csr_lens: bad lens tensor");
        ddstore::csr_lens(stream(), v.d_goff, idx.data_ptr<int64_t>(), idx.numel(),
                          v.prefix[nparts_], lens.data_ptr<int64_t>(), v.d_oob,
                          nullptr);
    }

    void scatter_local(const std::string& name, const at::Tensor& local_idx,
                       const at::Tensor& src) {
        // reshuffle placement: src row r -> local row local_idx[r]
        DeviceVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore scatter_local: CSR not supported");
        check_idx(local_idx);
        TORCH_CHECK(src.is_contiguous() && src.device().is_cuda() &&
                        dds_type_of(src) == v.dds_t,
                    "ddstore scatter_local: bad src");
        TORCH_CHECK(src.numel() == local_idx.numel() * v.row_elems,
                    "ddstore scatter_local: shape mismatch");
        check_peers(v);  // d_oob lives in the metadata block
        ddstore::scatter_rows_local(stream(), v.base, v.nrows_local, v.row_elems,
                                    v.dds_t, local_idx.data_ptr<int64_t>(),
                                    local_idx.numel(), src.data_ptr(), v.d_oob);
    }

    // Zero-copy view of the local shard as a torch tensor (does NOT own).
    at::Tensor local_shard(const std::string& name) {
        DeviceVar& v = var(name);
        int64_t n = v.is_csr ? v.nelems_local : v.nrows_local;
        auto opts = at::TensorOptions().dtype(v.st).device(at::kCUDA, device_);
        return at::from_blob(v.base, {n, v.row_elems}, opts);
    }

    void epoch_begin() {
        fsm_.begin();
        HIP_CHECK(hipStreamSynchronize(stream()));
    }
    void epoch_end() {
        fsm_.end();
        HIP_CHECK(hipStreamSynchronize(stream()));
        if (strict_mode()) check_strict();
    }
    bool epoch_active() const { return fsm_.fence_active; }

    // Raise if any variable has skipped samples (cumulative counters; reset
    // via reset_counters after an intentional OOB probe).
    void check_strict() {
        for (auto& kv : vars_) {
            DeviceVar& v = kv.second;
            if (!v.d_oob) continue;
            unsigned long long ctrs[DDS_NCTR] = {0, 0, 0};
            HIP_CHECK(hipMemcpy(ctrs, v.d_oob, DDS_NCTR * 8,
                                hipMemcpyDeviceToHost));
            TORCH_CHECK(ctrs[DDS_CTR_OOB] == 0, "ddstore strict: variable '",
                        kv.first, "': ", ctrs[DDS_CTR_OOB],
                        " out-of-range sample indices were skipped "
                        "(DDSTORE_STRICT=1)");
            TORCH_CHECK(ctrs[DDS_CTR_CAP] == 0, "ddstore strict: variable '",
                        kv.first, "': ", ctrs[DDS_CTR_CAP],
                        " CSR samples skipped: output capacity buffer too "
                        "small (DDSTORE_STRICT=1)");
        }
    }

    py::dict query(const std::string& name) {
        DeviceVar& v = var(name);
        py::dict d;
        d["nrows_local"] = v.nrows_local;
        d["nrows_total"] = v.prefix[nparts_];
        d["disp"] = v.row_elems;
        d["itemsize"] = v.itemsize;
        d["is_csr"] = v.is_csr;
        d["prefix"] = v.prefix;
        if (v.is_csr) {
            d["nelems_local"] = v.nelems_local;
            d["elem_prefix"] = v.elem_prefix;
        }
        d["n_gather"] = v.n_gather;
        d["rows_gathered"] = v.rows_gathered;
        int64_t bytes = v.bytes_gathered;
        if (v.d_oob) {
            unsigned long long ctrs[DDS_NCTR] = {0, 0, 0};
            HIP_CHECK(hipStreamSynchronize(stream()));
            HIP_CHECK(hipMemcpy(ctrs, v.d_oob, DDS_NCTR * 8,
                                hipMemcpyDeviceToHost));
            d["oob_skipped"] = (int64_t)ctrs[DDS_CTR_OOB];
            d["cap_skipped"] = (int64_t)ctrs[DDS_CTR_CAP];
            // gather_csr_fast accounts true elements device-side (the
            // capacity-buffer size would over-report)
            bytes += (int64_t)ctrs[DDS_CTR_ELEMS] * v.row_elems * v.itemsize;
        }
        d["bytes_gathered"] = bytes;
        return d;
    }

    // Zero a variable's skip/stat counters (after an intentional OOB test,
    // or between bench phases).
    void reset_counters(const std::string& name) {
        DeviceVar& v = var(name);
        if (v.d_oob) {
            HIP_CHECK(hipStreamSynchronize(stream()));
            HIP_CHECK(hipMemset(v.d_oob, 0, DDS_NCTR * 8));
        }
        v.n_gather = v.rows_gathered = v.bytes_gathered = 0;
    }

    bool has(const std::string& name) const { return vars_.count(name) > 0; }

    void free_var(const std::string& name) {
        auto it = vars_.find(name);
        TORCH_CHECK(it != vars_.end(), "ddstore: unknown variable '", name, "'");
        release(it->second);
        vars_.erase(it);
    }

    void free_all() {
        for (auto& kv : vars_) release(kv.second);
        vars_.clear();
    }

private:
    void free_all_noexcept() noexcept {
        try { free_all(); } catch (...) { /* after runtime teardown */ }
    }

    void check_new(const std::string& name) {
        TORCH_CHECK(!vars_.count(name), "ddstore: variable '", name, "' already exists");
    }
    DeviceVar& var(const std::string& name) {
        auto it = vars_.find(name);
        TORCH_CHECK(it != vars_.end(), "ddstore: unknown variable '", name, "'");
        return it->second;
    }
    void check_peers(const DeviceVar& v) {
        TORCH_CHECK(v.d_peers != nullptr,
                    "ddstore: variable not finalized (open_peers not called)");
    }
    void check_idx(const at::Tensor& idx) {
        TORCH_CHECK(idx.scalar_type() == at::kLong && idx.is_contiguous() &&
                        idx.device().is_cuda() && idx.device().index() == device_,
                    "ddstore: indices must be a contiguous int64 tensor on the store device");
    }

    void alloc_base(DeviceVar& v, size_t bytes) {
        hipError_t e = hipSetDevice(device_);
        TORCH_CHECK(e == hipSuccess, "ddstore: hipSetDevice failed");
        v.base_bytes = bytes < 256 ? 256 : bytes;
        HIP_CHECK(hipMalloc(&v.base, v.base_bytes));
    }

    void ingest(void* dst_base, const at::Tensor& src, int64_t byte_off) {
        char* dst = (char*)dst_base + byte_off;
        size_t nb = (size_t)src.numel() * src.element_size();
        if (nb == 0) return;
        if (src.device().is_cpu()) {
            HIP_CHECK(hipMemcpy(dst, src.data_ptr(), nb, hipMemcpyHostToDevice));
        } else {
            TORCH_CHECK(src.device().index() == device_, "ddstore: wrong source device");
            HIP_CHECK(hipMemcpyAsync(dst, src.data_ptr(), nb,
                                     hipMemcpyDeviceToDevice, stream()));
            HIP_CHECK(hipStreamSynchronize(stream()));
        }
    }

    void release(DeviceVar& v) {
        if (!v.active && !v.base) return;
        for (int r = 0; r < (int)v.peers.size(); ++r)
            if (v.opened.size() > (size_t)r && v.opened[r] && v.peers[r])
                (void)hipIpcCloseMemHandle(v.peers[r]);
        v.peers.clear();
        if (v.d_peers) (void)hipFree(v.d_peers);
        if (v.d_goff) (void)hipFree(v.d_goff);
        if (v.base) (void)hipFree(v.base);
        v.base = nullptr;
        v.d_peers = nullptr;
        v.d_goff = nullptr;
        v.active = false;
    }

    int device_;
    int rank_;
    int nparts_;
    EpochFSM fsm_;
    std::map<std::string, DeviceVar> vars_;
};

// ===========================================================================
// HostStore (POSIX shm, CPU compatibility path -- BASELINE config 1)
// ===========================================================================

struct HostVar {
    bool active = false;
    bool is_csr = false;
    bool creator = false;
    int dds_t = 0;
    at::ScalarType st = at::kFloat;
    int64_t row_elems = 0;
    int64_t itemsize = 0;
    int64_t nrows_local = 0;
    int64_t nelems_local = 0;
    std::vector<int64_t> prefix;
    std::vector<int64_t> elem_prefix;
    std::string shm_name;
    void* base = nullptr;
    size_t base_bytes = 0;
    std::vector<void*> peers;
    std::vector<size_t> peer_bytes;
    std::vector<std::string> peer_names;
    at::Tensor goff;  // CPU int64 [ntotal+1], replicated
    int64_t n_gather = 0, rows_gathered = 0, bytes_gathered = 0;
    int64_t oob_skipped = 0, cap_skipped = 0;
};

class HostStore {
public:
    HostStore(const std::string& session, int rank, int nparts)
        : session_(session), rank_(rank), nparts_(nparts) {
        TORCH_CHECK(nparts >= 1 && nparts <= DDS_MAX_PARTS,
                    "ddstore: world size must be in [1, ", DDS_MAX_PARTS, "]");
    }
    ~HostStore() {
        try { free_all(); } catch (...) {}
    }

    std::string shm_name(const std::string& var) const {
        return "/dds-" + session_ + "-" + var + "-" + std::to_string(rank_);
    }

    std::string add(const std::string& name, const at::Tensor& src, int64_t nrows,
                    int64_t row_elems, std::vector<int64_t> nrows_all) {
        check_new(name);
        TORCH_CHECK(src.is_contiguous() && src.device().is_cpu(),
                    "ddstore add: array must be C-contiguous on CPU");
        TORCH_CHECK(src.numel() == nrows * row_elems, "ddstore add: shape mismatch");
        HostVar v;
        v.st = src.scalar_type();
        v.dds_t = dds_type_of(src);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;
        v.nrows_local = nrows;
        v.prefix = make_prefix(nrows_all);
        create_shm(v, name, (size_t)(nrows * row_elems * v.itemsize));
        std::memcpy(v.base, src.data_ptr(), (size_t)src.numel() * v.itemsize);
        try_collapse_hugepages(v.base, v.base_bytes);
        v.active = true;
        std::string n = v.shm_name;
        vars_[name] = std::move(v);
        return n;
    }

    std::string init(const std::string& name, int64_t nrows, int64_t row_elems,
                     at::ScalarType st, std::vector<int64_t> nrows_all) {
        check_new(name);
        at::Tensor proto = at::empty({0}, at::TensorOptions().dtype(st));
        HostVar v;
        v.st = st;
        v.dds_t = dds_type_of(proto);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;
        v.nrows_local = nrows;
        v.prefix = make_prefix(nrows_all);
        create_shm(v, name, (size_t)(nrows * row_elems * v.itemsize));
        std::memset(v.base, 0, v.base_bytes);
        try_collapse_hugepages(v.base, v.base_bytes);
        v.active = true;
        std::string n = v.shm_name;
        vars_[name] = std::move(v);
        return n;
    }

    std::string add_csr(const std::string& name, const at::Tensor& values,
                        int64_t nsamples, int64_t nelems, int64_t row_elems,
                        std::vector<int64_t> nsamples_all,
                        std::vector<int64_t> nelems_all, const at::Tensor& goff_cpu) {
        check_new(name);
        HostVar v;
        v.is_csr = true;
        v.st = values.scalar_type();
        v.dds_t = dds_type_of(values);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;
        v.nrows_local = nsamples;
        v.nelems_local = nelems;
        v.prefix = make_prefix(nsamples_all);
        v.elem_prefix = make_prefix(nelems_all);
        TORCH_CHECK(values.is_contiguous() && values.device().is_cpu(),
                    "ddstore add_csr: values must be contiguous on CPU");
        TORCH_CHECK(values.numel() == nelems * row_elems, "ddstore add_csr: shape mismatch");
        TORCH_CHECK(goff_cpu.scalar_type() == at::kLong && goff_cpu.is_contiguous() &&
                        goff_cpu.numel() == v.prefix[nparts_] + 1,
                    "ddstore add_csr: bad global offsets");
        v.goff = goff_cpu;
        create_shm(v, name, (size_t)(nelems * row_elems * v.itemsize));
        std::memcpy(v.base, values.data_ptr(), (size_t)values.numel() * v.itemsize);
        try_collapse_hugepages(v.base, v.base_bytes);
        v.active = true;
        std::string n = v.shm_name;
        vars_[name] = std::move(v);
        return n;
    }

    std::string init_csr(const std::string& name, int64_t nsamples,
                         int64_t nelems, int64_t row_elems, at::ScalarType st,
                         std::vector<int64_t> nsamples_all,
                         std::vector<int64_t> nelems_all,
                         const at::Tensor& goff_cpu) {
        check_new(name);
        at::Tensor proto = at::empty({0}, at::TensorOptions().dtype(st));
        HostVar v;
        v.is_csr = true;
        v.st = st;
        v.dds_t = dds_type_of(proto);
        v.itemsize = dds_itemsize(v.dds_t);
        v.row_elems = row_elems;
        v.nrows_local = nsamples;
        v.nelems_local = nelems;
        v.prefix = make_prefix(nsamples_all);
        v.elem_prefix = make_prefix(nelems_all);
        TORCH_CHECK(goff_cpu.scalar_type() == at::kLong && goff_cpu.is_contiguous() &&
                        goff_cpu.numel() == v.prefix[nparts_] + 1,
                    "ddstore init_csr: bad global offsets");
        v.goff = goff_cpu;
        create_shm(v, name, (size_t)(nelems * row_elems * v.itemsize));
        std::memset(v.base, 0, v.base_bytes);
        try_collapse_hugepages(v.base, v.base_bytes);
        v.active = true;
        std::string n = v.shm_name;
        vars_[name] = std::move(v);
        return n;
    }

    void update_elems(const std::string& name, const at::Tensor& src,
                      int64_t elem_offset) {
        HostVar& v = var(name);
        TORCH_CHECK(v.is_csr, "ddstore update_elems: CSR variables only");
        TORCH_CHECK(src.is_contiguous() && src.device().is_cpu(),
                    "ddstore update: array must be C-contiguous on CPU");
        TORCH_CHECK(dds_itemsize(dds_type_of(src)) == v.itemsize,
                    "ddstore update: itemsize mismatch");
        int64_t nelems = src.numel() / (v.row_elems > 0 ? v.row_elems : 1);
        TORCH_CHECK(src.numel() == nelems * v.row_elems,
                    "ddstore update: shape mismatch");
        TORCH_CHECK(elem_offset >= 0 && elem_offset + nelems <= v.nelems_local,
                    "ddstore update: out of range");
        std::memcpy((char*)v.base + elem_offset * v.row_elems * v.itemsize,
                    src.data_ptr(), (size_t)src.numel() * v.itemsize);
    }

    void open_peers(const std::string& name, const std::vector<std::string>& names) {
        HostVar& v = var(name);
        TORCH_CHECK((int)names.size() == nparts_, "ddstore open_peers: size mismatch");
        v.peers.assign(nparts_, nullptr);
        v.peer_bytes.assign(nparts_, 0);
        v.peer_names = names;
        const std::vector<int64_t>& cnt_prefix = v.is_csr ? v.elem_prefix : v.prefix;
        for (int r = 0; r < nparts_; ++r) {
            size_t nb = (size_t)((cnt_prefix[r + 1] - cnt_prefix[r]) * v.row_elems *
                                 v.itemsize);
            if (nb < 1) nb = 1;
            if (r == rank_) {
                v.peers[r] = v.base;
                v.peer_bytes[r] = 0;  // not mapped, do not munmap
                continue;
            }
            int fd = shm_open(names[r].c_str(), O_RDONLY, 0600);
            TORCH_CHECK(fd >= 0, "ddstore: shm_open failed for ", names[r]);
            void* p = mmap(nullptr, nb, PROT_READ, MAP_SHARED, fd, 0);
            close(fd);
            TORCH_CHECK(p != MAP_FAILED, "ddstore: mmap failed for ", names[r]);
#if defined(MADV_HUGEPAGE)
            (void)madvise(p, nb, MADV_HUGEPAGE);
#endif
            try_collapse_hugepages(p, nb);  // peer VMA: PMD-map shared pages
            v.peers[r] = p;
            v.peer_bytes[r] = nb;
        }
    }

    void update(const std::string& name, const at::Tensor& src, int64_t offset) {
        HostVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore update: not supported for CSR variables");
        TORCH_CHECK(src.is_contiguous() && src.device().is_cpu(),
                    "ddstore update: array must be C-contiguous on CPU");
        TORCH_CHECK(dds_itemsize(dds_type_of(src)) == v.itemsize,
                    "ddstore update: itemsize mismatch");
        int64_t nrows = src.numel() / v.row_elems;
        TORCH_CHECK(src.numel() == nrows * v.row_elems, "ddstore update: shape mismatch");
        TORCH_CHECK(offset >= 0 && offset + nrows <= v.nrows_local,
                    "ddstore update: out of range");
        std::memcpy((char*)v.base + offset * v.row_elems * v.itemsize, src.data_ptr(),
                    (size_t)src.numel() * v.itemsize);
    }

    void get_range(const std::string& name, int64_t start, int64_t count,
                   at::Tensor out) {
        HostVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore get: use get_csr for CSR variables");
        TORCH_CHECK(out.is_contiguous() && out.device().is_cpu(),
                    "ddstore get: output must be C-contiguous on CPU");
        TORCH_CHECK(dds_itemsize(dds_type_of(out)) == v.itemsize,
                    "ddstore get: itemsize mismatch");
        TORCH_CHECK(out.numel() == count * v.row_elems, "ddstore get: shape mismatch");
        check_peers(v);
        int64_t ntotal = v.prefix[nparts_];
        TORCH_CHECK(start >= 0 && start < ntotal, "Invalid start on target");
        int target = owner_of_host(v.prefix, start);
        TORCH_CHECK(start + count <= v.prefix[target + 1], "Invalid count on target");
        const char* src = (const char*)v.peers[target] +
                          (start - v.prefix[target]) * v.row_elems * v.itemsize;
        size_t nb = (size_t)count * v.row_elems * v.itemsize;
        std::memcpy(out.data_ptr(), src, nb);
        v.n_gather += 1;
        v.rows_gathered += count;
        v.bytes_gathered += nb;
    }

    void gather(const std::string& name, const at::Tensor& idx, at::Tensor out) {
        HostVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore gather: use gather_csr for CSR variables");
        check_peers(v);
        TORCH_CHECK(idx.scalar_type() == at::kLong && idx.is_contiguous() &&
                        idx.device().is_cpu(),
                    "ddstore gather: indices must be contiguous int64 on CPU");
        int64_t nidx = idx.numel();
        // dtype (not just itemsize) must match: a same-size different dtype
        // (bf16 out from an f16 store) would silently bit-reinterpret here
        // while the GPU path converts numerically; the Python
        // layer converts via a staging tensor when dtypes differ.
        TORCH_CHECK(out.is_contiguous() && out.device().is_cpu() &&
                        dds_type_of(out) == v.dds_t,
                    "ddstore gather: output must be a contiguous CPU tensor of the store dtype");
        TORCH_CHECK(out.numel() == nidx * v.row_elems, "ddstore gather: shape mismatch");
        const int64_t* ip = idx.data_ptr<int64_t>();
        const int64_t rb = v.row_elems * v.itemsize;
        char* op = (char*)out.data_ptr();
        const int64_t ntotal = v.prefix[nparts_];
        // random source rows are DRAM-latency-bound: software-prefetch the
        // row PF iterations ahead (the extra owner lookup is ~free), and
        // write the packed output with non-temporal stores
        const int64_t PF = host_prefetch_dist();
        host_parallel_for(0, nidx, 1024, [&](int64_t b, int64_t e) {
            auto src_of = [&](int64_t g) -> const char* {
                int p = owner_of_host(v.prefix, g);
                return (const char*)v.peers[p] + (g - v.prefix[p]) * rb;
            };
            for (int64_t i = b; i < e; ++i) {
                if (PF > 0 && i + PF < e) {
                    int64_t gp = ip[i + PF];
                    if (gp >= 0 && gp < ntotal) {
                        const char* ps = src_of(gp);
                        for (int64_t o = 0; o < rb; o += 64)
                            __builtin_prefetch(ps + o, 0, 1);
                    }
                }
                int64_t g = ip[i];
                TORCH_CHECK(g >= 0 && g < ntotal, "ddstore gather: index out of range");
                copy_row_nt(op + i * rb, src_of(g), (size_t)rb);
            }
        });
        copy_rows_fence();
        v.n_gather += 1;
        v.rows_gathered += nidx;
        v.bytes_gathered += nidx * rb;
    }

    void gather_csr(const std::string& name, const at::Tensor& idx,
                    const at::Tensor& out_off, at::Tensor out, int64_t total_elems) {
        HostVar& v = var(name);
        TORCH_CHECK(v.is_csr, "ddstore gather_csr: not a CSR variable");
        check_peers(v);
        TORCH_CHECK(idx.scalar_type() == at::kLong && idx.is_contiguous() &&
                        idx.device().is_cpu(), "ddstore gather_csr: bad indices");
        TORCH_CHECK(out_off.scalar_type() == at::kLong && out_off.is_contiguous() &&
                        out_off.numel() == idx.numel() + 1,
                    "ddstore gather_csr: bad out_off");
        TORCH_CHECK(out.is_contiguous() && out.device().is_cpu() &&
                        dds_type_of(out) == v.dds_t &&
                        out.numel() >= total_elems * v.row_elems,
                    "ddstore gather_csr: bad output tensor");
        const int64_t* ip = idx.data_ptr<int64_t>();
        const int64_t* oo = out_off.data_ptr<int64_t>();
        const int64_t* go = v.goff.data_ptr<int64_t>();
        const int64_t eb = v.row_elems * v.itemsize;
        const int64_t ntotal = v.prefix[nparts_];
        // real buffer capacity in elements: a sample whose slice would end
        // past it (or an out-of-range id) is skipped + counted, mirroring
        // the device kernels -- never read/written out of bounds
        const int64_t cap =
            out.numel() / (v.row_elems > 0 ? v.row_elems : (int64_t)1);
        char* op = (char*)out.data_ptr();
        std::atomic<int64_t> oob(0), capskip(0), elems(0);
        host_parallel_for(0, idx.numel(), 64, [&](int64_t b, int64_t e) {
            int64_t my_oob = 0, my_cap = 0, my_elems = 0;
            for (int64_t s = b; s < e; ++s) {
                int64_t g = ip[s];
                if (g < 0 || g >= ntotal) {
                    ++my_oob;
                    continue;
                }
                int p = owner_of_host(v.prefix, g);
                int64_t e0 = go[g], n = go[g + 1] - go[g];
                if (oo[s] < 0 || oo[s] + n > cap) {
                    ++my_cap;
                    continue;
                }
                std::memcpy(op + oo[s] * eb,
                            (const char*)v.peers[p] + (e0 - v.elem_prefix[p]) * eb,
                            (size_t)(n * eb));
                my_elems += n;
            }
            oob += my_oob;
            capskip += my_cap;
            elems += my_elems;
        });
        v.oob_skipped += oob.load();
        v.cap_skipped += capskip.load();
        v.n_gather += 1;
        v.rows_gathered += idx.numel();
        v.bytes_gathered += elems.load() * eb;  // true gathered bytes
    }

    void scatter_local(const std::string& name, const at::Tensor& local_idx,
                       const at::Tensor& src) {
        HostVar& v = var(name);
        TORCH_CHECK(!v.is_csr, "ddstore scatter_local: CSR not supported");
        TORCH_CHECK(local_idx.scalar_type() == at::kLong && local_idx.is_contiguous() &&
                        local_idx.device().is_cpu(), "ddstore scatter_local: bad indices");
        TORCH_CHECK(src.is_contiguous() && src.device().is_cpu() &&
                        dds_type_of(src) == v.dds_t &&
                        src.numel() == local_idx.numel() * v.row_elems,
                    "ddstore scatter_local: bad src");
        const int64_t* ip = local_idx.data_ptr<int64_t>();
        const int64_t rb = v.row_elems * v.itemsize;
        const char* sp = (const char*)src.data_ptr();
        host_parallel_for(0, local_idx.numel(), 1024, [&](int64_t b, int64_t e) {
            for (int64_t i = b; i < e; ++i) {
                TORCH_CHECK(ip[i] >= 0 && ip[i] < v.nrows_local,
                            "ddstore scatter_local: index out of range");
                std::memcpy((char*)v.base + ip[i] * rb, sp + i * rb, (size_t)rb);
            }
        });
    }

    at::Tensor local_shard(const std::string& name) {
        HostVar& v = var(name);
        int64_t n = v.is_csr ? v.nelems_local : v.nrows_local;
        return at::from_blob(v.base, {n, v.row_elems}, at::TensorOptions().dtype(v.st));
    }

    void epoch_begin() { fsm_.begin(); }
    void epoch_end() {
        fsm_.end();
        if (strict_mode()) check_strict();
    }
    bool epoch_active() const { return fsm_.fence_active; }

    void check_strict() {
        for (auto& kv : vars_) {
            HostVar& v = kv.second;
            TORCH_CHECK(v.oob_skipped == 0, "ddstore strict: variable '",
                        kv.first, "': ", v.oob_skipped,
                        " out-of-range sample indices were skipped "
                        "(DDSTORE_STRICT=1)");
            TORCH_CHECK(v.cap_skipped == 0, "ddstore strict: variable '",
                        kv.first, "': ", v.cap_skipped,
                        " CSR samples skipped: output capacity buffer too "
                        "small (DDSTORE_STRICT=1)");
        }
    }

    void reset_counters(const std::string& name) {
        HostVar& v = var(name);
        v.oob_skipped = v.cap_skipped = 0;
        v.n_gather = v.rows_gathered = v.bytes_gathered = 0;
    }

    py::dict query(const std::string& name) {
        HostVar& v = var(name);
        py::dict d;
        d["nrows_local"] = v.nrows_local;
        d["nrows_total"] = v.prefix[nparts_];
        d["disp"] = v.row_elems;
        d["itemsize"] = v.itemsize;
        d["is_csr"] = v.is_csr;
        d["prefix"] = v.prefix;
        if (v.is_csr) {
            d["nelems_local"] = v.nelems_local;
            d["elem_prefix"] = v.elem_prefix;
        }
        d["n_gather"] = v.n_gather;
        d["rows_gathered"] = v.rows_gathered;
        d["bytes_gathered"] = v.bytes_gathered;
        d["oob_skipped"] = v.oob_skipped;
        d["cap_skipped"] = v.cap_skipped;
        return d;
    }

    bool has(const std::string& name) const { return vars_.count(name) > 0; }

    void free_var(const std::string& name) {
        auto it = vars_.find(name);
        TORCH_CHECK(it != vars_.end(), "ddstore: unknown variable '", name, "'");
        release(it->second);
        vars_.erase(it);
    }

    void free_all() {
        for (auto& kv : vars_) release(kv.second);
        vars_.clear();
    }

private:
    void check_new(const std::string& name) {
        TORCH_CHECK(!vars_.count(name), "ddstore: variable '", name, "' already exists");
    }
    HostVar& var(const std::string& name) {
        auto it = vars_.find(name);
        TORCH_CHECK(it != vars_.end(), "ddstore: unknown variable '", name, "'");
        return it->second;
    }
    void check_peers(const HostVar& v) {
        TORCH_CHECK(!v.peers.empty(),
                    "ddstore: variable not finalized (open_peers not called)");
    }

    void create_shm(HostVar& v, const std::string& name, size_t bytes) {
        v.shm_name = shm_name(name);
        TORCH_CHECK(v.shm_name.size() < 250, "ddstore: variable name too long");
        v.base_bytes = bytes < 1 ? 1 : bytes;
        shm_unlink(v.shm_name.c_str());  // stale segment from a crashed run
        int fd = shm_open(v.shm_name.c_str(), O_CREAT | O_EXCL | O_RDWR, 0600);
        TORCH_CHECK(fd >= 0, "ddstore: shm_open(create) failed for ", v.shm_name);
        int rc = ftruncate(fd, (off_t)v.base_bytes);
        if (rc != 0) {
            close(fd);
            shm_unlink(v.shm_name.c_str());
            TORCH_CHECK(false, "ddstore: ftruncate failed (shm too large?)");
        }
        v.base = mmap(nullptr, v.base_bytes, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
        close(fd);
        TORCH_CHECK(v.base != MAP_FAILED, "ddstore: mmap failed");
#if defined(MADV_HUGEPAGE)
        // shm defaults to 4-KB pages; where shmem THP is enabled ("advise")
        // this collapses the random-gather TLB pressure
        (void)madvise(v.base, v.base_bytes, MADV_HUGEPAGE);
#endif
        v.creator = true;
    }

    void release(HostVar& v) {
        for (size_t r = 0; r < v.peers.size(); ++r)
            if (v.peer_bytes[r] > 0 && v.peers[r]) munmap(v.peers[r], v.peer_bytes[r]);
        v.peers.clear();
        if (v.base && v.base != MAP_FAILED) munmap(v.base, v.base_bytes);
        if (v.creator) shm_unlink(v.shm_name.c_str());
        v.base = nullptr;
        v.active = false;
    }

    std::string session_;
    int rank_;
    int nparts_;
    EpochFSM fsm_;
    std::map<std::string, HostVar> vars_;
};

// ===========================================================================
// Cycle traversal of a permutation (host): the in-place chunked reshuffle
// processes slots in cycle order so a bounded buffer suffices -- slot j_i is
// written from slot j_{i+1}=perm[j_i]'s OLD row, and within a contiguous
// cycle traversal every read slot is only overwritten at the same or a later
// position. Returns (order[n], starts[ncycles+1]): `order` is the
// concatenated cycle traversal, `starts` its cycle start positions
// (starts[ncycles] = n). Single O(n) pointer-chasing walk.
// ===========================================================================
template <typename I>
inline void cycle_walk(const I* p, int64_t n, int64_t* order,
                       std::vector<int64_t>& starts) {
    // the walk is a dependent random-access chain (unprefetchable); a
    // 32-bit working copy halves its memory traffic for n < 2^31
    std::vector<bool> visited((size_t)n, false);
    int64_t w = 0;
    for (int64_t s0 = 0; s0 < n; ++s0) {
        if (visited[(size_t)s0]) continue;
        starts.push_back(w);
        int64_t j = s0;
        while (!visited[(size_t)j]) {
            visited[(size_t)j] = true;
            order[w++] = j;
            j = (int64_t)p[j];
            TORCH_CHECK(j >= 0 && j < n,
                        "ddstore cycle_order: not a permutation (value out of range)");
        }
        TORCH_CHECK(j == s0,
                    "ddstore cycle_order: not a permutation (duplicate value)");
    }
}

inline std::pair<at::Tensor, at::Tensor> cycle_order(const at::Tensor& perm) {
    TORCH_CHECK(perm.scalar_type() == at::kLong && perm.is_contiguous() &&
                    perm.device().is_cpu(),
                "ddstore cycle_order: perm must be a contiguous int64 CPU tensor");
    const int64_t n = perm.numel();
    const int64_t* p = perm.data_ptr<int64_t>();
    at::Tensor order_t = at::empty({n}, at::TensorOptions().dtype(at::kLong));
    int64_t* order = order_t.data_ptr<int64_t>();
    std::vector<int64_t> starts;
    if (n <= (int64_t)INT32_MAX) {
        std::vector<int32_t> p32((size_t)n);
        host_parallel_for(0, n, 1 << 20, [&](int64_t b, int64_t e) {
            for (int64_t i = b; i < e; ++i) p32[(size_t)i] = (int32_t)p[i];
        });
        cycle_walk(p32.data(), n, order, starts);
    } else {
        cycle_walk(p, n, order, starts);
    }
    starts.push_back(n);
    at::Tensor starts_t = at::empty({(int64_t)starts.size()},
                                    at::TensorOptions().dtype(at::kLong));
    std::memcpy(starts_t.data_ptr<int64_t>(), starts.data(),
                starts.size() * sizeof(int64_t));
    return {order_t, starts_t};
}

} // namespace ddstore

