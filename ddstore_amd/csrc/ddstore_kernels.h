// ddstore_amd kernel launcher declarations (host-side interface).
//
// MI355X-native hot path of the store: the reference performs one synchronous
// MPI_Get / fi_read per row (reference: include/ddstore.hpp:229-237,
// src/common.cxx:332-343); here a whole minibatch of rows is gathered by ONE
// kernel launch that reads local + xGMI-peer HBM pointers directly and packs
// (optionally dtype-casting) into a contiguous output buffer.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

// Element dtype enum shared between host dispatch and kernels.
// bool is stored/moved as u8 (values 0/1), matching NumPy's memory layout.
enum DDSType : int {
    DDS_U8 = 0,
    DDS_I32 = 1,
    DDS_I64 = 2,
    DDS_F32 = 3,
    DDS_F64 = 4,
    DDS_F16 = 5,
    DDS_BF16 = 6,
    DDS_F8E4M3 = 7,   // OCP e4m3fn (gfx950-native fp8, NOT MI300X fnuz)
    DDS_F8E5M2 = 8,
    DDS_NUM_TYPES = 9
};

// Max world size the in-kernel directory supports (fits in LDS; one node of
// 8 GPUs is the design point, 128 leaves headroom for oversubscribed tests).
#define DDS_MAX_PARTS 128

// Per-variable device counter block layout (unsigned long long[DDS_NCTR]):
//   [0] oob_skipped  -- out-of-range sample indices skipped by a gather
//   [1] cap_skipped  -- CSR samples skipped because the output capacity
//                       buffer was too small (never write past it)
//   [2] csr_elems    -- true elements gathered by gather_csr_fast (stats;
//                       the capacity-buffer size would over-report)
#define DDS_CTR_OOB 0
#define DDS_CTR_CAP 1
#define DDS_CTR_ELEMS 2
#define DDS_NCTR 3

namespace ddstore {

// Gather `nidx` fixed-stride rows (row = `row_elems` elements of dtype
// `in_t`) addressed by global row ids `d_idx` from the sharded store
// described by (`d_peer_base`, `d_prefix`, `nparts`), packing them
// contiguously into `d_out` with dtype `out_t`.
//   d_peer_base : device array[nparts]   -- shard base pointers (self + IPC peers)
//   d_prefix    : device array[nparts+1] -- global row prefix sums, prefix[0]=0
// d_oob: device counter incremented once per out-of-range index (the row is
// skipped instead of read out of bounds; checked lazily via query()).
void gather_rows(hipStream_t stream,
                 const void* const* d_peer_base,
                 const int64_t* d_prefix, int nparts,
                 const int64_t* d_idx, int64_t nidx,
                 int64_t row_elems, int in_t, int out_t,
                 void* d_out, unsigned long long* d_oob);

// Fused affine gather: out = cast(in) * scale + shift (f32 math; float
// output dtypes only) -- data-loader normalization folded into the fetch.
void gather_rows_affine(hipStream_t stream,
                        const void* const* d_peer_base,
                        const int64_t* d_prefix, int nparts,
                        const int64_t* d_idx, int64_t nidx,
                        int64_t row_elems, int in_t, int out_t,
                        float scale, float shift,
                        void* d_out, unsigned long long* d_oob);

// CSR (variable-length record) gather: sample `g` owns elements
// [d_goff[g], d_goff[g+1]) of the global element space; each element is
// `elem_bytes` bytes (= disp * itemsize). Output element offsets per sample
// are provided in d_out_off (exclusive scan, [nidx+1]). `cap_elems` is the
// output buffer's capacity in elements: a sample whose slice would end past
// it is SKIPPED and counted in d_ctrs[DDS_CTR_CAP] (never written OOB).
//   d_sample_prefix : array[nparts+1] -- per-rank sample-count prefix sums
//   d_elem_prefix   : array[nparts+1] -- per-rank element-count prefix sums
void gather_csr(hipStream_t stream,
                const void* const* d_peer_base,
                const int64_t* d_sample_prefix,
                const int64_t* d_elem_prefix, int nparts,
                const int64_t* d_goff,
                const int64_t* d_idx, int64_t nidx,
                const int64_t* d_out_off,
                int64_t elem_bytes, int64_t cap_elems,
                void* d_out, unsigned long long* d_ctrs);

// lens[i] = goff[idx[i]+1] - goff[idx[i]] (CSR gather plan helper). If
// d_elems != nullptr the true total (sum of lens) is accumulated into it
// (wave-reduced; used by gather_csr_fast for exact byte stats).
void csr_lens(hipStream_t stream, const int64_t* d_goff, const int64_t* d_idx,
              int64_t nidx, int64_t nsamples, int64_t* d_lens,
              unsigned long long* d_oob, unsigned long long* d_elems);

// Balanced one-call CSR fetch: three tiny plan kernels (per-tile lens+scan,
// single-block tile-aggregate scan, finalize + work-item emission) followed
// by an item-parallel gather. Samples are split into <=~256-B work items so
// gather waves get near-uniform work (the per-sample kernel loses ~1.5x to
// max-of-N-samples-per-wave imbalance; r2 fixed-len A/B). Writes
// d_out_off[nidx+1]; accumulates true gathered elements into
// d_ctrs[DDS_CTR_ELEMS]; over-capacity samples are skipped + counted.
// d_scratch: csr_plan_scratch_bytes(nidx) bytes (no zeroing needed);
// d_desc: int64 work-item descriptors, capacity
// cap_elems / csr_item_elems(elem_bytes) + nidx entries.
size_t csr_plan_scratch_bytes(int64_t nidx);
int64_t csr_item_elems(int64_t elem_bytes);
void gather_csr_balanced(hipStream_t stream, const void* const* d_peer_base,
                         const int64_t* d_sample_prefix,
                         const int64_t* d_elem_prefix, int nparts,
                         const int64_t* d_goff, int64_t nsamples_total,
                         const int64_t* d_idx, int64_t nidx,
                         int64_t elem_bytes, int64_t cap_elems,
                         int64_t* d_out_off, void* d_out,
                         unsigned long long* d_ctrs,
                         void* d_scratch, int64_t* d_desc, int64_t desc_cap);


// Scatter rows of a packed buffer into the local shard at arbitrary local row
// ids (inverse of gather_rows with nparts==1). Used by the epoch reshuffle to
// place all-to-all-received rows.
void scatter_rows_local(hipStream_t stream,
                        void* d_base, int64_t nrows_local,
                        int64_t row_elems, int elem_t,
                        const int64_t* d_local_idx, int64_t nidx,
                        const void* d_src, unsigned long long* d_oob);

} // namespace ddstore
