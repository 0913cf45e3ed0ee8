// ddstore_amd CDNA4 (gfx950) gather/pack kernels.
//
// Design notes (MI355X):
//  - These kernels are bandwidth-bound data movers, no MFMA. The store's
//    reference counterpart moved one row per blocking MPI_Get/fi_read
//    (reference: include/ddstore.hpp:229-237, src/common.cxx:332-343). Here
//    one launch moves a whole minibatch: peer shards are directly readable
//    through hipIpc-mapped pointers, so a remote row is a plain vector load
//    that the fabric routes over xGMI.
//  - 16 B/lane loads (uint4) on the fast path: 64-lane wave x 16 B = 1 KiB
//    per instruction, the gfx950 coalescing sweet spot.
//  - The owner directory (world_size+1 prefix sums, <=8 on one node) is
//    staged in LDS; owner lookup is a ~3-7 step binary search over LDS.
//  - Grids are grid-stride with a cap: >>256 workgroups to fill 8 XCDs,
//    capped so small launches don't pay dispatch for idle blocks.

#include "ddstore_kernels.h"

#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>
#include <type_traits>
#include <cstdlib>

namespace ddstore {

namespace {

constexpr int kBlock = 256;
constexpr int64_t kMaxBlocks = 32768;   // 128 blocks/CU worth of headroom

inline int n_blocks(int64_t work_items) {
    int64_t b = (work_items + kBlock - 1) / kBlock;
    if (b < 1) b = 1;
    if (b > kMaxBlocks) b = kMaxBlocks;
    return (int)b;
}

template <typename T, int N>
struct alignas(sizeof(T) * N) VecT {
    T v[N];
};

__device__ __forceinline__ int owner_of(const int64_t* prefix, int nparts, int64_t row) {
    // prefix[0] = 0; find p with prefix[p] <= row < prefix[p+1].
    // (reference equivalent: linear `sortedsearch`, src/ddstore.cxx:5-17)
    int lo = 0, hi = nparts - 1;
    while (lo < hi) {
        int mid = (lo + hi + 1) >> 1;
        if (prefix[mid] <= row) lo = mid; else hi = mid - 1;
    }
    return lo;
}

template <typename T>
constexpr bool is_fp8_v =
    std::is_same_v<T, __hip_fp8_e4m3> || std::is_same_v<T, __hip_fp8_e5m2>;

template <typename Tout, typename Tin>
__device__ __forceinline__ Tout cvt(Tin v) {
    if constexpr (std::is_same_v<Tout, Tin>) {
        return v;
    } else if constexpr (std::is_same_v<Tin, __half>) {
        return cvt<Tout, float>(__half2float(v));
    } else if constexpr (std::is_same_v<Tin, __hip_bfloat16>) {
        return cvt<Tout, float>(__bfloat162float(v));
    } else if constexpr (is_fp8_v<Tin>) {
        return cvt<Tout, float>(static_cast<float>(v));
    } else if constexpr (std::is_same_v<Tout, __half>) {
        return __float2half(static_cast<float>(v));
    } else if constexpr (std::is_same_v<Tout, __hip_bfloat16>) {
        return __float2bfloat16(static_cast<float>(v));
    } else if constexpr (is_fp8_v<Tout>) {
        return Tout(static_cast<float>(v));
    } else {
        return static_cast<Tout>(v);
    }
}

// ---------------------------------------------------------------------------
// Fixed-stride row gather, same dtype, row size a multiple of 16 B.
// Thread t moves 16-B chunk (t % chunks_per_row) of row (t / chunks_per_row):
// consecutive lanes walk consecutive chunks of one row -> coalesced loads and
// stores; the per-thread i64 divide is hidden under the memory traffic.
// ---------------------------------------------------------------------------
// CHUNK = 16 or 32 bytes per thread. 32 B (two dwordx4 loads in flight per
// thread) measured faster than 16 B when rows allow -- same finding as the
// cast kernel, where 32B-read threads beat 16B-read threads on byte rate.
template <int CHUNK>
__global__ void __launch_bounds__(kBlock)
k_gather_rows_b16(const void* const* peer_base, const int64_t* gprefix, int nparts,
                  const int64_t* idx, int64_t nidx, int64_t chunks_per_row,
                  uint4* __restrict__ out, unsigned long long* oob) {
    using V = VecT<uint4, CHUNK / 16>;
    __shared__ int64_t s_prefix[DDS_MAX_PARTS + 1];
    __shared__ const V* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) s_prefix[i] = gprefix[i];
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const V*>(peer_base[i]);
    __syncthreads();

    V* __restrict__ vout = reinterpret_cast<V*>(out);
    const int64_t total = nidx * chunks_per_row;
    for (int64_t t = (int64_t)blockIdx.x * kBlock + threadIdx.x; t < total;
         t += (int64_t)gridDim.x * kBlock) {
        const int64_t r = t / chunks_per_row;
        const int64_t c = t - r * chunks_per_row;
        const int64_t g = idx[r];
        if (g < 0 || g >= s_prefix[nparts]) {  // skip instead of OOB read
            if (c == 0) atomicAdd(oob, 1ull);
            continue;
        }
        const int p = owner_of(s_prefix, nparts, g);
        vout[t] = s_base[p][(g - s_prefix[p]) * chunks_per_row + c];
    }
}

// ---------------------------------------------------------------------------
// Fixed-stride row gather with fused dtype cast, vectorized: each thread
// moves VEC = 16/max(sizeof(Tin), sizeof(Tout)) elements so the wider side
// issues full 16-B accesses and the narrower side 16/ratio-B accesses
// (scalar bf16/f16 loads cost ~2-2.5x -- cdna_hip_programming.md G13).
// Requires row_elems % VEC == 0 (host falls back to the scalar kernel).
// ---------------------------------------------------------------------------
template <typename Tin, typename Tout>
__global__ void __launch_bounds__(kBlock)
k_gather_rows_castv(const void* const* peer_base, const int64_t* gprefix, int nparts,
                    const int64_t* idx, int64_t nidx, int64_t row_elems,
                    Tout* __restrict__ out, unsigned long long* oob) {
    // measured optimum on MI355X: the OUTPUT side exactly 16 B per thread,
    // input side 16*in/out bytes (f32->bf16: 32B-in/16B-out = 21.2 us vs
    // 27.4 us for 16B/8B; u8->f32: 4B-in/16B-out = 20.9 us vs 26.0 us for
    // 16B/64B) -- stores are the side that cannot be split by the memory
    // system, so keep them at dwordx4.
    constexpr int VEC = 16 / sizeof(Tout);
    using Vin = VecT<Tin, VEC>;
    using Vout = VecT<Tout, VEC>;
    __shared__ int64_t s_prefix[DDS_MAX_PARTS + 1];
    __shared__ const Tin* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) s_prefix[i] = gprefix[i];
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const Tin*>(peer_base[i]);
    __syncthreads();

    const int64_t cpr = row_elems / VEC;  // vector chunks per row
    const int64_t total = nidx * cpr;
    for (int64_t t = (int64_t)blockIdx.x * kBlock + threadIdx.x; t < total;
         t += (int64_t)gridDim.x * kBlock) {
        const int64_t r = t / cpr;
        const int64_t c = t - r * cpr;
        const int64_t g = idx[r];
        if (g < 0 || g >= s_prefix[nparts]) {
            if (c == 0) atomicAdd(oob, 1ull);
            continue;
        }
        const int p = owner_of(s_prefix, nparts, g);
        const Vin vin = *reinterpret_cast<const Vin*>(
            s_base[p] + (g - s_prefix[p]) * row_elems + c * VEC);
        Vout vout;
#pragma unroll
        for (int k = 0; k < VEC; ++k) vout.v[k] = cvt<Tout>(vin.v[k]);
        *reinterpret_cast<Vout*>(out + t * VEC) = vout;
    }
}

// ---------------------------------------------------------------------------
// Fused affine gather: out = cast(in) * scale + shift, float outputs only
// (data-loader normalization fused into the fetch: store u8/fp8/f16 samples,
// gather normalized f32/bf16/f16 minibatches in one pass). Math in f32.
// ---------------------------------------------------------------------------
template <typename T>
__device__ __forceinline__ float to_f32(T v) {
    if constexpr (std::is_same_v<T, __half>) return __half2float(v);
    else if constexpr (std::is_same_v<T, __hip_bfloat16>) return __bfloat162float(v);
    else if constexpr (is_fp8_v<T>) return static_cast<float>(v);
    else return (float)v;
}

template <typename Tin, typename Tout>
__global__ void __launch_bounds__(kBlock)
k_gather_rows_affine(const void* const* peer_base, const int64_t* gprefix, int nparts,
                     const int64_t* idx, int64_t nidx, int64_t row_elems,
                     float scale, float shift,
                     Tout* __restrict__ out, unsigned long long* oob) {
    constexpr int VEC = 16 / sizeof(Tout);
    using Vin = VecT<Tin, VEC>;
    using Vout = VecT<Tout, VEC>;
    __shared__ int64_t s_prefix[DDS_MAX_PARTS + 1];
    __shared__ const Tin* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) s_prefix[i] = gprefix[i];
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const Tin*>(peer_base[i]);
    __syncthreads();

    const int64_t cpr = row_elems / VEC;
    const int64_t total = nidx * cpr;
    for (int64_t t = (int64_t)blockIdx.x * kBlock + threadIdx.x; t < total;
         t += (int64_t)gridDim.x * kBlock) {
        const int64_t r = t / cpr;
        const int64_t c = t - r * cpr;
        const int64_t g = idx[r];
        if (g < 0 || g >= s_prefix[nparts]) {
            if (c == 0) atomicAdd(oob, 1ull);
            continue;
        }
        const int p = owner_of(s_prefix, nparts, g);
        const Vin vin = *reinterpret_cast<const Vin*>(
            s_base[p] + (g - s_prefix[p]) * row_elems + c * VEC);
        Vout vout;
#pragma unroll
        for (int k = 0; k < VEC; ++k)
            vout.v[k] = cvt<Tout>(fmaf(to_f32(vin.v[k]), scale, shift));
        *reinterpret_cast<Vout*>(out + t * VEC) = vout;
    }
}

// element-wise fallback for row_elems not divisible by VEC
template <typename Tin, typename Tout>
__global__ void __launch_bounds__(kBlock)
k_gather_rows_affine_s(const void* const* peer_base, const int64_t* gprefix, int nparts,
                       const int64_t* idx, int64_t nidx, int64_t row_elems,
                       float scale, float shift,
                       Tout* __restrict__ out, unsigned long long* oob) {
    __shared__ int64_t s_prefix[DDS_MAX_PARTS + 1];
    __shared__ const Tin* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) s_prefix[i] = gprefix[i];
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const Tin*>(peer_base[i]);
    __syncthreads();
    const int64_t total = nidx * row_elems;
    for (int64_t t = (int64_t)blockIdx.x * kBlock + threadIdx.x; t < total;
         t += (int64_t)gridDim.x * kBlock) {
        const int64_t r = t / row_elems;
        const int64_t c = t - r * row_elems;
        const int64_t g = idx[r];
        if (g < 0 || g >= s_prefix[nparts]) {
            if (c == 0) atomicAdd(oob, 1ull);
            continue;
        }
        const int p = owner_of(s_prefix, nparts, g);
        out[t] = cvt<Tout>(fmaf(
            to_f32(s_base[p][(g - s_prefix[p]) * row_elems + c]), scale, shift));
    }
}

// ---------------------------------------------------------------------------
// Fixed-stride row gather, general: per-element loop with dtype cast.
// ---------------------------------------------------------------------------
template <typename Tin, typename Tout>
__global__ void __launch_bounds__(kBlock)
k_gather_rows_cast(const void* const* peer_base, const int64_t* gprefix, int nparts,
                   const int64_t* idx, int64_t nidx, int64_t row_elems,
                   Tout* __restrict__ out, unsigned long long* oob) {
    __shared__ int64_t s_prefix[DDS_MAX_PARTS + 1];
    __shared__ const Tin* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) s_prefix[i] = gprefix[i];
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const Tin*>(peer_base[i]);
    __syncthreads();

    const int64_t total = nidx * row_elems;
    for (int64_t t = (int64_t)blockIdx.x * kBlock + threadIdx.x; t < total;
         t += (int64_t)gridDim.x * kBlock) {
        const int64_t r = t / row_elems;
        const int64_t c = t - r * row_elems;
        const int64_t g = idx[r];
        if (g < 0 || g >= s_prefix[nparts]) {
            if (c == 0) atomicAdd(oob, 1ull);
            continue;
        }
        const int p = owner_of(s_prefix, nparts, g);
        out[t] = cvt<Tout>(s_base[p][(g - s_prefix[p]) * row_elems + c]);
    }
}

// ---------------------------------------------------------------------------
// CSR gather, chunked: GROUP lanes cooperate per sample (see gather_csr's
// heuristic note; small groups amortize per-sample setup across the wave).
// ---------------------------------------------------------------------------
template <typename T, int GROUP>
__global__ void __launch_bounds__(kBlock)
k_gather_csr_wave(const void* const* peer_base,
                  const int64_t* sample_prefix, const int64_t* elem_prefix, int nparts,
                  const int64_t* goff,
                  const int64_t* idx, int64_t nidx,
                  const int64_t* out_off, int64_t chunks_per_elem,
                  int64_t cap_elems,
                  T* __restrict__ out, unsigned long long* ctrs) {
    __shared__ int64_t s_sprefix[DDS_MAX_PARTS + 1];
    __shared__ int64_t s_eprefix[DDS_MAX_PARTS + 1];
    __shared__ const T* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) {
        s_sprefix[i] = sample_prefix[i];
        s_eprefix[i] = elem_prefix[i];
    }
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const T*>(peer_base[i]);
    __syncthreads();

    constexpr int GPB = kBlock / GROUP;  // sample groups per block
    const int64_t first = (int64_t)blockIdx.x * GPB + threadIdx.x / GROUP;
    const int64_t step = (int64_t)gridDim.x * GPB;
    const int tid = threadIdx.x % GROUP;
    for (int64_t s = first; s < nidx; s += step) {
        const int64_t g = idx[s];
        if (g < 0 || g >= s_sprefix[nparts]) {
            if (tid == 0) atomicAdd(ctrs + DDS_CTR_OOB, 1ull);
            continue;
        }
        const int p = owner_of(s_sprefix, nparts, g);
        const int64_t e0 = goff[g];
        const int64_t len = goff[g + 1] - e0;
        const int64_t o0 = out_off[s];
        if (o0 < 0 || o0 + len > cap_elems) {  // undersized capacity buffer:
            if (tid == 0) {                    // never write OOB
                atomicAdd(ctrs + DDS_CTR_CAP, 1ull);
                // keep the true-bytes counter honest (see k_csr_plan2)
                atomicAdd(ctrs + DDS_CTR_ELEMS,
                          (unsigned long long)(-(long long)len));
            }
            continue;
        }
        const int64_t nch = len * chunks_per_elem;
        const T* src = s_base[p] + (e0 - s_eprefix[p]) * chunks_per_elem;
        T* dst = out + o0 * chunks_per_elem;
        for (int64_t c = tid; c < nch; c += GROUP) dst[c] = src[c];
    }
}

// Per-sample lengths for a CSR gather plan: lens[i] = goff[idx[i]+1] -
// goff[idx[i]] (one kernel instead of the 3-4 elementwise torch launches
// the equivalent `goff[idx+1]-goff[idx]` costs per step).
__global__ void __launch_bounds__(kBlock)
k_csr_lens(const int64_t* goff, const int64_t* idx, int64_t nidx, int64_t nsamples,
           int64_t* __restrict__ lens, unsigned long long* oob,
           unsigned long long* elems) {
    int64_t acc = 0;
    for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nidx;
         i += (int64_t)gridDim.x * kBlock) {
        const int64_t g = idx[i];
        if (g < 0 || g >= nsamples) {
            lens[i] = 0;
            atomicAdd(oob, 1ull);
            continue;
        }
        const int64_t L = goff[g + 1] - goff[g];
        lens[i] = L;
        acc += L;
    }
    if (elems) {
        // wave-reduce (64-wide) then one atomic per wave: exact gathered-
        // element stats without per-thread atomic contention
        for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off);
        if ((threadIdx.x & 63) == 0 && acc)
            atomicAdd(elems, (unsigned long long)acc);
    }
}

// Copy nd dwords src->dst by `nthreads` cooperating threads (tid strided),
// with the STORE side aligned up to dwordx4 (store width is the side the
// memory system cannot split -- same measured rule as the cast kernels):
// head dwords to 16-B store alignment, then a uint4-store body fed by
// align(4) uint4 LOADS -- gfx950 global loads only require dword alignment
// for dwordx4, so the load side also runs at 16 B/lane even though sample
// payload starts are only 4-B-granular (the compiler would scalarize to
// dword loads on targets without unaligned vector access).
struct __attribute__((aligned(4))) DW4A4 {  // 16-B payload, 4-B alignment
    uint32_t v[4];
};

// Copy-body variants for the CSR dword path (selected by DDSTORE_CSR_COPY,
// A/B evidence in profiles/): 0 = dword loads feeding dwordx4 stores
// (round-1 shape), 1 = align(4) dwordx4 loads + dwordx4 stores,
// 2 = variant 1 with nontemporal stores (write-once output bypasses L2).
template <int VAR>
__device__ __forceinline__ void copy_dwords_store16(
    uint32_t* __restrict__ dst, const uint32_t* __restrict__ src,
    int64_t nd, int tid, int nthreads) {
    const int64_t h_align = (int64_t)((4 - ((((uintptr_t)dst) >> 2) & 3)) & 3);
    const int64_t h = h_align < nd ? h_align : nd;
    for (int64_t i = tid; i < h; i += nthreads) dst[i] = src[i];
    const int64_t nb = (nd - h) >> 2;
    uint4* d4 = reinterpret_cast<uint4*>(dst + h);
    if constexpr (VAR == 0) {
        const uint32_t* s2 = src + h;
        for (int64_t i = tid; i < nb; i += nthreads) {
            uint4 o;
            o.x = s2[4 * i];
            o.y = s2[4 * i + 1];
            o.z = s2[4 * i + 2];
            o.w = s2[4 * i + 3];
            d4[i] = o;
        }
    } else {
        const DW4A4* s4 = reinterpret_cast<const DW4A4*>(src + h);
        for (int64_t i = tid; i < nb; i += nthreads) {
            const DW4A4 t = s4[i];  // one align(4) dwordx4 load on gfx950
            uint4 o;
            o.x = t.v[0];
            o.y = t.v[1];
            o.z = t.v[2];
            o.w = t.v[3];
            if constexpr (VAR == 2) {
                typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));
                u32x4 ev = {o.x, o.y, o.z, o.w};
                __builtin_nontemporal_store(
                    ev, reinterpret_cast<u32x4*>(&d4[i]));
            } else {
                d4[i] = o;
            }
        }
    }
    for (int64_t i = h + (nb << 2) + tid; i < nd; i += nthreads) dst[i] = src[i];
}

// CSR gather for 4-B-granular (but not 16-B-aligned) elements: dword
// addressing with 16-B-aligned stores. GROUP lanes cooperate per sample
// (16 for small samples so many samples stay in flight per CU, 64 or the
// whole block for larger payloads).
template <int GROUP, int VAR>
__global__ void __launch_bounds__(kBlock)
k_gather_csr_dw(const void* const* peer_base,
                const int64_t* sample_prefix, const int64_t* elem_prefix, int nparts,
                const int64_t* goff,
                const int64_t* idx, int64_t nidx,
                const int64_t* out_off, int64_t dwords_per_elem,
                int64_t cap_elems,
                uint32_t* __restrict__ out, unsigned long long* ctrs) {
    __shared__ int64_t s_sprefix[DDS_MAX_PARTS + 1];
    __shared__ int64_t s_eprefix[DDS_MAX_PARTS + 1];
    __shared__ const uint32_t* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) {
        s_sprefix[i] = sample_prefix[i];
        s_eprefix[i] = elem_prefix[i];
    }
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const uint32_t*>(peer_base[i]);
    __syncthreads();

    constexpr int GPB = kBlock / GROUP;  // sample groups per block
    const int64_t first = (int64_t)blockIdx.x * GPB + threadIdx.x / GROUP;
    const int64_t step = (int64_t)gridDim.x * GPB;
    const int tid = threadIdx.x % GROUP;
    for (int64_t s = first; s < nidx; s += step) {
        const int64_t g = idx[s];
        if (g < 0 || g >= s_sprefix[nparts]) {
            if (tid == 0) atomicAdd(ctrs + DDS_CTR_OOB, 1ull);
            continue;
        }
        const int p = owner_of(s_sprefix, nparts, g);
        const int64_t e0 = goff[g];
        const int64_t len = goff[g + 1] - e0;
        const int64_t o0 = out_off[s];
        if (o0 < 0 || o0 + len > cap_elems) {
            if (tid == 0) {
                atomicAdd(ctrs + DDS_CTR_CAP, 1ull);
                atomicAdd(ctrs + DDS_CTR_ELEMS,
                          (unsigned long long)(-(long long)len));
            }
            continue;
        }
        copy_dwords_store16<VAR>(out + o0 * dwords_per_elem,
                                 s_base[p] + (e0 - s_eprefix[p]) * dwords_per_elem,
                                 len * dwords_per_elem, tid, GROUP);
    }
}

// ---------------------------------------------------------------------------
// CSR plan: per-sample lens + exclusive scan -> out_off, PLUS a balanced
// work-item list for the gather.
//
// History of this path (all measured at B=262144, ~512 B mean samples):
//   r1: zeros + lens kernel + torch cumsum               ~25 us plan
//   r2a: single fused lens+scan+gather kernel            2x SLOWER overall
//        (block-wide phase barriers serialize the payload copies)
//   r2b: decoupled-lookback scan kernel                  57-83 us (the
//        cross-tile flag traffic dominates regardless of spin flavor)
//   r2c (this): THREE tiny deterministic kernels, no cross-block waiting:
//        k_csr_plan1  per-tile lens + block scan -> partial out_off,
//                     per-tile (elems, items) aggregates
//        k_csr_plan2  one block scans the tile aggregates -> bases, totals
//        k_csr_plan3  finalize out_off (+tile base), capacity check, and
//                     EMIT work items: samples are split into <=ITEM-element
//                     pieces so every gather group gets near-uniform work
//                     (static sample-per-group scheduling loses ~1.5x to
//                     the max-of-4-samples-per-wave effect; fixed-len A/B
//                     r2: 46 vs 75 us for the same mean bytes).
//
// Item descriptor: (sample_index << 20) | piece_index (piece < 2^20 =>
// samples up to 2^20 * ITEM elements; guarded in plan3).
// Scratch layout (csr_plan_scratch_bytes): [aggs 2*ntiles][lens nidx]
// [meta 2] int64 each; meta = {total_items, total_elems}.
// ---------------------------------------------------------------------------
// Work-item size: A/B on MI355X (B=262144, 16..240-elem f32 samples), with
// self-contained descriptors: 256 B -> 2.39G, 512 B -> 2.65G, 1024 B ->
// 2.91G samples/s -- per-item fixed cost beats the wave-balance gain, so
// items hold most whole samples and splitting only trims outliers.
// Also tried and REVERTED (commit history has both sides): (a) fusing the
// aggregate scan into plan1's last block -- the per-block __threadfence
// costs an L2 writeback each, plan1 8 -> 41 us; (b) size-banded item
// emission for wave uniformity -- the per-block same-address atomicAdd
// allocation serializes at ~60 ns/op (plan3 6 -> 58 us) AND the banded
// dst scatter loses write locality (gather 76 -> 85 us).
constexpr int kItemBytes = 1024;  // env override: DDSTORE_CSR_ITEM

__device__ __forceinline__ void block_scan_pair(int64_t& x, int64_t& y,
                                                int64_t* s_wsum2, int lane,
                                                int wave, int64_t* tile_tot) {
    // inclusive block scan of (x, y) over kBlock threads (wave shfl + LDS)
    for (int off = 1; off < 64; off <<= 1) {
        int64_t ax = __shfl_up(x, off);
        int64_t ay = __shfl_up(y, off);
        if (lane >= off) {
            x += ax;
            y += ay;
        }
    }
    if (lane == 63) {
        s_wsum2[2 * wave] = x;
        s_wsum2[2 * wave + 1] = y;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        int64_t ax = 0, ay = 0;
        for (int w = 0; w < kBlock / 64; ++w) {
            int64_t tx = s_wsum2[2 * w], ty = s_wsum2[2 * w + 1];
            s_wsum2[2 * w] = ax;
            s_wsum2[2 * w + 1] = ay;
            ax += tx;
            ay += ty;
        }
        if (tile_tot) {
            tile_tot[0] = ax;
            tile_tot[1] = ay;
        }
    }
    __syncthreads();
    x += s_wsum2[2 * wave];
    y += s_wsum2[2 * wave + 1];
}

__global__ void __launch_bounds__(kBlock)
k_csr_plan1(const int64_t* goff, int64_t nsamples,
            const int64_t* idx, int64_t nidx, int64_t item_elems,
            int64_t* __restrict__ lens_tmp, int64_t* __restrict__ e0_tmp,
            int64_t* __restrict__ aggs, unsigned long long* ctrs) {
    __shared__ int64_t s_wsum2[2 * (kBlock / 64)];
    __shared__ int64_t s_tot[2];
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int64_t ntiles = (nidx + kBlock - 1) / kBlock;
    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t i = tile * kBlock + threadIdx.x;
        int64_t L = 0, e0 = -1;
        if (i < nidx) {
            const int64_t g = idx[i];
            if (g < 0 || g >= nsamples) {
                atomicAdd(ctrs + DDS_CTR_OOB, 1ull);
            } else {
                e0 = goff[g];
                L = goff[g + 1] - e0;
            }
            lens_tmp[i] = L;
            e0_tmp[i] = e0;  // global element start (src side), cached for plan3
        }
        int64_t x = L;
        int64_t y = L > 0 ? (L + item_elems - 1) / item_elems : 0;
        block_scan_pair(x, y, s_wsum2, lane, wave, s_tot);
        // (out_off is written complete, with bases, by plan3)
        if (threadIdx.x == 0) {
            aggs[2 * tile] = s_tot[0];
            aggs[2 * tile + 1] = s_tot[1];
        }
        __syncthreads();  // LDS reused next tile
    }
}

__global__ void __launch_bounds__(kBlock)
k_csr_plan2(int64_t* __restrict__ aggs, int64_t ntiles,
            int64_t* __restrict__ meta, unsigned long long* ctrs) {
    // ONE block: running exclusive scan of the (elems, items) tile
    // aggregates in chunks of kBlock; writes totals into meta
    __shared__ int64_t s_wsum2[2 * (kBlock / 64)];
    __shared__ int64_t s_tot[2];
    __shared__ int64_t s_carry[2];
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    if (threadIdx.x == 0) {
        s_carry[0] = 0;
        s_carry[1] = 0;
    }
    __syncthreads();
    for (int64_t base = 0; base < ntiles; base += kBlock) {
        const int64_t t = base + threadIdx.x;
        int64_t x = t < ntiles ? aggs[2 * t] : 0;
        int64_t y = t < ntiles ? aggs[2 * t + 1] : 0;
        const int64_t mx = x, my = y;
        block_scan_pair(x, y, s_wsum2, lane, wave, s_tot);
        if (t < ntiles) {
            aggs[2 * t] = s_carry[0] + x - mx;      // exclusive base
            aggs[2 * t + 1] = s_carry[1] + y - my;
        }
        __syncthreads();
        if (threadIdx.x == 0) {
            s_carry[0] += s_tot[0];
            s_carry[1] += s_tot[1];
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        meta[0] = s_carry[1];  // total items
        meta[1] = s_carry[0];  // total elems
        // true-bytes stats: requested elements (plan3 subtracts lens of
        // capacity-skipped samples)
        atomicAdd(ctrs + DDS_CTR_ELEMS, (unsigned long long)s_carry[0]);
    }
}

// plan3 emits SELF-CONTAINED item descriptors (2 int64 each) so the gather
// does zero random metadata loads (per-item idx/goff/out_off/owner lookups
// were measured to cost MORE than the imbalance they fixed: 110 vs 75 us):
//   desc[2j]   = (peer << 56) | src_unit_offset   (units: dwords or bytes)
//   desc[2j+1] = (n_units << 44) | dst_unit_offset
__global__ void __launch_bounds__(kBlock)
k_csr_plan3(const int64_t* __restrict__ lens_tmp,
            const int64_t* __restrict__ e0_tmp, int64_t nidx,
            const int64_t* sample_prefix, const int64_t* elem_prefix,
            int nparts, const int64_t* idx,
            int64_t item_elems, int64_t units_per_elem, int64_t cap_elems,
            const int64_t* __restrict__ aggs,
            int64_t* __restrict__ out_off,
            int64_t* __restrict__ desc, int64_t desc_cap,
            unsigned long long* ctrs) {
    __shared__ int64_t s_wsum2[2 * (kBlock / 64)];
    __shared__ int64_t s_sprefix[DDS_MAX_PARTS + 1];
    __shared__ int64_t s_eprefix[DDS_MAX_PARTS + 1];
    for (int i = threadIdx.x; i <= nparts; i += kBlock) {
        s_sprefix[i] = sample_prefix[i];
        s_eprefix[i] = elem_prefix[i];
    }
    __syncthreads();
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int64_t ntiles = (nidx + kBlock - 1) / kBlock;
    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t i = tile * kBlock + threadIdx.x;
        const int64_t L = i < nidx ? lens_tmp[i] : 0;
        int64_t x = L;
        int64_t items = L > 0 ? (L + item_elems - 1) / item_elems : 0;
        int64_t y = items;
        block_scan_pair(x, y, s_wsum2, lane, wave, nullptr);
        const int64_t e_base = aggs[2 * tile], i_base = aggs[2 * tile + 1];
        if (i < nidx) {
            const int64_t excl_e = e_base + x - L;   // global elem offset
            const int64_t excl_i = i_base + y - items;
            out_off[i + 1] = e_base + x;
            if (i == 0) out_off[0] = 0;
            if (L > 0) {
                if (excl_e + L > cap_elems ||
                    excl_i + items > desc_cap ||
                    L * units_per_elem >= (int64_t(1) << 44)) {
                    // over-capacity (or pathological): no items emitted,
                    // never written out of bounds; counted
                    atomicAdd(ctrs + DDS_CTR_CAP, 1ull);
                    atomicAdd(ctrs + DDS_CTR_ELEMS,
                              (unsigned long long)(-(long long)L));
                } else {
                    const int64_t g = idx[i];
                    const int64_t p = owner_of(s_sprefix, nparts, g);
                    const int64_t src0 =
                        (e0_tmp[i] - s_eprefix[p]) * units_per_elem;
                    const int64_t dst0 = excl_e * units_per_elem;
                    const int64_t iu = item_elems * units_per_elem;
                    const int64_t nu = L * units_per_elem;
                    for (int64_t k = 0; k < items; ++k) {
                        const int64_t off = k * iu;
                        const int64_t n = nu - off < iu ? nu - off : iu;
                        desc[2 * (excl_i + k)] = (p << 56) | (src0 + off);
                        desc[2 * (excl_i + k) + 1] = (n << 44) | (dst0 + off);
                    }
                }
            }
        }
        __syncthreads();
    }
}

// Balanced CSR gather: GROUP lanes per work item (one <=ITEM-byte piece of
// one sample, self-contained descriptor); near-uniform work per wave kills
// the max-of-N-samples imbalance of the per-sample kernel, and the
// coalesced 16-B descriptor read is the ONLY metadata access. Item count
// is read from meta on device (the host never syncs for it).
template <int GROUP, bool DW, int VAR>
__global__ void __launch_bounds__(kBlock)
k_gather_csr_items(const void* const* peer_base, int nparts,
                   const int64_t* __restrict__ desc,
                   const int64_t* __restrict__ meta,
                   void* __restrict__ out_, unsigned long long* ctrs) {
    __shared__ const char* s_base[DDS_MAX_PARTS];
    for (int i = threadIdx.x; i < nparts; i += kBlock)
        s_base[i] = reinterpret_cast<const char*>(peer_base[i]);
    __syncthreads();
    const int64_t total = meta[0];
    constexpr int GPB = kBlock / GROUP;
    const int64_t first = (int64_t)blockIdx.x * GPB + threadIdx.x / GROUP;
    const int64_t step = (int64_t)gridDim.x * GPB;
    const int tid = threadIdx.x % GROUP;
    (void)ctrs;
    for (int64_t j = first; j < total; j += step) {
        const int64_t d0 = desc[2 * j];
        const int64_t d1 = desc[2 * j + 1];
        const int p = (int)(d0 >> 56);
        const int64_t src_u = d0 & ((int64_t(1) << 56) - 1);
        const int64_t n_u = d1 >> 44;
        const int64_t dst_u = d1 & ((int64_t(1) << 44) - 1);
        if constexpr (DW) {
            copy_dwords_store16<VAR>(
                reinterpret_cast<uint32_t*>(out_) + dst_u,
                reinterpret_cast<const uint32_t*>(s_base[p]) + src_u,
                n_u, tid, GROUP);
        } else {
            char* dst = reinterpret_cast<char*>(out_) + dst_u;
            const char* src = s_base[p] + src_u;
            for (int64_t c = tid; c < n_u; c += GROUP) dst[c] = src[c];
        }
    }
}

// ---------------------------------------------------------------------------
// Local scatter (reshuffle placement): row r of src -> local row
// local_idx[r] of base. Same chunk mapping as gather.
// ---------------------------------------------------------------------------
template <int CHUNK>
__global__ void __launch_bounds__(kBlock)
k_scatter_rows_b16(uint4* __restrict__ base_, int64_t nrows, int64_t chunks_per_row,
                   const int64_t* local_idx, int64_t nidx,
                   const uint4* __restrict__ src_, unsigned long long* oob) {
    using V = VecT<uint4, CHUNK / 16>;
    V* __restrict__ base = reinterpret_cast<V*>(base_);
    const V* __restrict__ src = reinterpret_cast<const V*>(src_);
    const int64_t total = nidx * chunks_per_row;
    for (int64_t t = (int64_t)blockIdx.x * kBlock + threadIdx.x; t < total;
         t += (int64_t)gridDim.x * kBlock) {
        const int64_t r = t / chunks_per_row;
        const int64_t c = t - r * chunks_per_row;
        const int64_t l = local_idx[r];
        if (l < 0 || l >= nrows) {
            if (c == 0) atomicAdd(oob, 1ull);
            continue;
        }
        base[l * chunks_per_row + c] = src[t];
    }
}

template <typename T>
__global__ void __launch_bounds__(kBlock)
k_scatter_rows_elem(T* __restrict__ base, int64_t nrows, int64_t row_elems,
                    const int64_t* local_idx, int64_t nidx,
                    const T* __restrict__ src, unsigned long long* oob) {
    const int64_t total = nidx * row_elems;
    for (int64_t t = (int64_t)blockIdx.x * kBlock + threadIdx.x; t < total;
         t += (int64_t)gridDim.x * kBlock) {
        const int64_t r = t / row_elems;
        const int64_t c = t - r * row_elems;
        const int64_t l = local_idx[r];
        if (l < 0 || l >= nrows) {
            if (c == 0) atomicAdd(oob, 1ull);
            continue;
        }
        base[l * row_elems + c] = src[t];
    }
}

inline int dds_itemsize(int t) {
    switch (t) {
        case DDS_U8: case DDS_F8E4M3: case DDS_F8E5M2: return 1;
        case DDS_F16: case DDS_BF16: return 2;
        case DDS_I32: case DDS_F32: return 4;
        default: return 8;
    }
}

template <typename Tin, typename Tout>
void launch_gather_cast_one(hipStream_t stream, const void* const* pb,
                            const int64_t* pf, int np, const int64_t* idx,
                            int64_t n, int64_t re, Tout* out,
                            unsigned long long* oob) {
    constexpr int VEC = 16 / sizeof(Tout);
    if (re % VEC == 0 && ((uintptr_t)out % 16) == 0) {
        const int grid = n_blocks(n * (re / VEC));
        hipLaunchKernelGGL((k_gather_rows_castv<Tin, Tout>), dim3(grid), dim3(kBlock),
                           0, stream, pb, pf, np, idx, n, re, out, oob);
    } else {
        const int grid = n_blocks(n * re);
        hipLaunchKernelGGL((k_gather_rows_cast<Tin, Tout>), dim3(grid), dim3(kBlock),
                           0, stream, pb, pf, np, idx, n, re, out, oob);
    }
}

template <typename Tin>
void launch_gather_cast_out(hipStream_t stream, const void* const* pb,
                            const int64_t* pf, int np, const int64_t* idx,
                            int64_t n, int64_t re, int out_t, void* out,
                            unsigned long long* oob) {
    switch (out_t) {
#define DDS_OUT(tag, T)                                                               \
    case tag:                                                                         \
        launch_gather_cast_one<Tin, T>(stream, pb, pf, np, idx, n, re, (T*)out, oob); \
        break;
        DDS_OUT(DDS_U8, uint8_t)
        DDS_OUT(DDS_I32, int32_t)
        DDS_OUT(DDS_I64, int64_t)
        DDS_OUT(DDS_F32, float)
        DDS_OUT(DDS_F64, double)
        DDS_OUT(DDS_F16, __half)
        DDS_OUT(DDS_BF16, __hip_bfloat16)
        DDS_OUT(DDS_F8E4M3, __hip_fp8_e4m3)
        DDS_OUT(DDS_F8E5M2, __hip_fp8_e5m2)
#undef DDS_OUT
    }
}

} // namespace

template <typename Tin>
void launch_affine_out(hipStream_t stream, const void* const* pb, const int64_t* pf,
                       int np, const int64_t* idx, int64_t n, int64_t re,
                       float a, float b, int out_t, void* out,
                       unsigned long long* oob) {
#define DDS_AFF(tag, T)                                                              \
    case tag: {                                                                      \
        constexpr int VEC = 16 / sizeof(T);                                          \
        if (re % VEC == 0 && ((uintptr_t)out % 16) == 0) {                           \
            const int grid = n_blocks(n * (re / VEC));                               \
            hipLaunchKernelGGL((k_gather_rows_affine<Tin, T>), dim3(grid),           \
                               dim3(kBlock), 0, stream, pb, pf, np, idx, n, re, a,   \
                               b, (T*)out, oob);                                     \
        } else {                                                                     \
            const int grid = n_blocks(n * re);                                       \
            hipLaunchKernelGGL((k_gather_rows_affine_s<Tin, T>), dim3(grid),         \
                               dim3(kBlock), 0, stream, pb, pf, np, idx, n, re, a,   \
                               b, (T*)out, oob);                                     \
        }                                                                            \
        break;                                                                       \
    }
    switch (out_t) {
        DDS_AFF(DDS_F32, float)
        DDS_AFF(DDS_F16, __half)
        DDS_AFF(DDS_BF16, __hip_bfloat16)
    }
#undef DDS_AFF
}

void gather_rows_affine(hipStream_t stream,
                        const void* const* d_peer_base,
                        const int64_t* d_prefix, int nparts,
                        const int64_t* d_idx, int64_t nidx,
                        int64_t row_elems, int in_t, int out_t,
                        float scale, float shift,
                        void* d_out, unsigned long long* d_oob) {
    if (nidx == 0 || row_elems == 0) return;
    switch (in_t) {
#define DDS_AIN(tag, T)                                                          \
    case tag:                                                                    \
        launch_affine_out<T>(stream, d_peer_base, d_prefix, nparts, d_idx,       \
                             nidx, row_elems, scale, shift, out_t, d_out,        \
                             d_oob);                                             \
        break;
        DDS_AIN(DDS_U8, uint8_t)
        DDS_AIN(DDS_I32, int32_t)
        DDS_AIN(DDS_I64, int64_t)
        DDS_AIN(DDS_F32, float)
        DDS_AIN(DDS_F64, double)
        DDS_AIN(DDS_F16, __half)
        DDS_AIN(DDS_BF16, __hip_bfloat16)
        DDS_AIN(DDS_F8E4M3, __hip_fp8_e4m3)
        DDS_AIN(DDS_F8E5M2, __hip_fp8_e5m2)
#undef DDS_AIN
    }
}

void gather_rows(hipStream_t stream,
                 const void* const* d_peer_base,
                 const int64_t* d_prefix, int nparts,
                 const int64_t* d_idx, int64_t nidx,
                 int64_t row_elems, int in_t, int out_t,
                 void* d_out, unsigned long long* d_oob) {
    if (nidx == 0 || row_elems == 0) return;
    const int64_t row_bytes = row_elems * dds_itemsize(in_t);
    // the uint4/VecT paths require a 16/32-B-aligned output pointer; an
    // offset view (e.g. buf[1:]) falls through to the cast/scalar kernels,
    // whose own vector path re-checks alignment
    const uintptr_t out_align = (uintptr_t)d_out;
    if (in_t == out_t && row_bytes % 16 == 0 && out_align % 16 == 0) {
        const int64_t cpr = row_bytes / 16;
        // NB: a thread-per-row variant (one idx load, all chunks in flight
        // per thread) was measured SLOWER on MI355X for <=64 B rows (7.5 vs
        // 6.1 us at B=131072): fewer, fatter threads lose more to the
        // row-id load+search serialization than they gain in ILP. The
        // chunk-per-thread mapping is used for every row size, with 32-B
        // chunks when the row allows (two dwordx4 loads in flight/thread).
        // A/B on MI355X (B=131072): 512 B rows -- 16B 30.1us / 32B 28.5 /
        // 64B 33.9; 64 B rows -- 16B 6.1 / 32B 6.7 (too few threads per
        // row). 32-B chunks win once a row has >=4 of them.
        if (row_bytes % 32 == 0 && row_bytes >= 128 && out_align % 32 == 0) {
            const int64_t cpr32 = row_bytes / 32;
            const int grid = n_blocks(nidx * cpr32);
            hipLaunchKernelGGL((k_gather_rows_b16<32>), dim3(grid), dim3(kBlock), 0,
                               stream, d_peer_base, d_prefix, nparts, d_idx, nidx,
                               cpr32, (uint4*)d_out, d_oob);
        } else {
            const int grid = n_blocks(nidx * cpr);
            hipLaunchKernelGGL((k_gather_rows_b16<16>), dim3(grid), dim3(kBlock), 0,
                               stream, d_peer_base, d_prefix, nparts, d_idx, nidx,
                               cpr, (uint4*)d_out, d_oob);
        }
        return;
    }
    switch (in_t) {
#define DDS_IN(tag, T)                                                           \
    case tag:                                                                    \
        launch_gather_cast_out<T>(stream, d_peer_base, d_prefix, nparts, d_idx,  \
                                  nidx, row_elems, out_t, d_out, d_oob);         \
        break;
        DDS_IN(DDS_U8, uint8_t)
        DDS_IN(DDS_I32, int32_t)
        DDS_IN(DDS_I64, int64_t)
        DDS_IN(DDS_F32, float)
        DDS_IN(DDS_F64, double)
        DDS_IN(DDS_F16, __half)
        DDS_IN(DDS_BF16, __hip_bfloat16)
        DDS_IN(DDS_F8E4M3, __hip_fp8_e4m3)
        DDS_IN(DDS_F8E5M2, __hip_fp8_e5m2)
#undef DDS_IN
    }
}

void csr_lens(hipStream_t stream, const int64_t* d_goff, const int64_t* d_idx,
              int64_t nidx, int64_t nsamples, int64_t* d_lens,
              unsigned long long* d_oob, unsigned long long* d_elems) {
    if (nidx == 0) return;
    hipLaunchKernelGGL(k_csr_lens, dim3(n_blocks(nidx)), dim3(kBlock), 0, stream,
                       d_goff, d_idx, nidx, nsamples, d_lens, d_oob, d_elems);
}


void gather_csr(hipStream_t stream,
                const void* const* d_peer_base,
                const int64_t* d_sample_prefix,
                const int64_t* d_elem_prefix, int nparts,
                const int64_t* d_goff,
                const int64_t* d_idx, int64_t nidx,
                const int64_t* d_out_off,
                int64_t elem_bytes, int64_t cap_elems,
                void* d_out, unsigned long long* d_ctrs) {
    if (nidx == 0) return;
    // lanes cooperating per sample: small groups amortize the per-sample
    // setup (row-id load + directory search) across the wave and keep more
    // samples in flight. A/B on MI355X: GROUP=16 wins or ties from 512 B to
    // 32 KiB average samples (2.23 vs 1.44 G samples/s at 512 B vs GROUP=64)
    // -- the only reason to grow the group is chip COVERAGE when the batch
    // has few samples (need >=1024 workgroups across 256 CUs).
    static const int g_override = [] {
        const char* e = getenv("DDSTORE_CSR_GROUP");
        return e ? atoi(e) : 0;
    }();
    int group = nidx >= 16384 ? 16 : (nidx >= 4096 ? 64 : 256);
    if (g_override) group = g_override;
    int64_t b = (nidx + (kBlock / group) - 1) / (kBlock / group);
    const int grid = (int)(b < kMaxBlocks ? b : kMaxBlocks);
    // vector paths reinterpret d_out; a misaligned output (e.g. an offset
    // view) falls back to the next-narrower granularity
    const uintptr_t oa = (uintptr_t)d_out;
#define DDS_CSR_G(T, div, G)                                                         \
    hipLaunchKernelGGL((k_gather_csr_wave<T, G>), dim3(grid), dim3(kBlock), 0,       \
                       stream, d_peer_base, d_sample_prefix, d_elem_prefix, nparts,  \
                       d_goff, d_idx, nidx, d_out_off, elem_bytes / div, cap_elems,  \
                       (T*)d_out, d_ctrs)
#define DDS_CSR_T(T, div)                                                            \
    do {                                                                             \
        if (group == 8) DDS_CSR_G(T, div, 8);                                        \
        else if (group == 16) DDS_CSR_G(T, div, 16);                                  \
        else if (group == 32) DDS_CSR_G(T, div, 32);                                 \
        else if (group == 64) DDS_CSR_G(T, div, 64);                                 \
        else DDS_CSR_G(T, div, 256);                                                 \
    } while (0)
    static const int copy_var = [] {
        const char* e = getenv("DDSTORE_CSR_COPY");
        int v = e ? atoi(e) : 0;  // A/B r2: dword loads won (1.845 vs
        return (v >= 0 && v <= 2) ? v : 0;  // 1.802/1.773 G samples/s)
    }();
#define DDS_CSR_DW_V(G, V)                                                           \
    hipLaunchKernelGGL((k_gather_csr_dw<G, V>), dim3(grid), dim3(kBlock), 0, stream, \
                       d_peer_base, d_sample_prefix, d_elem_prefix, nparts, d_goff,  \
                       d_idx, nidx, d_out_off, elem_bytes / 4, cap_elems,            \
                       (uint32_t*)d_out, d_ctrs)
#define DDS_CSR_DW(G)                                                                \
    do {                                                                             \
        if (copy_var == 0) DDS_CSR_DW_V(G, 0);                                       \
        else if (copy_var == 2) DDS_CSR_DW_V(G, 2);                                  \
        else DDS_CSR_DW_V(G, 1);                                                     \
    } while (0)
    if (elem_bytes % 16 == 0 && oa % 16 == 0) {
        DDS_CSR_T(uint4, 16);
    } else if (elem_bytes % 4 == 0 && oa % 4 == 0) {
        // 4/8-B-granular elements: dword addressing, stores re-aligned to
        // dwordx4 inside each sample's payload
        if (group == 8) DDS_CSR_DW(8);
        else if (group == 16) DDS_CSR_DW(16);
        else if (group == 32) DDS_CSR_DW(32);
        else if (group == 64) DDS_CSR_DW(64);
        else DDS_CSR_DW(256);
    } else {
        DDS_CSR_T(uint8_t, 1);
    }
#undef DDS_CSR_G
#undef DDS_CSR_T
#undef DDS_CSR_DW
#undef DDS_CSR_DW_V
}
size_t csr_plan_scratch_bytes(int64_t nidx) {
    const int64_t ntiles = (nidx + kBlock - 1) / kBlock;
    return (size_t)(2 * (ntiles > 0 ? ntiles : 1) + 2 * (nidx > 0 ? nidx : 1)
                    + 2) * sizeof(int64_t);
}

int64_t csr_item_elems(int64_t elem_bytes) {
    static const int item_bytes = [] {
        const char* e = getenv("DDSTORE_CSR_ITEM");
        int v = e ? atoi(e) : kItemBytes;
        return v >= 16 ? v : kItemBytes;
    }();
    int64_t ie = item_bytes / (elem_bytes > 0 ? elem_bytes : 1);
    return ie > 0 ? ie : 1;
}

void gather_csr_balanced(hipStream_t stream, const void* const* d_peer_base,
                         const int64_t* d_sample_prefix,
                         const int64_t* d_elem_prefix, int nparts,
                         const int64_t* d_goff, int64_t nsamples_total,
                         const int64_t* d_idx, int64_t nidx,
                         int64_t elem_bytes, int64_t cap_elems,
                         int64_t* d_out_off, void* d_out,
                         unsigned long long* d_ctrs,
                         void* d_scratch, int64_t* d_desc, int64_t desc_cap) {
    if (nidx == 0) return;
    const int64_t ntiles = (nidx + kBlock - 1) / kBlock;
    int64_t* aggs = reinterpret_cast<int64_t*>(d_scratch);
    int64_t* lens_tmp = aggs + 2 * ntiles;
    int64_t* e0_tmp = lens_tmp + nidx;
    int64_t* meta = e0_tmp + nidx;
    const int64_t item_elems = csr_item_elems(elem_bytes);
    const uintptr_t oa = (uintptr_t)d_out;
    const bool dw = elem_bytes % 4 == 0 && oa % 4 == 0;
    const int64_t upe = dw ? elem_bytes / 4 : elem_bytes;  // units per elem
    const int g1 = (int)(ntiles < kMaxBlocks ? ntiles : kMaxBlocks);
    hipLaunchKernelGGL(k_csr_plan1, dim3(g1), dim3(kBlock), 0, stream, d_goff,
                       nsamples_total, d_idx, nidx, item_elems, lens_tmp,
                       e0_tmp, aggs, d_ctrs);
    hipLaunchKernelGGL(k_csr_plan2, dim3(1), dim3(kBlock), 0, stream, aggs,
                       ntiles, meta, d_ctrs);
    hipLaunchKernelGGL(k_csr_plan3, dim3(g1), dim3(kBlock), 0, stream,
                       lens_tmp, e0_tmp, nidx, d_sample_prefix, d_elem_prefix,
                       nparts, d_idx, item_elems, upe, cap_elems, aggs,
                       d_out_off, d_desc, desc_cap, d_ctrs);
    static const int item_group = [] {
        const char* e = getenv("DDSTORE_CSR_ITEM_GROUP");
        int v = e ? atoi(e) : 16;
        return (v == 8 || v == 16 || v == 32) ? v : 16;
    }();
    static const int item_var = [] {
        const char* e = getenv("DDSTORE_CSR_COPY");
        int v = e ? atoi(e) : 0;
        return (v >= 0 && v <= 2) ? v : 0;
    }();
    const int64_t gi = (desc_cap + (kBlock / item_group) - 1)
                       / (kBlock / item_group);
    const int g2 = (int)(gi < kMaxBlocks ? gi : kMaxBlocks);
#define DDS_ITEMS_GV(G, V)                                                     \
    hipLaunchKernelGGL((k_gather_csr_items<G, true, V>), dim3(g2),             \
                       dim3(kBlock), 0, stream, d_peer_base, nparts, d_desc,   \
                       meta, d_out, d_ctrs)
#define DDS_ITEMS_G(G)                                                         \
    do {                                                                       \
        if (item_var == 1) DDS_ITEMS_GV(G, 1);                                 \
        else if (item_var == 2) DDS_ITEMS_GV(G, 2);                            \
        else DDS_ITEMS_GV(G, 0);                                               \
    } while (0)
    if (dw) {
        if (item_group == 8) DDS_ITEMS_G(8);
        else if (item_group == 32) DDS_ITEMS_G(32);
        else DDS_ITEMS_G(16);
    } else {
        hipLaunchKernelGGL((k_gather_csr_items<16, false, 0>), dim3(g2),
                           dim3(kBlock), 0, stream, d_peer_base, nparts,
                           d_desc, meta, d_out, d_ctrs);
    }
#undef DDS_ITEMS_GV
#undef DDS_ITEMS_G
}

void scatter_rows_local(hipStream_t stream,
                        void* d_base, int64_t nrows_local,
                        int64_t row_elems, int elem_t,
                        const int64_t* d_local_idx, int64_t nidx,
                        const void* d_src, unsigned long long* d_oob) {
    if (nidx == 0 || row_elems == 0) return;
    const int64_t row_bytes = row_elems * dds_itemsize(elem_t);
    const uintptr_t al = (uintptr_t)d_src | (uintptr_t)d_base;
    if (row_bytes % 32 == 0 && row_bytes >= 128 && al % 32 == 0) {
        const int64_t cpr = row_bytes / 32;
        const int grid = n_blocks(nidx * cpr);
        hipLaunchKernelGGL((k_scatter_rows_b16<32>), dim3(grid), dim3(kBlock), 0,
                           stream, (uint4*)d_base, nrows_local, cpr, d_local_idx,
                           nidx, (const uint4*)d_src, d_oob);
    } else if (row_bytes % 16 == 0 && al % 16 == 0) {
        const int64_t cpr = row_bytes / 16;
        const int grid = n_blocks(nidx * cpr);
        hipLaunchKernelGGL((k_scatter_rows_b16<16>), dim3(grid), dim3(kBlock), 0,
                           stream, (uint4*)d_base, nrows_local, cpr, d_local_idx,
                           nidx, (const uint4*)d_src, d_oob);
    } else {
        const int64_t total_bytes = nidx * row_bytes;
        const int grid = n_blocks(total_bytes);
        hipLaunchKernelGGL((k_scatter_rows_elem<uint8_t>), dim3(grid), dim3(kBlock), 0,
                           stream, (uint8_t*)d_base, nrows_local, row_bytes,
                           d_local_idx, nidx, (const uint8_t*)d_src, d_oob);
    }
}

} // namespace ddstore
