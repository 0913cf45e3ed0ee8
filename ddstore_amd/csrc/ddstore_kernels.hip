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
                // keep the true-bytes counter honest (see k_csr_scan)
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
// Fused CSR plan: per-sample lens + exclusive scan -> out_off in ONE kernel
// with a decoupled lookback across workgroup tiles of kBlock samples.
//
// Round-1 measured the separate plan (zeros + lens kernel + torch cumsum) at
// ~25 us/step (B=262144); a serial-lens full fusion was tried and reverted
// (latency-bound), and a lens+scan+GATHER single kernel was measured 2x
// SLOWER than the pipeline (the block-wide phase barriers serialize the
// payload copies against the scan; r2 A/B: 186 vs 98 us/step). Scan-only
// fusion keeps the payload gather streaming in its own kernel while cutting
// the plan to one small launch + an 8-KB memset.
//
// tile_state[t] encodes (value << 2) | flag, flag 1 = aggregate ready,
// 2 = inclusive prefix ready; it must be zeroed before launch. The grid is
// capped to the occupancy-resident block count so every spinning tile's
// producer is guaranteed to be scheduled (forward progress).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(kBlock)
k_csr_scan(const int64_t* goff, int64_t nsamples,
           const int64_t* idx, int64_t nidx,
           int64_t* __restrict__ out_off,
           unsigned long long* ctrs,
           unsigned long long* __restrict__ tile_state) {
    __shared__ int64_t s_wsum[kBlock / 64];
    __shared__ int64_t s_tile_base;
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int64_t ntiles = (nidx + kBlock - 1) / kBlock;
    for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
        const int64_t i = tile * kBlock + threadIdx.x;
        int64_t L = 0;
        if (i < nidx) {
            const int64_t g = idx[i];
            if (g < 0 || g >= nsamples) {
                atomicAdd(ctrs + DDS_CTR_OOB, 1ull);
            } else {
                L = goff[g + 1] - goff[g];
            }
        }
        // 64-wide inclusive wave scan of L, then cross-wave bases via LDS
        int64_t x = L;
        for (int off = 1; off < 64; off <<= 1) {
            int64_t y = __shfl_up(x, off);
            if (lane >= off) x += y;
        }
        if (lane == 63) s_wsum[wave] = x;
        __syncthreads();
        if (threadIdx.x == 0) {
            int64_t acc = 0;
            for (int w = 0; w < kBlock / 64; ++w) {
                int64_t t = s_wsum[w];
                s_wsum[w] = acc;
                acc += t;
            }
            s_tile_base = acc;  // stash the tile aggregate for wave 0
            __hip_atomic_store(&tile_state[tile],
                               ((unsigned long long)acc << 2) | 1ull,
                               __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
        }
        __syncthreads();
        if (wave == 0) {
            // WAVE-PARALLEL lookback: 64 predecessor tiles per round trip.
            // A single-lane walk serializes on L2 atomic latency (~60 ns x
            // ntiles: measured 63 us at 1024 tiles, r2); here lane l reads
            // tile t-l, the wave sums aggregates back to the nearest
            // INCLUSIVE entry, and only continues past a full window of
            // bare aggregates -- the chain cost drops by ~64x.
            const int64_t acc = s_tile_base;
            int64_t excl = 0;
            int64_t t = tile - 1;
            while (t >= 0) {
                const int64_t pos = t - lane;
                unsigned long long v = 0;
                if (pos >= 0) {
                    // agent-scope atomic LOAD: a plain cache-bypassing read
                    // that coalesces across lanes -- an atomicAdd(...,0)
                    // spin is 64 uncoalesced L2 RMWs per window and was
                    // measured to keep the scan at ~57 us (r2)
                    for (;;) {
                        v = __hip_atomic_load(&tile_state[pos],
                                              __ATOMIC_ACQUIRE,
                                              __HIP_MEMORY_SCOPE_AGENT);
                        if ((v & 3ull) != 0ull) break;
                        __builtin_amdgcn_s_sleep(1);
                    }
                }
                // nearest lane holding an inclusive prefix (if any)
                const unsigned long long ball =
                    __ballot((pos >= 0) && (v & 3ull) == 2ull);
                const int incl_lane = ball ? (__ffsll((long long)ball) - 1) : 64;
                int64_t c = (pos >= 0 && lane <= incl_lane) ? (int64_t)(v >> 2) : 0;
                for (int off = 32; off > 0; off >>= 1) c += __shfl_down(c, off);
                c = __shfl(c, 0);
                excl += c;
                if (incl_lane < 64) break;
                t -= 64;
            }
            if (lane == 0) {
                __hip_atomic_store(&tile_state[tile],
                                   ((unsigned long long)(excl + acc) << 2) | 2ull,
                                   __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
                s_tile_base = excl;
                // true-bytes stats: requested elements; the gather kernels
                // subtract the lens of capacity-skipped samples (rare path)
                atomicAdd(ctrs + DDS_CTR_ELEMS, (unsigned long long)acc);
                if (tile == 0) out_off[0] = 0;
            }
        }
        __syncthreads();
        if (i < nidx) out_off[i + 1] = s_tile_base + s_wsum[wave] + x;
        __syncthreads();  // s_wsum reused next tile iteration
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
size_t csr_fused_scratch_bytes(int64_t nidx) {
    const int64_t ntiles = (nidx + kBlock - 1) / kBlock;
    return (size_t)(ntiles > 0 ? ntiles : 1) * sizeof(unsigned long long);
}

void csr_scan(hipStream_t stream,
              const int64_t* d_goff, int64_t nsamples,
              const int64_t* d_idx, int64_t nidx,
              int64_t* d_out_off, unsigned long long* d_ctrs,
              void* d_tiles) {
    if (nidx == 0) return;
    // the decoupled-lookback spin requires every earlier tile's producer
    // block to be scheduled: cap the grid at the occupancy-resident count
    static const int resident = [] {
        int dev = 0;
        (void)hipGetDevice(&dev);
        hipDeviceProp_t prop{};
        (void)hipGetDeviceProperties(&prop, dev);
        int occ = 0;
        (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(
            &occ, reinterpret_cast<const void*>(&k_csr_scan), kBlock, 0);
        if (occ < 1) occ = 1;
        int sms = prop.multiProcessorCount > 0 ? prop.multiProcessorCount : 64;
        return occ * sms;
    }();
    const int64_t ntiles = (nidx + kBlock - 1) / kBlock;
    const int grid = (int)(ntiles < resident ? ntiles : (int64_t)resident);
    hipLaunchKernelGGL(k_csr_scan, dim3(grid), dim3(kBlock), 0, stream,
                       d_goff, nsamples, d_idx, nidx, d_out_off, d_ctrs,
                       reinterpret_cast<unsigned long long*>(d_tiles));
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
