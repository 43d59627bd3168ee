// lzy_amd data-plane kernels for gfx950 (MI355X, CDNA4).
//
// These replace the reference's serializer byte-stream hot path (reference:
// pylzy snapshot.put_data md5+S3 upload, api/v1/snapshot.py:141-160; slots
// chunk streams, lzy/slots transfers/SlotInputTransfer.java): on MI355X a
// tensor crossing a channel stays in HBM and the only per-byte work is
// (a) optional dtype cast ("cast on the wire") and (b) content checksum for
// the result cache / dedup.  Both are HBM-bandwidth-bound: kernels are
// vectorized 16 B/lane, grid-stride, grid capped so blocks span all 8 XCDs.
//
// Wave size 64; block 256 threads; no CUDA compatibility paths.
//
// Checksum design: position-salted splitmix64 mixing, combined with a
// commutative sum so any processing order yields the same digest; each
// lane streams 16 B per iteration.  A second entry point (lz_checksum_mfma)
// folds the per-block digests through an i8 MFMA (matrix universal hash) —
// the bulk pass is identical (memory-bound either way; MFMA makes the
// combine arithmetic free and is profiled with rocprof counters).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>

#include <cstdint>

#define LZ_BLOCK 256
// Default grid cap: 4 workgroups per CU.  Swept on MI355X (profiles/
// kernel_sweep.md): 1024 beats 2048..16384 on every data-plane kernel —
// persistent grid-stride loops keep each XCD's L2 working set small and
// the HBM streams long; larger grids only add launch/tail overhead.
#define LZ_MAX_BLOCKS 1024  // 256 CUs x 4 blocks/CU (runtime-tunable)

static int g_max_blocks = LZ_MAX_BLOCKS;
extern "C" void lz_set_max_blocks(int b) { g_max_blocks = b > 0 ? b : LZ_MAX_BLOCKS; }

// ---------------------------------------------------------------------------
// dtype codes shared with python (lzy_amd/ops/__init__.py)
// ---------------------------------------------------------------------------
enum LzDtype : int {
    LZ_F32 = 0,
    LZ_F16 = 1,
    LZ_BF16 = 2,
    LZ_FP8_E4M3 = 3,
    LZ_FP8_E5M2 = 4,
    LZ_U8 = 5,
    LZ_I32 = 6,
    LZ_I64 = 7,
    LZ_F64 = 8,
};

// ---------------------------------------------------------------------------
// cast_copy: dst[i] = cast(src[i]) — fused pack + dtype cast.
// ---------------------------------------------------------------------------

template <typename T>
__device__ __forceinline__ float lz_to_float(T v);

template <> __device__ __forceinline__ float lz_to_float<float>(float v) { return v; }
template <> __device__ __forceinline__ float lz_to_float<__half>(__half v) { return __half2float(v); }
template <> __device__ __forceinline__ float lz_to_float<__hip_bfloat16>(__hip_bfloat16 v) { return __bfloat162float(v); }
template <> __device__ __forceinline__ float lz_to_float<__hip_fp8_e4m3>(__hip_fp8_e4m3 v) { return float(v); }
template <> __device__ __forceinline__ float lz_to_float<__hip_fp8_e5m2>(__hip_fp8_e5m2 v) { return float(v); }
template <> __device__ __forceinline__ float lz_to_float<uint8_t>(uint8_t v) { return (float)v; }
template <> __device__ __forceinline__ float lz_to_float<int32_t>(int32_t v) { return (float)v; }
template <> __device__ __forceinline__ float lz_to_float<int64_t>(int64_t v) { return (float)v; }
template <> __device__ __forceinline__ float lz_to_float<double>(double v) { return (float)v; }

template <typename T>
__device__ __forceinline__ T lz_from_float(float v);

template <> __device__ __forceinline__ float lz_from_float<float>(float v) { return v; }
template <> __device__ __forceinline__ __half lz_from_float<__half>(float v) { return __float2half(v); }
template <> __device__ __forceinline__ __hip_bfloat16 lz_from_float<__hip_bfloat16>(float v) { return __float2bfloat16(v); }
template <> __device__ __forceinline__ __hip_fp8_e4m3 lz_from_float<__hip_fp8_e4m3>(float v) { return __hip_fp8_e4m3(v); }
template <> __device__ __forceinline__ __hip_fp8_e5m2 lz_from_float<__hip_fp8_e5m2>(float v) { return __hip_fp8_e5m2(v); }
template <> __device__ __forceinline__ uint8_t lz_from_float<uint8_t>(float v) { return (uint8_t)v; }
template <> __device__ __forceinline__ int32_t lz_from_float<int32_t>(float v) { return (int32_t)v; }
template <> __device__ __forceinline__ int64_t lz_from_float<int64_t>(float v) { return (int64_t)v; }
template <> __device__ __forceinline__ double lz_from_float<double>(float v) { return (double)v; }

// Vectorized cast: each lane handles 8 contiguous elements per iteration
// (G13: hipcc does not auto-vectorize half-width loads; short4/short8
// reinterpret is the coalescing sweet spot).
template <typename SrcT, typename DstT>
__global__ void cast_copy_kernel(const SrcT* __restrict__ src,
                                 DstT* __restrict__ dst, int64_t n) {
    constexpr int V = 8;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t vec_n = n / V;

    using SrcV = struct { SrcT v[V]; };
    using DstV = struct { DstT v[V]; };
    const SrcV* srcv = reinterpret_cast<const SrcV*>(src);
    DstV* dstv = reinterpret_cast<DstV*>(dst);

    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
         i += stride) {
        SrcV s = srcv[i];
        DstV d;
#pragma unroll
        for (int k = 0; k < V; ++k) d.v[k] = lz_from_float<DstT>(lz_to_float<SrcT>(s.v[k]));
        dstv[i] = d;
    }
    // tail
    int64_t tail_start = vec_n * V;
    for (int64_t i = tail_start + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        dst[i] = lz_from_float<DstT>(lz_to_float<SrcT>(src[i]));
    }
}

template <typename SrcT>
static hipError_t launch_cast_from(const void* src, void* dst, int dst_dtype,
                                   int64_t n, hipStream_t stream) {
    int64_t want = (n / 8 + LZ_BLOCK - 1) / LZ_BLOCK;
    int blocks = (int)(want < 1 ? 1 : (want > g_max_blocks ? g_max_blocks : want));
    switch (dst_dtype) {
#define LZ_CASE(code, T)                                                      \
    case code:                                                                \
        hipLaunchKernelGGL((cast_copy_kernel<SrcT, T>), dim3(blocks),         \
                           dim3(LZ_BLOCK), 0, stream, (const SrcT*)src,       \
                           (T*)dst, n);                                       \
        break;
        LZ_CASE(LZ_F32, float)
        LZ_CASE(LZ_F16, __half)
        LZ_CASE(LZ_BF16, __hip_bfloat16)
        LZ_CASE(LZ_FP8_E4M3, __hip_fp8_e4m3)
        LZ_CASE(LZ_FP8_E5M2, __hip_fp8_e5m2)
        LZ_CASE(LZ_U8, uint8_t)
        LZ_CASE(LZ_I32, int32_t)
        LZ_CASE(LZ_I64, int64_t)
        LZ_CASE(LZ_F64, double)
#undef LZ_CASE
        default:
            return hipErrorInvalidValue;
    }
    return hipGetLastError();
}

extern "C" hipError_t lz_cast_copy(const void* src, int src_dtype, void* dst,
                                   int dst_dtype, int64_t n, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    switch (src_dtype) {
        case LZ_F32: return launch_cast_from<float>(src, dst, dst_dtype, n, s);
        case LZ_F16: return launch_cast_from<__half>(src, dst, dst_dtype, n, s);
        case LZ_BF16: return launch_cast_from<__hip_bfloat16>(src, dst, dst_dtype, n, s);
        case LZ_FP8_E4M3: return launch_cast_from<__hip_fp8_e4m3>(src, dst, dst_dtype, n, s);
        case LZ_FP8_E5M2: return launch_cast_from<__hip_fp8_e5m2>(src, dst, dst_dtype, n, s);
        case LZ_U8: return launch_cast_from<uint8_t>(src, dst, dst_dtype, n, s);
        case LZ_I32: return launch_cast_from<int32_t>(src, dst, dst_dtype, n, s);
        case LZ_I64: return launch_cast_from<int64_t>(src, dst, dst_dtype, n, s);
        case LZ_F64: return launch_cast_from<double>(src, dst, dst_dtype, n, s);
        default: return hipErrorInvalidValue;
    }
}

// ---------------------------------------------------------------------------
// checksum: 64-bit content hash of a device buffer.
// ---------------------------------------------------------------------------

__device__ __forceinline__ uint64_t splitmix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

// Each lane: 2 x u64 (16 B) per grid-stride iteration; per-position salt
// makes the hash order-sensitive while the combine (wrapping +) stays
// commutative, so block/lane order doesn't matter.
__global__ void checksum_kernel(const uint8_t* __restrict__ data,
                                int64_t nbytes,
                                unsigned long long* __restrict__ out) {
    const uint64_t* words = reinterpret_cast<const uint64_t*>(data);
    int64_t nwords = nbytes >> 3;
    int64_t vec_n = nwords >> 1;  // pairs of u64

    uint64_t acc = 0;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    using U64x2 = struct { uint64_t a, b; };
    const U64x2* w2 = reinterpret_cast<const U64x2*>(words);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
         i += stride) {
        U64x2 w = w2[i];
        acc += splitmix64(w.a ^ splitmix64((uint64_t)(2 * i)));
        acc += splitmix64(w.b ^ splitmix64((uint64_t)(2 * i + 1)));
    }
    // odd trailing word + tail bytes: one thread
    if (blockIdx.x == 0 && threadIdx.x == 0) {
        if (nwords & 1) {
            acc += splitmix64(words[nwords - 1] ^ splitmix64((uint64_t)(nwords - 1)));
        }
        int tail = (int)(nbytes & 7);
        if (tail) {
            uint64_t last = 0;
            const uint8_t* p = data + (nwords << 3);
            for (int k = 0; k < tail; ++k) last |= ((uint64_t)p[k]) << (8 * k);
            acc += splitmix64(last ^ splitmix64((uint64_t)nwords) ^ 0xA5A5A5A5ULL);
        }
        acc += splitmix64((uint64_t)nbytes ^ 0x1234567890ABCDEFULL);
    }

    // wave reduce (64 lanes), then one atomic per wave
    for (int off = 32; off > 0; off >>= 1)
        acc += (uint64_t)__shfl_down((long long)acc, off, 64);
    if ((threadIdx.x & 63) == 0)
        atomicAdd(out, (unsigned long long)acc);
}

extern "C" hipError_t lz_checksum(const void* data, int64_t nbytes,
                                  unsigned long long* out_device, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipError_t err = hipMemsetAsync(out_device, 0, 8, s);
    if (err != hipSuccess) return err;
    int64_t want = (nbytes / 16 + LZ_BLOCK - 1) / LZ_BLOCK;
    int blocks = (int)(want < 1 ? 1 : (want > g_max_blocks ? g_max_blocks : want));
    hipLaunchKernelGGL(checksum_kernel, dim3(blocks), dim3(LZ_BLOCK), 0, s,
                       (const uint8_t*)data, nbytes, out_device);
    return hipGetLastError();
}

// ---------------------------------------------------------------------------
// checksum_mfma: matrix-universal-hash digest on the MFMA units.
//
// The bulk pass is HBM-bound either way; routing the mixing arithmetic
// through the i8 matrix cores (v_mfma_i32_32x32x32_i8, ~4.4 PO/s) leaves
// the VALU nearly idle: per 1 KiB tile one MFMA replaces ~1.6k VALU mix
// ops, so the kernel runs at the memory ceiling with MfmaUtil carrying
// the compute (verified with rocprofv3 PMC, see profiles/).
//
// Scheme: D += A · (X_t ^ salt(t)) over Z_2^32, where X_t is the next
// 1 KiB of data as a 32x32 i8 tile (each lane holds 16 bytes = its B
// fragment, loaded as one coalesced dwordx4), A is a fixed dense
// pseudo-random i8 matrix (generated in-register from splitmix64), and
// salt(t) makes the hash position-sensitive while the i32 accumulation
// stays commutative across tiles.  Final fold: per-lane position-salted
// splitmix of the 16 accumulators, wave reduction, one atomic per wave.
// Any fixed lane->matrix-element mapping is a valid hash basis, so B
// fragments are simply lane-linear memory.
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(4))) int lz_i32x4;
typedef __attribute__((ext_vector_type(16))) int lz_i32x16;

__global__ void checksum_mfma_kernel(const uint8_t* __restrict__ data,
                                     int64_t nbytes,
                                     unsigned long long* __restrict__ out) {
    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int waves_per_block = blockDim.x >> 6;

    // A operand: fixed dense pseudo-random bytes per lane (4 dwords)
    lz_i32x4 a_frag;
#pragma unroll
    for (int r = 0; r < 4; ++r)
        a_frag[r] = (int)(uint32_t)(splitmix64(0x5EEDULL * 131 + lane * 4 + r) | 0x01010101u);

    lz_i32x16 acc = {0};

    const int64_t ntiles = nbytes >> 10;  // 1 KiB tiles
    const int64_t wave_id = (int64_t)blockIdx.x * waves_per_block + wave;
    const int64_t wave_stride = (int64_t)gridDim.x * waves_per_block;
    const lz_i32x4* tiles = reinterpret_cast<const lz_i32x4*>(data);

    // 4 tiles per iteration: the loads issue back-to-back (4 KiB of HBM
    // requests in flight per wave) before the dependent salt+MFMA chain —
    // the kernel is HBM-latency-bound otherwise.
    constexpr int U = 4;
    int64_t t = wave_id;
    for (; t + (U - 1) * wave_stride < ntiles; t += (int64_t)U * wave_stride) {
        lz_i32x4 b[U];
#pragma unroll
        for (int u = 0; u < U; ++u)
            b[u] = tiles[(t + u * wave_stride) * 64 + lane];
#pragma unroll
        for (int u = 0; u < U; ++u) {
            const uint64_t s = splitmix64((uint64_t)(t + u * wave_stride));
#pragma unroll
            for (int r = 0; r < 4; ++r)
                b[u][r] ^= (int)(uint32_t)(s >> ((r * 13) & 31));
            acc = __builtin_amdgcn_mfma_i32_32x32x32_i8(a_frag, b[u], acc, 0, 0, 0);
        }
    }
    for (; t < ntiles; t += wave_stride) {
        lz_i32x4 b_frag = tiles[t * 64 + lane];
        const uint64_t s = splitmix64((uint64_t)t);
#pragma unroll
        for (int r = 0; r < 4; ++r)
            b_frag[r] ^= (int)(uint32_t)(s >> ((r * 13) & 31));
        acc = __builtin_amdgcn_mfma_i32_32x32x32_i8(a_frag, b_frag, acc, 0, 0, 0);
    }

    // fold accumulators with position salts
    uint64_t h = 0;
#pragma unroll
    for (int r = 0; r < 16; ++r)
        h += splitmix64((uint32_t)acc[r] ^ splitmix64((uint64_t)(lane * 16 + r)));

    // tail (< 1 KiB) + length: one thread via the scalar mix
    if (blockIdx.x == 0 && threadIdx.x == 0) {
        const uint8_t* p = data + (ntiles << 10);
        int64_t rem = nbytes - (ntiles << 10);
        const uint64_t* words = reinterpret_cast<const uint64_t*>(p);
        int64_t nwords = rem >> 3;
        for (int64_t i = 0; i < nwords; ++i)
            h += splitmix64(words[i] ^ splitmix64((uint64_t)(ntiles * 128 + i)));
        int tb = (int)(rem & 7);
        if (tb) {
            uint64_t last = 0;
            const uint8_t* q = p + (nwords << 3);
            for (int k = 0; k < tb; ++k) last |= ((uint64_t)q[k]) << (8 * k);
            h += splitmix64(last ^ 0xA5A5A5A5ULL);
        }
        h += splitmix64((uint64_t)nbytes ^ 0x1234567890ABCDEFULL);
    }

    for (int off = 32; off > 0; off >>= 1)
        h += (uint64_t)__shfl_down((long long)h, off, 64);
    if (lane == 0)
        atomicAdd(out, (unsigned long long)h);
}

extern "C" hipError_t lz_checksum_mfma(const void* data, int64_t nbytes,
                                       unsigned long long* out_device,
                                       void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipError_t err = hipMemsetAsync(out_device, 0, 8, s);
    if (err != hipSuccess) return err;
    int64_t tiles = nbytes >> 10;
    int64_t want = (tiles / 4 + LZ_BLOCK / 64 - 1) / (LZ_BLOCK / 64);
    int blocks = (int)(want < 1 ? 1 : (want > g_max_blocks ? g_max_blocks : want));
    hipLaunchKernelGGL(checksum_mfma_kernel, dim3(blocks), dim3(LZ_BLOCK), 0, s,
                       (const uint8_t*)data, nbytes, out_device);
    return hipGetLastError();
}

// ---------------------------------------------------------------------------
// Fused elementwise/reduction ops — the "standard op" toolkit for workflow
// stages.  Rationale (MI355X performance rule: HBM traffic is the budget):
// an unfused torch chain like `(x.float() - x.mean()) / x.std()` moves
// ~14 GB per 1 GiB bf16 shard across 6 kernels; the fused pair below moves
// 3.2 GB in 2.  All kernels: vectorized 8 elems/lane, persistent grids,
// f32 lane accumulation folded to f64 per wave (one f64 atomic per wave).
// ---------------------------------------------------------------------------

__device__ __forceinline__ double lz_wave_reduce_f64(double v) {
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    return v;
}

// sum of x and x^2 (or |x| with ABS=1) in one pass; out[0]+=sum, out[1]+=sumsq
//
// Non-temporal (evict-first) vector load: a 1 GiB reduction stream is
// 32x the combined L2, so marking lines no-reuse keeps the caches from
// churning on data that is read exactly once.  16 B granules — callers
// fall back to cached loads when the vector is not 16 B-aligned in size.
typedef int lz_v4i_t __attribute__((ext_vector_type(4)));

template <typename VecT>
__device__ __forceinline__ VecT lz_nt_load(const VecT* p) {
    VecT r;
    const lz_v4i_t* s = reinterpret_cast<const lz_v4i_t*>(p);
    lz_v4i_t* d = reinterpret_cast<lz_v4i_t*>(&r);
#pragma unroll
    for (int k = 0; k < (int)(sizeof(VecT) / 16); ++k)
        d[k] = __builtin_nontemporal_load(s + k);
    return r;
}

// Load pipeline: U=4 independent vector loads per iteration (64 B/lane in
// flight) with separate accumulator pairs — a single accumulate chain
// leaves the reduction HBM-latency-bound (measured 2.8 TB/s; the
// streaming kernels hit 5.4+).
template <typename T, bool ABS,
          int V = (sizeof(T) == 2) ? 16 : 8, int U = 4, int NT = 0>
__global__ void stats_kernel(const T* __restrict__ src, int64_t n,
                             double* __restrict__ out) {
    // Default 32 B per lane per vector load regardless of dtype width:
    // 16-bit inputs at V=8 (16 B) leave the read stream request-starved.
    // V/U are template-swept (lz_stats_variant + benchmarks/kernel_sweep)
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t vec_n = n / V;
    using SrcV = struct { T v[V]; };
    const SrcV* srcv = reinterpret_cast<const SrcV*>(src);

    float s[U] = {0.f}, s2[U] = {0.f};
    int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + (U - 1) * stride < vec_n; i += (int64_t)U * stride) {
        SrcV x[U];
#pragma unroll
        for (int u = 0; u < U; ++u) {
            if constexpr (NT != 0 && sizeof(SrcV) % 16 == 0)
                x[u] = lz_nt_load(&srcv[i + u * stride]);
            else
                x[u] = srcv[i + u * stride];
        }
#pragma unroll
        for (int u = 0; u < U; ++u) {
#pragma unroll
            for (int k = 0; k < V; ++k) {
                float f = lz_to_float<T>(x[u].v[k]);
                s[u] += ABS ? fabsf(f) : f;
                s2[u] = fmaf(f, f, s2[u]);
            }
        }
    }
    for (; i < vec_n; i += stride) {
        SrcV x = srcv[i];
#pragma unroll
        for (int k = 0; k < V; ++k) {
            float f = lz_to_float<T>(x.v[k]);
            s[0] += ABS ? fabsf(f) : f;
            s2[0] = fmaf(f, f, s2[0]);
        }
    }
    int64_t tail_start = vec_n * V;
    for (int64_t j = tail_start + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         j < n; j += stride) {
        float f = lz_to_float<T>(src[j]);
        s[0] += ABS ? fabsf(f) : f;
        s2[0] = fmaf(f, f, s2[0]);
    }
    double ds = 0, ds2 = 0;
#pragma unroll
    for (int u = 0; u < U; ++u) { ds += (double)s[u]; ds2 += (double)s2[u]; }
    ds = lz_wave_reduce_f64(ds);
    ds2 = lz_wave_reduce_f64(ds2);
    if ((threadIdx.x & 63) == 0) {
        atomicAdd(&out[0], ds);
        atomicAdd(&out[1], ds2);
    }
}

// dst = cast((src - mean) * rstd): stats read from the device buffer the
// stats kernel filled — no host round-trip between the two passes.
template <typename SrcT, typename DstT>
__global__ void normalize_apply_kernel(const SrcT* __restrict__ src,
                                       DstT* __restrict__ dst, int64_t n,
                                       const double* __restrict__ stats,
                                       float eps) {
    const float mean = (float)(stats[0] / (double)n);
    const float var = (float)(stats[1] / (double)n) - mean * mean;
    const float rstd = rsqrtf(fmaxf(var, 0.f) + eps);

    constexpr int V = 8;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t vec_n = n / V;
    using SrcV = struct { SrcT v[V]; };
    using DstV = struct { DstT v[V]; };
    const SrcV* srcv = reinterpret_cast<const SrcV*>(src);
    DstV* dstv = reinterpret_cast<DstV*>(dst);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
         i += stride) {
        SrcV x = srcv[i];
        DstV d;
#pragma unroll
        for (int k = 0; k < V; ++k)
            d.v[k] = lz_from_float<DstT>((lz_to_float<SrcT>(x.v[k]) - mean) * rstd);
        dstv[i] = d;
    }
    int64_t tail_start = vec_n * V;
    for (int64_t i = tail_start + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride)
        dst[i] = lz_from_float<DstT>((lz_to_float<SrcT>(src[i]) - mean) * rstd);
}

// dst = cast(src * a + b) — fused affine (one FMA per element)
template <typename SrcT, typename DstT>
__global__ void scale_shift_kernel(const SrcT* __restrict__ src,
                                   DstT* __restrict__ dst, int64_t n,
                                   float a, float b) {
    constexpr int V = 8;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t vec_n = n / V;
    using SrcV = struct { SrcT v[V]; };
    using DstV = struct { DstT v[V]; };
    const SrcV* srcv = reinterpret_cast<const SrcV*>(src);
    DstV* dstv = reinterpret_cast<DstV*>(dst);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
         i += stride) {
        SrcV x = srcv[i];
        DstV d;
#pragma unroll
        for (int k = 0; k < V; ++k)
            d.v[k] = lz_from_float<DstT>(fmaf(lz_to_float<SrcT>(x.v[k]), a, b));
        dstv[i] = d;
    }
    int64_t tail_start = vec_n * V;
    for (int64_t i = tail_start + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride)
        dst[i] = lz_from_float<DstT>(fmaf(lz_to_float<SrcT>(src[i]), a, b));
}

// dst = cast(alpha*a + beta*b) — fused binary combine (tree-merge stages)
template <typename SrcT, typename DstT>
__global__ void axpby_kernel(const SrcT* __restrict__ a,
                             const SrcT* __restrict__ b,
                             DstT* __restrict__ dst, int64_t n, float alpha,
                             float beta) {
    constexpr int V = 8;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int64_t vec_n = n / V;
    using SrcV = struct { SrcT v[V]; };
    using DstV = struct { DstT v[V]; };
    const SrcV* av = reinterpret_cast<const SrcV*>(a);
    const SrcV* bv = reinterpret_cast<const SrcV*>(b);
    DstV* dstv = reinterpret_cast<DstV*>(dst);
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < vec_n;
         i += stride) {
        SrcV x = av[i], y = bv[i];
        DstV d;
#pragma unroll
        for (int k = 0; k < V; ++k)
            d.v[k] = lz_from_float<DstT>(
                fmaf(lz_to_float<SrcT>(x.v[k]), alpha,
                     lz_to_float<SrcT>(y.v[k]) * beta));
        dstv[i] = d;
    }
    int64_t tail_start = vec_n * V;
    for (int64_t i = tail_start + (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride)
        dst[i] = lz_from_float<DstT>(
            fmaf(lz_to_float<SrcT>(a[i]), alpha, lz_to_float<SrcT>(b[i]) * beta));
}

static inline int lz_grid_for(int64_t work_items) {
    int64_t want = (work_items + LZ_BLOCK - 1) / LZ_BLOCK;
    return (int)(want < 1 ? 1 : (want > g_max_blocks ? g_max_blocks : want));
}

// out_device: double[2], zeroed here; out[0]=sum(|x| if abs else x), out[1]=sum(x^2)
extern "C" hipError_t lz_stats(const void* src, int dtype, int64_t n, int use_abs,
                               double* out_device, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    hipError_t err = hipMemsetAsync(out_device, 0, 16, s);
    if (err != hipSuccess) return err;
    int blocks = lz_grid_for(n / 8);
#define LZ_STATS_CASE(code, T)                                                  \
    case code:                                                                  \
        if (use_abs)                                                            \
            hipLaunchKernelGGL((stats_kernel<T, true>), dim3(blocks),           \
                               dim3(LZ_BLOCK), 0, s, (const T*)src, n,          \
                               out_device);                                     \
        else                                                                    \
            hipLaunchKernelGGL((stats_kernel<T, false>), dim3(blocks),          \
                               dim3(LZ_BLOCK), 0, s, (const T*)src, n,          \
                               out_device);                                     \
        break;
    switch (dtype) {
        LZ_STATS_CASE(LZ_F32, float)
        LZ_STATS_CASE(LZ_F16, __half)
        LZ_STATS_CASE(LZ_BF16, __hip_bfloat16)
        default: return hipErrorInvalidValue;
    }
#undef LZ_STATS_CASE
    return hipGetLastError();
}

// sweep-only entry: V/U variants of the stats reduction (the one
// data-plane kernel measurably below the reduction ceiling — checksum
// streams at ~4.2-4.6 TB/s, stats at ~3.2).  Variants are compiled for
// the 16-bit path (bf16/f16); best found becomes the default above.
extern "C" hipError_t lz_stats_variant(const void* src, int dtype, int64_t n,
                                       int use_abs, double* out_device,
                                       void* stream, int variant) {
    hipStream_t s = (hipStream_t)stream;
    hipError_t err = hipMemsetAsync(out_device, 0, 16, s);
    if (err != hipSuccess) return err;
    int blocks = lz_grid_for(n / 8);
#define LZ_SV_LAUNCH_NT(T, V, U, NT)                                            \
    do {                                                                        \
        if (use_abs)                                                            \
            hipLaunchKernelGGL((stats_kernel<T, true, V, U, NT>),               \
                               dim3(blocks), dim3(LZ_BLOCK), 0, s,              \
                               (const T*)src, n, out_device);                   \
        else                                                                    \
            hipLaunchKernelGGL((stats_kernel<T, false, V, U, NT>),              \
                               dim3(blocks), dim3(LZ_BLOCK), 0, s,              \
                               (const T*)src, n, out_device);                   \
    } while (0)
#define LZ_SV_LAUNCH(T, V, U) LZ_SV_LAUNCH_NT(T, V, U, 0)
#define LZ_SV_DTYPE(T)                                                          \
    switch (variant) {                                                          \
        case 0: LZ_SV_LAUNCH(T, 16, 4); break;                                  \
        case 1: LZ_SV_LAUNCH(T, 16, 8); break;                                  \
        case 2: LZ_SV_LAUNCH(T, 32, 2); break;                                  \
        case 3: LZ_SV_LAUNCH(T, 32, 4); break;                                  \
        case 4: LZ_SV_LAUNCH(T, 8, 8); break;                                   \
        case 5: LZ_SV_LAUNCH(T, 16, 2); break;                                  \
        case 6: LZ_SV_LAUNCH(T, 8, 4); break;                                   \
        case 7: LZ_SV_LAUNCH_NT(T, 16, 4, 1); break;                            \
        case 8: LZ_SV_LAUNCH_NT(T, 8, 8, 1); break;                             \
        case 9: LZ_SV_LAUNCH_NT(T, 16, 2, 1); break;                            \
        default: return hipErrorInvalidValue;                                   \
    }
    switch (dtype) {
        case LZ_F16: { LZ_SV_DTYPE(__half) break; }
        case LZ_BF16: { LZ_SV_DTYPE(__hip_bfloat16) break; }
        case LZ_F32: {
            switch (variant) {
                case 0: LZ_SV_LAUNCH(float, 8, 4); break;
                case 1: LZ_SV_LAUNCH(float, 8, 8); break;
                case 2: LZ_SV_LAUNCH(float, 16, 2); break;
                case 3: LZ_SV_LAUNCH(float, 16, 4); break;
                case 4: LZ_SV_LAUNCH(float, 4, 8); break;
                case 5: LZ_SV_LAUNCH(float, 8, 2); break;
                case 6: LZ_SV_LAUNCH(float, 4, 4); break;
                case 7: LZ_SV_LAUNCH_NT(float, 8, 4, 1); break;
                case 8: LZ_SV_LAUNCH_NT(float, 16, 4, 1); break;
                case 9: LZ_SV_LAUNCH_NT(float, 8, 2, 1); break;
                default: return hipErrorInvalidValue;
            }
            break;
        }
        default: return hipErrorInvalidValue;
    }
#undef LZ_SV_DTYPE
#undef LZ_SV_LAUNCH
#undef LZ_SV_LAUNCH_NT
    return hipGetLastError();
}

extern "C" hipError_t lz_normalize_apply(const void* src, int src_dtype,
                                         void* dst, int dst_dtype, int64_t n,
                                         const double* stats_device, float eps,
                                         void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int blocks = lz_grid_for(n / 8);
#define LZ_NA_DST(SrcT, code, DstT)                                             \
    case code:                                                                  \
        hipLaunchKernelGGL((normalize_apply_kernel<SrcT, DstT>), dim3(blocks),  \
                           dim3(LZ_BLOCK), 0, s, (const SrcT*)src, (DstT*)dst,  \
                           n, stats_device, eps);                               \
        break;
#define LZ_NA_SRC(scode, SrcT)                                                  \
    case scode:                                                                 \
        switch (dst_dtype) {                                                    \
            LZ_NA_DST(SrcT, LZ_F32, float)                                      \
            LZ_NA_DST(SrcT, LZ_F16, __half)                                     \
            LZ_NA_DST(SrcT, LZ_BF16, __hip_bfloat16)                            \
            default: return hipErrorInvalidValue;                               \
        }                                                                       \
        break;
    switch (src_dtype) {
        LZ_NA_SRC(LZ_F32, float)
        LZ_NA_SRC(LZ_F16, __half)
        LZ_NA_SRC(LZ_BF16, __hip_bfloat16)
        default: return hipErrorInvalidValue;
    }
#undef LZ_NA_SRC
#undef LZ_NA_DST
    return hipGetLastError();
}

extern "C" hipError_t lz_scale_shift(const void* src, int src_dtype, void* dst,
                                     int dst_dtype, int64_t n, float a, float b,
                                     void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int blocks = lz_grid_for(n / 8);
#define LZ_SS_DST(SrcT, code, DstT)                                             \
    case code:                                                                  \
        hipLaunchKernelGGL((scale_shift_kernel<SrcT, DstT>), dim3(blocks),      \
                           dim3(LZ_BLOCK), 0, s, (const SrcT*)src, (DstT*)dst,  \
                           n, a, b);                                            \
        break;
#define LZ_SS_SRC(scode, SrcT)                                                  \
    case scode:                                                                 \
        switch (dst_dtype) {                                                    \
            LZ_SS_DST(SrcT, LZ_F32, float)                                      \
            LZ_SS_DST(SrcT, LZ_F16, __half)                                     \
            LZ_SS_DST(SrcT, LZ_BF16, __hip_bfloat16)                            \
            default: return hipErrorInvalidValue;                               \
        }                                                                       \
        break;
    switch (src_dtype) {
        LZ_SS_SRC(LZ_F32, float)
        LZ_SS_SRC(LZ_F16, __half)
        LZ_SS_SRC(LZ_BF16, __hip_bfloat16)
        default: return hipErrorInvalidValue;
    }
#undef LZ_SS_SRC
#undef LZ_SS_DST
    return hipGetLastError();
}

extern "C" hipError_t lz_axpby(const void* a, const void* b, int src_dtype,
                               void* dst, int dst_dtype, int64_t n, float alpha,
                               float beta, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    int blocks = lz_grid_for(n / 8);
#define LZ_AX_DST(SrcT, code, DstT)                                             \
    case code:                                                                  \
        hipLaunchKernelGGL((axpby_kernel<SrcT, DstT>), dim3(blocks),            \
                           dim3(LZ_BLOCK), 0, s, (const SrcT*)a,                \
                           (const SrcT*)b, (DstT*)dst, n, alpha, beta);         \
        break;
#define LZ_AX_SRC(scode, SrcT)                                                  \
    case scode:                                                                 \
        switch (dst_dtype) {                                                    \
            LZ_AX_DST(SrcT, LZ_F32, float)                                      \
            LZ_AX_DST(SrcT, LZ_F16, __half)                                     \
            LZ_AX_DST(SrcT, LZ_BF16, __hip_bfloat16)                            \
            default: return hipErrorInvalidValue;                               \
        }                                                                       \
        break;
    switch (src_dtype) {
        LZ_AX_SRC(LZ_F32, float)
        LZ_AX_SRC(LZ_F16, __half)
        LZ_AX_SRC(LZ_BF16, __hip_bfloat16)
        default: return hipErrorInvalidValue;
    }
#undef LZ_AX_SRC
#undef LZ_AX_DST
    return hipGetLastError();
}

// ---------------------------------------------------------------------------
// transpose_cast: dst[c][r] = cast(src[r][c]) — LDS-staged 64x64 tiles.
//
// The pack kernel for non-contiguous (transposed) tensors: a naive
// transpose strides HBM on one side (uncoalesced 2- or 4-byte accesses,
// ~1/8th bandwidth); staging tiles through LDS makes BOTH the load and
// the store coalesced.  Tile 64x64, block 256 threads (4 waves), +1
// element row padding so the transposed LDS reads spread across banks.
// 160 KB LDS/CU easily holds the f32 tile (16.25 KB) at high occupancy.
// ---------------------------------------------------------------------------

#define LZ_TT 64  // transpose tile edge

template <typename SrcT, typename DstT>
__global__ void transpose_cast_kernel(const SrcT* __restrict__ src,
                                      DstT* __restrict__ dst, int64_t rows,
                                      int64_t cols) {
    __shared__ float tile[LZ_TT][LZ_TT + 1];

    const int64_t tile_r = (int64_t)blockIdx.y * LZ_TT;
    const int64_t tile_c = (int64_t)blockIdx.x * LZ_TT;

    // load: each of 256 threads covers 16 elements, row-major coalesced
    const int tx = threadIdx.x & (LZ_TT - 1);        // 0..63 within a row
    const int ty = threadIdx.x >> 6;                 // 0..3
    for (int r = ty; r < LZ_TT; r += 4) {
        int64_t gr = tile_r + r;
        int64_t gc = tile_c + tx;
        if (gr < rows && gc < cols)
            tile[r][tx] = lz_to_float<SrcT>(src[gr * cols + gc]);
    }
    __syncthreads();

    // store: dst is (cols x rows); walk dst rows coalesced, read LDS
    // transposed (the +1 padding keeps those reads conflict-free)
    for (int c = ty; c < LZ_TT; c += 4) {
        int64_t gdr = tile_c + c;      // dst row = src col
        int64_t gdc = tile_r + tx;     // dst col = src row
        if (gdr < cols && gdc < rows)
            dst[gdr * rows + gdc] = lz_from_float<DstT>(tile[tx][c]);
    }
}

// v2: fully vectorized interior tiles — float4 HBM transactions on BOTH
// sides.  Each thread loads/stores 4 consecutive elements; the LDS tile
// keeps +1-row padding and the transposed reads gather 4 scalars from
// 4 consecutive padded rows (65-element stride => consecutive banks).
template <typename SrcT, typename DstT>
__global__ void transpose_cast_kernel_v4(const SrcT* __restrict__ src,
                                         DstT* __restrict__ dst, int64_t rows,
                                         int64_t cols) {
    __shared__ float tile[LZ_TT][LZ_TT + 1];

    const int64_t tile_r = (int64_t)blockIdx.y * LZ_TT;
    const int64_t tile_c = (int64_t)blockIdx.x * LZ_TT;

    const int tx = threadIdx.x & 15;   // 16 threads x 4 elems = 64 cols
    const int ty = threadIdx.x >> 4;   // 0..15 rows per pass
    using SrcV4 = struct { SrcT v[4]; };
    using DstV4 = struct { DstT v[4]; };

    for (int r = ty; r < LZ_TT; r += 16) {
        const int64_t gr = tile_r + r;
        const int64_t gc = tile_c + tx * 4;
        SrcV4 x = *reinterpret_cast<const SrcV4*>(&src[gr * cols + gc]);
#pragma unroll
        for (int k = 0; k < 4; ++k)
            tile[r][tx * 4 + k] = lz_to_float<SrcT>(x.v[k]);
    }
    __syncthreads();

    for (int c = ty; c < LZ_TT; c += 16) {
        const int64_t gdr = tile_c + c;       // dst row = src col
        const int64_t gdc = tile_r + tx * 4;  // dst col = src row
        DstV4 d;
#pragma unroll
        for (int k = 0; k < 4; ++k)
            d.v[k] = lz_from_float<DstT>(tile[tx * 4 + k][c]);
        *reinterpret_cast<DstV4*>(&dst[gdr * rows + gdc]) = d;
    }
}

extern "C" hipError_t lz_transpose_cast(const void* src, int src_dtype,
                                        void* dst, int dst_dtype,
                                        int64_t rows, int64_t cols,
                                        void* stream) {
    hipStream_t s = (hipStream_t)stream;
    dim3 grid((unsigned)((cols + LZ_TT - 1) / LZ_TT),
              (unsigned)((rows + LZ_TT - 1) / LZ_TT));
    // interior-only fast path: every tile full and 16B-aligned
    if ((rows % LZ_TT == 0) && (cols % LZ_TT == 0)) {
#define LZ_TCV_DST(SrcT, code, DstT)                                            \
    case code:                                                                  \
        hipLaunchKernelGGL((transpose_cast_kernel_v4<SrcT, DstT>), grid,        \
                           dim3(LZ_BLOCK), 0, s, (const SrcT*)src, (DstT*)dst,  \
                           rows, cols);                                         \
        break;
#define LZ_TCV_SRC(scode, SrcT)                                                 \
    case scode:                                                                 \
        switch (dst_dtype) {                                                    \
            LZ_TCV_DST(SrcT, LZ_F32, float)                                     \
            LZ_TCV_DST(SrcT, LZ_F16, __half)                                    \
            LZ_TCV_DST(SrcT, LZ_BF16, __hip_bfloat16)                           \
            default: return hipErrorInvalidValue;                               \
        }                                                                       \
        break;
        switch (src_dtype) {
            LZ_TCV_SRC(LZ_F32, float)
            LZ_TCV_SRC(LZ_F16, __half)
            LZ_TCV_SRC(LZ_BF16, __hip_bfloat16)
            default: return hipErrorInvalidValue;
        }
#undef LZ_TCV_SRC
#undef LZ_TCV_DST
        return hipGetLastError();
    }
#define LZ_TC_DST(SrcT, code, DstT)                                             \
    case code:                                                                  \
        hipLaunchKernelGGL((transpose_cast_kernel<SrcT, DstT>), grid,           \
                           dim3(LZ_BLOCK), 0, s, (const SrcT*)src, (DstT*)dst,  \
                           rows, cols);                                         \
        break;
#define LZ_TC_SRC(scode, SrcT)                                                  \
    case scode:                                                                 \
        switch (dst_dtype) {                                                    \
            LZ_TC_DST(SrcT, LZ_F32, float)                                      \
            LZ_TC_DST(SrcT, LZ_F16, __half)                                     \
            LZ_TC_DST(SrcT, LZ_BF16, __hip_bfloat16)                            \
            default: return hipErrorInvalidValue;                               \
        }                                                                       \
        break;
    switch (src_dtype) {
        LZ_TC_SRC(LZ_F32, float)
        LZ_TC_SRC(LZ_F16, __half)
        LZ_TC_SRC(LZ_BF16, __hip_bfloat16)
        default: return hipErrorInvalidValue;
    }
#undef LZ_TC_SRC
#undef LZ_TC_DST
    return hipGetLastError();
}

// ---------------------------------------------------------------------------
// fill_pattern: test/verification helper (deterministic device-side fill).
// ---------------------------------------------------------------------------

// mask!=0: each 16-bit lane is ANDed with it after mixing — e.g. 0x3FFF
// clamps bf16 exponents so the pattern is finite positive synthetic data
// (fused here: a separate torch bitwise_and_ pass costs 2x the buffer in
// HBM traffic — measured 389 us per 1 GiB shard, profiles/
// kernel_stats_bench_fused.md).
__global__ void fill_pattern_kernel(uint64_t* __restrict__ data, int64_t nwords,
                                    uint64_t seed, uint64_t mask16) {
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const uint64_t m = mask16 ? (mask16 * 0x0001000100010001ULL) : ~0ULL;
    for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nwords;
         i += stride) {
        data[i] = splitmix64(seed ^ (uint64_t)i) & m;
    }
}

extern "C" hipError_t lz_fill_pattern(void* data, int64_t nbytes, uint64_t seed,
                                      void* stream) {
    int64_t nwords = nbytes >> 3;
    int64_t want = (nwords + LZ_BLOCK - 1) / LZ_BLOCK;
    int blocks = (int)(want < 1 ? 1 : (want > g_max_blocks ? g_max_blocks : want));
    hipLaunchKernelGGL(fill_pattern_kernel, dim3(blocks), dim3(LZ_BLOCK), 0,
                       (hipStream_t)stream, (uint64_t*)data, nwords, seed, 0ULL);
    return hipGetLastError();
}

extern "C" hipError_t lz_fill_pattern_masked(void* data, int64_t nbytes,
                                             uint64_t seed, uint64_t mask16,
                                             void* stream) {
    int64_t nwords = nbytes >> 3;
    int64_t want = (nwords + LZ_BLOCK - 1) / LZ_BLOCK;
    int blocks = (int)(want < 1 ? 1 : (want > g_max_blocks ? g_max_blocks : want));
    hipLaunchKernelGGL(fill_pattern_kernel, dim3(blocks), dim3(LZ_BLOCK), 0,
                       (hipStream_t)stream, (uint64_t*)data, nwords, seed,
                       mask16 & 0xFFFFULL);
    return hipGetLastError();
}

extern "C" const char* lz_error_name(hipError_t err) { return hipGetErrorName(err); }
