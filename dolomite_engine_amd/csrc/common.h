// Shared device helpers for the dolomite_hip kernels (gfx950 / CDNA4).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#include "../../include/dolomite_hip.h"

#define WAVE 64  // CDNA wavefront width (hard-coded per guide §1)

#define HIP_CHECK_RET(expr)                \
    do {                                   \
        hipError_t _e = (expr);            \
        if (_e != hipSuccess) return (int)_e; \
    } while (0)

static inline int dol_last_error() {
    hipError_t e = hipGetLastError();
    return (int)e;
}

// ---- dtype load/store: fp32 (code 0) or bf16 (code 1) ---------------------

__device__ __forceinline__ float bf16_to_f32(uint16_t u) {
    union { uint32_t u; float f; } c;
    c.u = ((uint32_t)u) << 16;
    return c.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16(float f) {
    // round-to-nearest-even, matching torch's float->bfloat16 cast
    union { float f; uint32_t u; } c;
    c.f = f;
    uint32_t u = c.u;
    if ((u & 0x7fffffffu) > 0x7f800000u) return (uint16_t)((u >> 16) | 0x0040u);  // NaN
    uint32_t rounding = 0x7fffu + ((u >> 16) & 1u);
    return (uint16_t)((u + rounding) >> 16);
}

template <typename T>
__device__ __forceinline__ float load_as_f32(const T* p);
template <>
__device__ __forceinline__ float load_as_f32<float>(const float* p) { return *p; }
template <>
__device__ __forceinline__ float load_as_f32<uint16_t>(const uint16_t* p) { return bf16_to_f32(*p); }

template <typename T>
__device__ __forceinline__ void store_from_f32(T* p, float v);
template <>
__device__ __forceinline__ void store_from_f32<float>(float* p, float v) { *p = v; }
template <>
__device__ __forceinline__ void store_from_f32<uint16_t>(uint16_t* p, float v) { *p = f32_to_bf16(v); }

// ---- wave / block reductions ---------------------------------------------

// 16-lane-internal steps as DPP row_ror (pure VALU; __shfl_xor lowers to
// ds_bpermute LDS instructions), cross-row steps as shuffles.
template <int N>
__device__ __forceinline__ float dpp_ror_f32(float v) {
    return __int_as_float(__builtin_amdgcn_update_dpp(
        0, __float_as_int(v), 0x120 + N, 0xF, 0xF, false));
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
    v += dpp_ror_f32<8>(v);
    v += dpp_ror_f32<4>(v);
    v += dpp_ror_f32<2>(v);
    v += dpp_ror_f32<1>(v);
    v += __shfl_xor(v, 16, 64);
    v += __shfl_xor(v, 32, 64);
    return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = 32; off >= 1; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
    return v;
}

// Block-level sum over `nthreads` (multiple of 64) using `red` LDS scratch
// (size nthreads/64 floats). Every thread returns the total.
__device__ __forceinline__ float block_reduce_sum(float v, float* red, int nthreads) {
    int lane = threadIdx.x & 63;
    int wid = threadIdx.x >> 6;
    int nw = nthreads >> 6;
    v = wave_reduce_sum(v);
    if (lane == 0) red[wid] = v;
    __syncthreads();
    float total = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i)
        if (i < nw) total += red[i];
    __syncthreads();
    return total;
}

// ---- vectorized row I/O: one wide load/store per V elements --------------
// (hipcc does not merge scalar bf16 accesses well — guide §5 common mistake 2)

typedef unsigned short us8_t __attribute__((ext_vector_type(8)));
typedef float f4_t __attribute__((ext_vector_type(4)));

template <typename T, int V>
struct VecIO;

template <>
struct VecIO<uint16_t, 8> {
    static __device__ __forceinline__ void load(const uint16_t* p, float* out) {
        us8_t v = *(const us8_t*)p;
#pragma unroll
        for (int e = 0; e < 8; ++e) out[e] = bf16_to_f32((uint16_t)v[e]);
    }
    static __device__ __forceinline__ void store(uint16_t* p, const float* in) {
        us8_t v;
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = f32_to_bf16(in[e]);
        *(us8_t*)p = v;
    }
    // single-touch streams: bypass L2 allocation (and the write-allocate
    // read-for-ownership a plain store of a full line still pays)
    static __device__ __forceinline__ void load_nt(const uint16_t* p, float* out) {
        us8_t v = __builtin_nontemporal_load((const us8_t*)p);
#pragma unroll
        for (int e = 0; e < 8; ++e) out[e] = bf16_to_f32((uint16_t)v[e]);
    }
    static __device__ __forceinline__ void store_nt(uint16_t* p, const float* in) {
        us8_t v;
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = f32_to_bf16(in[e]);
        __builtin_nontemporal_store(v, (us8_t*)p);
    }
};

template <>
struct VecIO<float, 4> {
    static __device__ __forceinline__ void load(const float* p, float* out) {
        f4_t v = *(const f4_t*)p;
#pragma unroll
        for (int e = 0; e < 4; ++e) out[e] = v[e];
    }
    static __device__ __forceinline__ void store(float* p, const float* in) {
        f4_t v;
#pragma unroll
        for (int e = 0; e < 4; ++e) v[e] = in[e];
        *(f4_t*)p = v;
    }
    static __device__ __forceinline__ void load_nt(const float* p, float* out) {
        f4_t v = __builtin_nontemporal_load((const f4_t*)p);
#pragma unroll
        for (int e = 0; e < 4; ++e) out[e] = v[e];
    }
    static __device__ __forceinline__ void store_nt(float* p, const float* in) {
        f4_t v;
#pragma unroll
        for (int e = 0; e < 4; ++e) v[e] = in[e];
        __builtin_nontemporal_store(v, (f4_t*)p);
    }
};

template <>
struct VecIO<uint16_t, 2> {
    static __device__ __forceinline__ void load(const uint16_t* p, float* out) {
        uint32_t v = *(const uint32_t*)p;
        out[0] = bf16_to_f32((uint16_t)(v & 0xffff));
        out[1] = bf16_to_f32((uint16_t)(v >> 16));
    }
    static __device__ __forceinline__ void store(uint16_t* p, const float* in) {
        uint32_t v = (uint32_t)f32_to_bf16(in[0]) | ((uint32_t)f32_to_bf16(in[1]) << 16);
        *(uint32_t*)p = v;
    }
    static __device__ __forceinline__ void load_nt(const uint16_t* p, float* out) { load(p, out); }
    static __device__ __forceinline__ void store_nt(uint16_t* p, const float* in) { store(p, in); }
};

template <>
struct VecIO<float, 2> {
    static __device__ __forceinline__ void load(const float* p, float* out) {
        out[0] = p[0];
        out[1] = p[1];
    }
    static __device__ __forceinline__ void store(float* p, const float* in) {
        p[0] = in[0];
        p[1] = in[1];
    }
    static __device__ __forceinline__ void load_nt(const float* p, float* out) { load(p, out); }
    static __device__ __forceinline__ void store_nt(float* p, const float* in) { store(p, in); }
};

template <>
struct VecIO<uint16_t, 1> {
    static __device__ __forceinline__ void load(const uint16_t* p, float* out) { out[0] = bf16_to_f32(*p); }
    static __device__ __forceinline__ void store(uint16_t* p, const float* in) { *p = f32_to_bf16(in[0]); }
    static __device__ __forceinline__ void load_nt(const uint16_t* p, float* out) { load(p, out); }
    static __device__ __forceinline__ void store_nt(uint16_t* p, const float* in) { store(p, in); }
};

template <>
struct VecIO<float, 1> {
    static __device__ __forceinline__ void load(const float* p, float* out) { out[0] = *p; }
    static __device__ __forceinline__ void store(float* p, const float* in) { *p = in[0]; }
    static __device__ __forceinline__ void load_nt(const float* p, float* out) { load(p, out); }
    static __device__ __forceinline__ void store_nt(float* p, const float* in) { store(p, in); }
};

template <>
struct VecIO<float, 8> {
    static __device__ __forceinline__ void load(const float* p, float* out) {
        VecIO<float, 4>::load(p, out);
        VecIO<float, 4>::load(p + 4, out + 4);
    }
    static __device__ __forceinline__ void store(float* p, const float* in) {
        VecIO<float, 4>::store(p, in);
        VecIO<float, 4>::store(p + 4, in + 4);
    }
    static __device__ __forceinline__ void load_nt(const float* p, float* out) { load(p, out); }
    static __device__ __forceinline__ void store_nt(float* p, const float* in) { store(p, in); }
};
