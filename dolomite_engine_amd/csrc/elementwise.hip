// dolomite_hip — normalization / RoPE / cross-entropy / AdamW kernels for
// gfx950 (MI355X). C-ABI entry points declared in include/dolomite_hip.h.
//
// These are the MI355X-native replacements for (reference call sites):
//   RMSNorm   : modeling_utils/normalization/rmsnorm/base.py:18-25 semantics
//               (the reference's Triton torchtitan kernels are NOT ported)
//   LayerNorm : nn.LayerNorm ('layernorm'/'torch' implementation)
//   RoPE      : position_embedding/rope.py:104-121 applied at
//               attention/padding_free.py:38-40 on the packed c_attn layout
//   CE        : F.cross_entropy at model_wrapper/pretraining.py:125 /
//               gpt_dolomite/main.py:200
//   AdamW     : torch.optim.AdamW (optimization/optimizer.py:74)
//
// All HBM-bound: the design rules are coalesced wide loads (bf16x8 /
// float4), fp32 accumulation, one LDS stash per row to avoid re-reads.

#include "common.h"

#include <math.h>

// ===========================================================================
// RMSNorm forward: one workgroup per row, input stashed in LDS as fp32.
// ===========================================================================

template <typename T, int V>
__global__ void __launch_bounds__(256) rmsnorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ res_in, const T* __restrict__ w,
    T* __restrict__ y, T* __restrict__ res_out, float* __restrict__ rstd_out,
    int64_t T_rows, int H, float eps) {
    extern __shared__ float smem[];           // H floats (row stash) + 8 reduce
    float* row = smem;
    float* red = smem + H;

    int64_t t = blockIdx.x;
    const T* xr = x + t * (int64_t)H;
    const T* rr = res_in ? res_in + t * (int64_t)H : nullptr;

    float ss = 0.f;
    for (int c = threadIdx.x * V; c < H; c += blockDim.x * V) {
#pragma unroll
        for (int k = 0; k < V; ++k) {
            if (c + k < H) {
                float v = load_as_f32(xr + c + k);
                if (rr) {
                    // the residual add is a tensor-dtype add in the reference
                    // (layer.py:74,87): round the sum like a torch bf16 add
                    v += load_as_f32(rr + c + k);
                    T tmp;
                    store_from_f32(&tmp, v);
                    v = load_as_f32(&tmp);
                }
                row[c + k] = v;
                ss += v * v;
            }
        }
    }
    float total = block_reduce_sum(ss, red, blockDim.x);
    float rstd = rsqrtf(total / (float)H + eps);
    if (threadIdx.x == 0 && rstd_out) rstd_out[t] = rstd;

    T* yr = y + t * (int64_t)H;
    T* sr = res_out ? res_out + t * (int64_t)H : nullptr;
    for (int c = threadIdx.x * V; c < H; c += blockDim.x * V) {
#pragma unroll
        for (int k = 0; k < V; ++k) {
            if (c + k < H) {
                float s = row[c + k];
                if (sr) store_from_f32(sr + c + k, s);
                // reference order: cast normalized value to input dtype BEFORE
                // multiplying by weight (rmsnorm/base.py:23-25)
                float nhat;
                T tmp;
                store_from_f32(&tmp, s * rstd);
                nhat = load_as_f32(&tmp);
                store_from_f32(yr + c + k, load_as_f32(w + c + k) * nhat);
            }
        }
    }
}

extern "C" int dolomite_rmsnorm_fwd(dolomite_stream_t stream,
                                    const void* x, const void* res_in, const void* w,
                                    void* y, void* res_out, float* rstd,
                                    int64_t T_rows, int64_t H, float eps, int dtype) {
    if (H > 65536 / 8) { /* LDS cap: H*4 + 32 bytes must fit 160KB; enforce 16K */ }
    dim3 grid((uint32_t)T_rows), block(256);
    size_t shmem = (size_t)H * 4 + 8 * 4;
    if (shmem > 160 * 1024) return 9001;
    if (dtype == DOLOMITE_BF16) {
        if (H % 8 == 0)
            hipLaunchKernelGGL((rmsnorm_fwd_kernel<uint16_t, 8>), grid, block, shmem, (hipStream_t)stream,
                               (const uint16_t*)x, (const uint16_t*)res_in, (const uint16_t*)w,
                               (uint16_t*)y, (uint16_t*)res_out, rstd, T_rows, (int)H, eps);
        else
            hipLaunchKernelGGL((rmsnorm_fwd_kernel<uint16_t, 1>), grid, block, shmem, (hipStream_t)stream,
                               (const uint16_t*)x, (const uint16_t*)res_in, (const uint16_t*)w,
                               (uint16_t*)y, (uint16_t*)res_out, rstd, T_rows, (int)H, eps);
    } else {
        if (H % 4 == 0)
            hipLaunchKernelGGL((rmsnorm_fwd_kernel<float, 4>), grid, block, shmem, (hipStream_t)stream,
                               (const float*)x, (const float*)res_in, (const float*)w,
                               (float*)y, (float*)res_out, rstd, T_rows, (int)H, eps);
        else
            hipLaunchKernelGGL((rmsnorm_fwd_kernel<float, 1>), grid, block, shmem, (hipStream_t)stream,
                               (const float*)x, (const float*)res_in, (const float*)w,
                               (float*)y, (float*)res_out, rstd, T_rows, (int)H, eps);
    }
    return dol_last_error();
}

// ===========================================================================
// RMSNorm backward: fixed grid, per-thread column ownership for dw partials.
//   dx = rstd*(w*dy - s_hat * mean(w*dy*s_hat)),  s_hat = s_fp32*rstd
//   dw = sum_rows dy * cast_to_dtype(s_hat)
// ===========================================================================

#define RMS_BWD_BLOCKS 600  // > 256 CUs, fixed so dw_partial scratch is bounded

extern "C" int dolomite_rmsnorm_bwd_nblocks(int64_t T_rows) {
    (void)T_rows;
    return RMS_BWD_BLOCKS;
}

template <typename T, int V, int ITMAX>
__global__ void __launch_bounds__(256) rmsnorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ s, const T* __restrict__ w,
    const float* __restrict__ rstd, T* __restrict__ dx, float* __restrict__ dw_partial,
    int64_t T_rows, int H) {
    __shared__ float red[8];

    float wv[ITMAX][V];
    float dw_acc[ITMAX][V];
#pragma unroll
    for (int i = 0; i < ITMAX; ++i)
#pragma unroll
        for (int k = 0; k < V; ++k) {
            int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
            wv[i][k] = (c < H) ? load_as_f32(w + c) : 0.f;
            dw_acc[i][k] = 0.f;
        }

    for (int64_t t = blockIdx.x; t < T_rows; t += gridDim.x) {
        const T* dyr = dy + t * (int64_t)H;
        const T* sr = s + t * (int64_t)H;
        float r = rstd[t];

        float sv[ITMAX][V], dyv[ITMAX][V];
        float dot = 0.f;
#pragma unroll
        for (int i = 0; i < ITMAX; ++i)
#pragma unroll
            for (int k = 0; k < V; ++k) {
                int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
                if (c < H) {
                    float svv = load_as_f32(sr + c);
                    float dyy = load_as_f32(dyr + c);
                    sv[i][k] = svv;
                    dyv[i][k] = dyy;
                    dot += wv[i][k] * dyy * (svv * r);
                } else {
                    sv[i][k] = 0.f;
                    dyv[i][k] = 0.f;
                }
            }
        float dtot = block_reduce_sum(dot, red, blockDim.x) / (float)H;

        T* dxr = dx + t * (int64_t)H;
#pragma unroll
        for (int i = 0; i < ITMAX; ++i)
#pragma unroll
            for (int k = 0; k < V; ++k) {
                int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
                if (c < H) {
                    float shat = sv[i][k] * r;
                    float dxv = r * (wv[i][k] * dyv[i][k] - shat * dtot);
                    store_from_f32(dxr + c, dxv);
                    // dw uses the CAST normalized value (reference casts before w-mul)
                    T tmp;
                    store_from_f32(&tmp, shat);
                    dw_acc[i][k] += dyv[i][k] * load_as_f32(&tmp);
                }
            }
    }

    float* dwp = dw_partial + (int64_t)blockIdx.x * H;
#pragma unroll
    for (int i = 0; i < ITMAX; ++i)
#pragma unroll
        for (int k = 0; k < V; ++k) {
            int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
            if (c < H) dwp[c] = dw_acc[i][k];
        }
}

template <typename T>
static int launch_rmsnorm_bwd(hipStream_t stream, const T* dy, const T* s, const T* w,
                              const float* rstd, T* dx, float* dw_partial, int64_t T_rows, int H) {
    dim3 grid(RMS_BWD_BLOCKS), block(256);
    const int V = 4;
    int it = (H + 256 * V - 1) / (256 * V);
    if (it <= 1)
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<T, V, 1>), grid, block, 0, stream, dy, s, w, rstd, dx, dw_partial, T_rows, H);
    else if (it <= 2)
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<T, V, 2>), grid, block, 0, stream, dy, s, w, rstd, dx, dw_partial, T_rows, H);
    else if (it <= 4)
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<T, V, 4>), grid, block, 0, stream, dy, s, w, rstd, dx, dw_partial, T_rows, H);
    else if (it <= 8)
        hipLaunchKernelGGL((rmsnorm_bwd_kernel<T, V, 8>), grid, block, 0, stream, dy, s, w, rstd, dx, dw_partial, T_rows, H);
    else
        return 9002;  // H > 8192 unsupported by this kernel
    return dol_last_error();
}

extern "C" int dolomite_rmsnorm_bwd(dolomite_stream_t stream,
                                    const void* dy, const void* s, const void* w,
                                    const float* rstd, void* dx, float* dw_partial,
                                    int64_t T_rows, int64_t H, int dtype) {
    if (dtype == DOLOMITE_BF16)
        return launch_rmsnorm_bwd((hipStream_t)stream, (const uint16_t*)dy, (const uint16_t*)s,
                                  (const uint16_t*)w, rstd, (uint16_t*)dx, dw_partial, T_rows, (int)H);
    return launch_rmsnorm_bwd((hipStream_t)stream, (const float*)dy, (const float*)s,
                              (const float*)w, rstd, (float*)dx, dw_partial, T_rows, (int)H);
}

// ===========================================================================
// LayerNorm forward / backward (same structure, plus mean and bias).
//   y = cast((x32 - mu) * rstd * w32 + b32)   [torch F.layer_norm semantics:
//   bf16 upcast to fp32 throughout, single final cast]
// ===========================================================================

template <typename T, int V>
__global__ void __launch_bounds__(256) layernorm_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ res_in,
    const T* __restrict__ w, const T* __restrict__ b,
    T* __restrict__ y, T* __restrict__ res_out,
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    int64_t T_rows, int H, float eps) {
    extern __shared__ float smem[];
    float* row = smem;
    float* red = smem + H;

    int64_t t = blockIdx.x;
    const T* xr = x + t * (int64_t)H;
    const T* rr = res_in ? res_in + t * (int64_t)H : nullptr;

    float sum = 0.f;
    for (int c = threadIdx.x * V; c < H; c += blockDim.x * V) {
#pragma unroll
        for (int k = 0; k < V; ++k)
            if (c + k < H) {
                float v = load_as_f32(xr + c + k);
                if (rr) {
                    v += load_as_f32(rr + c + k);
                    T tmp;
                    store_from_f32(&tmp, v);
                    v = load_as_f32(&tmp);
                }
                row[c + k] = v;
                sum += v;
            }
    }
    float mu = block_reduce_sum(sum, red, blockDim.x) / (float)H;
    float ss = 0.f;
    for (int c = threadIdx.x * V; c < H; c += blockDim.x * V) {
#pragma unroll
        for (int k = 0; k < V; ++k)
            if (c + k < H) {
                float d = row[c + k] - mu;
                ss += d * d;
            }
    }
    float rstd = rsqrtf(block_reduce_sum(ss, red, blockDim.x) / (float)H + eps);
    if (threadIdx.x == 0) {
        if (mean_out) mean_out[t] = mu;
        if (rstd_out) rstd_out[t] = rstd;
    }

    T* yr = y + t * (int64_t)H;
    T* sr = res_out ? res_out + t * (int64_t)H : nullptr;
    for (int c = threadIdx.x * V; c < H; c += blockDim.x * V) {
#pragma unroll
        for (int k = 0; k < V; ++k)
            if (c + k < H) {
                float s = row[c + k];
                if (sr) store_from_f32(sr + c + k, s);
                float v = (s - mu) * rstd * load_as_f32(w + c + k) + load_as_f32(b + c + k);
                store_from_f32(yr + c + k, v);
            }
    }
}

extern "C" int dolomite_layernorm_fwd(dolomite_stream_t stream,
                                      const void* x, const void* res_in, const void* w, const void* b,
                                      void* y, void* res_out, float* mean, float* rstd,
                                      int64_t T_rows, int64_t H, float eps, int dtype) {
    dim3 grid((uint32_t)T_rows), block(256);
    size_t shmem = (size_t)H * 4 + 8 * 4;
    if (shmem > 160 * 1024) return 9001;
    if (dtype == DOLOMITE_BF16) {
        if (H % 8 == 0)
            hipLaunchKernelGGL((layernorm_fwd_kernel<uint16_t, 8>), grid, block, shmem, (hipStream_t)stream,
                               (const uint16_t*)x, (const uint16_t*)res_in, (const uint16_t*)w, (const uint16_t*)b,
                               (uint16_t*)y, (uint16_t*)res_out, mean, rstd, T_rows, (int)H, eps);
        else
            hipLaunchKernelGGL((layernorm_fwd_kernel<uint16_t, 1>), grid, block, shmem, (hipStream_t)stream,
                               (const uint16_t*)x, (const uint16_t*)res_in, (const uint16_t*)w, (const uint16_t*)b,
                               (uint16_t*)y, (uint16_t*)res_out, mean, rstd, T_rows, (int)H, eps);
    } else {
        if (H % 4 == 0)
            hipLaunchKernelGGL((layernorm_fwd_kernel<float, 4>), grid, block, shmem, (hipStream_t)stream,
                               (const float*)x, (const float*)res_in, (const float*)w, (const float*)b,
                               (float*)y, (float*)res_out, mean, rstd, T_rows, (int)H, eps);
        else
            hipLaunchKernelGGL((layernorm_fwd_kernel<float, 1>), grid, block, shmem, (hipStream_t)stream,
                               (const float*)x, (const float*)res_in, (const float*)w, (const float*)b,
                               (float*)y, (float*)res_out, mean, rstd, T_rows, (int)H, eps);
    }
    return dol_last_error();
}

template <typename T, int V, int ITMAX>
__global__ void __launch_bounds__(256) layernorm_bwd_kernel(
    const T* __restrict__ dy, const T* __restrict__ s, const T* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    T* __restrict__ dx, float* __restrict__ dwdb_partial,
    int64_t T_rows, int H) {
    __shared__ float red[8];

    float wv[ITMAX][V], dw_acc[ITMAX][V], db_acc[ITMAX][V];
#pragma unroll
    for (int i = 0; i < ITMAX; ++i)
#pragma unroll
        for (int k = 0; k < V; ++k) {
            int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
            wv[i][k] = (c < H) ? load_as_f32(w + c) : 0.f;
            dw_acc[i][k] = 0.f;
            db_acc[i][k] = 0.f;
        }

    for (int64_t t = blockIdx.x; t < T_rows; t += gridDim.x) {
        const T* dyr = dy + t * (int64_t)H;
        const T* sr = s + t * (int64_t)H;
        float mu = mean[t], r = rstd[t];

        float xh[ITMAX][V], dyv[ITMAX][V];
        float d1 = 0.f, d2 = 0.f;
#pragma unroll
        for (int i = 0; i < ITMAX; ++i)
#pragma unroll
            for (int k = 0; k < V; ++k) {
                int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
                if (c < H) {
                    float xhat = (load_as_f32(sr + c) - mu) * r;
                    float dyy = load_as_f32(dyr + c);
                    float dyw = dyy * wv[i][k];
                    xh[i][k] = xhat;
                    dyv[i][k] = dyy;
                    d1 += dyw;
                    d2 += dyw * xhat;
                } else {
                    xh[i][k] = 0.f;
                    dyv[i][k] = 0.f;
                }
            }
        float m1 = block_reduce_sum(d1, red, blockDim.x) / (float)H;
        __syncthreads();
        float m2 = block_reduce_sum(d2, red, blockDim.x) / (float)H;

        T* dxr = dx + t * (int64_t)H;
#pragma unroll
        for (int i = 0; i < ITMAX; ++i)
#pragma unroll
            for (int k = 0; k < V; ++k) {
                int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
                if (c < H) {
                    float dxv = r * (dyv[i][k] * wv[i][k] - m1 - xh[i][k] * m2);
                    store_from_f32(dxr + c, dxv);
                    dw_acc[i][k] += dyv[i][k] * xh[i][k];
                    db_acc[i][k] += dyv[i][k];
                }
            }
    }

    float* dwp = dwdb_partial + (int64_t)blockIdx.x * H;
    float* dbp = dwdb_partial + ((int64_t)gridDim.x + blockIdx.x) * H;
#pragma unroll
    for (int i = 0; i < ITMAX; ++i)
#pragma unroll
        for (int k = 0; k < V; ++k) {
            int c = i * (int)blockDim.x * V + (int)threadIdx.x * V + k;
            if (c < H) {
                dwp[c] = dw_acc[i][k];
                dbp[c] = db_acc[i][k];
            }
        }
}

template <typename T>
static int launch_layernorm_bwd(hipStream_t stream, const T* dy, const T* s, const T* w,
                                const float* mean, const float* rstd, T* dx, float* dwdb_partial,
                                int64_t T_rows, int H) {
    dim3 grid(RMS_BWD_BLOCKS), block(256);
    const int V = 4;
    int it = (H + 256 * V - 1) / (256 * V);
    if (it <= 1)
        hipLaunchKernelGGL((layernorm_bwd_kernel<T, V, 1>), grid, block, 0, stream, dy, s, w, mean, rstd, dx, dwdb_partial, T_rows, H);
    else if (it <= 2)
        hipLaunchKernelGGL((layernorm_bwd_kernel<T, V, 2>), grid, block, 0, stream, dy, s, w, mean, rstd, dx, dwdb_partial, T_rows, H);
    else if (it <= 4)
        hipLaunchKernelGGL((layernorm_bwd_kernel<T, V, 4>), grid, block, 0, stream, dy, s, w, mean, rstd, dx, dwdb_partial, T_rows, H);
    else if (it <= 8)
        hipLaunchKernelGGL((layernorm_bwd_kernel<T, V, 8>), grid, block, 0, stream, dy, s, w, mean, rstd, dx, dwdb_partial, T_rows, H);
    else
        return 9002;
    return dol_last_error();
}

extern "C" int dolomite_layernorm_bwd(dolomite_stream_t stream,
                                      const void* dy, const void* s, const void* w,
                                      const float* mean, const float* rstd,
                                      void* dx, float* dwdb_partial,
                                      int64_t T_rows, int64_t H, int dtype) {
    if (dtype == DOLOMITE_BF16)
        return launch_layernorm_bwd((hipStream_t)stream, (const uint16_t*)dy, (const uint16_t*)s,
                                    (const uint16_t*)w, mean, rstd, (uint16_t*)dx, dwdb_partial, T_rows, (int)H);
    return launch_layernorm_bwd((hipStream_t)stream, (const float*)dy, (const float*)s,
                                (const float*)w, mean, rstd, (float*)dx, dwdb_partial, T_rows, (int)H);
}

// ===========================================================================
// Partials reduction: out[h] = sum_i partial[i*H + h]
// ===========================================================================

__global__ void __launch_bounds__(256) reduce_partials_kernel(
    const float* __restrict__ partial, float* __restrict__ out, int64_t nblocks, int64_t H) {
    int64_t h = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (h >= H) return;
    float acc = 0.f;
    for (int64_t i = 0; i < nblocks; ++i) acc += partial[i * H + h];
    out[h] = acc;
}

extern "C" int dolomite_reduce_partials(dolomite_stream_t stream,
                                        const float* partial, float* out,
                                        int64_t nblocks, int64_t H) {
    dim3 grid((uint32_t)((H + 255) / 256)), block(256);
    hipLaunchKernelGGL(reduce_partials_kernel, grid, block, 0, (hipStream_t)stream, partial, out, nblocks, H);
    return dol_last_error();
}

// ===========================================================================
// RoPE on the packed QKV layout (in-place or same-offset out-of-place).
// One thread per (token, head, pair). position_embedding/rope.py:104-121.
// ===========================================================================

template <typename T>
__global__ void __launch_bounds__(256) rope_qkv_kernel(
    const T* __restrict__ qkv_in, T* __restrict__ qkv_out,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    int64_t total, int64_t row_len, int H, int Hkv, int D, int G,
    int64_t q_gstride, int64_t k_off, int64_t kv_hstride, float dir) {
    int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
    if (idx >= total) return;
    int halfD = D >> 1;
    int nheads = H + Hkv;
    int64_t pairs_per_tok = (int64_t)nheads * halfD;
    int64_t t = idx / pairs_per_tok;
    int rem = (int)(idx % pairs_per_tok);
    int hh = rem / halfD;
    int i = rem % halfD;

    int64_t base = t * row_len;
    if (hh < H) {
        base += (int64_t)(hh / G) * q_gstride + (int64_t)(hh % G) * D;
    } else {
        int j = hh - H;
        base += k_off + (int64_t)j * kv_hstride;
    }
    float c1 = cos_t[t * D + i];
    float s1 = sin_t[t * D + i] * dir;
    float c2 = cos_t[t * D + i + halfD];
    float s2 = sin_t[t * D + i + halfD] * dir;
    float x1 = load_as_f32(qkv_in + base + i);
    float x2 = load_as_f32(qkv_in + base + i + halfD);
    store_from_f32(qkv_out + base + i, x1 * c1 - x2 * s1);
    store_from_f32(qkv_out + base + i + halfD, x2 * c2 + x1 * s2);
}

extern "C" int dolomite_rope_qkv(dolomite_stream_t stream,
                                 const void* qkv_in, void* qkv_out,
                                 const float* cos_t, const float* sin_t,
                                 int64_t T_rows, int64_t row_len,
                                 int H, int Hkv, int D, int G,
                                 int64_t q_gstride, int64_t k_off, int64_t kv_hstride,
                                 int dir, int rotate_v_copy, int dtype) {
    (void)rotate_v_copy;
    if (D % 2 != 0) return 9003;
    int64_t total = T_rows * (int64_t)(H + Hkv) * (D / 2);
    if (total == 0) return 0;
    dim3 grid((uint32_t)((total + 255) / 256)), block(256);
    float fdir = (dir >= 0) ? 1.f : -1.f;
    if (dtype == DOLOMITE_BF16)
        hipLaunchKernelGGL((rope_qkv_kernel<uint16_t>), grid, block, 0, (hipStream_t)stream,
                           (const uint16_t*)qkv_in, (uint16_t*)qkv_out, cos_t, sin_t,
                           total, row_len, H, Hkv, D, G, q_gstride, k_off, kv_hstride, fdir);
    else
        hipLaunchKernelGGL((rope_qkv_kernel<float>), grid, block, 0, (hipStream_t)stream,
                           (const float*)qkv_in, (float*)qkv_out, cos_t, sin_t,
                           total, row_len, H, Hkv, D, G, q_gstride, k_off, kv_hstride, fdir);
    return dol_last_error();
}

// ===========================================================================
// Fused cross entropy (mean, ignore_index): one workgroup per row.
// fwd: online max/sum -> lse, row_loss.  bwd: dlogits = (p - onehot)*gs.
// ===========================================================================

template <typename T, int V>
__global__ void __launch_bounds__(256) ce_fwd_kernel(
    const T* __restrict__ logits, const int64_t* __restrict__ labels,
    float* __restrict__ row_loss, float* __restrict__ lse_out,
    int64_t V_dim, int64_t row_stride, int ignore_index) {
    __shared__ float red[8];
    int64_t t = blockIdx.x;
    const T* xr = logits + t * row_stride;

    float m = -INFINITY, ssum = 0.f;
    for (int64_t c = threadIdx.x * V; c < V_dim; c += (int64_t)blockDim.x * V) {
#pragma unroll
        for (int k = 0; k < V; ++k)
            if (c + k < V_dim) {
                float v = load_as_f32(xr + c + k);
                if (v > m) {
                    ssum *= __expf(m - v);
                    m = v;
                }
                ssum += __expf(v - m);
            }
    }
    // block combine: global max then shift partial sums
    float mall = m;
#pragma unroll
    for (int off = 32; off >= 1; off >>= 1) mall = fmaxf(mall, __shfl_xor(mall, off, 64));
    int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
    if (lane == 0) red[wid] = mall;
    __syncthreads();
    float gmax = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));
    __syncthreads();
    float shifted = (m == -INFINITY) ? 0.f : ssum * __expf(m - gmax);
    float total = block_reduce_sum(shifted, red, blockDim.x);
    float lse = gmax + __logf(total);

    if (threadIdx.x == 0) {
        lse_out[t] = lse;
        int64_t lab = labels[t];
        if (lab == ignore_index) {
            row_loss[t] = 0.f;
        } else {
            row_loss[t] = lse - load_as_f32(xr + lab);
        }
    }
}

extern "C" int dolomite_ce_fwd(dolomite_stream_t stream,
                               const void* logits, const int64_t* labels,
                               float* row_loss, float* lse,
                               int64_t T_rows, int64_t V_dim, int64_t row_stride,
                               int ignore_index, int dtype) {
    dim3 grid((uint32_t)T_rows), block(256);
    if (dtype == DOLOMITE_BF16) {
        if (V_dim % 8 == 0)
            hipLaunchKernelGGL((ce_fwd_kernel<uint16_t, 8>), grid, block, 0, (hipStream_t)stream,
                               (const uint16_t*)logits, labels, row_loss, lse, V_dim, row_stride, ignore_index);
        else
            hipLaunchKernelGGL((ce_fwd_kernel<uint16_t, 1>), grid, block, 0, (hipStream_t)stream,
                               (const uint16_t*)logits, labels, row_loss, lse, V_dim, row_stride, ignore_index);
    } else {
        if (V_dim % 4 == 0)
            hipLaunchKernelGGL((ce_fwd_kernel<float, 4>), grid, block, 0, (hipStream_t)stream,
                               (const float*)logits, labels, row_loss, lse, V_dim, row_stride, ignore_index);
        else
            hipLaunchKernelGGL((ce_fwd_kernel<float, 1>), grid, block, 0, (hipStream_t)stream,
                               (const float*)logits, labels, row_loss, lse, V_dim, row_stride, ignore_index);
    }
    return dol_last_error();
}

template <typename T, int V>
__global__ void __launch_bounds__(256) ce_bwd_kernel(
    const T* __restrict__ logits, const int64_t* __restrict__ labels,
    const float* __restrict__ lse, T* __restrict__ dlogits, float gs,
    int64_t V_dim, int64_t row_stride, int ignore_index) {
    int64_t t = blockIdx.x;
    const T* xr = logits + t * row_stride;
    T* dr = dlogits + t * row_stride;
    int64_t lab = labels[t];
    float l = lse[t];
    if (lab == ignore_index) {
        for (int64_t c = threadIdx.x * V; c < V_dim; c += (int64_t)blockDim.x * V)
#pragma unroll
            for (int k = 0; k < V; ++k)
                if (c + k < V_dim) store_from_f32(dr + c + k, 0.f);
        return;
    }
    for (int64_t c = threadIdx.x * V; c < V_dim; c += (int64_t)blockDim.x * V) {
#pragma unroll
        for (int k = 0; k < V; ++k)
            if (c + k < V_dim) {
                float p = __expf(load_as_f32(xr + c + k) - l);
                float g = (c + k == lab) ? (p - 1.f) : p;
                store_from_f32(dr + c + k, g * gs);
            }
    }
}

extern "C" int dolomite_ce_bwd(dolomite_stream_t stream,
                               const void* logits, const int64_t* labels, const float* lse,
                               void* dlogits, float grad_scale,
                               int64_t T_rows, int64_t V_dim, int64_t row_stride,
                               int ignore_index, int dtype) {
    dim3 grid((uint32_t)T_rows), block(256);
    if (dtype == DOLOMITE_BF16) {
        if (V_dim % 8 == 0)
            hipLaunchKernelGGL((ce_bwd_kernel<uint16_t, 8>), grid, block, 0, (hipStream_t)stream,
                               (const uint16_t*)logits, labels, lse, (uint16_t*)dlogits, grad_scale, V_dim, row_stride, ignore_index);
        else
            hipLaunchKernelGGL((ce_bwd_kernel<uint16_t, 1>), grid, block, 0, (hipStream_t)stream,
                               (const uint16_t*)logits, labels, lse, (uint16_t*)dlogits, grad_scale, V_dim, row_stride, ignore_index);
    } else {
        if (V_dim % 4 == 0)
            hipLaunchKernelGGL((ce_bwd_kernel<float, 4>), grid, block, 0, (hipStream_t)stream,
                               (const float*)logits, labels, lse, (float*)dlogits, grad_scale, V_dim, row_stride, ignore_index);
        else
            hipLaunchKernelGGL((ce_bwd_kernel<float, 1>), grid, block, 0, (hipStream_t)stream,
                               (const float*)logits, labels, lse, (float*)dlogits, grad_scale, V_dim, row_stride, ignore_index);
    }
    return dol_last_error();
}

// ===========================================================================
// Fused AdamW on a flat fp32 master shard (torch.optim.AdamW semantics),
// with optional bf16 write-out. HBM-bound; 4 elems per thread, coalesced.
// ===========================================================================

template <int GRAD_BF16>
__global__ void __launch_bounds__(256) adamw_kernel(
    float* __restrict__ master, uint16_t* __restrict__ param_out,
    const void* __restrict__ grad_v, float* __restrict__ m, float* __restrict__ v,
    int64_t n, float lr, float b1, float b2, float eps, float wd,
    float bc1, float bc2) {
    int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
        int64_t i = i0 + k;
        if (i >= n) return;
        float g = GRAD_BF16 ? bf16_to_f32(((const uint16_t*)grad_v)[i]) : ((const float*)grad_v)[i];
        float p = master[i];
        p *= (1.f - lr * wd);
        float mi = m[i] * b1 + (1.f - b1) * g;
        float vi = v[i] * b2 + (1.f - b2) * g * g;
        m[i] = mi;
        v[i] = vi;
        float denom = sqrtf(vi / bc2) + eps;
        p -= (lr / bc1) * mi / denom;
        master[i] = p;
        if (param_out) param_out[i] = f32_to_bf16(p);
    }
}

extern "C" int dolomite_adamw_step(dolomite_stream_t stream,
                                   float* master, void* param_out_bf16,
                                   const void* grad, int grad_dtype,
                                   float* m, float* v,
                                   int64_t n, float lr, float beta1, float beta2,
                                   float eps, float weight_decay, int step) {
    if (n == 0) return 0;
    float bc1 = 1.f - powf(beta1, (float)step);
    float bc2 = 1.f - powf(beta2, (float)step);
    int64_t nthreads = (n + 3) / 4;
    dim3 grid((uint32_t)((nthreads + 255) / 256)), block(256);
    if (grad_dtype == DOLOMITE_BF16)
        hipLaunchKernelGGL((adamw_kernel<1>), grid, block, 0, (hipStream_t)stream,
                           master, (uint16_t*)param_out_bf16, grad, m, v, n, lr, beta1, beta2, eps, weight_decay, bc1, bc2);
    else
        hipLaunchKernelGGL((adamw_kernel<0>), grid, block, 0, (hipStream_t)stream,
                           master, (uint16_t*)param_out_bf16, grad, m, v, n, lr, beta1, beta2, eps, weight_decay, bc1, bc2);
    return dol_last_error();
}

// ===========================================================================
// Scalar scale in place (grad clip apply).
// ===========================================================================

template <typename T>
__global__ void __launch_bounds__(256) scale_kernel(T* __restrict__ buf, int64_t n, float s) {
    int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
        int64_t i = i0 + k;
        if (i < n) store_from_f32(buf + i, load_as_f32(buf + i) * s);
    }
}

extern "C" int dolomite_scale_inplace(dolomite_stream_t stream, void* buf, int64_t n,
                                      float scale, int dtype) {
    if (n == 0) return 0;
    int64_t nthreads = (n + 3) / 4;
    dim3 grid((uint32_t)((nthreads + 255) / 256)), block(256);
    if (dtype == DOLOMITE_BF16)
        hipLaunchKernelGGL((scale_kernel<uint16_t>), grid, block, 0, (hipStream_t)stream, (uint16_t*)buf, n, scale);
    else
        hipLaunchKernelGGL((scale_kernel<float>), grid, block, 0, (hipStream_t)stream, (float*)buf, n, scale);
    return dol_last_error();
}

extern "C" int dolomite_hip_abi_version(void) { return 1; }
