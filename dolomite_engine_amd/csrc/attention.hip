// dolomite_hip — varlen causal flash attention for gfx950 (MI355X).
//
// Replaces flash_attn_varlen_func at the reference call site
// attention/padding_free.py:51-62 (and flash.py:112-123): causal attention
// over packed (total_q, heads, head_dim) with cu_seqlens, MHA/GQA/MQA head
// layouts addressed in place on the fused c_attn projection output
// (attention/base.py:72-81) via (t_stride, group_stride) addressing.
//
// Design (CDNA4-first):
//   - MFMA v_mfma_f32_16x16x32_bf16 for QK^T and PV; 64-wide waves.
//   - forward: workgroup = 8 waves = 512 threads; q-tile 128 rows (16 per
//     wave), k-tile 64 keys streamed through LDS as row-major [key][d]
//     images (stride DPAD+16, conflict-free b128 B-frag reads); the PV
//     B-fragment comes from the PI23-rowed V image via ds_read_b64_tr_b16.
//   - online softmax in fp32 VGPRs, m/l carried per row, row reductions as
//     DPP row_ror rotations (the C-fragment's 16 columns of one row).
//   - P routed through a per-wave LDS tile to re-shape C-layout -> A-layout;
//     the PV B-fragment comes from the PI23 V image via depth-3 counted
//     tr16 ladders.
//   - LSE (H,T) fp32 written for backward; backward recomputes P. dkv
//     writes per-q-head (T, H, D) fp32 partials (exclusive stores, no
//     atomics) reduced in fixed order by the finalize kernel —
//     deterministic gradients; dq accumulates dQ in registers and stores
//     bf16 straight into the packed dqkv.
//   - dropout unsupported (hot-path configs run attn_pdrop = 0).
//
// Fragment layout assumption for mfma_f32_16x16x32_bf16 (verified on
// hardware by dolomite_mfma_probe, tests/test_gpu_kernels.py):
//   A[i][k]: lane l holds i = l&15, k = (l>>4)*8 + e   (e = 0..7)
//   B[k][j]: lane l holds k = (l>>4)*8 + e, j = l&15
//   C[r][c]: lane l, reg x holds r = (l>>4)*4 + x, c = l&15

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define MFMA16(a, b, c) __builtin_amdgcn_mfma_f32_16x16x32_bf16((a), (b), (c), 0, 0, 0)

// softmax runs in the exp2 domain: v_exp_f32 natively computes 2^x, so
// folding log2(e) into the softmax scale removes one multiply per element
#define DOL_LOG2E 1.44269504088896340736f
#define DOL_LN2 0.69314718055994530942f

// ---------------------------------------------------------------------------
// ds_read_b64_tr_b16 — gfx950 hardware transpose read (semantics pinned on
// HW by dolomite_tr16_probe): each lane reads 64b at its own 8B-aligned LDS
// address; within a 16-lane group, output lane l slot j receives element
// (l&3) of the word read by lane (l&48) + 4j + ((l>>2)&3).
//
// Feeding an MFMA B-fragment B[k=(l>>4)*8+e][j=l&15] for a 32-row
// contraction chunk at `rowbase`, 16-col block at `d0`, from a ROW-MAJOR
// image with stride S: lane l passes the address of
//   row = rowbase + 8*(l>>4) + 4h + ((l>>2)&3),  col = d0 + 4*(l&3)
// and read h delivers fragment elements e = 4h..4h+3.
//
// Conflicts (tools_lds_sim.py, validated against SQ_LDS_BANK_CONFLICT):
// with S=112 the natural row order 2-way conflicts (rows 0 and 8 alias);
// storing image rows with bits 2 and 3 SWAPPED makes both the tr reads and
// the b128 row-major B-frag reads conflict-free. All images consumed by tr
// reads therefore use PI23 row placement (writers and readers alike).
// ---------------------------------------------------------------------------
#define PI23(q) (((q) & ~12) | ((((q) >> 2) & 1) << 3) | ((((q) >> 3) & 1) << 2))

typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));

// The reads and their waitcnt MUST live in one asm block: the compiler
// assumes an asm statement completes synchronously, so with a separate wait
// it is free to copy/shuffle the "ready" result registers before the data
// actually lands (observed as silent corruption of one of two fragments).
// Earlyclobber outputs keep the dest registers disjoint from the address
// operands of later reads in the same block.

// B-fragment from a PI23 row-major image: rowbase = 32-row chunk start
// (multiple of 32), d0 = 16-col block start.
__device__ __forceinline__ bf16x8 lds_tr16_bfrag(const __bf16* img, int rowbase, int S, int d0, int lane) {
    const int g8 = (lane >> 4) * 8;
    const int jj = (lane >> 2) & 3;
    const int cc = (lane & 3) * 4;
    unsigned a0 = (unsigned)(size_t)(img + PI23(rowbase + g8 + jj) * S + d0 + cc);
    unsigned a1 = (unsigned)(size_t)(img + PI23(rowbase + g8 + 4 + jj) * S + d0 + cc);
    bf16x4 lo, hi;
    asm volatile(
        "ds_read_b64_tr_b16 %0, %2\n\t"
        "ds_read_b64_tr_b16 %1, %3\n\t"
        "s_waitcnt lgkmcnt(0)"
        : "=&v"(lo), "=&v"(hi)
        : "v"(a0), "v"(a1)
        : "memory");
    bf16x8 out;
#pragma unroll
    for (int t = 0; t < 4; ++t) { out[t] = lo[t]; out[4 + t] = hi[t]; }
    return out;
}

// ---------------------------------------------------------------------------
// Pipelined tr16 ladder (guide §5.7 forms (i)/(ii) + T3/T4 counted waits).
//
// Round 1 measured that per-fragment `lgkmcnt(0)` drains serialize the MFMA
// stream (each tr16 read's 50-cycle latency lands on the critical path).
// The ladder splits issue and wait: fragment i+1's reads are issued BEFORE
// waiting on fragment i with a counted `lgkmcnt(N)` (N = the just-issued
// reads), so LDS latency hides under the current fragment's MFMAs and the
// counter never drains to 0 inside the loop.
//
// Count discipline: the ladder's waits are exact only if no OTHER lgkm op
// is outstanding, so each region (a) opens with a full drain that names the
// already-loaded operands (strip A-fragments), and (b) is fenced with
// sched_barrier(0) so the compiler cannot move its own ds/smem ops inside.
// ---------------------------------------------------------------------------

// Issue one B-fragment's two transpose reads (no wait): OFF = byte offset of
// the (rowbase, d0) chunk inside the image, a0/a1 = the lane's two PI23 row
// base addresses (computed once per kernel).
template <int OFF>
__device__ __forceinline__ void tr16_issue(unsigned a0, unsigned a1, bf16x4& lo, bf16x4& hi) {
    asm volatile(
        "ds_read_b64_tr_b16 %0, %2 offset:%c4\n\t"
        "ds_read_b64_tr_b16 %1, %3 offset:%c4"
        : "=&v"(lo), "=&v"(hi)
        : "v"(a0), "v"(a1), "i"(OFF)
        : "memory");
}

// Counted wait naming the registers it guarantees (guide §5.7 form (ii)).
template <int N>
__device__ __forceinline__ void lgkm_wait2(bf16x4& a, bf16x4& b) {
    asm volatile("s_waitcnt lgkmcnt(%c2)" : "+v"(a), "+v"(b) : "i"(N));
}
template <int N>
__device__ __forceinline__ void lgkm_wait4(bf16x4& a, bf16x4& b, bf16x4& c, bf16x4& d) {
    asm volatile("s_waitcnt lgkmcnt(%c4)" : "+v"(a), "+v"(b), "+v"(c), "+v"(d) : "i"(N));
}
template <int N>
__device__ __forceinline__ void lgkm_wait8(bf16x4& a, bf16x4& b, bf16x4& c, bf16x4& d,
                                           bf16x4& e, bf16x4& f, bf16x4& g, bf16x4& h) {
    asm volatile("s_waitcnt lgkmcnt(%c8)"
                 : "+v"(a), "+v"(b), "+v"(c), "+v"(d), "+v"(e), "+v"(f), "+v"(g), "+v"(h)
                 : "i"(N));
}

__device__ __forceinline__ bf16x8 tr16_join8(bf16x4 lo, bf16x4 hi) {
    bf16x8 out;
#pragma unroll
    for (int t = 0; t < 4; ++t) { out[t] = lo[t]; out[4 + t] = hi[t]; }
    return out;
}

// Full drain opening a ladder region; names the strip fragments the region's
// MFMAs consume so their (compiler-issued) reads are complete and counted out.
__device__ __forceinline__ void lgkm_drain2x8(bf16x8& a, bf16x8& b) {
    asm volatile("s_waitcnt lgkmcnt(0)" : "+v"(a), "+v"(b));
}
__device__ __forceinline__ void lgkm_drain4x8(bf16x8& a, bf16x8& b, bf16x8& c, bf16x8& d) {
    asm volatile("s_waitcnt lgkmcnt(0)" : "+v"(a), "+v"(b), "+v"(c), "+v"(d));
}

// Branchless guarded 16B load. Requires D % 8 == 0 (every head dim on this
// path): a chunk is then fully inside [0, D) or fully in the pad, so the
// guard reduces to ONE wave-divergent-free vector load from a clamped
// address plus per-element selects. (The original branchy version emitted
// 8 scalar loads + a vector load per fragment and forced hipcc to drain
// vmcnt(0) before every dependent MFMA — the dominant stall of round 1's
// backward kernel.)
__device__ __forceinline__ bf16x8 load_bf16x8_guard(const __bf16* p, int d0, int D, bool valid) {
    const __bf16* q = (d0 < D) ? p : (p - d0);  // clamp into the row
    bool ok = valid && (d0 < D);
    bf16x8 r = *(const bf16x8*)q;
#pragma unroll
    for (int e = 0; e < 8; ++e) r[e] = ok ? r[e] : (__bf16)0.f;
    return r;
}

// 16-lane row reduction (the 16 columns of one C-fragment row live in the
// 16 lanes of one quarter-wave). DPP row_ror rotations instead of
// __shfl_xor: hipcc lowers the xor shuffles to ds_bpermute — 32 LDS
// instructions (plus their lgkm waits) per fwd loop iteration for an
// issue-bound kernel. row_ror by 8/4/2/1 inside the 16-lane DPP row is
// pure VALU and reduces the same 16-lane set.
__device__ __forceinline__ float qwave_reduce_max(float v) {
    v = fmaxf(v, dpp_ror_f32<8>(v));
    v = fmaxf(v, dpp_ror_f32<4>(v));
    v = fmaxf(v, dpp_ror_f32<2>(v));
    v = fmaxf(v, dpp_ror_f32<1>(v));
    return v;
}
__device__ __forceinline__ float qwave_reduce_sum(float v) {
    v += dpp_ror_f32<8>(v);
    v += dpp_ror_f32<4>(v);
    v += dpp_ror_f32<2>(v);
    v += dpp_ror_f32<1>(v);
    return v;
}


// XCD-aware block remap (guide T1): the dispatcher places flat block id b on
// XCD b%8, so the raw grid (tile, seq, head) spreads one (seq, head)'s tiles
// over all 8 L2s and every tile re-fetches the same Q/dO/K/V slices from HBM
// (measured: 16 GB/launch vs ~1 GB algorithmic on the backward dkv kernel).
// Remap so ALL tiles of one (seq, head) stay on ONE XCD and run
// back-to-back: its ~1.3 MB working set then lives in that XCD's 4 MB L2.
// Pure permutation — correctness never depends on placement.
__device__ __forceinline__ void xcd_remap_tile_bh(int& tile, int& b, int& h) {
    const int ntile = gridDim.x, nb = gridDim.y, nh = gridDim.z;
    const int64_t nbh = (int64_t)nb * nh;
    if (nbh % 8 != 0) return;  // identity fallback for tiny grids
    int64_t raw = blockIdx.x + (int64_t)ntile * (blockIdx.y + (int64_t)nb * blockIdx.z);
    int xcd = (int)(raw % 8);
    int64_t idx = raw / 8;             // per-XCD sequence number
    int64_t pair_local = idx / ntile;  // which (seq, head) pair on this XCD
    tile = (int)(idx % ntile);
    int64_t pair = pair_local * 8 + xcd;
    b = (int)(pair % nb);
    h = (int)(pair / nb);
}

// ===========================================================================
// Forward
// ===========================================================================

template <int DPAD>
__global__ void __launch_bounds__(512, 4) fa_fwd_kernel(
    const __bf16* __restrict__ q, const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    __bf16* __restrict__ o, float* __restrict__ lse,
    const int32_t* __restrict__ cu, int H, int Hkv, int D, int G,
    int64_t q_ts, int64_t q_gs, int64_t k_ts, int64_t k_hs, int64_t v_ts, int64_t v_hs,
    int64_t o_ts, int64_t T_total, float scale) {
    constexpr int KCH = DPAD / 32;   // contraction chunks for QK^T
    constexpr int DCH = DPAD / 16;   // output col blocks for PV
    // Strides chosen with tools_lds_sim.py (exact gfx950 bank model):
    //   row-major K image: stride DPAD+16 makes every b128 B-frag read
    //     conflict-free (the b128 lane groups mix two lg half-rows offset by
    //     4 dwords; +8 padding left them 2-way conflicted);
    //   transposed V image: stride 96 + block swizzle (col>>3)^(3*(row>>3)&7)
    //     -> conflict-free b128 reads;
    //   P strips: stride 72 (b16 writes conflict-free; reads 2-way — the
    //     reverse trade costs more write cycles than it saves).
    constexpr int SK = DPAD + 16;    // K/V LDS row stride (elems)
    constexpr int SV = 64 + 8;       // P strip row stride

    int tile_id = blockIdx.x, b = blockIdx.y, h = blockIdx.z;
    xcd_remap_tile_bh(tile_id, b, h);
    // cu-derived values are wave-uniform: pin them to SGPRs so per-lane
    // address arithmetic built on them does not hold VGPR pairs across the
    // main loop (measured: the compiler otherwise spills pointer pairs)
    const int s0 = __builtin_amdgcn_readfirstlane(cu[b]);
    const int L = __builtin_amdgcn_readfirstlane(cu[b + 1]) - s0;
    // heaviest tiles (largest qs -> most k-tiles) dispatch FIRST: in-order
    // dispatch otherwise schedules the long-pole causal workgroups last
    // 8 waves x 16 rows = 128 q rows per workgroup: K/V staging and
    // barriers amortize over twice the compute at the same per-wave
    // register budget
    const int ntile_seq = (L + 127) / 128;
    if (tile_id >= ntile_seq) return;
    const int qs = (ntile_seq - 1 - tile_id) * 128;

    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int lr = lane & 15;   // fragment col / A-row index
    const int lg = lane >> 4;   // fragment k-group / C-row group

    // Two double-buffered staging pipelines were built and MEASURED OUT:
    //   (a) register-staged issue-early/write-late: 5.44 -> 5.84 ms (the
    //       write pass serializes behind the MFMAs, guide T14's warning);
    //   (b) LDS-DMA (buffer_load..lds) into the inactive buffer: wall-time
    //       NEUTRAL (5.65 ms) and NUMERICALLY UNSAFE as compiled — hipcc
    //       waterfalls the buffer descriptors it cannot prove wave-uniform
    //       (T20) and issues some chunk DMAs under partial exec masks, so
    //       inactive lanes leave stale LDS in the image (flaky parity
    //       failures across runs; see git history for the full variant).
    // At 2 workgroups/CU the co-resident workgroup already hides staging
    // latency, so the single-buffer two-barrier schedule stays.
    constexpr int NBUF = 1;
    extern __shared__ char smem_raw[];
    __bf16* Klds = (__bf16*)smem_raw;              // [NBUF][64][SK]
    __bf16* Vlds = Klds + NBUF * 64 * SK;          // [NBUF][64 key][SK] (PI23 rows)
    __bf16* Plds = Vlds + NBUF * 64 * SK;          // [8 waves][16][SV]
    __bf16* Pw = Plds + wave * 16 * SV;
    // V is stored row-major (coalesced staging) and consumed as the PV
    // B-fragment via ds_read_b64_tr_b16 — no transposed image, no scatter.

    const int kvh = h / G;
    const int64_t q_hoff = (int64_t)(h / G) * q_gs + (int64_t)(h % G) * D;

    // --- load this wave's Q fragments (A-layout: i = lr, k = lg*8+e) ---
    const int qrow = qs + wave * 16 + lr;
    const bool qvalid = qrow < L;
    bf16x8 qf[KCH];
#pragma unroll
    for (int kc = 0; kc < KCH; ++kc) {
        int d0 = kc * 32 + lg * 8;
        const __bf16* p = q + (int64_t)(s0 + (qvalid ? qrow : 0)) * q_ts + q_hoff + d0;
        qf[kc] = load_bf16x8_guard(p, d0, D, qvalid);
    }

    float m_run[4], l_run[4];
    f32x4 o_acc[DCH];
#pragma unroll
    for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
#pragma unroll
    for (int dc = 0; dc < DCH; ++dc) o_acc[dc] = {0.f, 0.f, 0.f, 0.f};

    // lane base addresses for the PV tr16 ladder (PI23 row placement; the
    // per-step (rowbase, d0) displacement goes in the ds_read immediate)
    const int tr_jj = (lane >> 2) & 3;
    const int tr_g8 = (lane >> 4) * 8;
    const int tr_cc = (lane & 3) * 4;
    const unsigned aV0 = (unsigned)(size_t)(Vlds + PI23(tr_g8 + tr_jj) * SK + tr_cc);
    const unsigned aV1 = (unsigned)(size_t)(Vlds + PI23(tr_g8 + 4 + tr_jj) * SK + tr_cc);

    const int kend = min(L, qs + 128);
    const int ntiles = (kend + 63) / 64;

    if (D < DPAD) {  // pad columns zeroed once (see dkv note)
        for (int pidx = threadIdx.x; pidx < 64 * (DPAD - D) / 8; pidx += 512) {
            int key = pidx / ((DPAD - D) / 8);
            int d0 = D + (pidx % ((DPAD - D) / 8)) * 8;
            bf16x8 z = {};
            *(bf16x8*)&Klds[key * SK + d0] = z;
            *(bf16x8*)&Vlds[PI23(key) * SK + d0] = z;
        }
    }

    // tile-invariant staging piece coordinates (see dkv note)
    int st_koff[2], st_voff[2];
    int st_klds[2], st_vlds[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        int pidx = (int)threadIdx.x + j * 512;
        int pp = pidx < 64 * D / 8 ? pidx : 0;
        int key = pp / (D / 8);
        int d0 = (pp % (D / 8)) * 8;
        st_koff[j] = (int)(key * k_ts + kvh * k_hs) + d0;   // fits 32 bits
        st_voff[j] = (int)(key * v_ts + kvh * v_hs) + d0;
        st_klds[j] = key * SK + d0;
        st_vlds[j] = PI23(key) * SK + d0;
    }

    // T5 static priority: the later-dispatched half of an 8-wave workgroup
    // loses VALU arbitration; one setprio for it, no per-cluster flips.
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256) __builtin_amdgcn_s_setprio(1);

    constexpr int PIECES = 64 * DPAD / 8;  // 8-elem staging pieces

    // direct cooperative staging (prologue and the NBUF==1 path)
    auto stage_direct = [&](int ks_, __bf16* Kw, __bf16* Vw) {
        if (ks_ + 64 <= kend) {
            const int64_t disp = (int64_t)(s0 + ks_);
#pragma unroll
            for (int j = 0; j < 2; ++j) {
                if ((int)threadIdx.x + j * 512 < 64 * D / 8) {
                    *(bf16x8*)&Kw[st_klds[j]] = *(const bf16x8*)(k + disp * k_ts + st_koff[j]);
                    *(bf16x8*)&Vw[st_vlds[j]] = *(const bf16x8*)(v + disp * v_ts + st_voff[j]);
                }
            }
        } else {
            for (int pidx = threadIdx.x; pidx < PIECES; pidx += 512) {
                int key = pidx / (DPAD / 8);
                int d0 = (pidx % (DPAD / 8)) * 8;
                bool kv_valid = (ks_ + key) < kend;
                const __bf16* kp = k + (int64_t)(s0 + (kv_valid ? ks_ + key : 0)) * k_ts + (int64_t)kvh * k_hs + d0;
                *(bf16x8*)&Kw[key * SK + d0] = load_bf16x8_guard(kp, d0, D, kv_valid);
                const __bf16* vp = v + (int64_t)(s0 + (kv_valid ? ks_ + key : 0)) * v_ts + (int64_t)kvh * v_hs + d0;
                *(bf16x8*)&Vw[PI23(key) * SK + d0] = load_bf16x8_guard(vp, d0, D, kv_valid);
            }
        }
    };

    auto compute_tile = [&](const __bf16* Kb, unsigned aV0c, unsigned aV1c, int ks) {
        // --- QK^T: 4 key-blocks of 16, accumulate over KCH chunks ---
        f32x4 sc[4];
#pragma unroll
        for (int cb = 0; cb < 4; ++cb) {
            sc[cb] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kc = 0; kc < KCH; ++kc) {
                bf16x8 kf = *(const bf16x8*)&Kb[(cb * 16 + lr) * SK + kc * 32 + lg * 8];
                sc[cb] = MFMA16(qf[kc], kf, sc[cb]);
            }
        }

        // --- mask + online softmax (fp32), streamed per C-row: the row's
        // 4 scores live in 4 registers and P goes straight to the LDS strip
        // (keeps the per-tile live set small for the 128-VGPR cap).
        // Wave-uniform mask split: full tiles skip the per-element compare
        // chain entirely (the kernel is issue-bound); the -INF guards for
        // the FIRST tile's running max stay in both branches. ---
        const float scale2 = scale * DOL_LOG2E;  // exp2-domain scores
        // One body, two cheap reductions of per-element work:
        //  - the mask-compare chain runs only on edge tiles (wave-uniform
        //    branch; a full dual-body split cost fwd 32 VGPRs + spills);
        //  - the -INFINITY select guards are gone: v_exp_f32(-inf) is
        //    exactly +0, and mnew is never -inf after the fully-masked-row
        //    force, so exp2f does the zeroing for free.
        const bool edge_tile =
            !((ks + 63 <= qs + wave * 16) && (ks + 64 <= kend) && (qs + wave * 16 + 16 <= L));
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            float pv[4];
#pragma unroll
            for (int cb = 0; cb < 4; ++cb) pv[cb] = sc[cb][r] * scale2;
            if (__builtin_amdgcn_readfirstlane(edge_tile ? 1 : 0)) {
                const int qpos = qs + wave * 16 + lg * 4 + r;
#pragma unroll
                for (int cb = 0; cb < 4; ++cb) {
                    const int kpos = ks + cb * 16 + lr;
                    if (kpos > qpos || kpos >= kend || qpos >= L) pv[cb] = -INFINITY;
                }
            }
            float rowmax = fmaxf(fmaxf(pv[0], pv[1]), fmaxf(pv[2], pv[3]));
            rowmax = qwave_reduce_max(rowmax);
            float mnew = fmaxf(m_run[r], rowmax);
            if (mnew == -INFINITY) mnew = 0.f;  // fully-masked row guard
            float alpha = exp2f(m_run[r] - mnew);
            float rsum = 0.f;
#pragma unroll
            for (int cb = 0; cb < 4; ++cb) {
                float e = exp2f(pv[cb] - mnew);
                Pw[(lg * 4 + r) * SV + cb * 16 + lr] = (__bf16)e;
                rsum += e;
            }
            rsum = qwave_reduce_sum(rsum);
            l_run[r] = l_run[r] * alpha + rsum;
            m_run[r] = mnew;
#pragma unroll
            for (int dc = 0; dc < DCH; ++dc) o_acc[dc][r] *= alpha;
        }

        // --- PV: A = P (this wave's rows), B via the pipelined tr16 ladder
        // over the row-major PI23 V image (issue frag i+1, counted-wait
        // frag i, MFMA — LDS latency hides under the matrix pipe) ---
        {
            bf16x8 pf0 = *(const bf16x8*)&Pw[lr * SV + lg * 8];
            bf16x8 pf1 = *(const bf16x8*)&Pw[lr * SV + 32 + lg * 8];
            lgkm_drain2x8(pf0, pf1);
            __builtin_amdgcn_sched_barrier(0);
            // depth-3 pipeline: two fragments stay in flight — ONE MFMA
            // (~17 cyc) does not cover the ~50-cycle LDS read latency that
            // depth-2 exposed at every counted wait
            bf16x4 vlo[3], vhi[3];
            tr16_issue<0>(aV0c, aV1c, vlo[0], vhi[0]);
            tr16_issue<((1 / DCH) * 32 * SK + (1 % DCH) * 16) * 2>(aV0c, aV1c, vlo[1], vhi[1]);
#define FWD_PV_STEP(i)                                                                                  \
    if constexpr ((i) < 2 * DCH) {                                                                      \
        constexpr int kc2_ = (i) / DCH, dc_ = (i) % DCH;                                                \
        if constexpr ((i) + 2 < 2 * DCH) {                                                              \
            constexpr int kn_ = ((i) + 2) / DCH, dn_ = ((i) + 2) % DCH;                                 \
            tr16_issue<(kn_ * 32 * SK + dn_ * 16) * 2>(aV0c, aV1c, vlo[((i) + 2) % 3], vhi[((i) + 2) % 3]); \
            lgkm_wait2<4>(vlo[(i) % 3], vhi[(i) % 3]);                                                  \
        } else if constexpr ((i) + 1 < 2 * DCH) {                                                       \
            lgkm_wait2<2>(vlo[(i) % 3], vhi[(i) % 3]);                                                  \
        } else {                                                                                        \
            lgkm_wait2<0>(vlo[(i) % 3], vhi[(i) % 3]);                                                  \
        }                                                                                               \
        o_acc[dc_] = MFMA16(kc2_ ? pf1 : pf0, tr16_join8(vlo[(i) % 3], vhi[(i) % 3]), o_acc[dc_]);      \
    }
            FWD_PV_STEP(0) FWD_PV_STEP(1) FWD_PV_STEP(2) FWD_PV_STEP(3)
            FWD_PV_STEP(4) FWD_PV_STEP(5) FWD_PV_STEP(6) FWD_PV_STEP(7)
            FWD_PV_STEP(8) FWD_PV_STEP(9) FWD_PV_STEP(10) FWD_PV_STEP(11)
            FWD_PV_STEP(12) FWD_PV_STEP(13) FWD_PV_STEP(14) FWD_PV_STEP(15)
#undef FWD_PV_STEP
            __builtin_amdgcn_sched_barrier(0);
        }
    };

    if constexpr (NBUF == 2) {
        // ---- LDS-DMA staging setup ----
        // Chunk = one buffer_load_dwordx4..lds instruction: 64 lanes x 16 B
        // of contiguous LDS. Per tensor image: 64*SK*2/1024 chunks; this
        // wave owns chunks c = wave, wave+8, ... across K then V.
        constexpr int CH = (64 * SK * 2) / 1024;   // chunks per image (SK%8==0)
        constexpr int NC = (2 * CH + 7) / 8;       // chunks this wave may own
        // Per-sequence, per-kv-head buffer descriptors: OOB voffsets
        // (key >= L, or MFMA pad columns sent out of range) read as zero.
        auto rsrcK = __builtin_amdgcn_make_buffer_rsrc(
            (void*)(k + (int64_t)s0 * k_ts + (int64_t)kvh * k_hs), 0,
            (unsigned)(((int64_t)(L - 1) * k_ts + D) * 2), 0);
        auto rsrcV = __builtin_amdgcn_make_buffer_rsrc(
            (void*)(v + (int64_t)s0 * v_ts + (int64_t)kvh * v_hs), 0,
            (unsigned)(((int64_t)(L - 1) * v_ts + D) * 2), 0);
        // Loop-invariant per-lane voffsets (bytes) for key tile 0; per tile
        // add ks*stride*2. Pad columns get 0x40000000 (stays OOB after the
        // per-tile add — tensors on this path are < 1 GiB).
        unsigned choff[NC];
        bool ch_is_v[NC];
#pragma unroll
        for (int i = 0; i < NC; ++i) {
            int c = wave + i * 8;
            bool isv = c >= CH;
            int cc = isv ? c - CH : c;
            int e0 = cc * 512 + lane * 8;
            int row = e0 / SK;
            int col = e0 % SK;
            int key = isv ? PI23(row) : row;   // V image rows are PI23-placed
            int64_t ts = isv ? v_ts : k_ts;
            unsigned off = (unsigned)((key * ts + col) * 2);
            if (col >= D || c >= 2 * CH) off = 0x40000000u;
            choff[i] = off;
            ch_is_v[i] = isv;
        }
        const unsigned kstep2 = (unsigned)(k_ts * 2) * 64u;  // voffset delta per 64-key tile
        const unsigned vstep2 = (unsigned)(v_ts * 2) * 64u;

        auto stage_dma = [&](int kt_, int buf_) {
            __bf16* Kb = Klds + buf_ * 64 * SK;
            __bf16* Vb = Vlds + buf_ * 64 * SK;
#pragma unroll
            for (int i = 0; i < NC; ++i) {
                int c = wave + i * 8;
                if (c >= 2 * CH) break;
                bool isv = ch_is_v[i];
                int cc = isv ? c - CH : c;
                char* ldsp = (char*)(isv ? Vb : Kb) + cc * 1024;
                unsigned vo = choff[i] + (unsigned)kt_ * (isv ? vstep2 : kstep2);
                __builtin_amdgcn_raw_ptr_buffer_load_lds(
                    isv ? rsrcV : rsrcK,
                    (__attribute__((address_space(3))) void*)ldsp, 16, vo, 0, 0, 0);
            }
        };

        stage_dma(0, 0);
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __syncthreads();
        for (int kt = 0; kt < ntiles; ++kt) {
            const int cur = kt & 1;
            if (kt + 1 < ntiles) stage_dma(kt + 1, cur ^ 1);  // in flight under compute
            compute_tile(Klds + cur * 64 * SK,
                         aV0 + (unsigned)(cur * 64 * SK * 2),
                         aV1 + (unsigned)(cur * 64 * SK * 2), kt * 64);
            // the DMA into buf cur^1 must land before the next iteration
            // reads it; the barrier then publishes it workgroup-wide
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __syncthreads();
        }
    } else {
        for (int kt = 0; kt < ntiles; ++kt) {
            stage_direct(kt * 64, Klds, Vlds);
            __syncthreads();
            compute_tile(Klds, aV0, aV1, kt * 64);
            __syncthreads();  // K/V LDS reused next tile
        }
    }

    // --- epilogue: normalize, store O and LSE ---
    float inv_l[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        inv_l[r] = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
        const int qpos = qs + wave * 16 + lg * 4 + r;
        if (lr == 0 && qpos < L)
            lse[(int64_t)h * T_total + s0 + qpos] = (m_run[r] + log2f(l_run[r])) * DOL_LN2;
    }
#pragma unroll
    for (int dc = 0; dc < DCH; ++dc) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int qpos = qs + wave * 16 + lg * 4 + r;
            const int d = dc * 16 + lr;
            if (qpos < L && d < D)
                o[(int64_t)(s0 + qpos) * o_ts + (int64_t)h * D + d] = (__bf16)(o_acc[dc][r] * inv_l[r]);
        }
    }
}

template <int DPAD>
static int launch_fa_fwd(hipStream_t stream, const __bf16* q, const __bf16* k, const __bf16* v,
                         __bf16* o, float* lse, const int32_t* cu, int batch, int64_t T,
                         int H, int Hkv, int D, int G,
                         int64_t q_ts, int64_t q_gs, int64_t k_ts, int64_t k_hs,
                         int64_t v_ts, int64_t v_hs, int max_tiles, float scale) {
    dim3 grid(max_tiles, batch, H), block(512);
    size_t shmem = ((size_t)(DPAD <= 96 ? 2 : 1) * 64 * (DPAD + 16) * 2 + 8 * 16 * 72) * sizeof(__bf16);
    hipLaunchKernelGGL((fa_fwd_kernel<DPAD>), grid, block, shmem, stream,
                       q, k, v, o, lse, cu, H, Hkv, D, G,
                       q_ts, q_gs, k_ts, k_hs, v_ts, v_hs, (int64_t)H * D, T, scale);
    return dol_last_error();
}

extern "C" int dolomite_fa_varlen_fwd(dolomite_stream_t stream,
                                      const void* q, const void* k, const void* v,
                                      void* o, float* lse,
                                      const int32_t* cu_seqlens, int batch, int max_seqlen, int64_t T,
                                      int H, int Hkv, int D, int G,
                                      int64_t q_tstride, int64_t q_gstride,
                                      int64_t k_tstride, int64_t k_hstride,
                                      int64_t v_tstride, int64_t v_hstride,
                                      float scale, int dtype) {
    if (dtype != DOLOMITE_BF16) return 9010;  // bf16 only (north-star dtype)
    if (D > 128) return 9011;
    // grid.x = tiles of the longest sequence; shorter sequences' surplus
    // workgroups exit on the cu_seqlens check.
    int max_tiles = (max_seqlen + 127) / 128;
    hipStream_t s = (hipStream_t)stream;
#define CASE(DP)                                                                                     \
    return launch_fa_fwd<DP>(s, (const __bf16*)q, (const __bf16*)k, (const __bf16*)v, (__bf16*)o,    \
                             lse, cu_seqlens, batch, T, H, Hkv, D, G, q_tstride, q_gstride,          \
                             k_tstride, k_hstride, v_tstride, v_hstride, max_tiles, scale)
    if (D <= 32) CASE(32);
    if (D <= 64) CASE(64);
    if (D <= 96) CASE(96);
    CASE(128);
#undef CASE
}

// ===========================================================================
// Backward pass 1: delta[h,t] = sum_d dO[t,h,d] * O[t,h,d]   (one wave/row)
// ===========================================================================

template <typename T>
__global__ void __launch_bounds__(256) fa_bwd_preprocess_kernel(
    const T* __restrict__ o, const T* __restrict__ dout, float* __restrict__ delta,
    int64_t T_total, int H, int D, int64_t o_ts, int64_t do_ts) {
    int64_t row = ((int64_t)blockIdx.x * 4) + (threadIdx.x >> 6);
    int lane = threadIdx.x & 63;
    if (row >= T_total * H) return;
    int64_t t = row / H;
    int h = (int)(row % H);
    const T* op = o + t * o_ts + (int64_t)h * D;
    const T* dp = dout + t * do_ts + (int64_t)h * D;
    float acc = 0.f;
    // element pairs per lane: coalesced 4-byte loads instead of scalar b16
    for (int d = lane * 2; d + 1 < D; d += 128) {
        float o0[2], d0[2];
        VecIO<T, 2>::load(op + d, o0);
        VecIO<T, 2>::load(dp + d, d0);
        acc += o0[0] * d0[0] + o0[1] * d0[1];
    }
    if (D & 1) {  // odd head dim tail (not on any named config)
        int d = D - 1;
        if (lane == 0) acc += load_as_f32(op + d) * load_as_f32(dp + d);
    }
    acc = wave_reduce_sum(acc);
    if (lane == 0) delta[(int64_t)h * T_total + t] = acc;
}

// Fast path for the contiguous bf16 layout (every named config): the
// one-wave-per-(t,h) kernel above issues only two b32 loads per lane for an
// 80-element row — far too shallow to cover HBM latency (measured 0.36 ms
// vs 0.11 ms algorithmic). Here a block streams a TOK-token tile with b128
// loads (8-element groups never cross a head boundary since D % 8 == 0),
// parks one fp32 partial dot per group in LDS, then one thread per (t, h)
// slot sums its D/8 contiguous partials in FIXED ascending order — the
// reduction stays bit-deterministic for the resume tests.
__global__ void __launch_bounds__(256) fa_bwd_preprocess_tiled(
    const __bf16* __restrict__ o, const __bf16* __restrict__ dout, float* __restrict__ delta,
    int64_t T_total, int H, int D, int TOK) {
    extern __shared__ float part[];  // [TOK*H*D/8]
    const int64_t hd = (int64_t)H * D;
    const int64_t base = (int64_t)blockIdx.x * TOK * hd;
    const int64_t total = T_total * hd;
    const int np = (int)(TOK * hd / 8);
#pragma unroll 4
    for (int p = threadIdx.x; p < np; p += 256) {
        int64_t flat = base + (int64_t)p * 8;
        float acc = 0.f;
        if (flat + 8 <= total) {
            bf16x8 ov = *(const bf16x8*)(o + flat);
            bf16x8 dv = *(const bf16x8*)(dout + flat);
#pragma unroll
            for (int e = 0; e < 8; ++e) acc += (float)ov[e] * (float)dv[e];
        }
        part[p] = acc;
    }
    __syncthreads();
    const int slots = TOK * H;
    const int dp8 = D / 8;
    for (int s = threadIdx.x; s < slots; s += 256) {
        float a = 0.f;
        const float* ps = &part[s * dp8];
        for (int i = 0; i < dp8; ++i) a += ps[i];
        int tl = s / H, h = s - tl * H;
        int64_t t = (int64_t)blockIdx.x * TOK + tl;
        if (t < T_total) delta[(int64_t)h * T_total + t] = a;
    }
}

extern "C" int dolomite_fa_bwd_preprocess(dolomite_stream_t stream,
                                          const void* o, const void* dout, float* delta,
                                          int64_t T, int H, int D,
                                          int64_t o_tstride, int64_t do_tstride, int dtype) {
    const int64_t hd = (int64_t)H * D;
    int TOK = 16;
    while (TOK > 1 && TOK * hd * 4 / 8 > 65536) TOK /= 2;  // LDS partial budget
    if (dtype == DOLOMITE_BF16 && D % 8 == 0 && o_tstride == hd && do_tstride == hd && T > 0 &&
        TOK * hd * 4 / 8 <= 65536) {
        dim3 grid((uint32_t)((T + TOK - 1) / TOK)), block(256);
        size_t shmem = (size_t)(TOK * hd / 8) * sizeof(float);
        hipLaunchKernelGGL((fa_bwd_preprocess_tiled), grid, block, shmem, (hipStream_t)stream,
                           (const __bf16*)o, (const __bf16*)dout, delta, T, H, D, TOK);
        return dol_last_error();
    }
    int64_t rows = T * H;
    dim3 grid((uint32_t)((rows + 3) / 4)), block(256);
    if (dtype == DOLOMITE_BF16)
        hipLaunchKernelGGL((fa_bwd_preprocess_kernel<uint16_t>), grid, block, 0, (hipStream_t)stream,
                           (const uint16_t*)o, (const uint16_t*)dout, delta, T, H, D, o_tstride, do_tstride);
    else
        hipLaunchKernelGGL((fa_bwd_preprocess_kernel<float>), grid, block, 0, (hipStream_t)stream,
                           (const float*)o, (const float*)dout, delta, T, H, D, o_tstride, do_tstride);
    return dol_last_error();
}

// ===========================================================================
// Backward pass 2: per (kv-tile, seq, kv-head) workgroup.
// ===========================================================================

// Backward is split FA2-style into two kernels (each recomputes P from
// q/k/lse — cheaper than the atomics+barriers a fused version needs):
//   fa_bwd_dkv_kernel : grid (kv-tile, seq, q-head); dK/dV accumulate in
//     registers over the q-tile loop, joined into fp32 buffers by one atomic
//     pass per workgroup (G q-head contributors per kv strip).
//   fa_bwd_dq_kernel  : grid (q-tile, seq, q-head); dQ accumulates in
//     registers over the kv-tile loop — no atomics at all — and is stored
//     once. K^T is the only LDS image (swizzled), staged per kv tile.

// dkv round-2 structure: NO transposed LDS images. The Q/dO images are
// stored row-major with PI23 row placement only, read two ways:
//   - S^T/dP^T B-fragments: b128 row reads (row cb*16+PI23(lr), conflict-free
//     at stride DPAD+16 per tools_lds_sim.py);
//   - dK/dV B-fragments: ds_read_b64_tr_b16 ladder (same PI23 image).
// This removes 2*DPAD*96 elements of LDS (102 KB -> 64 KB at DPAD=96) and
// the 16 scalar b16 scatter-writes per staged piece. Together with the
// __launch_bounds__(512, 4) register cap (<=128 VGPR) the kernel reaches
// 2 workgroups/CU: while one workgroup stages Q/dO from HBM the co-resident
// one runs its MFMA segments — the cross-workgroup latency hiding the
// round-1 single-occupancy version lacked.
// Occupancy bound: <=96 head-dim fits the 128-VGPR / 64-KB-LDS budget for
// 2 workgroups/CU (measured: 128 VGPR, 3 prologue spills); the 128 variant
// does not (43 spills into the hot loop at the cap) and keeps 1 WG/CU.
template <int DPAD>
__global__ void __launch_bounds__(512, DPAD <= 96 ? 4 : 2) fa_bwd_dkv_kernel(
    const __bf16* __restrict__ q, const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    const __bf16* __restrict__ dout, const float* __restrict__ lse, const float* __restrict__ delta,
    float* __restrict__ dk_acc, float* __restrict__ dv_acc,
    const int32_t* __restrict__ cu, int H, int Hkv, int D, int G,
    int64_t q_ts, int64_t q_gs, int64_t k_ts, int64_t k_hs, int64_t v_ts, int64_t v_hs,
    int64_t do_ts, int64_t T_total, float scale) {
    constexpr int KCH = DPAD / 32;
    constexpr int DCH = DPAD / 16;

    int tile_id = blockIdx.x, b = blockIdx.y, h = blockIdx.z;
    xcd_remap_tile_bh(tile_id, b, h);
    const int kvh = h / G;
    // cu-derived values are wave-uniform: pin them to SGPRs so per-lane
    // address arithmetic built on them does not hold VGPR pairs across the
    // main loop (measured: the compiler otherwise spills pointer pairs)
    const int s0 = __builtin_amdgcn_readfirstlane(cu[b]);
    const int L = __builtin_amdgcn_readfirstlane(cu[b + 1]) - s0;
    // 8 waves x 16 keys = 128-key strip per workgroup
    const int ks = tile_id * 128;
    if (ks >= L) return;

    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int lr = lane & 15;
    const int lg = lane >> 4;

    constexpr int SQ = DPAD + 16;              // PI23-rowed image stride
    // Transposed [q][key] strips: each lane's 4 C-rows (keys) are contiguous
    // along the key axis, so the softmax emits ONE ds_write_b64 per strip
    // per q-block instead of 4 scalar b16 stores (32 -> 8 write
    // instructions per wave per tile; the LDS store path was a measured
    // co-bottleneck). Read back as MFMA A-fragments via the tr16 ladder.
    // Stride 132: rows stay 8-byte aligned (ds_write_b64 / tr16 reads need
    // natural alignment — stride*2 must be a multiple of 8; the odd
    // half-stride that would fully de-conflict the write groups is
    // misaligned). 132/2=66 leaves a 2-way write-group conflict, which on
    // ds_write_b64 costs ~2 LDS-array cycles per instruction.
    constexpr int STS = 128 + 4;
    extern __shared__ char smem_raw[];
    __bf16* Qlds = (__bf16*)smem_raw;          // [64 q][SQ] (PI23 rows)
    __bf16* dOl = Qlds + 64 * SQ;              // [64 q][SQ] (PI23 rows)
    __bf16* dSTl = dOl + 64 * SQ;              // [64 q][STS] (dS^T, PI23 rows)
    __bf16* PTl = dSTl + 64 * STS;             // [64 q][STS] (P^T, PI23 rows)

    const int kend = min(L, ks + 128);

    const int krow = ks + wave * 16 + lr;
    const bool kvalid = krow < kend;
    bf16x8 kfr[KCH], vfr[KCH];
#pragma unroll
    for (int kc = 0; kc < KCH; ++kc) {
        int d0 = kc * 32 + lg * 8;
        const __bf16* kp = k + (int64_t)(s0 + (kvalid ? krow : 0)) * k_ts + (int64_t)kvh * k_hs + d0;
        kfr[kc] = load_bf16x8_guard(kp, d0, D, kvalid);
        const __bf16* vp = v + (int64_t)(s0 + (kvalid ? krow : 0)) * v_ts + (int64_t)kvh * v_hs + d0;
        vfr[kc] = load_bf16x8_guard(vp, d0, D, kvalid);
    }

    f32x4 dvr[DCH], dkr[DCH];
#pragma unroll
    for (int dc = 0; dc < DCH; ++dc) {
        dvr[dc] = {0.f, 0.f, 0.f, 0.f};
        dkr[dc] = {0.f, 0.f, 0.f, 0.f};
    }

    // lane base addresses for the dK/dV tr16 ladders
    const int tr_jj = (lane >> 2) & 3;
    const int tr_g8 = (lane >> 4) * 8;
    const int tr_cc = (lane & 3) * 4;
    // Two lane bases per image PAIR: dOl sits 64*SQ elements above Qlds and
    // PTl 64*STS above dSTl — both displacements fit the 16-bit ds_read
    // immediate, so the second image of each pair rides the offset field
    // instead of holding two more address registers (dkv is register-bound).
    const unsigned aQ0 = (unsigned)(size_t)(Qlds + PI23(tr_g8 + tr_jj) * SQ + tr_cc);
    const unsigned aQ1 = (unsigned)(size_t)(Qlds + PI23(tr_g8 + 4 + tr_jj) * SQ + tr_cc);
    const unsigned aDS0 = (unsigned)(size_t)(dSTl + PI23(tr_g8 + tr_jj) * STS + wave * 16 + tr_cc);
    const unsigned aDS1 = (unsigned)(size_t)(dSTl + PI23(tr_g8 + 4 + tr_jj) * STS + wave * 16 + tr_cc);
    constexpr int DO_IMM = 64 * SQ * 2;        // dOl = Qlds + this (bytes)
    constexpr int PT_IMM = 64 * STS * 2;       // PTl = dSTl + this (bytes)
    static_assert(DO_IMM + 32 * SQ * 2 + 16 * 2 < 65536, "ds offset range");
    static_assert(PT_IMM + 32 * STS * 2 + 16 * 2 < 65536, "ds offset range");

    // T5 static priority (guide §5.5): the later-dispatched half of an
    // 8-wave workgroup loses VALU arbitration to the older half; one
    // setprio(1) for that half removes its start-of-segment penalty.
    // The guard must be provably wave-uniform (readfirstlane) or s_setprio
    // is emitted unconditionally.
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256) __builtin_amdgcn_s_setprio(1);

    const int qt0 = ks / 64;
    const int nqt = (L + 63) / 64;
    const int64_t q_hoff = (int64_t)(h / G) * q_gs + (int64_t)(h % G) * D;
    const int64_t do_hoff = (int64_t)h * D;

    // MFMA pad columns [D, DPAD) zeroed ONCE: the fast staging path below
    // then only writes the real D columns per tile (guards and pad-fill
    // gone from the per-tile loop; the guarded edge path rewrites pads
    // with zeros, so the invariant holds under any tile interleave)
    if (D < DPAD) {
        for (int pidx = threadIdx.x; pidx < 64 * (DPAD - D) / 8; pidx += 512) {
            int qq = pidx / ((DPAD - D) / 8);
            int d0 = D + (pidx % ((DPAD - D) / 8)) * 8;
            bf16x8 z = {};
            *(bf16x8*)&Qlds[PI23(qq) * SQ + d0] = z;
            *(bf16x8*)&dOl[PI23(qq) * SQ + d0] = z;
        }
    }

    // (hoisting the staging coordinates like dq/fwd costs 3 scratch
    // pointer reloads per iteration at dkv's 128-VGPR cap and measured
    // slower than recomputing addresses — dkv recomputes, with the
    // runtime division replaced by an exact magic multiply: pidx < 2048
    // and D/8 <= 16, so (pidx * (2^22/(D/8) + 1)) >> 22 is exact)
    const unsigned st_magic = (1u << 22) / (unsigned)(D / 8) + 1;
    for (int qt = qt0; qt < nqt; ++qt) {
        const int qs = qt * 64;
        __syncthreads();  // previous iteration's image reads done
        {
            const int pieces = 64 * DPAD / 8;
            if (qs + 64 <= L) {
                // full interior tile: guard-free staging of the real D cols
                for (int pidx = threadIdx.x; pidx < 64 * D / 8; pidx += 512) {
                    int qq = (int)(((unsigned)pidx * st_magic) >> 22);
                    int d0 = (pidx - qq * (D / 8)) * 8;
                    *(bf16x8*)&Qlds[PI23(qq) * SQ + d0] =
                        *(const bf16x8*)(q + (int64_t)(s0 + qs + qq) * q_ts + q_hoff + d0);
                    *(bf16x8*)&dOl[PI23(qq) * SQ + d0] =
                        *(const bf16x8*)(dout + (int64_t)(s0 + qs + qq) * do_ts + do_hoff + d0);
                }
            } else {
                for (int pidx = threadIdx.x; pidx < pieces; pidx += 512) {
                    int qq = pidx / (DPAD / 8);
                    int d0 = (pidx % (DPAD / 8)) * 8;
                    bool valid = (qs + qq) < L;
                    const __bf16* qp = q + (int64_t)(s0 + (valid ? qs + qq : 0)) * q_ts + q_hoff + d0;
                    *(bf16x8*)&Qlds[PI23(qq) * SQ + d0] = load_bf16x8_guard(qp, d0, D, valid);
                    const __bf16* dp = dout + (int64_t)(s0 + (valid ? qs + qq : 0)) * do_ts + do_hoff + d0;
                    *(bf16x8*)&dOl[PI23(qq) * SQ + d0] = load_bf16x8_guard(dp, d0, D, valid);
                }
            }
        }
        __syncthreads();

        // Per q-column block: S^T = K*Q^T and dP^T = V*dO^T MFMAs, then that
        // block's softmax straight into the LDS strips. Merging the two
        // loops keeps only ONE C-fragment pair (8 VGPRs) live instead of
        // four (32) — the register headroom the (512,4) cap needs.
        // The kernels are ISSUE-bound (SQ counters): the causal-mask
        // compare/select chain runs only on diagonal/edge tiles via a
        // wave-uniform branch; ~94% of tiles take the maskless body.
        const bool full_tile = (ks + wave * 16 + 16 <= qs + 1) && (kend == ks + 64) && (qs + 64 <= L);
#pragma unroll
        for (int cb = 0; cb < 4; ++cb) {
            f32x4 st = {0.f, 0.f, 0.f, 0.f};
            f32x4 dpt = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kc = 0; kc < KCH; ++kc) {
                int d0 = kc * 32 + lg * 8;
                bf16x8 qb = *(const bf16x8*)&Qlds[(cb * 16 + PI23(lr)) * SQ + d0];
                st = MFMA16(kfr[kc], qb, st);
                bf16x8 db = *(const bf16x8*)&dOl[(cb * 16 + PI23(lr)) * SQ + d0];
                dpt = MFMA16(vfr[kc], db, dpt);
            }
            const int qpos = qs + cb * 16 + lr;
            const bool qok = qpos < L;
            float lsev = qok ? lse[(int64_t)h * T_total + s0 + qpos] * DOL_LOG2E : 0.f;
            float delv = qok ? delta[(int64_t)h * T_total + s0 + qpos] : 0.f;
            bf16x4 pvv, dsv;
            if (__builtin_amdgcn_readfirstlane(full_tile ? 1 : 0)) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    float pv = exp2f(st[r] * (scale * DOL_LOG2E) - lsev);
                    pvv[r] = (__bf16)pv;
                    dsv[r] = (__bf16)(pv * (dpt[r] - delv) * scale);
                }
            } else {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int kpos = ks + wave * 16 + lg * 4 + r;
                    bool ok = qok && kpos < kend && kpos <= qpos;
                    float pv = ok ? exp2f(st[r] * (scale * DOL_LOG2E) - lsev) : 0.f;
                    float ds = ok ? pv * (dpt[r] - delv) * scale : 0.f;
                    pvv[r] = (__bf16)pv;
                    dsv[r] = (__bf16)ds;
                }
            }
            const int srow = PI23(cb * 16 + lr) * STS + wave * 16 + lg * 4;
            *(bf16x4*)&PTl[srow] = pvv;
            *(bf16x4*)&dSTl[srow] = dsv;
        }

        // dV += P^T*dO then dK += dS^T*Q (contraction over q) — TWO
        // independent depth-3 tr16 ladders: dO feeds only dV and Q only dK,
        // so splitting halves the live A-fragment registers (one strip pair
        // instead of two) and the freed budget funds a third in-flight
        // fragment — one MFMA per wait no longer exposes LDS latency.
        {
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");  // exact counts from here
            __builtin_amdgcn_sched_barrier(0);
            bf16x4 a_lo[2], a_hi[2];
            tr16_issue<PT_IMM>(aDS0, aDS1, a_lo[0], a_hi[0]);
            tr16_issue<PT_IMM + 32 * STS * 2>(aDS0, aDS1, a_lo[1], a_hi[1]);
            bf16x4 blo[3], bhi[3];
            tr16_issue<DO_IMM>(aQ0, aQ1, blo[0], bhi[0]);
            tr16_issue<DO_IMM + ((1 / DCH) * 32 * SQ + (1 % DCH) * 16) * 2>(aQ0, aQ1, blo[1], bhi[1]);
            lgkm_wait4<4>(a_lo[0], a_hi[0], a_lo[1], a_hi[1]);
            bf16x8 af0 = tr16_join8(a_lo[0], a_hi[0]);
            bf16x8 af1 = tr16_join8(a_lo[1], a_hi[1]);
#define DKV_HALF(i, IMM, ACC, AF0, AF1)                                                                 \
    if constexpr ((i) < 2 * DCH) {                                                                      \
        constexpr int kc2_ = (i) / DCH, dc_ = (i) % DCH;                                                \
        if constexpr ((i) + 2 < 2 * DCH) {                                                              \
            constexpr int kn_ = ((i) + 2) / DCH, dn_ = ((i) + 2) % DCH;                                 \
            tr16_issue<(kn_ * 32 * SQ + dn_ * 16) * 2 + (IMM)>(aQ0, aQ1, blo[((i) + 2) % 3], bhi[((i) + 2) % 3]); \
            lgkm_wait2<4>(blo[(i) % 3], bhi[(i) % 3]);                                                  \
        } else if constexpr ((i) + 1 < 2 * DCH) {                                                       \
            lgkm_wait2<2>(blo[(i) % 3], bhi[(i) % 3]);                                                  \
        } else {                                                                                        \
            lgkm_wait2<0>(blo[(i) % 3], bhi[(i) % 3]);                                                  \
        }                                                                                               \
        ACC[dc_] = MFMA16(kc2_ ? (AF1) : (AF0), tr16_join8(blo[(i) % 3], bhi[(i) % 3]), ACC[dc_]);      \
    }
#define DKV_A(i) DKV_HALF(i, DO_IMM, dvr, af0, af1)
            DKV_A(0) DKV_A(1) DKV_A(2) DKV_A(3)
            DKV_A(4) DKV_A(5) DKV_A(6) DKV_A(7)
            DKV_A(8) DKV_A(9) DKV_A(10) DKV_A(11)
            DKV_A(12) DKV_A(13) DKV_A(14) DKV_A(15)
#undef DKV_A
            // half B: dK += dS^T * Q. The strip A-fragments and the first
            // two Q B-fragments are issued, then one counted wait.
            tr16_issue<0>(aDS0, aDS1, a_lo[0], a_hi[0]);
            tr16_issue<32 * STS * 2>(aDS0, aDS1, a_lo[1], a_hi[1]);
            tr16_issue<0>(aQ0, aQ1, blo[0], bhi[0]);
            tr16_issue<((1 / DCH) * 32 * SQ + (1 % DCH) * 16) * 2>(aQ0, aQ1, blo[1], bhi[1]);
            lgkm_wait4<4>(a_lo[0], a_hi[0], a_lo[1], a_hi[1]);
            af0 = tr16_join8(a_lo[0], a_hi[0]);
            af1 = tr16_join8(a_lo[1], a_hi[1]);
#define DKV_B(i) DKV_HALF(i, 0, dkr, af0, af1)
            DKV_B(0) DKV_B(1) DKV_B(2) DKV_B(3)
            DKV_B(4) DKV_B(5) DKV_B(6) DKV_B(7)
            DKV_B(8) DKV_B(9) DKV_B(10) DKV_B(11)
            DKV_B(12) DKV_B(13) DKV_B(14) DKV_B(15)
#undef DKV_B
#undef DKV_HALF
            __builtin_amdgcn_sched_barrier(0);
        }
    }

    // Deterministic join: each (kv-tile, q-head) workgroup owns its keys'
    // (t, h) slot of the per-head partial buffers exclusively, so dK/dV
    // are plain stores (half the traffic of atomic RMW); fa_grad_finalize
    // reduces the G q-head contributions in a FIXED order. The round-1
    // fp32 atomicAdd join was arrival-order nondeterministic, which made
    // bit-exact checkpoint resume a coin flip.
#pragma unroll
    for (int dc = 0; dc < DCH; ++dc) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int kpos = ks + wave * 16 + lg * 4 + r;
            const int d = dc * 16 + lr;
            if (kpos < kend && d < D) {
                int64_t idx = ((int64_t)(s0 + kpos) * H + h) * D + d;
                // non-temporal: the 1.3 GB of fp32 partials per launch must
                // not evict the XCD-resident Q/dO slices (PMC: dkv read
                // traffic 3.6 GB/launch vs ~1.4 GB ideal with plain stores)
                __builtin_nontemporal_store(dkr[dc][r], &dk_acc[idx]);
                __builtin_nontemporal_store(dvr[dc][r], &dv_acc[idx]);
            }
        }
    }
}

template <int DPAD>
__global__ void __launch_bounds__(512, 4) fa_bwd_dq_kernel(
    const __bf16* __restrict__ q, const __bf16* __restrict__ k, const __bf16* __restrict__ v,
    const __bf16* __restrict__ dout, const float* __restrict__ lse, const float* __restrict__ delta,
    __bf16* __restrict__ dqkv_q,
    const int32_t* __restrict__ cu, int H, int Hkv, int D, int G,
    int64_t q_ts, int64_t q_gs, int64_t k_ts, int64_t k_hs, int64_t v_ts, int64_t v_hs,
    int64_t do_ts, int64_t T_total, float scale) {
    constexpr int KCH = DPAD / 32;
    constexpr int DCH = DPAD / 16;
    constexpr int ST = 64 + 8;                 // strip stride

    int tile_id = blockIdx.x, b = blockIdx.y, h = blockIdx.z;
    xcd_remap_tile_bh(tile_id, b, h);
    const int kvh = h / G;
    // cu-derived values are wave-uniform: pin them to SGPRs so per-lane
    // address arithmetic built on them does not hold VGPR pairs across the
    // main loop (measured: the compiler otherwise spills pointer pairs)
    const int s0 = __builtin_amdgcn_readfirstlane(cu[b]);
    const int L = __builtin_amdgcn_readfirstlane(cu[b + 1]) - s0;
    // 8 waves x 16 rows = 128 q rows per workgroup; heaviest tiles first
    const int ntile_seq = (L + 127) / 128;
    if (tile_id >= ntile_seq) return;
    const int qs = (ntile_seq - 1 - tile_id) * 128;

    const int lane = threadIdx.x & 63;
    const int wave = threadIdx.x >> 6;
    const int lr = lane & 15;
    const int lg = lane >> 4;

    // Round-2: the swizzled K^T image is gone — the dQ+=dS*K B-fragments
    // come from the PI23-rowed row-major K image via the tr16 ladder, like
    // dkv. Saves DPAD*96 LDS elements and the per-piece scatter writes.
    constexpr int SQ = DPAD + 16;              // image stride
    extern __shared__ char smem_raw[];
    __bf16* Klds = (__bf16*)smem_raw;          // [64 key][SQ] (PI23 rows)
    __bf16* Vlds = Klds + 64 * SQ;             // [64 key][SQ] (natural rows)
    __bf16* dSl = Vlds + 64 * SQ;              // [8 waves][16 q][ST] (dS strips)
    __bf16* dSw = dSl + wave * 16 * ST;

    const int64_t q_hoff = (int64_t)(h / G) * q_gs + (int64_t)(h % G) * D;
    const int64_t do_hoff = (int64_t)h * D;

    // lane base addresses for the dQ tr16 ladder over Klds
    const int tr_jj = (lane >> 2) & 3;
    const int tr_g8 = (lane >> 4) * 8;
    const int tr_cc = (lane & 3) * 4;
    const unsigned aK0 = (unsigned)(size_t)(Klds + PI23(tr_g8 + tr_jj) * SQ + tr_cc);
    const unsigned aK1 = (unsigned)(size_t)(Klds + PI23(tr_g8 + 4 + tr_jj) * SQ + tr_cc);

    // this wave's Q and dO fragments (A-layout: i = lr -> q row)
    const int qrow = qs + wave * 16 + lr;
    const bool qvalid = qrow < L;
    bf16x8 qfr[KCH], dfr[KCH];
#pragma unroll
    for (int kc = 0; kc < KCH; ++kc) {
        int d0 = kc * 32 + lg * 8;
        const __bf16* qp = q + (int64_t)(s0 + (qvalid ? qrow : 0)) * q_ts + q_hoff + d0;
        qfr[kc] = load_bf16x8_guard(qp, d0, D, qvalid);
        const __bf16* dp = dout + (int64_t)(s0 + (qvalid ? qrow : 0)) * do_ts + do_hoff + d0;
        dfr[kc] = load_bf16x8_guard(dp, d0, D, qvalid);
    }

    float lsev[4], delv[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int qp_ = qs + wave * 16 + lg * 4 + r;
        bool ok = qp_ < L;
        lsev[r] = ok ? lse[(int64_t)h * T_total + s0 + qp_] * DOL_LOG2E : 0.f;
        delv[r] = ok ? delta[(int64_t)h * T_total + s0 + qp_] : 0.f;
    }

    f32x4 dq[DCH];
#pragma unroll
    for (int dc = 0; dc < DCH; ++dc) dq[dc] = {0.f, 0.f, 0.f, 0.f};

    if (D < DPAD) {  // pad columns zeroed once (see dkv note)
        for (int pidx = threadIdx.x; pidx < 64 * (DPAD - D) / 8; pidx += 512) {
            int key = pidx / ((DPAD - D) / 8);
            int d0 = D + (pidx % ((DPAD - D) / 8)) * 8;
            bf16x8 z = {};
            *(bf16x8*)&Klds[PI23(key) * SQ + d0] = z;
            *(bf16x8*)&Vlds[key * SQ + d0] = z;
        }
    }

    // tile-invariant staging piece coordinates (see dkv note)
    int st_koff[2], st_voff[2];
    int st_klds[2], st_vlds[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
        int pidx = (int)threadIdx.x + j * 512;
        int pp = pidx < 64 * D / 8 ? pidx : 0;
        int key = pp / (D / 8);
        int d0 = (pp % (D / 8)) * 8;
        st_koff[j] = (int)(key * k_ts + kvh * k_hs) + d0;   // fits 32 bits
        st_voff[j] = (int)(key * v_ts + kvh * v_hs) + d0;
        st_klds[j] = PI23(key) * SQ + d0;
        st_vlds[j] = key * SQ + d0;
    }

    const int kend_total = min(L, qs + 128);
    const int nkt = (kend_total + 63) / 64;

    // T5 static priority for the younger workgroup half (see dkv note)
    if (__builtin_amdgcn_readfirstlane(threadIdx.x) >= 256) __builtin_amdgcn_s_setprio(1);

    for (int kt = 0; kt < nkt; ++kt) {
        const int ks = kt * 64;
        __syncthreads();  // previous tile's image reads done
        // stage row-major K (PI23 rows) + V images, cooperative
        {
            const int pieces = 64 * DPAD / 8;
            if (ks + 64 <= kend_total) {
                const int64_t disp = (int64_t)(s0 + ks);
#pragma unroll
                for (int j = 0; j < 2; ++j) {
                    if ((int)threadIdx.x + j * 512 < 64 * D / 8) {
                        *(bf16x8*)&Klds[st_klds[j]] = *(const bf16x8*)(k + disp * k_ts + st_koff[j]);
                        *(bf16x8*)&Vlds[st_vlds[j]] = *(const bf16x8*)(v + disp * v_ts + st_voff[j]);
                    }
                }
            } else {
                for (int pidx = threadIdx.x; pidx < pieces; pidx += 512) {
                    int key = pidx / (DPAD / 8);
                    int d0 = (pidx % (DPAD / 8)) * 8;
                    bool valid = (ks + key) < kend_total;
                    const __bf16* kp = k + (int64_t)(s0 + (valid ? ks + key : 0)) * k_ts + (int64_t)kvh * k_hs + d0;
                    *(bf16x8*)&Klds[PI23(key) * SQ + d0] = load_bf16x8_guard(kp, d0, D, valid);
                    const __bf16* vp = v + (int64_t)(s0 + (valid ? ks + key : 0)) * v_ts + (int64_t)kvh * v_hs + d0;
                    *(bf16x8*)&Vlds[key * SQ + d0] = load_bf16x8_guard(vp, d0, D, valid);
                }
            }
        }
        __syncthreads();

        // Per key-column block: S = Q*K^T and dP = dO*V^T MFMAs, then that
        // block's dS straight into the wave's strip (one C-pair live).
        // Wave-uniform mask split (see dkv note).
        const bool full_tile = (ks + 64 <= qs + wave * 16 + 1) && (ks + 64 <= kend_total) && (qs + wave * 16 + 16 <= L);
#pragma unroll
        for (int cb = 0; cb < 4; ++cb) {
            f32x4 sc = {0.f, 0.f, 0.f, 0.f};
            f32x4 dp = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int kc = 0; kc < KCH; ++kc) {
                int d0 = kc * 32 + lg * 8;
                bf16x8 kb = *(const bf16x8*)&Klds[(cb * 16 + PI23(lr)) * SQ + d0];
                sc = MFMA16(qfr[kc], kb, sc);
                bf16x8 vb = *(const bf16x8*)&Vlds[(cb * 16 + lr) * SQ + d0];
                dp = MFMA16(dfr[kc], vb, dp);
            }
            if (__builtin_amdgcn_readfirstlane(full_tile ? 1 : 0)) {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    float pv = exp2f(sc[r] * (scale * DOL_LOG2E) - lsev[r]);
                    dSw[(lg * 4 + r) * ST + cb * 16 + lr] = (__bf16)(pv * (dp[r] - delv[r]) * scale);
                }
            } else {
#pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int qpos = qs + wave * 16 + lg * 4 + r;
                    const int kpos = ks + cb * 16 + lr;
                    bool ok = qpos < L && kpos < kend_total && kpos <= qpos;
                    float pv = ok ? exp2f(sc[r] * (scale * DOL_LOG2E) - lsev[r]) : 0.f;
                    float ds = ok ? pv * (dp[r] - delv[r]) * scale : 0.f;
                    dSw[(lg * 4 + r) * ST + cb * 16 + lr] = (__bf16)ds;
                }
            }
        }

        // dQ += dS*K (contraction over this tile's keys) — pipelined tr16
        // ladder over the PI23 K image
        {
            bf16x8 dsf0 = *(const bf16x8*)&dSw[lr * ST + lg * 8];
            bf16x8 dsf1 = *(const bf16x8*)&dSw[lr * ST + 32 + lg * 8];
            lgkm_drain2x8(dsf0, dsf1);
            __builtin_amdgcn_sched_barrier(0);
            // depth-4 rotation: 3 fragments in flight per wait (the dq cap
            // has ~16 VGPR of headroom the dkv cap does not)
            bf16x4 klo[4], khi[4];
            tr16_issue<0>(aK0, aK1, klo[0], khi[0]);
            tr16_issue<((1 / DCH) * 32 * SQ + (1 % DCH) * 16) * 2>(aK0, aK1, klo[1], khi[1]);
            tr16_issue<((2 / DCH) * 32 * SQ + (2 % DCH) * 16) * 2>(aK0, aK1, klo[2], khi[2]);
#define DQ_STEP(i)                                                                                      \
    if constexpr ((i) < 2 * DCH) {                                                                      \
        constexpr int kc2_ = (i) / DCH, dc_ = (i) % DCH;                                                \
        if constexpr ((i) + 3 < 2 * DCH) {                                                              \
            constexpr int kn_ = ((i) + 3) / DCH, dn_ = ((i) + 3) % DCH;                                 \
            tr16_issue<(kn_ * 32 * SQ + dn_ * 16) * 2>(aK0, aK1, klo[((i) + 3) % 4], khi[((i) + 3) % 4]); \
            lgkm_wait2<6>(klo[(i) % 4], khi[(i) % 4]);                                                  \
        } else if constexpr ((i) + 2 < 2 * DCH) {                                                       \
            lgkm_wait2<4>(klo[(i) % 4], khi[(i) % 4]);                                                  \
        } else if constexpr ((i) + 1 < 2 * DCH) {                                                       \
            lgkm_wait2<2>(klo[(i) % 4], khi[(i) % 4]);                                                  \
        } else {                                                                                        \
            lgkm_wait2<0>(klo[(i) % 4], khi[(i) % 4]);                                                  \
        }                                                                                               \
        dq[dc_] = MFMA16(kc2_ ? dsf1 : dsf0, tr16_join8(klo[(i) % 4], khi[(i) % 4]), dq[dc_]);          \
    }
            DQ_STEP(0) DQ_STEP(1) DQ_STEP(2) DQ_STEP(3)
            DQ_STEP(4) DQ_STEP(5) DQ_STEP(6) DQ_STEP(7)
            DQ_STEP(8) DQ_STEP(9) DQ_STEP(10) DQ_STEP(11)
            DQ_STEP(12) DQ_STEP(13) DQ_STEP(14) DQ_STEP(15)
#undef DQ_STEP
            __builtin_amdgcn_sched_barrier(0);
        }
    }

    // single bf16 store of the accumulated dQ straight into the packed dqkv
    // (each (t, h, d) is owned by exactly one q-tile workgroup)
#pragma unroll
    for (int dc = 0; dc < DCH; ++dc) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int qpos = qs + wave * 16 + lg * 4 + r;
            const int d = dc * 16 + lr;
            if (qpos < L && d < D)
                dqkv_q[(int64_t)(s0 + qpos) * q_ts + q_hoff + d] = (__bf16)dq[dc][r];
        }
    }
}

template <int DPAD>
static int launch_fa_bwd(hipStream_t stream, const __bf16* q, const __bf16* k, const __bf16* v,
                         const __bf16* dout, const float* lse, const float* delta,
                         __bf16* dqkv_q, float* dk_acc, float* dv_acc,
                         const int32_t* cu, int batch, int64_t T, int H, int Hkv, int D, int G,
                         int64_t q_ts, int64_t q_gs, int64_t k_ts, int64_t k_hs,
                         int64_t v_ts, int64_t v_hs, int64_t do_ts, int max_tiles, float scale) {
    constexpr int ST = 64 + 8;
    dim3 block(512);
    dim3 grid(max_tiles, batch, H);
    constexpr int SQ = DPAD + 16;
    size_t shmem_dkv = (size_t)(64 * (128 + 4) * 2 + 64 * SQ * 2) * sizeof(__bf16);
    hipLaunchKernelGGL((fa_bwd_dkv_kernel<DPAD>), grid, block, shmem_dkv, stream,
                       q, k, v, dout, lse, delta, dk_acc, dv_acc, cu, H, Hkv, D, G,
                       q_ts, q_gs, k_ts, k_hs, v_ts, v_hs, do_ts, T, scale);
    int err = dol_last_error();
    if (err) return err;
    size_t shmem_dq = (size_t)(64 * SQ * 2 + 8 * 16 * ST) * sizeof(__bf16);
    hipLaunchKernelGGL((fa_bwd_dq_kernel<DPAD>), grid, block, shmem_dq, stream,
                       q, k, v, dout, lse, delta, dqkv_q, cu, H, Hkv, D, G,
                       q_ts, q_gs, k_ts, k_hs, v_ts, v_hs, do_ts, T, scale);
    return dol_last_error();
}

extern "C" int dolomite_fa_varlen_bwd(dolomite_stream_t stream,
                                      const void* q, const void* k, const void* v,
                                      const void* dout, const float* lse,
                                      const float* delta, void* dqkv_q, float* dk_acc, float* dv_acc,
                                      const int32_t* cu_seqlens, int batch, int max_seqlen, int64_t T,
                                      int H, int Hkv, int D, int G,
                                      int64_t q_tstride, int64_t q_gstride,
                                      int64_t k_tstride, int64_t k_hstride,
                                      int64_t v_tstride, int64_t v_hstride,
                                      int64_t do_tstride,
                                      float scale, int dtype) {
    if (dtype != DOLOMITE_BF16) return 9010;
    if (D > 128) return 9011;
    int max_tiles = (max_seqlen + 127) / 128;
    hipStream_t s = (hipStream_t)stream;
#define CASE(DP)                                                                                        \
    return launch_fa_bwd<DP>(s, (const __bf16*)q, (const __bf16*)k, (const __bf16*)v,                   \
                             (const __bf16*)dout, lse, delta, (__bf16*)dqkv_q, dk_acc, dv_acc,          \
                             cu_seqlens, batch, T, H, Hkv, D, G, q_tstride, q_gstride, k_tstride,       \
                             k_hstride, v_tstride, v_hstride, do_tstride, max_tiles, scale)
    if (D <= 32) CASE(32);
    if (D <= 64) CASE(64);
    if (D <= 96) CASE(96);
    CASE(128);
#undef CASE
}

// ===========================================================================
// Backward pass 3: cast dq/dk/dv fp32 accumulators into the packed dqkv.
// ===========================================================================

// Reduce the (T, H, D) per-q-head fp32 partials over each kv head's G
// contributors in fixed order and cast into the packed dqkv. One thread
// per FOUR d-columns (D % 4 == 0 on this path): float4 streaming loads —
// the scalar version sat at 95% wave-wait on load latency.
template <typename T>
__global__ void __launch_bounds__(256) fa_grad_finalize_kernel(
    const float* __restrict__ dk_acc, const float* __restrict__ dv_acc,
    T* __restrict__ dqkv, int64_t total_kv,
    int Hkv, int D, int G, int64_t row_ts, int64_t k_off, int64_t kv_hs, int64_t v_off) {
    int64_t quad = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;  // 4-elem groups
    int64_t total_q4 = total_kv / 4;
    bool is_v = quad >= total_q4;
    if (is_v) quad -= total_q4;
    if (quad >= total_q4) return;
    int64_t kidx = quad * 4;
    int64_t t = kidx / ((int64_t)Hkv * D);
    int rem = (int)(kidx % ((int64_t)Hkv * D));
    int j = rem / D;
    int d = rem % D;
    const float* src = is_v ? dv_acc : dk_acc;
    int64_t base = ((t * Hkv + j) * (int64_t)G) * D + d;
    f4_t acc = {0.f, 0.f, 0.f, 0.f};
    for (int g = 0; g < G; ++g) {
        f4_t v4 = __builtin_nontemporal_load((const f4_t*)&src[base + (int64_t)g * D]);
        acc += v4;
    }
    T* out = &dqkv[t * row_ts + (is_v ? v_off : k_off) + (int64_t)j * kv_hs + d];
#pragma unroll
    for (int e = 0; e < 4; ++e) store_from_f32(out + e, acc[e]);
}

extern "C" int dolomite_fa_grad_finalize(dolomite_stream_t stream,
                                         const float* dk_acc, const float* dv_acc,
                                         void* dqkv, int64_t T, int Hkv, int D, int G,
                                         int64_t row_tstride,
                                         int64_t k_off, int64_t kv_hstride, int64_t v_off, int dtype) {
    int64_t total_kv = T * (int64_t)Hkv * D;
    if (D % 4 != 0) return 9012;  // D % 8 == 0 on every supported path
    int64_t total = 2 * (total_kv / 4);
    if (total == 0) return 0;
    dim3 grid((uint32_t)((total + 255) / 256)), block(256);
    if (dtype == DOLOMITE_BF16)
        hipLaunchKernelGGL((fa_grad_finalize_kernel<uint16_t>), grid, block, 0, (hipStream_t)stream,
                           dk_acc, dv_acc, (uint16_t*)dqkv, total_kv,
                           Hkv, D, G, row_tstride, k_off, kv_hstride, v_off);
    else
        hipLaunchKernelGGL((fa_grad_finalize_kernel<float>), grid, block, 0, (hipStream_t)stream,
                           dk_acc, dv_acc, (float*)dqkv, total_kv,
                           Hkv, D, G, row_tstride, k_off, kv_hstride, v_off);
    return dol_last_error();
}

// ===========================================================================
// MFMA fragment-layout self-test (see header comment).
// ===========================================================================

// ds_read_b64_tr_b16 semantics probe: LDS holds bf16 values = element index
// (0..255, exact in bf16). Each variant issues the transpose-read with a
// different per-lane address convention; the host inspects which elements
// landed in which lane/slot to pin the gather pattern (guide T10 says
// lane l elem j <- lds[(l&15) + j*16 + (l>>4)*64] but is not a spec).
typedef __bf16 bf16x4p __attribute__((ext_vector_type(4)));
__global__ void __launch_bounds__(64) tr16_probe_kernel(float* out) {
    __shared__ __bf16 s[256];
    const int lane = threadIdx.x & 63;
    for (int i = threadIdx.x; i < 256; i += 64) s[i] = (__bf16)(float)i;
    __syncthreads();
    const unsigned base = (unsigned)(size_t)(__bf16*)s;  // LDS byte address of s[0]
    unsigned addrs[3] = {
        base + (unsigned)((lane >> 4) * 128),                        // group base only
        base + (unsigned)(lane * 8),                                 // lane-linear b64
        base + (unsigned)(((lane & 15) * 2) + ((lane >> 4) * 128)),  // +column offset
    };
#pragma unroll
    for (int v = 0; v < 3; ++v) {
        bf16x4p r;
        asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
                     : "=v"(r)
                     : "v"(addrs[v])
                     : "memory");
#pragma unroll
        for (int j = 0; j < 4; ++j) out[(v * 64 + lane) * 4 + j] = (float)r[j];
    }
}

extern "C" int dolomite_tr16_probe(dolomite_stream_t stream, float* out) {
    hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream, out);
    return dol_last_error();
}

// End-to-end check of lds_tr16_bfrag under kernel conditions (dynamic LDS,
// PI23-staged row-major image): src is a row-major [64][cols] bf16 matrix;
// out[l][e] (float) = the B-fragment element e lane l received for the
// chunk (rowbase, d0). Host asserts out[l][e] == src[rowbase+(l>>4)*8+e][d0+(l&15)].
__global__ void __launch_bounds__(64) tr16_bfrag_probe_kernel(
    const __bf16* src, float* out, int cols, int S, int rowbase, int d0) {
    extern __shared__ char smem_raw[];
    __bf16* img = (__bf16*)smem_raw;
    const int lane = threadIdx.x & 63;
    for (int i = threadIdx.x; i < 64 * (cols / 8); i += 64) {
        int row = i / (cols / 8), c8 = (i % (cols / 8)) * 8;
        *(bf16x8*)&img[PI23(row) * S + c8] = *(const bf16x8*)&src[row * cols + c8];
    }
    __syncthreads();
    bf16x8 f = lds_tr16_bfrag(img, rowbase, S, d0, lane);
#pragma unroll
    for (int e = 0; e < 8; ++e) out[lane * 8 + e] = (float)f[e];
}

extern "C" int dolomite_tr16_bfrag_probe(dolomite_stream_t stream, const void* src, float* out,
                                         int cols, int S, int rowbase, int d0) {
    size_t shmem = (size_t)64 * S * sizeof(__bf16);
    hipLaunchKernelGGL(tr16_bfrag_probe_kernel, dim3(1), dim3(64), shmem, (hipStream_t)stream,
                       (const __bf16*)src, out, cols, S, rowbase, d0);
    return dol_last_error();
}

__global__ void __launch_bounds__(64) mfma_probe_kernel(const __bf16* A, const __bf16* B, float* C) {
    int lane = threadIdx.x & 63;
    int lr = lane & 15, lg = lane >> 4;
    bf16x8 a, b;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        a[e] = A[lr * 32 + lg * 8 + e];      // A[i][k], i=lr, k=lg*8+e
        b[e] = B[(lg * 8 + e) * 16 + lr];    // B[k][j], k=lg*8+e, j=lr
    }
    f32x4 c = {0.f, 0.f, 0.f, 0.f};
    c = MFMA16(a, b, c);
#pragma unroll
    for (int r = 0; r < 4; ++r) C[(lg * 4 + r) * 16 + lr] = c[r];
}

extern "C" int dolomite_mfma_probe(dolomite_stream_t stream, const void* A, const void* B, float* C) {
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, (hipStream_t)stream,
                       (const __bf16*)A, (const __bf16*)B, C);
    return dol_last_error();
}
