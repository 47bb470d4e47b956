// MI355X (gfx950 / CDNA4) kernels for the LSTM cell hot path.
//
// Replaces the reference's per-timestep TF graph ops:
//   k_lstm_rec_fwd : fused recurrent-GEMM + pointwise cell update [K1b+K2]
//                    gates = pre_t + h_{t-1} @ W_h  (MFMA bf16, fp32 accum)
//                    i,f,o = sigmoid, g = tanh; c_t = f*c + i*g;
//                    h_t = o*tanh(c_t)  -- one kernel per timestep.
//   k_lstm_pw_bwd  : fused pointwise backward [K4] -> raw-gate grads dgates
//   (recurrent dh GEMM [K5a] and the batched dW/dx GEMMs [K5b] go through
//    rocBLAS/hipBLASLt from the C++ sequence driver in bindings.cpp)
//   k_l2norm / k_clip_scale [K6], k_sgd / k_adam [K7]: optimizer path.
//
// Layout contract (must match lstm_tensorspark_amd/ops/functional.py):
//   * gate order i,f,g,o BLOCKED along the 4H axis;
//   * W_h passed TRANSPOSED as (4H, H) row-major so each lane's MFMA
//     B-fragment (8 k-contiguous bf16) is one contiguous 16-byte load;
//   * cell state fp32, h/activations bf16, MFMA accumulation fp32.
//
// Wavefront size is 64 (CDNA), block = 256 threads = 4 waves.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define DEVINL __device__ __forceinline__

// Chunk swizzle for 64-B LDS rows (4x 16-B chunks) that is bank-conflict-
// free for ds_read_b128's fixed lane groups ({0-3,12-15,20-27} etc.): the
// physical slot of k-chunk lk at row r is lk ^ M64[(r>>2)&3] with
// M64 = {0,3,2,1}.  Derivation: a lane group spans rows {c,c+4,c+8,c+12}
// x two k-groups; this permutation gives the 16 lanes 16 distinct
// (row%4, slot) bank positions.  XOR is involutive, so staging uses the
// same formula on the SOURCE chunk index.  (The row&3 / row&7 XOR used for
// 128-B rows does NOT transfer to 64-B rows: rows 12 apart collide and
// k-groups 0/2 collapse -- measured 157M bank-conflict cycles.)
DEVINL int swz64(int row) { return (0x1230 >> (((row >> 2) & 3) * 4)) & 3; }

// Fast activations: v_exp + v_rcp (1-ulp) -- bf16 outputs cannot see the
// difference vs libcalls, and __ocml_tanh_f32 is a branchy libcall.
DEVINL float rcp_(float x) { return __builtin_amdgcn_rcpf(x); }
DEVINL float sigmoidf_(float x) {
    x = fminf(fmaxf(x, -30.f), 30.f);
    return rcp_(1.0f + __expf(-x));
}
DEVINL float tanhf_(float x) {
    x = fminf(fmaxf(x, -15.f), 15.f);
    const float e = __expf(2.0f * x);
    return (e - 1.0f) * rcp_(e + 1.0f);
}

// ---------------------------------------------------------------------------
// K1b+K2: fused recurrent GEMM + LSTM pointwise, one timestep.
//
// Grid: (ceil(B/BM), H/64).  Block: 256 threads = 4 waves; wave w owns
// h-columns [blockIdx.y*64 + w*16, +16) and computes FOUR 16-col MFMA
// accumulators (one per gate) over BM rows, so the cell update needs no
// cross-lane traffic: after the K-loop, lane (l) holds, for each of its
// 4 fragment rows, the i/f/g/o raw-gate values of ONE (row, h-column).
//
// MFMA: v_mfma_f32_16x16x32_bf16.  A-fragment: lane l = h_prev[row=l&15]
// [k=(l>>4)*8 ..+8]; B-fragment: w_t[n=l&15][same k] with w_t=(4H,H)=W_h^T;
// C/D: col=l&15, row=(l>>4)*4+r.
// TAIL HANDLING: rows (batch) bounds-checked; requires H % 64 == 0.
//
// Variants (within-probe A/B sweep, tools/kernel_bench.py):
//   STAGE_A: stage the h tile in LDS once per BK=64 K-step (the 4 waves
//            otherwise each load identical A fragments = 4x redundant L2
//            traffic).  LDS rows padded +8 elems so the ds_read_b128
//            fragment reads are bank-conflict-free (16 lanes' start banks
//            4*(9r mod 16) all distinct).
//   XCD_SWZ: bijective blockIdx remap so blocks sharing a W_h panel
//            (same n-block) land on the same XCD's private L2 (T1).
// ---------------------------------------------------------------------------
template <int BM, bool RAW_DEBUG, int STAGE, bool XCD_SWZ>
__global__ __launch_bounds__(256) void k_lstm_rec_fwd(
    const __hip_bfloat16* __restrict__ h_prev, long hp_stride,
    const __hip_bfloat16* __restrict__ w_t,     // (4H, H) row-major = W_h^T
    const __hip_bfloat16* __restrict__ pre, long pre_stride,   // (B, 4H)
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ h_out, long ho_stride,
    float* __restrict__ c_out, long co_stride,
    __hip_bfloat16* __restrict__ acts, long act_stride,        // (B, 4H)
    int B, int H)
{
    constexpr int MF = BM / 16;
    const int tid  = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr   = lane & 15;   // fragment row (A) / col (B, C/D)
    const int lk   = lane >> 4;   // k-group 0..3

    int m_blk = blockIdx.x, n_blk = blockIdx.y;
    if (XCD_SWZ) {
        const int nm = gridDim.x, nwg = gridDim.x * gridDim.y;
        const int bid = blockIdx.x + blockIdx.y * nm;
        const int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, loc = bid >> 3;
        const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                        + loc;
        m_blk = swz % nm;      // m fastest: one XCD covers few W panels
        n_blk = swz / nm;
    }
    const int m_base = m_blk * BM;
    const int n_base = n_blk * 64 + wave * 16;  // h-column base

    f32x4 acc[4][MF];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int mf = 0; mf < MF; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

    const long koff0 = (long)lk * 8;

    if constexpr (STAGE == 2) {
        // All-glds 2-phase: BOTH tiles staged by global_load_lds into a
        // double-buffered dynamic-LDS image (no VGPR staging, no register
        // double-buffer copies -- the fix for the accvgpr-shuffle stall).
        // Images are lane-linear (glds writes base + lane*16); bank
        // conflicts on the b128 fragment reads are broken by XOR-ing the
        // 16B-chunk index with (row & 7) on the SOURCE address (rule 21).
        constexpr int BK = 64;
        constexpr int ASLICES = BM / 32;       // A tile = BM x 64 x 2B
        constexpr int ABYTES = BM * 128;
        extern __shared__ __attribute__((aligned(16))) char smem[];
        char* const a_base = smem;                       // [2][ABYTES]
        char* const b_base = smem + 2 * ABYTES;          // [2][32768]
        const int nsteps = H / BK;
        const long n_col0 = (long)n_blk * 64;

        auto stage = [&](int buf, int k0) {
#pragma unroll
            for (int s2 = 0; s2 < ASLICES; ++s2) {
                const int ci = s2 * 256 + tid;
                const int row = ci >> 3, ch = ci & 7;
                int grow = m_base + row;
                grow = grow < B ? grow : B - 1;  // clamp: extra rows unused
                const void* src = h_prev + (long)grow * hp_stride + k0
                                  + ((ch ^ (row & 7)) * 8);
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        a_base + buf * ABYTES + s2 * 4096 + wave * 1024),
                    16, 0, 0);
            }
#pragma unroll
            for (int s2 = 0; s2 < 8; ++s2) {
                const int ci = s2 * 256 + tid;
                const int gc = ci >> 3, ch = ci & 7;
                const long grow = (long)(gc >> 6) * H + n_col0 + (gc & 63);
                const void* src = w_t + grow * H + k0 + ((ch ^ (gc & 7)) * 8);
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        b_base + buf * 32768 + s2 * 4096 + wave * 1024),
                    16, 0, 0);
            }
        };

        stage(0, 0);
        __syncthreads();  // carries the vmcnt(0) glds drain
        for (int j = 0; j < nsteps; ++j) {
            const int buf = j & 1;
            if (j + 1 < nsteps) stage(buf ^ 1, (j + 1) * BK);  // issue early
            const char* ab = a_base + buf * ABYTES;
            const char* bb = b_base + buf * 32768;
#pragma unroll
            for (int s = 0; s < 2; ++s) {
                bf16x8 a[MF], bfr[4];
#pragma unroll
                for (int mf = 0; mf < MF; ++mf) {
                    const int row = mf * 16 + lr;
                    a[mf] = *(const bf16x8*)(
                        ab + row * 128 + (((s * 4 + lk) ^ (row & 7)) * 16));
                }
#pragma unroll
                for (int g = 0; g < 4; ++g) {
                    const int gc = g * 64 + wave * 16 + lr;
                    bfr[g] = *(const bf16x8*)(
                        bb + gc * 128 + (((s * 4 + lk) ^ (gc & 7)) * 16));
                }
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int mf = 0; mf < MF; ++mf)
                        acc[g][mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a[mf], bfr[g], acc[g][mf], 0, 0, 0);
            }
            __syncthreads();  // drains next tile's glds + protects buf reuse
        }
    } else if constexpr (STAGE == 3) {
        // Counted-vmcnt 3-buffer glds pipeline (T3+T4): BK=32 tiles, two
        // tiles in flight across RAW barriers (never vmcnt(0) in the main
        // loop), 60 KiB LDS at BM=64 -> 2 blocks/CU.  Per-wave glds per
        // tile: A = BM/64, B = 4.
        constexpr int BK = 32;
        constexpr int ABYTES = BM * BK * 2;          // 4 KiB at BM=64
        constexpr int BBYTES = 256 * BK * 2;         // 16 KiB
        constexpr int ASLICES = ABYTES / 4096;       // glds instrs (>=1)
        constexpr int TILE_GLDS = (ASLICES < 1 ? 1 : ASLICES) + 4;
        extern __shared__ __attribute__((aligned(16))) char smem[];
        const int nsteps = H / BK;
        const long n_col0 = (long)n_blk * 64;

        auto stage = [&](int buf, int k0) {
            char* const ab = smem + buf * (ABYTES + BBYTES);
            char* const bb = ab + ABYTES;
#pragma unroll
            for (int s2 = 0; s2 < (ASLICES < 1 ? 1 : ASLICES); ++s2) {
                const int ci = s2 * 256 + tid;
                const int row = ci >> 2, ch = ci & 3;   // 4 chunks of 16B per 64-byte row
                int grow = m_base + row;
                grow = grow < B ? grow : B - 1;
                const void* src = h_prev + (long)grow * hp_stride + k0
                                  + ((ch ^ swz64(row)) * 8);
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        ab + s2 * 4096 + wave * 1024), 16, 0, 0);
            }
#pragma unroll
            for (int s2 = 0; s2 < 4; ++s2) {
                const int ci = s2 * 256 + tid;
                const int gc = ci >> 2, ch = ci & 3;
                const long grow = (long)(gc >> 6) * H + n_col0 + (gc & 63);
                const void* src = w_t + grow * H + k0 + ((ch ^ swz64(gc)) * 8);
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        bb + s2 * 4096 + wave * 1024), 16, 0, 0);
            }
        };

        stage(0, 0);
        if (nsteps > 1) stage(1, BK);
        for (int j = 0; j < nsteps; ++j) {
            const int buf = j % 3;
            // tile j landed when at most the NEXT tile's glds are pending
            if (j + 1 < nsteps)
                asm volatile("s_waitcnt vmcnt(%0)" :: "i"(TILE_GLDS) : "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
            // issue tile j+2 into the third buffer (not in use by j, j+1)
            if (j + 2 < nsteps) stage((j + 2) % 3, (j + 2) * BK);
            const char* ab = smem + buf * (ABYTES + BBYTES);
            const char* bb = ab + ABYTES;
            bf16x8 a[MF], bfr[4];
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
                const int row = mf * 16 + lr;
                a[mf] = *(const bf16x8*)(ab + row * 64 + ((lk ^ swz64(row)) * 16));
            }
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const int gc = g * 64 + wave * 16 + lr;
                bfr[g] = *(const bf16x8*)(bb + gc * 64 + ((lk ^ swz64(gc)) * 16));
            }
#pragma unroll
            for (int g = 0; g < 4; ++g)
#pragma unroll
                for (int mf = 0; mf < MF; ++mf)
                    acc[g][mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[mf], bfr[g], acc[g][mf], 0, 0, 0);
        }
    } else if constexpr (STAGE == 1) {
        constexpr int LDA = 72;  // 64 + 8 pad: conflict-free b128 reads
        __shared__ __hip_bfloat16 a_lds[2 * BM][LDA];  // double-buffered
        // BK=64 K-steps, double-buffered LDS A tile, ONE barrier per step.
        // A and B are both register-prefetched one K-step ahead so the
        // global-load latency hides under the previous step's MFMAs (T14
        // issue-early / write-late).
        constexpr int BK = 64;
        constexpr int SLICES = BM * BK / 8 / 256;  // 16B loads per thread
        static_assert(SLICES >= 1, "block too small for tile");
        const int ar = tid >> 3, ac = (tid & 7) * 8;  // staging coords
        const int nsteps = H / BK;

        auto load_a = [&](int k0, bf16x8 (&a_reg)[SLICES]) {
#pragma unroll
            for (int s = 0; s < SLICES; ++s) {
                const int row = m_base + ar + s * (256 / 8);
                a_reg[s] = (row < B)
                    ? *(const bf16x8*)(h_prev + (long)row * hp_stride + k0 + ac)
                    : bf16x8{};
            }
        };
        auto store_a = [&](int buf, const bf16x8 (&a_reg)[SLICES]) {
#pragma unroll
            for (int s = 0; s < SLICES; ++s)
                *(bf16x8*)(&a_lds[buf * BM + ar + s * (256 / 8)][ac]) = a_reg[s];
        };
        auto load_b = [&](int k0, bf16x8 (&b_reg)[4][2]) {
#pragma unroll
            for (int g = 0; g < 4; ++g)
#pragma unroll
                for (int s = 0; s < 2; ++s)
                    b_reg[g][s] = *(const bf16x8*)(
                        w_t + ((long)g * H + n_base + lr) * H + k0 + s * 32 + koff0);
        };

        bf16x8 a_reg[SLICES], b_cur[4][2], b_nxt[4][2];
        load_a(0, a_reg);
        load_b(0, b_cur);
        store_a(0, a_reg);
        if (nsteps > 1) {
            load_a(BK, a_reg);   // tile 1 -> a_reg (stored at bottom of j=0)
            load_b(BK, b_nxt);
        }
        __syncthreads();
        for (int j = 0; j < nsteps; ++j) {
            const int buf = j & 1;
#pragma unroll
            for (int s = 0; s < 2; ++s) {
                bf16x8 a[MF];
#pragma unroll
                for (int mf = 0; mf < MF; ++mf)
                    a[mf] = *(const bf16x8*)(
                        &a_lds[buf * BM + mf * 16 + lr][s * 32 + lk * 8]);
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int mf = 0; mf < MF; ++mf)
                        acc[g][mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a[mf], b_cur[g][s], acc[g][mf], 0, 0, 0);
            }
            if (j + 1 < nsteps) {
                // write tile j+1 (loads issued one MFMA phase ago), then
                // issue tile j+2's loads to hide under the NEXT MFMA phase.
                store_a(buf ^ 1, a_reg);
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int s = 0; s < 2; ++s) b_cur[g][s] = b_nxt[g][s];
                if (j + 2 < nsteps) {
                    load_a((j + 2) * BK, a_reg);
                    load_b((j + 2) * BK, b_nxt);
                }
                __syncthreads();
            }
        }
    } else {
        for (int k0 = 0; k0 < H; k0 += 32) {
            bf16x8 a[MF];
#pragma unroll
            for (int mf = 0; mf < MF; ++mf) {
                const int row = m_base + mf * 16 + lr;
                a[mf] = (row < B)
                    ? *(const bf16x8*)(h_prev + (long)row * hp_stride + k0 + koff0)
                    : bf16x8{};
            }
            bf16x8 bfr[4];
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const long n = (long)g * H + n_base + lr;
                bfr[g] = *(const bf16x8*)(w_t + n * H + k0 + koff0);
            }
#pragma unroll
            for (int g = 0; g < 4; ++g)
#pragma unroll
                for (int mf = 0; mf < MF; ++mf)
                    acc[g][mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[mf], bfr[g], acc[g][mf], 0, 0, 0);
        }
    }

    // Epilogue: fused bias/pre add + activations + cell update.
    const int hcol = n_base + lr;
#pragma unroll
    for (int mf = 0; mf < MF; ++mf) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int row = m_base + mf * 16 + lk * 4 + r;
            if (row >= B) continue;
            float raw[4];
#pragma unroll
            for (int g = 0; g < 4; ++g)
                raw[g] = acc[g][mf][r] +
                         __bfloat162float(pre[(long)row * pre_stride + (long)g * H + hcol]);
            if (RAW_DEBUG) {
                // debug/numerics mode: write raw gate pre-activations only
#pragma unroll
                for (int g = 0; g < 4; ++g)
                    acts[(long)row * act_stride + (long)g * H + hcol] =
                        __float2bfloat16(raw[g]);
                continue;
            }
            const float gi = sigmoidf_(raw[0]);
            const float gf = sigmoidf_(raw[1]);
            const float gg = tanhf_(raw[2]);
            const float go = sigmoidf_(raw[3]);
            const float cp = c_prev[(long)row * cp_stride + hcol];
            const float c  = gf * cp + gi * gg;
            const float h  = go * tanhf_(c);
            h_out[(long)row * ho_stride + hcol] = __float2bfloat16(h);
            c_out[(long)row * co_stride + hcol] = c;
            __hip_bfloat16* ap = acts + (long)row * act_stride + hcol;
            ap[0]           = __float2bfloat16(gi);
            ap[(long)H]     = __float2bfloat16(gf);
            ap[(long)2 * H] = __float2bfloat16(gg);
            ap[(long)3 * H] = __float2bfloat16(go);
        }
    }
}

// ---------------------------------------------------------------------------
// MT256 fused recurrent GEMM + cell update: the 256x256-tile structure of
// the verified 8-phase GEMM template (guide §5 "256² 8-phase"), adapted to
// the LSTM epilogue:
//   - block = 512 threads = 8 waves laid out 2M x 4N.  Wave (wm, wn)
//     computes rows [wm*128, +128) x h-columns [n_blk*64 + wn*16, +16) for
//     ALL FOUR gates (gate-interleaved N): its 64 output gate-columns are
//     {g*H + hcol} for g=0..3, so the cell update stays lane-local exactly
//     as in k_lstm_rec_fwd.
//   - BK=32 K-tiles (one mfma_16x16x32 depth), THREE 32 KiB LDS buffers
//     (A 256x32 + B 256x32 bf16), glds-staged two tiles ahead with counted
//     vmcnt and raw s_barrier (never vmcnt(0) in the main loop) -- the
//     STAGE=3 discipline at 4 glds/tile/wave.
//   - XOR-chunk source swizzle (ch ^ (row & 3) on 64-B rows) keeps the
//     ds_read_b128 fragment reads bank-conflict-free.
//   - 96 KiB LDS -> 1 block/CU, 2 waves/SIMD; acc = 4 gates x 8 m-frags
//     (128 VGPRs).  grid (B/256, H/64) = 256 blocks at the headline shape:
//     exactly one block per CU.
// Per-output operand traffic is (256+256)/(256*256) = 2.5x less than the
// BM64 variants; measured MFMA:LDS cycle ratio leaves LDS at ~37% load.
// Requires B % 256 == 0 (launcher falls back to the BM64 path otherwise).
// ---------------------------------------------------------------------------
// ABL (ablation, µbench only): 0 normal; 1 = no staging/vmcnt (reads one
// pre-staged buffer every tile -- WRONG results, isolates compute+barrier
// cost); 2 = staging but no MFMA (isolates the glds/vmcnt pipeline cost).
// STYLE 0 = 3-buffer BK=32 two-phase counted-vmcnt; STYLE 1 = 2-buffer
// BK=64, one vmcnt(0)-draining __syncthreads per K-tile (guide: ties the
// best register pipeline on the 256^2 tile at 1/4 the LDS instructions).
template <bool RAW_DEBUG, bool XCD_SWZ, int ABL = 0, int STYLE = 0>
__global__ __launch_bounds__(512) void k_lstm_rec_fwd_mt256(
    const __hip_bfloat16* __restrict__ h_prev, long hp_stride,
    const __hip_bfloat16* __restrict__ w_t,     // (4H, H) row-major = W_h^T
    const __hip_bfloat16* __restrict__ pre, long pre_stride,   // (B, 4H)
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ h_out, long ho_stride,
    float* __restrict__ c_out, long co_stride,
    __hip_bfloat16* __restrict__ acts, long act_stride,
    int B, int H)
{
    constexpr int BK = 32;
    constexpr int ABYTES = 256 * BK * 2;   // 16 KiB
    constexpr int TILE_BYTES = 2 * ABYTES; // A + B = 32 KiB
    constexpr int TILE_GLDS = 4;           // 2 A + 2 B glds per tile per wave
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr = lane & 15;   // fragment row (A) / col (B, C/D)
    const int lk = lane >> 4;   // k-group 0..3
    const int wm = wave & 1;    // M half: rows [wm*128, +128)
    const int wn = wave >> 1;   // N group: h-cols [wn*16, +16)

    int m_blk = blockIdx.x, n_blk = blockIdx.y;
    if (XCD_SWZ) {
        const int nm = gridDim.x, nwg = gridDim.x * gridDim.y;
        const int bid = blockIdx.x + blockIdx.y * nm;
        const int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, loc = bid >> 3;
        const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                        + loc;
        m_blk = swz % nm;
        n_blk = swz / nm;
    }
    const int m_base = m_blk * 256;
    const long n_col0 = (long)n_blk * 64;

    // stage K-tile k0 into buffer buf: A rows 256 x BK, B gate-cols 256 x BK,
    // 64-B rows, XOR-chunk source swizzle so fragment reads are conflict-free
    auto stage_a = [&](int buf, int k0) {
        char* const ab = smem + buf * TILE_BYTES;
#pragma unroll
        for (int s2 = 0; s2 < 2; ++s2) {
            const int ci = s2 * 512 + tid;
            const int row = ci >> 2, ch = ci & 3;
            const void* src = h_prev + (long)(m_base + row) * hp_stride + k0
                              + ((ch ^ swz64(row)) * 8);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    ab + s2 * 8192 + wave * 1024), 16, 0, 0);
        }
    };
    auto stage_b = [&](int buf, int k0) {
        char* const bb = smem + buf * TILE_BYTES + ABYTES;
#pragma unroll
        for (int s2 = 0; s2 < 2; ++s2) {
            const int ci = s2 * 512 + tid;
            const int gc = ci >> 2, ch = ci & 3;
            const long grow = (long)(gc >> 6) * H + n_col0 + (gc & 63);
            const void* src = w_t + grow * H + k0 + ((ch ^ swz64(gc)) * 8);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    bb + s2 * 8192 + wave * 1024), 16, 0, 0);
        }
    };
    auto stage = [&](int buf, int k0) { stage_a(buf, k0); stage_b(buf, k0); };

    f32x4 acc[4][8];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int mf = 0; mf < 8; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

    if (STYLE == 1) {
        // 2-buffer BK=64: images are [256][128] bytes with the proven
        // ch^(row&7) swizzle; ONE __syncthreads per K-tile carries the
        // vmcnt(0) glds drain (STAGE2 discipline at MT256 scale).
        constexpr int BK64 = 64;
        constexpr int A64 = 256 * BK64 * 2;        // 32 KiB
        constexpr int T64 = 2 * A64;               // A+B = 64 KiB
        auto stage64 = [&](int buf, int k0) {
            char* const ab = smem + buf * T64;
            char* const bb = ab + A64;
#pragma unroll
            for (int s2 = 0; s2 < 4; ++s2) {
                const int ci = s2 * 512 + tid;
                const int row = ci >> 3, ch = ci & 7;
                const void* src = h_prev + (long)(m_base + row) * hp_stride
                                  + k0 + ((ch ^ (row & 7)) * 8);
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        ab + s2 * 8192 + wave * 1024), 16, 0, 0);
            }
#pragma unroll
            for (int s2 = 0; s2 < 4; ++s2) {
                const int ci = s2 * 512 + tid;
                const int gc = ci >> 3, ch = ci & 7;
                const long grow = (long)(gc >> 6) * H + n_col0 + (gc & 63);
                const void* src = w_t + grow * H + k0 + ((ch ^ (gc & 7)) * 8);
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        bb + s2 * 8192 + wave * 1024), 16, 0, 0);
            }
        };
        const int nst = H / BK64;
        stage64(0, 0);
        __syncthreads();   // drains the glds (vmcnt0) + releases reads
        for (int j = 0; j < nst; ++j) {
            const int buf = j & 1;
            if (j + 1 < nst) stage64(buf ^ 1, (j + 1) * BK64);
            const char* ab = smem + buf * T64;
            const char* bb = ab + A64;
#pragma unroll
            for (int kc = 0; kc < 2; ++kc) {
                bf16x8 bfr[4], a[8];
#pragma unroll
                for (int g = 0; g < 4; ++g) {
                    const int gc = g * 64 + wn * 16 + lr;
                    bfr[g] = *(const bf16x8*)(
                        bb + gc * 128 + (((kc * 4 + lk) ^ (gc & 7)) * 16));
                }
#pragma unroll
                for (int mi = 0; mi < 8; ++mi) {
                    const int row = wm * 128 + mi * 16 + lr;
                    a[mi] = *(const bf16x8*)(
                        ab + row * 128 + (((kc * 4 + lk) ^ (row & 7)) * 16));
                }
                __builtin_amdgcn_s_setprio(1);
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int mi = 0; mi < 8; ++mi)
                        acc[g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            bfr[g], a[mi], acc[g][mi], 0, 0, 0);
                __builtin_amdgcn_s_setprio(0);
            }
            __syncthreads();  // next tile's glds drained + buf reuse safe
        }
        goto epilogue;
    }
    {
    const int nsteps = H / BK;
    stage(0, 0);
    if (ABL != 1 && nsteps > 1) stage(1, BK);
    // Phase discipline (guide §5 8-phase template, T3+T4): per phase
    //   {ds_read subtile; 2 glds; s_barrier; lgkmcnt(0); setprio(1);
    //    16 MFMA; setprio(0); s_barrier}
    // with vmcnt counted ONCE per K-tile (never 0 in the main loop).
    // CRITICAL ORDER: vmcnt is PER-WAVE, but fragments are staged by ALL
    // waves' glds -- so every tile-landing wait must be followed by an
    // s_barrier BEFORE any wave reads that tile's LDS.  The waits therefore
    // sit at the TAIL of the previous tile (just before its trailing
    // barrier), counting only the tile staged in that same iteration.
    if (nsteps > 1 && ABL != 1)
        asm volatile("s_waitcnt vmcnt(%0)" :: "i"(TILE_GLDS) : "memory");
    else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    for (int j = 0; j < nsteps; ++j) {
        const int buf = (ABL == 1) ? 0 : (j % 3);
        const char* ab = smem + buf * TILE_BYTES;
        const char* bb = ab + ABYTES;
        const bool pf = (ABL != 1) && (j + 2 < nsteps);
        const int pbuf = (j + 2) % 3, pk0 = (j + 2) * BK;
        // ---- phase A: B frags + A frags 0-3; stage tile j+2's A half
        bf16x8 bfr[4], a[4];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            const int gc = g * 64 + wn * 16 + lr;
            bfr[g] = *(const bf16x8*)(bb + gc * 64 + ((lk ^ swz64(gc)) * 16));
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
            const int row = wm * 128 + mi * 16 + lr;
            a[mi] = *(const bf16x8*)(ab + row * 64 + ((lk ^ swz64(row)) * 16));
        }
        if (pf) stage_a(pbuf, pk0);
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_setprio(1);
        // OPERAND ORDER: A = weight fragment, B = hidden fragment, so the
        // D layout is (gate-col, batch): lane (lk,lr) holds 4 CONSECUTIVE
        // h-columns (lk*4+j) of one batch row (lr) -> the epilogue's
        // pre/acts/h/c accesses are bf16x4/f32x4, not scalar bf16 (the
        // scalar-epilogue version measured 50 of its 71 us in the epilogue).
        if (ABL != 2)
#pragma unroll
            for (int g = 0; g < 4; ++g)
#pragma unroll
                for (int mi = 0; mi < 4; ++mi)
                    acc[g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        bfr[g], a[mi], acc[g][mi], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        __builtin_amdgcn_s_barrier();
        // ---- phase B: A frags 4-7; stage tile j+2's B half
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
            const int row = wm * 128 + (4 + mi) * 16 + lr;
            a[mi] = *(const bf16x8*)(ab + row * 64 + ((lk ^ swz64(row)) * 16));
        }
        if (pf) stage_b(pbuf, pk0);
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_setprio(1);
        if (ABL != 2)
#pragma unroll
            for (int g = 0; g < 4; ++g)
#pragma unroll
                for (int mi = 0; mi < 4; ++mi)
                    acc[g][4 + mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        bfr[g], a[mi], acc[g][4 + mi], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        // tail wait: tile j+1 must be landed before the next iteration's
        // ds_reads; only tile j+2's glds (issued THIS iteration) may fly.
        if (ABL != 1 && j + 1 < nsteps) {
            if (pf)
                asm volatile("s_waitcnt vmcnt(%0)" :: "i"(TILE_GLDS) : "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
    }
    }
epilogue:

    // Fused epilogue, fully vectorized: lane owns h-columns
    // [hcol0, hcol0+4) of batch rows (wm*128 + nf*16 + lr) -- every global
    // access is bf16x4 (8 B) or f32x4 (16 B).
    const int hcol0 = (int)n_col0 + wn * 16 + lk * 4;
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
        const long row = m_base + wm * 128 + nf * 16 + lr;
        f32x4 raw[4];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            const bf16x4 pv = *(const bf16x4*)(
                pre + row * pre_stride + (long)g * H + hcol0);
#pragma unroll
            for (int j = 0; j < 4; ++j)
                raw[g][j] = acc[g][nf][j] + (float)pv[j];
        }
        __hip_bfloat16* ap = acts + row * act_stride + hcol0;
        if (RAW_DEBUG) {
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                bf16x4 o;
#pragma unroll
                for (int j = 0; j < 4; ++j) o[j] = (__bf16)raw[g][j];
                *(bf16x4*)(ap + (long)g * H) = o;
            }
            continue;
        }
        const f32x4 cp = *(const f32x4*)(c_prev + row * cp_stride + hcol0);
        bf16x4 oi, of, og, oo, oh;
        f32x4 oc;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const float gi = sigmoidf_(raw[0][j]);
            const float gf = sigmoidf_(raw[1][j]);
            const float gg = tanhf_(raw[2][j]);
            const float go = sigmoidf_(raw[3][j]);
            const float c  = gf * cp[j] + gi * gg;
            oc[j] = c;
            oh[j] = (__bf16)(go * tanhf_(c));
            oi[j] = (__bf16)gi;
            of[j] = (__bf16)gf;
            og[j] = (__bf16)gg;
            oo[j] = (__bf16)go;
        }
        *(bf16x4*)(h_out + row * ho_stride + hcol0) = oh;
        *(f32x4*)(c_out + row * co_stride + hcol0) = oc;
        *(bf16x4*)(ap)               = oi;
        *(bf16x4*)(ap + (long)H)     = of;
        *(bf16x4*)(ap + (long)2 * H) = og;
        *(bf16x4*)(ap + (long)3 * H) = oo;
    }
}

// ---------------------------------------------------------------------------
// 8-PHASE 256x256 fused recurrent GEMM + cell update (guide §5 "256² 8-phase
// template", the 1.3 PF/s-class structure, adapted to the LSTM epilogue).
// Differences from k_lstm_rec_fwd_mt256 (the ~500 TF/s round-1 kernel):
//   - BK=64 K-tiles (stored as TWO BK=32 sub-images so the PROVEN
//     conflict-free swz64 64-B-row chunk swizzle applies unchanged),
//     2 LDS buffers x (A 32 KiB + B 32 KiB) = 128 KiB;
//   - 4 phases per K-tile, each = {ds_read quadrant frags; glds-stage ONE
//     half-tile; s_barrier; lgkmcnt(0); setprio(1); 16 MFMA; setprio(0);
//     s_barrier} -- the fine ds_read || glds || MFMA interleave that the
//     guide measures as THE lever (coarse phase-split = -7..-27%);
//   - counted vmcnt ONCE per K-tile (never 0 in the main loop): the stage
//     schedule {P1: A-lo(j+1), P2: B-lo(j+2), P3: A-hi(j+2), P4: B-hi(j+2)}
//     leaves exactly 3 half-tiles (vmcnt(6)) in flight at each boundary and
//     guarantees ALL of tile j+1 landed there:
//       A-lo(j+1) staged at phase 4j (this tile's P1)    <= boundary-3
//       B-lo(j+1)/A-hi(j+1)/B-hi(j+1) staged in tile j-1 <= boundary-4
//     (slot-freeness: a half's LDS slot is reused exactly 2 tiles later,
//      and its last reader phase precedes the re-stage phase: B-lo read
//      only in P1, A-* in P1/P2, B-hi in P3 -- frags then live in VGPRs.)
//   - per-wave output unchanged: 128 batch rows x 4 gates x 16 h-cols,
//     gate-interleaved so the fused cell update stays lane-local; the
//     epilogue is byte-identical to k_lstm_rec_fwd_mt256's.
// Requires B % 256 == 0, H % 64 == 0.  Variant bit 512 in the launcher.
// ---------------------------------------------------------------------------
// ABL bitflags (µbench only): 1 = no staging/vmcnt in the main loop (reads
// stale buffer 0 -- WRONG results; isolates compute+barrier+read cost),
// 2 = staging but no MFMA (isolates the glds/vmcnt pipeline cost),
// 4 = no setprio around the MFMA clusters.
template <bool RAW_DEBUG, bool XCD_SWZ, int ABL = 0>
__global__ __launch_bounds__(512, 2) void k_lstm_rec_fwd_8p(
    const __hip_bfloat16* __restrict__ h_prev, long hp_stride,
    const __hip_bfloat16* __restrict__ w_t,     // (4H, H) row-major = W_h^T
    const __hip_bfloat16* __restrict__ pre, long pre_stride,   // (B, 4H)
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ h_out, long ho_stride,
    float* __restrict__ c_out, long co_stride,
    __hip_bfloat16* __restrict__ acts, long act_stride,
    int B, int H)
{
    constexpr int BK = 64;          // K-tile depth (2 x 32-deep sub-images)
    constexpr int OPBYTES = 32768;  // one operand tile: 2 sub-images x 16 KiB
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr = lane & 15;   // fragment row (A) / col (B, C/D)
    const int lk = lane >> 4;   // k-group 0..3
    const int wm = wave & 1;    // M half: rows [wm*128, +128)
    const int wn = wave >> 1;   // N group: h-cols [wn*16, +16)

    int m_blk = blockIdx.x, n_blk = blockIdx.y;
    if (XCD_SWZ) {
        const int nm = gridDim.x, nwg = gridDim.x * gridDim.y;
        const int bid = blockIdx.x + blockIdx.y * nm;
        const int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, loc = bid >> 3;
        const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                        + loc;
        m_blk = swz % nm;
        n_blk = swz / nm;
    }
    const int m_base = m_blk * 256;
    const long n_col0 = (long)n_blk * 64;
    const int nsteps = H / BK;

    // Stage ONE half-tile (128 rows x 64 k) of operand A (h) or B (W^T)
    // into buffer `buf` at K offset k0: 2 glds per thread (one per 32-deep
    // sub-image).  LDS image per operand: [kb][row][chunk] with 64-B rows
    // and the swz64 chunk permutation applied on the SOURCE address
    // (glds writes lane-linear; reads un-swizzle -- rule 21).
    auto stage_half = [&](int is_b, int hf, int buf, int k0) {
        char* const base = smem + buf * (2 * OPBYTES) + is_b * OPBYTES;
        const int r = hf * 128 + (tid >> 2);
        const int ch = tid & 3;
        const int chs = (ch ^ swz64(r)) * 8;
#pragma unroll
        for (int kb = 0; kb < 2; ++kb) {
            const void* src;
            if (!is_b) {
                src = h_prev + (long)(m_base + r) * hp_stride + (k0 + kb * 32)
                      + chs;
            } else {
                const long grow = (long)(r >> 6) * H + n_col0 + (r & 63);
                src = w_t + grow * H + (k0 + kb * 32) + chs;
            }
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    base + kb * 16384 + hf * 8192 + wave * 1024), 16, 0, 0);
        }
    };

    f32x4 acc[4][8];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int mf = 0; mf < 8; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

    // ---- prologue: tile 0 fully + {B-lo, A-hi, B-hi} of tile 1, counted
    // drains so tile 0 is landed and exactly 3 half-tiles stay in flight.
    stage_half(0, 0, 0, 0);              // A-lo(0)
    stage_half(1, 0, 0, 0);              // B-lo(0)
    stage_half(0, 1, 0, 0);              // A-hi(0)
    stage_half(1, 1, 0, 0);              // B-hi(0)
    if (nsteps > 1 && !(ABL & 1)) {
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");  // pace: 2 landed
        stage_half(1, 0, 1, BK);         // B-lo(1)
        stage_half(0, 1, 1, BK);         // A-hi(1)
        stage_half(1, 1, 1, BK);         // B-hi(1)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");  // tile 0 landed
    } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    if (!(ABL & 8)) __builtin_amdgcn_s_barrier();

    bf16x8 wfr[2][2];    // gate pair frags (g within pair, kb) -- reused
    bf16x8 hlo[4][2];    // wave's m-half rows 0-63  (mi, kb)
    bf16x8 hhi[4][2];    // wave's m-half rows 64-127

    for (int j = 0; j < nsteps; ++j) {
        const int buf = (ABL & 1) ? 0 : (j & 1);
        const char* ab = smem + buf * (2 * OPBYTES);
        const char* bb = ab + OPBYTES;
        // ---- P1: W gates 0-1 + h mi 0-3 (12 ds_reads); stage A-lo(j+1)
#pragma unroll
        for (int g = 0; g < 2; ++g) {
            const int gc = g * 64 + wn * 16 + lr;
            const int co = (lk ^ swz64(gc)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                wfr[g][kb] = *(const bf16x8*)(bb + kb * 16384 + gc * 64 + co);
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
            const int row = wm * 128 + mi * 16 + lr;
            const int co = (lk ^ swz64(row)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                hlo[mi][kb] = *(const bf16x8*)(ab + kb * 16384 + row * 64 + co);
        }
        if (!(ABL & 1) && j + 1 < nsteps) stage_half(0, 0, (j + 1) & 1, (j + 1) * BK);
        asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(1);
if constexpr (!(ABL & 2)) {
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hlo[mi][kb], acc[g][mi], 0, 0, 0);
        } else {
            // keep the phase's ds_reads live (guide rule 17)
            for (int q = 0; q < 2; ++q) {
                asm volatile("" :: "v"(wfr[q][0]), "v"(wfr[q][1]));
                asm volatile("" :: "v"(hlo[q][0]), "v"(hhi[q][0]));
            }
        }
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(0);
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
        // ---- P2: h mi 4-7 (8 ds_reads); stage B-lo(j+2)
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
            const int row = wm * 128 + (4 + mi) * 16 + lr;
            const int co = (lk ^ swz64(row)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                hhi[mi][kb] = *(const bf16x8*)(ab + kb * 16384 + row * 64 + co);
        }
        if (!(ABL & 1) && j + 2 < nsteps) stage_half(1, 0, buf, (j + 2) * BK);
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(1);
if constexpr (!(ABL & 2)) {
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[g][4 + mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hhi[mi][kb], acc[g][4 + mi], 0, 0, 0);
        } else {
            // keep the phase's ds_reads live (guide rule 17)
            for (int q = 0; q < 2; ++q) {
                asm volatile("" :: "v"(wfr[q][0]), "v"(wfr[q][1]));
                asm volatile("" :: "v"(hlo[q][0]), "v"(hhi[q][0]));
            }
        }
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(0);
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
        // ---- P3: W gates 2-3 (4 ds_reads); stage A-hi(j+2)
#pragma unroll
        for (int g = 0; g < 2; ++g) {
            const int gc = (2 + g) * 64 + wn * 16 + lr;
            const int co = (lk ^ swz64(gc)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                wfr[g][kb] = *(const bf16x8*)(bb + kb * 16384 + gc * 64 + co);
        }
        if (!(ABL & 1) && j + 2 < nsteps) stage_half(0, 1, buf, (j + 2) * BK);
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(1);
if constexpr (!(ABL & 2)) {
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[2 + g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hlo[mi][kb], acc[2 + g][mi], 0, 0, 0);
        } else {
            // keep the phase's ds_reads live (guide rule 17)
            for (int q = 0; q < 2; ++q) {
                asm volatile("" :: "v"(wfr[q][0]), "v"(wfr[q][1]));
                asm volatile("" :: "v"(hlo[q][0]), "v"(hhi[q][0]));
            }
        }
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(0);
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
        // ---- P4: no ds_reads; stage B-hi(j+2); boundary vmcnt
        if (!(ABL & 1) && j + 2 < nsteps) stage_half(1, 1, buf, (j + 2) * BK);
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(1);
if constexpr (!(ABL & 2)) {
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < 4; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[2 + g][4 + mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hhi[mi][kb], acc[2 + g][4 + mi], 0, 0, 0);
        } else {
            // keep the phase's ds_reads live (guide rule 17)
            for (int q = 0; q < 2; ++q) {
                asm volatile("" :: "v"(wfr[q][0]), "v"(wfr[q][1]));
                asm volatile("" :: "v"(hlo[q][0]), "v"(hhi[q][0]));
            }
        }
        if (!(ABL & 4)) __builtin_amdgcn_s_setprio(0);
        if (!(ABL & 1) && j + 1 < nsteps) {
            if (j + 2 < nsteps)
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        if (!(ABL & 8)) __builtin_amdgcn_s_barrier();
    }

    // Fused epilogue (identical to k_lstm_rec_fwd_mt256): lane owns
    // h-columns [hcol0, hcol0+4) of batch rows (wm*128 + nf*16 + lr).
    const int hcol0 = (int)n_col0 + wn * 16 + lk * 4;
#pragma unroll
    for (int nf = 0; nf < 8; ++nf) {
        const long row = m_base + wm * 128 + nf * 16 + lr;
        f32x4 raw[4];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            const bf16x4 pv = *(const bf16x4*)(
                pre + row * pre_stride + (long)g * H + hcol0);
#pragma unroll
            for (int j = 0; j < 4; ++j)
                raw[g][j] = acc[g][nf][j] + (float)pv[j];
        }
        __hip_bfloat16* ap = acts + row * act_stride + hcol0;
        if (RAW_DEBUG) {
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                bf16x4 o;
#pragma unroll
                for (int j = 0; j < 4; ++j) o[j] = (__bf16)raw[g][j];
                *(bf16x4*)(ap + (long)g * H) = o;
            }
            continue;
        }
        const f32x4 cp = *(const f32x4*)(c_prev + row * cp_stride + hcol0);
        bf16x4 oi, of, og, oo, oh;
        f32x4 oc;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
            const float gi = sigmoidf_(raw[0][j]);
            const float gf = sigmoidf_(raw[1][j]);
            const float gg = tanhf_(raw[2][j]);
            const float go = sigmoidf_(raw[3][j]);
            const float c  = gf * cp[j] + gi * gg;
            oc[j] = c;
            oh[j] = (__bf16)(go * tanhf_(c));
            oi[j] = (__bf16)gi;
            of[j] = (__bf16)gf;
            og[j] = (__bf16)gg;
            oo[j] = (__bf16)go;
        }
        *(bf16x4*)(h_out + row * ho_stride + hcol0) = oh;
        *(f32x4*)(c_out + row * co_stride + hcol0) = oc;
        *(bf16x4*)(ap)               = oi;
        *(bf16x4*)(ap + (long)H)     = of;
        *(bf16x4*)(ap + (long)2 * H) = og;
        *(bf16x4*)(ap + (long)3 * H) = oo;
    }
}

// ---------------------------------------------------------------------------
// 2-PHASE merged variant of the 8-phase kernel: same 256x256 BK=64 tile and
// LDS images, but TWO phases per K-tile of 32 MFMA each -- half the barrier
// crossings, double-length matrix segments (the 4-phase ablation showed the
// barrier+read scaffold, not MFMA, dominating).  No setprio (measured
// neutral-to-negative on this GEMM shape, guide T5 "NOT GEMM").
//   Ph1: ds_read A mi0-7 (16) + W g01 (4); stage {A-lo(j+1), B-hi(j+1)};
//        barrier; lgkmcnt(0); 32 MFMA (g01 x mi0-7 x kb); barrier.
//   Ph2: ds_read W g23 (4); stage {B-lo(j+2), A-hi(j+2)};
//        barrier; lgkmcnt(0); 32 MFMA (g23); boundary vmcnt(4); barrier.
// Slot-freeness: B-hi(j+1) stages in Ph1(j) because its slot (B-hi(j-1))
// was last read in Ph2(j-1); B-lo/A-hi(j+2) stage in Ph2(j) after their
// slots' last reads in Ph1(j).  Boundary vmcnt(4) leaves exactly Ph2(j)'s
// two stages in flight and guarantees ALL of tile j+1 landed.
// ---------------------------------------------------------------------------
template <bool RAW_DEBUG, bool XCD_SWZ, int TUNE = 0, int BM = 256>
__global__ __launch_bounds__(512, 2) void k_lstm_rec_fwd_2ph(
    const __hip_bfloat16* __restrict__ h_prev, long hp_stride,
    const __hip_bfloat16* __restrict__ w_t,     // (4H, H) row-major = W_h^T
    const __hip_bfloat16* __restrict__ pre, long pre_stride,   // (B, 4H)
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ h_out, long ho_stride,
    float* __restrict__ c_out, long co_stride,
    __hip_bfloat16* __restrict__ acts, long act_stride,
    int B, int H)
{
    constexpr int BK = 64;
    constexpr int MI = BM / 32;        // m-fragments per wave
    constexpr int MHALF = BM / 2;      // rows per wave m-half
    constexpr int ABYTES = BM * 128;   // A operand tile (2 kb sub-images)
    constexpr int BUFB = ABYTES + 32768;
    // per-wave glds per stage: B halves 2, A halves BM/128 (at BM=128 each
    // wave issues ONE -- waves 0-3 fill kb=0, waves 4-7 kb=1)
    constexpr int NB = 2 + BM / 128;   // boundary vmcnt: Ph2's stages
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr = lane & 15;
    const int lk = lane >> 4;
    const int wm = wave & 1;
    const int wn = wave >> 1;

    int m_blk = blockIdx.x, n_blk = blockIdx.y;
    if (XCD_SWZ) {
        const int nm = gridDim.x, nwg = gridDim.x * gridDim.y;
        const int bid = blockIdx.x + blockIdx.y * nm;
        const int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, loc = bid >> 3;
        const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                        + loc;
        m_blk = swz % nm;
        n_blk = swz / nm;
    }
    const int m_base = m_blk * BM;
    const long n_col0 = (long)n_blk * 64;
    const int nsteps = H / BK;

    auto stage_half = [&](int is_b, int hf, int buf, int k0) {
        char* const base = smem + buf * BUFB + is_b * ABYTES;
        if (is_b) {
            const int r = hf * 128 + (tid >> 2);
            const int ch = tid & 3;
            const int chs = (ch ^ swz64(r)) * 8;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb) {
                const long grow = (long)(r >> 6) * H + n_col0 + (r & 63);
                const void* src = w_t + grow * H + (k0 + kb * 32) + chs;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        base + kb * 16384 + hf * 8192 + wave * 1024),
                    16, 0, 0);
            }
        } else if (BM == 256) {
            const int r = hf * 128 + (tid >> 2);
            const int chs = ((tid & 3) ^ swz64(r)) * 8;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb) {
                const void* src = h_prev + (long)(m_base + r) * hp_stride
                                  + (k0 + kb * 32) + chs;
                __builtin_amdgcn_global_load_lds(
                    (const __attribute__((address_space(1))) void*)src,
                    (__attribute__((address_space(3))) void*)(
                        base + kb * 16384 + hf * 8192 + wave * 1024),
                    16, 0, 0);
            }
        } else {
            // BM=128: one glds per thread; waves 0-3 fill kb=0, 4-7 kb=1
            const int kb = wave >> 2;
            const int t2 = tid & 255;
            const int r = hf * 64 + (t2 >> 2);
            const int chs = ((t2 & 3) ^ swz64(r)) * 8;
            const void* src = h_prev + (long)(m_base + r) * hp_stride
                              + (k0 + kb * 32) + chs;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    base + kb * (ABYTES / 2) + hf * (ABYTES / 4)
                    + (wave & 3) * 1024), 16, 0, 0);
        }
    };

    f32x4 acc[4][MI];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int mf = 0; mf < MI; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

    // prologue: tile 0 fully + Ph2(-1)'s {B-lo(1), A-hi(1)}
    stage_half(0, 0, 0, 0);
    stage_half(1, 1, 0, 0);
    stage_half(1, 0, 0, 0);
    stage_half(0, 1, 0, 0);
    if (nsteps > 1) {
        asm volatile("s_waitcnt vmcnt(%0)" :: "i"(NB) : "memory");
        stage_half(1, 0, 1, BK);         // B-lo(1)
        stage_half(0, 1, 1, BK);         // A-hi(1)
        // tile 0 landed (NB = Ph2-equivalent stages left in flight)
        asm volatile("s_waitcnt vmcnt(%0)" :: "i"(NB) : "memory");
    } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    bf16x8 wfr[2][2];
    bf16x8 hfr[MI][2];

    for (int j = 0; j < nsteps; ++j) {
        const int buf = j & 1;
        const char* ab = smem + buf * BUFB;
        const char* bb = ab + ABYTES;
        // ---- Ph1: A mi0-7 + W g01 (20 ds_reads); stage A-lo/B-hi(j+1)
        if ((TUNE & 2) && j + 1 < nsteps) {
            stage_half(0, 0, (j + 1) & 1, (j + 1) * BK);  // A-lo(j+1)
            stage_half(1, 1, (j + 1) & 1, (j + 1) * BK);  // B-hi(j+1)
        }
#pragma unroll
        for (int g = 0; g < 2; ++g) {
            const int gc = g * 64 + wn * 16 + lr;
            const int co = (lk ^ swz64(gc)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                wfr[g][kb] = *(const bf16x8*)(bb + kb * 16384 + gc * 64 + co);
        }
#pragma unroll
        for (int mi = 0; mi < MI; ++mi) {
            const int row = wm * MHALF + mi * 16 + lr;
            const int co = (lk ^ swz64(row)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                hfr[mi][kb] = *(const bf16x8*)(
                    ab + kb * (ABYTES / 2) + row * 64 + co);
        }
        if (!(TUNE & 2) && j + 1 < nsteps) {
            stage_half(0, 0, (j + 1) & 1, (j + 1) * BK);  // A-lo(j+1)
            stage_half(1, 1, (j + 1) & 1, (j + 1) * BK);  // B-hi(j+1)
        }
        if (!(TUNE & 1))
            asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < MI; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hfr[mi][kb], acc[g][mi], 0, 0, 0);
        __builtin_amdgcn_s_barrier();
        // ---- Ph2: W g23 (4 ds_reads); stage B-lo/A-hi(j+2)
        if ((TUNE & 2) && j + 2 < nsteps) {
            stage_half(1, 0, buf, (j + 2) * BK);  // B-lo(j+2)
            stage_half(0, 1, buf, (j + 2) * BK);  // A-hi(j+2)
        }
#pragma unroll
        for (int g = 0; g < 2; ++g) {
            const int gc = (2 + g) * 64 + wn * 16 + lr;
            const int co = (lk ^ swz64(gc)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                wfr[g][kb] = *(const bf16x8*)(bb + kb * 16384 + gc * 64 + co);
        }
        if (!(TUNE & 2) && j + 2 < nsteps) {
            stage_half(1, 0, buf, (j + 2) * BK);  // B-lo(j+2)
            stage_half(0, 1, buf, (j + 2) * BK);  // A-hi(j+2)
        }
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < MI; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[2 + g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hfr[mi][kb], acc[2 + g][mi], 0, 0, 0);
        if (j + 1 < nsteps) {
            if (j + 2 < nsteps)
                asm volatile("s_waitcnt vmcnt(%0)" :: "i"(NB) : "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
    }

    // Epilogue. RAW mode keeps the direct per-lane form (numerics tests).
    // The CELL path stages acc through LDS (free after the K-loop) so that
    // every global access is a coalesced >=16-B op: the direct form's 8-B
    // accesses (lane = 4 cols of one row, rows 16 apart across lanes) were
    // store-issue-bound -- measured ~60 us of the B=16384 step.
    if (RAW_DEBUG) {
        const int hcol0 = (int)n_col0 + wn * 16 + lk * 4;
#pragma unroll
        for (int nf = 0; nf < MI; ++nf) {
            const long row = m_base + wm * MHALF + nf * 16 + lr;
            __hip_bfloat16* ap = acts + row * act_stride + hcol0;
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const bf16x4 pv = *(const bf16x4*)(
                    pre + row * pre_stride + (long)g * H + hcol0);
                bf16x4 o;
#pragma unroll
                for (int jj = 0; jj < 4; ++jj)
                    o[jj] = (__bf16)(acc[g][nf][jj] + (float)pv[jj]);
                *(bf16x4*)(ap + (long)g * H) = o;
            }
        }
        return;
    }
    float* const img = (float*)smem;        // [64 rows][256 gate-cols] fp32
#pragma unroll
    for (int rd = 0; rd < MI / 2; ++rd) {
        if (rd) __syncthreads();            // previous round's reads done
        // write rows nf = 2rd, 2rd+1 of this wave's m-half; chunk index
        // XORed with (row & 3) << 2 to spread ds_write bank slots
#pragma unroll
        for (int nn = 0; nn < 2; ++nn) {
            const int nf = rd * 2 + nn;
            const int irow = wm * 32 + nn * 16 + lr;
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const int chunk = (g * 16 + wn * 4 + lk) ^ ((irow & 3) << 2);
                *(f32x4*)(img + irow * 256 + chunk * 4) = acc[g][nf];
            }
        }
        __syncthreads();
        // read pass: thread = (row = tid>>3, 8 h-cols = (tid&7)*8)
        const int irow = tid >> 3;
        const int hc = (tid & 7) * 8;
        const int wm_r = irow >> 5, rem = irow & 31;
        const long grow = m_base + wm_r * MHALF + (rd * 2 + (rem >> 4)) * 16
                          + (rem & 15);
        float raw[4][8];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
#pragma unroll
            for (int hh = 0; hh < 2; ++hh) {
                const int chunk = (g * 16 + (hc >> 2) + hh)
                                  ^ ((irow & 3) << 2);
                const f32x4 v = *(const f32x4*)(img + irow * 256 + chunk * 4);
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) raw[g][hh * 4 + jj] = v[jj];
            }
            const bf16x8 pv = *(const bf16x8*)(
                pre + grow * pre_stride + (long)g * H + n_col0 + hc);
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) raw[g][jj] += (float)pv[jj];
        }
        const long hcol = n_col0 + hc;
        const f32x4 cp0 = *(const f32x4*)(c_prev + grow * cp_stride + hcol);
        const f32x4 cp1 = *(const f32x4*)(c_prev + grow * cp_stride + hcol + 4);
        bf16x8 oi, of, og, oo, oh;
        f32x4 oc0, oc1;
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
            const float gi = sigmoidf_(raw[0][jj]);
            const float gf = sigmoidf_(raw[1][jj]);
            const float gg = tanhf_(raw[2][jj]);
            const float go = sigmoidf_(raw[3][jj]);
            const float cpv = jj < 4 ? cp0[jj] : cp1[jj - 4];
            const float c = gf * cpv + gi * gg;
            if (jj < 4) oc0[jj] = c; else oc1[jj - 4] = c;
            oh[jj] = (__bf16)(go * tanhf_(c));
            oi[jj] = (__bf16)gi;
            of[jj] = (__bf16)gf;
            og[jj] = (__bf16)gg;
            oo[jj] = (__bf16)go;
        }
        *(bf16x8*)(h_out + grow * ho_stride + hcol) = oh;
        *(f32x4*)(c_out + grow * co_stride + hcol) = oc0;
        *(f32x4*)(c_out + grow * co_stride + hcol + 4) = oc1;
        __hip_bfloat16* ap = acts + grow * act_stride + hcol;
        *(bf16x8*)(ap)               = oi;
        *(bf16x8*)(ap + (long)H)     = of;
        *(bf16x8*)(ap + (long)2 * H) = og;
        *(bf16x8*)(ap + (long)3 * H) = oo;
    }
}

// ---------------------------------------------------------------------------
// OVERLAP variant [r2 experiment]: 128x256 tile, BK=32, 256 threads
// (4 waves, 1M x 4N), THREE 24-KiB LDS buffers = 72 KiB/block -> TWO
// blocks co-resident per CU (VGPR ~200 = 2 waves/SIMD).  While one
// block's waves sit in an MFMA segment, the partner block's waves run
// their HBM-bound staging / LDS-bounce epilogue -- the mem-beside-matrix
// pairing the pipeline-locked 512-thread kernels cannot get.  One phase
// per K-tile: {12 ds_reads; stage tile j+2 (6 glds, 3rd buffer);
// s_barrier; lgkmcnt(0); 32 MFMA; vmcnt(6) boundary; s_barrier}.
// ---------------------------------------------------------------------------
template <bool RAW_DEBUG, bool XCD_SWZ, int TUNE = 0>
__global__ __launch_bounds__(256, 2) void k_lstm_rec_fwd_ov(
    const __hip_bfloat16* __restrict__ h_prev, long hp_stride,
    const __hip_bfloat16* __restrict__ w_t,     // (4H, H) row-major = W_h^T
    const __hip_bfloat16* __restrict__ pre, long pre_stride,   // (B, 4H)
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ h_out, long ho_stride,
    float* __restrict__ c_out, long co_stride,
    __hip_bfloat16* __restrict__ acts, long act_stride,
    int B, int H)
{
    constexpr int BK = 32;
    constexpr int ABYTES = 128 * 64;      // 8 KiB (128 rows x 32 k bf16)
    constexpr int BBYTES = 256 * 64;      // 16 KiB (256 gate-cols x 32 k)
    constexpr int BUFB = ABYTES + BBYTES; // 24 KiB; x3 buffers = 72 KiB
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr = lane & 15;
    const int lk = lane >> 4;
    const int wn = wave;               // 4 N groups of 16 h-cols

    int m_blk = blockIdx.x, n_blk = blockIdx.y;
    if (XCD_SWZ) {
        const int nm = gridDim.x, nwg = gridDim.x * gridDim.y;
        const int bid = blockIdx.x + blockIdx.y * nm;
        const int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, loc = bid >> 3;
        const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                        + loc;
        m_blk = swz % nm;
        n_blk = swz / nm;
    }
    const int m_base = m_blk * 128;
    const long n_col0 = (long)n_blk * 64;
    const int nsteps = H / BK;

    // stage one whole BK=32 K-tile (A 8 KiB + B 16 KiB, 6 glds per thread)
    auto stage = [&](int buf, int k0) {
        char* const ab = smem + buf * BUFB;
        char* const bb = ab + ABYTES;
#pragma unroll
        for (int s2 = 0; s2 < 2; ++s2) {
            const int ci = s2 * 256 + tid;
            const int r = ci >> 2, ch = ci & 3;
            const void* src = h_prev + (long)(m_base + r) * hp_stride + k0
                              + ((ch ^ swz64(r)) * 8);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    ab + s2 * 4096 + wave * 1024), 16, 0, 0);
        }
#pragma unroll
        for (int s2 = 0; s2 < 4; ++s2) {
            const int ci = s2 * 256 + tid;
            const int gc = ci >> 2, ch = ci & 3;
            const long grow = (long)(gc >> 6) * H + n_col0 + (gc & 63);
            const void* src = w_t + grow * H + k0 + ((ch ^ swz64(gc)) * 8);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    bb + s2 * 4096 + wave * 1024), 16, 0, 0);
        }
    };

    f32x4 acc[4][8];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int mf = 0; mf < 8; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

    stage(0, 0);
    if (nsteps > 1) stage(1, BK);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");  // tile 0 landed
    __builtin_amdgcn_s_barrier();

    for (int j = 0; j < nsteps; ++j) {
        const int buf = j % 3;
        const char* ab = smem + buf * BUFB;
        const char* bb = ab + ABYTES;
        // glds first by default (TUNE2 lesson from the 2ph kernel);
        // TUNE&1 = reads-first for A/B.  3 buffers make either order safe.
        if (!(TUNE & 1) && j + 2 < nsteps) stage((j + 2) % 3, (j + 2) * BK);
        bf16x8 wfr[4], hfr[8];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            const int gc = g * 64 + wn * 16 + lr;
            wfr[g] = *(const bf16x8*)(bb + gc * 64 + ((lk ^ swz64(gc)) * 16));
        }
#pragma unroll
        for (int mi = 0; mi < 8; ++mi) {
            const int row = mi * 16 + lr;
            hfr[mi] = *(const bf16x8*)(ab + row * 64 + ((lk ^ swz64(row)) * 16));
        }
        if ((TUNE & 1) && j + 2 < nsteps) stage((j + 2) % 3, (j + 2) * BK);
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
            for (int mi = 0; mi < 8; ++mi)
                acc[g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    wfr[g], hfr[mi], acc[g][mi], 0, 0, 0);
        if (j + 1 < nsteps) {
            if (j + 2 < nsteps)
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
    }

    // Epilogue. RAW: direct stores; CELL: LDS-bounce (own 72 KiB now free)
    if (RAW_DEBUG) {
        const int hcol0 = (int)n_col0 + wn * 16 + lk * 4;
#pragma unroll
        for (int nf = 0; nf < 8; ++nf) {
            const long row = m_base + nf * 16 + lr;
            __hip_bfloat16* ap = acts + row * act_stride + hcol0;
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const bf16x4 pv = *(const bf16x4*)(
                    pre + row * pre_stride + (long)g * H + hcol0);
                bf16x4 o;
#pragma unroll
                for (int jj = 0; jj < 4; ++jj)
                    o[jj] = (__bf16)(acc[g][nf][jj] + (float)pv[jj]);
                *(bf16x4*)(ap + (long)g * H) = o;
            }
        }
        return;
    }
    float* const img = (float*)smem;        // [32 rows][256 gate-cols] fp32
#pragma unroll
    for (int rd = 0; rd < 4; ++rd) {
        if (rd) __syncthreads();
#pragma unroll
        for (int nn = 0; nn < 2; ++nn) {
            const int nf = rd * 2 + nn;
            const int irow = nn * 16 + lr;
#pragma unroll
            for (int g = 0; g < 4; ++g) {
                const int chunk = (g * 16 + wn * 4 + lk) ^ ((irow & 3) << 2);
                *(f32x4*)(img + irow * 256 + chunk * 4) = acc[g][nf];
            }
        }
        __syncthreads();
        // 32 rows x 8 h-col chunks = 256 tasks, one per thread
        const int irow = tid >> 3;
        const int hc = (tid & 7) * 8;
        const long grow = m_base + (rd * 2 + (irow >> 4)) * 16 + (irow & 15);
        float raw[4][8];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
#pragma unroll
            for (int hh = 0; hh < 2; ++hh) {
                const int chunk = (g * 16 + (hc >> 2) + hh)
                                  ^ ((irow & 3) << 2);
                const f32x4 v = *(const f32x4*)(img + irow * 256 + chunk * 4);
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) raw[g][hh * 4 + jj] = v[jj];
            }
            const bf16x8 pv = *(const bf16x8*)(
                pre + grow * pre_stride + (long)g * H + n_col0 + hc);
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) raw[g][jj] += (float)pv[jj];
        }
        const long hcol = n_col0 + hc;
        const f32x4 cp0 = *(const f32x4*)(c_prev + grow * cp_stride + hcol);
        const f32x4 cp1 = *(const f32x4*)(c_prev + grow * cp_stride + hcol + 4);
        bf16x8 oi, of, og, oo, oh;
        f32x4 oc0, oc1;
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) {
            const float gi = sigmoidf_(raw[0][jj]);
            const float gf = sigmoidf_(raw[1][jj]);
            const float gg = tanhf_(raw[2][jj]);
            const float go = sigmoidf_(raw[3][jj]);
            const float cpv = jj < 4 ? cp0[jj] : cp1[jj - 4];
            const float c = gf * cpv + gi * gg;
            if (jj < 4) oc0[jj] = c; else oc1[jj - 4] = c;
            oh[jj] = (__bf16)(go * tanhf_(c));
            oi[jj] = (__bf16)gi;
            of[jj] = (__bf16)gf;
            og[jj] = (__bf16)gg;
            oo[jj] = (__bf16)go;
        }
        *(bf16x8*)(h_out + grow * ho_stride + hcol) = oh;
        *(f32x4*)(c_out + grow * co_stride + hcol) = oc0;
        *(f32x4*)(c_out + grow * co_stride + hcol + 4) = oc1;
        __hip_bfloat16* ap = acts + grow * act_stride + hcol;
        *(bf16x8*)(ap)               = oi;
        *(bf16x8*)(ap + (long)H)     = of;
        *(bf16x8*)(ap + (long)2 * H) = og;
        *(bf16x8*)(ap + (long)3 * H) = oo;
    }
}

// ---------------------------------------------------------------------------
// FUSED BACKWARD STEP [K5a+K4]: dh_t = dgates_{t+1} @ W_h^T on MFMA, with
// the pointwise backward of step t fused into the epilogue -- one kernel
// replaces the hipBLASLt dh GEMM + k_lstm_pw_bwd pair per timestep AND
// keeps the (B,H) dh carry in registers (no bf16 round trip; the carry is
// fp32 = strictly better numerics than the split path's bf16 dh_buf).
//
// GEMM mapping mirrors k_lstm_rec_fwd_2ph exactly (256x256 output tile,
// BK=64 as two swz64 BK=32 sub-images, 2 phases/tile, counted vmcnt):
//   - M = batch rows of dgates_{t+1} (the "A image"),
//   - N = 256 dh columns; the "B image" rows are W_h rows n_col0+gc
//     ((H, 4H) row-major: k = gate-dim is contiguous -- NO transpose copy),
//   - K = 4H (64 K-tiles at H=1024).
// The wave's 4 column-frags are cf*64 + wn*16 (+lr), NOT gates: the
// epilogue loops cf to cover 4 dh column groups, each lane owning 4
// consecutive dh columns -> every pw access is bf16x4/f32x4.
// RAW_DEBUG: write dh only (used for the chain-end dh0 and numerics tests).
// ---------------------------------------------------------------------------
template <bool RAW_DEBUG, bool XCD_SWZ>
__global__ __launch_bounds__(512, 2) void k_lstm_dh_pw_bwd(
    const __hip_bfloat16* __restrict__ dg_next, long dgn_stride,  // (B,4H)
    const __hip_bfloat16* __restrict__ w_h, long wh_stride,       // (H,4H)
    const __hip_bfloat16* __restrict__ dh_up, long up_stride,     // nullable
    const __hip_bfloat16* __restrict__ acts, long act_stride,     // (B,4H)
    const float* __restrict__ c_cur, long cc_stride,
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ dgates_out, long dgo_stride,     // (B,4H)
    float* __restrict__ dc, long dc_stride,                       // in-place
    __hip_bfloat16* __restrict__ dh_out, long dho_stride,         // RAW only
    float* __restrict__ db_partial,   // nullable: (B/256, 4H) fp32, this
                                      // block += its column sums (bias grad
                                      // accumulated at write -- saves the
                                      // separate 17 GB/layer colsum pass)
    int B, int H)
{
    constexpr int BK = 64;
    constexpr int OPBYTES = 32768;
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr = lane & 15;
    const int lk = lane >> 4;
    const int wm = wave & 1;
    const int wn = wave >> 1;

    int m_blk = blockIdx.x, n_blk = blockIdx.y;
    if (XCD_SWZ) {
        const int nm = gridDim.x, nwg = gridDim.x * gridDim.y;
        const int bid = blockIdx.x + blockIdx.y * nm;
        const int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, loc = bid >> 3;
        const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                        + loc;
        m_blk = swz % nm;
        n_blk = swz / nm;
    }
    const int m_base = m_blk * 256;
    const long n_col0 = (long)n_blk * 256;   // 256 dh columns per tile
    const int K = 4 * H;
    const int nsteps = K / BK;

    auto stage_half = [&](int is_b, int hf, int buf, int k0) {
        char* const base = smem + buf * (2 * OPBYTES) + is_b * OPBYTES;
        const int r = hf * 128 + (tid >> 2);
        const int ch = tid & 3;
        const int chs = (ch ^ swz64(r)) * 8;
#pragma unroll
        for (int kb = 0; kb < 2; ++kb) {
            const void* src;
            if (!is_b) {
                src = dg_next + (long)(m_base + r) * dgn_stride + (k0 + kb * 32)
                      + chs;
            } else {
                src = w_h + (n_col0 + r) * wh_stride + (k0 + kb * 32) + chs;
            }
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    base + kb * 16384 + hf * 8192 + wave * 1024), 16, 0, 0);
        }
    };

    f32x4 acc[4][8];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int mf = 0; mf < 8; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

    stage_half(0, 0, 0, 0);
    stage_half(1, 1, 0, 0);
    stage_half(1, 0, 0, 0);
    stage_half(0, 1, 0, 0);
    if (nsteps > 1) {
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        stage_half(1, 0, 1, BK);
        stage_half(0, 1, 1, BK);
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    } else {
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();

    bf16x8 wfr[2][2];
    bf16x8 hfr[8][2];

    for (int j = 0; j < nsteps; ++j) {
        const int buf = j & 1;
        const char* ab = smem + buf * (2 * OPBYTES);
        const char* bb = ab + OPBYTES;
        if (j + 1 < nsteps) {
            stage_half(0, 0, (j + 1) & 1, (j + 1) * BK);
            stage_half(1, 1, (j + 1) & 1, (j + 1) * BK);
        }
#pragma unroll
        for (int g = 0; g < 2; ++g) {
            const int gc = g * 64 + wn * 16 + lr;
            const int co = (lk ^ swz64(gc)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                wfr[g][kb] = *(const bf16x8*)(bb + kb * 16384 + gc * 64 + co);
        }
#pragma unroll
        for (int mi = 0; mi < 8; ++mi) {
            const int row = wm * 128 + mi * 16 + lr;
            const int co = (lk ^ swz64(row)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                hfr[mi][kb] = *(const bf16x8*)(ab + kb * 16384 + row * 64 + co);
        }
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < 8; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hfr[mi][kb], acc[g][mi], 0, 0, 0);
        __builtin_amdgcn_s_barrier();
        if (j + 2 < nsteps) {
            stage_half(1, 0, buf, (j + 2) * BK);
            stage_half(0, 1, buf, (j + 2) * BK);
        }
#pragma unroll
        for (int g = 0; g < 2; ++g) {
            const int gc = (2 + g) * 64 + wn * 16 + lr;
            const int co = (lk ^ swz64(gc)) * 16;
#pragma unroll
            for (int kb = 0; kb < 2; ++kb)
                wfr[g][kb] = *(const bf16x8*)(bb + kb * 16384 + gc * 64 + co);
        }
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int g = 0; g < 2; ++g)
#pragma unroll
            for (int mi = 0; mi < 8; ++mi)
#pragma unroll
                for (int kb = 0; kb < 2; ++kb)
                    acc[2 + g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        wfr[g][kb], hfr[mi][kb], acc[2 + g][mi], 0, 0, 0);
        if (j + 1 < nsteps) {
            if (j + 2 < nsteps)
                asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
    }

    // Epilogue. RAW: direct per-lane dh store (chain-end dh0 + tests).
    // CELL path: LDS-bounce (see k_lstm_rec_fwd_2ph) -- the direct form's
    // 8-B scattered accesses made the fused step slower than split.
    if (RAW_DEBUG) {
#pragma unroll
        for (int cf = 0; cf < 4; ++cf) {
            const long hcol0 = n_col0 + cf * 64 + wn * 16 + lk * 4;
#pragma unroll
            for (int nf = 0; nf < 8; ++nf) {
                const long row = m_base + wm * 128 + nf * 16 + lr;
                bf16x4 o;
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) o[jj] = (__bf16)acc[cf][nf][jj];
                *(bf16x4*)(dh_out + row * dho_stride + hcol0) = o;
            }
        }
        return;
    }
    float* const img = (float*)smem;        // [64 rows][256 dh-cols] fp32
    // bias-grad accumulation: this thread's fixed 8 dh-cols x 4 gates
    float acc_db[4][8];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) acc_db[g][jj] = 0.f;
#pragma unroll
    for (int rd = 0; rd < 4; ++rd) {
        if (rd) __syncthreads();
#pragma unroll
        for (int nn = 0; nn < 2; ++nn) {
            const int nf = rd * 2 + nn;
            const int irow = wm * 32 + nn * 16 + lr;
#pragma unroll
            for (int cf = 0; cf < 4; ++cf) {
                const int chunk = (cf * 16 + wn * 4 + lk) ^ ((irow & 3) << 2);
                *(f32x4*)(img + irow * 256 + chunk * 4) = acc[cf][nf];
            }
        }
        __syncthreads();
        // 64 rows x 32 8-col chunks = 2048 tasks, 4 per thread
#pragma unroll
        for (int sub = 0; sub < 4; ++sub) {
            const int irow = sub * 16 + (tid >> 5);
            const int hc = (tid & 31) * 8;
            const int wm_r = irow >> 5, rem = irow & 31;
            const long grow = m_base + wm_r * 128 + (rd * 2 + (rem >> 4)) * 16
                              + (rem & 15);
            float dhv[8];
#pragma unroll
            for (int hh = 0; hh < 2; ++hh) {
                const int chunk = ((hc >> 2) + hh) ^ ((irow & 3) << 2);
                const f32x4 v = *(const f32x4*)(img + irow * 256 + chunk * 4);
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) dhv[hh * 4 + jj] = v[jj];
            }
            const long hcol = n_col0 + hc;
            if (dh_up) {
                const bf16x8 uv = *(const bf16x8*)(
                    dh_up + grow * up_stride + hcol);
#pragma unroll
                for (int jj = 0; jj < 8; ++jj) dhv[jj] += (float)uv[jj];
            }
            const f32x4 cc0 = *(const f32x4*)(c_cur + grow * cc_stride + hcol);
            const f32x4 cc1 = *(const f32x4*)(c_cur + grow * cc_stride + hcol + 4);
            const f32x4 cp0 = *(const f32x4*)(c_prev + grow * cp_stride + hcol);
            const f32x4 cp1 = *(const f32x4*)(c_prev + grow * cp_stride + hcol + 4);
            const f32x4 dv0 = *(const f32x4*)(dc + grow * dc_stride + hcol);
            const f32x4 dv1 = *(const f32x4*)(dc + grow * dc_stride + hcol + 4);
            const __hip_bfloat16* ap = acts + grow * act_stride + hcol;
            const bf16x8 vi = *(const bf16x8*)(ap);
            const bf16x8 vf = *(const bf16x8*)(ap + (long)H);
            const bf16x8 vg = *(const bf16x8*)(ap + (long)2 * H);
            const bf16x8 vo = *(const bf16x8*)(ap + (long)3 * H);
            bf16x8 out_i, out_f, out_g, out_o;
            f32x4 dc0, dc1;
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) {
                const float i = (float)vi[jj], f = (float)vf[jj];
                const float g = (float)vg[jj], o = (float)vo[jj];
                const float ccv = jj < 4 ? cc0[jj] : cc1[jj - 4];
                const float cpv = jj < 4 ? cp0[jj] : cp1[jj - 4];
                const float dcv = jj < 4 ? dv0[jj] : dv1[jj - 4];
                const float tc = tanhf_(ccv);
                const float d_o = dhv[jj] * tc;
                float d_c = dcv + dhv[jj] * o * (1.0f - tc * tc);
                const float d_i = d_c * g;
                const float d_g = d_c * i;
                const float d_f = d_c * cpv;
                const float dco = d_c * f;
                if (jj < 4) dc0[jj] = dco; else dc1[jj - 4] = dco;
                out_i[jj] = (__bf16)(d_i * i * (1.0f - i));
                out_f[jj] = (__bf16)(d_f * f * (1.0f - f));
                out_g[jj] = (__bf16)(d_g * (1.0f - g * g));
                out_o[jj] = (__bf16)(d_o * o * (1.0f - o));
            }
            __hip_bfloat16* dgp = dgates_out + grow * dgo_stride + hcol;
            *(bf16x8*)(dgp)               = out_i;
            *(bf16x8*)(dgp + (long)H)     = out_f;
            *(bf16x8*)(dgp + (long)2 * H) = out_g;
            *(bf16x8*)(dgp + (long)3 * H) = out_o;
            *(f32x4*)(dc + grow * dc_stride + hcol) = dc0;
            *(f32x4*)(dc + grow * dc_stride + hcol + 4) = dc1;
            if (db_partial) {
#pragma unroll
                for (int jj = 0; jj < 8; ++jj) {
                    acc_db[0][jj] += (float)out_i[jj];
                    acc_db[1][jj] += (float)out_f[jj];
                    acc_db[2][jj] += (float)out_g[jj];
                    acc_db[3][jj] += (float)out_o[jj];
                }
            }
        }
    }
    if (db_partial) {
        // deterministic block reduce: 16 ordered contributor groups into
        // a 4-KiB LDS psum (placed after the 64-KiB img), then one
        // read-modify-write of this block's own partials row (launches
        // are stream-serialized across timesteps, so += is race-free and
        // run-to-run reproducible).
        float* const psum = (float*)(smem + 65536);   // [4 gates][256 cols]
        for (int i = tid; i < 1024; i += 512) psum[i] = 0.f;
        __syncthreads();
        for (int c = 0; c < 16; ++c) {
            if ((tid >> 5) == c) {
                const int hc = (tid & 31) * 8;
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int jj = 0; jj < 8; ++jj)
                        psum[g * 256 + hc + jj] += acc_db[g][jj];
            }
            __syncthreads();
        }
        for (int i = tid; i < 1024; i += 512) {
            const int g = i >> 8, col = i & 255;
            float* p = db_partial + (long)m_blk * (4 * H) + (long)g * H
                       + n_col0 + col;
            *p += psum[i];
        }
    }
}

// Overlap variant of the fused backward step: 128x256 dh tile, BK=32,
// 256 threads, 3 LDS buffers (72 KiB) -> 2 blocks/CU so the pointwise
// epilogue's HBM traffic hides under the partner block's MFMA segments.
// Same operand mapping as k_lstm_dh_pw_bwd.
template <bool RAW_DEBUG, bool XCD_SWZ, int TUNE = 0>
__global__ __launch_bounds__(256, 2) void k_lstm_dh_pw_bwd_ov(
    const __hip_bfloat16* __restrict__ dg_next, long dgn_stride,  // (B,4H)
    const __hip_bfloat16* __restrict__ w_h, long wh_stride,       // (H,4H)
    const __hip_bfloat16* __restrict__ dh_up, long up_stride,     // nullable
    const __hip_bfloat16* __restrict__ acts, long act_stride,     // (B,4H)
    const float* __restrict__ c_cur, long cc_stride,
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ dgates_out, long dgo_stride,     // (B,4H)
    float* __restrict__ dc, long dc_stride,                       // in-place
    __hip_bfloat16* __restrict__ dh_out, long dho_stride,         // RAW only
    float* __restrict__ db_partial,   // nullable: (B/256, 4H) fp32, this
                                      // block += its column sums (bias grad
                                      // accumulated at write -- saves the
                                      // separate 17 GB/layer colsum pass)
    int B, int H)
{
    constexpr int BK = 32;
    constexpr int ABYTES = 128 * 64;
    constexpr int BBYTES = 256 * 64;
    constexpr int BUFB = ABYTES + BBYTES;
    extern __shared__ __attribute__((aligned(16))) char smem[];

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr = lane & 15;
    const int lk = lane >> 4;
    const int wn = wave;

    int m_blk = blockIdx.x, n_blk = blockIdx.y;
    if (XCD_SWZ) {
        const int nm = gridDim.x, nwg = gridDim.x * gridDim.y;
        const int bid = blockIdx.x + blockIdx.y * nm;
        const int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, loc = bid >> 3;
        const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q)
                        + loc;
        m_blk = swz % nm;
        n_blk = swz / nm;
    }
    const int m_base = m_blk * 128;
    const long n_col0 = (long)n_blk * 256;
    const int K = 4 * H;
    const int nsteps = K / BK;

    auto stage = [&](int buf, int k0) {
        char* const ab = smem + buf * BUFB;
        char* const bb = ab + ABYTES;
#pragma unroll
        for (int s2 = 0; s2 < 2; ++s2) {
            const int ci = s2 * 256 + tid;
            const int r = ci >> 2, ch = ci & 3;
            const void* src = dg_next + (long)(m_base + r) * dgn_stride + k0
                              + ((ch ^ swz64(r)) * 8);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    ab + s2 * 4096 + wave * 1024), 16, 0, 0);
        }
#pragma unroll
        for (int s2 = 0; s2 < 4; ++s2) {
            const int ci = s2 * 256 + tid;
            const int gc = ci >> 2, ch = ci & 3;
            const void* src = w_h + (n_col0 + gc) * wh_stride + k0
                              + ((ch ^ swz64(gc)) * 8);
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) void*)src,
                (__attribute__((address_space(3))) void*)(
                    bb + s2 * 4096 + wave * 1024), 16, 0, 0);
        }
    };

    f32x4 acc[4][8];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int mf = 0; mf < 8; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

    stage(0, 0);
    if (nsteps > 1) stage(1, BK);
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    for (int j = 0; j < nsteps; ++j) {
        const int buf = j % 3;
        const char* ab = smem + buf * BUFB;
        const char* bb = ab + ABYTES;
        // glds first by default (TUNE2 lesson from the 2ph kernel);
        // TUNE&1 = reads-first for A/B.  3 buffers make either order safe.
        if (!(TUNE & 1) && j + 2 < nsteps) stage((j + 2) % 3, (j + 2) * BK);
        bf16x8 wfr[4], hfr[8];
#pragma unroll
        for (int g = 0; g < 4; ++g) {
            const int gc = g * 64 + wn * 16 + lr;
            wfr[g] = *(const bf16x8*)(bb + gc * 64 + ((lk ^ swz64(gc)) * 16));
        }
#pragma unroll
        for (int mi = 0; mi < 8; ++mi) {
            const int row = mi * 16 + lr;
            hfr[mi] = *(const bf16x8*)(ab + row * 64 + ((lk ^ swz64(row)) * 16));
        }
        __builtin_amdgcn_s_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
            for (int mi = 0; mi < 8; ++mi)
                acc[g][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    wfr[g], hfr[mi], acc[g][mi], 0, 0, 0);
        if (j + 1 < nsteps) {
            if (j + 2 < nsteps)
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
    }

    if (RAW_DEBUG) {
#pragma unroll
        for (int cf = 0; cf < 4; ++cf) {
            const long hcol0 = n_col0 + cf * 64 + wn * 16 + lk * 4;
#pragma unroll
            for (int nf = 0; nf < 8; ++nf) {
                const long row = m_base + nf * 16 + lr;
                bf16x4 o;
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) o[jj] = (__bf16)acc[cf][nf][jj];
                *(bf16x4*)(dh_out + row * dho_stride + hcol0) = o;
            }
        }
        return;
    }
    float* const img = (float*)smem;        // [32 rows][256 dh-cols] fp32
    float acc_db[4][8];
#pragma unroll
    for (int g = 0; g < 4; ++g)
#pragma unroll
        for (int jj = 0; jj < 8; ++jj) acc_db[g][jj] = 0.f;
#pragma unroll
    for (int rd = 0; rd < 4; ++rd) {
        if (rd) __syncthreads();
#pragma unroll
        for (int nn = 0; nn < 2; ++nn) {
            const int nf = rd * 2 + nn;
            const int irow = nn * 16 + lr;
#pragma unroll
            for (int cf = 0; cf < 4; ++cf) {
                const int chunk = (cf * 16 + wn * 4 + lk) ^ ((irow & 3) << 2);
                *(f32x4*)(img + irow * 256 + chunk * 4) = acc[cf][nf];
            }
        }
        __syncthreads();
        // 32 rows x 32 8-col chunks = 1024 tasks, 4 per thread
#pragma unroll
        for (int sub = 0; sub < 4; ++sub) {
            const int irow = sub * 8 + (tid >> 5);
            const int hc = (tid & 31) * 8;
            const long grow = m_base + (rd * 2 + (irow >> 4)) * 16
                              + (irow & 15);
            float dhv[8];
#pragma unroll
            for (int hh = 0; hh < 2; ++hh) {
                const int chunk = ((hc >> 2) + hh) ^ ((irow & 3) << 2);
                const f32x4 v = *(const f32x4*)(img + irow * 256 + chunk * 4);
#pragma unroll
                for (int jj = 0; jj < 4; ++jj) dhv[hh * 4 + jj] = v[jj];
            }
            const long hcol = n_col0 + hc;
            if (dh_up) {
                const bf16x8 uv = *(const bf16x8*)(
                    dh_up + grow * up_stride + hcol);
#pragma unroll
                for (int jj = 0; jj < 8; ++jj) dhv[jj] += (float)uv[jj];
            }
            const f32x4 cc0 = *(const f32x4*)(c_cur + grow * cc_stride + hcol);
            const f32x4 cc1 = *(const f32x4*)(c_cur + grow * cc_stride + hcol + 4);
            const f32x4 cp0 = *(const f32x4*)(c_prev + grow * cp_stride + hcol);
            const f32x4 cp1 = *(const f32x4*)(c_prev + grow * cp_stride + hcol + 4);
            const f32x4 dv0 = *(const f32x4*)(dc + grow * dc_stride + hcol);
            const f32x4 dv1 = *(const f32x4*)(dc + grow * dc_stride + hcol + 4);
            const __hip_bfloat16* ap = acts + grow * act_stride + hcol;
            const bf16x8 vi = *(const bf16x8*)(ap);
            const bf16x8 vf = *(const bf16x8*)(ap + (long)H);
            const bf16x8 vg = *(const bf16x8*)(ap + (long)2 * H);
            const bf16x8 vo = *(const bf16x8*)(ap + (long)3 * H);
            bf16x8 out_i, out_f, out_g, out_o;
            f32x4 dc0, dc1;
#pragma unroll
            for (int jj = 0; jj < 8; ++jj) {
                const float i = (float)vi[jj], f = (float)vf[jj];
                const float g = (float)vg[jj], o = (float)vo[jj];
                const float ccv = jj < 4 ? cc0[jj] : cc1[jj - 4];
                const float cpv = jj < 4 ? cp0[jj] : cp1[jj - 4];
                const float dcv = jj < 4 ? dv0[jj] : dv1[jj - 4];
                const float tc = tanhf_(ccv);
                const float d_o = dhv[jj] * tc;
                float d_c = dcv + dhv[jj] * o * (1.0f - tc * tc);
                const float d_i = d_c * g;
                const float d_g = d_c * i;
                const float d_f = d_c * cpv;
                const float dco = d_c * f;
                if (jj < 4) dc0[jj] = dco; else dc1[jj - 4] = dco;
                out_i[jj] = (__bf16)(d_i * i * (1.0f - i));
                out_f[jj] = (__bf16)(d_f * f * (1.0f - f));
                out_g[jj] = (__bf16)(d_g * (1.0f - g * g));
                out_o[jj] = (__bf16)(d_o * o * (1.0f - o));
            }
            __hip_bfloat16* dgp = dgates_out + grow * dgo_stride + hcol;
            *(bf16x8*)(dgp)               = out_i;
            *(bf16x8*)(dgp + (long)H)     = out_f;
            *(bf16x8*)(dgp + (long)2 * H) = out_g;
            *(bf16x8*)(dgp + (long)3 * H) = out_o;
            *(f32x4*)(dc + grow * dc_stride + hcol) = dc0;
            *(f32x4*)(dc + grow * dc_stride + hcol + 4) = dc1;
            if (db_partial) {
#pragma unroll
                for (int jj = 0; jj < 8; ++jj) {
                    acc_db[0][jj] += (float)out_i[jj];
                    acc_db[1][jj] += (float)out_f[jj];
                    acc_db[2][jj] += (float)out_g[jj];
                    acc_db[3][jj] += (float)out_o[jj];
                }
            }
        }
    }
    if (db_partial) {
        // 256-thread variant of the 512-thread kernel's deterministic
        // reduce (8 contributor groups); NOTE the partials row index is
        // in B/128 units here -- a caller enabling this must allocate
        // (B/128, 4H) partials (the binding currently passes null).
        float* const psum = (float*)(smem + 65536);   // [4 gates][256 cols]
        for (int i = tid; i < 1024; i += 256) psum[i] = 0.f;
        __syncthreads();
        for (int c = 0; c < 8; ++c) {
            if ((tid >> 5) == c) {
                const int hc = (tid & 31) * 8;
#pragma unroll
                for (int g = 0; g < 4; ++g)
#pragma unroll
                    for (int jj = 0; jj < 8; ++jj)
                        psum[g * 256 + hc + jj] += acc_db[g][jj];
            }
            __syncthreads();
        }
        for (int i = tid; i < 1024; i += 256) {
            const int g = i >> 8, col = i & 255;
            float* p = db_partial + (long)m_blk * (4 * H) + (long)g * H
                       + n_col0 + col;
            *p += psum[i];
        }
    }
}

extern "C" int launch_lstm_dh_pw_bwd_ov(
    const void* dg_next, long dgn_stride, const void* w_h, long wh_stride,
    const void* dh_up, long up_stride, const void* acts, long act_stride,
    const float* c_cur, long cc_stride, const float* c_prev, long cp_stride,
    void* dgates_out, long dgo_stride, float* dc, long dc_stride,
    void* dh_out, long dho_stride, int B, int H, int raw, int xcd_swz,
    hipStream_t stream)
{
    if (B % 128 != 0 || H % 256 != 0) return -1;
    const size_t smem = 3 * (128 * 64 + 256 * 64);
    dim3 grid(B / 128, H / 256), block(256);
#define DHOVLAUNCH(RAW_, SW_)                                                 \
    do {                                                                      \
        hipFuncSetAttribute((const void*)(k_lstm_dh_pw_bwd_ov<RAW_, SW_>),    \
                            hipFuncAttributeMaxDynamicSharedMemorySize,       \
                            (int)smem);                                       \
        hipLaunchKernelGGL((k_lstm_dh_pw_bwd_ov<RAW_, SW_>), grid, block,     \
                           smem, stream,                                      \
            (const __hip_bfloat16*)dg_next, dgn_stride,                       \
            (const __hip_bfloat16*)w_h, wh_stride,                            \
            (const __hip_bfloat16*)dh_up, up_stride,                          \
            (const __hip_bfloat16*)acts, act_stride, c_cur, cc_stride,        \
            c_prev, cp_stride, (__hip_bfloat16*)dgates_out, dgo_stride,       \
            dc, dc_stride, (__hip_bfloat16*)dh_out, dho_stride,               \
            (float*)nullptr, B, H);                                           \
    } while (0)
    if (raw) {
        if (xcd_swz) DHOVLAUNCH(true, true); else DHOVLAUNCH(true, false);
    } else {
        if (xcd_swz) DHOVLAUNCH(false, true); else DHOVLAUNCH(false, false);
    }
#undef DHOVLAUNCH
    return 0;
}

extern "C" int launch_lstm_dh_pw_bwd(
    const void* dg_next, long dgn_stride, const void* w_h, long wh_stride,
    const void* dh_up, long up_stride, const void* acts, long act_stride,
    const float* c_cur, long cc_stride, const float* c_prev, long cp_stride,
    void* dgates_out, long dgo_stride, float* dc, long dc_stride,
    void* dh_out, long dho_stride, float* db_partial, int B, int H, int raw,
    int xcd_swz, hipStream_t stream)
{
    if (B % 256 != 0 || H % 64 != 0) return -1;
    const size_t smem = 131072;
    dim3 grid(B / 256, H / 256), block(512);
    if (H % 256 != 0) return -1;
#define DHLAUNCH(RAW_, SW_)                                                   \
    do {                                                                      \
        hipFuncSetAttribute((const void*)(k_lstm_dh_pw_bwd<RAW_, SW_>),       \
                            hipFuncAttributeMaxDynamicSharedMemorySize,       \
                            (int)smem);                                       \
        hipLaunchKernelGGL((k_lstm_dh_pw_bwd<RAW_, SW_>), grid, block, smem,  \
                           stream,                                            \
            (const __hip_bfloat16*)dg_next, dgn_stride,                       \
            (const __hip_bfloat16*)w_h, wh_stride,                            \
            (const __hip_bfloat16*)dh_up, up_stride,                          \
            (const __hip_bfloat16*)acts, act_stride, c_cur, cc_stride,        \
            c_prev, cp_stride, (__hip_bfloat16*)dgates_out, dgo_stride,       \
            dc, dc_stride, (__hip_bfloat16*)dh_out, dho_stride, db_partial,   \
            B, H);                                                            \
    } while (0)
    if (raw) {
        if (xcd_swz) DHLAUNCH(true, true); else DHLAUNCH(true, false);
    } else {
        if (xcd_swz) DHLAUNCH(false, true); else DHLAUNCH(false, false);
    }
#undef DHLAUNCH
    return 0;
}

// ---------------------------------------------------------------------------
// PERSISTENT whole-sequence forward: ONE kernel for the entire T-step loop.
//
// The recurrent weight slice stays RESIDENT in LDS for all T timesteps and
// the cell state stays in registers; only h crosses steps (through L2,
// published with an agent-scope release / acquire grid barrier per step --
// guide §6 Guideline 16).  This is the SURVEY.md §7 "persistent-kernel
// fusion across timesteps" design:
//   - block = 256 threads = 4 waves; block tile = 256 batch rows x 16
//     h-columns (= 64 gate-columns);  wave w owns rows [w*64, +64).
//   - W_lds[64][H+8]: the block's 64 gate-columns of W_h^T, padded +8
//     elements/row so ds_read_b128 B-fragments are bank-conflict-free
//     (start banks 4*gcol mod 64, distinct within each 16-lane group).
//   - c kept in registers (same lane owns the same (row,hcol) across steps
//     because the C/D fragment map is step-invariant).
//   - grid = (B/256) x (H/16) <= 256 blocks -> fully resident at 1
//     block/CU (LDS-limited).  Natural dispatch places all n-blocks of one
//     m-block on one XCD (grid.x = m), so each XCD's L2 caches its own
//     256-row h panel.
//   - per-step grid barrier: plain h stores -> per-wave vmcnt(0) drain ->
//     __syncthreads -> one-lane agent RELEASE fence (+ restated asm wait,
//     G16 pitfall 12) -> relaxed arrival count on cnt[t];  consumer polls
//     relaxed with s_sleep + bounded spin (NO hang: on timeout sets
//     cnt[T] flag and every block returns), then ONE agent ACQUIRE.
// Eligibility: B % 256 == 0, H % 64 == 0, (B/256)*(H/16) <= 256,
//              64*(H+8)*2 <= 160 KiB (H <= 1144).  Others use the
//              per-step kernel path.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_lstm_seq_fwd_persistent(
    __hip_bfloat16* __restrict__ h_all,         // (T+1,B,H); h0 pre-written
                                                // at slot 0 (fwd) / T (rev)
    const __hip_bfloat16* __restrict__ w_t,     // (4H,H) = W_h^T
    const __hip_bfloat16* __restrict__ pre,     // (T,B,4H)
    const float* __restrict__ c0,               // (B,H)
    float* __restrict__ c_seq,                  // (T,B,H)
    __hip_bfloat16* __restrict__ acts,          // (T,B,4H)
    unsigned* __restrict__ cnt,                 // [T+1] zeroed; [T] = timeout flag
    int B, int H, int T, int reverse)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    __hip_bfloat16* w_lds = (__hip_bfloat16*)smem;   // [64][H+8]
    const int ldw = H + 8;

    const int tid = threadIdx.x;
    const int wave = tid >> 6;
    const int lane = tid & 63;
    const int lr = lane & 15;
    const int lk = lane >> 4;
    const int m_blk = blockIdx.x;   // fast dim -> XCD gets one m-panel
    const int n_blk = blockIdx.y;
    const int n0 = n_blk * 16;                  // first h-column
    const int m_base = m_blk * 256 + wave * 64; // wave's first batch row
    const int nblocks = gridDim.x * gridDim.y;
    // TIME-MAJOR: per-step blocks are dense; row stride H (h/c), 4H (pre/acts)
    const long BH = (long)B * H, BH4 = (long)B * 4 * H;

    // ---- fill the resident W slice: local gate-col gc = g*16+c  <-
    //      global w_t row g*H + n0 + c, all K
    for (int idx = tid; idx < 64 * (H / 8); idx += 256) {
        const int gc = idx / (H / 8);
        const int kc = (idx % (H / 8)) * 8;
        const long grow = (long)(gc >> 4) * H + n0 + (gc & 15);
        *(bf16x8*)(w_lds + gc * ldw + kc) = *(const bf16x8*)(w_t + grow * H + kc);
    }

    // ---- load this lane's resident cell state: rows mf*16 + lk*4 + r
    const int hcol = n0 + lr;
    float c_reg[4][4];
#pragma unroll
    for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int r = 0; r < 4; ++r)
            c_reg[mf][r] = c0[(long)(m_base + mf * 16 + lk * 4 + r) * H + hcol];

    __syncthreads();   // W_lds ready (block-local)

    const long koff0 = (long)lk * 8;
    int tmo = 0;
    for (int step = 0; step < T; ++step) {
        const int t = reverse ? (T - 1 - step) : step;
        // slot convention: h_prev(t) at slot t (fwd) / t+1 (rev); h0 was
        // pre-written at its slot by the driver, so no special-casing.
        const __hip_bfloat16* hsrc = h_all + (long)(reverse ? t + 1 : t) * BH;
        const long hstride = H;

        f32x4 acc[4][4];
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
            for (int mf = 0; mf < 4; ++mf) acc[g][mf] = f32x4{0, 0, 0, 0};

        // K loop: A fragments direct (L2-resident h panel), B from LDS.
        // A is register-prefetched one k-step ahead.
        bf16x8 a_cur[4], a_nxt[4];
#pragma unroll
        for (int mf = 0; mf < 4; ++mf)
            a_cur[mf] = *(const bf16x8*)(
                hsrc + (long)(m_base + mf * 16 + lr) * hstride + koff0);
        for (int k0 = 0; k0 < H; k0 += 32) {
            if (k0 + 32 < H) {
#pragma unroll
                for (int mf = 0; mf < 4; ++mf)
                    a_nxt[mf] = *(const bf16x8*)(
                        hsrc + (long)(m_base + mf * 16 + lr) * hstride
                        + k0 + 32 + koff0);
            }
            bf16x8 bfr[4];
#pragma unroll
            for (int g = 0; g < 4; ++g)
                bfr[g] = *(const bf16x8*)(
                    w_lds + (g * 16 + lr) * ldw + k0 + koff0);
#pragma unroll
            for (int g = 0; g < 4; ++g)
#pragma unroll
                for (int mf = 0; mf < 4; ++mf)
                    acc[g][mf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a_cur[mf], bfr[g], acc[g][mf], 0, 0, 0);
#pragma unroll
            for (int mf = 0; mf < 4; ++mf) a_cur[mf] = a_nxt[mf];
        }

        // ---- fused epilogue: pre + activations + in-register cell update
        const __hip_bfloat16* pre_t = pre + (long)t * BH4;
        __hip_bfloat16* h_t = h_all + (long)(reverse ? t : t + 1) * BH;
        float* c_t = c_seq + (long)t * BH;
        __hip_bfloat16* act_t = acts + (long)t * BH4;
#pragma unroll
        for (int mf = 0; mf < 4; ++mf) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const long row = m_base + mf * 16 + lk * 4 + r;
                float raw[4];
#pragma unroll
                for (int g = 0; g < 4; ++g)
                    raw[g] = acc[g][mf][r] +
                             __bfloat162float(pre_t[row * 4 * H + (long)g * H + hcol]);
                const float gi = sigmoidf_(raw[0]);
                const float gf = sigmoidf_(raw[1]);
                const float gg = tanhf_(raw[2]);
                const float go = sigmoidf_(raw[3]);
                const float c = gf * c_reg[mf][r] + gi * gg;
                c_reg[mf][r] = c;
                const float h = go * tanhf_(c);
                h_t[row * H + hcol] = __float2bfloat16(h);
                c_t[row * H + hcol] = c;
                __hip_bfloat16* ap = act_t + row * 4 * H + hcol;
                ap[0] = __float2bfloat16(gi);
                ap[(long)H] = __float2bfloat16(gf);
                ap[(long)2 * H] = __float2bfloat16(gg);
                ap[(long)3 * H] = __float2bfloat16(go);
            }
        }

        // ---- grid barrier: publish h_t, wait for all blocks (G16)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // every wave drains
        __syncthreads();
        if (tid == 0) {
            __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // pitfall 12
            __hip_atomic_fetch_add(&cnt[step], 1u, __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
            unsigned spins = 0;
            while (__hip_atomic_load(&cnt[step], __ATOMIC_RELAXED,
                                     __HIP_MEMORY_SCOPE_AGENT) < (unsigned)nblocks) {
                if (__hip_atomic_load(&cnt[T], __ATOMIC_RELAXED,
                                      __HIP_MEMORY_SCOPE_AGENT) != 0u) {
                    tmo = 1;  // another block timed out: abort too
                    break;
                }
                __builtin_amdgcn_s_sleep(8);
                if (++spins > 10u * 1000u * 1000u) {  // a few seconds: give up
                    __hip_atomic_store(&cnt[T], 1u, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT);
                    tmo = 1;
                    break;
                }
            }
            __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
        }
        if (__syncthreads_count(tmo) != 0) return;  // a lane timed out: abort
    }
}

// ---------------------------------------------------------------------------
// K2 (split path): pointwise LSTM cell update for one timestep, consuming a
// PRE-COMPUTED recurrent GEMM output raw_rec = h_{t-1} @ W_h (hipBLASLt,
// measured 908 TF/s on the (4096,1024)x(1024,4096) step shape vs ~475 for
// the best fused hand kernel -- the split's extra raw round-trip costs less
// than the GEMM gap).  Fully vectorized: 4 h-columns per lane.
// ---------------------------------------------------------------------------
template <int W>  // columns per lane: 8 (dwordx4 stores) when H%8==0
__global__ __launch_bounds__(256) void k_lstm_pw_fwd(
    const __hip_bfloat16* __restrict__ raw_rec, long rr_stride,  // (B,4H)
    const __hip_bfloat16* __restrict__ pre, long pre_stride,     // (B,4H)
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ h_out, long ho_stride,
    float* __restrict__ c_out, long co_stride,
    __hip_bfloat16* __restrict__ acts, long act_stride,
    int B, int H)
{
    typedef __bf16 bfv __attribute__((ext_vector_type(W)));
    typedef float fv __attribute__((ext_vector_type(W)));
    // flattened (row, col) indexing: every thread active regardless of H/W
    // vs blockDim (a (H/W)-per-block geometry left half the block idle at
    // H=1024, W=8 -- measured 3.0 TB/s instead of ~6).
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const int cpr = H / W;            // vec-chunks per row
    const int row = (int)(idx / cpr);
    if (row >= B) return;
    const int col0 = (int)(idx % cpr) * W;
    const __hip_bfloat16* rp = raw_rec + (long)row * rr_stride + col0;
    const __hip_bfloat16* pp = pre ? pre + (long)row * pre_stride + col0
                                   : nullptr;   // null = zero (decode path)
    fv raw[4];
#pragma unroll
    for (int g = 0; g < 4; ++g) {
        const bfv rv = *(const bfv*)(rp + (long)g * H);
#pragma unroll
        for (int j = 0; j < W; ++j) raw[g][j] = (float)rv[j];
        if (pp) {
            const bfv pv = *(const bfv*)(pp + (long)g * H);
#pragma unroll
            for (int j = 0; j < W; ++j) raw[g][j] += (float)pv[j];
        }
    }
    const fv cp = *(const fv*)(c_prev + (long)row * cp_stride + col0);
    bfv oi, of, og, oo, oh;
    fv oc;
#pragma unroll
    for (int j = 0; j < W; ++j) {
        const float gi = sigmoidf_(raw[0][j]);
        const float gf = sigmoidf_(raw[1][j]);
        const float gg = tanhf_(raw[2][j]);
        const float go = sigmoidf_(raw[3][j]);
        const float c = gf * cp[j] + gi * gg;
        oc[j] = c;
        oh[j] = (__bf16)(go * tanhf_(c));
        oi[j] = (__bf16)gi;
        of[j] = (__bf16)gf;
        og[j] = (__bf16)gg;
        oo[j] = (__bf16)go;
    }
    *(bfv*)(h_out + (long)row * ho_stride + col0) = oh;
    *(fv*)(c_out + (long)row * co_stride + col0) = oc;
    if (acts) {   // null = decode path (no backward, no saved activations)
        __hip_bfloat16* ap = acts + (long)row * act_stride + col0;
        *(bfv*)(ap)               = oi;
        *(bfv*)(ap + (long)H)     = of;
        *(bfv*)(ap + (long)2 * H) = og;
        *(bfv*)(ap + (long)3 * H) = oo;
    }
}

extern "C" void launch_lstm_pw_fwd(
    const void* raw_rec, long rr_stride, const void* pre, long pre_stride,
    const float* c_prev, long cp_stride, void* h_out, long ho_stride,
    float* c_out, long co_stride, void* acts, long act_stride,
    int B, int H, hipStream_t stream)
{
    dim3 block(256);
    if (H % 8 == 0) {
        dim3 grid((unsigned)(((long)B * (H / 8) + 255) / 256));
        hipLaunchKernelGGL(k_lstm_pw_fwd<8>, grid, block, 0, stream,
            (const __hip_bfloat16*)raw_rec, rr_stride,
            (const __hip_bfloat16*)pre, pre_stride, c_prev, cp_stride,
            (__hip_bfloat16*)h_out, ho_stride, c_out, co_stride,
            (__hip_bfloat16*)acts, act_stride, B, H);
    } else {
        dim3 grid((unsigned)(((long)B * (H / 4) + 255) / 256));
        hipLaunchKernelGGL(k_lstm_pw_fwd<4>, grid, block, 0, stream,
            (const __hip_bfloat16*)raw_rec, rr_stride,
            (const __hip_bfloat16*)pre, pre_stride, c_prev, cp_stride,
            (__hip_bfloat16*)h_out, ho_stride, c_out, co_stride,
            (__hip_bfloat16*)acts, act_stride, B, H);
    }
}

// ---------------------------------------------------------------------------
// K8: bias-grad column sum  out(C) = sum_r in(R,C) -- ONE fp32-accum pass
// over the (T*B, 4H) dgates (torch's generic two-stage reduce measured
// ~3x the roofline on this shape).  DETERMINISTIC two-stage: each block
// writes its fixed-order chunk partial to partials[gy][C]; a final kernel
// sums the gy partials per column in fixed order (the earlier fp32
// atomicAdd across grid-y blocks made bias grads non-bitwise run-to-run).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_colsum_bf16(
    const __hip_bfloat16* __restrict__ in, float* __restrict__ partials,
    long R, long C, int chunk)
{
    const long c0 = ((long)blockIdx.x * 256 + threadIdx.x) * 8;
    if (c0 >= C) return;
    const long r0 = (long)blockIdx.y * chunk;
    const long r1 = min(r0 + (long)chunk, R);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    for (long r = r0; r < r1; ++r) {
        const bf16x8 v = *(const bf16x8*)(in + r * C + c0);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += (float)v[j];
    }
    *(f32x4*)(&partials[(long)blockIdx.y * C + c0]) =
        f32x4{acc[0], acc[1], acc[2], acc[3]};
    *(f32x4*)(&partials[(long)blockIdx.y * C + c0 + 4]) =
        f32x4{acc[4], acc[5], acc[6], acc[7]};
}

__global__ __launch_bounds__(256) void k_colsum_final(
    const float* __restrict__ partials, float* __restrict__ out,
    long C, int gy)
{
    const long c = (long)blockIdx.x * 256 + threadIdx.x;  // one column/thread
    if (c >= C) return;
    float s = 0.f;
    for (int y = 0; y < gy; ++y) s += partials[(long)y * C + c];
    out[c] = s;
}

// Row-chunking: pick gy so grid.x * gy fills the chip (the fixed
// 4096-row chunk of r1 left a (16384, 4096) input at 8 blocks -> 1.8 ms
// for a 134 MB pass).  partials must hold colsum_gy(R, C) * C floats.
extern "C" int colsum_gy(long R, long C) {
    const long gx = (C / 8 + 255) / 256;
    long gy = 1024 / (gx < 1 ? 1 : gx);
    const long gy_max = (R + 63) / 64;   // >= 64 rows per chunk
    if (gy > gy_max) gy = gy_max;
    if (gy < 1) gy = 1;
    return (int)gy;
}

extern "C" void launch_colsum_final(const float* partials, float* out, long C,
                                    int gy, hipStream_t stream)
{
    dim3 block(256), grid((unsigned)((C + 255) / 256));
    hipLaunchKernelGGL(k_colsum_final, grid, block, 0, stream,
                       partials, out, C, gy);
}

extern "C" int launch_colsum_bf16(const void* in, float* partials, float* out,
                                  long R, long C, hipStream_t stream)
{
    if (C % 8 != 0) return -1;
    const int gy = colsum_gy(R, C);
    const int chunk = (int)((R + gy - 1) / gy);
    dim3 block(256);
    dim3 grid((unsigned)((C / 8 + 255) / 256), (unsigned)gy);
    hipLaunchKernelGGL(k_colsum_bf16, grid, block, 0, stream,
                       (const __hip_bfloat16*)in, partials, R, C, chunk);
    dim3 grid2((unsigned)((C + 255) / 256));
    hipLaunchKernelGGL(k_colsum_final, grid2, block, 0, stream,
                       partials, out, C, gy);
    return 0;
}

// ---------------------------------------------------------------------------
// Serving GEMV: out (B,N) = A (N,K row-major) @ x^T (B,K) + bias, for the
// SMALL-B decode path where hipBLASLt's GEMV leaves ~7x bandwidth on the
// table (per-token decode is pure weight streaming).  One WAVE per output
// row n: 64 lanes stride bf16x8 chunks over K, B accumulators per lane
// (x is LDS-resident), cross-lane shfl reduce.  A is the PRE-TRANSPOSED
// weight (serving layout, built once per model by the decode cache).
// ---------------------------------------------------------------------------
// NT: non-temporal weight loads (guide "nt-weights": a decode stream is
// read ONCE by one CU per token -- nt loads landed 18% faster there; the
// prefill-reuse caveat does not apply to this kernel's weight rows).
template <int BMAX, bool NT = false>
__global__ __launch_bounds__(256) void k_gemv_bf16(
    const __hip_bfloat16* __restrict__ A,    // (N, K) row-major
    const __hip_bfloat16* __restrict__ x,    // (B, K) row-major
    const float* __restrict__ bias,          // (N,) fp32 or null
    __hip_bfloat16* __restrict__ out,        // (B, N)
    float* __restrict__ out_f32,             // (B, N) or null (either/or)
    int B, long N, int K)
{
    extern __shared__ __attribute__((aligned(16))) char smem[];
    __hip_bfloat16* xs = (__hip_bfloat16*)smem;      // [B][K]
    const int tid = threadIdx.x;
    for (int i = tid * 8; i < B * K; i += 256 * 8) {
        *(bf16x8*)(xs + i) = *(const bf16x8*)(x + i);
    }
    __syncthreads();
    const int wave = tid >> 6, lane = tid & 63;
    const long wave_gid = (long)blockIdx.x * 4 + wave;
    const long nwaves = (long)gridDim.x * 4;
    // persistent wave-loop over output rows: keeps the weight stream
    // continuous instead of paying per-row wave ramp-up (K/1024 is only a
    // handful of iterations per row)
    for (long n = wave_gid; n < N; n += nwaves) {
        const __hip_bfloat16* ar = A + n * K;
        float acc[BMAX];
#pragma unroll
        for (int b = 0; b < BMAX; ++b) acc[b] = 0.f;
        // two bf16x8 loads in flight per lane per iteration (deeper MLP --
        // a single outstanding 16B load per lane leaves the row stream
        // latency-exposed)
        for (int k0 = lane * 16; k0 < K; k0 += 64 * 16) {
            bf16x8 av0, av1;
            if (NT) {
                av0 = __builtin_nontemporal_load((const bf16x8*)(ar + k0));
                av1 = __builtin_nontemporal_load((const bf16x8*)(ar + k0 + 8));
            } else {
                av0 = *(const bf16x8*)(ar + k0);
                av1 = *(const bf16x8*)(ar + k0 + 8);
            }
#pragma unroll
            for (int b = 0; b < BMAX; ++b) {
                if (b < B) {
                    const bf16x8 xv0 = *(const bf16x8*)(xs + b * K + k0);
                    const bf16x8 xv1 = *(const bf16x8*)(xs + b * K + k0 + 8);
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        acc[b] += (float)av0[j] * (float)xv0[j]
                                + (float)av1[j] * (float)xv1[j];
                }
            }
        }
#pragma unroll
        for (int b = 0; b < BMAX; ++b)
#pragma unroll
            for (int off = 32; off > 0; off >>= 1)
                acc[b] += __shfl_down(acc[b], off, 64);
        if (lane == 0) {
            const float bv = bias ? bias[n] : 0.f;
#pragma unroll
            for (int b = 0; b < BMAX; ++b)
                if (b < B) {
                    const float v = acc[b] + bv;
                    if (out) out[(long)b * N + n] = __float2bfloat16(v);
                    else out_f32[(long)b * N + n] = v;
                }
        }
    }
}

static int g_gemv_nt = 1;  // nt weight loads on by default (A/B'd r2)
extern "C" void set_gemv_nt(int v) { g_gemv_nt = v; }

// Returns 0 on launch, -1 when the shape is outside the fast-path
// envelope (the kernel template caps B at 16, K must satisfy the
// lane-stride contract, and x must fit LDS) -- the caller must NOT read
// `out` on -1.  Signature matches the `int` declaration in bindings.cpp
// (it was `void` here before: the rc check read an undefined register).
extern "C" int launch_gemv_bf16(const void* A, const void* x,
                                const float* bias, void* out, float* out_f32,
                                int B, long N, int K, hipStream_t stream)
{
    const size_t smem = (size_t)B * K * 2;
    // correctness envelope: the lane loop strides bf16x8 PAIRS (K % 16);
    // decode_gemv_ok additionally demands K % 1024 (full-lane PERF
    // envelope) for the auto path -- an explicit use_gemv with smaller K
    // is valid, just underutilizes lanes.
    if (B < 1 || B > 16 || K % 16 != 0 || smem > 160 * 1024 || N < 1)
        return -1;
    dim3 block(256), grid((unsigned)((N + 3) / 4));
#define GEMV_LAUNCH(BM)                                                       \
    do {                                                                      \
        if (g_gemv_nt)                                                        \
            hipLaunchKernelGGL((k_gemv_bf16<BM, true>), grid, block, smem,    \
                               stream, (const __hip_bfloat16*)A,              \
                               (const __hip_bfloat16*)x, bias,                \
                               (__hip_bfloat16*)out, out_f32, B, N, K);       \
        else                                                                  \
            hipLaunchKernelGGL((k_gemv_bf16<BM, false>), grid, block, smem,   \
                               stream, (const __hip_bfloat16*)A,              \
                               (const __hip_bfloat16*)x, bias,                \
                               (__hip_bfloat16*)out, out_f32, B, N, K);       \
    } while (0)
    if (B <= 1) GEMV_LAUNCH(1);
    else if (B <= 4) GEMV_LAUNCH(4);
    else if (B <= 8) GEMV_LAUNCH(8);
    else GEMV_LAUNCH(16);
#undef GEMV_LAUNCH
    return 0;
}

// ---------------------------------------------------------------------------
// FP8 quantization for the split-path recurrent GEMM: h is tanh-bounded to
// (-1,1), so a FIXED scale of 448 (e4m3 max) needs no amax pass; the GEMM
// dequantizes with scale_a = 1/448.  OCP e4m3 (gfx950), not fnuz.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_quant_fp8_fixed(
    const __hip_bfloat16* __restrict__ in, unsigned char* __restrict__ out,
    long n, float scale)
{
    const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    if (i0 + 8 <= n) {
        const bf16x8 v = *(const bf16x8*)(in + i0);
        unsigned long long pk = 0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const __hip_fp8_e4m3 q((float)v[j] * scale);
            pk |= (unsigned long long)q.__x << (8 * j);
        }
        *(unsigned long long*)(out + i0) = pk;
    } else {
        for (long i = i0; i < n; ++i) {
            const __hip_fp8_e4m3 q(__bfloat162float(in[i]) * scale);
            out[i] = q.__x;
        }
    }
}

extern "C" void launch_quant_fp8_fixed(const void* in, void* out, long n,
                                       float scale, hipStream_t stream)
{
    dim3 block(256), grid((unsigned)((n / 8 + 255) / 256));
    hipLaunchKernelGGL(k_quant_fp8_fixed, grid, block, 0, stream,
                       (const __hip_bfloat16*)in, (unsigned char*)out, n,
                       scale);
}

// Dynamic-scale variant: the multiplier (448/amax) is a DEVICE scalar so
// the amax -> quantize chain never syncs the host.
__global__ __launch_bounds__(256) void k_quant_fp8_dyn(
    const __hip_bfloat16* __restrict__ in, unsigned char* __restrict__ out,
    long n, const float* __restrict__ mult)
{
    const float scale = *mult;
    const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
    if (i0 + 8 <= n) {
        const bf16x8 v = *(const bf16x8*)(in + i0);
        unsigned long long pk = 0;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const __hip_fp8_e4m3 q((float)v[j] * scale);
            pk |= (unsigned long long)q.__x << (8 * j);
        }
        *(unsigned long long*)(out + i0) = pk;
    } else {
        for (long i = i0; i < n; ++i) {
            const __hip_fp8_e4m3 q(__bfloat162float(in[i]) * scale);
            out[i] = q.__x;
        }
    }
}

extern "C" void launch_quant_fp8_dyn(const void* in, void* out, long n,
                                     const float* mult, hipStream_t stream)
{
    dim3 block(256), grid((unsigned)((n / 8 + 255) / 256));
    hipLaunchKernelGGL(k_quant_fp8_dyn, grid, block, 0, stream,
                       (const __hip_bfloat16*)in, (unsigned char*)out, n,
                       mult);
}

// ---------------------------------------------------------------------------
// Fused fp8 wgrad pack [DEFERRED r1 item 2]: ONE pass over the (R, C)
// dgates producing, together,
//   out8  (R, C) e4m3 row-major      (dx GEMM operand)
//   out8T (C, R) e4m3 row-major      (dW GEMM operand, transposed)
//   partials (R/512, C) fp32         (column-sum chunks -> db via
//                                     k_colsum_final, deterministic)
// replacing the measured-53-ms chain {amax reuse + k_quant_fp8_dyn (17 GB
// re-read) + 2x k_quant_fp8_dyn_T (64-B transposed writes)} with an
// ~34 GB single pass whose transposed writes are 256-B contiguous.
// Block = 512 rows x 128 cols; per thread 4 sub-blocks of 8x8 (register
// transpose); LDS stages the transposed fp8 image [128][512] (+16-B pad).
// Requires R % 512 == 0 && C % 128 == 0 (caller falls back otherwise).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_quant_fp8_pack(
    const __hip_bfloat16* __restrict__ in, unsigned char* __restrict__ out8,
    unsigned char* __restrict__ out8T, float* __restrict__ partials,
    long R, long C, const float* __restrict__ mult)
{
    constexpr int LDT = 528;                   // 512 + 16-B pad, 16-B aligned
    // 76 KiB > the 64 KiB static-LDS cap -> dynamic shared, ONE object
    extern __shared__ __attribute__((aligned(16))) char smem[];
    unsigned char* const tr = (unsigned char*)smem;         // [128][LDT]
    float* const psum = (float*)(smem + 128 * LDT);         // [128][17]
    const float scale = *mult;
    const int tid = threadIdx.x;
    const long r0 = (long)blockIdx.y * 512;
    const long c0 = (long)blockIdx.x * 128;
    const int c8 = (tid & 15) * 8;             // this thread's 8 columns
    const int rgrp = tid >> 4;                 // 16 row groups
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
        const int r8 = (rgrp * 4 + s) * 8;     // sub-block top row
        unsigned long long rowpk[8];
#pragma unroll
        for (int rr = 0; rr < 8; ++rr) {
            const bf16x8 v = *(const bf16x8*)(in + (r0 + r8 + rr) * C + c0 + c8);
            unsigned long long pk = 0;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const float fv = (float)v[j];
                acc[j] += fv;
                const __hip_fp8_e4m3 q(fv * scale);
                pk |= (unsigned long long)q.__x << (8 * j);
            }
            rowpk[rr] = pk;
            *(unsigned long long*)(out8 + (r0 + r8 + rr) * C + c0 + c8) = pk;
        }
        // 8x8 byte transpose in registers -> 8 column packs
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            unsigned long long cp = 0;
#pragma unroll
            for (int rr = 0; rr < 8; ++rr)
                cp |= ((rowpk[rr] >> (8 * j)) & 0xffull) << (8 * rr);
            *(unsigned long long*)(&tr[(c8 + j) * LDT + r8]) = cp;
        }
    }
    // block-level deterministic column partial
#pragma unroll
    for (int j = 0; j < 8; ++j) psum[(c8 + j) * 17 + rgrp] = acc[j];
    __syncthreads();
    if (tid < 128) {
        float s = 0.f;
#pragma unroll
        for (int gq = 0; gq < 16; ++gq) s += psum[tid * 17 + gq];
        partials[(long)blockIdx.y * C + c0 + tid] = s;
    }
    // transposed drain: 256-B contiguous runs per thread
    const int tc = tid >> 1, half = tid & 1;
    const unsigned char* src = &tr[tc * LDT + half * 256];
    unsigned char* dst = out8T + (c0 + tc) * R + r0 + half * 256;
#pragma unroll
    for (int k = 0; k < 16; ++k)
        *(f32x4*)(dst + k * 16) = *(const f32x4*)(src + k * 16);
}

extern "C" int launch_quant_fp8_pack(const void* in, void* out8, void* out8T,
                                     float* partials, long R, long C,
                                     const float* mult, hipStream_t stream)
{
    if (R % 512 != 0 || C % 128 != 0) return -1;
    dim3 block(256), grid((unsigned)(C / 128), (unsigned)(R / 512));
    const size_t smem = 128 * 528 + 128 * 17 * sizeof(float);
    static bool attr_done = false;
    if (!attr_done) {
        hipFuncSetAttribute((const void*)k_quant_fp8_pack,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            (int)smem);
        attr_done = true;
    }
    hipLaunchKernelGGL(k_quant_fp8_pack, grid, block, smem, stream,
                       (const __hip_bfloat16*)in, (unsigned char*)out8,
                       (unsigned char*)out8T, partials, R, C, mult);
    return 0;
}

// Transposing quantizer: in (R, C) bf16 row-major -> out (C, R) fp8
// row-major, via a 64x64 LDS tile (fp8 in LDS, +8 pad kills write/read
// bank conflicts).  Feeds _scaled_mm's layout rules for the dW GEMMs
// (mat1 row-major (C, R), built from a row-major (R, C) gradient/input).
__global__ __launch_bounds__(256) void k_quant_fp8_dyn_T(
    const __hip_bfloat16* __restrict__ in, unsigned char* __restrict__ out,
    long R, long C, const float* __restrict__ mult)
{
    const float scale = *mult;
    __shared__ unsigned char tile[64][72];
    const long r0 = (long)blockIdx.y * 64;
    const long c0 = (long)blockIdx.x * 64;
    const int tid = threadIdx.x;
    // load+quantize: 8 rows per iteration, 8 cols per thread
    const int lr = tid >> 3, lc = (tid & 7) * 8;   // 32 rows x 64 cols / iter
#pragma unroll
    for (int it = 0; it < 2; ++it) {
        const long r = r0 + it * 32 + lr;
        if (r < R && c0 + lc < C) {
            const bf16x8 v = *(const bf16x8*)(in + r * C + c0 + lc);
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                const __hip_fp8_e4m3 q((float)v[j] * scale);
                tile[lc + j][it * 32 + lr] = q.__x;
            }
        }
    }
    __syncthreads();
    // store transposed: out row = column c, 8 consecutive R values/thread
    const int oc = tid >> 3, orr = (tid & 7) * 8;  // 32 out-rows x 64 / iter
#pragma unroll
    for (int it = 0; it < 2; ++it) {
        const long c = c0 + it * 32 + oc;
        const long r = r0 + orr;
        if (c < C && r + 8 <= R + (R & 7)) {
            unsigned long long pk = 0;
#pragma unroll
            for (int j = 0; j < 8; ++j)
                pk |= (unsigned long long)tile[it * 32 + oc][orr + j] << (8 * j);
            if (r + 8 <= R)
                *(unsigned long long*)(out + c * R + r) = pk;
            else
                for (int j = 0; j < 8 && r + j < R; ++j)
                    out[c * R + r + j] = tile[it * 32 + oc][orr + j];
        }
    }
}

extern "C" void launch_quant_fp8_dyn_T(const void* in, void* out, long R,
                                       long C, const float* mult,
                                       hipStream_t stream)
{
    dim3 block(256), grid((unsigned)((C + 63) / 64), (unsigned)((R + 63) / 64));
    hipLaunchKernelGGL(k_quant_fp8_dyn_T, grid, block, 0, stream,
                       (const __hip_bfloat16*)in, (unsigned char*)out, R, C,
                       mult);
}

// ---------------------------------------------------------------------------
// K4: fused pointwise backward for one timestep (elementwise over B x H).
// Produces raw-gate grads dgates_t (bf16) and the fp32 cell-grad carry
// IN PLACE (dc reads+writes the same (row,col)).
// dh_total = dh_up[t] + dh_carry (either may be null = zero).
// Vectorized W=8 per lane (dwordx4 stores) when H % 8 == 0, else W=4.
// ---------------------------------------------------------------------------
template <int W>
__global__ __launch_bounds__(256) void k_lstm_pw_bwd(
    const __hip_bfloat16* __restrict__ dh_up, long up_stride,   // nullable
    const __hip_bfloat16* __restrict__ dh_carry, long hc_stride, // nullable
    const __hip_bfloat16* __restrict__ acts, long act_stride,   // (B,4H)
    const float* __restrict__ c_cur, long cc_stride,
    const float* __restrict__ c_prev, long cp_stride,
    __hip_bfloat16* __restrict__ dgates, long dg_stride,        // (B,4H)
    float* __restrict__ dc, long dc_stride,                     // in-place carry
    int B, int H, int first_step /* dc uninitialised: treat as dcT or 0 */,
    const float* __restrict__ dcT, long dcT_stride)             // nullable
{
    typedef __bf16 bfv __attribute__((ext_vector_type(W)));
    typedef float fv __attribute__((ext_vector_type(W)));
    const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const int cpr = H / W;
    const int row = (int)(idx / cpr);
    if (row >= B) return;
    const int col0 = (int)(idx % cpr) * W;

    fv dh;
#pragma unroll
    for (int j = 0; j < W; ++j) dh[j] = 0.f;
    if (dh_up) {
        bfv v = *(const bfv*)(dh_up + (long)row * up_stride + col0);
#pragma unroll
        for (int j = 0; j < W; ++j) dh[j] += (float)v[j];
    }
    if (dh_carry) {
        bfv v = *(const bfv*)(dh_carry + (long)row * hc_stride + col0);
#pragma unroll
        for (int j = 0; j < W; ++j) dh[j] += (float)v[j];
    }
    fv dcv;
    if (first_step) {
        if (dcT) dcv = *(const fv*)(dcT + (long)row * dcT_stride + col0);
        else
#pragma unroll
            for (int j = 0; j < W; ++j) dcv[j] = 0.f;
    } else {
        dcv = *(const fv*)(dc + (long)row * dc_stride + col0);
    }
    const fv cc = *(const fv*)(c_cur + (long)row * cc_stride + col0);
    const fv cp = *(const fv*)(c_prev + (long)row * cp_stride + col0);
    const __hip_bfloat16* ap = acts + (long)row * act_stride + col0;
    bfv vi = *(const bfv*)(ap);
    bfv vf = *(const bfv*)(ap + (long)H);
    bfv vg = *(const bfv*)(ap + (long)2 * H);
    bfv vo = *(const bfv*)(ap + (long)3 * H);

    bfv out_i, out_f, out_g, out_o;
    fv dc_out;
#pragma unroll
    for (int j = 0; j < W; ++j) {
        const float i = (float)vi[j], f = (float)vf[j];
        const float g = (float)vg[j], o = (float)vo[j];
        const float tc = tanhf_(cc[j]);
        const float d_o = dh[j] * tc;
        float d_c = dcv[j] + dh[j] * o * (1.0f - tc * tc);
        const float d_i = d_c * g;
        const float d_g = d_c * i;
        const float d_f = d_c * cp[j];
        dc_out[j] = d_c * f;
        out_i[j] = (__bf16)(d_i * i * (1.0f - i));
        out_f[j] = (__bf16)(d_f * f * (1.0f - f));
        out_g[j] = (__bf16)(d_g * (1.0f - g * g));
        out_o[j] = (__bf16)(d_o * o * (1.0f - o));
    }
    __hip_bfloat16* dgp = dgates + (long)row * dg_stride + col0;
    *(bfv*)(dgp)               = out_i;
    *(bfv*)(dgp + (long)H)     = out_f;
    *(bfv*)(dgp + (long)2 * H) = out_g;
    *(bfv*)(dgp + (long)3 * H) = out_o;
    *(fv*)(dc + (long)row * dc_stride + col0) = dc_out;
}

// ---------------------------------------------------------------------------
// K6: global-norm clip.  Per-tensor partial sum-of-squares -> atomicAdd into
// a single fp32 accumulator; then scale-if-needed reading the final norm.
// ---------------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(256) void k_l2norm_partial(
    const T* __restrict__ x, long n, float* __restrict__ out_sumsq)
{
    __shared__ float red[4];  // one partial per wave
    float s = 0.f;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const float v = (float)x[i];
        s += v * v;
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off, 64);
    const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
    if (lane == 0) red[wave] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = red[0] + red[1] + red[2] + red[3];
        atomicAdd(out_sumsq, t);
    }
}

template <typename T>
__global__ __launch_bounds__(256) void k_clip_scale(
    T* __restrict__ x, long n, const float* __restrict__ norm, float max_norm)
{
    const float nv = *norm;
    const float coef = max_norm / (nv + 1e-6f);
    if (coef >= 1.0f) return;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x)
        x[i] = (T)((float)x[i] * coef);
}

// ---------------------------------------------------------------------------
// K7: fused optimizer apply (per tensor; fp32 master weights).
// Math matches ops/optim.py exactly (the CPU oracle).
// ---------------------------------------------------------------------------
template <typename PT>
__global__ __launch_bounds__(256) void k_sgd(
    PT* __restrict__ p, float* __restrict__ m, const PT* __restrict__ g,
    float* __restrict__ buf /* null if momentum==0 */, long n,
    float lr, float mom, float wd)
{
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        float gv = (float)g[i];
        float mv = m[i];
        if (wd != 0.f) gv += wd * mv;
        if (buf) {
            const float b = buf[i] * mom + gv;
            buf[i] = b;
            gv = b;
        }
        mv -= lr * gv;
        m[i] = mv;
        p[i] = (PT)mv;
    }
}

template <typename PT>
__global__ __launch_bounds__(256) void k_adam(
    PT* __restrict__ p, float* __restrict__ m, const PT* __restrict__ g,
    float* __restrict__ ea, float* __restrict__ eas, long n,
    float lr, float b1, float b2, float eps, float wd, float bc1, float bc2)
{
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        float gv = (float)g[i];
        float mv = m[i];
        if (wd != 0.f) gv += wd * mv;
        const float a1 = ea[i] * b1 + (1.f - b1) * gv;
        const float a2 = eas[i] * b2 + (1.f - b2) * gv * gv;
        ea[i] = a1;
        eas[i] = a2;
        const float denom = sqrtf(a2 / bc2) + eps;
        mv -= lr * (a1 / bc1) / denom;
        m[i] = mv;
        p[i] = (PT)mv;
    }
}

// ---------------------------------------------------------------------------
// C-linkage launchers (called from bindings.cpp)
// ---------------------------------------------------------------------------
extern "C" {

// variant encoding: (v & 3) = STAGE {0 direct, 1 LDS-A, 2 all-glds, 3
// counted-vmcnt}, bit2 = BM32 (else BM64), bit3 = XCD_SWZ, bit4 = BM128
// (overrides bit2; halves per-output W_h panel L2 traffic vs BM64)
void launch_lstm_rec_fwd_v(
    const void* h_prev, long hp_stride, const void* w_t,
    const void* pre, long pre_stride, const float* c_prev, long cp_stride,
    void* h_out, long ho_stride, float* c_out, long co_stride,
    void* acts, long act_stride, int B, int H, int raw_debug, int variant,
    hipStream_t stream)
{
    // overlap path (bit12): 128x256 BK=32, 256 threads, 2 blocks/CU
    if ((variant & 4096) && (B % 128 == 0)) {
        const size_t smem = 3 * (128 * 64 + 256 * 64);  // 72 KiB
        dim3 grid(B / 128, H / 64), block(256);
#define OVLAUNCH_T(RAW_, SW_, TN_)                                            \
    do {                                                                      \
        hipFuncSetAttribute((const void*)(k_lstm_rec_fwd_ov<RAW_, SW_, TN_>), \
                            hipFuncAttributeMaxDynamicSharedMemorySize,       \
                            (int)smem);                                       \
        hipLaunchKernelGGL((k_lstm_rec_fwd_ov<RAW_, SW_, TN_>), grid, block,  \
                           smem, stream,                                      \
            (const __hip_bfloat16*)h_prev, hp_stride,                         \
            (const __hip_bfloat16*)w_t, (const __hip_bfloat16*)pre,           \
            pre_stride, c_prev, cp_stride, (__hip_bfloat16*)h_out, ho_stride, \
            c_out, co_stride, (__hip_bfloat16*)acts, act_stride, B, H);       \
    } while (0)
#define OVLAUNCH(RAW_, SW_)                                                   \
    do {                                                                      \
        if (variant & 1) OVLAUNCH_T(RAW_, SW_, 1);                            \
        else OVLAUNCH_T(RAW_, SW_, 0);                                        \
    } while (0)
        if (raw_debug) {
            if (variant & 8) OVLAUNCH(true, true); else OVLAUNCH(true, false);
        } else {
            if (variant & 8) OVLAUNCH(false, true); else OVLAUNCH(false, false);
        }
#undef OVLAUNCH
#undef OVLAUNCH_T
        return;
    }
    // 2-phase merged path (bit10): BMx256 BK=64, 32-MFMA segments;
    // bit11 selects the BM=128 row tile (grids that under-fill at BM=256)
    if ((variant & 1024)
        && (B % ((variant & 2048) ? 128 : 256) == 0)) {
        const int bm = (variant & 2048) ? 128 : 256;
        const size_t smem = (size_t)2 * (bm * 128 + 32768);
        dim3 grid(B / bm, H / 64), block(512);
#define P2LAUNCH(RAW_, SW_, TN_, BM_)                                         \
    do {                                                                      \
        hipFuncSetAttribute(                                                  \
            (const void*)(k_lstm_rec_fwd_2ph<RAW_, SW_, TN_, BM_>),           \
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem);           \
        hipLaunchKernelGGL((k_lstm_rec_fwd_2ph<RAW_, SW_, TN_, BM_>), grid,   \
                           block, smem, stream,                               \
            (const __hip_bfloat16*)h_prev, hp_stride,                         \
            (const __hip_bfloat16*)w_t, (const __hip_bfloat16*)pre,           \
            pre_stride, c_prev, cp_stride, (__hip_bfloat16*)h_out, ho_stride, \
            c_out, co_stride, (__hip_bfloat16*)acts, act_stride, B, H);       \
    } while (0)
#define P2DISPATCH(RAW_, SW_, BM_)                                            \
    switch (variant & 3) {                                                    \
        case 1: P2LAUNCH(RAW_, SW_, 1, BM_); break;                           \
        case 2: P2LAUNCH(RAW_, SW_, 2, BM_); break;                           \
        case 3: P2LAUNCH(RAW_, SW_, 3, BM_); break;                           \
        default: P2LAUNCH(RAW_, SW_, 0, BM_); break;                          \
    }
#define P2DISPATCH2(RAW_, SW_)                                                \
    do {                                                                      \
        if (variant & 2048) { P2DISPATCH(RAW_, SW_, 128); }                   \
        else { P2DISPATCH(RAW_, SW_, 256); }                                  \
    } while (0)
        if (raw_debug) {
            if (variant & 8) { P2DISPATCH2(true, true); }
            else { P2DISPATCH2(true, false); }
        } else {
            if (variant & 8) { P2DISPATCH2(false, true); }
            else { P2DISPATCH2(false, false); }
        }
#undef P2DISPATCH2
#undef P2DISPATCH
#undef P2LAUNCH
        return;
    }
    // 8-phase path (bit9): 256x256 tile, BK=64 counted-vmcnt glds pipeline
    // (guide §5 template), requires B % 256 == 0.
    if ((variant & 512) && (B % 256 == 0)) {
        const size_t smem = 131072;  // 2 buffers x (A 32K + B 32K)
        dim3 grid(B / 256, H / 64), block(512);
#define P8LAUNCH(RAW_, SW_, ABL_)                                             \
    do {                                                                      \
        hipFuncSetAttribute((const void*)(k_lstm_rec_fwd_8p<RAW_, SW_, ABL_>),\
                            hipFuncAttributeMaxDynamicSharedMemorySize,       \
                            (int)smem);                                       \
        hipLaunchKernelGGL((k_lstm_rec_fwd_8p<RAW_, SW_, ABL_>), grid, block, \
                           smem, stream,                                      \
            (const __hip_bfloat16*)h_prev, hp_stride,                         \
            (const __hip_bfloat16*)w_t, (const __hip_bfloat16*)pre,           \
            pre_stride, c_prev, cp_stride, (__hip_bfloat16*)h_out, ho_stride, \
            c_out, co_stride, (__hip_bfloat16*)acts, act_stride, B, H);       \
    } while (0)
        if ((variant & 192) == 192) { P8LAUNCH(true, false, 3); return; }
        if (variant & 64) { P8LAUNCH(true, false, 1); return; }   // no-stage
        if (variant & 128) { P8LAUNCH(true, false, 2); return; }  // no-MFMA
        if (variant & 256) { P8LAUNCH(true, false, 8); return; }  // no-barrier
        if (variant & 4) {                                        // no-setprio
            if (raw_debug) P8LAUNCH(true, false, 4);
            else P8LAUNCH(false, false, 4);
            return;
        }
        if (raw_debug) {
            if (variant & 8) P8LAUNCH(true, true, 0);
            else P8LAUNCH(true, false, 0);
        } else {
            if (variant & 8) P8LAUNCH(false, true, 0);
            else P8LAUNCH(false, false, 0);
        }
#undef P8LAUNCH
        return;
    }
    // MT256 path (bit5): 256x256 tile, 512 threads, requires B % 256 == 0.
    if ((variant & 32) && (B % 256 == 0)) {
        const size_t smem = (variant & 1)
            ? (size_t)(2 * 2 * 256 * 64 * 2)                 // 128 KiB (STYLE1)
            : 3 * 2 * (size_t)(256 * 32 * 2);                // 96 KiB
        static bool attr_done[4] = {false, false, false, false};
        const int ai = (variant & 8 ? 1 : 0) | (raw_debug ? 2 : 0);
        dim3 grid(B / 256, H / 64), block(512);
#define MTLAUNCH4(RAW_, SW_, ABL_, ST_)                                       \
    do {                                                                      \
        hipError_t e_ = hipFuncSetAttribute(                                  \
            (const void*)(k_lstm_rec_fwd_mt256<RAW_, SW_, ABL_, ST_>),        \
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem);           \
        (void)e_;                                                             \
        hipLaunchKernelGGL((k_lstm_rec_fwd_mt256<RAW_, SW_, ABL_, ST_>),      \
                           grid, block, smem, stream,                         \
            (const __hip_bfloat16*)h_prev, hp_stride,                         \
            (const __hip_bfloat16*)w_t, (const __hip_bfloat16*)pre,           \
            pre_stride, c_prev, cp_stride, (__hip_bfloat16*)h_out, ho_stride, \
            c_out, co_stride, (__hip_bfloat16*)acts, act_stride, B, H);       \
    } while (0)
#define MTLAUNCH3(RAW_, SW_, ABL_)                                            \
    do {                                                                      \
        hipError_t e_ = hipFuncSetAttribute(                                  \
            (const void*)(k_lstm_rec_fwd_mt256<RAW_, SW_, ABL_>),             \
            hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem);           \
        (void)e_;                                                             \
        hipLaunchKernelGGL((k_lstm_rec_fwd_mt256<RAW_, SW_, ABL_>), grid,     \
                           block, smem, stream,                               \
            (const __hip_bfloat16*)h_prev, hp_stride,                         \
            (const __hip_bfloat16*)w_t, (const __hip_bfloat16*)pre,           \
            pre_stride, c_prev, cp_stride, (__hip_bfloat16*)h_out, ho_stride, \
            c_out, co_stride, (__hip_bfloat16*)acts, act_stride, B, H);       \
    } while (0)
#define MTLAUNCH(RAW_, SW_)                                                   \
    do {                                                                      \
        if (!attr_done[ai]) {                                                 \
            hipError_t e2_ = hipFuncSetAttribute(                             \
                (const void*)(k_lstm_rec_fwd_mt256<RAW_, SW_>),               \
                hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem);       \
            (void)e2_;                                                        \
            attr_done[ai] = true;                                             \
        }                                                                     \
        hipLaunchKernelGGL((k_lstm_rec_fwd_mt256<RAW_, SW_>), grid, block,    \
                           smem, stream,                                      \
            (const __hip_bfloat16*)h_prev, hp_stride,                         \
            (const __hip_bfloat16*)w_t, (const __hip_bfloat16*)pre,           \
            pre_stride, c_prev, cp_stride, (__hip_bfloat16*)h_out, ho_stride, \
            c_out, co_stride, (__hip_bfloat16*)acts, act_stride, B, H);       \
    } while (0)
        if (variant & 64) { MTLAUNCH3(true, false, 1); return; }    // no-stage
        if (variant & 128) { MTLAUNCH3(true, false, 2); return; }   // no-MFMA
        if (variant & 1) {  // STYLE1: 2-buffer BK=64 (needs 128 KiB LDS)
            if (raw_debug) {
                if (variant & 8) MTLAUNCH4(true, true, 0, 1);
                else MTLAUNCH4(true, false, 0, 1);
            } else {
                if (variant & 8) MTLAUNCH4(false, true, 0, 1);
                else MTLAUNCH4(false, false, 0, 1);
            }
            return;
        }
        if (raw_debug) {
            if (variant & 8) MTLAUNCH(true, true); else MTLAUNCH(true, false);
        } else {
            if (variant & 8) MTLAUNCH(false, true); else MTLAUNCH(false, false);
        }
#undef MTLAUNCH
#undef MTLAUNCH3
#undef MTLAUNCH4
        return;
    }
    // STAGE3 requires BM >= 64: its staging writes full 4 KiB glds slices,
    // which would overrun the 2 KiB A image at BM=32 (corrupts B tile).
    if ((variant & 7) == 7) variant &= ~1;  // BM32+STAGE3 -> BM32+STAGE2
    const int BM = (variant & 16) ? 128 : ((variant & 4) ? 32 : 64);
    const int stage = variant & 3;
    dim3 grid((B + BM - 1) / BM, H / 64);
    dim3 block(256);
    size_t smem = 0;
    if (stage == 2) smem = (size_t)(2 * BM * 128 + 2 * 32768);
    if (stage == 3) smem = (size_t)(3 * (BM * 64 + 16384));
#define ARGS                                                                  \
    (const __hip_bfloat16*)h_prev, hp_stride, (const __hip_bfloat16*)w_t,     \
    (const __hip_bfloat16*)pre, pre_stride, c_prev, cp_stride,                \
    (__hip_bfloat16*)h_out, ho_stride, c_out, co_stride,                      \
    (__hip_bfloat16*)acts, act_stride, B, H
#define LAUNCH(BM_, RAW_, ST_, SW_)                                           \
    do {                                                                      \
        if (ST_ == 2 || ST_ == 3) {                                           \
            static bool attr_done_##BM_##RAW_##SW_ = false;                   \
            if (!attr_done_##BM_##RAW_##SW_ && smem > 64 * 1024) {            \
                hipFuncSetAttribute(                                          \
                    (const void*)(k_lstm_rec_fwd<BM_, RAW_, ST_, SW_>),       \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)smem);   \
                attr_done_##BM_##RAW_##SW_ = true;                            \
            }                                                                 \
        }                                                                     \
        hipLaunchKernelGGL((k_lstm_rec_fwd<BM_, RAW_, ST_, SW_>), grid,       \
                           block, smem, stream, ARGS);                        \
    } while (0)
#define DISPATCH(RAW_)                                                        \
    switch (variant & 31) {                                                   \
        case 18: LAUNCH(128, RAW_, 2, false); break;                          \
        case 19: LAUNCH(128, RAW_, 3, false); break;                          \
        case 26: LAUNCH(128, RAW_, 2, true); break;                           \
        case 27: LAUNCH(128, RAW_, 3, true); break;                           \
        case 0: LAUNCH(64, RAW_, 0, false); break;                            \
        case 1: LAUNCH(64, RAW_, 1, false); break;                            \
        case 2: LAUNCH(64, RAW_, 2, false); break;                            \
        case 3: LAUNCH(64, RAW_, 3, false); break;                            \
        case 4: LAUNCH(32, RAW_, 0, false); break;                            \
        case 5: LAUNCH(32, RAW_, 1, false); break;                            \
        case 6: LAUNCH(32, RAW_, 2, false); break;                            \
        case 7: LAUNCH(32, RAW_, 3, false); break;                            \
        case 8: LAUNCH(64, RAW_, 0, true); break;                             \
        case 9: LAUNCH(64, RAW_, 1, true); break;                             \
        case 10: LAUNCH(64, RAW_, 2, true); break;                            \
        case 11: LAUNCH(64, RAW_, 3, true); break;                            \
        case 12: LAUNCH(32, RAW_, 0, true); break;                            \
        case 13: LAUNCH(32, RAW_, 1, true); break;                            \
        case 14: LAUNCH(32, RAW_, 2, true); break;                            \
        case 15: LAUNCH(32, RAW_, 3, true); break;                            \
        default: LAUNCH(64, RAW_, 2, true); break;                            \
    }
    if (raw_debug) { DISPATCH(true); } else { DISPATCH(false); }
#undef DISPATCH
#undef LAUNCH
#undef ARGS
}

static int g_fwd_variant = -1;  // -1 = auto: pick by batch size (kbench
// sweeps: B >= 2048 -> BM64+glds+swz (10); smaller -> BM32+glds+swz (14))

void set_fwd_variant(int v) { g_fwd_variant = v; }
int get_fwd_variant() { return g_fwd_variant; }

// Persistent whole-sequence forward.  Returns 0 if launched, -1 if the
// shape is ineligible (caller falls back to the per-step loop).
int launch_lstm_seq_fwd_persistent(
    void* h_all, const void* w_t, const void* pre, const float* c0,
    float* c_seq, void* acts, unsigned* cnt,
    int B, int H, int T, int reverse, hipStream_t stream)
{
    if (B % 256 != 0 || H % 64 != 0) return -1;
    const int nm = B / 256, nn = H / 16;
    if (nm * nn > 256) return -1;
    const size_t smem = (size_t)64 * (H + 8) * 2;
    if (smem > 160 * 1024) return -1;
    static size_t attr_set_for = 0;
    if (smem > 64 * 1024 && attr_set_for < smem) {
        if (hipFuncSetAttribute((const void*)k_lstm_seq_fwd_persistent,
                                hipFuncAttributeMaxDynamicSharedMemorySize,
                                (int)smem) != hipSuccess)
            return -1;
        attr_set_for = smem;
    }
    dim3 grid(nm, nn);  // m fastest: each XCD L2 caches one 256-row h panel
    dim3 block(256);
    hipLaunchKernelGGL(k_lstm_seq_fwd_persistent, grid, block, smem, stream,
        (__hip_bfloat16*)h_all, (const __hip_bfloat16*)w_t,
        (const __hip_bfloat16*)pre, c0, c_seq,
        (__hip_bfloat16*)acts, cnt, B, H, T, reverse);
    return 0;
}

void launch_lstm_rec_fwd(
    const void* h_prev, long hp_stride, const void* w_t,
    const void* pre, long pre_stride, const float* c_prev, long cp_stride,
    void* h_out, long ho_stride, float* c_out, long co_stride,
    void* acts, long act_stride, int B, int H, int raw_debug,
    hipStream_t stream)
{
    int v = g_fwd_variant;
    // auto: the 2ph kernel where its grid fills the chip and K=H is in
    // the measured-win regime (see bindings.cpp fill2ph).  The 4-round
    // interleaved A/B picked 2ph over the 2-blocks/CU ov variant
    // (median 200.0 vs 208.6 us/step at the headline shape); ov stays
    // as variants 4096/4097 for A/B.
    if (v < 0) v = (B % 256 == 0 && H <= 1536
                    && (B / 256) * (H / 64) >= 256)
                     ? 1027                        // 2ph TUNE3 (kbench r2)
                 : (B >= 2048) ? 10 : 14;
    launch_lstm_rec_fwd_v(h_prev, hp_stride, w_t, pre, pre_stride, c_prev,
                          cp_stride, h_out, ho_stride, c_out, co_stride, acts,
                          act_stride, B, H, raw_debug, v, stream);
}

void launch_lstm_pw_bwd(
    const void* dh_up, long up_stride, const void* dh_carry, long hc_stride,
    const void* acts, long act_stride, const float* c_cur, long cc_stride,
    const float* c_prev, long cp_stride, void* dgates, long dg_stride,
    float* dc, long dc_stride, int B, int H, int first_step,
    const float* dcT, long dcT_stride, hipStream_t stream)
{
    dim3 block(256);
    if (H % 8 == 0) {
        dim3 grid((unsigned)(((long)B * (H / 8) + 255) / 256));
        hipLaunchKernelGGL(k_lstm_pw_bwd<8>, grid, block, 0, stream,
            (const __hip_bfloat16*)dh_up, up_stride,
            (const __hip_bfloat16*)dh_carry, hc_stride,
            (const __hip_bfloat16*)acts, act_stride, c_cur, cc_stride,
            c_prev, cp_stride, (__hip_bfloat16*)dgates, dg_stride,
            dc, dc_stride, B, H, first_step, dcT, dcT_stride);
    } else {
        dim3 grid((unsigned)(((long)B * (H / 4) + 255) / 256));
        hipLaunchKernelGGL(k_lstm_pw_bwd<4>, grid, block, 0, stream,
            (const __hip_bfloat16*)dh_up, up_stride,
            (const __hip_bfloat16*)dh_carry, hc_stride,
            (const __hip_bfloat16*)acts, act_stride, c_cur, cc_stride,
            c_prev, cp_stride, (__hip_bfloat16*)dgates, dg_stride,
            dc, dc_stride, B, H, first_step, dcT, dcT_stride);
    }
}

static inline int norm_grid(long n) {
    long g = (n + 255) / 256;
    return (int)(g > 2048 ? 2048 : g);
}

void launch_l2norm(const void* x, long n, float* out, int is_bf16,
                   hipStream_t stream)
{
    dim3 grid(norm_grid(n)), block(256);
    if (is_bf16)
        hipLaunchKernelGGL(k_l2norm_partial<__hip_bfloat16>, grid, block, 0,
                           stream, (const __hip_bfloat16*)x, n, out);
    else
        hipLaunchKernelGGL(k_l2norm_partial<float>, grid, block, 0, stream,
                           (const float*)x, n, out);
}

void launch_clip_scale(void* x, long n, const float* norm, float max_norm,
                       int is_bf16, hipStream_t stream)
{
    dim3 grid(norm_grid(n)), block(256);
    if (is_bf16)
        hipLaunchKernelGGL(k_clip_scale<__hip_bfloat16>, grid, block, 0,
                           stream, (__hip_bfloat16*)x, n, norm, max_norm);
    else
        hipLaunchKernelGGL(k_clip_scale<float>, grid, block, 0, stream,
                           (float*)x, n, norm, max_norm);
}

void launch_sgd(void* p, float* m, const void* g, float* buf, long n,
                float lr, float mom, float wd, int is_bf16, hipStream_t stream)
{
    dim3 grid(norm_grid(n)), block(256);
    if (is_bf16)
        hipLaunchKernelGGL(k_sgd<__hip_bfloat16>, grid, block, 0, stream,
            (__hip_bfloat16*)p, m, (const __hip_bfloat16*)g, buf, n, lr, mom, wd);
    else
        hipLaunchKernelGGL(k_sgd<float>, grid, block, 0, stream,
            (float*)p, m, (const float*)g, buf, n, lr, mom, wd);
}

void launch_adam(void* p, float* m, const void* g, float* ea, float* eas,
                 long n, float lr, float b1, float b2, float eps, float wd,
                 float bc1, float bc2, int is_bf16, hipStream_t stream)
{
    dim3 grid(norm_grid(n)), block(256);
    if (is_bf16)
        hipLaunchKernelGGL(k_adam<__hip_bfloat16>, grid, block, 0, stream,
            (__hip_bfloat16*)p, m, (const __hip_bfloat16*)g, ea, eas, n,
            lr, b1, b2, eps, wd, bc1, bc2);
    else
        hipLaunchKernelGGL(k_adam<float>, grid, block, 0, stream,
            (float*)p, m, (const float*)g, ea, eas, n,
            lr, b1, b2, eps, wd, bc1, bc2);
}

}  // extern "C"

// ---------------------------------------------------------------------------
// K3: fused log-softmax + cross-entropy for the LM head (V up to 50k+).
// Forward: one workgroup per row, single online pass (running max + sum-exp
// with rescale), vectorized bf16x8 loads; writes per-row loss and lse.
// Backward: dlogits = (exp(l - lse) - onehot) * row_scale, grid-stride.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void k_xent_fwd(
    const __hip_bfloat16* __restrict__ logits,  // (N, V)
    const long* __restrict__ targets,           // (N,)
    float* __restrict__ loss,                   // (N,)
    float* __restrict__ lse,                    // (N,)
    long N, int V)
{
    const long row = blockIdx.x;
    if (row >= N) return;
    const __hip_bfloat16* lp = logits + row * V;
    const int tid = threadIdx.x;

    float m = -3.0e38f, s = 0.f;
    const int VV = V & ~7;  // vectorized body
    for (int v0 = tid * 8; v0 < VV; v0 += 256 * 8) {
        bf16x8 x = *(const bf16x8*)(lp + v0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const float xv = (float)x[j];
            if (xv > m) { s *= __expf(m - xv); m = xv; }
            s += __expf(xv - m);
        }
    }
    for (int v = VV + tid; v < V; v += 256) {
        const float xv = __bfloat162float(lp[v]);
        if (xv > m) { s *= __expf(m - xv); m = xv; }
        s += __expf(xv - m);
    }
    // wave reduce (running-max merge), then cross-wave via LDS
    __shared__ float red_m[4], red_s[4];
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        const float om = __shfl_down(m, off, 64);
        const float os = __shfl_down(s, off, 64);
        if (om > m) { s = s * __expf(m - om) + os; m = om; }
        else        { s = s + os * __expf(om - m); }
    }
    const int lane = tid & 63, wave = tid >> 6;
    if (lane == 0) { red_m[wave] = m; red_s[wave] = s; }
    __syncthreads();
    if (tid == 0) {
        float M = red_m[0], S = red_s[0];
#pragma unroll
        for (int w = 1; w < 4; ++w) {
            const float om = red_m[w], os = red_s[w];
            if (om > M) { S = S * __expf(M - om) + os; M = om; }
            else        { S = S + os * __expf(om - M); }
        }
        const float l = M + __logf(S);
        lse[row] = l;
        loss[row] = l - __bfloat162float(lp[targets[row]]);
    }
}

__global__ __launch_bounds__(256) void k_xent_bwd(
    const __hip_bfloat16* __restrict__ logits, const long* __restrict__ targets,
    const float* __restrict__ lse, __hip_bfloat16* __restrict__ dlogits,
    const float* __restrict__ gscale, float inv_n, long N, int V)
{
    const long row = blockIdx.y;
    if (row >= N) return;
    const float scale = (gscale ? *gscale : 1.0f) * inv_n;
    const float l = lse[row];
    const long tgt = targets[row];
    const __hip_bfloat16* lp = logits + row * V;
    __hip_bfloat16* dp = dlogits + row * V;
    for (int v0 = (blockIdx.x * 256 + threadIdx.x) * 8; v0 < V;
         v0 += gridDim.x * 256 * 8) {
        if (v0 + 8 <= V) {
            bf16x8 x = *(const bf16x8*)(lp + v0);
            bf16x8 o;
#pragma unroll
            for (int j = 0; j < 8; ++j) {
                float p = __expf((float)x[j] - l);
                if (v0 + j == tgt) p -= 1.0f;
                o[j] = (__bf16)(p * scale);
            }
            *(bf16x8*)(dp + v0) = o;
        } else {
            for (int v = v0; v < V; ++v) {
                float p = __expf(__bfloat162float(lp[v]) - l);
                if (v == tgt) p -= 1.0f;
                dp[v] = __float2bfloat16(p * scale);
            }
        }
    }
}

extern "C" {
void launch_xent_fwd(const void* logits, const long* targets, float* loss,
                     float* lse, long N, int V, hipStream_t stream)
{
    dim3 grid((unsigned)N), block(256);
    hipLaunchKernelGGL(k_xent_fwd, grid, block, 0, stream,
                       (const __hip_bfloat16*)logits, targets, loss, lse, N, V);
}
void launch_xent_bwd(const void* logits, const long* targets, const float* lse,
                     void* dlogits, const float* gscale, float inv_n, long N,
                     int V, hipStream_t stream)
{
    int gx = (V / 8 + 255) / 256;
    gx = gx > 16 ? 16 : (gx < 1 ? 1 : gx);
    dim3 grid(gx, (unsigned)N), block(256);
    hipLaunchKernelGGL(k_xent_bwd, grid, block, 0, stream,
                       (const __hip_bfloat16*)logits, targets, lse,
                       (__hip_bfloat16*)dlogits, gscale, inv_n, N, V);
}
}  // extern "C"
