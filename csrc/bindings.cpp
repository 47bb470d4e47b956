// C++ sequence drivers + pybind bindings for the MI355X LSTM kernels.
//
// The time loops run HERE (one Python call per layer per direction), not in
// Python: at ~3.5us host launch overhead per kernel, a 128-step recurrent
// loop must not round-trip through the interpreter per step (SURVEY.md §7
// hard part #1).  Library GEMMs (recurrent dh, and the batched dW/dx in
// ops/lstm_layer.py) go through rocBLAS/hipBLASLt via at::mm.

#include <torch/extension.h>
#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <set>
#include <vector>

extern "C" {
void launch_lstm_rec_fwd(const void*, long, const void*, const void*, long,
                         const float*, long, void*, long, float*, long,
                         void*, long, int, int, int, hipStream_t);
void launch_lstm_pw_bwd(const void*, long, const void*, long, const void*,
                        long, const float*, long, const float*, long, void*,
                        long, float*, long, int, int, int, const float*, long,
                        hipStream_t);
void launch_l2norm(const void*, long, float*, int, hipStream_t);
void launch_clip_scale(void*, long, const float*, float, int, hipStream_t);
void launch_sgd(void*, float*, const void*, float*, long, float, float, float,
                int, hipStream_t);
void launch_adam(void*, float*, const void*, float*, float*, long, float,
                 float, float, float, float, float, float, int, hipStream_t);
void set_fwd_variant(int);
int get_fwd_variant();
void launch_lstm_pw_fwd(const void*, long, const void*, long, const float*,
                        long, void*, long, float*, long, void*, long, int,
                        int, hipStream_t);
void launch_quant_fp8_fixed(const void*, void*, long, float, hipStream_t);
void launch_quant_fp8_dyn(const void*, void*, long, const float*, hipStream_t);
int launch_gemv_bf16(const void*, const void*, const float*, void*, float*,
                     int, long, int, hipStream_t);
void set_gemv_nt(int);
int launch_colsum_bf16(const void*, float*, float*, long, long, hipStream_t);
int colsum_gy(long, long);
void launch_quant_fp8_dyn_T(const void*, void*, long, long, const float*,
                            hipStream_t);
int launch_quant_fp8_pack(const void*, void*, void*, float*, long, long,
                          const float*, hipStream_t);
void launch_colsum_final(const float*, float*, long, int, hipStream_t);
void launch_xent_fwd(const void*, const long*, float*, float*, long, int,
                     hipStream_t);
void launch_xent_bwd(const void*, const long*, const float*, void*,
                     const float*, float, long, int, hipStream_t);
int launch_lstm_seq_fwd_persistent(void*, const void*, const void*,
                                   const float*, float*, void*,
                                   unsigned*, int, int, int, int, hipStream_t);
int launch_lstm_dh_pw_bwd(const void*, long, const void*, long, const void*,
                          long, const void*, long, const float*, long,
                          const float*, long, void*, long, float*, long,
                          void*, long, float*, int, int, int, int,
                          hipStream_t);
int launch_lstm_dh_pw_bwd_ov(const void*, long, const void*, long,
                             const void*, long, const void*, long,
                             const float*, long, const float*, long, void*,
                             long, float*, long, void*, long, int, int, int,
                             int, hipStream_t);
}

namespace {

using at::Tensor;

bool g_persistent_enabled = false;  // per-step glds path is faster (kbench6)
// Split forward: hipBLASLt recurrent GEMM + k_lstm_pw_fwd pointwise.
// -1 = auto (r2): the FUSED 2ph hand kernel where its grid fills the chip
// and H is in its measured-win regime (fill2ph in lstm_seq_fwd); split
// where B*H >= 2M otherwise; small-tile fused kernels below that.
// 0 forces fused, 1 forces split (tests/autotune).
int g_split_fwd = -1;
// Fused backward step: -1 auto (on where eligible AND H in the measured
// regime), 1 = force the 512-thread kernel on any eligible shape,
// 2 = force the 2-blocks/CU ov variant (A/B only: measured slower --
// docs/KERNELS.md), 0 = off (split hipBLASLt dh GEMM + k_lstm_pw_bwd).
int g_fused_bwd = -1;
// FP8 recurrent GEMM (experimental, MI355X 2x-bf16 MFMA rate): quantize
// h (tanh-bounded, FIXED scale 448) and W_h (per-call amax scale) to OCP
// e4m3 and run the per-step GEMM through hipBLASLt _scaled_mm (measured
// 1939 vs 963 TF/s on the (16384,1024)x(1024,4096) step shape).  Only the
// recurrent GEMM runs fp8; pre/activations/backward stay bf16.
bool g_fp8_fwd = false;
std::set<long> g_validated_grids;

hipStream_t cur_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}

void check_bf16_cuda(const Tensor& t, const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
    TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
}

// ---------------------------------------------------------------------
// Forward: whole-sequence driver.  TIME-MAJOR: pre (T,B,4H) bf16,
// w_h (H,4H) bf16, h0 (B,H) bf16, c0 (B,H) fp32
// -> (h_all (T+1,B,H), c_seq (T,B,H), acts (T,B,4H), hT, cT).
// h_all slot convention (ops/functional.py): fwd h_all[0]=h0,
// h_all[t+1]=h_t; rev h_all[T]=h0, h_all[t]=h_t — so h_{t-1} per step AND
// the layer output are both contiguous leading-dim slices (no cat/copies).
// ---------------------------------------------------------------------
std::vector<Tensor> lstm_seq_fwd(Tensor pre, Tensor w_h, Tensor h0, Tensor c0,
                                 bool reverse) {
    check_bf16_cuda(pre, "pre");
    check_bf16_cuda(w_h, "w_h");
    check_bf16_cuda(h0, "h0");
    TORCH_CHECK(c0.scalar_type() == at::kFloat && c0.is_cuda(), "c0 fp32 GPU");
    TORCH_CHECK(pre.dim() == 3, "pre must be (T,B,4H)");
    const long T = pre.size(0), B = pre.size(1), H4 = pre.size(2);
    const long H = H4 / 4;
    TORCH_CHECK(w_h.size(0) == H && w_h.size(1) == H4, "w_h must be (H,4H)");
    TORCH_CHECK(H % 64 == 0, "fused kernel requires H % 64 == 0, got ", H);
    pre = pre.contiguous();
    auto h0c = h0.contiguous();
    auto c0c = c0.contiguous();
    // Forward path choice: the hand-written fused 2ph kernel (256x256
    // tile) beats the split hipBLASLt+pointwise path where its grid fills
    // the 256-CU chip (measured 204 vs 258 us/step at B=16384 H=1024);
    // below that, split wins at B*H >= 2M and the BM64/BM32 fused kernels
    // win smaller still (round-1 sweeps).
    // The fused 256-col-tile kernels win where (a) their grid fills the
    // chip AND (b) K=H is small enough that hipBLASLt's deep-K advantage
    // doesn't apply (measured: fused wins H=1024 for B=4096..32768, ties
    // H=2048 at B=2048, LOSES H=8192 -- library GEMM ~1 PF/s at K>=2048).
    const bool fill2ph = (B % 256 == 0) && (H % 64 == 0) && (H <= 1536)
                         && (B / 256) * (H / 64) >= 256;
    // fp8 mode's e4m3 recurrent GEMM lives on the split path -- keep it
    // there (measured faster than the bf16 fused kernel at 2x MFMA rate)
    const bool split_auto = (B * H >= 2097152) && (!fill2ph || g_fp8_fwd);
    // W_h^T (4H, H): one transposed copy per layer call so each lane's MFMA
    // B-fragment is a contiguous 16-byte load (fused path only).
    const bool need_wt = g_persistent_enabled
        || !((g_split_fwd < 0) ? split_auto : (g_split_fwd != 0));
    auto w_t = need_wt ? w_h.t().contiguous() : w_h;

    auto opts_bf = pre.options();
    auto opts_f32 = pre.options().dtype(at::kFloat);
    auto h_all = at::empty({T + 1, B, H}, opts_bf);
    auto c_seq = at::empty({T, B, H}, opts_f32);
    auto acts = at::empty({T, B, H4}, opts_bf);
    h_all.select(0, reverse ? T : 0).copy_(h0c, /*non_blocking=*/true);

    auto* pre_p = (const char*)pre.data_ptr();
    auto* ha_p = (char*)h_all.data_ptr();
    auto* cs_p = (float*)c_seq.data_ptr();
    auto* act_p = (char*)acts.data_ptr();
    auto stream = cur_stream();

    // Persistent whole-sequence path (W_h resident in LDS across all T
    // steps, grid barrier per step).  First use of each grid shape is
    // validated synchronously against the bounded-spin timeout flag.
    if (g_persistent_enabled) {
        auto cnt = at::zeros({T + 1}, pre.options().dtype(at::kInt));
        int rc = launch_lstm_seq_fwd_persistent(
            h_all.data_ptr(), w_t.data_ptr(), pre.data_ptr(),
            (const float*)c0c.data_ptr(),
            (float*)c_seq.data_ptr(), acts.data_ptr(),
            (unsigned*)cnt.data_ptr(), (int)B, (int)H, (int)T, reverse ? 1 : 0,
            stream);
        if (rc == 0) {
            const long key = (B / 256) * 1000000 + H;
            if (!g_validated_grids.count(key)) {
                const int flag = cnt[T].item<int>();  // syncs once per shape
                TORCH_CHECK(flag == 0,
                            "persistent LSTM kernel timed out (grid not "
                            "resident?) B=", B, " H=", H, " T=", T);
                g_validated_grids.insert(key);
            }
            auto hT = h_all.select(0, reverse ? 0 : T).clone();
            auto cT = c_seq.select(0, reverse ? 0 : T - 1).clone();
            return {h_all, c_seq, acts, hT, cT};
        }
    }

    const bool split = (g_split_fwd < 0) ? split_auto : (g_split_fwd != 0);
    const bool fp8 = g_fp8_fwd && split;
    at::Tensor raw_buf, h8, w8_cm, sc_a, sc_b;
    if (split) raw_buf = at::empty({B, H4}, opts_bf);
    if (fp8) {
        auto opts_f8 = pre.options().dtype(at::kFloat8_e4m3fn);
        h8 = at::empty({B, H}, opts_f8);
        // W_h: per-call amax scale, quantized once; _scaled_mm wants the
        // second operand COLUMN-major
        auto w_amax = w_h.abs().amax().to(at::kFloat).clamp_min(1e-8);
        sc_b = w_amax / 448.0;
        w8_cm = at::mul(w_h, 448.0 / w_amax).to(at::kFloat8_e4m3fn)
                    .t().contiguous().t();
        sc_a = at::full({}, 1.0 / 448.0, pre.options().dtype(at::kFloat));
    }
    for (long k = 0; k < T; ++k) {
        const long t = reverse ? (T - 1 - k) : k;
        // prev slot holds h_{t-1} (h0 at the first iteration by the slot
        // convention); out slot receives h_t.
        const long slot_prev = reverse ? (t + 1) : t;
        const long slot_out = reverse ? t : (t + 1);
        const float* cp;
        if (k == 0) {
            cp = (const float*)c0c.data_ptr();
        } else {
            const long t_prev = reverse ? (t + 1) : (t - 1);
            cp = cs_p + t_prev * B * H;
        }
        if (fp8) {
            launch_quant_fp8_fixed(ha_p + 2 * slot_prev * B * H,
                                   h8.data_ptr(), B * H, 448.0f, stream);
            at::_scaled_mm_out(raw_buf, h8, w8_cm, sc_a, sc_b, {}, {},
                               at::kBFloat16, true);
            launch_lstm_pw_fwd(
                raw_buf.data_ptr(), H4,
                pre_p + 2 * t * B * H4, H4,
                cp, H,
                ha_p + 2 * slot_out * B * H, H,
                cs_p + t * B * H, H,
                act_p + 2 * t * B * H4, H4,
                (int)B, (int)H, stream);
        } else if (split) {
            at::mm_out(raw_buf, h_all.select(0, slot_prev), w_h);
            launch_lstm_pw_fwd(
                raw_buf.data_ptr(), H4,
                pre_p + 2 * t * B * H4, H4,
                cp, H,
                ha_p + 2 * slot_out * B * H, H,
                cs_p + t * B * H, H,
                act_p + 2 * t * B * H4, H4,
                (int)B, (int)H, stream);
        } else {
            launch_lstm_rec_fwd(
                ha_p + 2 * slot_prev * B * H, H, w_t.data_ptr(),
                pre_p + 2 * t * B * H4, H4,  // bf16 = 2 bytes
                cp, H,
                ha_p + 2 * slot_out * B * H, H,
                cs_p + t * B * H, H,
                act_p + 2 * t * B * H4, H4,
                (int)B, (int)H, /*raw_debug=*/0, stream);
        }
    }
    auto hT = h_all.select(0, reverse ? 0 : T).clone();
    auto cT = c_seq.select(0, reverse ? 0 : T - 1).clone();
    return {h_all, c_seq, acts, hT, cT};
}

// Single fused step with optional raw-gate debug output (numerics tests).
std::vector<Tensor> lstm_rec_fwd_step(Tensor h_prev, Tensor w_h, Tensor pre_t,
                                      Tensor c_prev, bool raw_debug) {
    check_bf16_cuda(h_prev, "h_prev");
    const long B = pre_t.size(0), H4 = pre_t.size(1), H = H4 / 4;
    TORCH_CHECK(H % 64 == 0, "H % 64 != 0");
    auto w_t = w_h.t().contiguous();
    auto hp = h_prev.contiguous();
    auto pr = pre_t.contiguous();
    auto cp = c_prev.contiguous();
    auto h_out = at::empty({B, H}, pre_t.options());
    auto c_out = at::empty({B, H}, pre_t.options().dtype(at::kFloat));
    auto acts = at::empty({B, H4}, pre_t.options());
    launch_lstm_rec_fwd(hp.data_ptr(), H, w_t.data_ptr(), pr.data_ptr(), H4,
                        (const float*)cp.data_ptr(), H, h_out.data_ptr(), H,
                        (float*)c_out.data_ptr(), H, acts.data_ptr(), H4,
                        (int)B, (int)H, raw_debug ? 1 : 0, cur_stream());
    return {h_out, c_out, acts};
}

// ---------------------------------------------------------------------
// Backward: whole-sequence driver.
// ---------------------------------------------------------------------
std::vector<Tensor> lstm_seq_bwd(c10::optional<Tensor> dh_seq_opt,
                                 c10::optional<Tensor> d_hT_opt,
                                 c10::optional<Tensor> d_cT_opt,
                                 Tensor w_h, Tensor acts, Tensor c_seq,
                                 Tensor c0, bool reverse) {
    // TIME-MAJOR: acts (T,B,4H), c_seq (T,B,H), dh_seq (T,B,H).
    check_bf16_cuda(acts, "acts");
    const long T = acts.size(0), B = acts.size(1), H4 = acts.size(2);
    const long H = H4 / 4;
    TORCH_CHECK(H % 4 == 0, "H % 4 != 0");
    acts = acts.contiguous();
    c_seq = c_seq.contiguous();
    auto c0c = c0.contiguous();
    Tensor dh_seq;
    const bool has_up = dh_seq_opt.has_value();
    if (has_up) dh_seq = dh_seq_opt->contiguous();
    Tensor d_hT;
    const bool has_hT = d_hT_opt.has_value();
    if (has_hT) d_hT = d_hT_opt->contiguous();
    Tensor d_cT;
    const bool has_cT = d_cT_opt.has_value();
    if (has_cT) d_cT = d_cT_opt->contiguous();

    auto dgates = at::empty({T, B, H4}, acts.options());
    auto dc = at::empty({B, H}, acts.options().dtype(at::kFloat));
    auto dh_buf = at::empty({B, H}, acts.options());
    auto w_hc = w_h.contiguous();
    // Fused backward step [K5a+K4]: hand MFMA dh GEMM with the pointwise
    // backward fused into its epilogue (k_lstm_dh_pw_bwd) -- one kernel
    // per timestep, fp32 dh carry in registers, and no W_h^T transpose
    // copy (the kernel reads the (H,4H) layout directly).
    // auto (-1): only where the (B/256, H/256) grid fills enough of the
    // 256-CU chip AND H is in the measured-parity regime (H=1024: tie
    // with split + fewer launches; large H unmeasured -> stay split);
    // explicit 1 forces it on any eligible shape (tests).
    const bool fused = (B % 256 == 0) && (H % 256 == 0)
        && (g_fused_bwd > 0
            || (g_fused_bwd < 0 && H <= 1536
                && (B / 256) * (H / 256) >= 192));
    // bias-grad partials: (B/256, 4H) fp32, accumulated by the fused
    // kernels at dgates-write time (saves the separate 17 GB/layer colsum
    // read).  Only the 512-thread kernel implements it; the ov A/B arm
    // (mode 2) skips it and the caller falls back to colsum.
    Tensor db_partial;
    const bool want_db = fused && g_fused_bwd != 2 && T > 1;
    if (want_db)
        db_partial = at::zeros({B / 256, H4},
                               acts.options().dtype(at::kFloat));
    float* dbp = want_db ? (float*)db_partial.data_ptr() : nullptr;
    Tensor w_h_t;
    if (!fused) {
        // split fallback: W_h^T as a CONTIGUOUS copy (one 8 MB transpose
        // per layer-backward): hipBLASLt runs the per-step (B,4H)x(4H,H)
        // grad GEMM ~1.4x faster on the plain-NN layout than on the
        // transposed view (980 vs 670 TF/s).
        w_h_t = w_h.t().contiguous();
    }

    auto* act_p = (const char*)acts.data_ptr();
    auto* cs_p = (const float*)c_seq.data_ptr();
    auto* dg_p = (char*)dgates.data_ptr();
    auto* up_p = has_up ? (const char*)dh_seq.data_ptr() : nullptr;
    auto stream = cur_stream();

    for (long k = T - 1; k >= 0; --k) {
        const long t = reverse ? (T - 1 - k) : k;
        const long t_prev = reverse ? (T - k) : (k - 1);  // index of c_{t-1}
        const bool first = (k == T - 1);
        const float* cprev_p;
        if (k > 0) {
            cprev_p = cs_p + t_prev * B * H;
        } else {
            cprev_p = (const float*)c0c.data_ptr();
        }
        if (fused && !first) {
            // dgates of the step handled in the PREVIOUS iteration
            const long t_nx = reverse ? (t - 1) : (t + 1);
            // 512-thread 256x256 kernel (measured: ties split and beats
            // the 2-blocks/CU ov variant, whose 128-row tiles double the
            // W-panel traffic at K=4H -- 248 vs 236 us/step at headline)
            int rc = -1;
            // interleaved A/B (4 rounds): swz-off 206.7 vs swz-on 212.3
            // us/step median -> no-swizzle is the default; mode 3 kept as
            // the swz-on A/B arm
            const int bswz = (g_fused_bwd == 3) ? 1 : 0;
            if (g_fused_bwd == 2)
                rc = launch_lstm_dh_pw_bwd_ov(
                    dg_p + 2 * t_nx * B * H4, H4, w_hc.data_ptr(), H4,
                    has_up ? up_p + 2 * t * B * H : nullptr, H,
                    act_p + 2 * t * B * H4, H4,
                    cs_p + t * B * H, H, cprev_p, H,
                    dg_p + 2 * t * B * H4, H4,
                    (float*)dc.data_ptr(), H,
                    nullptr, 0, (int)B, (int)H, /*raw=*/0, /*swz=*/0,
                    stream);
            if (rc != 0) rc = launch_lstm_dh_pw_bwd(
                dg_p + 2 * t_nx * B * H4, H4, w_hc.data_ptr(), H4,
                has_up ? up_p + 2 * t * B * H : nullptr, H,
                act_p + 2 * t * B * H4, H4,
                cs_p + t * B * H, H, cprev_p, H,
                dg_p + 2 * t * B * H4, H4,
                (float*)dc.data_ptr(), H,
                nullptr, 0, dbp, (int)B, (int)H, /*raw=*/0, bswz, stream);
            TORCH_CHECK(rc == 0, "fused bwd launch failed");
            continue;
        }
        const void* carry_p = nullptr;
        long carry_stride = H;
        if (first) {
            if (has_hT) carry_p = d_hT.data_ptr();
        } else {
            carry_p = dh_buf.data_ptr();
        }
        launch_lstm_pw_bwd(
            has_up ? up_p + 2 * t * B * H : nullptr, H,
            carry_p, carry_stride,
            act_p + 2 * t * B * H4, H4,
            cs_p + t * B * H, H, cprev_p, H,
            dg_p + 2 * t * B * H4, H4,
            (float*)dc.data_ptr(), H, (int)B, (int)H,
            first ? 1 : 0,
            (first && has_cT) ? (const float*)d_cT.data_ptr() : nullptr, H,
            stream);
        if (!fused) {
            // recurrent hidden-grad GEMM [K5a]: dh_{t-1} = dgates_t @ W_h^T
            at::mm_out(dh_buf, dgates.select(0, t), w_h_t);
        }
    }
    if (fused) {
        // chain-end dh0 = dgates_{t_last} @ W_h^T via the RAW fused kernel
        const long t0 = reverse ? (T - 1) : 0;
        int rc = launch_lstm_dh_pw_bwd(
            dg_p + 2 * t0 * B * H4, H4, w_hc.data_ptr(), H4,
            nullptr, 0, nullptr, 0, nullptr, 0, nullptr, 0,
            nullptr, 0, nullptr, 0,
            dh_buf.data_ptr(), H, nullptr, (int)B, (int)H, /*raw=*/1,
            /*swz=*/0, stream);
        TORCH_CHECK(rc == 0, "fused bwd raw launch failed");
    }
    if (!want_db)
        db_partial = at::empty({0, H4}, acts.options().dtype(at::kFloat));
    // (dgates, dh0, dc0, bias-grad partials [0 rows when unavailable --
    //  covers all timesteps EXCEPT the first processed step, which ran
    //  through k_lstm_pw_bwd; the python side adds that step's colsum])
    return {dgates, dh_buf, dc, db_partial};
}

// ---------------------------------------------------------------------
// Serving GEMV: out (B,N) = x (B,K) @ A_t(N,K)^T + bias, the small-B
// decode fast path (one wave per output row, full-rate weight streaming).
// A_t is the PRE-TRANSPOSED (serving-layout) weight.  Returns the result
// in bf16; raises if the shape is outside the fast path's envelope --
// the python caller guards with decode_gemv_ok().
// ---------------------------------------------------------------------
Tensor decode_gemv(Tensor A_t, Tensor x, c10::optional<Tensor> bias) {
    check_bf16_cuda(A_t, "A_t");
    check_bf16_cuda(x, "x");
    TORCH_CHECK(A_t.dim() == 2 && x.dim() == 2 && A_t.size(1) == x.size(1),
                "decode_gemv: A_t (N,K), x (B,K)");
    A_t = A_t.contiguous();
    x = x.contiguous();
    const long N = A_t.size(0), K = A_t.size(1), B = x.size(0);
    const float* bp = nullptr;
    Tensor bc;
    if (bias.has_value()) {
        bc = bias->to(at::kFloat).contiguous();
        bp = (const float*)bc.data_ptr();
    }
    auto out = at::empty({B, N}, x.options());
    int rc = launch_gemv_bf16(A_t.data_ptr(), x.data_ptr(), bp,
                              out.data_ptr(), nullptr, (int)B, N, (int)K,
                              cur_stream());
    TORCH_CHECK(rc == 0, "decode_gemv: shape outside fast-path envelope "
                "(B<=16, B*K*2 <= 160 KiB LDS, K % 16 == 0)");
    return out;
}

void decode_cell_(Tensor raw, Tensor h, Tensor c) {
    // In-place decode cell update: raw (B,4H) bf16 = full pre-activation
    // gates; h (B,H) bf16 and c (B,H) fp32 are the state buffers, updated
    // in place (ONE kernel replaces the ~12 elementwise ops of the eager
    // cell math -- per-token decode is small-kernel-latency bound).
    check_bf16_cuda(raw, "raw");
    check_bf16_cuda(h, "h");
    const long B = raw.size(0), H4 = raw.size(1), H = H4 / 4;
    TORCH_CHECK(h.size(0) == B && h.size(1) == H && c.size(1) == H,
                "state shape mismatch");
    TORCH_CHECK(raw.is_contiguous() && h.is_contiguous() && c.is_contiguous());
    TORCH_CHECK(c.scalar_type() == at::kFloat);
    launch_lstm_pw_fwd(raw.data_ptr(), H4, nullptr, 0,
                       (const float*)c.data_ptr(), H,
                       h.data_ptr(), H, (float*)c.data_ptr(), H,
                       nullptr, 0, (int)B, (int)H, cur_stream());
}

Tensor colsum_bf16(Tensor t) {
    // out (C,) fp32 = sum over rows of (R, C) bf16, one bf16 pass +
    // deterministic fixed-order partial combine (bitwise run-to-run)
    check_bf16_cuda(t, "t");
    TORCH_CHECK(t.dim() == 2, "colsum wants (R,C)");
    t = t.contiguous();
    const long R = t.size(0), C = t.size(1);
    auto partials = at::empty({(long)colsum_gy(R, C), C},
                              t.options().dtype(at::kFloat));
    auto out = at::empty({C}, t.options().dtype(at::kFloat));
    int rc = launch_colsum_bf16(t.data_ptr(), (float*)partials.data_ptr(),
                                (float*)out.data_ptr(), R, C, cur_stream());
    TORCH_CHECK(rc == 0, "colsum: C % 8 != 0");
    return out;
}

bool decode_gemv_ok(long B, long K) {
    return B <= 16 && B * K * 2 <= 160 * 1024 && K % 1024 == 0;
}

// ---------------------------------------------------------------------
// FP8 quantization helpers (experimental fp8 mode, python side: the big
// dW/dx GEMMs in ops/lstm_layer.py).  mult = 448/amax, a DEVICE scalar.
// ---------------------------------------------------------------------
Tensor quant_fp8(Tensor t, Tensor mult) {
    check_bf16_cuda(t, "t");
    t = t.contiguous();
    auto out = at::empty_like(t, t.options().dtype(at::kFloat8_e4m3fn));
    launch_quant_fp8_dyn(t.data_ptr(), out.data_ptr(), t.numel(),
                         (const float*)mult.data_ptr(), cur_stream());
    return out;
}

std::vector<Tensor> quant_fp8_pack(Tensor t, Tensor mult) {
    // (R,C) bf16 -> {out8 (R,C) e4m3, out8T (C,R) e4m3, db (C,) fp32} in
    // ONE pass (fp8 wgrad path).  Raises outside the divisibility
    // envelope -- the python caller guards.
    check_bf16_cuda(t, "t");
    TORCH_CHECK(t.dim() == 2, "quant_fp8_pack wants (R,C)");
    t = t.contiguous();
    const long R = t.size(0), C = t.size(1);
    TORCH_CHECK(R % 512 == 0 && C % 128 == 0,
                "quant_fp8_pack: R % 512 == 0 && C % 128 == 0 required");
    auto opts8 = t.options().dtype(at::kFloat8_e4m3fn);
    auto out8 = at::empty({R, C}, opts8);
    auto out8T = at::empty({C, R}, opts8);
    auto partials = at::empty({R / 512, C}, t.options().dtype(at::kFloat));
    auto db = at::empty({C}, t.options().dtype(at::kFloat));
    int rc = launch_quant_fp8_pack(t.data_ptr(), out8.data_ptr(),
                                   out8T.data_ptr(),
                                   (float*)partials.data_ptr(), R, C,
                                   (const float*)mult.data_ptr(),
                                   cur_stream());
    TORCH_CHECK(rc == 0, "quant_fp8_pack launch failed");
    launch_colsum_final((const float*)partials.data_ptr(),
                        (float*)db.data_ptr(), C, (int)(R / 512),
                        cur_stream());
    return {out8, out8T, db};
}

Tensor quant_fp8_T(Tensor t, Tensor mult) {
    check_bf16_cuda(t, "t");
    TORCH_CHECK(t.dim() == 2, "quant_fp8_T wants (R,C)");
    t = t.contiguous();
    const long R = t.size(0), C = t.size(1);
    auto out = at::empty({C, R}, t.options().dtype(at::kFloat8_e4m3fn));
    launch_quant_fp8_dyn_T(t.data_ptr(), out.data_ptr(), R, C,
                           (const float*)mult.data_ptr(), cur_stream());
    return out;
}

// ---------------------------------------------------------------------
// K3: fused log-softmax cross-entropy (LM head, vocab up to 50k+)
// ---------------------------------------------------------------------
std::vector<Tensor> xent_fwd(Tensor logits, Tensor targets) {
    check_bf16_cuda(logits, "logits");
    TORCH_CHECK(logits.dim() == 2 && targets.dim() == 1, "logits (N,V), targets (N,)");
    TORCH_CHECK(targets.scalar_type() == at::kLong, "targets int64");
    logits = logits.contiguous();
    auto tgt = targets.contiguous();
    const long N = logits.size(0);
    const int V = (int)logits.size(1);
    auto loss = at::empty({N}, logits.options().dtype(at::kFloat));
    auto lse = at::empty({N}, logits.options().dtype(at::kFloat));
    launch_xent_fwd(logits.data_ptr(), (const long*)tgt.data_ptr(),
                    (float*)loss.data_ptr(), (float*)lse.data_ptr(), N, V,
                    cur_stream());
    return {loss, lse};
}

Tensor xent_bwd(Tensor logits, Tensor targets, Tensor lse,
                c10::optional<Tensor> grad_out) {
    logits = logits.contiguous();
    auto tgt = targets.contiguous();
    const long N = logits.size(0);
    const int V = (int)logits.size(1);
    auto dlogits = at::empty_like(logits);
    const float* gp = nullptr;
    Tensor g;
    if (grad_out.has_value()) {
        g = grad_out->to(at::kFloat).contiguous();
        gp = (const float*)g.data_ptr();
    }
    launch_xent_bwd(logits.data_ptr(), (const long*)tgt.data_ptr(),
                    (const float*)lse.data_ptr(), dlogits.data_ptr(),
                    gp, 1.0f / (float)N, N, V, cur_stream());
    return dlogits;
}

// ---------------------------------------------------------------------
// K6/K7: multi-tensor clip + optimizers
// ---------------------------------------------------------------------
Tensor multi_tensor_l2norm(std::vector<Tensor> grads) {
    TORCH_CHECK(!grads.empty());
    auto out = at::zeros({}, grads[0].options().dtype(at::kFloat));
    auto stream = cur_stream();
    for (auto& g : grads) {
        auto gc = g.contiguous();
        TORCH_CHECK(gc.is_cuda());
        launch_l2norm(gc.data_ptr(), gc.numel(), (float*)out.data_ptr(),
                      gc.scalar_type() == at::kBFloat16 ? 1 : 0, stream);
    }
    return out.sqrt_();
}

void multi_tensor_clip_scale(std::vector<Tensor> grads, Tensor norm,
                             double max_norm) {
    auto stream = cur_stream();
    for (auto& g : grads) {
        TORCH_CHECK(g.is_contiguous(), "grad must be contiguous for clip");
        launch_clip_scale(g.data_ptr(), g.numel(),
                          (const float*)norm.data_ptr(), (float)max_norm,
                          g.scalar_type() == at::kBFloat16 ? 1 : 0, stream);
    }
}

void fused_sgd(std::vector<Tensor> params, std::vector<Tensor> masters,
               std::vector<Tensor> grads, std::vector<Tensor> bufs,
               double lr, double mom, double wd) {
    auto stream = cur_stream();
    const bool has_mom = !bufs.empty();
    for (size_t i = 0; i < params.size(); ++i) {
        auto& p = params[i];
        auto g = grads[i].contiguous();
        TORCH_CHECK(p.is_contiguous() && masters[i].is_contiguous());
        TORCH_CHECK(g.numel() == p.numel() && masters[i].numel() == p.numel());
        TORCH_CHECK(g.scalar_type() == p.scalar_type(), "grad/param dtype mismatch");
        launch_sgd(p.data_ptr(), (float*)masters[i].data_ptr(), g.data_ptr(),
                   has_mom ? (float*)bufs[i].data_ptr() : nullptr, p.numel(),
                   (float)lr, (float)mom, (float)wd,
                   p.scalar_type() == at::kBFloat16 ? 1 : 0, stream);
    }
}

void fused_adam(std::vector<Tensor> params, std::vector<Tensor> masters,
                std::vector<Tensor> grads, std::vector<Tensor> ea,
                std::vector<Tensor> eas, double lr, double b1, double b2,
                double eps, double wd, double bc1, double bc2) {
    auto stream = cur_stream();
    for (size_t i = 0; i < params.size(); ++i) {
        auto& p = params[i];
        auto g = grads[i].contiguous();
        TORCH_CHECK(g.scalar_type() == p.scalar_type(), "grad/param dtype mismatch");
        launch_adam(p.data_ptr(), (float*)masters[i].data_ptr(), g.data_ptr(),
                    (float*)ea[i].data_ptr(), (float*)eas[i].data_ptr(),
                    p.numel(), (float)lr, (float)b1, (float)b2, (float)eps,
                    (float)wd, (float)bc1, (float)bc2,
                    p.scalar_type() == at::kBFloat16 ? 1 : 0, stream);
    }
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.doc() = "MI355X (gfx950) HIP kernels for lstm_tensorspark_amd";
    m.def("lstm_seq_fwd", &lstm_seq_fwd, "fused LSTM sequence forward");
    m.def("lstm_seq_bwd", &lstm_seq_bwd, "fused LSTM sequence backward",
          py::arg("dh_seq") = py::none(), py::arg("d_hT") = py::none(),
          py::arg("d_cT") = py::none(), py::arg("w_h"), py::arg("acts"),
          py::arg("c_seq"), py::arg("c0"), py::arg("reverse"));
    m.def("lstm_rec_fwd_step", &lstm_rec_fwd_step,
          "single fused recurrent step (raw_debug for GEMM-mapping tests)");
    m.def("set_persistent_fwd", [](bool v) { g_persistent_enabled = v; });
    m.def("colsum_bf16", &colsum_bf16, "single-pass fp32 column sum");
    m.def("decode_cell_", &decode_cell_, "in-place fused decode cell update");
    m.def("decode_gemv", &decode_gemv, "small-B serving GEMV",
          py::arg("A_t"), py::arg("x"), py::arg("bias") = py::none());
    m.def("decode_gemv_ok", &decode_gemv_ok);
    m.def("set_gemv_nt", &set_gemv_nt,
          "non-temporal weight loads in the decode GEMV (1 default)");
    m.def("quant_fp8", &quant_fp8, "bf16 -> e4m3, device-scalar multiplier");
    m.def("quant_fp8_T", &quant_fp8_T, "bf16 (R,C) -> e4m3 (C,R) transposed");
    m.def("quant_fp8_pack", &quant_fp8_pack,
          "one-pass bf16 (R,C) -> {e4m3 (R,C), e4m3 (C,R), colsum fp32}");
    m.def("set_fp8_fwd", [](bool v) { g_fp8_fwd = v; },
          "experimental: fp8(e4m3) recurrent GEMM in the split forward");
    m.def("get_fp8_fwd", []() { return g_fp8_fwd; });
    m.def("set_split_fwd", [](int v) { g_split_fwd = v; },
          "-1 auto (B>=2048), 0 fused hand kernel, 1 hipBLASLt+pointwise");
    m.def("get_split_fwd", []() { return g_split_fwd; });
    m.def("set_fused_bwd", [](int v) { g_fused_bwd = v; },
          "fused dh-GEMM+pointwise backward: -1/1 on where eligible, 0 off");
    m.def("get_fused_bwd", []() { return g_fused_bwd; });
    m.def("get_persistent_fwd", []() { return g_persistent_enabled; });
    m.def("set_fwd_variant", &set_fwd_variant,
          "select rec_fwd kernel variant (bit0 STAGE_A, bit1 XCD_SWZ, bit2 BM32)");
    m.def("get_fwd_variant", &get_fwd_variant);
    m.def("xent_fwd", &xent_fwd);
    m.def("xent_bwd", &xent_bwd);
    m.def("multi_tensor_l2norm", &multi_tensor_l2norm);
    m.def("multi_tensor_clip_scale", &multi_tensor_clip_scale);
    m.def("fused_sgd", &fused_sgd);
    m.def("fused_adam", &fused_adam);
    m.attr("BUILT_FOR") = "gfx950";
}
