// MyCNN fused inference kernels — hand-written CDNA4 (gfx950 / MI355X) HIP.
//
// Replaces the reference's PyTorch ATen CPU ops (reference bin/models.py:22-36;
// SURVEY.md §2.6 K1-K9) with two fused kernels:
//
//   1. conv_stack_*_kernel: (SN, CIN, 120) windows -> (SN, LIN) features.
//      Fuses Conv1d(CIN,4,K1) + tanh + MaxPool1d(PK,PS) + Conv1d(4,1,5) +
//      tanh + pool into ONE kernel: one 64-lane wavefront per window.
//      Two compute paths:
//        - bf16 input (the serving path): conv1 as an MFMA im2col-GEMM.
//          The window is staged TRANSPOSED in LDS as bf16 (xt[t][i]), which
//          makes the im2col matrix A[s][kk] (kk = k*CIN + i) a simple
//          shifted view xt_flat[s*CIN + kk] — A fragments are contiguous
//          LDS reads, no gather. B = conv weights pre-packed host-side in
//          MFMA fragment order (4 used columns of a 16-col tile); the
//          v_mfma_f32_16x16x32_bf16 accumulator chains over KK/32 k-steps,
//          epilogue adds bias + tanh and lands rows in LDS for the pool.
//        - fp32 input (exact-numerics path): direct VALU cross-correlation.
//      Dropout sites are identity in eval (K5).
//
//   2. lstm_head_kernel: (S, N, LIN) features -> (S, N) logits/probs.
//      Fuses the 2-layer LSTM(LIN,16) *batch-axis-as-time* scan (the
//      reference's 2-D-input quirk: hidden state flows across the N windows
//      of a batch — SURVEY.md §2.3) with the Linear(16,1) head, the age gate
//      relu(age*eps+1) and optional sigmoid. One wavefront per sequence:
//      the 64 lanes are the 64 gate-unit rows (4 gates x 16 hidden),
//      per-lane weight rows live in VGPRs, the full h vectors of both layers
//      are collected into per-lane registers with 16 INDEPENDENT shuffles
//      per layer (dependency-chain broken: the serial-scan critical path is
//      ~2 shuffle round-trips per layer, not 32), features prefetched into
//      LDS in CHUNK-step blocks.
//
// Numerics: fp32 accumulation throughout (MFMA accumulator is fp32).
// Wavefront size is 64 on CDNA4 (not 32) — all lane math assumes it.

#include <hip/hip_runtime.h>
#include <stdlib.h>

#define WAVE 64
#define WG_WAVES 4
#define WG_THREADS (WAVE * WG_WAVES)

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// ---------------------------------------------------------------------------
// Geometry per model variant (lengths from SURVEY.md §2.3 forward semantics)
// ---------------------------------------------------------------------------
template <int CIN_, int K1_, int PK_, int PS_, int L_ = 120>
struct Geom {
    static constexpr int CIN = CIN_;
    static constexpr int K1  = K1_;
    static constexpr int PK  = PK_;
    static constexpr int PS  = PS_;
    static constexpr int L   = L_;
    static constexpr int C1  = L - K1 + 1;            // conv1 out len
    static constexpr int P1  = (C1 - PK) / PS + 1;    // pool1 out len
    static constexpr int C2  = P1 - 5 + 1;            // conv2 out len (k=5)
    static constexpr int LIN = (C2 - PK) / PS + 1;    // pool2 out len = LSTM in
    // MFMA im2col geometry
    static constexpr int KK     = CIN * K1;           // reduction length
    static constexpr int KSTEPS = (KK + 31) / 32;     // mfma k-steps (K=32)
    static constexpr int MTILES = (C1 + 15) / 16;     // 16-row output tiles
    static constexpr int XTN    = CIN * L;            // transposed window elems
    static constexpr int XTSZ   = ((MTILES * 16 - 1) * CIN + KSTEPS * 32 + 7)
                                  / 8 * 8;            // padded LDS extent
    // fp32 weight-pack offsets (must match tskd_amd/ops/pack.py)
    static constexpr int OW1   = 0;                   // [4][CIN][K1]
    static constexpr int OB1   = OW1 + 4 * CIN * K1;  // [4]
    static constexpr int OW2   = OB1 + 4;             // [4][5]
    static constexpr int OB2   = OW2 + 20;            // [1]
    static constexpr int OWIH1 = OB2 + 1;             // [64][LIN]
    static constexpr int OWHH1 = OWIH1 + 64 * LIN;    // [64][16]
    static constexpr int OBL1  = OWHH1 + 64 * 16;     // [64] (b_ih + b_hh)
    static constexpr int OWIH2 = OBL1 + 64;           // [64][16]
    static constexpr int OWHH2 = OWIH2 + 64 * 16;     // [64][16]
    static constexpr int OBL2  = OWHH2 + 64 * 16;     // [64]
    static constexpr int OOUTW = OBL2 + 64;           // [16]
    static constexpr int OOUTB = OOUTW + 16;          // [1]
    static constexpr int NPACK = OOUTB + 1;
};

using GeomCNN5 = Geom<10, 10, 3, 2>;  // MyCNN5: 111/55/51/25
using GeomCNN2 = Geom<7, 5, 2, 2>;    // MyCNN2/3: 116/58/54/27
using GeomCNN4 = Geom<10, 5, 2, 2>;   // MyCNN4: same lens as CNN2, 10 ch

static_assert(GeomCNN5::LIN == 25, "MyCNN5 feature length must be 25");
static_assert(GeomCNN2::LIN == 27, "MyCNN2 feature length must be 27");
static_assert(GeomCNN5::KSTEPS == 4 && GeomCNN2::KSTEPS == 2, "");

// Same-wave LDS read-after-write fence: LDS ops of one wave complete in
// order once lgkmcnt drains; the asm "memory" clobber stops compiler
// reordering, wave_barrier stops scheduler migration across it.
__device__ __forceinline__ void wave_sync() {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
}

__device__ __forceinline__ float bf16_to_f32(unsigned short u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

__device__ __forceinline__ float sigmoidf_(float x) {
    return 1.0f / (1.0f + __expf(-x));
}

__device__ __forceinline__ float tanhf_(float x) {
    // tanh(x) = 2*sigmoid(2x) - 1; fast-exp based, fp32-accurate to ~1e-7 rel.
    return 2.0f / (1.0f + __expf(-2.0f * x)) - 1.0f;
}

// ---------------------------------------------------------------------------
// Shared epilogue: pool1 -> conv2 -> tanh -> pool2 -> feature store.
// Reads lds_c1 (4 x C1 conv1 rows), uses lds_p1/lds_c2 scratch.
//
// TANH_AT_POOL: tanh is monotone, so tanh(max(a,b)) == max(tanh(a), tanh(b))
// — the MFMA path stores RAW conv1 pre-activations (bias folded) and applies
// tanh AFTER each pool, cutting transcendental count ~2x (444->220, 51->25)
// and keeping every lane active.
// ---------------------------------------------------------------------------
template <class G, bool TANH_AT_POOL>
__device__ __forceinline__ void conv_tail(
    int lane, const float* __restrict__ lds_w, float* c1w, float* p1w,
    float* c2w, float* __restrict__ feat, long win)
{
    const float* w2 = lds_w + G::OW2;
    const float  b2 = lds_w[G::OB2];
    // pool1 (PK, PS)
    for (int o = lane; o < 4 * G::P1; o += WAVE) {
        const int c = o / G::P1, q = o % G::P1;
        const float* src = c1w + c * G::C1 + q * G::PS;
        float m = src[0];
        #pragma unroll
        for (int k = 1; k < G::PK; ++k) m = fmaxf(m, src[k]);
        p1w[o] = TANH_AT_POOL ? tanhf_(m) : m;
    }
    wave_sync();
    // conv2 (+ tanh here only when pool1 already produced tanh'd inputs and
    // pool2 will apply the second tanh)
    for (int s = lane; s < G::C2; s += WAVE) {
        float acc = b2;
        #pragma unroll
        for (int c = 0; c < 4; ++c) {
            const float* pr = p1w + c * G::P1 + s;
            #pragma unroll
            for (int k = 0; k < 5; ++k)
                acc = fmaf(w2[c * 5 + k], pr[k], acc);
        }
        c2w[s] = TANH_AT_POOL ? acc : tanhf_(acc);
    }
    wave_sync();
    // pool2 -> feature vector
    for (int q = lane; q < G::LIN; q += WAVE) {
        const float* src = c2w + q * G::PS;
        float m = src[0];
        #pragma unroll
        for (int k = 1; k < G::PK; ++k) m = fmaxf(m, src[k]);
        feat[win * G::LIN + q] = TANH_AT_POOL ? tanhf_(m) : m;
    }
    wave_sync();  // before the next window overwrites scratch
}

// ---------------------------------------------------------------------------
// Kernel 1a (fp32 path): direct VALU conv stack — exact fp32 numerics.
// ---------------------------------------------------------------------------
template <class G>
__global__ __launch_bounds__(WG_THREADS) void conv_stack_kernel(
    const float* __restrict__ x,  // (SN, CIN, L)
    float* __restrict__ feat,     // (SN, LIN)
    const float* __restrict__ wpack,
    int SN)
{
    constexpr int NW = 4 * G::CIN * G::K1 + 4 + 20 + 1;
    __shared__ float lds_w[NW];
    __shared__ float lds_x[WG_WAVES][G::CIN * G::L];
    __shared__ float lds_c1[WG_WAVES][4 * G::C1];
    __shared__ float lds_p1[WG_WAVES][4 * G::P1];
    __shared__ float lds_c2[WG_WAVES][G::C2];

    for (int i = threadIdx.x; i < NW; i += WG_THREADS) lds_w[i] = wpack[i];
    __syncthreads();

    const float* w1 = lds_w + G::OW1;
    const float* b1 = lds_w + G::OB1;
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    float* xw = lds_x[wave];

    for (long win = blockIdx.x * WG_WAVES + wave; win < SN;
         win += (long)gridDim.x * WG_WAVES) {
        const float* xin = x + win * (G::CIN * G::L);
        for (int i = lane; i < G::CIN * G::L; i += WAVE) xw[i] = xin[i];
        wave_sync();
        for (int o = lane; o < 4 * G::C1; o += WAVE) {
            const int c = o / G::C1, s = o % G::C1;
            // two independent partial accumulators (even/odd input channel)
            // halve the 100-deep dependent fma chain
            float acc = b1[c], acc2 = 0.f;
            const float* wr = w1 + c * (G::CIN * G::K1);
            #pragma unroll
            for (int i = 0; i < G::CIN; i += 2) {
                const float* xr = xw + i * G::L + s;
                const float* xr2 = xw + (i + 1) * G::L + s;
                #pragma unroll
                for (int k = 0; k < G::K1; ++k) {
                    acc = fmaf(wr[i * G::K1 + k], xr[k], acc);
                    if (i + 1 < G::CIN)
                        acc2 = fmaf(wr[(i + 1) * G::K1 + k], xr2[k], acc2);
                }
            }
            lds_c1[wave][o] = tanhf_(acc + acc2);
        }
        wave_sync();
        conv_tail<G, false>(lane, lds_w, lds_c1[wave], lds_p1[wave],
                            lds_c2[wave], feat, win);
    }
}

// ---------------------------------------------------------------------------
// Kernel 1b (bf16 path): conv1 as MFMA im2col-GEMM (16x16x32 bf16).
//
// Fragment maps (gfx950 v_mfma_f32_16x16x32_bf16; verified on-device by
// tskd_debug_mfma16x16x32 + tests/test_hip_mycnn.py):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C: lane l holds C[row = (l>>4)*4 + r][col = l&15], r = 0..3
// ---------------------------------------------------------------------------
template <class G>
__global__ __launch_bounds__(WG_THREADS) void conv_stack_mfma_kernel(
    const unsigned short* __restrict__ x,   // (SN, CIN, L) bf16 bits
    float* __restrict__ feat,               // (SN, LIN)
    const float* __restrict__ wpack,
    const unsigned short* __restrict__ bfrag,  // [KSTEPS][64][8] bf16 bits
    int SN)
{
    constexpr int NW = 4 * G::CIN * G::K1 + 4 + 20 + 1;
    __shared__ float lds_w[NW];
    __shared__ unsigned short lds_xt[WG_WAVES][G::XTSZ];  // transposed window
    __shared__ float lds_c1[WG_WAVES][4 * G::C1];
    __shared__ float lds_p1[WG_WAVES][4 * G::P1];
    __shared__ float lds_c2[WG_WAVES][G::C2];

    for (int i = threadIdx.x; i < NW; i += WG_THREADS) lds_w[i] = wpack[i];
    __syncthreads();

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const float* b1 = lds_w + G::OB1;
    unsigned short* xt = lds_xt[wave];

    // Preload B fragments (per-lane, all k-steps): one dwordx4 per step.
    bf16x8 bfr[G::KSTEPS];
    union BU { unsigned int d[4]; unsigned short u[8]; bf16x8 v; };
    #pragma unroll
    for (int st = 0; st < G::KSTEPS; ++st) {
        BU bu;
        bu.d[0] = ((const unsigned int*)bfrag)[(st * WAVE + lane) * 4 + 0];
        bu.d[1] = ((const unsigned int*)bfrag)[(st * WAVE + lane) * 4 + 1];
        bu.d[2] = ((const unsigned int*)bfrag)[(st * WAVE + lane) * 4 + 2];
        bu.d[3] = ((const unsigned int*)bfrag)[(st * WAVE + lane) * 4 + 3];
        bfr[st] = bu.v;
    }

    for (long win = blockIdx.x * WG_WAVES + wave; win < SN;
         win += (long)gridDim.x * WG_WAVES) {
        const unsigned short* xin = x + win * (G::CIN * G::L);
        // Stage TRANSPOSED: xt[t*CIN + i] = x[i*L + t]. Coalesced 8-byte
        // global reads (L % 4 == 0 so a quad never crosses a channel row),
        // scattered 2-byte LDS writes.
        static_assert(G::L % 4 == 0, "quad staging needs L % 4 == 0");
        for (int q = lane; q < G::CIN * G::L / 4; q += WAVE) {
            const int f = q * 4;
            const int i = f / G::L, t0 = f % G::L;
            union { unsigned long long u; unsigned short h[4]; } v;
            v.u = *(const unsigned long long*)(xin + f);
            #pragma unroll
            for (int j = 0; j < 4; ++j)
                xt[(t0 + j) * G::CIN + i] = v.h[j];
        }
        for (int i = G::XTN + lane; i < G::XTSZ; i += WAVE) xt[i] = 0;
        wave_sync();

        // conv1 = A(im2col view of xt) x B(weights), one 16-row tile at a time
        #pragma unroll 1
        for (int mt = 0; mt < G::MTILES; ++mt) {
            const int s = mt * 16 + (lane & 15);
            f32x4 acc = {0.f, 0.f, 0.f, 0.f};
            #pragma unroll
            for (int st = 0; st < G::KSTEPS; ++st) {
                const int base = s * G::CIN + st * 32 + (lane >> 4) * 8;
                BU au;
                if constexpr (G::CIN % 2 == 0) {
                    // base is even: 4-byte-aligned LDS reads land directly
                    // in the fragment dwords (little-endian bf16 pairs).
                    const unsigned int* xtu = (const unsigned int*)xt;
                    au.d[0] = xtu[base / 2];
                    au.d[1] = xtu[base / 2 + 1];
                    au.d[2] = xtu[base / 2 + 2];
                    au.d[3] = xtu[base / 2 + 3];
                } else {
                    #pragma unroll
                    for (int j = 0; j < 8; ++j) au.u[j] = xt[base + j];
                }
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(au.v, bfr[st],
                                                              acc, 0, 0, 0);
            }
            // epilogue: bias-folded RAW pre-activations; tanh is applied
            // after pool1 (monotonicity) in conv_tail<., true>
            const int c = lane & 15;
            if (c < 4) {
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int row = mt * 16 + (lane >> 4) * 4 + r;
                    if (row < G::C1)
                        lds_c1[wave][c * G::C1 + row] = acc[r] + b1[c];
                }
            }
        }
        wave_sync();
        conv_tail<G, true>(lane, lds_w, lds_c1[wave], lds_p1[wave],
                           lds_c2[wave], feat, win);
    }
}

// ---------------------------------------------------------------------------
// Kernel 1c (bf16 TIME-LAST layout): conv1 MFMA im2col with A-fragments read
// STRAIGHT FROM GLOBAL — no LDS staging, no transpose pass.
//
// Input layout (SN, L, CIN) ("timelast"): the im2col row A[s][kk]
// (kk = k*CIN + i) is the contiguous global range x[win*L*CIN + s*CIN + kk],
// so each lane's 8-element fragment is 4 dword loads through the vector-
// memory path (L1-resident: a window is 2.4 KB). PMC showed the staged
// variant spends 39% of cycles waiting on LDS (profiles/r01) — this variant
// keeps LDS only for the small pooled tail.
//
// NOTE: fragment reads overrun a window by up to (KSTEPS*32 - KK + C1%16
// rows) elements into the NEXT window (zero B-columns make the values
// irrelevant); callers must allocate the x buffer with TLAST_SLACK trailing
// elements (tskd_amd/ops wrapper does).
// ---------------------------------------------------------------------------
#define TLAST_SLACK 256  // trailing bf16 elements required after the tensor
template <class G>
__global__ __launch_bounds__(WG_THREADS) void conv_stack_mfma_tlast_kernel(
    const unsigned short* __restrict__ x,   // (SN, L, CIN) bf16 bits
    float* __restrict__ feat,               // (SN, LIN)
    const float* __restrict__ wpack,
    const unsigned short* __restrict__ bfrag,  // [KSTEPS][64][8] bf16 bits
    int SN)
{
    constexpr int NW = 4 * G::CIN * G::K1 + 4 + 20 + 1;
    __shared__ float lds_w[NW];
    __shared__ float lds_c1[WG_WAVES][4 * G::C1];
    __shared__ float lds_p1[WG_WAVES][4 * G::P1];
    __shared__ float lds_c2[WG_WAVES][G::C2];

    for (int i = threadIdx.x; i < NW; i += WG_THREADS) lds_w[i] = wpack[i];
    __syncthreads();

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const float* b1 = lds_w + G::OB1;

    bf16x8 bfr[G::KSTEPS];
    union BU { unsigned int d[4]; unsigned short u[8]; bf16x8 v; };
    #pragma unroll
    for (int st = 0; st < G::KSTEPS; ++st) {
        BU bu;
        #pragma unroll
        for (int q = 0; q < 4; ++q)
            bu.d[q] = ((const unsigned int*)bfrag)[(st * WAVE + lane) * 4 + q];
        bfr[st] = bu.v;
    }
    // per-lane fragment base within a window (dwords): rows l&15, k-group l>>4
    const int lane_dw = ((lane & 15) * G::CIN + (lane >> 4) * 8) / 2;
    __shared__ float lds_c1b[WG_WAVES][4 * G::C1];  // 2nd window of the pair

    // TWO windows per wave: the pair's MFMA/load chains are independent, so
    // the in-order wave covers one chain's global-load latency with the
    // other's MFMAs (the single-window chain was latency-exposed).
    for (long win = (blockIdx.x * WG_WAVES + wave) * 2; win < SN;
         win += (long)gridDim.x * WG_WAVES * 2) {
        const long win1 = win + 1;
        const bool has1 = win1 < SN;
        const unsigned int* xwa =
            (const unsigned int*)(x + win * (long)(G::L * G::CIN));
        const unsigned int* xwb =
            (const unsigned int*)(x + (has1 ? win1 : win) *
                                  (long)(G::L * G::CIN));
        #pragma unroll 1
        for (int mt = 0; mt < G::MTILES; ++mt) {
            f32x4 acca = {0.f, 0.f, 0.f, 0.f};
            f32x4 accb = {0.f, 0.f, 0.f, 0.f};
            const unsigned int* basea = xwa + (long)mt * (16 * G::CIN / 2)
                                        + lane_dw;
            const unsigned int* baseb = xwb + (long)mt * (16 * G::CIN / 2)
                                        + lane_dw;
            #pragma unroll
            for (int st = 0; st < G::KSTEPS; ++st) {
                BU aa, ab;
                aa.d[0] = basea[st * 16 + 0];
                aa.d[1] = basea[st * 16 + 1];
                aa.d[2] = basea[st * 16 + 2];
                aa.d[3] = basea[st * 16 + 3];
                ab.d[0] = baseb[st * 16 + 0];
                ab.d[1] = baseb[st * 16 + 1];
                ab.d[2] = baseb[st * 16 + 2];
                ab.d[3] = baseb[st * 16 + 3];
                acca = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aa.v, bfr[st],
                                                               acca, 0, 0, 0);
                accb = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ab.v, bfr[st],
                                                               accb, 0, 0, 0);
            }
            const int c = lane & 15;
            if (c < 4) {
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int row = mt * 16 + (lane >> 4) * 4 + r;
                    if (row < G::C1) {
                        lds_c1[wave][c * G::C1 + row] = acca[r] + b1[c];
                        lds_c1b[wave][c * G::C1 + row] = accb[r] + b1[c];
                    }
                }
            }
        }
        wave_sync();
        conv_tail<G, true>(lane, lds_w, lds_c1[wave], lds_p1[wave],
                           lds_c2[wave], feat, win);
        if (has1)
            conv_tail<G, true>(lane, lds_w, lds_c1b[wave], lds_p1[wave],
                               lds_c2[wave], feat, win1);
    }
}

// ---------------------------------------------------------------------------
// Kernel 1d (bf16 TIME-LAST, workgroup-per-window): all 4 waves of a WG
// cooperate on ONE window — waves split the 7 conv1 M-tiles, the pooled
// tail runs WG-wide — dividing the per-window serial-latency chain ~4x at
// the same instruction count. Tiny LDS (one window's intermediates) keeps
// occupancy at the wave cap.
// ---------------------------------------------------------------------------
template <class G>
__global__ __launch_bounds__(WG_THREADS) void conv_stack_mfma_wg_kernel(
    const unsigned short* __restrict__ x,   // (SN, L, CIN) bf16 bits
    float* __restrict__ feat,               // (SN, LIN)
    const float* __restrict__ wpack,
    const unsigned short* __restrict__ bfrag,
    int SN)
{
    constexpr int NW = 4 * G::CIN * G::K1 + 4 + 20 + 1;
    __shared__ float lds_w[NW];
    __shared__ float lds_c1[4 * G::C1];
    __shared__ float lds_p1[4 * G::P1];
    __shared__ float lds_c2[G::C2];

    for (int i = threadIdx.x; i < NW; i += WG_THREADS) lds_w[i] = wpack[i];
    __syncthreads();

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const float* b1 = lds_w + G::OB1;
    const float* w2 = lds_w + G::OW2;
    const float b2 = lds_w[G::OB2];

    bf16x8 bfr[G::KSTEPS];
    union BU { unsigned int d[4]; unsigned short u[8]; bf16x8 v; };
    #pragma unroll
    for (int st = 0; st < G::KSTEPS; ++st) {
        BU bu;
        #pragma unroll
        for (int q = 0; q < 4; ++q)
            bu.d[q] = ((const unsigned int*)bfrag)[(st * WAVE + lane) * 4 + q];
        bfr[st] = bu.v;
    }
    const int lane_dw = ((lane & 15) * G::CIN + (lane >> 4) * 8) / 2;

    for (long win = blockIdx.x; win < SN; win += gridDim.x) {
        const unsigned int* xw =
            (const unsigned int*)(x + win * (long)(G::L * G::CIN));
        // waves split the M-tiles
        for (int mt = wave; mt < G::MTILES; mt += WG_WAVES) {
            f32x4 acc = {0.f, 0.f, 0.f, 0.f};
            const unsigned int* base = xw + (long)mt * (16 * G::CIN / 2)
                                       + lane_dw;
            #pragma unroll
            for (int st = 0; st < G::KSTEPS; ++st) {
                BU au;
                au.d[0] = base[st * 16 + 0];
                au.d[1] = base[st * 16 + 1];
                au.d[2] = base[st * 16 + 2];
                au.d[3] = base[st * 16 + 3];
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(au.v, bfr[st],
                                                              acc, 0, 0, 0);
            }
            const int c = lane & 15;
            if (c < 4) {
                #pragma unroll
                for (int r = 0; r < 4; ++r) {
                    const int row = mt * 16 + (lane >> 4) * 4 + r;
                    if (row < G::C1)
                        lds_c1[c * G::C1 + row] = acc[r] + b1[c];
                }
            }
        }
        __syncthreads();
        // WG-wide pooled tail (tanh after pool — monotonicity)
        for (int o = threadIdx.x; o < 4 * G::P1; o += WG_THREADS) {
            const int c = o / G::P1, q = o % G::P1;
            const float* src = lds_c1 + c * G::C1 + q * G::PS;
            float m = src[0];
            #pragma unroll
            for (int k = 1; k < G::PK; ++k) m = fmaxf(m, src[k]);
            lds_p1[o] = tanhf_(m);
        }
        __syncthreads();
        for (int s = threadIdx.x; s < G::C2; s += WG_THREADS) {
            float acc = b2;
            #pragma unroll
            for (int c = 0; c < 4; ++c) {
                const float* pr = lds_p1 + c * G::P1 + s;
                #pragma unroll
                for (int k = 0; k < 5; ++k)
                    acc = fmaf(w2[c * 5 + k], pr[k], acc);
            }
            lds_c2[s] = acc;
        }
        __syncthreads();
        for (int q = threadIdx.x; q < G::LIN; q += WG_THREADS) {
            const float* src = lds_c2 + q * G::PS;
            float m = src[0];
            #pragma unroll
            for (int k = 1; k < G::PK; ++k) m = fmaxf(m, src[k]);
            feat[win * G::LIN + q] = tanhf_(m);
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------------
// Kernel 2: fused LSTM (batch-as-time) + Linear head + age gate (+sigmoid)
// (K6+K7+K8+K9 of SURVEY.md §2.6)
// ---------------------------------------------------------------------------
// Lane layout: lane l owns gate-unit row l of both LSTM layers, where rows
// 0-15 = input gate i, 16-31 = forget f, 32-47 = cell g, 48-63 = output o
// (PyTorch gate order) for hidden units u = l & 15.  After each step the
// full 16-wide h vectors are collected into per-lane register arrays with
// 16 independent shuffles, so the next step's h-dots are register FMAs.
template <class G, int CHUNK = 32>
__global__ __launch_bounds__(WG_THREADS) void lstm_head_kernel(
    const float* __restrict__ feat,  // (S, N, LIN)
    const float* __restrict__ age,   // (S, N) or nullptr
    float* __restrict__ out,         // (S, N)
    const float* __restrict__ wpack,
    int S, int N, float age_eps, int apply_sigmoid)
{
    constexpr int LIN = G::LIN;
    constexpr int PRE_R = (CHUNK * LIN + WAVE - 1) / WAVE;  // regs per lane
    __shared__ float lds_feat[WG_WAVES][CHUNK * LIN];

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int unit = lane & 15;

    // Per-lane weight rows -> registers (broadcast through L2; once).
    float wih1[LIN], whh1[16], wih2[16], whh2[16];
    const float bl1 = wpack[G::OBL1 + lane];
    const float bl2 = wpack[G::OBL2 + lane];
    #pragma unroll
    for (int i = 0; i < LIN; ++i) wih1[i] = wpack[G::OWIH1 + lane * LIN + i];
    #pragma unroll
    for (int i = 0; i < 16; ++i) {
        whh1[i] = wpack[G::OWHH1 + lane * 16 + i];
        wih2[i] = wpack[G::OWIH2 + lane * 16 + i];
        whh2[i] = wpack[G::OWHH2 + lane * 16 + i];
    }
    const float outw = wpack[G::OOUTW + unit];
    const float outb = wpack[G::OOUTB];
    const bool is_cell_gate = (lane >= 32 && lane < 48);

    float* fw = lds_feat[wave];

    for (long seq = blockIdx.x * WG_WAVES + wave; seq < S;
         seq += (long)gridDim.x * WG_WAVES) {
        const float* fseq = feat + seq * (long)N * LIN;
        const float* aseq = age ? age + seq * N : nullptr;
        float h1v[16], h2v[16];   // full h vectors, replicated per lane
        float c1 = 0.f, c2 = 0.f; // own-unit cell state
        #pragma unroll
        for (int u = 0; u < 16; ++u) h1v[u] = h2v[u] = 0.f;

        // Double-buffered chunk pipeline: chunk c+1's global loads are
        // ISSUED before chunk c's compute (their s_waitcnt lands at the LDS
        // write after the compute), so the scan never stalls on HBM latency
        // at chunk boundaries.
        const long ntot = (long)N * LIN;
        float pre[PRE_R];
        #pragma unroll
        for (int r = 0; r < PRE_R; ++r) {
            const long idx = (long)r * WAVE + lane;
            pre[r] = idx < ntot ? fseq[idx] : 0.f;
        }
        for (int t0 = 0; t0 < N; t0 += CHUNK) {
            const int tn = min(CHUNK, N - t0);
            #pragma unroll
            for (int r = 0; r < PRE_R; ++r) {
                const int j = r * WAVE + lane;
                if (j < CHUNK * LIN) fw[j] = pre[r];
            }
            wave_sync();
            if (t0 + CHUNK < N) {
                const long base = (long)(t0 + CHUNK) * LIN;
                #pragma unroll
                for (int r = 0; r < PRE_R; ++r) {
                    const long idx = base + (long)r * WAVE + lane;
                    pre[r] = idx < ntot ? fseq[idx] : 0.f;
                }
            }

            for (int tt = 0; tt < tn; ++tt) {
                const float* xt = fw + tt * LIN;
                // ----- layer 1: LIN-dim x-dot (LDS broadcast, 2 partial
                //       accumulators) + 16-dim h-dot (register FMA) -----
                float ga = bl1, gb = 0.f;
                #pragma unroll
                for (int i = 0; i < LIN; i += 2) {
                    ga = fmaf(wih1[i], xt[i], ga);
                    if (i + 1 < LIN) gb = fmaf(wih1[i + 1], xt[i + 1], gb);
                }
                #pragma unroll
                for (int u = 0; u < 16; u += 2) {
                    ga = fmaf(whh1[u], h1v[u], ga);
                    gb = fmaf(whh1[u + 1], h1v[u + 1], gb);
                }
                float g = ga + gb;
                float a = is_cell_gate ? tanhf_(g) : sigmoidf_(g);
                {
                    const float iu = __shfl(a, unit);
                    const float fu = __shfl(a, unit + 16);
                    const float gu = __shfl(a, unit + 32);
                    const float ou = __shfl(a, unit + 48);
                    c1 = fmaf(fu, c1, iu * gu);
                    const float h = ou * tanhf_(c1);
                    #pragma unroll
                    for (int u = 0; u < 16; ++u) h1v[u] = __shfl(h, u);
                }
                // ----- layer 2: both dots are register FMAs -----
                ga = bl2; gb = 0.f;
                #pragma unroll
                for (int u = 0; u < 16; u += 2) {
                    ga = fmaf(wih2[u], h1v[u], ga);
                    gb = fmaf(wih2[u + 1], h1v[u + 1], gb);
                    ga = fmaf(whh2[u], h2v[u], ga);
                    gb = fmaf(whh2[u + 1], h2v[u + 1], gb);
                }
                g = ga + gb;
                a = is_cell_gate ? tanhf_(g) : sigmoidf_(g);
                {
                    const float iu = __shfl(a, unit);
                    const float fu = __shfl(a, unit + 16);
                    const float gu = __shfl(a, unit + 32);
                    const float ou = __shfl(a, unit + 48);
                    c2 = fmaf(fu, c2, iu * gu);
                    const float h = ou * tanhf_(c2);
                    #pragma unroll
                    for (int u = 0; u < 16; ++u) h2v[u] = __shfl(h, u);
                }
                // ----- head (every lane has h2v: reduce-free dot) -----
                if (lane == 0) {
                    float y = outb;
                    #pragma unroll
                    for (int u = 0; u < 16; ++u)
                        y = fmaf(wpack[G::OOUTW + u], h2v[u], y);
                    const float ag = aseq ? aseq[t0 + tt] : 0.f;
                    y *= fmaxf(fmaf(ag, age_eps, 1.0f), 0.0f);
                    if (apply_sigmoid) y = sigmoidf_(y);
                    out[seq * N + t0 + tt] = y;
                }
            }
            wave_sync();  // before overwriting the feature chunk
        }
    }
    (void)outw;
}

// ---------------------------------------------------------------------------
// Debug: one 16x16x32 bf16 MFMA tile (fragment-map ground truth for tests)
// ---------------------------------------------------------------------------
__global__ void debug_mfma_kernel(const unsigned short* __restrict__ A,
                                  const unsigned short* __restrict__ B,
                                  float* __restrict__ C) {
    const int lane = threadIdx.x;
    union BU { unsigned short u[8]; bf16x8 v; } a, b;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        const int k = (lane >> 4) * 8 + j;
        a.u[j] = A[(lane & 15) * 32 + k];   // A[row][k], row-major 16x32
        b.u[j] = B[k * 16 + (lane & 15)];   // B[k][col], row-major 32x16
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v, acc, 0, 0, 0);
    #pragma unroll
    for (int r = 0; r < 4; ++r)
        C[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

// ---------------------------------------------------------------------------
// extern "C" launchers (ctypes API; stream owned by caller -> hipGraph-safe)
// ---------------------------------------------------------------------------
namespace {

template <class G>
int launch_conv(const void* x, int x_is_bf16, int x_timelast, float* feat,
                const float* wpack, const void* bfrag, int SN,
                hipStream_t stream) {
    if (SN <= 0) return 0;
    int grid = (SN + WG_WAVES - 1) / WG_WAVES;
    if (grid > 8192) grid = 8192;
    if (x_is_bf16) {
        if (!bfrag) return -2;  // bf16 path requires packed B fragments
        if (x_timelast) {
            if constexpr (G::CIN % 2 == 0) {
                // DEFAULT: 1-wave-per-2-windows. The 4-wave-WG variant
                // measured equal at 4096x256 but 17% SLOWER at the bench
                // shape 8192x1024 (gpurun_out/val_infknobs.log) — always
                // re-A/B at the production shape.
                int onewave = 1;
                if (const char* e = getenv("TSKD_CONV_TLAST_1WAVE"))
                    onewave = atoi(e);
                if (onewave) {
                    int g1 = (SN / 2 + WG_WAVES - 1) / WG_WAVES;
                    if (g1 < 1) g1 = 1;
                    if (g1 > 8192) g1 = 8192;
                    hipLaunchKernelGGL((conv_stack_mfma_tlast_kernel<G>),
                                       dim3(g1), dim3(WG_THREADS), 0, stream,
                                       (const unsigned short*)x, feat, wpack,
                                       (const unsigned short*)bfrag, SN);
                } else {
                    // 4-wave workgroup per window (A/B reference)
                    int g2 = SN > 16384 ? 16384 : SN;
                    hipLaunchKernelGGL((conv_stack_mfma_wg_kernel<G>),
                                       dim3(g2), dim3(WG_THREADS), 0, stream,
                                       (const unsigned short*)x, feat, wpack,
                                       (const unsigned short*)bfrag, SN);
                }
                return (int)hipGetLastError();
            }
            return -4;  // odd-CIN variants: use the staged layout
        }
        hipLaunchKernelGGL((conv_stack_mfma_kernel<G>), dim3(grid),
                           dim3(WG_THREADS), 0, stream,
                           (const unsigned short*)x, feat, wpack,
                           (const unsigned short*)bfrag, SN);
    } else {
        if (x_timelast) return -4;  // fp32 path is (C, L) only
        hipLaunchKernelGGL((conv_stack_kernel<G>), dim3(grid),
                           dim3(WG_THREADS), 0, stream, (const float*)x, feat,
                           wpack, SN);
    }
    return (int)hipGetLastError();
}

template <class G>
int launch_lstm(const float* feat, const float* age, float* out,
                const float* wpack, int S, int N, float age_eps,
                int apply_sigmoid, hipStream_t stream) {
    if (S <= 0 || N <= 0) return 0;
    int grid = (S + WG_WAVES - 1) / WG_WAVES;
    if (grid > 8192) grid = 8192;
    int chunk = 32;
    if (const char* e = getenv("TSKD_LSTM_CHUNK")) chunk = atoi(e);
    if (chunk >= 64)
        hipLaunchKernelGGL((lstm_head_kernel<G, 64>), dim3(grid),
                           dim3(WG_THREADS), 0, stream, feat, age, out,
                           wpack, S, N, age_eps, apply_sigmoid);
    else
        hipLaunchKernelGGL((lstm_head_kernel<G, 32>), dim3(grid),
                           dim3(WG_THREADS), 0, stream, feat, age, out,
                           wpack, S, N, age_eps, apply_sigmoid);
    return (int)hipGetLastError();
}

}  // namespace

extern "C" {

// variant: 0 = MyCNN5, 1 = MyCNN2/3, 2 = MyCNN4
int tskd_conv_fwd(const void* x, int x_is_bf16, int x_timelast, float* feat,
                  const float* wpack, const void* bfrag, int SN, int variant,
                  void* stream) {
    hipStream_t s = (hipStream_t)stream;
    switch (variant) {
        case 0: return launch_conv<GeomCNN5>(x, x_is_bf16, x_timelast, feat,
                                             wpack, bfrag, SN, s);
        case 1: return launch_conv<GeomCNN2>(x, x_is_bf16, x_timelast, feat,
                                             wpack, bfrag, SN, s);
        case 2: return launch_conv<GeomCNN4>(x, x_is_bf16, x_timelast, feat,
                                             wpack, bfrag, SN, s);
    }
    return -1;
}

int tskd_lstm_head_fwd(const float* feat, const float* age, float* out,
                       const float* wpack, int S, int N, float age_eps,
                       int apply_sigmoid, int variant, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    switch (variant) {
        case 0: return launch_lstm<GeomCNN5>(feat, age, out, wpack, S, N,
                                             age_eps, apply_sigmoid, s);
        case 1: return launch_lstm<GeomCNN2>(feat, age, out, wpack, S, N,
                                             age_eps, apply_sigmoid, s);
        case 2: return launch_lstm<GeomCNN4>(feat, age, out, wpack, S, N,
                                             age_eps, apply_sigmoid, s);
    }
    return -1;
}

int tskd_debug_mfma16x16x32(const unsigned short* A, const unsigned short* B,
                            float* C, void* stream) {
    hipLaunchKernelGGL(debug_mfma_kernel, dim3(1), dim3(WAVE), 0,
                       (hipStream_t)stream, A, B, C);
    return (int)hipGetLastError();
}

int tskd_pack_size(int variant) {
    switch (variant) {
        case 0: return GeomCNN5::NPACK;
        case 1: return GeomCNN2::NPACK;
        case 2: return GeomCNN4::NPACK;
    }
    return -1;
}

int tskd_feat_len(int variant) {
    switch (variant) {
        case 0: return GeomCNN5::LIN;
        case 1: return GeomCNN2::LIN;
        case 2: return GeomCNN4::LIN;
    }
    return -1;
}

// MFMA B-fragment pack geometry for python (ksteps per variant).
int tskd_conv_ksteps(int variant) {
    switch (variant) {
        case 0: return GeomCNN5::KSTEPS;
        case 1: return GeomCNN2::KSTEPS;
        case 2: return GeomCNN4::KSTEPS;
    }
    return -1;
}

}  // extern "C"
