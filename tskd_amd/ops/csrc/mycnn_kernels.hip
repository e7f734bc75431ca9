// MyCNN fused inference kernels — hand-written CDNA4 (gfx950 / MI355X) HIP.
//
// Replaces the reference's PyTorch ATen CPU ops (reference bin/models.py:22-36;
// SURVEY.md §2.6 K1-K9) with two fused kernels:
//
//   1. conv_stack_kernel:  (SN, CIN, 120) windows -> (SN, LIN) features.
//      Fuses Conv1d(CIN,4,K1) + tanh + MaxPool1d(PK,PS) + Conv1d(4,1,5) +
//      tanh + pool into ONE kernel: one 64-lane wavefront per window, the
//      window and all intermediates staged in LDS, weights staged in LDS
//      once per workgroup. Dropout sites are identity in eval (K5).
//
//   2. lstm_head_kernel:   (S, N, LIN) features -> (S, N) logits/probs.
//      Fuses the 2-layer LSTM(LIN,16) *batch-axis-as-time* scan (the
//      reference's 2-D-input quirk: hidden state flows across the N windows
//      of a batch — SURVEY.md §2.3) with the Linear(16,1) head, the
//      age gate relu(age*eps+1) and optional sigmoid. One wavefront per
//      sequence: the 64 lanes are the 64 LSTM gate-units (4 gates x 16
//      hidden), per-lane weight rows live in VGPRs, h/c in registers,
//      cross-lane traffic via __shfl only; features prefetched into LDS in
//      CHUNK-step blocks so the sequential scan is never global-latency
//      bound.
//
// Numerics: fp32 accumulation throughout; input windows bf16 or fp32.
// Wavefront size is 64 on CDNA4 (not 32) — all lane math below assumes it.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define WG_WAVES 4
#define WG_THREADS (WAVE * WG_WAVES)

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
// Kernel 1: fused conv stack (K1+K2+K3+K4 of SURVEY.md §2.6)
// ---------------------------------------------------------------------------
// One wave per window; WG_WAVES windows per workgroup; grid-stride over SN.
// DT: input element (unsigned short = bf16 bits, or float).
template <class G, class DT>
__global__ __launch_bounds__(WG_THREADS) void conv_stack_kernel(
    const DT* __restrict__ x,    // (SN, CIN, L)
    float* __restrict__ feat,    // (SN, LIN)
    const float* __restrict__ wpack,
    int SN)
{
    constexpr int NW = 4 * G::CIN * G::K1 + 4 + 20 + 1;  // conv weights+biases
    __shared__ float lds_w[NW];
    __shared__ float lds_x[WG_WAVES][G::CIN * G::L];
    __shared__ float lds_c1[WG_WAVES][4 * G::C1];
    __shared__ float lds_p1[WG_WAVES][4 * G::P1];
    __shared__ float lds_c2[WG_WAVES][G::C2];

    // Stage conv weights once per workgroup.
    for (int i = threadIdx.x; i < NW; i += WG_THREADS)
        lds_w[i] = wpack[i];
    __syncthreads();

    const float* w1 = lds_w + G::OW1;
    const float* b1 = lds_w + G::OB1;
    const float* w2 = lds_w + G::OW2;
    const float  b2 = lds_w[G::OB2];

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    float* xw  = lds_x[wave];
    float* c1w = lds_c1[wave];
    float* p1w = lds_p1[wave];
    float* c2w = lds_c2[wave];

    for (int win = blockIdx.x * WG_WAVES + wave; win < SN;
         win += gridDim.x * WG_WAVES) {
        // --- stage window into LDS as fp32 ---
        const DT* xin = x + (long)win * (G::CIN * G::L);
        for (int i = lane; i < G::CIN * G::L; i += WAVE) {
            if constexpr (sizeof(DT) == 2)
                xw[i] = bf16_to_f32((unsigned short)xin[i]);
            else
                xw[i] = (float)xin[i];
        }
        wave_sync();

        // --- conv1 + tanh: 4 x C1 outputs ---
        for (int o = lane; o < 4 * G::C1; o += WAVE) {
            const int c = o / G::C1, s = o % G::C1;
            float acc = b1[c];
            const float* wr = w1 + c * (G::CIN * G::K1);
            #pragma unroll
            for (int i = 0; i < G::CIN; ++i) {
                const float* xr = xw + i * G::L + s;
                #pragma unroll
                for (int k = 0; k < G::K1; ++k)
                    acc = fmaf(wr[i * G::K1 + k], xr[k], acc);
            }
            c1w[o] = tanhf_(acc);
        }
        wave_sync();

        // --- pool1 (PK, PS) ---
        for (int o = lane; o < 4 * G::P1; o += WAVE) {
            const int c = o / G::P1, q = o % G::P1;
            const float* src = c1w + c * G::C1 + q * G::PS;
            float m = src[0];
            #pragma unroll
            for (int k = 1; k < G::PK; ++k) m = fmaxf(m, src[k]);
            p1w[o] = m;
        }
        wave_sync();

        // --- conv2 + tanh: C2 outputs (k=5, 4 in-ch) ---
        for (int s = lane; s < G::C2; s += WAVE) {
            float acc = b2;
            #pragma unroll
            for (int c = 0; c < 4; ++c) {
                const float* pr = p1w + c * G::P1 + s;
                #pragma unroll
                for (int k = 0; k < 5; ++k)
                    acc = fmaf(w2[c * 5 + k], pr[k], acc);
            }
            c2w[s] = tanhf_(acc);
        }
        wave_sync();

        // --- pool2 -> feature vector ---
        for (int q = lane; q < G::LIN; q += WAVE) {
            const float* src = c2w + q * G::PS;
            float m = src[0];
            #pragma unroll
            for (int k = 1; k < G::PK; ++k) m = fmaxf(m, src[k]);
            feat[(long)win * G::LIN + q] = m;
        }
        // next grid-stride window re-stages; wave_sync before overwrite
        wave_sync();
    }
}

// ---------------------------------------------------------------------------
// Kernel 2: fused LSTM (batch-as-time) + Linear head + age gate (+sigmoid)
// (K6+K7+K8+K9 of SURVEY.md §2.6)
// ---------------------------------------------------------------------------
// Lane layout: lane l owns gate-unit row l of both LSTM layers, where rows
// 0-15 = input gate i, 16-31 = forget f, 32-47 = cell g, 48-63 = output o
// (PyTorch gate order) for hidden units u = l & 15.  h/c live in lanes 0-15.
template <class G, int CHUNK = 32>
__global__ __launch_bounds__(WG_THREADS) void lstm_head_kernel(
    const float* __restrict__ feat,  // (S, N, LIN)
    const float* __restrict__ age,   // (S, N) or nullptr
    float* __restrict__ out,         // (S, N)
    const float* __restrict__ wpack,
    int S, int N, float age_eps, int apply_sigmoid)
{
    constexpr int LIN = G::LIN;
    __shared__ float lds_feat[WG_WAVES][CHUNK * LIN];

    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int unit = lane & 15;

    // Per-lane weight rows -> registers (broadcast through L2; once per seq).
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
    const float outw = wpack[G::OOUTW + unit];  // head weight for own unit
    const float outb = wpack[G::OOUTB];

    float* fw = lds_feat[wave];

    for (int seq = blockIdx.x * WG_WAVES + wave; seq < S;
         seq += gridDim.x * WG_WAVES) {
        const float* fseq = feat + (long)seq * N * LIN;
        const float* aseq = age ? age + (long)seq * N : nullptr;
        float h1 = 0.f, c1 = 0.f, h2 = 0.f, c2 = 0.f;  // lanes 0-15 hold state

        for (int t0 = 0; t0 < N; t0 += CHUNK) {
            const int tn = min(CHUNK, N - t0);
            // Prefetch a CHUNK of features into LDS (amortizes HBM latency
            // over CHUNK sequential steps).
            for (int i = lane; i < tn * LIN; i += WAVE)
                fw[i] = fseq[(long)t0 * LIN + i];
            wave_sync();

            for (int tt = 0; tt < tn; ++tt) {
                const float* xt = fw + tt * LIN;
                // ----- layer 1 gates: LIN-dim x-dot (LDS broadcast reads)
                //       + 16-dim h-dot (register shuffle) -----
                float g = bl1;
                #pragma unroll
                for (int i = 0; i < LIN; ++i) g = fmaf(wih1[i], xt[i], g);
                #pragma unroll
                for (int u = 0; u < 16; ++u)
                    g = fmaf(whh1[u], __shfl(h1, u), g);
                float a = (lane >= 32 && lane < 48) ? tanhf_(g) : sigmoidf_(g);
                {
                    const float iu = __shfl(a, unit);
                    const float fu = __shfl(a, unit + 16);
                    const float gu = __shfl(a, unit + 32);
                    const float ou = __shfl(a, unit + 48);
                    const float cn = fmaf(fu, c1, iu * gu);
                    c1 = cn;
                    h1 = ou * tanhf_(cn);
                }
                // ----- layer 2 -----
                g = bl2;
                #pragma unroll
                for (int u = 0; u < 16; ++u) {
                    const float h1u = __shfl(h1, u);
                    g = fmaf(wih2[u], h1u, g);
                    g = fmaf(whh2[u], __shfl(h2, u), g);
                }
                a = (lane >= 32 && lane < 48) ? tanhf_(g) : sigmoidf_(g);
                {
                    const float iu = __shfl(a, unit);
                    const float fu = __shfl(a, unit + 16);
                    const float gu = __shfl(a, unit + 32);
                    const float ou = __shfl(a, unit + 48);
                    const float cn = fmaf(fu, c2, iu * gu);
                    c2 = cn;
                    h2 = ou * tanhf_(cn);
                }
                // ----- head: logit = sum_u outw[u]*h2[u] + outb,
                //       then age gate (+ optional sigmoid) -----
                float p = (lane < 16) ? outw * h2 : 0.f;
                #pragma unroll
                for (int off = 8; off > 0; off >>= 1)
                    p += __shfl_xor(p, off);
                if (lane == 0) {
                    float y = p + outb;
                    const float ag = aseq ? aseq[t0 + tt] : 0.f;
                    const float scale = fmaxf(fmaf(ag, age_eps, 1.0f), 0.0f);
                    y *= scale;
                    if (apply_sigmoid) y = sigmoidf_(y);
                    out[(long)seq * N + t0 + tt] = y;
                }
            }
            wave_sync();  // before overwriting the feature chunk
        }
    }
}

// ---------------------------------------------------------------------------
// extern "C" launchers (ctypes API; stream owned by caller -> hipGraph-safe)
// ---------------------------------------------------------------------------
namespace {

template <class G>
int launch_conv(const void* x, int x_is_bf16, float* feat, const float* wpack,
                int SN, hipStream_t stream) {
    if (SN <= 0) return 0;
    int grid = (SN + WG_WAVES - 1) / WG_WAVES;
    if (grid > 8192) grid = 8192;  // grid-stride beyond (Guideline 11)
    if (x_is_bf16)
        hipLaunchKernelGGL((conv_stack_kernel<G, unsigned short>), dim3(grid),
                           dim3(WG_THREADS), 0, stream,
                           (const unsigned short*)x, feat, wpack, SN);
    else
        hipLaunchKernelGGL((conv_stack_kernel<G, float>), dim3(grid),
                           dim3(WG_THREADS), 0, stream,
                           (const float*)x, feat, wpack, SN);
    return (int)hipGetLastError();
}

template <class G>
int launch_lstm(const float* feat, const float* age, float* out,
                const float* wpack, int S, int N, float age_eps,
                int apply_sigmoid, hipStream_t stream) {
    if (S <= 0 || N <= 0) return 0;
    int grid = (S + WG_WAVES - 1) / WG_WAVES;
    if (grid > 8192) grid = 8192;
    hipLaunchKernelGGL((lstm_head_kernel<G>), dim3(grid), dim3(WG_THREADS), 0,
                       stream, feat, age, out, wpack, S, N, age_eps,
                       apply_sigmoid);
    return (int)hipGetLastError();
}

}  // namespace

extern "C" {

// variant: 0 = MyCNN5, 1 = MyCNN2/3, 2 = MyCNN4
int tskd_conv_fwd(const void* x, int x_is_bf16, float* feat,
                  const float* wpack, int SN, int variant, void* stream) {
    hipStream_t s = (hipStream_t)stream;
    switch (variant) {
        case 0: return launch_conv<GeomCNN5>(x, x_is_bf16, feat, wpack, SN, s);
        case 1: return launch_conv<GeomCNN2>(x, x_is_bf16, feat, wpack, SN, s);
        case 2: return launch_conv<GeomCNN4>(x, x_is_bf16, feat, wpack, SN, s);
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

}  // extern "C"
