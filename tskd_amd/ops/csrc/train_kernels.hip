// MyCNN training kernels — hand-written CDNA4 (gfx950) HIP.
//
// The training configuration of SURVEY.md §2.6 (K10-K12): fused
// BCE-with-logits loss + gradient, full backward through the conv stack and
// the 2-layer batch-axis-as-time LSTM (BPTT reverse scan), and a fused Adam
// step over the packed parameter buffer. The packed layout is the inference
// Geom<> layout (mycnn_kernels.hip / tskd_amd/ops/pack.py).
//
// Parameterization note (documented divergence): the packed model stores the
// LSTM bias as ONE vector b = b_ih + b_hh (only the sum enters the math).
// Gradients match torch's d(b_ih) == d(b_hh); exporting splits b/2 + b/2.
//
// Kernel inventory:
//   train_conv_fwd   — conv stack forward that stashes tanh outputs and
//                      maxpool argmax indices (one wave per window).
//   train_lstm_fwd   — LSTM scan forward stashing per-step activations
//                      (i,f,g,o), c, h for both layers + head logits
//                      (one wave per sequence; same lane layout as
//                      inference: lane = gate-unit row).
//   train_loss_bwd   — elementwise BCEWithLogits(pos_weight): per-window
//                      loss sum (atomic) + d(logit_base) incl. the age-gate
//                      scale and 1/N mean factor.
//   train_lstm_bwd   — reverse BPTT scan: per-wave weight-gradient
//                      accumulators live in LDS rows (lane-private, no
//                      conflicts), column reductions through an LDS copy of
//                      the weights; emits dfeat and atomically folds weight
//                      grads into the global grad buffer.
//   train_conv_bwd   — maxpool scatter + tanh' + conv weight/data grads.
//   train_adam       — fused Adam with bias correction + grad zeroing.

#include <hip/hip_runtime.h>
#include <stdlib.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(4))) float f32x4;

// ---- shared geometry (must match mycnn_kernels.hip) ----
template <int CIN_, int K1_, int PK_, int PS_, int L_ = 120>
struct TGeom {
    static constexpr int CIN = CIN_, K1 = K1_, PK = PK_, PS = PS_, L = L_;
    static constexpr int C1 = L - K1 + 1;
    static constexpr int P1 = (C1 - PK) / PS + 1;
    static constexpr int C2 = P1 - 5 + 1;
    static constexpr int LIN = (C2 - PK) / PS + 1;
    static constexpr int OW1 = 0;
    static constexpr int OB1 = OW1 + 4 * CIN * K1;
    static constexpr int OW2 = OB1 + 4;
    static constexpr int OB2 = OW2 + 20;
    static constexpr int OWIH1 = OB2 + 1;
    static constexpr int OWHH1 = OWIH1 + 64 * LIN;
    static constexpr int OBL1 = OWHH1 + 64 * 16;
    static constexpr int OWIH2 = OBL1 + 64;
    static constexpr int OWHH2 = OWIH2 + 64 * 16;
    static constexpr int OBL2 = OWHH2 + 64 * 16;
    static constexpr int OOUTW = OBL2 + 64;
    static constexpr int OOUTB = OOUTW + 16;
    static constexpr int NPACK = OOUTB + 1;
    // conv stash layout (per window, fp32/int32 words)
    static constexpr int SC_C1T = 0;                 // [4*C1] tanh(conv1)
    static constexpr int SC_C2T = SC_C1T + 4 * C1;   // [C2]  tanh(conv2)
    static constexpr int SC_I1 = SC_C2T + C2;        // [4*P1] pool1 argmax
    static constexpr int SC_I2 = SC_I1 + 4 * P1;     // [LIN] pool2 argmax
    static constexpr int SC_M1 = SC_I2 + LIN;        // [4*P1] dropout1 mult
    static constexpr int SC_M2 = SC_M1 + 4 * P1;     // [LIN] dropout2 mult
    static constexpr int SC_SIZE = SC_M2 + LIN;
    // lstm stash layout (per step, fp32 words)
    static constexpr int SL_A1 = 0;    // [64] layer1 activated gates
    static constexpr int SL_C1 = 64;   // [16]
    static constexpr int SL_H1 = 80;   // [16]
    static constexpr int SL_A2 = 96;   // [64]
    static constexpr int SL_C2 = 160;  // [16]
    static constexpr int SL_H2 = 176;  // [16]
    static constexpr int SL_SIZE = 192;
};

using TG5 = TGeom<10, 10, 3, 2>;
using TG2 = TGeom<7, 5, 2, 2>;
using TG4 = TGeom<10, 5, 2, 2>;

__device__ __forceinline__ void twsync() {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
}

__device__ __forceinline__ float sigm_(float x) {
    return 1.0f / (1.0f + __expf(-x));
}
__device__ __forceinline__ float tanh_(float x) {
    return 2.0f / (1.0f + __expf(-2.0f * x)) - 1.0f;
}

// Counter-based RNG for dropout masks (K5): stateless splitmix64 hash of
// (seed, element index) -> uniform [0,1). Not ATen's Philox stream — the
// reference's train-mode Bernoulli semantics, reproducible per seed.
__device__ __forceinline__ float rnd_uniform_(unsigned long long seed,
                                              unsigned long long idx) {
    unsigned long long z = seed + idx * 0x9E3779B97F4A7C15ull;
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    z ^= z >> 31;
    return (float)(unsigned int)(z >> 33) * (1.0f / 2147483648.0f);
}

// inverted-dropout multiplier: 0 with prob p, else 1/(1-p)
__device__ __forceinline__ float drop_mult_(float p, unsigned long long seed,
                                            unsigned long long idx) {
    if (p <= 0.f) return 1.f;
    return (rnd_uniform_(seed, idx) < p) ? 0.f : 1.f / (1.f - p);
}

// ---------------------------------------------------------------------------
// conv forward with stash (fp32; one wave per window, 4 waves per block)
// ---------------------------------------------------------------------------
template <class G, bool STAGE_X = true, int PAIR = 0>
__global__ __launch_bounds__(256) void train_conv_fwd_kernel(
    const float* __restrict__ x,      // (SN, CIN, L)
    float* __restrict__ feat,         // (SN, LIN)
    float* __restrict__ stash,        // (SN, SC_SIZE) (int words for argmax)
    const float* __restrict__ wpack, int SN,
    float drop1_p, float drop2_p, unsigned long long seed)
{
    constexpr int NW = 4 * G::CIN * G::K1 + 4 + 20 + 1;
    __shared__ float lw[NW];
    // lx only exists in the staged instantiation (19 KB of LDS otherwise
    // wasted against occupancy)
    __shared__ float lx[STAGE_X ? 4 : 1][STAGE_X ? G::CIN * G::L : 1];
    __shared__ float lp1[4][4 * G::P1];
    for (int i = threadIdx.x; i < NW; i += 256) lw[i] = wpack[i];
    __syncthreads();
    const int wave = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    const float* w1 = lw + G::OW1;
    const float* b1 = lw + G::OB1;
    const float* w2 = lw + G::OW2;
    const float b2 = lw[G::OB2];

    for (long win = blockIdx.x * 4 + wave; win < SN;
         win += (long)gridDim.x * 4) {
        const float* xin = x + win * (G::CIN * G::L);
        float* st = stash + win * G::SC_SIZE;
        float* c1t = st + G::SC_C1T;
        float* c2t = st + G::SC_C2T;
        int* i1 = (int*)(st + G::SC_I1);
        int* i2 = (int*)(st + G::SC_I2);
        float* m1 = st + G::SC_M1;
        float* m2 = st + G::SC_M2;
        // STAGE_X=false reads the window straight from L1-resident global
        // memory (PMC: the LDS-staged variant spends 32% of wave cycles in
        // s_waitcnt lgkmcnt — the same lesson as the inference tlast conv);
        // the backward pass still stages (it sweeps x twice).
        const float* xw;
        if (STAGE_X) {
            float* xs = lx[STAGE_X ? wave : 0];
            for (int i = lane; i < G::CIN * G::L; i += WAVE) xs[i] = xin[i];
            twsync();
            xw = xs;
        } else {
            xw = xin;
        }
        if (PAIR) {
            // Paired outputs (round 2): one lane computes TWO adjacent s
            // positions, whose x windows overlap K1-1 of K1+1 values — the
            // shared x read cuts LDS/L1 traffic in the dominant loop ~1.8x.
            constexpr int CP = (G::C1 + 1) / 2;
            for (int o = lane; o < 4 * CP; o += WAVE) {
                const int c = o / CP, s0 = (o % CP) * 2;
                float aa = b1[c], ab = b1[c], a2 = 0.f, b2p = 0.f;
                const float* wr = w1 + c * (G::CIN * G::K1);
                const bool has_b = s0 + 1 < G::C1;
                #pragma unroll
                for (int i = 0; i < G::CIN; ++i) {
                    float xv[G::K1 + 1];
                    #pragma unroll
                    for (int k = 0; k <= G::K1; ++k)
                        xv[k] = xw[i * G::L + s0 + k];
                    #pragma unroll
                    for (int k = 0; k < G::K1; k += 2) {
                        aa = fmaf(wr[i * G::K1 + k], xv[k], aa);
                        ab = fmaf(wr[i * G::K1 + k], xv[k + 1], ab);
                        if (k + 1 < G::K1) {
                            a2 = fmaf(wr[i * G::K1 + k + 1], xv[k + 1], a2);
                            b2p = fmaf(wr[i * G::K1 + k + 1], xv[k + 2],
                                       b2p);
                        }
                    }
                }
                c1t[c * G::C1 + s0] = tanh_(aa + a2);
                if (has_b) c1t[c * G::C1 + s0 + 1] = tanh_(ab + b2p);
            }
        } else if (PAIR == 2) {
            // 4 independent accumulators (i parity × k parity): halves the
            // dependent-FMA chain depth vs the 2-accumulator default
            for (int o = lane; o < 4 * G::C1; o += WAVE) {
                const int c = o / G::C1, s = o % G::C1;
                float a0 = b1[c], a1 = 0.f, a2 = 0.f, a3 = 0.f;
                const float* wr = w1 + c * (G::CIN * G::K1);
                #pragma unroll
                for (int i = 0; i < G::CIN; i += 2) {
                    #pragma unroll
                    for (int k = 0; k < G::K1; k += 2) {
                        a0 = fmaf(wr[i * G::K1 + k],
                                  xw[i * G::L + s + k], a0);
                        if (k + 1 < G::K1)
                            a1 = fmaf(wr[i * G::K1 + k + 1],
                                      xw[i * G::L + s + k + 1], a1);
                        if (i + 1 < G::CIN) {
                            a2 = fmaf(wr[(i + 1) * G::K1 + k],
                                      xw[(i + 1) * G::L + s + k], a2);
                            if (k + 1 < G::K1)
                                a3 = fmaf(wr[(i + 1) * G::K1 + k + 1],
                                          xw[(i + 1) * G::L + s + k + 1],
                                          a3);
                        }
                    }
                }
                c1t[o] = tanh_((a0 + a1) + (a2 + a3));
            }
        } else {
        for (int o = lane; o < 4 * G::C1; o += WAVE) {
            const int c = o / G::C1, s = o % G::C1;
            // two independent partial accumulators (even/odd input channel)
            float acc = b1[c], acc2 = 0.f;
            const float* wr = w1 + c * (G::CIN * G::K1);
            #pragma unroll
            for (int i = 0; i < G::CIN; i += 2) {
                #pragma unroll
                for (int k = 0; k < G::K1; ++k) {
                    acc = fmaf(wr[i * G::K1 + k], xw[i * G::L + s + k], acc);
                    if (i + 1 < G::CIN)
                        acc2 = fmaf(wr[(i + 1) * G::K1 + k],
                                    xw[(i + 1) * G::L + s + k], acc2);
                }
            }
            c1t[o] = tanh_(acc + acc2);
        }
        }
        twsync();
        for (int o = lane; o < 4 * G::P1; o += WAVE) {
            const int c = o / G::P1, q = o % G::P1;
            const float* src = c1t + c * G::C1 + q * G::PS;
            float m = src[0];
            int am = 0;
            #pragma unroll
            for (int k = 1; k < G::PK; ++k)
                if (src[k] > m) { m = src[k]; am = k; }
            // dropout site 1 (after pool1; MyCNN5's first dropout)
            const float dm = drop_mult_(drop1_p, seed,
                                        (unsigned long long)win * 1024 + o);
            m1[o] = dm;
            lp1[wave][o] = m * dm;
            i1[o] = am;
        }
        twsync();
        for (int s = lane; s < G::C2; s += WAVE) {
            float acc = b2;
            #pragma unroll
            for (int c = 0; c < 4; ++c)
                #pragma unroll
                for (int k = 0; k < 5; ++k)
                    acc = fmaf(w2[c * 5 + k], lp1[wave][c * G::P1 + s + k],
                               acc);
            c2t[s] = tanh_(acc);
        }
        twsync();
        for (int q = lane; q < G::LIN; q += WAVE) {
            const float* src = c2t + q * G::PS;
            float m = src[0];
            int am = 0;
            #pragma unroll
            for (int k = 1; k < G::PK; ++k)
                if (src[k] > m) { m = src[k]; am = k; }
            // dropout site 2 (after pool2)
            const float dm = drop_mult_(drop2_p, seed,
                                        (unsigned long long)win * 1024 + 512 + q);
            m2[q] = dm;
            feat[win * G::LIN + q] = m * dm;
            i2[q] = am;
        }
        twsync();
    }
}

// ---------------------------------------------------------------------------
// LSTM forward with stash (+ head logits). One wave per sequence.
// ---------------------------------------------------------------------------
template <class G>
__global__ __launch_bounds__(WAVE) void train_lstm_fwd_kernel(
    const float* __restrict__ feat,   // (S, B, LIN)
    const float* __restrict__ age,    // (S, B) or nullptr
    const float* __restrict__ wpack,
    float* __restrict__ stash,        // (S, B, SL_SIZE)
    float* __restrict__ logits,       // (S, B) (pre-age-scale base * scale)
    int S, int B, float age_eps)
{
    constexpr int LIN = G::LIN;
    const int lane = threadIdx.x;
    const int unit = lane & 15;
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
    const float outb = wpack[G::OOUTB];
    const bool cell_gate = (lane >= 32 && lane < 48);

    for (long seq = blockIdx.x; seq < S; seq += gridDim.x) {
        const float* fs = feat + seq * (long)B * LIN;
        float* ss = stash + seq * (long)B * G::SL_SIZE;
        float h1v[16], h2v[16];
        float c1 = 0.f, c2 = 0.f;
        #pragma unroll
        for (int u = 0; u < 16; ++u) h1v[u] = h2v[u] = 0.f;
        for (int t = 0; t < B; ++t) {
            const float* xt = fs + (long)t * LIN;
            float* sp = ss + (long)t * G::SL_SIZE;
            float g = bl1;
            #pragma unroll
            for (int i = 0; i < LIN; ++i) g = fmaf(wih1[i], xt[i], g);
            #pragma unroll
            for (int u = 0; u < 16; ++u) g = fmaf(whh1[u], h1v[u], g);
            float a = cell_gate ? tanh_(g) : sigm_(g);
            sp[G::SL_A1 + lane] = a;
            {
                const float iu = __shfl(a, unit);
                const float fu = __shfl(a, unit + 16);
                const float gu = __shfl(a, unit + 32);
                const float ou = __shfl(a, unit + 48);
                c1 = fmaf(fu, c1, iu * gu);
                const float h = ou * tanh_(c1);
                if (lane < 16) {
                    sp[G::SL_C1 + lane] = c1;
                    sp[G::SL_H1 + lane] = h;
                }
                #pragma unroll
                for (int u = 0; u < 16; ++u) h1v[u] = __shfl(h, u);
            }
            g = bl2;
            #pragma unroll
            for (int u = 0; u < 16; ++u) {
                g = fmaf(wih2[u], h1v[u], g);
                g = fmaf(whh2[u], h2v[u], g);
            }
            a = cell_gate ? tanh_(g) : sigm_(g);
            sp[G::SL_A2 + lane] = a;
            {
                const float iu = __shfl(a, unit);
                const float fu = __shfl(a, unit + 16);
                const float gu = __shfl(a, unit + 32);
                const float ou = __shfl(a, unit + 48);
                c2 = fmaf(fu, c2, iu * gu);
                const float h = ou * tanh_(c2);
                if (lane < 16) {
                    sp[G::SL_C2 + lane] = c2;
                    sp[G::SL_H2 + lane] = h;
                }
                #pragma unroll
                for (int u = 0; u < 16; ++u) h2v[u] = __shfl(h, u);
            }
            if (lane == 0) {
                float z = outb;
                #pragma unroll
                for (int u = 0; u < 16; ++u)
                    z = fmaf(wpack[G::OOUTW + u], h2v[u], z);
                const float ag = age ? age[seq * B + t] : 0.f;
                z *= fmaxf(fmaf(ag, age_eps, 1.0f), 0.0f);
                logits[seq * B + t] = z;
            }
        }
    }
}

// ---------------------------------------------------------------------------
// loss (+ dlogit_base): BCEWithLogits with pos_weight, mean over N.
//   dlogit_base = dL/d(z_base) = dL/dz * age_scale   (z = z_base * scale)
// ---------------------------------------------------------------------------
__global__ void train_loss_kernel(
    const float* __restrict__ logits, const float* __restrict__ targets,
    const float* __restrict__ age, float* __restrict__ dlogit_base,
    float* __restrict__ loss_sum, long n, float pos_weight, float age_eps)
{
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const float z = logits[i];
        const float y = targets[i];
        // stable: softplus(-z) = log(1+exp(-z)) = max(-z,0)+log1p(exp(-| z|))
        const float sp = fmaxf(-z, 0.f) + __logf(1.f + __expf(-fabsf(z)));
        const float loss = pos_weight * y * sp + (1.f - y) * (z + sp);
        const float dz = (-pos_weight * y * sigm_(-z)
                          + (1.f - y) * sigm_(z)) / (float)n;
        const float ag = age ? age[i] : 0.f;
        const float scale = fmaxf(fmaf(ag, age_eps, 1.0f), 0.0f);
        dlogit_base[i] = dz * scale;
        atomicAdd(loss_sum, loss / (float)n);
    }
}

// ---------------------------------------------------------------------------
// LSTM backward (BPTT reverse scan). One WAVE per block per sequence.
// LDS: weight copies for column access + lane-private grad rows.
// ---------------------------------------------------------------------------
template <class G, int RACC = 1, int WPB = 1>
// RACC: grad accumulators in registers (A/B ref; measured +29%)
// WPB: sequences (waves) per block sharing ONE read-only weight copy —
//      the 19 KB LDS weight stage was per-sequence at WPB=1, capping
//      occupancy at ~4 single-wave workgroups/CU
__global__ __launch_bounds__(WAVE * WPB) void train_lstm_bwd_kernel(
    const float* __restrict__ feat,     // (S, B, LIN)
    const float* __restrict__ dlogit,   // (S, B) dL/d z_base
    const float* __restrict__ stash,    // (S, B, SL_SIZE)
    const float* __restrict__ wpack,
    float* __restrict__ grads,          // (NPACK) atomic accumulate
    float* __restrict__ dfeat,          // (S, B, LIN)
    int S, int B)
{
    constexpr int LIN = G::LIN;
    // 16-wide rows padded to 17 dwords: a 16-dword stride puts the whole
    // half-wave on 2 banks (16-way conflict); gcd(17, 32) = 1 spreads it.
    constexpr int P16 = 17;
    __shared__ float lwih1[64 * LIN], lwhh1[64 * P16];
    __shared__ float lwih2[64 * P16], lwhh2[64 * P16];
    // RACC=1: the per-lane grad rows live in REGISTERS. RACC=0 keeps the
    // r1 LDS-row layout (per wave when WPB > 1).
    __shared__ float sgwih1[WPB][RACC ? 1 : 64 * LIN];
    __shared__ float sgwhh1[WPB][RACC ? 1 : 64 * P16];
    __shared__ float sgb1[WPB][RACC ? 1 : 64];
    __shared__ float sgwih2[WPB][RACC ? 1 : 64 * P16];
    __shared__ float sgwhh2[WPB][RACC ? 1 : 64 * P16];
    __shared__ float sgb2[WPB][RACC ? 1 : 64];
    __shared__ float sgout[WPB][17];
    __shared__ float slda[WPB][64];  // per-step activated-gate grads

    const int lane = threadIdx.x % WAVE;
    const int wv = threadIdx.x / WAVE;
    const int unit = lane & 15;
    float* gwih1 = sgwih1[wv];
    float* gwhh1 = sgwhh1[wv];
    float* gb1 = sgb1[wv];
    float* gwih2 = sgwih2[wv];
    float* gwhh2 = sgwhh2[wv];
    float* gb2 = sgb2[wv];
    float* gout = sgout[wv];
    float* lda = slda[wv];
    float rwih1[LIN], rwhh1[16], rwih2[16], rwhh2[16];  // dead if !RACC
    float rb1 = 0.f, rb2 = 0.f, rout = 0.f, routb = 0.f;
    #pragma unroll
    for (int j = 0; j < LIN; ++j) rwih1[j] = 0.f;
    #pragma unroll
    for (int j = 0; j < 16; ++j) rwhh1[j] = rwih2[j] = rwhh2[j] = 0.f;
    for (int i = threadIdx.x; i < 64 * LIN; i += WAVE * WPB)
        lwih1[i] = wpack[G::OWIH1 + i];
    for (int i = threadIdx.x; i < 64 * 16; i += WAVE * WPB) {
        const int pi = (i / 16) * P16 + (i % 16);
        lwhh1[pi] = wpack[G::OWHH1 + i];
        lwih2[pi] = wpack[G::OWIH2 + i];
        lwhh2[pi] = wpack[G::OWHH2 + i];
    }
    if (!RACC) {
        for (int i = lane; i < 64 * LIN; i += WAVE) gwih1[i] = 0.f;
        for (int i = lane; i < 64 * 16; i += WAVE) {
            const int pi = (i / 16) * P16 + (i % 16);
            gwhh1[pi] = gwih2[pi] = gwhh2[pi] = 0.f;
        }
        gb1[lane] = gb2[lane] = 0.f;
    }
    if (lane < 17) gout[lane] = 0.f;
    const float outw = wpack[G::OOUTW + unit];
    if (WPB > 1) __syncthreads(); else twsync();

    for (long seq = blockIdx.x * WPB + wv; seq < S;
         seq += (long)gridDim.x * WPB) {
        const float* fs = feat + seq * (long)B * LIN;
        const float* dz = dlogit + seq * (long)B;
        const float* ss = stash + seq * (long)B * G::SL_SIZE;
        float* df = dfeat + seq * (long)B * LIN;
        // carried state (replicated per lane for unit = lane&15)
        float dh1n = 0.f, dc1n = 0.f, dh2n = 0.f, dc2n = 0.f;
        for (int t = B - 1; t >= 0; --t) {
            const float* sp = ss + (long)t * G::SL_SIZE;
            const float* spm = (t > 0) ? sp - G::SL_SIZE : nullptr;
            // ---- layer 2 ----
            const float a_l2 = sp[G::SL_A2 + lane];
            const float i2 = __shfl(a_l2, unit);
            const float f2 = __shfl(a_l2, unit + 16);
            const float g2 = __shfl(a_l2, unit + 32);
            const float o2 = __shfl(a_l2, unit + 48);
            const float c2 = sp[G::SL_C2 + unit];
            const float c2p = spm ? spm[G::SL_C2 + unit] : 0.f;
            const float h2p = spm ? spm[G::SL_H2 + unit] : 0.f;
            const float h1 = sp[G::SL_H1 + unit];
            const float tc2 = tanh_(c2);
            // dh2 total for own unit
            const float dzt = dz[t];
            float dh2 = dzt * outw + dh2n;
            float dc2 = dh2 * o2 * (1.f - tc2 * tc2) + dc2n;
            // per-lane activated-gate grad (row = lane)
            float da2;
            if (lane < 16) da2 = dc2 * g2 * i2 * (1.f - i2);
            else if (lane < 32) da2 = dc2 * c2p * f2 * (1.f - f2);
            else if (lane < 48) da2 = dc2 * i2 * (1.f - g2 * g2);
            else da2 = dh2 * tc2 * o2 * (1.f - o2);
            lda[lane] = da2;
            // weight grads: lane-private rows (registers when RACC)
            #pragma unroll
            for (int j = 0; j < 16; ++j) {
                const float h1j = sp[G::SL_H1 + j];
                const float h2pj = spm ? spm[G::SL_H2 + j] : 0.f;
                if (RACC) {
                    rwih2[j] = fmaf(da2, h1j, rwih2[j]);
                    rwhh2[j] = fmaf(da2, h2pj, rwhh2[j]);
                } else {
                    gwih2[lane * P16 + j] = fmaf(da2, h1j,
                                                 gwih2[lane * P16 + j]);
                    gwhh2[lane * P16 + j] = fmaf(da2, h2pj,
                                                 gwhh2[lane * P16 + j]);
                }
            }
            if (RACC) rb2 += da2; else gb2[lane] += da2;
            if (lane < 16) {
                if (RACC) rout = fmaf(dzt, sp[G::SL_H2 + lane], rout);
                else gout[lane] = fmaf(dzt, sp[G::SL_H2 + lane], gout[lane]);
            }
            if (lane == 0) { if (RACC) routb += dzt; else gout[16] += dzt; }
            twsync();
            // column reductions: dh1 (into layer1) and dh2_{t-1}
            // lane (u + 16k) sums rows [16k, 16k+16)
            {
                const int k0 = (lane >> 4) * 16;
                float p1 = 0.f, p2 = 0.f;
                #pragma unroll
                for (int r = 0; r < 16; ++r) {
                    const float d = lda[k0 + r];
                    p1 = fmaf(d, lwih2[(k0 + r) * P16 + unit], p1);
                    p2 = fmaf(d, lwhh2[(k0 + r) * P16 + unit], p2);
                }
                #pragma unroll
                for (int off = 16; off < 64; off <<= 1) {
                    p1 += __shfl_xor(p1, off);
                    p2 += __shfl_xor(p2, off);
                }
                // p1 = dL/dh1_t(unit) from layer2; p2 = dh2_{t-1}(unit)
                dh2n = p2;
                dc2n = dc2 * f2;
                // ---- layer 1 ----
                const float a_l1 = sp[G::SL_A1 + lane];
                const float i1 = __shfl(a_l1, unit);
                const float f1 = __shfl(a_l1, unit + 16);
                const float g1 = __shfl(a_l1, unit + 32);
                const float o1 = __shfl(a_l1, unit + 48);
                const float c1 = sp[G::SL_C1 + unit];
                const float c1p = spm ? spm[G::SL_C1 + unit] : 0.f;
                const float tc1 = tanh_(c1);
                float dh1 = p1 + dh1n;
                float dc1 = dh1 * o1 * (1.f - tc1 * tc1) + dc1n;
                float da1;
                if (lane < 16) da1 = dc1 * g1 * i1 * (1.f - i1);
                else if (lane < 32) da1 = dc1 * c1p * f1 * (1.f - f1);
                else if (lane < 48) da1 = dc1 * i1 * (1.f - g1 * g1);
                else da1 = dh1 * tc1 * o1 * (1.f - o1);
                twsync();  // lda still holds layer2 values until here
                lda[lane] = da1;
                const float* xt = fs + (long)t * LIN;
                #pragma unroll
                for (int j = 0; j < LIN; ++j) {
                    if (RACC) rwih1[j] = fmaf(da1, xt[j], rwih1[j]);
                    else gwih1[lane * LIN + j] = fmaf(da1, xt[j],
                                                      gwih1[lane * LIN + j]);
                }
                #pragma unroll
                for (int j = 0; j < 16; ++j) {
                    const float h1pj = spm ? spm[G::SL_H1 + j] : 0.f;
                    if (RACC) rwhh1[j] = fmaf(da1, h1pj, rwhh1[j]);
                    else gwhh1[lane * P16 + j] = fmaf(da1, h1pj,
                                                      gwhh1[lane * P16 + j]);
                }
                if (RACC) rb1 += da1; else gb1[lane] += da1;
                dc1n = dc1 * f1;
                twsync();
                // dh1_{t-1} and dfeat
                float q1 = 0.f;
                #pragma unroll
                for (int r = 0; r < 16; ++r)
                    q1 = fmaf(lda[k0 + r], lwhh1[(k0 + r) * P16 + unit], q1);
                #pragma unroll
                for (int off = 16; off < 64; off <<= 1)
                    q1 += __shfl_xor(q1, off);
                dh1n = q1;
                // dfeat_t[j] = sum_l da1_l * wih1[l][j]; lanes j < LIN
                if (lane < LIN) {
                    float s = 0.f;
                    for (int l = 0; l < 64; ++l)
                        s = fmaf(lda[l], lwih1[l * LIN + lane], s);
                    df[(long)t * LIN + lane] = s;
                }
                twsync();
            }
        }
    }
    // fold the per-block accumulators into global grads
    twsync();
    if (RACC) {
        #pragma unroll
        for (int j = 0; j < LIN; ++j)
            atomicAdd(&grads[G::OWIH1 + lane * LIN + j], rwih1[j]);
        #pragma unroll
        for (int j = 0; j < 16; ++j) {
            atomicAdd(&grads[G::OWHH1 + lane * 16 + j], rwhh1[j]);
            atomicAdd(&grads[G::OWIH2 + lane * 16 + j], rwih2[j]);
            atomicAdd(&grads[G::OWHH2 + lane * 16 + j], rwhh2[j]);
        }
        atomicAdd(&grads[G::OBL1 + lane], rb1);
        atomicAdd(&grads[G::OBL2 + lane], rb2);
        if (lane < 16) atomicAdd(&grads[G::OOUTW + lane], rout);
        if (lane == 0) atomicAdd(&grads[G::OOUTB], routb);
        return;
    }
    for (int i = lane; i < 64 * LIN; i += WAVE)
        atomicAdd(&grads[G::OWIH1 + i], gwih1[i]);
    for (int i = lane; i < 64 * 16; i += WAVE) {
        const int pi = (i / 16) * P16 + (i % 16);
        atomicAdd(&grads[G::OWHH1 + i], gwhh1[pi]);
        atomicAdd(&grads[G::OWIH2 + i], gwih2[pi]);
        atomicAdd(&grads[G::OWHH2 + i], gwhh2[pi]);
    }
    atomicAdd(&grads[G::OBL1 + lane], gb1[lane]);
    atomicAdd(&grads[G::OBL2 + lane], gb2[lane]);
    if (lane < 16) atomicAdd(&grads[G::OOUTW + lane], gout[lane]);
    if (lane == 0) atomicAdd(&grads[G::OOUTB], gout[16]);
}

// ---------------------------------------------------------------------------
// conv backward: dfeat -> conv/bias grads (atomics). One wave per window.
// ---------------------------------------------------------------------------
template <class G, int XP = 1, int SLIDE = 0>  // XP: lx row pad (bank
                                               // spread); SLIDE: r2 scheme
__global__ __launch_bounds__(256) void train_conv_bwd_kernel(
    const float* __restrict__ x,        // (SN, CIN, L)
    const float* __restrict__ stash,    // (SN, SC_SIZE)
    const float* __restrict__ dfeat,    // (SN, LIN)
    const float* __restrict__ wpack,
    float* __restrict__ grads, int SN)
{
    __shared__ float lw2[21];  // conv2 weights + bias
    // row stride L+1: L=120 dwords is 24 mod 32 banks (partial conflicts
    // on the stride-L xw reads in the conv1-grad loop); +1 spreads banks.
    // SLIDE=2 reads x straight from global (the sliding loop touches each
    // x element once, L1-streamed) — the 19 KB stage is then dead weight
    __shared__ float lx[SLIDE >= 2 ? 1 : 4]
                       [SLIDE >= 2 ? 1 : G::CIN * (G::L + XP)];
    __shared__ float lp1[4][4 * G::P1];
    __shared__ float lda1[4][4 * G::C1];
    __shared__ float lda2[4][G::C2];
    __shared__ float ldp1[4][4 * G::P1];
    __shared__ float gw1[4][4 * G::CIN * G::K1];  // per-wave conv1 w grads
    __shared__ float gsml[4][25];  // per-wave w2/b grads for the block fold
    for (int i = threadIdx.x; i < 21; i += 256) lw2[i] = wpack[G::OW2 + i];
    __syncthreads();
    const int wave = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
    // lane-local conv2 grad: each lane owns exactly ONE (c,k) element — a
    // scalar, NOT a lane-indexed array (runtime-indexed per-thread arrays
    // spill to scratch memory).
    float gw2_own = 0.f;
    float gb1l[4] = {0.f, 0.f, 0.f, 0.f};
    float gb2l = 0.f;
    for (int i = lane; i < 4 * G::CIN * G::K1; i += WAVE) gw1[wave][i] = 0.f;
    twsync();

    for (long win = blockIdx.x * 4 + wave; win < SN;
         win += (long)gridDim.x * 4) {
        const float* st = stash + win * G::SC_SIZE;
        const float* c1t = st + G::SC_C1T;
        const float* c2t = st + G::SC_C2T;
        const int* i1 = (const int*)(st + G::SC_I1);
        const int* i2 = (const int*)(st + G::SC_I2);
        const float* m1s = st + G::SC_M1;
        const float* m2s = st + G::SC_M2;
        const float* dfw = dfeat + win * G::LIN;
        const float* xin = x + win * (G::CIN * G::L);
        float* xw = lx[SLIDE >= 2 ? 0 : wave];
        if (SLIDE < 2)
            for (int i = lane; i < G::CIN * G::L; i += WAVE)
                xw[i + XP * (i / G::L)] = xin[i];  // padded-row store
        // recompute the (dropout-masked) pool1 output = conv2's input
        for (int o = lane; o < 4 * G::P1; o += WAVE) {
            const int c = o / G::P1, q = o % G::P1;
            lp1[wave][o] = c1t[c * G::C1 + q * G::PS + i1[o]] * m1s[o];
            ldp1[wave][o] = 0.f;
        }
        for (int sidx = lane; sidx < G::C2; sidx += WAVE) lda2[wave][sidx] = 0.f;
        twsync();
        // pool2 scatter: adjacent windows (stride 2, kernel 3) can collide
        // only between NEIGHBORING q — even/odd q phases write disjoint
        // ranges, so each phase is fully lane-parallel.
        #pragma unroll
        for (int phase = 0; phase < 2; ++phase) {
            for (int q = lane; q < G::LIN; q += WAVE)
                if ((q & 1) == phase) {
                    const int p = q * G::PS + i2[q];
                    const float d = dfw[q] * m2s[q];       // through dropout2
                    const float ct = c2t[p];
                    lda2[wave][p] += d * (1.f - ct * ct);  // through tanh'
                }
            twsync();
        }
        // conv2 grads: dW2[c][k] = sum_s da2[s] * p1[c][s+k]; db2 = sum da2
        if (lane < 20) {
            const int c = lane / 5, k = lane % 5;
            const float* da = lda2[wave];
            const float* pr = lp1[wave] + c * G::P1 + k;
            float a0 = 0.f, a1 = 0.f, a2 = 0.f;
            int s = 0;
            for (; s + 3 <= G::C2; s += 3) {
                a0 = fmaf(da[s + 0], pr[s + 0], a0);
                a1 = fmaf(da[s + 1], pr[s + 1], a1);
                a2 = fmaf(da[s + 2], pr[s + 2], a2);
            }
            for (; s < G::C2; ++s) a0 = fmaf(da[s], pr[s], a0);
            gw2_own += a0 + a1 + a2;
        }
        if (lane == 0) {
            float acc = 0.f;
            for (int s = 0; s < G::C2; ++s) acc += lda2[wave][s];
            gb2l += acc;
        }
        // dp1[c][p] = sum_k da2[p-k] * w2[c][k]
        for (int o = lane; o < 4 * G::P1; o += WAVE) {
            const int c = o / G::P1, p = o % G::P1;
            float acc = 0.f;
            #pragma unroll
            for (int k = 0; k < 5; ++k) {
                const int s = p - k;
                if (s >= 0 && s < G::C2)
                    acc = fmaf(lda2[wave][s], lw2[c * 5 + k], acc);
            }
            ldp1[wave][o] = acc;
        }
        // zero da1 then pool1 scatter (serialized per channel: 4 lanes)
        for (int o = lane; o < 4 * G::C1; o += WAVE) lda1[wave][o] = 0.f;
        twsync();
        // pool1 scatter, same even/odd-phase parallelization per channel
        #pragma unroll
        for (int phase = 0; phase < 2; ++phase) {
            for (int o = lane; o < 4 * G::P1; o += WAVE) {
                const int c = o / G::P1, q = o % G::P1;
                if ((q & 1) == phase) {
                    const int p = c * G::C1 + q * G::PS + i1[o];
                    const float ct = c1t[p];
                    lda1[wave][p] += ldp1[wave][o] *
                                     m1s[o] *           // through dropout1
                                     (1.f - ct * ct);
                }
            }
            twsync();
        }
        // conv1 grads: dW1[c][i][k] = sum_s da1[c][s] * x[i][s+k].
        if (SLIDE && 4 * G::CIN <= WAVE) {
            // Sliding-register scheme (round 2): lane owns (c, i) with all
            // K1 accumulators in registers and a K1-wide register window of
            // x sliding over s. LDS reads drop from 2 per FMA to 2 per
            // K1 FMAs (1 x + 1 da per s step; the da read is a broadcast
            // within each c group) — this loop was the kernel's LDS-
            // throughput bound (r1 PMC: LDS-wait + conflicts).
            const int c = lane / G::CIN;       // 4 groups
            const int i = lane % G::CIN;
            if (lane < 4 * G::CIN) {
                const float* da = lda1[wave] + c * G::C1;
                const float* xr = (SLIDE >= 2) ? (xin + i * G::L)
                                               : (xw + i * (G::L + XP));
                float acc[G::K1];
                #pragma unroll
                for (int k = 0; k < G::K1; ++k) acc[k] = 0.f;
                if (SLIDE == 3) {
                    // K1-deep double-buffered prefetch: the 1-ahead shift
                    // (below) exposes an L1-latency wait EVERY step (PMC:
                    // 76% SQ_WAIT_ANY); batching K1 independent loads per
                    // K1 steps amortizes it K1x.
                    float w[2 * G::K1];
                    #pragma unroll
                    for (int k = 0; k < 2 * G::K1; ++k)
                        w[k] = (k < G::L) ? xr[k] : 0.f;
                    for (int s0 = 0; s0 < G::C1; s0 += G::K1) {
                        #pragma unroll
                        for (int t = 0; t < G::K1; ++t) {
                            const int s = s0 + t;
                            if (s < G::C1) {
                                const float d = da[s];
                                #pragma unroll
                                for (int k = 0; k < G::K1; ++k)
                                    acc[k] = fmaf(d, w[t + k], acc[k]);
                            }
                        }
                        #pragma unroll
                        for (int k = 0; k < G::K1; ++k) w[k] = w[k + G::K1];
                        #pragma unroll
                        for (int t = 0; t < G::K1; ++t) {
                            const int idx = s0 + 2 * G::K1 + t;
                            w[G::K1 + t] = (idx < G::L) ? xr[idx] : 0.f;
                        }
                    }
                } else {
                    float w[G::K1];
                    #pragma unroll
                    for (int k = 0; k < G::K1; ++k) w[k] = xr[k];
                    for (int s = 0; s < G::C1; ++s) {
                        const float d = da[s];
                        #pragma unroll
                        for (int k = 0; k < G::K1; ++k)
                            acc[k] = fmaf(d, w[k], acc[k]);
                        #pragma unroll
                        for (int k = 0; k < G::K1 - 1; ++k) w[k] = w[k + 1];
                        w[G::K1 - 1] = (s + G::K1 < G::L) ? xr[s + G::K1]
                                                          : 0.f;
                    }
                }
                #pragma unroll
                for (int k = 0; k < G::K1; ++k)
                    gw1[wave][(c * G::CIN + i) * G::K1 + k] += acc[k];
            }
        } else {
            // 4 independent partial accumulators: keeps 8 LDS loads in
            // flight per iteration instead of exposing LDS latency every
            // element.
            for (int o = lane; o < 4 * G::CIN * G::K1; o += WAVE) {
                const int c = o / (G::CIN * G::K1);
                const int i = (o / G::K1) % G::CIN;
                const int k = o % G::K1;
                const float* da = lda1[wave] + c * G::C1;
                const float* xr = xw + i * (G::L + XP) + k;
                float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
                int s = 0;
                for (; s + 4 <= G::C1; s += 4) {
                    a0 = fmaf(da[s + 0], xr[s + 0], a0);
                    a1 = fmaf(da[s + 1], xr[s + 1], a1);
                    a2 = fmaf(da[s + 2], xr[s + 2], a2);
                    a3 = fmaf(da[s + 3], xr[s + 3], a3);
                }
                for (; s < G::C1; ++s) a0 = fmaf(da[s], xr[s], a0);
                gw1[wave][o] += (a0 + a1) + (a2 + a3);
            }
        }
        #pragma unroll
        for (int c = 0; c < 4; ++c) {
            // db1[c]: distribute the C1-sum across lanes then reduce
            float acc = 0.f;
            for (int s = lane; s < G::C1; s += WAVE)
                acc += lda1[wave][c * G::C1 + s];
            #pragma unroll
            for (int off = 32; off > 0; off >>= 1)
                acc += __shfl_xor(acc, off);
            if (lane == 0) gb1l[c] += acc;
        }
        twsync();
    }
    // fold the block's 4 per-wave accumulators first, then ONE atomicAdd
    // set per block: with tens of thousands of waves, per-wave atomics on
    // the 425 grad addresses serialize per-address — the block fold cuts
    // the atomic count 4x and the grid cap (launcher) another 4x.
    if (lane < 20) gsml[wave][lane] = gw2_own;
    if (lane == 0) {  // gb1l/gb2l accumulate in lane 0's registers
        #pragma unroll
        for (int c = 0; c < 4; ++c) gsml[wave][20 + c] = gb1l[c];
        gsml[wave][24] = gb2l;
    }
    __syncthreads();
    for (int i = threadIdx.x; i < 4 * G::CIN * G::K1; i += 256)
        atomicAdd(&grads[G::OW1 + i],
                  gw1[0][i] + gw1[1][i] + gw1[2][i] + gw1[3][i]);
    if (threadIdx.x < 25) {
        const float v = gsml[0][threadIdx.x] + gsml[1][threadIdx.x] +
                        gsml[2][threadIdx.x] + gsml[3][threadIdx.x];
        if (threadIdx.x < 20) atomicAdd(&grads[G::OW2 + threadIdx.x], v);
        else if (threadIdx.x < 24)
            atomicAdd(&grads[G::OB1 + threadIdx.x - 20], v);
        else atomicAdd(&grads[G::OB2], v);
    }
}

// ---------------------------------------------------------------------------
// fused Adam + grad zeroing
// ---------------------------------------------------------------------------
__global__ void train_adam_kernel(
    float* __restrict__ p, float* __restrict__ g, float* __restrict__ m,
    float* __restrict__ v, long n, float lr, float beta1, float beta2,
    float eps, int step)
{
    const float bc1 = 1.f - __powf(beta1, (float)step);
    const float bc2 = 1.f - __powf(beta2, (float)step);
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const float gi = g[i];
        const float mi = beta1 * m[i] + (1.f - beta1) * gi;
        const float vi = beta2 * v[i] + (1.f - beta2) * gi * gi;
        m[i] = mi;
        v[i] = vi;
        p[i] -= lr * (mi / bc1) / (sqrtf(vi / bc2) + eps);
        g[i] = 0.f;
    }
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
namespace {
template <class G>
int conv_fwd(const float* x, float* feat, float* stash, const float* wpack,
             int SN, float d1, float d2, unsigned long long seed,
             hipStream_t s) {
    if (SN <= 0) return 0;
    int grid = min((SN + 3) / 4, 8192);
    // Measured: the LDS-staged variant beats direct-global reads by ~8%
    // here (no transpose cost, and the dependent FMA chain prefers LDS
    // latency) — the OPPOSITE of the inference tlast conv. Default staged.
    const char* sx = getenv("TSKD_TRAIN_STAGE_X");
    // TSKD_CONVFWD_PAIR: 0 (default) = 2-accumulator loop; 1 = paired
    // outputs (measured +6.3%, A/B ref); 2 = 4-accumulator chain split
    const char* pp = getenv("TSKD_CONVFWD_PAIR");
    const int pair = pp ? atoi(pp) : 0;
    const bool staged = !(sx && sx[0] == '0');
    if (!staged) {
        if (pair == 1)
            hipLaunchKernelGGL((train_conv_fwd_kernel<G, false, 1>),
                               dim3(grid), dim3(256), 0, s, x, feat, stash,
                               wpack, SN, d1, d2, seed);
        else if (pair == 2)
            hipLaunchKernelGGL((train_conv_fwd_kernel<G, false, 2>),
                               dim3(grid), dim3(256), 0, s, x, feat, stash,
                               wpack, SN, d1, d2, seed);
        else
            hipLaunchKernelGGL((train_conv_fwd_kernel<G, false, 0>),
                               dim3(grid), dim3(256), 0, s, x, feat, stash,
                               wpack, SN, d1, d2, seed);
    } else {
        if (pair == 1)
            hipLaunchKernelGGL((train_conv_fwd_kernel<G, true, 1>),
                               dim3(grid), dim3(256), 0, s, x, feat, stash,
                               wpack, SN, d1, d2, seed);
        else if (pair == 2)
            hipLaunchKernelGGL((train_conv_fwd_kernel<G, true, 2>),
                               dim3(grid), dim3(256), 0, s, x, feat, stash,
                               wpack, SN, d1, d2, seed);
        else
            hipLaunchKernelGGL((train_conv_fwd_kernel<G, true, 0>),
                               dim3(grid), dim3(256), 0, s, x, feat, stash,
                               wpack, SN, d1, d2, seed);
    }
    return (int)hipGetLastError();
}
template <class G>
int lstm_fwd(const float* feat, const float* age, const float* wpack,
             float* stash, float* logits, int S, int B, float eps,
             hipStream_t s) {
    if (S <= 0) return 0;
    hipLaunchKernelGGL((train_lstm_fwd_kernel<G>), dim3(min(S, 32768)),
                       dim3(WAVE), 0, s, feat, age, wpack, stash, logits, S,
                       B, eps);
    return (int)hipGetLastError();
}
template <class G>
int lstm_bwd(const float* feat, const float* dlogit, const float* stash,
             const float* wpack, float* grads, float* dfeat, int S, int B,
             hipStream_t s) {
    if (S <= 0) return 0;
    // TSKD_LSTMBWD_RACC=1 moves the per-lane grad rows to registers —
    // measured +29% step time at the bench shape (ab_racc.log: the ~75
    // extra live VGPRs cost more occupancy than the LDS RMWs cost
    // bandwidth). Default stays the r1 LDS layout; variant kept as A/B.
    // TSKD_LSTMBWD_WPB=2 packs two sequences (waves) per block sharing
    // one LDS weight copy (halves the per-sequence weight-stage cost).
    const char* ra = getenv("TSKD_LSTMBWD_RACC");
    const char* wb = getenv("TSKD_LSTMBWD_WPB");
    const bool racc = ra && ra[0] == '1';
    const int wpb = wb ? atoi(wb) : 1;
    if (wpb == 2) {
        const int grid = min((S + 1) / 2, 32768);
        if (racc)
            hipLaunchKernelGGL((train_lstm_bwd_kernel<G, 1, 2>), dim3(grid),
                               dim3(2 * WAVE), 0, s, feat, dlogit, stash,
                               wpack, grads, dfeat, S, B);
        else
            hipLaunchKernelGGL((train_lstm_bwd_kernel<G, 0, 2>), dim3(grid),
                               dim3(2 * WAVE), 0, s, feat, dlogit, stash,
                               wpack, grads, dfeat, S, B);
    } else if (racc)
        hipLaunchKernelGGL((train_lstm_bwd_kernel<G, 1>),
                           dim3(min(S, 32768)), dim3(WAVE), 0, s, feat,
                           dlogit, stash, wpack, grads, dfeat, S, B);
    else
        hipLaunchKernelGGL((train_lstm_bwd_kernel<G, 0>),
                           dim3(min(S, 32768)), dim3(WAVE), 0, s, feat,
                           dlogit, stash, wpack, grads, dfeat, S, B);
    return (int)hipGetLastError();
}
template <class G>
int conv_bwd(const float* x, const float* stash, const float* dfeat,
             const float* wpack, float* grads, int SN, hipStream_t s) {
    if (SN <= 0) return 0;
    int cap = 8192;  // 2048 was a wash at S=512 but -2.3% at S=1024
    if (const char* e = getenv("TSKD_CONVBWD_GRID")) cap = atoi(e);
    int grid = min((SN + 3) / 4, cap);
    const char* xp = getenv("TSKD_CONVBWD_PAD");
    const char* sl = getenv("TSKD_CONVBWD_SLIDE");
    // 0 = r1 loop; 1 = sliding-register (-22.1% vs 0, ab_slide.log);
    // 2 (default) = sliding with global-x reads, no LDS stage (-4.5%
    // further, ab_slide2.log)
    const int slide = sl ? atoi(sl) : 2;
    if (xp && xp[0] == '0') {
        if (slide == 2)
            hipLaunchKernelGGL((train_conv_bwd_kernel<G, 0, 2>), dim3(grid),
                               dim3(256), 0, s, x, stash, dfeat, wpack,
                               grads, SN);
        else if (slide == 1)
            hipLaunchKernelGGL((train_conv_bwd_kernel<G, 0, 1>), dim3(grid),
                               dim3(256), 0, s, x, stash, dfeat, wpack,
                               grads, SN);
        else
            hipLaunchKernelGGL((train_conv_bwd_kernel<G, 0, 0>), dim3(grid),
                               dim3(256), 0, s, x, stash, dfeat, wpack,
                               grads, SN);
    } else {
        if (slide == 2)
            hipLaunchKernelGGL((train_conv_bwd_kernel<G, 1, 2>), dim3(grid),
                               dim3(256), 0, s, x, stash, dfeat, wpack,
                               grads, SN);
        else if (slide == 1)
            hipLaunchKernelGGL((train_conv_bwd_kernel<G, 1, 1>), dim3(grid),
                               dim3(256), 0, s, x, stash, dfeat, wpack,
                               grads, SN);
        else
            hipLaunchKernelGGL((train_conv_bwd_kernel<G, 1, 0>), dim3(grid),
                               dim3(256), 0, s, x, stash, dfeat, wpack,
                               grads, SN);
    }
    return (int)hipGetLastError();
}
// K14: fused batch accuracy — sigmoid -> round -> eq -> count
// in one pass (reference utils.py:122-134). round(sigmoid(x)) == 1 iff
// x > 0 (torch round-half-to-even sends sigmoid(0)=0.5 to 0), so the
// sigmoid never needs evaluating; NaN logits count incorrect (NaN.round()
// != target in torch). Wave shuffle reduction, one atomic per wave.
__global__ void batch_accuracy_kernel(const float* __restrict__ logits,
                                      const float* __restrict__ targets,
                                      float* __restrict__ out, long n) {
    int correct = 0;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const float v = logits[i];
        const float pred = (v > 0.0f) ? 1.0f : 0.0f;
        correct += (!__builtin_isnan(v) && pred == targets[i]) ? 1 : 0;
    }
    for (int off = WAVE / 2; off > 0; off >>= 1)
        correct += __shfl_down(correct, off);
    if ((threadIdx.x % WAVE) == 0 && correct)
        atomicAdd(out, (float)correct);
}

}  // namespace

extern "C" {

int tskd_train_batch_accuracy(const float* logits, const float* targets,
                              float* out, long n, void* stream) {
    if (n <= 0) return 0;
    int grid = (int)min((n + 255) / 256, (long)4096);
    hipLaunchKernelGGL(batch_accuracy_kernel, dim3(grid), dim3(256), 0,
                       (hipStream_t)stream, logits, targets, out, n);
    return (int)hipGetLastError();
}

int tskd_train_conv_fwd(const float* x, float* feat, float* stash,
                        const float* wpack, int SN, float drop1_p,
                        float drop2_p, unsigned long long seed, int variant,
                        void* s) {
    switch (variant) {
        case 0: return conv_fwd<TG5>(x, feat, stash, wpack, SN, drop1_p,
                                     drop2_p, seed, (hipStream_t)s);
        case 1: return conv_fwd<TG2>(x, feat, stash, wpack, SN, drop1_p,
                                     drop2_p, seed, (hipStream_t)s);
        case 2: return conv_fwd<TG4>(x, feat, stash, wpack, SN, drop1_p,
                                     drop2_p, seed, (hipStream_t)s);
    }
    return -1;
}

int tskd_train_lstm_fwd(const float* feat, const float* age,
                        const float* wpack, float* stash, float* logits,
                        int S, int B, float age_eps, int variant, void* s) {
    switch (variant) {
        case 0: return lstm_fwd<TG5>(feat, age, wpack, stash, logits, S, B,
                                     age_eps, (hipStream_t)s);
        case 1: return lstm_fwd<TG2>(feat, age, wpack, stash, logits, S, B,
                                     age_eps, (hipStream_t)s);
        case 2: return lstm_fwd<TG4>(feat, age, wpack, stash, logits, S, B,
                                     age_eps, (hipStream_t)s);
    }
    return -1;
}

int tskd_train_loss(const float* logits, const float* targets,
                    const float* age, float* dlogit_base, float* loss_sum,
                    long n, float pos_weight, float age_eps, void* s) {
    if (n <= 0) return 0;
    int grid = (int)min((n + 255) / 256, (long)4096);
    hipLaunchKernelGGL(train_loss_kernel, dim3(grid), dim3(256), 0,
                       (hipStream_t)s, logits, targets, age, dlogit_base,
                       loss_sum, n, pos_weight, age_eps);
    return (int)hipGetLastError();
}

int tskd_train_lstm_bwd(const float* feat, const float* dlogit,
                        const float* stash, const float* wpack, float* grads,
                        float* dfeat, int S, int B, int variant, void* s) {
    switch (variant) {
        case 0: return lstm_bwd<TG5>(feat, dlogit, stash, wpack, grads, dfeat,
                                     S, B, (hipStream_t)s);
        case 1: return lstm_bwd<TG2>(feat, dlogit, stash, wpack, grads, dfeat,
                                     S, B, (hipStream_t)s);
        case 2: return lstm_bwd<TG4>(feat, dlogit, stash, wpack, grads, dfeat,
                                     S, B, (hipStream_t)s);
    }
    return -1;
}

int tskd_train_conv_bwd(const float* x, const float* stash,
                        const float* dfeat, const float* wpack, float* grads,
                        int SN, int variant, void* s) {
    switch (variant) {
        case 0: return conv_bwd<TG5>(x, stash, dfeat, wpack, grads, SN,
                                     (hipStream_t)s);
        case 1: return conv_bwd<TG2>(x, stash, dfeat, wpack, grads, SN,
                                     (hipStream_t)s);
        case 2: return conv_bwd<TG4>(x, stash, dfeat, wpack, grads, SN,
                                     (hipStream_t)s);
    }
    return -1;
}

int tskd_train_adam(float* p, float* g, float* m, float* v, long n, float lr,
                    float beta1, float beta2, float eps, int step, void* s) {
    if (n <= 0) return 0;
    int grid = (int)min((n + 255) / 256, (long)1024);
    hipLaunchKernelGGL(train_adam_kernel, dim3(grid), dim3(256), 0,
                       (hipStream_t)s, p, g, m, v, n, lr, beta1, beta2, eps,
                       step);
    return (int)hipGetLastError();
}

int tskd_train_stash_sizes(int variant, int* conv_words, int* lstm_words) {
    switch (variant) {
        case 0: *conv_words = TG5::SC_SIZE; *lstm_words = TG5::SL_SIZE; return 0;
        case 1: *conv_words = TG2::SC_SIZE; *lstm_words = TG2::SL_SIZE; return 0;
        case 2: *conv_words = TG4::SC_SIZE; *lstm_words = TG4::SL_SIZE; return 0;
    }
    return -1;
}

}  // extern "C"
