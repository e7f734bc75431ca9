// Fused streaming-preprocess kernels — CDNA4 (gfx950) HIP.
//
// Replaces the reference's Spark Structured Streaming stage (reference
// bin/processStream.py:105-218 + predictStream.py window assembly :96-142;
// SURVEY.md §2.6 K13) with GPU-resident per-(stream, channel) ring buffers
// and three kernels:
//
//   1. ingest_dense / ingest_events: raw samples -> 5-s bucket (sum, count).
//      Dense path: one wavefront per bucket, coalesced strided loads +
//      wave shuffle reduction. Sparse path (irregular event batches, the
//      real numerics records at fs=1/60 Hz): atomicAdd into buckets.
//
//   2. window_fill: the Spark groupBy(key, channel, window(180 s, 5 s))
//      .agg(avg) — a window STARTING at grid g covers buckets [g, g+36) and
//      its average is sum(raw)/count(raw) over those buckets — followed by
//      the reference's ffill -> bfill -> fillna(0) (processStream.py:114-123).
//      ffill carries state ACROSS trigger batches (documented improvement
//      over the reference, whose ffill restarts per 60-s micro-batch).
//
//   3. window_gather: assemble model windows (S, B, C, 120) from the last
//      120 processed grid points per stream (predictStream.py's 600 s/60 s
//      window -> (1, 10, 120) tensor), batched B windows back at a given
//      grid stride. Missing channels come out zero (never-filled grid = 0).
//
// Ring-buffer layout (all fp32, indexed mod G):
//   bsum, bcnt : (S, C, G)  raw-sample sum / count per 5-s bucket
//   proc       : (S, C, G)  processed (filled) window averages by START grid
//   last_val   : (S, C)     ffill carry (NaN = no value seen yet)
// Head indices live on the host (python StreamEngine); kernels take them as
// arguments so the whole trigger step can be captured in a hipGraph.

#include <hip/hip_runtime.h>
#include <cstdlib>
#include <math.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_;
typedef __attribute__((ext_vector_type(4))) float f32x4_;

__device__ __forceinline__ float bf16_to_f32_(unsigned short u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

// Device-resident ring indices (hipGraph mode): dstate[0]=head,
// dstate[1]=nproc. A null dstate falls back to the host-passed values.
__device__ __forceinline__ long ring_head(const long long* dstate, long h) {
    return dstate ? (long)dstate[0] : h;
}
__device__ __forceinline__ long ring_nproc(const long long* dstate, long p) {
    return dstate ? (long)dstate[1] : p;
}

__global__ void advance_state_kernel(long long* dstate, int nb, int np) {
    if (threadIdx.x == 0 && blockIdx.x == 0) {
        dstate[0] += nb;
        dstate[1] += np;
    }
}

typedef float f32x2p_ __attribute__((ext_vector_type(2)));

// Same-wave LDS RAW fence (see mycnn_kernels.hip wave_sync).
__device__ __forceinline__ void wsync_() {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
}

// ---------------------------------------------------------------------------
// 1a. Dense ingest: raw (S, CIN, T) -> NB = T/bucket_len new buckets/channel
// ---------------------------------------------------------------------------
// One wave processes FOUR consecutive buckets of one (stream, channel) row
// in 16-lane groups — a 625-sample bucket is 78 oct-loads, so a whole-wave
// bucket wastes 39% of lanes on the ragged tail; 16-lane groups keep ~95%
// busy (78/16 = 4.9 balanced iterations per group).
#define ING_GRP 4   // buckets per wave
#define ING_GL 16   // lanes per bucket group
template <class DT, int IMODE = 5>
__global__ void ingest_dense_kernel(
    const DT* __restrict__ raw,     // (S, CIN, T)
    float* __restrict__ bsum,       // (S, C, G), element stride BST
    float* __restrict__ bcnt,       // (= bsum+1 when packed (sum,cnt) pairs)
    const int* __restrict__ chan_map,  // (CIN) raw row -> wire channel
    int S, int CIN, int C, int T, int G,
    int bucket_len, long head_in,   // buckets written at [head, head+NB)
    const long long* __restrict__ dstate, int BST)
{
    const long head = ring_head(dstate, head_in);
    const int head_mod = (int)(head % G);  // hoisted: no 64-bit mod per task
    const int NB = T / bucket_len;
    const int NBG = (NB + ING_GRP - 1) / ING_GRP;
    const long nwaves = (long)S * CIN * NBG;
    const int wlane = threadIdx.x % WAVE;
    const int grp = wlane / ING_GL;     // which of the wave's 4 buckets
    const int lane = wlane % ING_GL;    // lane within the bucket group
    for (long w = (long)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
         w < nwaves; w += (long)gridDim.x * (blockDim.x / WAVE)) {
        // 32-bit decode: nwaves < 2^31 in any real config; 64-bit div/mod
        // per task measured as a visible fraction of the wave's cycle
        // budget (12 tasks x ~5 KB each)
        const int wi = (int)w;
        const int bg = wi % NBG;
        const int wrem = wi / NBG;
        const int cin = wrem % CIN;
        const int s = wrem / CIN;
        const int b = bg * ING_GRP + grp;
        const bool active = b < NB;
        const DT* src = raw + ((long)s * CIN + cin) * T +
                        (long)(active ? b : 0) * bucket_len;
        float sum = 0.f, cnt = 0.f;
        // 16B-aligned vector body with scalar head/tail peel (bucket offsets
        // like 625 samples are not 16B-aligned).
        if (active) {
            if constexpr (IMODE == 5 && sizeof(DT) == 2) {
                // peel-free: cover the bucket with ALIGNED oct-chunks and
                // mask the (at most two) boundary chunks in-register —
                // removes both scalar peel loops per bucket
                const unsigned short* su = (const unsigned short*)src;
                const int mis = (int)(((size_t)su & 15) / 2);  // elems before
                const u32x4_* vp = (const u32x4_*)(su - mis);
                // ceil: the final partial chunk is loaded and masked
                // in-register (measured 4% faster than a scalar tail) —
                // EXCEPT for the very last bucket of the whole tensor,
                // where the <= 14 B over-read would pass the allocation:
                // that one task floors the chunk count and takes the
                // scalar tail. The masked head read-back (<= 14 B before
                // the bucket) stays inside the raw tensor for every bucket
                // but (0,0,0), which is 16 B-aligned (mis = 0).
                const bool at_end = s == S - 1 && cin == CIN - 1 &&
                                    b == NB - 1;
                const int nch = (mis + bucket_len + (at_end ? 0 : 7)) / 8;
                for (int base = lane; base < nch; base += 5 * ING_GL) {
                    union { u32x4_ q; unsigned short h[8]; } v[5];
                    #pragma unroll
                    for (int u = 0; u < 5; ++u) {
                        const int pp = base + u * ING_GL;
                        v[u].q = __builtin_nontemporal_load(
                            &vp[pp < nch ? pp : 0]);
                    }
                    #pragma unroll
                    for (int u = 0; u < 5; ++u) {
                        const int pp = base + u * ING_GL;
                        if (pp >= nch) continue;
                        const int idx0 = pp * 8 - mis;  // sample idx of h[0]
                        if (idx0 >= 0 && idx0 + 8 <= bucket_len) {
                            #pragma unroll
                            for (int j = 0; j < 8; ++j) {
                                const float f = bf16_to_f32_(v[u].h[j]);
                                if (!isnan(f)) { sum += f; cnt += 1.f; }
                            }
                        } else {  // boundary chunk: mask by sample index
                            #pragma unroll
                            for (int j = 0; j < 8; ++j) {
                                const unsigned k = (unsigned)(idx0 + j);
                                if (k < (unsigned)bucket_len) {
                                    const float f = bf16_to_f32_(v[u].h[j]);
                                    if (!isnan(f)) { sum += f; cnt += 1.f; }
                                }
                            }
                        }
                    }
                }
                if (at_end)
                    for (int i = nch * 8 - mis + lane; i < bucket_len;
                         i += ING_GL) {  // tensor-final bucket only
                        const float f = bf16_to_f32_(su[i]);
                        if (!isnan(f)) { sum += f; cnt += 1.f; }
                    }
            } else if constexpr (sizeof(DT) == 2) {
                int pre = (int)(((16 - ((size_t)src & 15)) & 15) / 2);
                if (pre > bucket_len) pre = bucket_len;
                for (int i = lane; i < pre; i += ING_GL) {
                    const float f = bf16_to_f32_((unsigned short)src[i]);
                    if (!isnan(f)) { sum += f; cnt += 1.f; }
                }
                const int oct = (bucket_len - pre) / 8;
                const u32x4_* vp = (const u32x4_*)((const unsigned short*)src + pre);
                if constexpr (IMODE == 1 || IMODE == 4) {
                    // 5-deep load ILP with per-element NaN selects; mode 4
                    // splits the accumulation across two chains (even/odd
                    // element) — the single 40-add chain is the lane's
                    // critical path once loads are in flight
                    float s1 = 0.f, n1 = 0.f;
                    for (int base = lane; base < oct; base += 5 * ING_GL) {
                        union { u32x4_ q; unsigned short h[8]; } v[5];
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            const int pp = base + u * ING_GL;
                            v[u].q = __builtin_nontemporal_load(
                                &vp[pp < oct ? pp : 0]);
                        }
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            if (base + u * ING_GL < oct) {
                                #pragma unroll
                                for (int j = 0; j < 8; ++j) {
                                    const float f = bf16_to_f32_(v[u].h[j]);
                                    if constexpr (IMODE == 4) {
                                        if (j & 1) {
                                            if (!isnan(f)) { s1 += f;
                                                             n1 += 1.f; }
                                            continue;
                                        }
                                    }
                                    if (!isnan(f)) { sum += f; cnt += 1.f; }
                                }
                            }
                        }
                    }
                    sum += s1;
                    cnt += n1;
                } else if constexpr (IMODE == 2) {
                    // 5 independent nontemporal loads in flight per lane
                    // (read-ceiling probe: 1-deep streams measure ~5.9 TB/s,
                    // 4..5-deep ~7.0 TB/s). Out-of-range lanes re-load chunk
                    // 0 (clamped index) so the loads issue unconditionally,
                    // and their contribution is masked after.
                    //
                    // NaN handling is DEFERRED: the fast path unpacks each
                    // dword into its two bf16 lanes with 2 bit-ops + 2 adds
                    // (vs 5 VALU/element for the select chain — at 7 TB/s
                    // the select version is VALU-issue-bound). A NaN input
                    // poisons the group sum; the group then redoes its
                    // bucket with the exact per-element masked loop (rare:
                    // NaN means a dropped sample).
                    float bA = 0.f, bB = 0.f;
                    for (int base = lane; base < oct; base += 5 * ING_GL) {
                        union { u32x4_ q; unsigned int d[4]; } v[5];
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            const int pp = base + u * ING_GL;
                            v[u].q = __builtin_nontemporal_load(
                                &vp[pp < oct ? pp : 0]);
                        }
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            if (base + u * ING_GL < oct) {
                                #pragma unroll
                                for (int j = 0; j < 4; ++j) {
                                    const unsigned int d = v[u].d[j];
                                    bA += __uint_as_float(d << 16);
                                    bB += __uint_as_float(d & 0xffff0000u);
                                }
                            }
                        }
                    }
                    float body = bA + bB;
                    float chk = body;
                    #pragma unroll
                    for (int off = 8; off > 0; off >>= 1)
                        chk += __shfl_xor(chk, off);
                    if (!isnan(chk)) {
                        sum += body;
                        cnt += 8.f * (float)(oct > lane
                                             ? (oct - 1 - lane) / ING_GL + 1
                                             : 0);
                    } else {  // rare: masked redo of this bucket's body
                        for (int p = lane; p < oct; p += ING_GL) {
                            union { u32x4_ q; unsigned short h[8]; } v;
                            v.q = __builtin_nontemporal_load(&vp[p]);
                            #pragma unroll
                            for (int j = 0; j < 8; ++j) {
                                const float f = bf16_to_f32_(v.h[j]);
                                if (!isnan(f)) { sum += f; cnt += 1.f; }
                            }
                        }
                    }
                } else {
                    for (int p = lane; p < oct; p += ING_GL) {
                        union { u32x4_ q; unsigned short h[8]; } v;
                        // raw samples are consumed once: stream past L2
                        v.q = __builtin_nontemporal_load(&vp[p]);
                        #pragma unroll
                        for (int j = 0; j < 8; ++j) {
                            const float f = bf16_to_f32_(v.h[j]);
                            if (!isnan(f)) { sum += f; cnt += 1.f; }
                        }
                    }
                }
                for (int i = pre + oct * 8 + lane; i < bucket_len; i += ING_GL) {
                    const float f = bf16_to_f32_((unsigned short)src[i]);
                    if (!isnan(f)) { sum += f; cnt += 1.f; }
                }
            } else {
                int pre = (int)(((16 - ((size_t)src & 15)) & 15) / 4);
                if (pre > bucket_len) pre = bucket_len;
                for (int i = lane; i < pre; i += ING_GL) {
                    const float f = (float)src[i];
                    if (!isnan(f)) { sum += f; cnt += 1.f; }
                }
                const int quad = (bucket_len - pre) / 4;
                const f32x4_* vp = (const f32x4_*)((const float*)src + pre);
                if constexpr (IMODE == 1) {
                    for (int base = lane; base < quad; base += 5 * ING_GL) {
                        f32x4_ v[5];
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            const int pp = base + u * ING_GL;
                            v[u] = __builtin_nontemporal_load(
                                &vp[pp < quad ? pp : 0]);
                        }
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            if (base + u * ING_GL < quad) {
                                if (!isnan(v[u].x)) { sum += v[u].x; cnt += 1.f; }
                                if (!isnan(v[u].y)) { sum += v[u].y; cnt += 1.f; }
                                if (!isnan(v[u].z)) { sum += v[u].z; cnt += 1.f; }
                                if (!isnan(v[u].w)) { sum += v[u].w; cnt += 1.f; }
                            }
                        }
                    }
                } else if constexpr (IMODE == 2) {
                    // same deferred-NaN fast path as the bf16 body
                    float bA = 0.f, bB = 0.f;
                    for (int base = lane; base < quad; base += 5 * ING_GL) {
                        f32x4_ v[5];
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            const int pp = base + u * ING_GL;
                            v[u] = __builtin_nontemporal_load(
                                &vp[pp < quad ? pp : 0]);
                        }
                        #pragma unroll
                        for (int u = 0; u < 5; ++u) {
                            if (base + u * ING_GL < quad) {
                                bA += v[u].x + v[u].z;
                                bB += v[u].y + v[u].w;
                            }
                        }
                    }
                    float body = bA + bB;
                    float chk = body;
                    #pragma unroll
                    for (int off = 8; off > 0; off >>= 1)
                        chk += __shfl_xor(chk, off);
                    if (!isnan(chk)) {
                        sum += body;
                        cnt += 4.f * (float)(quad > lane
                                             ? (quad - 1 - lane) / ING_GL + 1
                                             : 0);
                    } else {
                        for (int p = lane; p < quad; p += ING_GL) {
                            const f32x4_ v = __builtin_nontemporal_load(&vp[p]);
                            if (!isnan(v.x)) { sum += v.x; cnt += 1.f; }
                            if (!isnan(v.y)) { sum += v.y; cnt += 1.f; }
                            if (!isnan(v.z)) { sum += v.z; cnt += 1.f; }
                            if (!isnan(v.w)) { sum += v.w; cnt += 1.f; }
                        }
                    }
                } else {
                    for (int p = lane; p < quad; p += ING_GL) {
                        const f32x4_ v = __builtin_nontemporal_load(&vp[p]);
                        if (!isnan(v.x)) { sum += v.x; cnt += 1.f; }
                        if (!isnan(v.y)) { sum += v.y; cnt += 1.f; }
                        if (!isnan(v.z)) { sum += v.z; cnt += 1.f; }
                        if (!isnan(v.w)) { sum += v.w; cnt += 1.f; }
                    }
                }
                for (int i = pre + quad * 4 + lane; i < bucket_len; i += ING_GL) {
                    const float f = (float)src[i];
                    if (!isnan(f)) { sum += f; cnt += 1.f; }
                }
            }
        }
        // reduce within each 16-lane bucket group
        #pragma unroll
        for (int off = 8; off > 0; off >>= 1) {
            sum += __shfl_xor(sum, off);
            cnt += __shfl_xor(cnt, off);
        }
        if (lane == 0 && active) {
            const int c = chan_map[cin];
            int bi = head_mod + b;
            if (bi >= G) bi -= G;      // b < NB <= G: one subtract suffices
            const long idx = (((long)s * C + c) * G + bi) * BST;
            bsum[idx] = sum;
            bcnt[idx] = cnt;
        }
    }
}


// IMODE=3 experiment: each 16-lane group owns TWO buckets (b and b+4) so a
// wave keeps 10 independent oct-loads in flight and amortizes the per-task
// index math / reduce / store bubble over twice the bytes. bf16 only (the
// serving path); fp32 falls back to the 5-deep kernel in the launcher.
__global__ void ingest_dense_pair_kernel(
    const unsigned short* __restrict__ raw, float* __restrict__ bsum,
    float* __restrict__ bcnt, const int* __restrict__ chan_map,
    int S, int CIN, int C, int T, int G, int bucket_len, long head_in,
    const long long* __restrict__ dstate, int BST)
{
    const long head = ring_head(dstate, head_in);
    const int head_mod = (int)(head % G);
    const int NB = T / bucket_len;
    const int SPAN = 2 * ING_GRP;                 // 8 buckets per wave
    const int NBG = (NB + SPAN - 1) / SPAN;
    const long nwaves = (long)S * CIN * NBG;
    const int wlane = threadIdx.x % WAVE;
    const int grp = wlane / ING_GL;
    const int lane = wlane % ING_GL;
    for (long w = (long)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
         w < nwaves; w += (long)gridDim.x * (blockDim.x / WAVE)) {
        const int wi = (int)w;
        const int bg = wi % NBG;
        const int wrem = wi / NBG;
        const int cin = wrem % CIN;
        const int s = wrem / CIN;
        const int b0 = bg * SPAN + grp, b1 = b0 + ING_GRP;
        const bool a0 = b0 < NB, a1 = b1 < NB;
        const unsigned short* row = raw + ((long)s * CIN + cin) * T;
        const unsigned short* src0 = row + (long)(a0 ? b0 : 0) * bucket_len;
        const unsigned short* src1 = row + (long)(a1 ? b1 : 0) * bucket_len;
        float sum0 = 0.f, cnt0 = 0.f, sum1 = 0.f, cnt1 = 0.f;
        int pre0 = (int)(((16 - ((size_t)src0 & 15)) & 15) / 2);
        int pre1 = (int)(((16 - ((size_t)src1 & 15)) & 15) / 2);
        if (pre0 > bucket_len) pre0 = bucket_len;
        if (pre1 > bucket_len) pre1 = bucket_len;
        if (a0)
            for (int i = lane; i < pre0; i += ING_GL) {
                const float f = bf16_to_f32_(src0[i]);
                if (!isnan(f)) { sum0 += f; cnt0 += 1.f; }
            }
        if (a1)
            for (int i = lane; i < pre1; i += ING_GL) {
                const float f = bf16_to_f32_(src1[i]);
                if (!isnan(f)) { sum1 += f; cnt1 += 1.f; }
            }
        const int oct0 = a0 ? (bucket_len - pre0) / 8 : 0;
        const int oct1 = a1 ? (bucket_len - pre1) / 8 : 0;
        const u32x4_* vp0 = (const u32x4_*)(src0 + pre0);
        const u32x4_* vp1 = (const u32x4_*)(src1 + pre1);
        const int octm = oct0 > oct1 ? oct0 : oct1;
        for (int base = lane; base < octm; base += 5 * ING_GL) {
            union U { u32x4_ q; unsigned short h[8]; } v0[5], v1[5];
            #pragma unroll
            for (int u = 0; u < 5; ++u) {
                const int pp = base + u * ING_GL;
                v0[u].q = __builtin_nontemporal_load(&vp0[pp < oct0 ? pp : 0]);
                v1[u].q = __builtin_nontemporal_load(&vp1[pp < oct1 ? pp : 0]);
            }
            #pragma unroll
            for (int u = 0; u < 5; ++u) {
                const int pp = base + u * ING_GL;
                if (pp < oct0) {
                    #pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const float f = bf16_to_f32_(v0[u].h[j]);
                        if (!isnan(f)) { sum0 += f; cnt0 += 1.f; }
                    }
                }
                if (pp < oct1) {
                    #pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const float f = bf16_to_f32_(v1[u].h[j]);
                        if (!isnan(f)) { sum1 += f; cnt1 += 1.f; }
                    }
                }
            }
        }
        if (a0)
            for (int i = pre0 + oct0 * 8 + lane; i < bucket_len; i += ING_GL) {
                const float f = bf16_to_f32_(src0[i]);
                if (!isnan(f)) { sum0 += f; cnt0 += 1.f; }
            }
        if (a1)
            for (int i = pre1 + oct1 * 8 + lane; i < bucket_len; i += ING_GL) {
                const float f = bf16_to_f32_(src1[i]);
                if (!isnan(f)) { sum1 += f; cnt1 += 1.f; }
            }
        #pragma unroll
        for (int off = 8; off > 0; off >>= 1) {
            sum0 += __shfl_xor(sum0, off);
            cnt0 += __shfl_xor(cnt0, off);
            sum1 += __shfl_xor(sum1, off);
            cnt1 += __shfl_xor(cnt1, off);
        }
        if (lane == 0) {
            const int c = chan_map[cin];
            const long rowo = ((long)s * C + c) * G;
            if (a0) {
                int bi = head_mod + b0;
                if (bi >= G) bi -= G;
                bsum[(rowo + bi) * BST] = sum0;
                bcnt[(rowo + bi) * BST] = cnt0;
            }
            if (a1) {
                int bi = head_mod + b1;
                if (bi >= G) bi -= G;
                bsum[(rowo + bi) * BST] = sum1;
                bcnt[(rowo + bi) * BST] = cnt1;
            }
        }
    }
}

// ---------------------------------------------------------------------------
// 1b. Sparse ingest: event tuples (stream, chan, grid_bucket, value)
// ---------------------------------------------------------------------------
__global__ void ingest_events_kernel(
    const int* __restrict__ ev_stream, const int* __restrict__ ev_chan,
    const long* __restrict__ ev_bucket, const float* __restrict__ ev_val,
    float* __restrict__ bsum, float* __restrict__ bcnt,
    int C, int G, long n_events, long min_bucket, int BST)
{
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_events;
         i += (long)gridDim.x * blockDim.x) {
        const long b = ev_bucket[i];
        if (b < min_bucket) continue;  // behind the watermark: dropped (late data)
        const float v = ev_val[i];
        if (isnan(v)) continue;
        const long idx = (((long)ev_stream[i] * C + ev_chan[i]) * G
                          + b % G) * BST;
        atomicAdd(&bsum[idx], v);
        atomicAdd(&bcnt[idx], 1.f);
    }
}

// Zero the bucket slots about to be (re)used: [head, head+nb) mod G.
__global__ void clear_buckets_kernel(
    float* __restrict__ bsum, float* __restrict__ bcnt,
    int S, int C, int G, long head, int nb, int BST)
{
    const long n = (long)S * C * nb;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const int j = (int)(i % nb);
        const long sc = i / nb;
        const long idx = (sc * G + (head + j) % G) * BST;
        bsum[idx] = 0.f;
        bcnt[idx] = 0.f;
    }
}

// ---------------------------------------------------------------------------
// 2. Sliding-window average + ffill/bfill/zero-fill
//    Processes proc grid points [phead, phead+np): window starting at grid j
//    averages buckets [j, j+win_buckets). One thread per (s, c): the fill is
//    an order-dependent scan (tiny np per trigger; np can be large in replay
//    catch-up, still coalesced across the S*C threads).
// ---------------------------------------------------------------------------
// One WAVE per (stream, channel): coalesced bucket loads, LDS-staged sliding
// sums (each lane owns one output point per 64-point chunk), the
// order-dependent ffill scan runs on lane 0 over LDS (np is small per
// trigger; large only in replay catch-up, still chunked).
__global__ __launch_bounds__(256) void window_fill_kernel(
    const float* __restrict__ bsum, const float* __restrict__ bcnt,
    float* __restrict__ proc, float* __restrict__ last_val,
    int S, int C, int G, long phead_in, int np, int win_buckets,
    const long long* __restrict__ dstate, int BST)
{
    const long phead = ring_nproc(dstate, phead_in);
    constexpr int CHUNK = 64;             // output points per iteration
    const int MAXW = 64;                  // win_buckets <= 64 (default 36)
    __shared__ float lds_s[4][CHUNK + 64];  // bucket sums (CHUNK+MAXW-1 used)
    __shared__ float lds_c[4][CHUNK + 64];
    __shared__ float lds_v[4][CHUNK];       // window values / filled output
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    float* ls = lds_s[wave];
    float* lc = lds_c[wave];
    float* lv = lds_v[wave];
    (void)MAXW;

    const long nsc = (long)S * C;
    for (long sc = (long)blockIdx.x * 4 + wave; sc < nsc;
         sc += (long)gridDim.x * 4) {
        const float* bs = bsum + sc * G * BST;
        const float* bc = bcnt + sc * G * BST;
        float* pr = proc + sc * G;
        float carry = last_val[sc];
        // After ffill-with-carry, NaNs can only be a PREFIX of the batch
        // (once any value is seen, carry is set forever). bfill therefore
        // reduces to: fill the prefix with the first valid value (or 0).
        int nan_prefix = 0;
        float first_val = NAN;

        for (int j0 = 0; j0 < np; j0 += CHUNK) {
            const int jn = min(CHUNK, np - j0);
            const int nload = jn + win_buckets - 1;
            // coalesced bucket loads into LDS
            for (int i = lane; i < nload; i += WAVE) {
                const long idx = (phead + j0 + i) % G;
                if (BST == 2) {  // packed (sum,cnt) pair: one 8 B load
                    const f32x2p_ v = *(const f32x2p_*)(bs + idx * 2);
                    ls[i] = v.x;
                    lc[i] = v.y;
                } else {
                    ls[i] = bs[idx];
                    lc[i] = bc[idx];
                }
            }
            wsync_();
            // lane j: sliding raw-sample mean of window starting at j0+j
            if (lane < jn) {
                float sum = 0.f, cnt = 0.f;
                for (int k = 0; k < win_buckets; ++k) {
                    sum += ls[lane + k];
                    cnt += lc[lane + k];
                }
                lv[lane] = (cnt > 0.f) ? sum / cnt : NAN;
            }
            wsync_();
            // lane 0: ffill scan (order-dependent; jn <= 64 steps)
            if (lane == 0) {
                for (int j = 0; j < jn; ++j) {
                    const float v = lv[j];
                    if (isnan(v)) {
                        if (isnan(carry)) ++nan_prefix;  // still before 1st value
                        else lv[j] = carry;
                    } else {
                        if (isnan(carry)) first_val = v;
                        carry = v;
                    }
                }
            }
            wsync_();
            if (lane < jn) pr[(phead + j0 + lane) % G] = lv[lane];
            wsync_();
            carry = __shfl(carry, 0);
        }
        nan_prefix = __shfl(nan_prefix, 0);
        first_val = __shfl(first_val, 0);
        if (nan_prefix > 0) {
            const float fill = isnan(first_val) ? 0.f : first_val;  // bfill|0
            for (int i = lane; i < nan_prefix; i += WAVE)
                pr[(phead + i) % G] = fill;
        }
        if (lane == 0) last_val[sc] = carry;
    }
}



__device__ __forceinline__ unsigned short f32_to_bf16_(float f) {
    union { float f; unsigned int i; } u;
    u.f = f;
    const unsigned int r = u.i + 0x7fffu + ((u.i >> 16) & 1);  // RNE
    return (unsigned short)(r >> 16);
}

// Timelast gather v2: one thread per (s, b, t) writes ALL C channels of
// one timepoint. In the (S, B, WIN, C) layout the channel axis is
// CONTIGUOUS, so the stores are C/2 packed dwords (vs v1's 2-byte
// scattered stores — profile: 72 us vs the std-layout gather's 30 us at
// S=16384); the reads coalesce per channel (consecutive t lanes hit
// consecutive proc addresses). Even C only (wire space is 10).
template <class OT>
__global__ void window_gather_tlast2_kernel(
    const float* __restrict__ proc, OT* __restrict__ out,  // (S, B, WIN, C)
    int S, int C, int G, int B, int WIN, int stride, long end_in,
    const long long* __restrict__ dstate, int end_extra)
{
    const long end = dstate ? (long)dstate[1] + end_extra : end_in;
    const long n = (long)S * B * WIN;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const int t = (int)(i % WIN);
        const int b = (int)((i / WIN) % B);
        const int s = (int)(i / ((long)WIN * B));
        const long wend = end - (long)(B - 1 - b) * stride;
        const long g = wend - WIN + t;
        const bool ok = g >= 0 && g < end && wend - WIN >= 0 && wend <= end;
        const int gi = ok ? (int)(g % G) : 0;
        const float* pr = proc + (long)s * C * G + gi;
        if constexpr (sizeof(OT) == 2) {
            unsigned int* od = (unsigned int*)(out + i * C);
            for (int c = 0; c < C; c += 2) {
                const unsigned int lo =
                    ok ? f32_to_bf16_(pr[(long)c * G]) : 0u;
                const unsigned int hi =
                    ok ? f32_to_bf16_(pr[(long)(c + 1) * G]) : 0u;
                od[c >> 1] = lo | (hi << 16);
            }
        } else {
            OT* od = out + i * C;
            for (int c = 0; c < C; ++c)
                od[c] = ok ? (OT)pr[(long)c * G] : (OT)0;
        }
    }
}

// Steady-state variant (np <= 16): the general kernel leaves 52 of 64 lanes
// idle when a trigger yields only NB=12 new grid points. Pack FOUR
// (stream, channel) rows per wave in 16-lane groups; the <= 51 bucket
// reads per row are consecutive-address (coalesced within the group) and
// L1-resident, so no LDS staging — only the 16-value ffill scan goes
// through LDS. Dispatch in tskd_preproc_window_fill (knob TSKD_FILL16).
__global__ __launch_bounds__(256) void window_fill16_kernel(
    const float* __restrict__ bsum, const float* __restrict__ bcnt,
    float* __restrict__ proc, float* __restrict__ last_val,
    int S, int C, int G, long phead_in, int np, int win_buckets,
    const long long* __restrict__ dstate, int BST)
{
    const long phead = ring_nproc(dstate, phead_in);
    __shared__ float lds_v[4][64];        // per wave: 4 groups x 16 values
    const int wave = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int grp = lane / 16, gl = lane % 16;
    float* lv = lds_v[wave] + grp * 16;

    const long nsc = (long)S * C;
    for (long sc0 = ((long)blockIdx.x * 4 + wave) * 4; sc0 < nsc;
         sc0 += (long)gridDim.x * 16) {
        const long sc = sc0 + grp;
        const bool live = sc < nsc;
        float carry = NAN, first_val = NAN, val = NAN;
        int nan_prefix = 0;
        if (live) {
            carry = last_val[sc];
            if (gl < np && BST == 2) {
                // packed (sum,cnt) pairs: ONE f32x2 load and ONE packed add
                // per bucket — half the loads and adds of the split-array
                // layout (the whole point of TSKD_PACKED_BUCKETS)
                const f32x2p_* pk = (const f32x2p_*)(bsum + sc * G * 2);
                const int start = (int)((phead + gl) % G);
                const int first = min(win_buckets, G - start);
                const int rem = win_buckets - first;
                f32x2p_ a0 = {0.f, 0.f}, a1 = {0.f, 0.f};
                f32x2p_ a2 = {0.f, 0.f}, a3 = {0.f, 0.f};
                const f32x2p_* p1 = pk + start;
                int k = 0;
                for (; k + 4 <= first; k += 4) {
                    a0 += p1[k];     a1 += p1[k + 1];
                    a2 += p1[k + 2]; a3 += p1[k + 3];
                }
                for (; k < first; ++k) a0 += p1[k];
                int j = 0;
                for (; j + 4 <= rem; j += 4) {
                    a0 += pk[j];     a1 += pk[j + 1];
                    a2 += pk[j + 2]; a3 += pk[j + 3];
                }
                for (; j < rem; ++j) a0 += pk[j];
                const f32x2p_ t = (a0 + a1) + (a2 + a3);
                val = (t.y > 0.f) ? t.x / t.y : NAN;
            } else if (gl < np) {
                const float* bs = bsum + sc * G;
                const float* bc = bcnt + sc * G;
                // split the (possibly wrapping) 36-bucket window into two
                // linear segments and sum with 4 independent accumulators:
                // the single dependent add chain was the lane's critical
                // path (the loads are L1/L2 hits shared across the group)
                const int start = (int)((phead + gl) % G);
                const int first = min(win_buckets, G - start);
                const int rem = win_buckets - first;
                float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
                float c0 = 0.f, c1 = 0.f, c2 = 0.f, c3 = 0.f;
                const float* p1 = bs + start;
                const float* q1 = bc + start;
                int k = 0;
                for (; k + 4 <= first; k += 4) {
                    a0 += p1[k];     c0 += q1[k];
                    a1 += p1[k + 1]; c1 += q1[k + 1];
                    a2 += p1[k + 2]; c2 += q1[k + 2];
                    a3 += p1[k + 3]; c3 += q1[k + 3];
                }
                for (; k < first; ++k) { a0 += p1[k]; c0 += q1[k]; }
                int j = 0;
                for (; j + 4 <= rem; j += 4) {
                    a0 += bs[j];     c0 += bc[j];
                    a1 += bs[j + 1]; c1 += bc[j + 1];
                    a2 += bs[j + 2]; c2 += bc[j + 2];
                    a3 += bs[j + 3]; c3 += bc[j + 3];
                }
                for (; j < rem; ++j) { a0 += bs[j]; c0 += bc[j]; }
                const float sum = (a0 + a1) + (a2 + a3);
                const float cnt = (c0 + c1) + (c2 + c3);
                val = (cnt > 0.f) ? sum / cnt : NAN;
            }
        }
        lv[gl] = val;
        wsync_();
        if (live && gl == 0) {            // serial ffill scan per group
            for (int j = 0; j < np; ++j) {
                const float v = lv[j];
                if (isnan(v)) {
                    if (isnan(carry)) ++nan_prefix;   // before first value
                    else lv[j] = carry;
                } else {
                    if (isnan(carry)) first_val = v;
                    carry = v;
                }
            }
            last_val[sc] = carry;
        }
        wsync_();
        if (live && gl < np) {
            nan_prefix = __shfl(nan_prefix, grp * 16);
            first_val = __shfl(first_val, grp * 16);
            float outv = lv[gl];
            if (gl < nan_prefix)          // bfill | fillna(0) prefix
                outv = isnan(first_val) ? 0.f : first_val;
            proc[sc * G + (phead + gl) % G] = outv;
        }
        wsync_();
    }
}

// ---------------------------------------------------------------------------
// 3. Window gather: (S, B, C, WIN) model inputs from processed grid.
//    Window b (b = 0..B-1) covers grid [end - (B-1-b)*stride - WIN,
//    end - (B-1-b)*stride). Grid points never produced (g < 0) read as 0.
// ---------------------------------------------------------------------------

// Vectorized: each thread emits 4 consecutive time points (WIN % 4 == 0),
// reading contiguous proc and writing one 8 B (bf16) / 16 B (f32) store.
// TLAST: emit (S, B, WIN, C) instead — the layout the LDS-free MFMA conv
// kernel consumes (scalar scattered stores; the read side stays coalesced).
template <class OT, bool TLAST = false>
__global__ void window_gather_kernel(
    const float* __restrict__ proc,
    OT* __restrict__ out,            // (S, B, C, WIN) or (S, B, WIN, C)
    int S, int C, int G, int B, int WIN, int stride, long end_in,
    const long long* __restrict__ dstate, int end_extra)
{
    const long end = dstate ? (long)dstate[1] + end_extra : end_in;
    const int WQ = WIN / 4;
    const long n = (long)S * B * C * WQ;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
         i += (long)gridDim.x * blockDim.x) {
        const int tq = (int)(i % WQ);
        const int c = (int)((i / WQ) % C);
        const int b = (int)((i / ((long)WQ * C)) % B);
        const int s = (int)(i / ((long)WQ * C * B));
        const long wend = end - (long)(B - 1 - b) * stride;
        const long g0 = wend - WIN + tq * 4;
        const float* pr = proc + ((long)s * C + c) * G;
        float v[4];
        #pragma unroll
        for (int j = 0; j < 4; ++j) {
            const long g = g0 + j;
            v[j] = (g >= 0 && g < end && wend - WIN >= 0 && wend <= end)
                       ? pr[g % G] : 0.f;
        }
        if constexpr (TLAST) {
            const long ob = ((long)s * B + b) * C * WIN;
            #pragma unroll
            for (int j = 0; j < 4; ++j) {
                const long o = ob + (long)(tq * 4 + j) * C + c;
                if constexpr (sizeof(OT) == 2) out[o] = f32_to_bf16_(v[j]);
                else out[o] = (OT)v[j];
            }
        } else {
            const long o = (((long)s * B + b) * C + c) * WIN + (long)tq * 4;
            if constexpr (sizeof(OT) == 2) {
                union { unsigned short h[4]; unsigned long long u; } pk;
                #pragma unroll
                for (int j = 0; j < 4; ++j) pk.h[j] = f32_to_bf16_(v[j]);
                *(unsigned long long*)(out + o) = pk.u;
            } else {
                float4 pk = {v[0], v[1], v[2], v[3]};
                *(float4*)((float*)out + o) = pk;
            }
        }
    }
}

// ---------------------------------------------------------------------------
// Optional per-window z-normalization (BASELINE.json names a z-normalize
// stage; the REFERENCE has none — SURVEY.md §7 fidelity note — so this is
// an opt-in post-pass, off by default): each (stream, batch, channel) row of
// WIN points is normalized to (v - mean) / max(std, eps) in place.
// One wave per row: shuffle-reduced mean/var, vectorized rewrite.
// ---------------------------------------------------------------------------
template <class OT>
__global__ void window_znorm_kernel(OT* __restrict__ w,  // (rows, WIN)
                                    long rows, int WIN, float eps)
{
    const int lane = threadIdx.x % WAVE;
    for (long r = (long)blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
         r < rows; r += (long)gridDim.x * (blockDim.x / WAVE)) {
        OT* row = w + r * WIN;
        float sum = 0.f, sq = 0.f;
        for (int t = lane; t < WIN; t += WAVE) {
            float v;
            if constexpr (sizeof(OT) == 2) v = bf16_to_f32_((unsigned short)row[t]);
            else v = (float)row[t];
            sum += v;
            sq = fmaf(v, v, sq);
        }
        #pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            sum += __shfl_xor(sum, off);
            sq += __shfl_xor(sq, off);
        }
        const float mean = sum / WIN;
        const float var = fmaxf(sq / WIN - mean * mean, 0.f);
        const float inv = 1.0f / fmaxf(sqrtf(var), eps);
        for (int t = lane; t < WIN; t += WAVE) {
            float v;
            if constexpr (sizeof(OT) == 2) v = bf16_to_f32_((unsigned short)row[t]);
            else v = (float)row[t];
            v = (v - mean) * inv;
            if constexpr (sizeof(OT) == 2) row[t] = (OT)f32_to_bf16_(v);
            else row[t] = (OT)v;
        }
    }
}

// ---------------------------------------------------------------------------
// extern "C" launchers
// ---------------------------------------------------------------------------
static inline int grid_for(long n, int block) {
    long g = (n + block - 1) / block;
    if (g > 16384) g = 16384;
    if (g < 1) g = 1;
    return (int)g;
}

extern "C" {

int tskd_preproc_ingest_dense(const void* raw, int raw_is_bf16,
                              float* bsum, float* bcnt, const int* chan_map,
                              int S, int CIN, int C, int T, int G,
                              int bucket_len, long head,
                              const long long* dstate, int bst,
                              void* stream) {
    hipStream_t st = (hipStream_t)stream;
    const int NB = T / bucket_len;
    if (NB <= 0 || S <= 0) return 0;
    const long nwaves = (long)S * CIN * ((NB + ING_GRP - 1) / ING_GRP);
    // read-ceiling probe: 32768 blocks measures ~3% above the 16384 cap.
    // TSKD_INGEST_GRID caps the block count: the probe saturates HBM from
    // ~2048 blocks, so a small grid leaves CU wave slots free for a
    // co-scheduled model chain (TriggerGraph overlap mode).
    long cap = 32768;
    if (const char* gc = getenv("TSKD_INGEST_GRID")) cap = atol(gc);
    long blocks = (nwaves + 3) / 4;
    if (blocks > cap) blocks = cap;
    if (blocks < 1) blocks = 1;
    const int grid = (int)blocks;
    // TSKD_INGEST_ILP: 0 = 1-deep select loop, 1 = 5-deep ILP + scalar
    // peel, 2 = deferred-NaN packed, 3 = paired-bucket, 4 = dual-acc,
    // 5 (default, bf16) = 5-deep ILP + masked boundary chunks (no scalar
    // peel; within-run A/B: beats 1 by 6.9%; fp32 falls back to 1).
    int mode = 5;
    if (const char* ilp = getenv("TSKD_INGEST_ILP")) mode = atoi(ilp);
    if (raw_is_bf16) {
        const unsigned short* rp = (const unsigned short*)raw;
        if (mode == 3) {
            hipLaunchKernelGGL(ingest_dense_pair_kernel, dim3(grid), dim3(256),
                               0, st, rp, bsum, bcnt, chan_map, S, CIN, C, T,
                               G, bucket_len, head, dstate, bst);
            return (int)hipGetLastError();
        }
        if (mode == 0)
            hipLaunchKernelGGL((ingest_dense_kernel<unsigned short, 0>),
                               dim3(grid), dim3(256), 0, st, rp, bsum, bcnt,
                               chan_map, S, CIN, C, T, G, bucket_len, head,
                               dstate, bst);
        else if (mode == 1)
            hipLaunchKernelGGL((ingest_dense_kernel<unsigned short, 1>),
                               dim3(grid), dim3(256), 0, st, rp, bsum, bcnt,
                               chan_map, S, CIN, C, T, G, bucket_len, head,
                               dstate, bst);
        else if (mode == 4)
            hipLaunchKernelGGL((ingest_dense_kernel<unsigned short, 4>),
                               dim3(grid), dim3(256), 0, st, rp, bsum, bcnt,
                               chan_map, S, CIN, C, T, G, bucket_len, head,
                               dstate, bst);
        else if (mode == 5)
            hipLaunchKernelGGL((ingest_dense_kernel<unsigned short, 5>),
                               dim3(grid), dim3(256), 0, st, rp, bsum, bcnt,
                               chan_map, S, CIN, C, T, G, bucket_len, head,
                               dstate, bst);
        else
            hipLaunchKernelGGL((ingest_dense_kernel<unsigned short, 2>),
                               dim3(grid), dim3(256), 0, st, rp, bsum, bcnt,
                               chan_map, S, CIN, C, T, G, bucket_len, head,
                               dstate, bst);
    } else {
        const float* rp = (const float*)raw;
        if (mode >= 3) mode = 1;  // fp32: modes 3-5 are bf16-only
        if (mode == 0)
            hipLaunchKernelGGL((ingest_dense_kernel<float, 0>), dim3(grid),
                               dim3(256), 0, st, rp, bsum, bcnt, chan_map, S,
                               CIN, C, T, G, bucket_len, head, dstate, bst);
        else if (mode == 1)
            hipLaunchKernelGGL((ingest_dense_kernel<float, 1>), dim3(grid),
                               dim3(256), 0, st, rp, bsum, bcnt, chan_map, S,
                               CIN, C, T, G, bucket_len, head, dstate, bst);
        else
            hipLaunchKernelGGL((ingest_dense_kernel<float, 2>), dim3(grid),
                               dim3(256), 0, st, rp, bsum, bcnt, chan_map, S,
                               CIN, C, T, G, bucket_len, head, dstate, bst);
    }
    return (int)hipGetLastError();
}

int tskd_preproc_ingest_events(const int* ev_stream, const int* ev_chan,
                               const long* ev_bucket, const float* ev_val,
                               float* bsum, float* bcnt, int C, int G,
                               long n_events, long min_bucket, int bst,
                               void* stream) {
    if (n_events <= 0) return 0;
    hipLaunchKernelGGL(ingest_events_kernel, dim3(grid_for(n_events, 256)),
                       dim3(256), 0, (hipStream_t)stream, ev_stream, ev_chan,
                       ev_bucket, ev_val, bsum, bcnt, C, G, n_events,
                       min_bucket, bst);
    return (int)hipGetLastError();
}

int tskd_preproc_clear_buckets(float* bsum, float* bcnt, int S, int C, int G,
                               long head, int nb, int bst, void* stream) {
    if (nb <= 0) return 0;
    const long n = (long)S * C * nb;
    hipLaunchKernelGGL(clear_buckets_kernel, dim3(grid_for(n, 256)), dim3(256),
                       0, (hipStream_t)stream, bsum, bcnt, S, C, G, head, nb,
                       bst);
    return (int)hipGetLastError();
}

int tskd_preproc_window_fill(const float* bsum, const float* bcnt, float* proc,
                             float* last_val, int S, int C, int G, long phead,
                             int np, int win_buckets,
                             const long long* dstate, int bst, void* stream) {
    if (np <= 0) return 0;
    const char* f16 = getenv("TSKD_FILL16");
    if (np <= 16 && !(f16 && f16[0] == '0')) {
        // steady-state: 4 (stream, channel) rows per wave, 16-lane groups
        const long nsc4 = ((long)S * C + 15) / 16;
        hipLaunchKernelGGL(window_fill16_kernel, dim3(grid_for(nsc4, 1)),
                           dim3(256), 0, (hipStream_t)stream, bsum, bcnt,
                           proc, last_val, S, C, G, phead, np, win_buckets,
                           dstate, bst);
        return (int)hipGetLastError();
    }
    const long nsc = (long)S * C * WAVE;  // one wave per (stream, channel)
    hipLaunchKernelGGL(window_fill_kernel, dim3(grid_for(nsc, 256)), dim3(256),
                       0, (hipStream_t)stream, bsum, bcnt, proc, last_val, S,
                       C, G, phead, np, win_buckets, dstate, bst);
    return (int)hipGetLastError();
}

int tskd_preproc_advance_state(long long* dstate, int nb, int np,
                               void* stream) {
    hipLaunchKernelGGL(advance_state_kernel, dim3(1), dim3(64), 0,
                       (hipStream_t)stream, dstate, nb, np);
    return (int)hipGetLastError();
}

int tskd_preproc_window_gather(const float* proc, void* out, int out_is_bf16,
                               int out_timelast, int S, int C, int G, int B,
                               int WIN, int stride, long end,
                               const long long* dstate, int end_extra,
                               void* stream) {
    if (WIN % 4 != 0) return -3;  // vectorized gather needs WIN % 4 == 0
    const long n = (long)S * B * C * (WIN / 4);
    if (n <= 0) return 0;
    hipStream_t st = (hipStream_t)stream;
    const char* g2 = getenv("TSKD_GATHER_TLAST_V2");
    const bool tlast2 = C % 2 == 0 && !(g2 && g2[0] == '0');
    const long n2 = (long)S * B * WIN;  // v2: thread per (s, b, t)
    if (out_is_bf16) {
        if (out_timelast && tlast2)
            hipLaunchKernelGGL((window_gather_tlast2_kernel<unsigned short>),
                               dim3(grid_for(n2, 256)), dim3(256), 0, st,
                               proc, (unsigned short*)out, S, C, G, B, WIN,
                               stride, end, dstate, end_extra);
        else if (out_timelast)
            hipLaunchKernelGGL((window_gather_kernel<unsigned short, true>),
                               dim3(grid_for(n, 256)), dim3(256), 0, st, proc,
                               (unsigned short*)out, S, C, G, B, WIN, stride,
                               end, dstate, end_extra);
        else
            hipLaunchKernelGGL((window_gather_kernel<unsigned short>),
                               dim3(grid_for(n, 256)), dim3(256), 0, st, proc,
                               (unsigned short*)out, S, C, G, B, WIN, stride,
                               end, dstate, end_extra);
    } else {
        if (out_timelast && tlast2)
            hipLaunchKernelGGL((window_gather_tlast2_kernel<float>),
                               dim3(grid_for(n2, 256)), dim3(256), 0, st,
                               proc, (float*)out, S, C, G, B, WIN, stride,
                               end, dstate, end_extra);
        else if (out_timelast)
            hipLaunchKernelGGL((window_gather_kernel<float, true>),
                               dim3(grid_for(n, 256)), dim3(256), 0, st, proc,
                               (float*)out, S, C, G, B, WIN, stride, end,
                               dstate, end_extra);
        else
            hipLaunchKernelGGL((window_gather_kernel<float>),
                               dim3(grid_for(n, 256)), dim3(256), 0, st, proc,
                               (float*)out, S, C, G, B, WIN, stride, end,
                               dstate, end_extra);
    }
    return (int)hipGetLastError();
}

int tskd_preproc_window_znorm(void* w, int is_bf16, long rows, int win,
                              float eps, void* stream) {
    if (rows <= 0) return 0;
    const int grid = grid_for(rows * WAVE, 256);
    if (is_bf16)
        hipLaunchKernelGGL((window_znorm_kernel<unsigned short>), dim3(grid),
                           dim3(256), 0, (hipStream_t)stream,
                           (unsigned short*)w, rows, win, eps);
    else
        hipLaunchKernelGGL((window_znorm_kernel<float>), dim3(grid), dim3(256),
                           0, (hipStream_t)stream, (float*)w, rows, win, eps);
    return (int)hipGetLastError();
}

}  // extern "C"
