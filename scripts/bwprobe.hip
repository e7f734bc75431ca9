// HBM read-bandwidth ceiling probe (calibration tool, not a product kernel).
//
// Measures the best-achievable pure-read rate on gfx950 across load widths,
// nontemporal hints and grid sizes, to price the pipeline's ingest stage
// against the REAL platform ceiling (profiles/r01: ingest 4.7 TB/s vs ATen
// copy 4.62 TB/s — is there more?).
//
// Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC scripts/bwprobe.hip
//        -o scripts/_bwprobe.so
#include <hip/hip_runtime.h>

typedef float f32x4_ __attribute__((ext_vector_type(4)));
typedef unsigned int u32x4_ __attribute__((ext_vector_type(4)));

// Each variant sums `n` floats (16 B-aligned base) and folds the result to
// out[block] so the loads cannot be optimized away.
template <int VEC, bool NT>
__global__ void read_probe(const float* __restrict__ p, long n4,
                           float* __restrict__ out) {
    const f32x4_* v = (const f32x4_*)p;
    f32x4_ acc = {0.f, 0.f, 0.f, 0.f};
    const long stride = (long)gridDim.x * blockDim.x;
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    for (; i + (VEC - 1) * stride < n4; i += VEC * stride) {
        #pragma unroll
        for (int j = 0; j < VEC; ++j) {
            f32x4_ t = NT ? __builtin_nontemporal_load(&v[i + j * stride])
                          : v[i + j * stride];
            acc += t;
        }
    }
    for (; i < n4; i += stride) acc += NT
        ? __builtin_nontemporal_load(&v[i]) : v[i];
    float s = acc.x + acc.y + acc.z + acc.w;
    __shared__ float red[256];
    red[threadIdx.x] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t = 0.f;
        for (int j = 0; j < 256; ++j) t += red[j];
        out[blockIdx.x] = t;
    }
}


// Variant 6: mirror ingest_dense's request pattern — 16-lane groups each
// sweeping a separate 1248 B region (4 regions per wave-task), 5-deep.
// Variant 7: SAME task shape (4992 B span per wave-task) but whole-wave
// contiguous lane mapping. 6 vs 7 isolates the request-pattern variable.
__global__ void read_probe_grouped(const float* __restrict__ p, long nbytes,
                                   float* __restrict__ out, int contiguous) {
    const int wlane = threadIdx.x % 64;
    const int grp = wlane / 16, gl = wlane % 16;
    const long ntasks = nbytes / 4992;
    f32x4_ acc = {0.f, 0.f, 0.f, 0.f};
    for (long t = (long)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
         t < ntasks; t += (long)gridDim.x * (blockDim.x / 64)) {
        if (contiguous) {
            const f32x4_* base = (const f32x4_*)((const char*)p + t * 4992);
            const int oct = 312;
            for (int b = wlane; b < oct; b += 5 * 64) {
                #pragma unroll
                for (int u = 0; u < 5; ++u) {
                    const int pp = b + u * 64;
                    acc += __builtin_nontemporal_load(&base[pp < oct ? pp : 0]);
                }
            }
        } else {
            const f32x4_* base = (const f32x4_*)((const char*)p + t * 4992 +
                                                 grp * 1248);
            const int oct = 78;
            for (int b = gl; b < oct; b += 5 * 16) {
                #pragma unroll
                for (int u = 0; u < 5; ++u) {
                    const int pp = b + u * 16;
                    acc += __builtin_nontemporal_load(&base[pp < oct ? pp : 0]);
                }
            }
        }
    }
    float s = acc.x + acc.y + acc.z + acc.w;
    __shared__ float red[256];
    red[threadIdx.x] = s;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t2 = 0.f;
        for (int j = 0; j < 256; ++j) t2 += red[j];
        out[blockIdx.x] = t2;
    }
}


// Variant 8: grouped pattern + the REAL ingest consumption (bf16 unpack,
// NaN select, count) — isolates the VALU-consumption cost from the
// request pattern (variant 6) and the pure-read ceiling (variant 7).
__device__ __forceinline__ float bf16u_(unsigned short h) {
    union { unsigned int i; float f; } u;
    u.i = ((unsigned int)h) << 16;
    return u.f;
}
__global__ void read_probe_consume(const float* __restrict__ p, long nbytes,
                                   float* __restrict__ out) {
    const int wlane = threadIdx.x % 64;
    const int grp = wlane / 16, gl = wlane % 16;
    const long ntasks = nbytes / 4992;
    float sum = 0.f, cnt = 0.f;
    for (long t = (long)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
         t < ntasks; t += (long)gridDim.x * (blockDim.x / 64)) {
        const char* base = (const char*)p + t * 4992 + grp * 1248;
        const int oct = 78;
        for (int b = gl; b < oct; b += 5 * 16) {
            union { u32x4_ q; unsigned short h[8]; } v[5];
            #pragma unroll
            for (int u = 0; u < 5; ++u) {
                const int pp = b + u * 16;
                v[u].q = __builtin_nontemporal_load(
                    (const u32x4_*)(base + (pp < oct ? pp : 0) * 16));
            }
            #pragma unroll
            for (int u = 0; u < 5; ++u) {
                if (b + u * 16 < oct) {
                    #pragma unroll
                    for (int j = 0; j < 8; ++j) {
                        const float f = bf16u_(v[u].h[j]);
                        if (!isnan(f)) { sum += f; cnt += 1.f; }
                    }
                }
            }
        }
    }
    __shared__ float red[256];
    red[threadIdx.x] = sum + cnt;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t2 = 0.f;
        for (int j = 0; j < 256; ++j) t2 += red[j];
        out[blockIdx.x] = t2;
    }
}


// Variant 9: grouped + PACKED consumption (1 shift + 1 and + 2 adds per
// dword, no NaN selects, no counts) — the mode-2 ingest fast path's cost.
__global__ void read_probe_packed(const float* __restrict__ p, long nbytes,
                                  float* __restrict__ out) {
    const int wlane = threadIdx.x % 64;
    const int grp = wlane / 16, gl = wlane % 16;
    const long ntasks = nbytes / 4992;
    float a = 0.f, b2 = 0.f;
    for (long t = (long)blockIdx.x * (blockDim.x / 64) + threadIdx.x / 64;
         t < ntasks; t += (long)gridDim.x * (blockDim.x / 64)) {
        const char* base = (const char*)p + t * 4992 + grp * 1248;
        const int oct = 78;
        for (int b = gl; b < oct; b += 5 * 16) {
            union { u32x4_ q; unsigned int d[4]; } v[5];
            #pragma unroll
            for (int u = 0; u < 5; ++u) {
                const int pp = b + u * 16;
                v[u].q = __builtin_nontemporal_load(
                    (const u32x4_*)(base + (pp < oct ? pp : 0) * 16));
            }
            #pragma unroll
            for (int u = 0; u < 5; ++u) {
                if (b + u * 16 < oct) {
                    #pragma unroll
                    for (int j = 0; j < 4; ++j) {
                        const unsigned int d = v[u].d[j];
                        a += __uint_as_float(d << 16);
                        b2 += __uint_as_float(d & 0xffff0000u);
                    }
                }
            }
        }
    }
    __shared__ float red[256];
    red[threadIdx.x] = a + b2;
    __syncthreads();
    if (threadIdx.x == 0) {
        float t2 = 0.f;
        for (int j = 0; j < 256; ++j) t2 += red[j];
        out[blockIdx.x] = t2;
    }
}

extern "C" int bw_probe(const float* p, long n, float* out, int variant,
                        int grid, void* stream) {
    const long n4 = n / 4;
    hipStream_t st = (hipStream_t)stream;
    switch (variant) {
        case 0: hipLaunchKernelGGL((read_probe<1, true>), dim3(grid),
                                   dim3(256), 0, st, p, n4, out); break;
        case 1: hipLaunchKernelGGL((read_probe<2, true>), dim3(grid),
                                   dim3(256), 0, st, p, n4, out); break;
        case 2: hipLaunchKernelGGL((read_probe<4, true>), dim3(grid),
                                   dim3(256), 0, st, p, n4, out); break;
        case 3: hipLaunchKernelGGL((read_probe<1, false>), dim3(grid),
                                   dim3(256), 0, st, p, n4, out); break;
        case 4: hipLaunchKernelGGL((read_probe<2, false>), dim3(grid),
                                   dim3(256), 0, st, p, n4, out); break;
        case 5: hipLaunchKernelGGL((read_probe<4, false>), dim3(grid),
                                   dim3(256), 0, st, p, n4, out); break;
        case 6: hipLaunchKernelGGL(read_probe_grouped, dim3(grid), dim3(256),
                                   0, st, p, n * 4L, out, 0); break;
        case 7: hipLaunchKernelGGL(read_probe_grouped, dim3(grid), dim3(256),
                                   0, st, p, n * 4L, out, 1); break;
        case 8: hipLaunchKernelGGL(read_probe_consume, dim3(grid), dim3(256),
                                   0, st, p, n * 4L, out); break;
        case 9: hipLaunchKernelGGL(read_probe_packed, dim3(grid), dim3(256),
                                   0, st, p, n * 4L, out); break;
    }
    return (int)hipGetLastError();
}
