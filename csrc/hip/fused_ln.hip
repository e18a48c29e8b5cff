// Fused LayerNorm for bf16 activations with fp32 parameters — gfx950.
//
// Why: under torch autocast, LayerNorm runs in fp32, inserting
// bf16<->f32 cast passes around all 24 LNs of BERT-base (measured
// 1.6 ms of a 12.7 ms step) and a multi-kernel backward. These kernels
// keep activations bf16 end to end and accumulate in f32.
//
// Shape: x is [N rows][H] with H % 8 == 0 (hidden size, e.g. 768/3072).
// One wave (64 lanes) owns one row: vec-8 loads land each row in
// registers once; mean/var via __shfl_down tree; the normalize reuses the
// registers (single global read in forward). Backward is single-pass:
// per-row dx from wave-reduced stats, per-column dW/db partials
// accumulated in LDS and flushed through NSHADOW interleaved global
// accumulators (same contention pattern as fused_bn.hip).
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

namespace {

inline int env_int(const char *name, int dflt)
{
    const char *v = getenv(name);
    return v && *v ? atoi(v) : dflt;
}

constexpr int WAVE = 64;
constexpr int BLOCK = 256;              // 4 waves = 4 rows per block
constexpr int ROWS_PER_BLOCK = BLOCK / WAVE;
constexpr int NSHADOW = 8;
constexpr int MAX_K = 8;  // supports H <= 8 * 512 = 4096

typedef __attribute__((ext_vector_type(8))) unsigned short ushort8;

__device__ inline float b2f(unsigned short u)
{
    union {
        unsigned int i;
        float f;
    } v;
    v.i = (unsigned int)u << 16;
    return v.f;
}

__device__ inline unsigned short f2b(float f)
{
    union {
        unsigned int i;
        float f;
    } v;
    v.f = f;
    unsigned int r = v.i + 0x7fff + ((v.i >> 16) & 1);
    return (unsigned short)(r >> 16);
}

__device__ inline float wave_sum(float v)
{
    for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
    return __shfl(v, 0);
}

// ---- forward: y = (x - mean) * rstd * w + b ----
// K (vec8 iterations per lane) is a template parameter: runtime-indexed
// register arrays would be allocated in scratch memory (200x slower).
template <int K>
__global__ void ln_fwd_kernel(const unsigned short *__restrict__ x,
                              unsigned short *__restrict__ y,
                              const float *__restrict__ w,
                              const float *__restrict__ b,
                              float *__restrict__ save_mean,
                              float *__restrict__ save_rstd, long long N,
                              int H, float eps)
{
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;
    const long long row_step = (long long)gridDim.x * ROWS_PER_BLOCK;
    for (long long r = (long long)blockIdx.x * ROWS_PER_BLOCK + wid; r < N;
         r += row_step) {
        const unsigned short *xr = x + r * H;
        ushort8 v[K];
        float s = 0.f, q = 0.f;
#pragma unroll
        for (int k = 0; k < K; ++k) {
            const int h0 = (k * WAVE + lane) * 8;
            if (h0 < H) {
                v[k] = *(const ushort8 *)(xr + h0);
#pragma unroll
                for (int i = 0; i < 8; ++i) {
                    const float f = b2f(v[k][i]);
                    s += f;
                    q += f * f;
                }
            }
        }
        s = wave_sum(s);
        q = wave_sum(q);
        const float mean = s / (float)H;
        const float var = fmaxf(q / (float)H - mean * mean, 0.f);
        const float rstd = rsqrtf(var + eps);
        if (lane == 0) {
            save_mean[r] = mean;
            save_rstd[r] = rstd;
        }
#pragma unroll
        for (int k = 0; k < K; ++k) {
            const int h0 = (k * WAVE + lane) * 8;
            if (h0 < H) {
                ushort8 o;
#pragma unroll
                for (int i = 0; i < 8; ++i) {
                    const float xh = (b2f(v[k][i]) - mean) * rstd;
                    o[i] = f2b(fmaf(xh, w[h0 + i], b[h0 + i]));
                }
                *(ushort8 *)(y + r * H + h0) = o;
            }
        }
    }
}

// ---- backward (single pass) ----
// dx = rstd * (g - mean(g) - xhat * mean(g * xhat)),  g = dy * w
// dW[h] += sum_rows dy * xhat ; db[h] += sum_rows dy  (via LDS + shadows)
// ILP independent rows in flight per wave: the two serial phases per
// row (loads -> wave_sum shuffle chains -> store) pipeline across rows.
template <int K, int ILP>
__global__ void ln_bwd_kernel(const unsigned short *__restrict__ dy,
                              const unsigned short *__restrict__ x,
                              const float *__restrict__ w,
                              const float *__restrict__ save_mean,
                              const float *__restrict__ save_rstd,
                              long long N, int H,
                              unsigned short *__restrict__ dx,
                              float *__restrict__ wb_sums)
{
    extern __shared__ float lds[];  // 2*H floats {dw, db}
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;
    for (int i = threadIdx.x; i < 2 * H; i += BLOCK) lds[i] = 0.f;
    __syncthreads();

    // per-lane register accumulators for the column sums: every row
    // visits the SAME h-positions for a given (lane, k), so dW/db
    // partials stay in registers and hit LDS only once per block
    float accw[K][8] = {};
    float accb[K][8] = {};
    const long long row_step = (long long)gridDim.x * ROWS_PER_BLOCK;
    long long r = (long long)blockIdx.x * ROWS_PER_BLOCK + wid;
    for (; r + (ILP - 1) * row_step < N; r += ILP * row_step) {
        ushort8 dv[ILP][K], xv[ILP][K];
        float s1[ILP], s2[ILP], mean[ILP], rstd[ILP];
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
            const long long rj = r + j * row_step;
            mean[j] = save_mean[rj];
            rstd[j] = save_rstd[rj];
            s1[j] = s2[j] = 0.f;
#pragma unroll
            for (int k = 0; k < K; ++k) {
                const int h0 = (k * WAVE + lane) * 8;
                if (h0 < H) {
                    dv[j][k] = *(const ushort8 *)(dy + rj * H + h0);
                    xv[j][k] = *(const ushort8 *)(x + rj * H + h0);
                }
            }
        }
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
#pragma unroll
            for (int k = 0; k < K; ++k) {
                const int h0 = (k * WAVE + lane) * 8;
                if (h0 < H) {
#pragma unroll
                    for (int i = 0; i < 8; ++i) {
                        const float d = b2f(dv[j][k][i]);
                        const float xh =
                            (b2f(xv[j][k][i]) - mean[j]) * rstd[j];
                        s1[j] += d * w[h0 + i];
                        s2[j] += d * w[h0 + i] * xh;
                    }
                }
            }
        }
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
            s1[j] = wave_sum(s1[j]) / (float)H;
            s2[j] = wave_sum(s2[j]) / (float)H;
        }
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
            const long long rj = r + j * row_step;
#pragma unroll
            for (int k = 0; k < K; ++k) {
                const int h0 = (k * WAVE + lane) * 8;
                if (h0 < H) {
                    ushort8 o;
#pragma unroll
                    for (int i = 0; i < 8; ++i) {
                        const float d = b2f(dv[j][k][i]);
                        const float xh =
                            (b2f(xv[j][k][i]) - mean[j]) * rstd[j];
                        const float g = d * w[h0 + i];
                        o[i] = f2b(rstd[j] * (g - s1[j] - xh * s2[j]));
                        accw[k][i] += d * xh;
                        accb[k][i] += d;
                    }
                    *(ushort8 *)(dx + rj * H + h0) = o;
                }
            }
        }
    }
    for (; r < N; r += row_step) {
        const unsigned short *dyr = dy + r * H;
        const unsigned short *xr = x + r * H;
        const float mean = save_mean[r];
        const float rstd = save_rstd[r];
        ushort8 dv[K], xv[K];
        float s1 = 0.f, s2 = 0.f;
#pragma unroll
        for (int k = 0; k < K; ++k) {
            const int h0 = (k * WAVE + lane) * 8;
            if (h0 < H) {
                dv[k] = *(const ushort8 *)(dyr + h0);
                xv[k] = *(const ushort8 *)(xr + h0);
#pragma unroll
                for (int i = 0; i < 8; ++i) {
                    const float d = b2f(dv[k][i]);
                    const float xh = (b2f(xv[k][i]) - mean) * rstd;
                    const float g = d * w[h0 + i];
                    s1 += g;
                    s2 += g * xh;
                }
            }
        }
        s1 = wave_sum(s1) / (float)H;
        s2 = wave_sum(s2) / (float)H;
#pragma unroll
        for (int k = 0; k < K; ++k) {
            const int h0 = (k * WAVE + lane) * 8;
            if (h0 < H) {
                ushort8 o;
#pragma unroll
                for (int i = 0; i < 8; ++i) {
                    const float d = b2f(dv[k][i]);
                    const float xh = (b2f(xv[k][i]) - mean) * rstd;
                    const float g = d * w[h0 + i];
                    o[i] = f2b(rstd * (g - s1 - xh * s2));
                    accw[k][i] += d * xh;
                    accb[k][i] += d;
                }
                *(ushort8 *)(dx + r * H + h0) = o;
            }
        }
    }
#pragma unroll
    for (int k = 0; k < K; ++k) {
        const int h0 = (k * WAVE + lane) * 8;
        if (h0 < H) {
#pragma unroll
            for (int i = 0; i < 8; ++i) {
                atomicAdd(&lds[h0 + i], accw[k][i]);
                atomicAdd(&lds[H + h0 + i], accb[k][i]);
            }
        }
    }
    __syncthreads();
    float *shadow = wb_sums + (size_t)(blockIdx.x % NSHADOW) * 2 * H;
    for (int i = threadIdx.x; i < 2 * H; i += BLOCK) {
        atomicAdd(&shadow[i], lds[i]);
    }
}

// Fold the shadow copies of {dw, db} into copy 0.
__global__ void ln_fold_kernel(float *__restrict__ wb_sums, int H)
{
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= 2 * H) return;
    float acc = wb_sums[i];
    for (int k = 1; k < NSHADOW; ++k) acc += wb_sums[(size_t)k * 2 * H + i];
    wb_sums[i] = acc;
}

}  // namespace

extern "C" {

hipError_t kf_ln_fwd(const void *x, void *y, const void *w, const void *b,
                     void *save_mean, void *save_rstd, long long N, int H,
                     float eps, void *stream)
{
    if (H % 8 != 0 || H > WAVE * 8 * MAX_K) return hipErrorInvalidValue;
    long long blocks = (N + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK;
    if (blocks > 4096) blocks = 4096;
    if (blocks < 1) blocks = 1;
    const int K = (H + WAVE * 8 - 1) / (WAVE * 8);
#define CASE(KK)                                                            \
    hipLaunchKernelGGL((ln_fwd_kernel<KK>), dim3((uint32_t)blocks),         \
                       dim3(BLOCK), 0, (hipStream_t)stream,                 \
                       (const unsigned short *)x, (unsigned short *)y,      \
                       (const float *)w, (const float *)b,                  \
                       (float *)save_mean, (float *)save_rstd, N, H, eps)
    switch (K) {
    case 1: CASE(1); break;
    case 2: CASE(2); break;
    case 3: CASE(3); break;
    case 4: CASE(4); break;
    case 6: CASE(6); break;
    case 8: CASE(8); break;
    default: return hipErrorInvalidValue;
    }
#undef CASE
    return hipGetLastError();
}

hipError_t kf_ln_bwd(const void *dy, const void *x, const void *w,
                     const void *save_mean, const void *save_rstd,
                     long long N, int H, void *dx, void *wb_sums,
                     void *stream)
{
    if (H % 8 != 0 || H > WAVE * 8 * MAX_K) return hipErrorInvalidValue;
    long long blocks = (N + ROWS_PER_BLOCK - 1) / ROWS_PER_BLOCK;
    // KF_LN_CAPKB: min KiB of dy+x input per block (0 = off);
    // KF_LN_MAXBLK: hard cap. Same-box sweep on BERT-base (N=4096,
    // H=768): 64 KiB floor 3250 seq/s, 16 KiB 3196, 8 KiB 3133 — the
    // NSHADOW atomic tail outweighs extra occupancy on this small shape,
    // so the 64 KiB floor stays the default.
    static const int cap_kb = env_int("KF_LN_CAPKB", 64);
    static const int maxblk = env_int("KF_LN_MAXBLK", 2048);
    if (cap_kb > 0) {
        const long long by_bytes =
            (N * H * 4 + cap_kb * 1024 - 1) / (cap_kb * 1024);
        if (blocks > by_bytes) blocks = by_bytes;
    }
    if (blocks > maxblk) blocks = maxblk;
    if (blocks < 1) blocks = 1;
    const int K = (H + WAVE * 8 - 1) / (WAVE * 8);
    static const int ilp_env = env_int("KF_LN_ILP", 1);  // 2 measured -5% (VGPR pressure)
    const bool ilp2 = ilp_env >= 2;
#define CASE(KK)                                                            \
    do {                                                                    \
        if (ilp2)                                                           \
            hipLaunchKernelGGL((ln_bwd_kernel<KK, 2>),                      \
                               dim3((uint32_t)blocks), dim3(BLOCK),         \
                               2 * H * sizeof(float), (hipStream_t)stream,  \
                               (const unsigned short *)dy,                  \
                               (const unsigned short *)x,                   \
                               (const float *)w, (const float *)save_mean,  \
                               (const float *)save_rstd, N, H,              \
                               (unsigned short *)dx, (float *)wb_sums);     \
        else                                                                \
            hipLaunchKernelGGL((ln_bwd_kernel<KK, 1>),                      \
                               dim3((uint32_t)blocks), dim3(BLOCK),         \
                               2 * H * sizeof(float), (hipStream_t)stream,  \
                               (const unsigned short *)dy,                  \
                               (const unsigned short *)x,                   \
                               (const float *)w, (const float *)save_mean,  \
                               (const float *)save_rstd, N, H,              \
                               (unsigned short *)dx, (float *)wb_sums);     \
    } while (0)
    switch (K) {
    case 1: CASE(1); break;
    case 2: CASE(2); break;
    case 3: CASE(3); break;
    case 4: CASE(4); break;
    case 6: CASE(6); break;
    case 8: CASE(8); break;
    default: return hipErrorInvalidValue;
    }
#undef CASE
    return hipGetLastError();
}

hipError_t kf_ln_fold(void *wb_sums, int H, void *stream)
{
    hipLaunchKernelGGL(ln_fold_kernel, dim3((2 * H + 255) / 256), dim3(256),
                       0, (hipStream_t)stream, (float *)wb_sums, H);
    return hipGetLastError();
}

}  // extern "C"
