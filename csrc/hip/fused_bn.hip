// Fused BatchNorm(+residual)(+ReLU) kernels for NHWC bf16 activations,
// fp32 parameters — gfx950.
//
// Why: under torch autocast the ResNet hot path runs BN in fp32 (cast
// bf16->f32 copies + MIOpen spatial BN + separate residual-add and ReLU
// elementwise kernels + f32->bf16 casts) — measured 5.2 ms of a 13 ms
// ResNet-50 b64 step (profiles/resnet50_b64_bf16_steady_state_r01.md).
// These kernels read/write bf16 exactly once per pass, accumulate in f32,
// and fold residual-add + ReLU into the normalize pass.
//
// Layout: channels_last (NHWC): flat element i has channel i % C. All
// kernels vectorize 8 consecutive channels per lane (16 B loads, G13) and
// require C % 8 == 0 and C <= 8 * 256 (ResNet: 64..2048); the Python layer
// falls back to eager BN otherwise.
//
// Per-channel reductions: each lane owns one channel-octet in registers
// (f32x8 accumulators), reduces into LDS with shared-memory atomics, one
// global atomic per channel per block into NSHADOW interleaved shadow
// copies (Guideline 12). Measured alternatives that LOST: full-chip grids
// with the same atomics (2x slower on the big layers — atomic storms),
// and atomic-free per-block partial rows + a fold kernel (fold is
// latency-bound at ~21 us/call). Geometry stays env-tunable for sweeps:
//   KF_BN_CAPKB  — min KiB of input per block (0 = off); default 64
//   KF_BN_MAXBLK — hard block cap; default 2048
//   KF_BN_ILP    — rows in flight per lane in reduction kernels (2|4)
#include <hip/hip_bf16.h>
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdlib>

namespace {

constexpr int BLOCK = 256;
constexpr int NSHADOW = 8;

inline int env_int(const char *name, int dflt)
{
    const char *v = getenv(name);
    return v && *v ? atoi(v) : dflt;
}

inline long long bn_reduce_blocks(long long M, int C, int rows_per_blk)
{
    static const int cap_kb = env_int("KF_BN_CAPKB", 64);
    static const int maxblk = env_int("KF_BN_MAXBLK", 2048);
    long long blocks = (M + rows_per_blk - 1) / rows_per_blk;
    if (cap_kb > 0) {
        const long long by_bytes =
            (M * C * 2 + cap_kb * 1024 - 1) / (cap_kb * 1024);
        if (blocks > by_bytes) blocks = by_bytes;
    }
    if (blocks > maxblk) blocks = maxblk;
    if (blocks < 1) blocks = 1;
    return blocks;
}

inline int bn_ilp_stats()
{
    static const int ilp = env_int("KF_BN_ILP_STATS",
                                   env_int("KF_BN_ILP", 4));
    return ilp >= 4 ? 4 : 2;
}

inline int bn_ilp_bwd()
{
    static const int ilp = env_int("KF_BN_ILP_BWD",
                                   env_int("KF_BN_ILP", 2));
    return ilp >= 4 ? 4 : 2;
}

// channel-octets (8 ch / 16 B) per lane: 2 doubles the per-lane loads in
// flight (32 B) at the cost of 2x accumulator VGPRs. Requires C % 16 == 0.
inline int bn_oct()
{
    static const int oct = env_int("KF_BN_OCT", 1);
    return oct >= 2 ? 2 : 1;
}

inline int bn_ilp_stream()
{
    static const int ilp = env_int("KF_BN_ILP_STREAM", 2);
    return ilp >= 4 ? 4 : 2;
}

// non-temporal loads for the single-use activation streams (KF_BN_NT=1)
inline bool bn_nt()
{
    static const bool nt = env_int("KF_BN_NT", 0) != 0;
    return nt;
}

typedef __attribute__((ext_vector_type(8))) unsigned short ushort8;

template <bool NT>
__device__ inline ushort8 load8(const unsigned short *p)
{
    if (NT) return __builtin_nontemporal_load((const ushort8 *)p);
    return *(const ushort8 *)p;
}

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
    unsigned int r = v.i + 0x7fff + ((v.i >> 16) & 1);  // round-nearest-even
    return (unsigned short)(r >> 16);
}

// ---- pass 1: per-channel sum / sumsq ----
// x: [M][C] bf16; out sums: f32[NSHADOW][2*C] (pre-zeroed)
// OCT octets (8 channels / 16 B each) per lane; ILP rows in flight.
template <int ILP, int OCT, bool NT>
__global__ void bn_stats_kernel(const unsigned short *__restrict__ x,
                                long long M, int C,
                                float *__restrict__ sums)
{
    extern __shared__ float lds[];  // 2*C floats
    const int gpr = C / (8 * OCT);  // channel groups per row
    const int rows_per_blk = BLOCK / gpr;
    const int g = threadIdx.x % gpr;
    const int row_off = threadIdx.x / gpr;
    for (int i = threadIdx.x; i < 2 * C; i += BLOCK) lds[i] = 0.f;
    __syncthreads();

    float s[8 * OCT], q[8 * OCT];
#pragma unroll
    for (int k = 0; k < 8 * OCT; ++k) s[k] = q[k] = 0.f;
    if (row_off < rows_per_blk) {
        const long long row_step = (long long)gridDim.x * rows_per_blk;
        long long r = (long long)blockIdx.x * rows_per_blk + row_off;
        for (; r + (ILP - 1) * row_step < M; r += ILP * row_step) {
            ushort8 v[ILP][OCT];
#pragma unroll
            for (int j = 0; j < ILP; ++j) {
#pragma unroll
                for (int o = 0; o < OCT; ++o)
                    v[j][o] = load8<NT>(
                        x + (r + j * row_step) * C + g * 8 * OCT + o * 8);
            }
#pragma unroll
            for (int j = 0; j < ILP; ++j) {
#pragma unroll
                for (int o = 0; o < OCT; ++o) {
#pragma unroll
                    for (int k = 0; k < 8; ++k) {
                        const float f = b2f(v[j][o][k]);
                        s[o * 8 + k] += f;
                        q[o * 8 + k] += f * f;
                    }
                }
            }
        }
        for (; r < M; r += row_step) {
#pragma unroll
            for (int o = 0; o < OCT; ++o) {
                const ushort8 v = load8<NT>(x + r * C + g * 8 * OCT +
                                            o * 8);
#pragma unroll
                for (int k = 0; k < 8; ++k) {
                    const float f = b2f(v[k]);
                    s[o * 8 + k] += f;
                    q[o * 8 + k] += f * f;
                }
            }
        }
    }
#pragma unroll
    for (int k = 0; k < 8 * OCT; ++k) {
        atomicAdd(&lds[g * 8 * OCT + k], s[k]);
        atomicAdd(&lds[C + g * 8 * OCT + k], q[k]);
    }
    __syncthreads();
    float *shadow = sums + (size_t)(blockIdx.x % NSHADOW) * 2 * C;
    for (int i = threadIdx.x; i < 2 * C; i += BLOCK) {
        atomicAdd(&shadow[i], lds[i]);
    }
}

// ---- finalize: mean/var -> folded scale/shift + running stats ----
// sums: NSHADOW x {sum,sumsq}; outputs a = w*rstd, b = bias - mean*a;
// saves mean/rstd for backward; updates running stats (momentum).
__global__ void bn_finalize_kernel(float *__restrict__ sums,
                                   const float *__restrict__ weight,
                                   const float *__restrict__ bias,
                                   float *__restrict__ running_mean,
                                   float *__restrict__ running_var,
                                   float *__restrict__ save_mean,
                                   float *__restrict__ save_rstd,
                                   float *__restrict__ a,
                                   float *__restrict__ b, long long M,
                                   int C, float eps, float momentum)
{
    const int c = blockIdx.x * blockDim.x + threadIdx.x;
    if (c >= C) return;
    float s0 = 0.f, s1 = 0.f;
    for (int k = 0; k < NSHADOW; ++k) {
        s0 += sums[(size_t)k * 2 * C + c];
        s1 += sums[(size_t)k * 2 * C + C + c];
        // re-zero for the next step: kills the separate fill launch
        sums[(size_t)k * 2 * C + c] = 0.f;
        sums[(size_t)k * 2 * C + C + c] = 0.f;
    }
    const float mean = s0 / (float)M;
    const float var = fmaxf(s1 / (float)M - mean * mean, 0.f);
    const float rstd = rsqrtf(var + eps);
    save_mean[c] = mean;
    save_rstd[c] = rstd;
    const float ac = weight[c] * rstd;
    a[c] = ac;
    b[c] = bias[c] - mean * ac;
    if (momentum > 0.f) {
        running_mean[c] += momentum * (mean - running_mean[c]);
        const float unbiased =
            M > 1 ? var * (float)M / (float)(M - 1) : var;
        running_var[c] += momentum * (unbiased - running_var[c]);
    }
}

// ---- pass 2: y = [relu]( a*x + b [+ res] ) ----
// Fixed channel-octet per thread: per-channel params live in registers for
// the whole row loop; ILP rows in flight per lane (env KF_BN_ILP_STREAM).
template <bool RELU, bool RES, int ILP>
__global__ void bn_fwd_kernel(const unsigned short *__restrict__ x,
                              const unsigned short *__restrict__ res,
                              unsigned short *__restrict__ y,
                              const float *__restrict__ a,
                              const float *__restrict__ b, long long M,
                              int C, unsigned char *__restrict__ mask)
{
    const int gpr = C / 8;
    const int rows_per_blk = BLOCK / gpr;
    const int g = threadIdx.x % gpr;
    const int row_off = threadIdx.x / gpr;
    if (row_off >= rows_per_blk) return;
    float ar[8], br[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
        ar[k] = a[g * 8 + k];
        br[k] = b[g * 8 + k];
    }
    const long long row_step = (long long)gridDim.x * rows_per_blk;
    long long r = (long long)blockIdx.x * rows_per_blk + row_off;
    for (; r + (ILP - 1) * row_step < M; r += ILP * row_step) {
        ushort8 v[ILP], rv[ILP];
        long long base[ILP];
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
            base[j] = (r + j * row_step) * C + (long long)g * 8;
            v[j] = *(const ushort8 *)(x + base[j]);
            if (RES) rv[j] = *(const ushort8 *)(res + base[j]);
        }
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
            ushort8 out;
            unsigned char mb = 0;
#pragma unroll
            for (int k = 0; k < 8; ++k) {
                float u = fmaf(ar[k], b2f(v[j][k]), br[k]);
                if (RES) u += b2f(rv[j][k]);
                if (RELU) {
                    if (u > 0.f) mb |= (unsigned char)(1u << k);
                    u = fmaxf(u, 0.f);
                }
                out[k] = f2b(u);
            }
            *(ushort8 *)(y + base[j]) = out;
            if (RELU && mask) mask[(r + j * row_step) * gpr + g] = mb;
        }
    }
    for (; r < M; r += row_step) {
        const long long base = r * C + (long long)g * 8;
        const ushort8 v = *(const ushort8 *)(x + base);
        ushort8 rv;
        if (RES) rv = *(const ushort8 *)(res + base);
        ushort8 out;
        unsigned char mb = 0;
#pragma unroll
        for (int k = 0; k < 8; ++k) {
            float u = fmaf(ar[k], b2f(v[k]), br[k]);
            if (RES) u += b2f(rv[k]);
            if (RELU) {
                if (u > 0.f) mb |= (unsigned char)(1u << k);
                u = fmaxf(u, 0.f);
            }
            out[k] = f2b(u);
        }
        *(ushort8 *)(y + base) = out;
        // ReLU sign bits: 1 bit per element, so backward never re-reads
        // the residual or recomputes the pre-activation
        if (RELU && mask) mask[r * gpr + g] = mb;
    }
}

// ---- backward pass 1: per-channel sums of dy_m and dy_m * xhat ----
// dy_m = dy * relu_mask; xhat from save_mean/save_rstd.
// out: f32[NSHADOW][2*C] {sum_dy, sum_dyxhat} (pre-zeroed; folded copies
// ARE db and dw).
template <int ILP, int OCT, bool MASKED, bool NT>
__global__ void bn_bwd_reduce_kernel(
    const unsigned short *__restrict__ dy,
    const unsigned short *__restrict__ x,
    const unsigned char *__restrict__ mask,
    const float *__restrict__ mean, const float *__restrict__ rstd,
    long long M, int C, float *__restrict__ sums)
{
    extern __shared__ float lds[];  // 2*C floats
    const int gpr = C / (8 * OCT);
    const int rows_per_blk = BLOCK / gpr;
    const int g = threadIdx.x % gpr;
    const int row_off = threadIdx.x / gpr;
    for (int i = threadIdx.x; i < 2 * C; i += BLOCK) lds[i] = 0.f;
    __syncthreads();

    float s1[8 * OCT], s2[8 * OCT];
#pragma unroll
    for (int k = 0; k < 8 * OCT; ++k) s1[k] = s2[k] = 0.f;
    if (row_off < rows_per_blk) {
        float mr[8 * OCT], rr[8 * OCT];
#pragma unroll
        for (int k = 0; k < 8 * OCT; ++k) {
            const int c = g * 8 * OCT + k;
            mr[k] = mean[c];
            rr[k] = rstd[c];
        }
        const long long row_step = (long long)gridDim.x * rows_per_blk;
        long long r = (long long)blockIdx.x * rows_per_blk + row_off;
        for (; r + (ILP - 1) * row_step < M; r += ILP * row_step) {
            ushort8 dv[ILP][OCT], xv[ILP][OCT];
            unsigned short mk[ILP];
#pragma unroll
            for (int j = 0; j < ILP; ++j) {
                const long long base =
                    (r + j * row_step) * C + (long long)g * 8 * OCT;
#pragma unroll
                for (int o = 0; o < OCT; ++o) {
                    dv[j][o] = load8<NT>(dy + base + o * 8);
                    xv[j][o] = load8<NT>(x + base + o * 8);
                }
                if (MASKED) {
                    const unsigned char *mrow =
                        mask + (r + j * row_step) * (gpr * OCT) + g * OCT;
                    mk[j] = mrow[0];
                    if (OCT == 2) mk[j] |= (unsigned short)mrow[1] << 8;
                } else {
                    mk[j] = 0xffff;
                }
            }
#pragma unroll
            for (int j = 0; j < ILP; ++j) {
#pragma unroll
                for (int o = 0; o < OCT; ++o) {
#pragma unroll
                    for (int k = 0; k < 8; ++k) {
                        float d = b2f(dv[j][o][k]);
                        if (MASKED && !((mk[j] >> (o * 8 + k)) & 1))
                            d = 0.f;
                        const float xh =
                            (b2f(xv[j][o][k]) - mr[o * 8 + k]) *
                            rr[o * 8 + k];
                        s1[o * 8 + k] += d;
                        s2[o * 8 + k] += d * xh;
                    }
                }
            }
        }
        for (; r < M; r += row_step) {
            const long long base = r * C + (long long)g * 8 * OCT;
#pragma unroll
            for (int o = 0; o < OCT; ++o) {
                const ushort8 dv = load8<NT>(dy + base + o * 8);
                const ushort8 xv = load8<NT>(x + base + o * 8);
                const unsigned char mb =
                    MASKED ? mask[r * (gpr * OCT) + g * OCT + o]
                           : (unsigned char)0xff;
#pragma unroll
                for (int k = 0; k < 8; ++k) {
                    const float xf = b2f(xv[k]);
                    float d = b2f(dv[k]);
                    if (MASKED && !((mb >> k) & 1)) d = 0.f;
                    const float xh = (xf - mr[o * 8 + k]) * rr[o * 8 + k];
                    s1[o * 8 + k] += d;
                    s2[o * 8 + k] += d * xh;
                }
            }
        }
    }
#pragma unroll
    for (int k = 0; k < 8 * OCT; ++k) {
        atomicAdd(&lds[g * 8 * OCT + k], s1[k]);
        atomicAdd(&lds[C + g * 8 * OCT + k], s2[k]);
    }
    __syncthreads();
    float *shadow = sums + (size_t)(blockIdx.x % NSHADOW) * 2 * C;
    for (int i = threadIdx.x; i < 2 * C; i += BLOCK) {
        atomicAdd(&shadow[i], lds[i]);
    }
}

// Fold the NSHADOW accumulator copies into dbdw[2*C] and re-zero the
// shadows (no separate fill launch next step).
__global__ void bn_fold_kernel(float *__restrict__ sums, int C,
                               float *__restrict__ dbdw)
{
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= 2 * C) return;
    float acc = 0.f;
    for (int k = 0; k < NSHADOW; ++k) {
        acc += sums[(size_t)k * 2 * C + i];
        sums[(size_t)k * 2 * C + i] = 0.f;
    }
    dbdw[i] = acc;
}

// ---- backward pass 2: dx (and d_res when fused residual) ----
// dx = w*rstd * (dy_m - sum_dy/M - xhat * sum_dyxhat/M); d_res = dy_m.
// ILP rows in flight per lane (env KF_BN_ILP_STREAM).
template <bool MASKED, bool RES, int ILP>
__global__ void bn_bwd_dx_kernel(
    const unsigned short *__restrict__ dy,
    const unsigned short *__restrict__ x,
    const unsigned char *__restrict__ mask, const float *__restrict__ a,
    const float *__restrict__ mean, const float *__restrict__ rstd,
    const float *__restrict__ sums, long long M, int C,
    unsigned short *__restrict__ dx, unsigned short *__restrict__ dres)
{
    const int gpr = C / 8;
    const int rows_per_blk = BLOCK / gpr;
    const int g = threadIdx.x % gpr;
    const int row_off = threadIdx.x / gpr;
    if (row_off >= rows_per_blk) return;
    const float invM = 1.f / (float)M;
    float ar[8], mr[8], rr[8], t1[8], t2[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
        const int c = g * 8 + k;
        ar[k] = a[c];
        mr[k] = mean[c];
        rr[k] = rstd[c];
        t1[k] = sums[c] * invM;      // mean of dy_m
        t2[k] = sums[C + c] * invM;  // mean of dy_m * xhat
    }
    const long long row_step = (long long)gridDim.x * rows_per_blk;
    long long r = (long long)blockIdx.x * rows_per_blk + row_off;
    for (; r + (ILP - 1) * row_step < M; r += ILP * row_step) {
        ushort8 dv[ILP], xv[ILP];
        unsigned char mk[ILP];
        long long base[ILP];
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
            base[j] = (r + j * row_step) * C + (long long)g * 8;
            dv[j] = *(const ushort8 *)(dy + base[j]);
            xv[j] = *(const ushort8 *)(x + base[j]);
            mk[j] = MASKED ? mask[(r + j * row_step) * gpr + g]
                           : (unsigned char)0xff;
        }
#pragma unroll
        for (int j = 0; j < ILP; ++j) {
            ushort8 dxo, dro;
#pragma unroll
            for (int k = 0; k < 8; ++k) {
                float d = b2f(dv[j][k]);
                if (MASKED && !((mk[j] >> k) & 1)) d = 0.f;
                if (RES) dro[k] = f2b(d);
                const float xh = (b2f(xv[j][k]) - mr[k]) * rr[k];
                dxo[k] = f2b(ar[k] * (d - t1[k] - xh * t2[k]));
            }
            *(ushort8 *)(dx + base[j]) = dxo;
            if (RES) *(ushort8 *)(dres + base[j]) = dro;
        }
    }
    for (; r < M; r += row_step) {
        const long long base = r * C + (long long)g * 8;
        const ushort8 dv = *(const ushort8 *)(dy + base);
        const ushort8 xv = *(const ushort8 *)(x + base);
        const unsigned char mb =
            MASKED ? mask[r * gpr + g] : (unsigned char)0xff;
        ushort8 dxo, dro;
#pragma unroll
        for (int k = 0; k < 8; ++k) {
            const float xf = b2f(xv[k]);
            float d = b2f(dv[k]);
            if (MASKED && !((mb >> k) & 1)) d = 0.f;
            if (RES) dro[k] = f2b(d);
            const float xh = (xf - mr[k]) * rr[k];
            const float t = d - t1[k] - xh * t2[k];
            dxo[k] = f2b(ar[k] * t);
        }
        *(ushort8 *)(dx + base) = dxo;
        if (RES) *(ushort8 *)(dres + base) = dro;
    }
}

// ---- column sum (Linear bias gradients): dy [M,C] bf16 -> db bf16[C] ----
// Same shadow-atomic reduction pattern as bn_stats (sum only); grid.y
// stripes channels in BLOCK-octet (2048-channel) tiles so C can exceed
// the per-block limit (BERT ffn C=3072). torch's generic column-reduce
// runs this shape at ~0.34 TB/s; this kernel matches bn_stats (~2 TB/s).
template <int ILP>
__global__ void col_sum_kernel(const unsigned short *__restrict__ dy,
                               long long M, int C, int ld,
                               float *__restrict__ shadows)
{
    // C = aligned column count (multiple of 8); ld = row stride in
    // elements (>= C; callers with C % 8 != 0 pass the aligned prefix
    // here and reduce the tail columns separately)
    const int oct_total = C / 8;
    const int oct_base = blockIdx.y * BLOCK;
    const int gpr = min(oct_total - oct_base, BLOCK);
    if (gpr <= 0) return;
    const int rows_per_blk = BLOCK / gpr;
    const int g = (int)threadIdx.x % gpr;
    const int row_off = (int)threadIdx.x / gpr;
    extern __shared__ float lds[];  // gpr*8 <= 2048 floats
    for (int i = threadIdx.x; i < gpr * 8; i += BLOCK) lds[i] = 0.f;
    __syncthreads();
    float s[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    if (row_off < rows_per_blk) {
        const long long col = (long long)(oct_base + g) * 8;
        const long long row_step = (long long)gridDim.x * rows_per_blk;
        long long r = (long long)blockIdx.x * rows_per_blk + row_off;
        for (; r + (ILP - 1) * row_step < M; r += ILP * row_step) {
            ushort8 v[ILP];
#pragma unroll
            for (int j = 0; j < ILP; ++j)
                v[j] = *(const ushort8 *)(dy + (r + j * row_step) * ld +
                                          col);
#pragma unroll
            for (int j = 0; j < ILP; ++j) {
#pragma unroll
                for (int k = 0; k < 8; ++k) s[k] += b2f(v[j][k]);
            }
        }
        for (; r < M; r += row_step) {
            const ushort8 v = *(const ushort8 *)(dy + r * ld + col);
#pragma unroll
            for (int k = 0; k < 8; ++k) s[k] += b2f(v[k]);
        }
    }
#pragma unroll
    for (int k = 0; k < 8; ++k) atomicAdd(&lds[g * 8 + k], s[k]);
    __syncthreads();
    float *shadow = shadows + (size_t)(blockIdx.x % NSHADOW) * C +
                    (size_t)oct_base * 8;
    for (int i = threadIdx.x; i < gpr * 8; i += BLOCK) {
        atomicAdd(&shadow[i], lds[i]);
    }
}

// fold shadows -> bf16 db, re-zero shadows for the next call
__global__ void col_fold_kernel(float *__restrict__ shadows, int C,
                                unsigned short *__restrict__ db)
{
    const int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= C) return;
    float acc = 0.f;
    for (int k = 0; k < NSHADOW; ++k) {
        acc += shadows[(size_t)k * C + i];
        shadows[(size_t)k * C + i] = 0.f;
    }
    db[i] = f2b(acc);
}

}  // namespace

// ---------------------------------------------------------------------------
// extern "C" launchers
// ---------------------------------------------------------------------------

extern "C" {

hipError_t kf_bn_stats(const void *x, long long M, int C, void *sums,
                       void *stream)
{
    if (C % 8 != 0 || C / 8 > BLOCK) return hipErrorInvalidValue;
    const int oct = (C % 16 == 0) ? bn_oct() : 1;
    const int gpr = C / (8 * oct);
    const int rows_per_blk = BLOCK / gpr;
    const long long blocks = bn_reduce_blocks(M, C, rows_per_blk);
    const dim3 grid((uint32_t)blocks), block(BLOCK);
    const size_t lds = 2 * C * sizeof(float);
    const auto s = (hipStream_t)stream;
#define CASE(I, O, NT)                                                      \
    hipLaunchKernelGGL((bn_stats_kernel<I, O, NT>), grid, block, lds, s,    \
                       (const unsigned short *)x, M, C, (float *)sums)
    const bool nt = bn_nt();
    if (bn_ilp_stats() == 4) {
        if (oct == 2) { if (nt) CASE(4, 2, true); else CASE(4, 2, false); }
        else { if (nt) CASE(4, 1, true); else CASE(4, 1, false); }
    } else {
        if (oct == 2) { if (nt) CASE(2, 2, true); else CASE(2, 2, false); }
        else { if (nt) CASE(2, 1, true); else CASE(2, 1, false); }
    }
#undef CASE
    return hipGetLastError();
}

hipError_t kf_bn_finalize(void *sums, const void *weight,
                          const void *bias, void *running_mean,
                          void *running_var, void *save_mean,
                          void *save_rstd, void *a, void *b, long long M,
                          int C, float eps, float momentum, void *stream)
{
    hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 255) / 256), dim3(256),
                       0, (hipStream_t)stream, (float *)sums,
                       (const float *)weight, (const float *)bias,
                       (float *)running_mean, (float *)running_var,
                       (float *)save_mean, (float *)save_rstd, (float *)a,
                       (float *)b, M, C, eps, momentum);
    return hipGetLastError();
}

hipError_t kf_bn_fwd(const void *x, const void *res, void *y, const void *a,
                     const void *b, long long M, int C, int relu,
                     void *mask, void *stream)
{
    if (C % 8 != 0 || C / 8 > BLOCK) return hipErrorInvalidValue;
    const int rows_per_blk = BLOCK / (C / 8);
    long long blocks = (M + rows_per_blk - 1) / rows_per_blk;
    if (blocks > 4096) blocks = 4096;
    if (blocks < 1) blocks = 1;
    const dim3 grid((uint32_t)blocks), block(BLOCK);
    const auto s = (hipStream_t)stream;
#define CASE(R, E, I)                                                       \
    hipLaunchKernelGGL((bn_fwd_kernel<R, E, I>), grid, block, 0, s,         \
                       (const unsigned short *)x,                           \
                       (const unsigned short *)res, (unsigned short *)y,    \
                       (const float *)a, (const float *)b, M, C,            \
                       (unsigned char *)mask)
    if (bn_ilp_stream() == 4) {
        if (relu && res) CASE(true, true, 4);
        else if (relu) CASE(true, false, 4);
        else if (res) CASE(false, true, 4);
        else CASE(false, false, 4);
    } else {
        if (relu && res) CASE(true, true, 2);
        else if (relu) CASE(true, false, 2);
        else if (res) CASE(false, true, 2);
        else CASE(false, false, 2);
    }
#undef CASE
    return hipGetLastError();
}

hipError_t kf_bn_bwd_reduce(const void *dy, const void *x,
                            const void *mask, const void *mean,
                            const void *rstd, long long M, int C,
                            void *sums, void *stream)
{
    if (C % 8 != 0 || C / 8 > BLOCK) return hipErrorInvalidValue;
    const int oct = 1;  // OCT=2 measured 2x slower on every shape
    const int gpr = C / (8 * oct);
    const int rows_per_blk = BLOCK / gpr;
    const long long blocks = bn_reduce_blocks(M, C, rows_per_blk);
    const dim3 grid((uint32_t)blocks), block(BLOCK);
    const auto s = (hipStream_t)stream;
    const size_t lds = 2 * C * sizeof(float);
#define CASE(I, O, MK, NT)                                                  \
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<I, O, MK, NT>), grid, block,   \
                       lds, s, (const unsigned short *)dy,                  \
                       (const unsigned short *)x,                           \
                       (const unsigned char *)mask, (const float *)mean,    \
                       (const float *)rstd, M, C, (float *)sums)
    const bool nt = bn_nt();
    // OCT=2 measured 2x slower; only instantiate OCT=1 with the NT axis
    if (bn_ilp_bwd() == 4) {
        if (mask) { if (nt) CASE(4, 1, true, true); else CASE(4, 1, true, false); }
        else { if (nt) CASE(4, 1, false, true); else CASE(4, 1, false, false); }
    } else {
        if (mask) { if (nt) CASE(2, 1, true, true); else CASE(2, 1, true, false); }
        else { if (nt) CASE(2, 1, false, true); else CASE(2, 1, false, false); }
    }
#undef CASE
    return hipGetLastError();
}

hipError_t kf_bn_fold(void *sums, int C, void *dbdw, void *stream)
{
    hipLaunchKernelGGL(bn_fold_kernel, dim3((2 * C + 255) / 256), dim3(256),
                       0, (hipStream_t)stream, (float *)sums, C,
                       (float *)dbdw);
    return hipGetLastError();
}

hipError_t kf_bn_bwd_dx(const void *dy, const void *x, const void *mask,
                        const void *a, const void *mean, const void *rstd,
                        const void *sums, long long M, int C, void *dx,
                        void *dres, void *stream)
{
    if (C % 8 != 0 || C / 8 > BLOCK) return hipErrorInvalidValue;
    const int rows_per_blk = BLOCK / (C / 8);
    long long blocks = (M + rows_per_blk - 1) / rows_per_blk;
    if (blocks > 4096) blocks = 4096;
    if (blocks < 1) blocks = 1;
    const dim3 grid((uint32_t)blocks), block(BLOCK);
    const auto s = (hipStream_t)stream;
#define CASE(MK, E, I)                                                      \
    hipLaunchKernelGGL((bn_bwd_dx_kernel<MK, E, I>), grid, block, 0, s,     \
                       (const unsigned short *)dy,                          \
                       (const unsigned short *)x,                           \
                       (const unsigned char *)mask, (const float *)a,       \
                       (const float *)mean, (const float *)rstd,            \
                       (const float *)sums, M, C, (unsigned short *)dx,     \
                       (unsigned short *)dres)
    if (bn_ilp_stream() == 4) {
        if (mask && dres) CASE(true, true, 4);
        else if (mask) CASE(true, false, 4);
        else if (dres) CASE(false, true, 4);
        else CASE(false, false, 4);
    } else {
        if (mask && dres) CASE(true, true, 2);
        else if (mask) CASE(true, false, 2);
        else if (dres) CASE(false, true, 2);
        else CASE(false, false, 2);
    }
#undef CASE
    return hipGetLastError();
}

hipError_t kf_col_sum(const void *dy, long long M, int C, int ld,
                      void *shadows, void *db_bf16, void *stream)
{
    if (C % 8 != 0 || ld < C) return hipErrorInvalidValue;
    const int oct_total = C / 8;
    const int gpr = oct_total < BLOCK ? oct_total : BLOCK;
    const int rows_per_blk = BLOCK / gpr;
    const long long bx = bn_reduce_blocks(M, C < 2048 ? C : 2048,
                                          rows_per_blk);
    const uint32_t by = (uint32_t)((oct_total + BLOCK - 1) / BLOCK);
    const dim3 grid((uint32_t)bx, by), block(BLOCK);
    const size_t lds = (size_t)gpr * 8 * sizeof(float);
    const auto s = (hipStream_t)stream;
    hipLaunchKernelGGL((col_sum_kernel<4>), grid, block, lds, s,
                       (const unsigned short *)dy, M, C, ld,
                       (float *)shadows);
    hipLaunchKernelGGL(col_fold_kernel, dim3((C + 255) / 256), dim3(256),
                       0, s, (float *)shadows, C,
                       (unsigned short *)db_bf16);
    return hipGetLastError();
}

}  // extern "C"
