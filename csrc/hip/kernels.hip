// Hand-written CDNA4 (gfx950 / MI355X) kernels for the KungFu-AMD hot ops.
//
// These replace the reference's TF-graph / NCCL-internal GPU work
// (SURVEY.md §2.6): gradient-fusion pack/unpack (ref item 6: TF fuse/defuse
// concat ops), fused model averaging (item 8: 0.5*(v+other) and
// (1-alpha)*v+alpha*avg), gradient-noise-scale norm² reduction (item 9),
// elementwise aggregation (item 7), plus an MI355X-native fused SGD-momentum
// step over flat parameter buffers (per-param torch ops would launch ~800
// kernels/step on ResNet-50).
//
// Design per /opt/skills/guides/cdna_hip_programming.md:
//  - all ops are memory-bound: bf16/f16 loads vectorized 16 B/lane
//    (Guideline 13), grid-stride with grid capped (~2048 blocks, G11);
//  - reductions: wave-64 __shfl_down tree -> LDS -> one atomic per block
//    (Appendix B "Reduction"); no 32-wide warp idioms;
//  - no CUDA compatibility layer: this file is HIP-only, gfx950-only.
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_runtime.h>

#include <cstdint>

#define WAVE 64

namespace {

typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;  // 16 B

// ---- dtype traits: convert through f32 ----
template <typename T>
struct Elem;

template <>
struct Elem<float> {
    __device__ static float to_f(float x) { return x; }
    __device__ static float from_f(float x) { return x; }
};
template <>
struct Elem<__hip_bfloat16> {
    __device__ static float to_f(__hip_bfloat16 x)
    {
        return __bfloat162float(x);
    }
    __device__ static __hip_bfloat16 from_f(float x)
    {
        return __float2bfloat16(x);
    }
};
template <>
struct Elem<__half> {
    __device__ static float to_f(__half x) { return __half2float(x); }
    __device__ static __half from_f(float x) { return __float2half(x); }
};

constexpr int BLOCK = 256;

inline int elementwise_grid(long long n, int per_thread)
{
    long long blocks = (n + (long long)BLOCK * per_thread - 1) /
                       ((long long)BLOCK * per_thread);
    if (blocks < 1) blocks = 1;
    if (blocks > 2048) blocks = 2048;  // grid-stride the rest (G11)
    return (int)blocks;
}

}  // namespace

// ---------------------------------------------------------------------------
// Fusion pack/unpack (multi-tensor copy through a chunk table)
// ---------------------------------------------------------------------------

struct Chunk {
    const void *src;          // segment base (device)
    unsigned long long dst;   // element offset into the fused buffer
    unsigned int n;           // elements in this chunk
};

namespace {

// One block per chunk entry (grid-stride over entries). 16-B vector copies
// when both sides are aligned; scalar tail.
template <typename T>
__global__ void pack_kernel(const Chunk *__restrict__ cs, int nc,
                            T *__restrict__ out)
{
    for (int b = blockIdx.x; b < nc; b += gridDim.x) {
        const Chunk c = cs[b];
        const T *src = (const T *)c.src;
        T *dst = out + c.dst;
        const int n = (int)c.n;
        constexpr int V = 16 / sizeof(T);
        const bool aligned = ((((uintptr_t)src) | ((uintptr_t)dst)) & 15) == 0;
        if (aligned) {
            const int nv = n / V;
            const uint4v *vs = (const uint4v *)src;
            uint4v *vd = (uint4v *)dst;
            for (int i = threadIdx.x; i < nv; i += blockDim.x) vd[i] = vs[i];
            for (int i = nv * V + threadIdx.x; i < n; i += blockDim.x)
                dst[i] = src[i];
        } else {
            for (int i = threadIdx.x; i < n; i += blockDim.x)
                dst[i] = src[i];
        }
    }
}

// Unpack fused -> segments, with a scale folded in (grad /= np after
// all-reduce; reference does a separate division pass).
template <typename T>
__global__ void unpack_kernel(const Chunk *__restrict__ cs, int nc,
                              const T *__restrict__ fused, float scale)
{
    for (int b = blockIdx.x; b < nc; b += gridDim.x) {
        const Chunk c = cs[b];
        T *dst = (T *)c.src;
        const T *src = fused + c.dst;
        const int n = (int)c.n;
        if (scale == 1.0f) {
            constexpr int V = 16 / sizeof(T);
            const bool aligned =
                ((((uintptr_t)src) | ((uintptr_t)dst)) & 15) == 0;
            if (aligned) {
                const int nv = n / V;
                const uint4v *vs = (const uint4v *)src;
                uint4v *vd = (uint4v *)dst;
                for (int i = threadIdx.x; i < nv; i += blockDim.x)
                    vd[i] = vs[i];
                for (int i = nv * V + threadIdx.x; i < n; i += blockDim.x)
                    dst[i] = src[i];
            } else {
                for (int i = threadIdx.x; i < n; i += blockDim.x)
                    dst[i] = src[i];
            }
        } else {
            for (int i = threadIdx.x; i < n; i += blockDim.x)
                dst[i] = Elem<T>::from_f(Elem<T>::to_f(src[i]) * scale);
        }
    }
}

}  // namespace

// ---------------------------------------------------------------------------
// Elementwise: average / axpby / scale / reduce-aggregate
// ---------------------------------------------------------------------------

namespace {

// y = (1-alpha)*y + alpha*x   (SMA model averaging; alpha=0.5 = pair avg)
template <typename T>
__global__ void avg_inplace_kernel(T *__restrict__ y,
                                   const T *__restrict__ x, float alpha,
                                   long long n)
{
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const float yi = Elem<T>::to_f(y[i]);
        const float xi = Elem<T>::to_f(x[i]);
        y[i] = Elem<T>::from_f(yi + alpha * (xi - yi));
    }
}

template <typename T>
__global__ void scale_kernel(T *__restrict__ y, float s, long long n)
{
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        y[i] = Elem<T>::from_f(Elem<T>::to_f(y[i]) * s);
    }
}

// z = x op y elementwise (GPU-side aggregation)
template <typename T, int OP>  // 0 sum, 1 min, 2 max, 3 prod
__global__ void transform2_kernel(T *__restrict__ z,
                                  const T *__restrict__ x, long long n)
{
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const float a = Elem<T>::to_f(z[i]);
        const float b = Elem<T>::to_f(x[i]);
        float r;
        if (OP == 0) r = a + b;
        else if (OP == 1) r = fminf(a, b);
        else if (OP == 2) r = fmaxf(a, b);
        else r = a * b;
        z[i] = Elem<T>::from_f(r);
    }
}

}  // namespace

// ---------------------------------------------------------------------------
// norm² reduction (gradient noise scale / gradient variance monitors)
// ---------------------------------------------------------------------------

namespace {

__device__ inline float wave_reduce_sum(float v)
{
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        v += __shfl_down(v, off, WAVE);
    }
    return v;
}

// out[0] += sum(x[i]^2); caller zeroes out[0] first.
template <typename T>
__global__ void norm2_kernel(const T *__restrict__ x, long long n,
                             float *__restrict__ out)
{
    __shared__ float warp_sums[BLOCK / WAVE];
    float acc = 0.f;
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const float v = Elem<T>::to_f(x[i]);
        acc += v * v;
    }
    acc = wave_reduce_sum(acc);
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    if (lane == 0) warp_sums[wid] = acc;
    __syncthreads();
    if (wid == 0) {
        float v = lane < BLOCK / WAVE ? warp_sums[lane] : 0.f;
        v = wave_reduce_sum(v);
        if (lane == 0) atomicAdd(out, v);
    }
}

// out[0] += dot(x, y)
template <typename T>
__global__ void dot_kernel(const T *__restrict__ x, const T *__restrict__ y,
                           long long n, float *__restrict__ out)
{
    __shared__ float warp_sums[BLOCK / WAVE];
    float acc = 0.f;
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        acc += Elem<T>::to_f(x[i]) * Elem<T>::to_f(y[i]);
    }
    acc = wave_reduce_sum(acc);
    const int lane = threadIdx.x & (WAVE - 1);
    const int wid = threadIdx.x / WAVE;
    if (lane == 0) warp_sums[wid] = acc;
    __syncthreads();
    if (wid == 0) {
        float v = lane < BLOCK / WAVE ? warp_sums[lane] : 0.f;
        v = wave_reduce_sum(v);
        if (lane == 0) atomicAdd(out, v);
    }
}

}  // namespace

// ---------------------------------------------------------------------------
// Fused SGD with momentum over flat buffers
// ---------------------------------------------------------------------------

namespace {

// p/g dtype T (bf16 or f32), momentum state in f32.
// grad_scale folds the 1/np division (and loss scaling) into the step.
template <typename T>
__global__ void sgd_momentum_kernel(T *__restrict__ p,
                                    const T *__restrict__ g,
                                    float *__restrict__ m, long long n,
                                    float lr, float momentum,
                                    float weight_decay, float grad_scale,
                                    int nesterov)
{
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        float pi = Elem<T>::to_f(p[i]);
        float gi = Elem<T>::to_f(g[i]) * grad_scale + weight_decay * pi;
        float mi = momentum * m[i] + gi;
        m[i] = mi;
        const float upd = nesterov ? gi + momentum * mi : mi;
        p[i] = Elem<T>::from_f(pi - lr * upd);
    }
}

// Mixed-precision variant: model weights (and grads) in T (bf16), the
// authoritative copy + momentum in f32. One kernel replaces the autocast
// master-weight pattern's separate cast passes (bf16<->f32 copies every
// step); traffic is 20 B/elem either way but the per-step cast kernels
// (~0.5 ms on ResNet-50 b64) disappear and the all-reduce payload halves.
template <typename T>
__global__ void sgd_momentum_master_kernel(
    T *__restrict__ p, const T *__restrict__ g,
    float *__restrict__ master, float *__restrict__ m, long long n,
    float lr, float momentum, float weight_decay, float grad_scale,
    int nesterov)
{
    const long long stride = (long long)gridDim.x * blockDim.x;
    for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         i < n; i += stride) {
        const float pi = master[i];
        const float gi = Elem<T>::to_f(g[i]) * grad_scale +
                         weight_decay * pi;
        const float mi = momentum * m[i] + gi;
        m[i] = mi;
        const float upd = nesterov ? gi + momentum * mi : mi;
        const float np = pi - lr * upd;
        master[i] = np;
        p[i] = Elem<T>::from_f(np);
    }
}

}  // namespace

// ---------------------------------------------------------------------------
// extern "C" launchers (called from the pybind host module)
// dtype codes: 0 = f32, 1 = bf16, 2 = f16
// ---------------------------------------------------------------------------

#define DISPATCH(dtype, fn, ...)                                            \
    switch (dtype) {                                                        \
    case 0: fn<float>(__VA_ARGS__); break;                                  \
    case 1: fn<__hip_bfloat16>(__VA_ARGS__); break;                         \
    case 2: fn<__half>(__VA_ARGS__); break;                                 \
    default: return hipErrorInvalidValue;                                   \
    }

namespace {
template <typename T>
void launch_pack(const Chunk *cs, int nc, void *out, hipStream_t s)
{
    int grid = nc < 2048 ? (nc > 0 ? nc : 1) : 2048;
    hipLaunchKernelGGL(pack_kernel<T>, dim3(grid), dim3(BLOCK), 0, s, cs, nc,
                       (T *)out);
}
template <typename T>
void launch_unpack(const Chunk *cs, int nc, const void *fused, float scale,
                   hipStream_t s)
{
    int grid = nc < 2048 ? (nc > 0 ? nc : 1) : 2048;
    hipLaunchKernelGGL(unpack_kernel<T>, dim3(grid), dim3(BLOCK), 0, s, cs,
                       nc, (const T *)fused, scale);
}
template <typename T>
void launch_avg(void *y, const void *x, float alpha, long long n,
                hipStream_t s)
{
    hipLaunchKernelGGL(avg_inplace_kernel<T>, dim3(elementwise_grid(n, 8)),
                       dim3(BLOCK), 0, s, (T *)y, (const T *)x, alpha, n);
}
template <typename T>
void launch_scale(void *y, float f, long long n, hipStream_t s)
{
    hipLaunchKernelGGL(scale_kernel<T>, dim3(elementwise_grid(n, 8)),
                       dim3(BLOCK), 0, s, (T *)y, f, n);
}
template <typename T>
void launch_norm2(const void *x, long long n, float *out, hipStream_t s)
{
    hipLaunchKernelGGL(norm2_kernel<T>, dim3(elementwise_grid(n, 16)),
                       dim3(BLOCK), 0, s, (const T *)x, n, out);
}
template <typename T>
void launch_dot(const void *x, const void *y, long long n, float *out,
                hipStream_t s)
{
    hipLaunchKernelGGL(dot_kernel<T>, dim3(elementwise_grid(n, 16)),
                       dim3(BLOCK), 0, s, (const T *)x, (const T *)y, n,
                       out);
}
template <typename T>
void launch_sgd(void *p, const void *g, float *m, long long n, float lr,
                float mu, float wd, float gs, int nesterov, hipStream_t s)
{
    hipLaunchKernelGGL(sgd_momentum_kernel<T>, dim3(elementwise_grid(n, 8)),
                       dim3(BLOCK), 0, s, (T *)p, (const T *)g, m, n, lr, mu,
                       wd, gs, nesterov);
}
template <typename T>
void launch_sgd_master(void *p, const void *g, float *master, float *m,
                       long long n, float lr, float mu, float wd, float gs,
                       int nesterov, hipStream_t s)
{
    hipLaunchKernelGGL(sgd_momentum_master_kernel<T>,
                       dim3(elementwise_grid(n, 8)), dim3(BLOCK), 0, s,
                       (T *)p, (const T *)g, master, m, n, lr, mu, wd, gs,
                       nesterov);
}
template <typename T>
void launch_transform2(int op, void *z, const void *x, long long n,
                       hipStream_t s)
{
    const dim3 grid(elementwise_grid(n, 8)), block(BLOCK);
    switch (op) {
    case 0:
        hipLaunchKernelGGL((transform2_kernel<T, 0>), grid, block, 0, s,
                           (T *)z, (const T *)x, n);
        break;
    case 1:
        hipLaunchKernelGGL((transform2_kernel<T, 1>), grid, block, 0, s,
                           (T *)z, (const T *)x, n);
        break;
    case 2:
        hipLaunchKernelGGL((transform2_kernel<T, 2>), grid, block, 0, s,
                           (T *)z, (const T *)x, n);
        break;
    default:
        hipLaunchKernelGGL((transform2_kernel<T, 3>), grid, block, 0, s,
                           (T *)z, (const T *)x, n);
        break;
    }
}
}  // namespace

extern "C" {

hipError_t kf_pack(const void *chunks_dev, int nchunks, void *fused,
                   int dtype, void *stream)
{
    DISPATCH(dtype, launch_pack, (const Chunk *)chunks_dev, nchunks, fused,
             (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_unpack(const void *chunks_dev, int nchunks, const void *fused,
                     float scale, int dtype, void *stream)
{
    DISPATCH(dtype, launch_unpack, (const Chunk *)chunks_dev, nchunks, fused,
             scale, (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_avg_inplace(void *y, const void *x, float alpha, long long n,
                          int dtype, void *stream)
{
    DISPATCH(dtype, launch_avg, y, x, alpha, n, (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_scale(void *y, float s, long long n, int dtype, void *stream)
{
    DISPATCH(dtype, launch_scale, y, s, n, (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_norm2(const void *x, long long n, void *out_f32, int dtype,
                    void *stream)
{
    DISPATCH(dtype, launch_norm2, x, n, (float *)out_f32,
             (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_dot(const void *x, const void *y, long long n, void *out_f32,
                  int dtype, void *stream)
{
    DISPATCH(dtype, launch_dot, x, y, n, (float *)out_f32,
             (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_sgd_momentum(void *p, const void *g, void *m_f32, long long n,
                           float lr, float momentum, float weight_decay,
                           float grad_scale, int nesterov, int dtype,
                           void *stream)
{
    DISPATCH(dtype, launch_sgd, p, g, (float *)m_f32, n, lr, momentum,
             weight_decay, grad_scale, nesterov, (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_sgd_momentum_master(void *p, const void *g, void *master_f32,
                                  void *m_f32, long long n, float lr,
                                  float momentum, float weight_decay,
                                  float grad_scale, int nesterov, int dtype,
                                  void *stream)
{
    DISPATCH(dtype, launch_sgd_master, p, g, (float *)master_f32,
             (float *)m_f32, n, lr, momentum, weight_decay, grad_scale,
             nesterov, (hipStream_t)stream);
    return hipGetLastError();
}

hipError_t kf_transform2(void *z, const void *x, long long n, int op,
                         int dtype, void *stream)
{
    DISPATCH(dtype, launch_transform2, op, z, x, n, (hipStream_t)stream);
    return hipGetLastError();
}

}  // extern "C"
