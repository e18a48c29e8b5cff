// Elementwise reduction for the CPU collective engine.
// Reference parity: srcs/go/kungfu/base/op.cpp (std_transform_2) + f16.c
// (AVX _mm256_cvtph_ps f16 sum). Plain templated loops (auto-vectorized
// f32/i32) plus an F16C+AVX vector path for f16 sums when the host CPU
// supports it.
#include "common.hpp"

#if defined(__x86_64__)
#include <immintrin.h>
#endif

namespace kf {

namespace {

#if defined(__x86_64__)
// f16 sum via F16C 8-lane convert-add-convert (reference f16.c:17-24),
// gated on runtime CPU support so the binary stays portable.
bool cpu_has_f16c()
{
    static const bool ok = __builtin_cpu_supports("f16c") &&
                           __builtin_cpu_supports("avx");
    return ok;
}

__attribute__((target("avx,f16c"))) void f16_sum_avx(uint16_t *acc,
                                                     const uint16_t *in,
                                                     size_t n)
{
    size_t i = 0;
    for (; i + 8 <= n; i += 8) {
        __m256 a = _mm256_cvtph_ps(
            _mm_loadu_si128((const __m128i *)(acc + i)));
        __m256 b = _mm256_cvtph_ps(
            _mm_loadu_si128((const __m128i *)(in + i)));
        _mm_storeu_si128(
            (__m128i *)(acc + i),
            _mm256_cvtps_ph(_mm256_add_ps(a, b),
                            _MM_FROUND_TO_NEAREST_INT));
    }
    for (; i < n; ++i) {
        acc[i] = float_to_half(half_to_float(acc[i]) +
                               half_to_float(in[i]));
    }
}
#endif

template <typename T, typename F>
void loop(T *acc, const T *in, size_t n, F f)
{
    for (size_t i = 0; i < n; ++i) acc[i] = f(acc[i], in[i]);
}

template <typename T>
void dispatch_op(T *acc, const T *in, size_t n, ReduceOp op)
{
    switch (op) {
    case ReduceOp::SUM:
        loop(acc, in, n, [](T a, T b) { return (T)(a + b); });
        return;
    case ReduceOp::MIN:
        loop(acc, in, n, [](T a, T b) { return b < a ? b : a; });
        return;
    case ReduceOp::MAX:
        loop(acc, in, n, [](T a, T b) { return a < b ? b : a; });
        return;
    case ReduceOp::PROD:
        loop(acc, in, n, [](T a, T b) { return (T)(a * b); });
        return;
    }
    throw std::runtime_error("bad reduce op");
}

template <float (*To)(uint16_t), uint16_t (*From)(float)>
void dispatch_f16ish(uint16_t *acc, const uint16_t *in, size_t n, ReduceOp op)
{
    auto apply = [&](auto f) {
        for (size_t i = 0; i < n; ++i) acc[i] = From(f(To(acc[i]), To(in[i])));
    };
    switch (op) {
    case ReduceOp::SUM:
        apply([](float a, float b) { return a + b; });
        return;
    case ReduceOp::MIN:
        apply([](float a, float b) { return b < a ? b : a; });
        return;
    case ReduceOp::MAX:
        apply([](float a, float b) { return a < b ? b : a; });
        return;
    case ReduceOp::PROD:
        apply([](float a, float b) { return a * b; });
        return;
    }
    throw std::runtime_error("bad reduce op");
}

}  // namespace

void reduce_inplace(void *acc, const void *in, size_t count, DType dt,
                    ReduceOp op)
{
    switch (dt) {
    case DType::U8:
        return dispatch_op((uint8_t *)acc, (const uint8_t *)in, count, op);
    case DType::I8:
        return dispatch_op((int8_t *)acc, (const int8_t *)in, count, op);
    case DType::I16:
        return dispatch_op((int16_t *)acc, (const int16_t *)in, count, op);
    case DType::I32:
        return dispatch_op((int32_t *)acc, (const int32_t *)in, count, op);
    case DType::I64:
        return dispatch_op((int64_t *)acc, (const int64_t *)in, count, op);
    case DType::U16:
        return dispatch_op((uint16_t *)acc, (const uint16_t *)in, count, op);
    case DType::U32:
        return dispatch_op((uint32_t *)acc, (const uint32_t *)in, count, op);
    case DType::U64:
        return dispatch_op((uint64_t *)acc, (const uint64_t *)in, count, op);
    case DType::F16:
#if defined(__x86_64__)
        if (op == ReduceOp::SUM && cpu_has_f16c()) {
            return f16_sum_avx((uint16_t *)acc, (const uint16_t *)in,
                               count);
        }
#endif
        return dispatch_f16ish<half_to_float, float_to_half>(
            (uint16_t *)acc, (const uint16_t *)in, count, op);
    case DType::BF16:
        return dispatch_f16ish<bf16_to_float, float_to_bf16>(
            (uint16_t *)acc, (const uint16_t *)in, count, op);
    case DType::F32:
        return dispatch_op((float *)acc, (const float *)in, count, op);
    case DType::F64:
        return dispatch_op((double *)acc, (const double *)in, count, op);
    }
    throw std::runtime_error("bad dtype");
}

}  // namespace kf
