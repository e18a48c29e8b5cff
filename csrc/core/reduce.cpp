// Elementwise reduction for the CPU collective engine.
// Reference parity: srcs/go/kungfu/base/op.cpp (std_transform_2) + f16.c.
// Plain templated loops; the compiler auto-vectorizes the f32/i32 cases.
#include "common.hpp"

namespace kf {

namespace {

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
