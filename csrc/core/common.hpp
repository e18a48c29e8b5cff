// Core scalar types for the KungFu-AMD runtime.
//
// Replaces the reference's Go base layer (srcs/go/kungfu/base/{dtype,op}.go and
// the C++ reduction template srcs/go/kungfu/base/op.cpp) with a single C++17
// header: dtype registry + elementwise reduction used by the CPU collective
// engine's partial-aggregation receive path.
#pragma once

#include <cstddef>
#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>

namespace kf {

enum class DType : uint8_t {
    U8 = 0,
    I8,
    I16,
    I32,
    I64,
    U16,
    U32,
    U64,
    F16,
    BF16,
    F32,
    F64,
};

inline size_t dtype_size(DType d)
{
    switch (d) {
    case DType::U8:
    case DType::I8:
        return 1;
    case DType::I16:
    case DType::U16:
    case DType::F16:
    case DType::BF16:
        return 2;
    case DType::I32:
    case DType::U32:
    case DType::F32:
        return 4;
    case DType::I64:
    case DType::U64:
    case DType::F64:
        return 8;
    }
    throw std::runtime_error("bad dtype");
}

enum class ReduceOp : uint8_t {
    SUM = 0,
    MIN,
    MAX,
    PROD,
};

// f16/bf16 <-> f32 scalar conversion (CPU path only; the GPU path never
// touches these — RCCL reduces on-device).
inline float half_to_float(uint16_t h)
{
    uint32_t sign = (uint32_t)(h >> 15) << 31;
    uint32_t exp  = (h >> 10) & 0x1f;
    uint32_t man  = h & 0x3ff;
    uint32_t out;
    if (exp == 0) {
        if (man == 0) {
            out = sign;
        } else {  // subnormal
            int e = -1;
            do {
                ++e;
                man <<= 1;
            } while ((man & 0x400) == 0);
            out = sign | ((uint32_t)(127 - 15 - e) << 23) |
                  ((man & 0x3ff) << 13);
        }
    } else if (exp == 0x1f) {
        out = sign | 0x7f800000u | (man << 13);
    } else {
        out = sign | ((exp + 127 - 15) << 23) | (man << 13);
    }
    float f;
    std::memcpy(&f, &out, 4);
    return f;
}

inline uint16_t float_to_half(float f)
{
    uint32_t x;
    std::memcpy(&x, &f, 4);
    uint32_t sign = (x >> 16) & 0x8000;
    int32_t exp   = (int32_t)((x >> 23) & 0xff) - 127 + 15;
    uint32_t man  = x & 0x7fffff;
    if (((x >> 23) & 0xff) == 0xff) return (uint16_t)(sign | 0x7c00 | (man ? 0x200 : 0));
    if (exp >= 0x1f) return (uint16_t)(sign | 0x7c00);  // overflow -> inf
    if (exp <= 0) {
        if (exp < -10) return (uint16_t)sign;
        man |= 0x800000;
        uint32_t shift = (uint32_t)(14 - exp);
        uint32_t half  = (man >> shift) & 0x3ff;
        uint32_t rem   = man & ((1u << shift) - 1);
        if (rem > (1u << (shift - 1))) half++;  // round to nearest
        return (uint16_t)(sign | half);
    }
    // round to nearest even
    uint32_t half = man >> 13;
    uint32_t rem  = man & 0x1fff;
    if (rem > 0x1000 || (rem == 0x1000 && (half & 1))) half++;
    return (uint16_t)(sign | ((uint32_t)exp << 10) | half);
}

inline float bf16_to_float(uint16_t h)
{
    uint32_t out = (uint32_t)h << 16;
    float f;
    std::memcpy(&f, &out, 4);
    return f;
}

inline uint16_t float_to_bf16(float f)
{
    uint32_t x;
    std::memcpy(&x, &f, 4);
    // round to nearest even
    uint32_t r = x + 0x7fff + ((x >> 16) & 1);
    return (uint16_t)(r >> 16);
}

// acc[i] = acc[i] op in[i] — the hot loop of the CPU collective receive path
// (reference: std_transform_2, srcs/go/kungfu/base/op.cpp:60-100).
void reduce_inplace(void *acc, const void *in, size_t count, DType dt,
                    ReduceOp op);

}  // namespace kf
