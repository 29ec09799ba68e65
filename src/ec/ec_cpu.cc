#include "ec_cpu.h"

#include <cmath>
#include <complex>
#include <cstring>
#include <cstdint>

namespace ucc {
namespace ec_cpu {

/* ---------------------------------------------------- float conversions */
float bf16_to_float(uint16_t v)
{
    uint32_t u = (uint32_t)v << 16;
    float    f;
    memcpy(&f, &u, 4);
    return f;
}

uint16_t float_to_bf16(float f)
{
    uint32_t u;
    memcpy(&u, &f, 4);
    /* round-to-nearest-even */
    uint32_t rounding = 0x7fff + ((u >> 16) & 1);
    return (uint16_t)((u + rounding) >> 16);
}

float fp16_to_float(uint16_t h)
{
    uint32_t sign = (h >> 15) & 1, exp = (h >> 10) & 0x1f, man = h & 0x3ff;
    uint32_t u;
    if (exp == 0) {
        if (man == 0) {
            u = sign << 31;
        } else { /* subnormal */
            exp = 127 - 15 + 1;
            while (!(man & 0x400)) {
                man <<= 1;
                exp--;
            }
            man &= 0x3ff;
            u = (sign << 31) | (exp << 23) | (man << 13);
        }
    } else if (exp == 0x1f) {
        u = (sign << 31) | 0x7f800000 | (man << 13);
    } else {
        u = (sign << 31) | ((exp - 15 + 127) << 23) | (man << 13);
    }
    float f;
    memcpy(&f, &u, 4);
    return f;
}

uint16_t float_to_fp16(float f)
{
    uint32_t u;
    memcpy(&u, &f, 4);
    uint32_t sign = (u >> 31) & 1;
    int32_t  exp  = (int32_t)((u >> 23) & 0xff) - 127 + 15;
    uint32_t man  = u & 0x7fffff;
    if (((u >> 23) & 0xff) == 0xff) { /* inf/nan */
        return (uint16_t)((sign << 15) | 0x7c00 | (man ? 0x200 : 0));
    }
    if (exp >= 0x1f) {
        return (uint16_t)((sign << 15) | 0x7c00); /* overflow -> inf */
    }
    if (exp <= 0) { /* subnormal or zero */
        if (exp < -10) {
            return (uint16_t)(sign << 15);
        }
        man |= 0x800000;
        uint32_t shift = 14 - exp;
        uint32_t half  = 1u << (shift - 1);
        uint32_t r     = (man + half - 1 + ((man >> shift) & 1)) >> shift;
        return (uint16_t)((sign << 15) | r);
    }
    uint32_t r = (man + 0xfff + ((man >> 13) & 1)) >> 13;
    if (r & 0x400) { /* mantissa overflow */
        r = 0;
        exp++;
        if (exp >= 0x1f) {
            return (uint16_t)((sign << 15) | 0x7c00);
        }
    }
    return (uint16_t)((sign << 15) | (exp << 10) | (r & 0x3ff));
}

/* OCP fp8: e4m3fn (no inf, single NaN 0x7f/0xff), e5m2 (IEEE-like). */
float fp8e4m3_to_float(uint8_t v)
{
    uint32_t sign = (v >> 7) & 1, exp = (v >> 3) & 0xf, man = v & 7;
    if (exp == 0xf && man == 7) {
        return sign ? -NAN : NAN;
    }
    float val;
    if (exp == 0) {
        val = (float)man / 8.0f / 64.0f; /* 2^-6 * man/8 */
    } else {
        val = (1.0f + (float)man / 8.0f) * std::ldexp(1.0f, (int)exp - 7);
    }
    return sign ? -val : val;
}

uint8_t float_to_fp8e4m3(float f)
{
    if (std::isnan(f)) {
        return 0x7f;
    }
    uint8_t sign = f < 0 ? 0x80 : 0;
    float   a    = std::fabs(f);
    if (a >= 448.0f) { /* saturate to max finite (OCP e4m3fn) */
        return sign | 0x7e;
    }
    if (a < std::ldexp(1.0f, -9)) { /* below half of min subnormal */
        return sign;
    }
    int   e;
    float m = std::frexp(a, &e); /* a = m * 2^e, m in [0.5,1) */
    e -= 1;                      /* a = (2m) * 2^(e), 2m in [1,2) */
    int   exp = e + 7;
    float man;
    if (exp <= 0) { /* subnormal */
        man     = a / std::ldexp(1.0f, -6) * 8.0f;
        int mi  = (int)std::nearbyint(man);
        if (mi >= 8) {
            return sign | 0x08; /* rounds up to min normal */
        }
        return sign | (uint8_t)mi;
    }
    man    = (m * 2.0f - 1.0f) * 8.0f;
    int mi = (int)std::nearbyint(man);
    if (mi >= 8) {
        mi = 0;
        exp++;
        if (exp >= 0xf && mi > 6) {
            return sign | 0x7e;
        }
    }
    if (exp == 0xf && mi == 7) {
        return sign | 0x7e; /* avoid NaN encoding */
    }
    if (exp > 0xf) {
        return sign | 0x7e;
    }
    return sign | (uint8_t)(exp << 3) | (uint8_t)mi;
}

float fp8e5m2_to_float(uint8_t v)
{
    uint32_t sign = (v >> 7) & 1, exp = (v >> 2) & 0x1f, man = v & 3;
    if (exp == 0x1f) {
        if (man) {
            return NAN;
        }
        return sign ? -INFINITY : INFINITY;
    }
    float val;
    if (exp == 0) {
        val = (float)man / 4.0f * std::ldexp(1.0f, -14);
    } else {
        val = (1.0f + (float)man / 4.0f) * std::ldexp(1.0f, (int)exp - 15);
    }
    return sign ? -val : val;
}

uint8_t float_to_fp8e5m2(float f)
{
    if (std::isnan(f)) {
        return 0x7e | 1;
    }
    uint8_t sign = std::signbit(f) ? 0x80 : 0;
    float   a    = std::fabs(f);
    if (std::isinf(f) || a > 57344.0f) {
        return sign | 0x7c;
    }
    if (a < std::ldexp(1.0f, -17)) {
        return sign;
    }
    int   e;
    float m   = std::frexp(a, &e);
    e -= 1;
    int exp = e + 15;
    if (exp <= 0) {
        int mi = (int)std::nearbyint(a / std::ldexp(1.0f, -14) * 4.0f);
        if (mi >= 4) {
            return sign | 0x04;
        }
        return sign | (uint8_t)mi;
    }
    int mi = (int)std::nearbyint((m * 2.0f - 1.0f) * 4.0f);
    if (mi >= 4) {
        mi = 0;
        exp++;
    }
    if (exp >= 0x1f) {
        return sign | 0x7c;
    }
    return sign | (uint8_t)(exp << 2) | (uint8_t)mi;
}

/* -------------------------------------------------------------- kernels */
template <typename T> struct Id {
    static T    load(const void *p, size_t i) { return ((const T *)p)[i]; }
    static void store(void *p, size_t i, T v) { ((T *)p)[i] = v; }
};
struct Bf16 {
    static float load(const void *p, size_t i)
    {
        return bf16_to_float(((const uint16_t *)p)[i]);
    }
    static void store(void *p, size_t i, float v)
    {
        ((uint16_t *)p)[i] = float_to_bf16(v);
    }
};
struct Fp16 {
    static float load(const void *p, size_t i)
    {
        return fp16_to_float(((const uint16_t *)p)[i]);
    }
    static void store(void *p, size_t i, float v)
    {
        ((uint16_t *)p)[i] = float_to_fp16(v);
    }
};
struct Fp8e4m3 {
    static float load(const void *p, size_t i)
    {
        return fp8e4m3_to_float(((const uint8_t *)p)[i]);
    }
    static void store(void *p, size_t i, float v)
    {
        ((uint8_t *)p)[i] = float_to_fp8e4m3(v);
    }
};
struct Fp8e5m2 {
    static float load(const void *p, size_t i)
    {
        return fp8e5m2_to_float(((const uint8_t *)p)[i]);
    }
    static void store(void *p, size_t i, float v)
    {
        ((uint8_t *)p)[i] = float_to_fp8e5m2(v);
    }
};

template <class Cv, typename A, bool IsInt>
static ucc_status_t reduce_typed(void *dst, const void *const *srcs,
                                 int n_srcs, size_t count,
                                 ucc_reduction_op_t op, double alpha)
{
    switch (op) {
#define UCC_EC_LOOP(expr, post)                                              \
    for (size_t i = 0; i < count; i++) {                                     \
        A acc = Cv::load(srcs[0], i);                                        \
        for (int s = 1; s < n_srcs; s++) {                                   \
            A v = Cv::load(srcs[s], i);                                      \
            expr;                                                            \
        }                                                                    \
        post;                                                                \
        Cv::store(dst, i, acc);                                              \
    }                                                                        \
    break
    case UCC_OP_SUM:
        UCC_EC_LOOP(acc = acc + v, acc = (A)(acc * (A)alpha));
    case UCC_OP_AVG:
        UCC_EC_LOOP(acc = acc + v, acc = (A)(acc * (A)alpha));
    case UCC_OP_PROD:
        UCC_EC_LOOP(acc = acc * v, (void)0);
    case UCC_OP_MAX:
        UCC_EC_LOOP(acc = v > acc ? v : acc, (void)0);
    case UCC_OP_MIN:
        UCC_EC_LOOP(acc = v < acc ? v : acc, (void)0);
    case UCC_OP_LAND:
        UCC_EC_LOOP(acc = (A)((acc != (A)0) && (v != (A)0)), (void)0);
    case UCC_OP_LOR:
        UCC_EC_LOOP(acc = (A)((acc != (A)0) || (v != (A)0)), (void)0);
    case UCC_OP_LXOR:
        UCC_EC_LOOP(acc = (A)((acc != (A)0) != (v != (A)0)), (void)0);
    case UCC_OP_BAND:
        if constexpr (IsInt) {
            UCC_EC_LOOP(acc = acc & v, (void)0);
        } else {
            return UCC_ERR_NOT_SUPPORTED;
        }
    case UCC_OP_BOR:
        if constexpr (IsInt) {
            UCC_EC_LOOP(acc = acc | v, (void)0);
        } else {
            return UCC_ERR_NOT_SUPPORTED;
        }
    case UCC_OP_BXOR:
        if constexpr (IsInt) {
            UCC_EC_LOOP(acc = acc ^ v, (void)0);
        } else {
            return UCC_ERR_NOT_SUPPORTED;
        }
    default: return UCC_ERR_NOT_SUPPORTED;
    }
    return UCC_OK;
#undef UCC_EC_LOOP
}

template <typename C>
static ucc_status_t reduce_complex(void *dst, const void *const *srcs,
                                   int n_srcs, size_t count,
                                   ucc_reduction_op_t op, double alpha)
{
    switch (op) {
    case UCC_OP_SUM:
    case UCC_OP_AVG:
        for (size_t i = 0; i < count; i++) {
            C acc = ((const C *)srcs[0])[i];
            for (int s = 1; s < n_srcs; s++) {
                acc += ((const C *)srcs[s])[i];
            }
            ((C *)dst)[i] = acc * (typename C::value_type)alpha;
        }
        return UCC_OK;
    case UCC_OP_PROD:
        for (size_t i = 0; i < count; i++) {
            C acc = ((const C *)srcs[0])[i];
            for (int s = 1; s < n_srcs; s++) {
                acc *= ((const C *)srcs[s])[i];
            }
            ((C *)dst)[i] = acc;
        }
        return UCC_OK;
    default: return UCC_ERR_NOT_SUPPORTED;
    }
}

ucc_status_t reduce(void *dst, const void *const *srcs, int n_srcs,
                    size_t count, ucc_datatype_t dt, ucc_reduction_op_t op,
                    double alpha)
{
    if (n_srcs < 1) {
        return UCC_ERR_INVALID_PARAM;
    }
    if (!ucc_dt_is_predefined(dt)) {
        /* generic datatype: pairwise user reduce callback */
        const ucc_generic_dt_ops_t *g = ucc_dt_generic_ops(dt);
        if (!g || !(g->flags & UCC_GENERIC_DT_OPS_FLAG_REDUCE) ||
            !g->reduce || alpha != 1.0) {
            return UCC_ERR_NOT_SUPPORTED;
        }
        size_t esz = ucc_dt_size(dt);
        if (dst != srcs[0]) {
            memcpy(dst, srcs[0], count * esz);
        }
        for (int s = 1; s < n_srcs; s++) {
            ucc_status_t st =
                g->reduce(dst, srcs[s], dst, count, g->cookie);
            if (st != UCC_OK) {
                return st;
            }
        }
        return UCC_OK;
    }
    switch (dt) {
    case UCC_DT_INT8:
        return reduce_typed<Id<int8_t>, int8_t, true>(dst, srcs, n_srcs,
                                                      count, op, alpha);
    case UCC_DT_UINT8:
        return reduce_typed<Id<uint8_t>, uint8_t, true>(dst, srcs, n_srcs,
                                                        count, op, alpha);
    case UCC_DT_INT16:
        return reduce_typed<Id<int16_t>, int16_t, true>(dst, srcs, n_srcs,
                                                        count, op, alpha);
    case UCC_DT_UINT16:
        return reduce_typed<Id<uint16_t>, uint16_t, true>(dst, srcs, n_srcs,
                                                          count, op, alpha);
    case UCC_DT_INT32:
        return reduce_typed<Id<int32_t>, int32_t, true>(dst, srcs, n_srcs,
                                                        count, op, alpha);
    case UCC_DT_UINT32:
        return reduce_typed<Id<uint32_t>, uint32_t, true>(dst, srcs, n_srcs,
                                                          count, op, alpha);
    case UCC_DT_INT64:
        return reduce_typed<Id<int64_t>, int64_t, true>(dst, srcs, n_srcs,
                                                        count, op, alpha);
    case UCC_DT_UINT64:
        return reduce_typed<Id<uint64_t>, uint64_t, true>(dst, srcs, n_srcs,
                                                          count, op, alpha);
    case UCC_DT_INT128:
        return reduce_typed<Id<__int128>, __int128, true>(dst, srcs, n_srcs,
                                                          count, op, alpha);
    case UCC_DT_UINT128:
        return reduce_typed<Id<unsigned __int128>, unsigned __int128, true>(
            dst, srcs, n_srcs, count, op, alpha);
    case UCC_DT_FLOAT16:
        return reduce_typed<Fp16, float, false>(dst, srcs, n_srcs, count, op,
                                                alpha);
    case UCC_DT_BFLOAT16:
        return reduce_typed<Bf16, float, false>(dst, srcs, n_srcs, count, op,
                                                alpha);
    case UCC_DT_FLOAT8_E4M3:
        return reduce_typed<Fp8e4m3, float, false>(dst, srcs, n_srcs, count,
                                                   op, alpha);
    case UCC_DT_FLOAT8_E5M2:
        return reduce_typed<Fp8e5m2, float, false>(dst, srcs, n_srcs, count,
                                                   op, alpha);
    case UCC_DT_FLOAT32:
        return reduce_typed<Id<float>, float, false>(dst, srcs, n_srcs, count,
                                                     op, alpha);
    case UCC_DT_FLOAT64:
        return reduce_typed<Id<double>, double, false>(dst, srcs, n_srcs,
                                                       count, op, alpha);
    case UCC_DT_FLOAT128:
        return reduce_typed<Id<long double>, long double, false>(
            dst, srcs, n_srcs, count, op, alpha);
    case UCC_DT_FLOAT32_COMPLEX:
        return reduce_complex<std::complex<float>>(dst, srcs, n_srcs, count,
                                                   op, alpha);
    case UCC_DT_FLOAT64_COMPLEX:
        return reduce_complex<std::complex<double>>(dst, srcs, n_srcs, count,
                                                    op, alpha);
    case UCC_DT_FLOAT128_COMPLEX:
        return reduce_complex<std::complex<long double>>(dst, srcs, n_srcs,
                                                         count, op, alpha);
    default: return UCC_ERR_NOT_SUPPORTED;
    }
}

ucc_status_t reduce_strided(void *dst, const void *src1, const void *src2,
                            size_t stride_bytes, int n_src2, size_t count,
                            ucc_datatype_t dt, ucc_reduction_op_t op,
                            double alpha)
{
    const void *srcs[64];
    if (n_src2 + 1 > 64) {
        return UCC_ERR_NOT_SUPPORTED;
    }
    srcs[0] = src1;
    for (int i = 0; i < n_src2; i++) {
        srcs[i + 1] = (const uint8_t *)src2 + stride_bytes * (size_t)i;
    }
    return reduce(dst, srcs, n_src2 + 1, count, dt, op, alpha);
}

void copy(void *dst, const void *src, size_t bytes)
{
    memcpy(dst, src, bytes);
}

} // namespace ec_cpu
} // namespace ucc
