/* gfx950 execution-component kernels.
 *
 * Design (MI355X-first, see /opt/skills/guides/cdna_hip_programming.md):
 *  - 256-thread workgroups (4 waves of 64), grid-stride loops capped so the
 *    scheduler keeps the chip busy without oversubscription (G11).
 *  - 16 B per lane vector packs (G13): bf16/fp16 loaded as ushort8-class
 *    packs, converted to f32, accumulated in f32; fp64 in f64; ints in T.
 *  - fused allreduce: cross-GPU arrival flags are u64 system-scope atomics
 *    on fine-grained memory; payload ordering = __threadfence_system on the
 *    producer before the flag store, one system acquire on the consumer
 *    after the poll matches (the G16 protocol lifted to system scope for
 *    xGMI peers). Every spin is bounded and reports via error_word.
 */
#include <hip/hip_runtime.h>

#include <algorithm>
#include <type_traits>

#include "../ec_hip.h"

namespace ucc {
namespace ec_hip {

/* ------------------------------------------------------------ convert  */
__device__ __forceinline__ float d_bf16_to_f(uint16_t v)
{
    union {
        uint32_t u;
        float    f;
    } c;
    c.u = (uint32_t)v << 16;
    return c.f;
}
__device__ __forceinline__ uint16_t d_f_to_bf16(float f)
{
    union {
        uint32_t u;
        float    f;
    } c;
    c.f = f;
    uint32_t r = 0x7fff + ((c.u >> 16) & 1);
    return (uint16_t)((c.u + r) >> 16);
}

/* fp8 (OCP): gfx950 has native OCP fp8<->f32 converts (VOP1/VOP3
 * v_cvt_f32_fp8 / v_cvt_pk_fp8_f32) — use them on device; the bit-math
 * path below stays as the portable reference (and matches the host
 * codec bit-for-bit on finite values). */
#if defined(__gfx950__)
#define UCC_NATIVE_FP8 1
#endif

#ifdef UCC_NATIVE_FP8
/* paired converts: one VOP handles 2 elements (v_cvt_pk_f32_fp8 /
 * v_cvt_pk_fp8_f32) — halves the VALU cost of the fp8 reduce, which is
 * 2x the elements per byte of bf16 */
__device__ __forceinline__ void d_e4m3x2_to_f(uint8_t lo, uint8_t hi,
                                              float &a, float &b)
{
    int packed = (int)lo | ((int)hi << 8);
    auto v2    = __builtin_amdgcn_cvt_pk_f32_fp8(packed, false);
    a          = v2[0];
    b          = v2[1];
}
__device__ __forceinline__ void d_f_to_e4m3x2(float a, float b,
                                              uint8_t &lo, uint8_t &hi)
{
    int w = __builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0, false);
    lo    = (uint8_t)w;
    hi    = (uint8_t)(w >> 8);
}
__device__ __forceinline__ void d_e5m2x2_to_f(uint8_t lo, uint8_t hi,
                                              float &a, float &b)
{
    int packed = (int)lo | ((int)hi << 8);
    auto v2    = __builtin_amdgcn_cvt_pk_f32_bf8(packed, false);
    a          = v2[0];
    b          = v2[1];
}
__device__ __forceinline__ void d_f_to_e5m2x2(float a, float b,
                                              uint8_t &lo, uint8_t &hi)
{
    int w = __builtin_amdgcn_cvt_pk_bf8_f32(a, b, 0, false);
    lo    = (uint8_t)w;
    hi    = (uint8_t)(w >> 8);
}
/* dword-wide converts: the pk builtins take a word-select operand, so
 * one dword (4 fp8) costs exactly 2 VOPs each way with NO byte
 * repacking VALU (the paired path above still assembles a 16-bit
 * packed operand from two byte registers per pair). */
__device__ __forceinline__ void d_e4m3x4_to_f(uint32_t w, float &a,
                                              float &b, float &c,
                                              float &d)
{
    auto lo = __builtin_amdgcn_cvt_pk_f32_fp8((int)w, false);
    auto hi = __builtin_amdgcn_cvt_pk_f32_fp8((int)w, true);
    a       = lo[0];
    b       = lo[1];
    c       = hi[0];
    d       = hi[1];
}
__device__ __forceinline__ uint32_t d_f_to_e4m3x4(float a, float b,
                                                  float c, float d)
{
    int w = __builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0, false);
    w     = __builtin_amdgcn_cvt_pk_fp8_f32(c, d, w, true);
    return (uint32_t)w;
}
__device__ __forceinline__ void d_e5m2x4_to_f(uint32_t w, float &a,
                                              float &b, float &c,
                                              float &d)
{
    auto lo = __builtin_amdgcn_cvt_pk_f32_bf8((int)w, false);
    auto hi = __builtin_amdgcn_cvt_pk_f32_bf8((int)w, true);
    a       = lo[0];
    b       = lo[1];
    c       = hi[0];
    d       = hi[1];
}
__device__ __forceinline__ uint32_t d_f_to_e5m2x4(float a, float b,
                                                  float c, float d)
{
    int w = __builtin_amdgcn_cvt_pk_bf8_f32(a, b, 0, false);
    w     = __builtin_amdgcn_cvt_pk_bf8_f32(c, d, w, true);
    return (uint32_t)w;
}
__device__ __forceinline__ float d_e4m3_to_f(uint8_t v)
{
    return __builtin_amdgcn_cvt_f32_fp8((int)v, 0);
}
__device__ __forceinline__ uint8_t d_f_to_e4m3(float f)
{
    return (uint8_t)__builtin_amdgcn_cvt_pk_fp8_f32(f, f, 0, false);
}
__device__ __forceinline__ float d_e5m2_to_f(uint8_t v)
{
    return __builtin_amdgcn_cvt_f32_bf8((int)v, 0);
}
__device__ __forceinline__ uint8_t d_f_to_e5m2(float f)
{
    return (uint8_t)__builtin_amdgcn_cvt_pk_bf8_f32(f, f, 0, false);
}
#else
__device__ __forceinline__ float d_e4m3_to_f(uint8_t v)
{
    uint32_t sign = v >> 7, exp = (v >> 3) & 0xf, man = v & 7;
    float    val;
    if (exp == 0xf && man == 7) {
        return __builtin_nanf("");
    }
    if (exp == 0) {
        val = (float)man * (1.0f / 512.0f);
    } else {
        val = (1.0f + (float)man * 0.125f) * exp2f((float)exp - 7.0f);
    }
    return sign ? -val : val;
}
__device__ __forceinline__ uint8_t d_f_to_e4m3(float f)
{
    if (isnan(f)) {
        return 0x7f;
    }
    uint8_t sign = f < 0.0f ? 0x80 : 0;
    float   a    = fabsf(f);
    if (a >= 448.0f) {
        return sign | 0x7e;
    }
    if (a < 0.001953125f) { /* 2^-9 */
        return sign;
    }
    int   e;
    float m   = frexpf(a, &e);
    int   exp = e - 1 + 7;
    if (exp <= 0) {
        int mi = (int)rintf(a * 512.0f);
        if (mi >= 8) {
            return sign | 0x08;
        }
        return sign | (uint8_t)mi;
    }
    int mi = (int)rintf((m * 2.0f - 1.0f) * 8.0f);
    if (mi >= 8) {
        mi = 0;
        exp++;
    }
    if (exp > 0xf || (exp == 0xf && mi == 7)) {
        return sign | 0x7e;
    }
    return sign | (uint8_t)(exp << 3) | (uint8_t)mi;
}
__device__ __forceinline__ float d_e5m2_to_f(uint8_t v)
{
    uint32_t sign = v >> 7, exp = (v >> 2) & 0x1f, man = v & 3;
    if (exp == 0x1f) {
        return man ? __builtin_nanf("")
                   : (sign ? -__builtin_inff() : __builtin_inff());
    }
    float val;
    if (exp == 0) {
        val = (float)man * 0.25f * exp2f(-14.0f);
    } else {
        val = (1.0f + (float)man * 0.25f) * exp2f((float)exp - 15.0f);
    }
    return sign ? -val : val;
}
__device__ __forceinline__ uint8_t d_f_to_e5m2(float f)
{
    if (isnan(f)) {
        return 0x7f;
    }
    uint8_t sign = signbit(f) ? 0x80 : 0;
    float   a    = fabsf(f);
    if (isinf(f) || a > 57344.0f) {
        return sign | 0x7c;
    }
    if (a < exp2f(-17.0f)) {
        return sign;
    }
    int   e;
    float m   = frexpf(a, &e);
    int   exp = e - 1 + 15;
    if (exp <= 0) {
        int mi = (int)rintf(a * exp2f(14.0f) * 4.0f);
        if (mi >= 4) {
            return sign | 0x04;
        }
        return sign | (uint8_t)mi;
    }
    int mi = (int)rintf((m * 2.0f - 1.0f) * 4.0f);
    if (mi >= 4) {
        mi = 0;
        exp++;
    }
    if (exp >= 0x1f) {
        return sign | 0x7c;
    }
    return sign | (uint8_t)(exp << 2) | (uint8_t)mi;
}
#endif /* UCC_NATIVE_FP8 */

/* type traits: storage type T <-> accumulator A */
template <typename T> struct Cvt {
    using A = T;
    static __device__ __forceinline__ A load(T v) { return v; }
    static __device__ __forceinline__ T store(A v) { return (T)v; }
};
struct bf16_t {
    uint16_t v;
};
template <> struct Cvt<bf16_t> {
    using A = float;
    static __device__ __forceinline__ float load(bf16_t x)
    {
        return d_bf16_to_f(x.v);
    }
    static __device__ __forceinline__ bf16_t store(float f)
    {
        return {d_f_to_bf16(f)};
    }
};
struct fp16_t {
    _Float16 v;
};
template <> struct Cvt<fp16_t> {
    using A = float;
    static __device__ __forceinline__ float load(fp16_t x)
    {
        return (float)x.v;
    }
    static __device__ __forceinline__ fp16_t store(float f)
    {
        return {(_Float16)f};
    }
};
struct e4m3_t {
    uint8_t v;
};
template <> struct Cvt<e4m3_t> {
    using A = float;
    static __device__ __forceinline__ float load(e4m3_t x)
    {
        return d_e4m3_to_f(x.v);
    }
    static __device__ __forceinline__ e4m3_t store(float f)
    {
        return {d_f_to_e4m3(f)};
    }
};
struct e5m2_t {
    uint8_t v;
};
template <> struct Cvt<e5m2_t> {
    using A = float;
    static __device__ __forceinline__ float load(e5m2_t x)
    {
        return d_e5m2_to_f(x.v);
    }
    static __device__ __forceinline__ e5m2_t store(float f)
    {
        return {d_f_to_e5m2(f)};
    }
};

/* op ids match ucc_reduction_op_t */
template <typename A, int OP>
__device__ __forceinline__ A red(A a, A b)
{
    if constexpr (OP == 0 || OP == 12) { /* SUM / AVG */
        return a + b;
    } else if constexpr (OP == 1) {
        return a * b;
    } else if constexpr (OP == 2) {
        return a > b ? a : b;
    } else if constexpr (OP == 3) {
        return a < b ? a : b;
    } else if constexpr (OP == 4) {
        return (A)((a != (A)0) && (b != (A)0));
    } else if constexpr (OP == 5) {
        return (A)((a != (A)0) || (b != (A)0));
    } else if constexpr (OP == 6) {
        return (A)((a != (A)0) != (b != (A)0));
    } else if constexpr (OP == 7) {
        if constexpr (!std::is_floating_point<A>::value) {
            return a & b;
        }
    } else if constexpr (OP == 8) {
        if constexpr (!std::is_floating_point<A>::value) {
            return a | b;
        }
    } else if constexpr (OP == 9) {
        if constexpr (!std::is_floating_point<A>::value) {
            return a ^ b;
        }
    }
    return a;
}

template <typename A>
__device__ __forceinline__ A apply_alpha(A v, float alpha)
{
    if constexpr (std::is_floating_point<A>::value) {
        return (A)(v * (A)alpha);
    } else {
        (void)alpha;
        return v;
    }
}

/* ------------------------------------------------------------- reduce  */
/* VEC chosen so VEC*sizeof(T) == 16 bytes (one dwordx4 per lane). */
template <typename T> struct VecOf {
    static constexpr int value = 16 / (int)sizeof(T);
};

template <typename T, int OP, int VEC>
struct __align__(16) Pack {
    T v[VEC];
};

#ifdef UCC_NATIVE_FP8
template <typename T>
inline constexpr bool is_fp8_v = std::is_same<T, e4m3_t>::value ||
                                 std::is_same<T, e5m2_t>::value;

template <typename T>
__device__ __forceinline__ void cvt2_load(T lo, T hi, float &a, float &b)
{
    if constexpr (std::is_same<T, e4m3_t>::value) {
        d_e4m3x2_to_f(lo.v, hi.v, a, b);
    } else {
        d_e5m2x2_to_f(lo.v, hi.v, a, b);
    }
}
template <typename T>
__device__ __forceinline__ void cvt2_store(float a, float b, T &lo, T &hi)
{
    if constexpr (std::is_same<T, e4m3_t>::value) {
        d_f_to_e4m3x2(a, b, lo.v, hi.v);
    } else {
        d_f_to_e5m2x2(a, b, lo.v, hi.v);
    }
}
template <typename T>
__device__ __forceinline__ void cvt4_load(uint32_t w, float &a, float &b,
                                          float &c, float &d)
{
    if constexpr (std::is_same<T, e4m3_t>::value) {
        d_e4m3x4_to_f(w, a, b, c, d);
    } else {
        d_e5m2x4_to_f(w, a, b, c, d);
    }
}
template <typename T>
__device__ __forceinline__ uint32_t cvt4_store(float a, float b, float c,
                                               float d)
{
    if constexpr (std::is_same<T, e4m3_t>::value) {
        return d_f_to_e4m3x4(a, b, c, d);
    }
    return d_f_to_e5m2x4(a, b, c, d);
}
#else
template <typename T> inline constexpr bool is_fp8_v = false;
template <typename T>
__device__ __forceinline__ void cvt2_load(T, T, float &, float &) {}
template <typename T>
__device__ __forceinline__ void cvt2_store(float, float, T &, T &) {}
template <typename T>
__device__ __forceinline__ void cvt4_load(uint32_t, float &, float &,
                                          float &, float &)
{
}
template <typename T>
__device__ __forceinline__ uint32_t cvt4_store(float, float, float,
                                               float)
{
    return 0;
}
#endif

/* vector-wide load/accumulate/store with dword-wide fp8 fast paths
 * (4 elements per word-selected pk convert pair, zero repack VALU) */
template <int N> struct __align__(16) PackW {
    uint32_t w[N];
};

template <typename T, int OP, int VEC, typename A, typename P>
__device__ __forceinline__ void vload(const P &x, A (&r)[VEC])
{
    if constexpr (is_fp8_v<T> && std::is_same<A, float>::value &&
                  VEC % 4 == 0) {
        auto pw = __builtin_bit_cast(PackW<VEC / 4>, x);
#pragma unroll
        for (int j = 0; j < VEC / 4; j++) {
            cvt4_load<T>(pw.w[j], r[4 * j], r[4 * j + 1], r[4 * j + 2],
                         r[4 * j + 3]);
        }
    } else {
#pragma unroll
        for (int k = 0; k < VEC; k++) {
            r[k] = Cvt<T>::load(x.v[k]);
        }
    }
}
template <typename T, int OP, int VEC, typename A, typename P>
__device__ __forceinline__ void vaccum(A (&r)[VEC], const P &x)
{
    if constexpr (is_fp8_v<T> && std::is_same<A, float>::value &&
                  VEC % 4 == 0) {
        auto pw = __builtin_bit_cast(PackW<VEC / 4>, x);
#pragma unroll
        for (int j = 0; j < VEC / 4; j++) {
            float a, b, c, d;
            cvt4_load<T>(pw.w[j], a, b, c, d);
            r[4 * j]     = red<A, OP>(r[4 * j], a);
            r[4 * j + 1] = red<A, OP>(r[4 * j + 1], b);
            r[4 * j + 2] = red<A, OP>(r[4 * j + 2], c);
            r[4 * j + 3] = red<A, OP>(r[4 * j + 3], d);
        }
    } else {
#pragma unroll
        for (int k = 0; k < VEC; k++) {
            r[k] = red<A, OP>(r[k], Cvt<T>::load(x.v[k]));
        }
    }
}
template <typename T, int OP, int VEC, typename A, typename P>
__device__ __forceinline__ void vstore(P &o, const A (&r)[VEC],
                                       float alpha)
{
    if constexpr (is_fp8_v<T> && std::is_same<A, float>::value &&
                  VEC % 4 == 0) {
        PackW<VEC / 4> pw;
#pragma unroll
        for (int j = 0; j < VEC / 4; j++) {
            pw.w[j] = cvt4_store<T>(apply_alpha<A>(r[4 * j], alpha),
                                    apply_alpha<A>(r[4 * j + 1], alpha),
                                    apply_alpha<A>(r[4 * j + 2], alpha),
                                    apply_alpha<A>(r[4 * j + 3], alpha));
        }
        o = __builtin_bit_cast(P, pw);
    } else {
#pragma unroll
        for (int k = 0; k < VEC; k++) {
            o.v[k] = Cvt<T>::store(apply_alpha<A>(r[k], alpha));
        }
    }
}

template <typename T, int OP, int VEC>
__global__ void k_reduce(ReduceArgs a)
{
    using A            = typename Cvt<T>::A;
    const uint64_t nv  = a.count / VEC;
    const int      n   = a.n_srcs;
    uint64_t       i   = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const uint64_t str = (uint64_t)gridDim.x * blockDim.x;
    using P            = Pack<T, OP, VEC>;
    for (; i < nv; i += str) {
        P acc = ((const P *)a.srcs[0])[i];
        A  r[VEC];
        vload<T, OP, VEC, A>(acc, r);
        for (int s = 1; s < n; s++) {
            P x = ((const P *)a.srcs[s])[i];
            vaccum<T, OP, VEC, A>(r, x);
        }
        P out;
        vstore<T, OP, VEC, A>(out, r, a.alpha);
        ((P *)a.dst)[i] = out;
    }
    /* tail (scalar) */
    uint64_t t = nv * VEC + ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x);
    for (; t < a.count; t += str) {
        A r = Cvt<T>::load(((const T *)a.srcs[0])[t]);
        for (int s = 1; s < n; s++) {
            r = red<A, OP>(r, Cvt<T>::load(((const T *)a.srcs[s])[t]));
        }
        ((T *)a.dst)[t] = Cvt<T>::store(apply_alpha<A>(r, a.alpha));
    }
}

template <typename T, int OP>
__global__ void k_reduce_scalar(ReduceArgs a)
{
    using A            = typename Cvt<T>::A;
    uint64_t       i   = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const uint64_t str = (uint64_t)gridDim.x * blockDim.x;
    for (; i < a.count; i += str) {
        A r = Cvt<T>::load(((const T *)a.srcs[0])[i]);
        for (int s = 1; s < a.n_srcs; s++) {
            r = red<A, OP>(r, Cvt<T>::load(((const T *)a.srcs[s])[i]));
        }
        ((T *)a.dst)[i] = Cvt<T>::store(apply_alpha<A>(r, a.alpha));
    }
}

/* -------------------------------------------------------- gather copy  */
__device__ __forceinline__ void d_copy_bytes_part(uint8_t *dst,
                                                  const uint8_t *src,
                                                  uint64_t len,
                                                  uint64_t vblock,
                                                  uint64_t vgrid)
{
    const uint64_t tid = vblock * blockDim.x + threadIdx.x;
    const uint64_t str = vgrid * blockDim.x;
    if ((((uintptr_t)dst | (uintptr_t)src) & 15) == 0) {
        uint64_t nv = len / 16;
        uint64_t i  = tid;
        /* 4-deep load batching: all four 16B loads issue before the
         * first dependent store (64B outstanding per lane — G7). */
        for (; i + 3 * str < nv; i += 4 * str) {
            uint4 v0 = ((const uint4 *)src)[i];
            uint4 v1 = ((const uint4 *)src)[i + str];
            uint4 v2 = ((const uint4 *)src)[i + 2 * str];
            uint4 v3 = ((const uint4 *)src)[i + 3 * str];
            ((uint4 *)dst)[i]           = v0;
            ((uint4 *)dst)[i + str]     = v1;
            ((uint4 *)dst)[i + 2 * str] = v2;
            ((uint4 *)dst)[i + 3 * str] = v3;
        }
        for (; i < nv; i += str) {
            ((uint4 *)dst)[i] = ((const uint4 *)src)[i];
        }
        for (uint64_t t = nv * 16 + tid; t < len; t += str) {
            dst[t] = src[t];
        }
    } else {
        for (uint64_t i = tid; i < len; i += str) {
            dst[i] = src[i];
        }
    }
}

__device__ __forceinline__ void d_copy_bytes(uint8_t *dst,
                                             const uint8_t *src,
                                             uint64_t len)
{
    d_copy_bytes_part(dst, src, len, blockIdx.x, gridDim.x);
}


__global__ void k_gather_copy(GatherArgs a)
{
    /* blocks partitioned across sources: peers' xGMI links stream
     * concurrently (sequential per-peer copies would serialize on one
     * link at a time) */
    const int grp = (int)gridDim.x / a.n > 0 ? (int)gridDim.x / a.n : 1;
    int       s   = (int)blockIdx.x / grp;
    int       bid = (int)blockIdx.x % grp;
    if ((int)gridDim.x < a.n) { /* fewer blocks than sources: loop */
        for (s = (int)blockIdx.x; s < a.n; s += gridDim.x) {
            uint8_t       *dst = (uint8_t *)a.dst_base + a.offs[s];
            const uint8_t *src = (const uint8_t *)a.srcs[s];
            d_copy_bytes_part(dst, src, a.lens[s], 0, 1);
        }
        return;
    }
    if (s >= a.n) {
        return;
    }
    uint8_t       *dst = (uint8_t *)a.dst_base + a.offs[s];
    const uint8_t *src = (const uint8_t *)a.srcs[s];
    if ((((uintptr_t)dst | (uintptr_t)src) & 3) == 0 ||
        (((uintptr_t)dst | (uintptr_t)src) & 15) == 0) {
        d_copy_bytes_part(dst, src, a.lens[s], (uint64_t)bid,
                          (uint64_t)grp);
    } else {
        const uint64_t tid = (uint64_t)bid * blockDim.x + threadIdx.x;
        const uint64_t str = (uint64_t)grp * blockDim.x;
        for (uint64_t i = tid; i < a.lens[s]; i += str) {
            dst[i] = src[i];
        }
    }
}

/* ---------------------------------------------------- fused allreduce  */
__device__ __forceinline__ uint64_t
sys_load(const uint64_t *p)
{
    return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
}
__device__ __forceinline__ void sys_store(uint64_t *p, uint64_t v)
{
    __hip_atomic_store(p, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_SYSTEM);
}

constexpr uint64_t kSpinLimit = kDefaultSpinLimit;
__device__ __forceinline__ uint64_t spin_cap(uint64_t v)
{
    return v ? v : kSpinLimit;
}

template <typename T, int OP, int VEC>
__global__ void k_fused_allreduce(FusedArgs a)
{
    using A = typename Cvt<T>::A;
    /* 1. stage src -> my scratch (all blocks, grid-stride, 16B packs) */
    {
        const uint64_t tid =
            (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
        const uint64_t str = (uint64_t)gridDim.x * blockDim.x;
        const uint64_t nv  = a.count / VEC;
        using P            = Pack<T, OP, VEC>;
        for (uint64_t i = tid; i < nv; i += str) {
            ((P *)a.my_scratch)[i] = ((const P *)a.src)[i];
        }
        for (uint64_t i = nv * VEC + tid; i < a.count; i += str) {
            ((T *)a.my_scratch)[i] = ((const T *)a.src)[i];
        }
    }
    __threadfence_system();
    __syncthreads();
    /* 2. grid arrival: each block bumps the staging counter; block 0
     * waits for all blocks, then signals every peer (incl. self).
     * staging counter is u64 index kMaxRanks*kMaxSlots + slot (monotone,
     * grows by nblocks per use -> target seq*nblocks). */
    uint64_t *stage_cnt = a.local_flags + 8 * kMaxRanks + a.slot;
    __shared__ int s_err;
    if (threadIdx.x == 0) {
        s_err = 0;
        __hip_atomic_fetch_add(stage_cnt, 1, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
    }
    __syncthreads();
    if (blockIdx.x == 0) {
        if (threadIdx.x == 0) {
            uint64_t spins = 0;
            const uint64_t cap = spin_cap(a.spin_limit);
            while (sys_load(stage_cnt) < a.stage_target) {
                if (++spins > cap) {
                    sys_store(a.error_word, 1);
                    s_err = 1;
                    break;
                }
                __builtin_amdgcn_s_sleep(2);
            }
        }
        __syncthreads();
        if (!s_err && (int)threadIdx.x < a.nranks) {
            sys_store(a.peer_flags[threadIdx.x] +
                          (uint64_t)a.slot * kMaxRanks + a.rank,
                      a.seq);
        }
    }
    /* 3. all blocks wait for all ranks' arrivals (timeout: all threads
     * leave together — never return while siblings sit at a barrier) */
    if (!s_err && threadIdx.x < 64) {
        int      j     = (int)threadIdx.x;
        uint64_t spins = 0;
        const uint64_t cap = spin_cap(a.spin_limit);
        if (j < a.nranks) {
            const uint64_t *f =
                a.local_flags + (uint64_t)a.slot * kMaxRanks + j;
            while (sys_load(f) < a.seq) {
                if (++spins > cap) {
                    sys_store(a.error_word, 1);
                    s_err = 1;
                    break;
                }
                __builtin_amdgcn_s_sleep(2);
            }
        }
    }
    __syncthreads();
    if (s_err) {
        return;
    }
    __threadfence_system(); /* acquire: drop stale lines before peer reads */
    /* 4. reduce all peers' scratch into dst */
    {
        const uint64_t tid =
            (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
        const uint64_t str = (uint64_t)gridDim.x * blockDim.x;
        const uint64_t nv  = a.count / VEC;
        using P            = Pack<T, OP, VEC>;
        const int n        = a.nranks;
        for (uint64_t i = tid; i < nv; i += str) {
            P acc = ((const P *)a.peer_scratch[0])[i];
            A r[VEC];
            vload<T, OP, VEC, A>(acc, r);
            for (int s = 1; s < n; s++) {
                P x = ((const P *)a.peer_scratch[s])[i];
                vaccum<T, OP, VEC, A>(r, x);
            }
            P out;
            vstore<T, OP, VEC, A>(out, r, a.alpha);
            ((P *)a.dst)[i] = out;
        }
        for (uint64_t t = nv * VEC + tid; t < a.count; t += str) {
            A r = Cvt<T>::load(((const T *)a.peer_scratch[0])[t]);
            for (int s = 1; s < n; s++) {
                r = red<A, OP>(r, Cvt<T>::load(((const T *)a.peer_scratch[s])[t]));
            }
            ((T *)a.dst)[t] = Cvt<T>::store(apply_alpha<A>(r, a.alpha));
        }
    }
    /* 5. host-visible completion: the last arriving block publishes
     * done_seq to the pinned word — the host polls plain memory
     * instead of event record+query (see FusedArgs.done_host). dst
     * stores are fenced before the arrival add; the release store
     * makes them visible to the host with the flag. */
    if (a.done_host) {
        __threadfence_system();
        __syncthreads();
        if (threadIdx.x == 0) {
            uint64_t v = __hip_atomic_fetch_add(
                             a.local_flags + kFusedDoneBase + a.slot, 1,
                             __ATOMIC_ACQ_REL,
                             __HIP_MEMORY_SCOPE_AGENT) +
                         1;
            if (v == a.done_target) {
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                __hip_atomic_store(a.done_host, a.done_seq,
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
            }
        }
    }
}

/* ------------------------------------- graph-capturable fused allreduce */
template <typename T, int OP, int VEC>
__global__ void k_fused_allreduce_graph(GraphFusedArgs a)
{
    using A = typename Cvt<T>::A;
    /* 0. device-side iteration number: this block's launch count. Every
     * block of every launch/replay increments its own counter exactly
     * once, so all blocks of one launch observe the same value. */
    uint64_t *my_cnt = a.local_flags + kGraphCntBase +
                       (uint64_t)a.slot * kMaxGraphBlocks + blockIdx.x;
    __shared__ uint64_t s_seq;
    if (threadIdx.x == 0) {
        s_seq = __hip_atomic_fetch_add(my_cnt, 1, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_AGENT) +
                1;
    }
    __syncthreads();
    const uint64_t seq = s_seq;
    const uint64_t par = (seq & 1) * a.parity_stride;
    /* 1. stage src -> my scratch (parity area for this iteration) */
    {
        const uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
        const uint64_t str = (uint64_t)gridDim.x * blockDim.x;
        const uint64_t nv  = a.count / VEC;
        using P            = Pack<T, OP, VEC>;
        P *mysc = (P *)((uint8_t *)a.my_scratch + par);
        for (uint64_t i = tid; i < nv; i += str) {
            mysc[i] = ((const P *)a.src)[i];
        }
        for (uint64_t i = nv * VEC + tid; i < a.count; i += str) {
            ((T *)mysc)[i] = ((const T *)a.src)[i];
        }
    }
    __threadfence_system();
    __syncthreads();
    /* 2. grid arrival; block 0 signals all peers with seq */
    uint64_t *stage_cnt = a.local_flags + kStageCntBase + a.slot;
    __shared__ int s_err;
    if (threadIdx.x == 0) {
        s_err = 0;
        __hip_atomic_fetch_add(stage_cnt, 1, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
    }
    __syncthreads();
    if (blockIdx.x == 0) {
        if (threadIdx.x == 0) {
            uint64_t spins  = 0;
            uint64_t target = seq * (uint64_t)a.nblocks;
            while (sys_load(stage_cnt) < target) {
                if (++spins > kSpinLimit) {
                    sys_store(a.error_word, 1);
                    s_err = 1;
                    break;
                }
                __builtin_amdgcn_s_sleep(2);
            }
        }
        __syncthreads();
        if (!s_err && (int)threadIdx.x < a.nranks) {
            sys_store(a.peer_flags[threadIdx.x] +
                          (uint64_t)a.slot * kMaxRanks + a.rank,
                      seq);
        }
    }
    /* 3. wait all ranks arrived for this iteration (timeout: all threads
     * leave together) */
    if (!s_err && threadIdx.x < 64) {
        int      j     = (int)threadIdx.x;
        uint64_t spins = 0;
        if (j < a.nranks) {
            const uint64_t *f =
                a.local_flags + (uint64_t)a.slot * kMaxRanks + j;
            while (sys_load(f) < seq) {
                if (++spins > kSpinLimit) {
                    sys_store(a.error_word, 1);
                    s_err = 1;
                    break;
                }
                __builtin_amdgcn_s_sleep(2);
            }
        }
    }
    __syncthreads();
    if (s_err) {
        return;
    }
    __threadfence_system();
    /* 4. reduce all peers' scratch (this iteration's parity) into dst.
     * Why parity is sufficient: a peer can start staging replay i+1 while
     * I still read its replay-i data (staging precedes the handshake),
     * but it writes the other parity area. It cannot reach replay i+2
     * (same parity as i) until the i+1 handshake completed, which needs
     * MY i+1 launch — stream-ordered after my reduce of i finished. */
    {
        const uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
        const uint64_t str = (uint64_t)gridDim.x * blockDim.x;
        const uint64_t nv  = a.count / VEC;
        using P            = Pack<T, OP, VEC>;
        const int n        = a.nranks;
        for (uint64_t i = tid; i < nv; i += str) {
            P acc = ((const P *)((const uint8_t *)a.peer_scratch[0] + par))[i];
            A r[VEC];
            vload<T, OP, VEC, A>(acc, r);
            for (int s = 1; s < n; s++) {
                P x = ((const P *)((const uint8_t *)a.peer_scratch[s] +
                                   par))[i];
                vaccum<T, OP, VEC, A>(r, x);
            }
            P out;
            vstore<T, OP, VEC, A>(out, r, a.alpha);
            ((P *)a.dst)[i] = out;
        }
        for (uint64_t t = nv * VEC + tid; t < a.count; t += str) {
            A r = Cvt<T>::load(
                ((const T *)((const uint8_t *)a.peer_scratch[0] + par))[t]);
            for (int s = 1; s < n; s++) {
                r = red<A, OP>(
                    r, Cvt<T>::load(((const T *)((const uint8_t *)
                                                     a.peer_scratch[s] +
                                                 par))[t]));
            }
            ((T *)a.dst)[t] = Cvt<T>::store(apply_alpha<A>(r, a.alpha));
        }
    }
}

/* ------------------------------------------- device-gated staged colls */
__device__ __forceinline__ int gated_idx(int phase, int slot, int parity)
{
    return kGatedCntBase + (phase * kGatedSlots + slot) * 2 + parity;
}

__device__ __forceinline__ int gated_mirror_idx(int phase, int slot,
                                                int parity, int src)
{
    return kGatedMirrorBase +
           ((phase * kGatedSlots + slot) * 2 + parity) * kMaxRanks + src;
}

/* Wait targets for this launch. Host-target mode reads them straight
 * from the (const) kernarg struct; graph mode derives them from this
 * block's launch ordinal within the (slot,parity) pattern: every gated
 * kernel of the pattern increments the same per-block counter, so with
 * pp kernels per iteration, counter value v maps to iteration
 * u = (v-1)/pp + 1 on every block of the same launch (fixed grid).
 * IMPORTANT: `a` stays const — mutating the by-value kernel argument
 * would force the whole ~660-byte struct into scratch memory and put
 * scratch reloads inside the hot loops (observed in gfx950 ISA). */
struct GatedTargets {
    uint64_t sw_reduce, sw_gather, prev_gather, stage, gather_wait;
    uint64_t sig_stage, sig_reduce, sig_gather;
};

__device__ __forceinline__ GatedTargets gated_targets(const GatedArgs &a)
{
    if (!a.derive) {
        return {a.t_sw_reduce,  a.t_sw_gather,  a.t_prev_gather,
                a.t_stage,      a.t_gather_wait,
                a.t_sig_stage,  a.t_sig_reduce, a.t_sig_gather};
    }
    uint64_t *cnt = a.local_flags + kGatedGraphBase +
                    ((uint64_t)a.slot * 2 + a.parity) * kGatedMaxBlocks +
                    blockIdx.x;
    __shared__ uint64_t s_u;
    if (threadIdx.x == 0) {
        uint64_t v = __hip_atomic_fetch_add(cnt, 1, __ATOMIC_RELAXED,
                                            __HIP_MEMORY_SCOPE_AGENT) +
                     1;
        s_u = (v - 1) / (uint64_t)a.pp + 1;
    }
    __syncthreads();
    const uint64_t u = s_u;
    const uint64_t B = (uint64_t)(a.nblocks ? a.nblocks : kGatedBlocks);
    return {a.has_reduce ? (u - 1) * B : 0,
            a.has_gather ? (u - 1) * B : 0,
            a.has_gather ? (u - 1) * B : 0, u * B, u * B,
            u * B, u * B, u * B};
}

/* all blocks wait until every rank reached counter[idx] >= target; on
 * spin timeout ALL threads leave together (no thread may return while
 * others sit at __syncthreads — that would hang the block forever).
 * PUSH protocol: ranks publish their phase counters into every peer's
 * mirror slots (gated_signal), so this wait polls LOCAL memory only —
 * remote xGMI reads during the wait would steal link bandwidth from the
 * concurrently-running data phases (B x n remote pollers; see
 * MI355X_MICROARCH.md "polling-cost"). */
__device__ __forceinline__ bool
gated_wait(const GatedArgs &a, int phase, uint64_t target)
{
    __shared__ int s_err;
    if (threadIdx.x == 0) {
        s_err = 0;
    }
    __syncthreads();
    if (target != 0 && threadIdx.x < 64) {
        int j = (int)threadIdx.x;
        if (j < a.nranks) {
            const uint64_t *f =
                a.pull_wait
                    ? a.peer_flags[j] +
                          gated_idx(phase, a.slot, a.parity)
                    : a.local_flags +
                          gated_mirror_idx(phase, a.slot, a.parity, j);
            uint64_t spins = 0;
            const uint64_t cap = spin_cap(a.spin_limit);
            while (sys_load(f) < target) {
                if (++spins > cap) {
                    sys_store(a.error_word, 1);
                    s_err = 1;
                    break;
                }
                __builtin_amdgcn_s_sleep(2);
            }
        }
    }
    __syncthreads();
    if (s_err) {
        return false;
    }
    __threadfence_system(); /* acquire: see peers' payload writes */
    return true;
}

/* per-block completion signal on my own counter; the LAST arriving block
 * of this launch pushes the new cumulative value into every rank's
 * mirror slot (n system-scope stores over xGMI — the only remote flag
 * traffic in the whole pipeline). acq_rel on the counter makes every
 * earlier block's release (and thus its payload writes, xGMI ones
 * included) visible-before the mirror store. */
__device__ __forceinline__ void gated_signal(const GatedArgs &a, int phase,
                                             uint64_t sig_target,
                                             bool pub_done = true)
{
    __threadfence_system();
    __syncthreads();
    if (threadIdx.x == 0) {
        uint64_t v = __hip_atomic_fetch_add(
                         a.local_flags +
                             gated_idx(phase, a.slot, a.parity),
                         1, __ATOMIC_ACQ_REL,
                         __HIP_MEMORY_SCOPE_SYSTEM) +
                     1;
        if (v == sig_target) { /* last arriver of this launch */
            /* guard against the ROCm 7.2 dropped-vmcnt hazard
             * (MI355X_MICROARCH.md "Compiler hazard"): the release
             * write-back must drain before the mirror stores leave */
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            for (int j = 0; j < a.nranks; j++) {
                __hip_atomic_store(
                    a.peer_flags[j] +
                        gated_mirror_idx(phase, a.slot, a.parity,
                                         a.rank),
                    v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_SYSTEM);
            }
            if (pub_done && a.done_host) {
                /* final kernel of a host-posted collective */
                __hip_atomic_store(a.done_host, a.done_seq,
                                   __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_SYSTEM);
            }
        }
    }
}

__global__ void k_staged_stage(const GatedArgs a)
{
    const GatedTargets t = gated_targets(a);
    if (!gated_wait(a, 1, t.sw_reduce) ||
        !gated_wait(a, 2, t.sw_gather)) { /* in-area reuse */
        return;
    }
    if (a.n_cells > 0) { /* per-dest cells (alltoall) */
        for (int c = 0; c < a.n_cells; c++) {
            if (a.c_len[c]) {
                d_copy_bytes((uint8_t *)a.my_in + a.c_dst_off[c],
                             (const uint8_t *)a.src + a.c_src_off[c],
                             a.c_len[c]);
            }
        }
    } else if (a.len) {
        d_copy_bytes((uint8_t *)a.my_in, (const uint8_t *)a.src, a.len);
    }
    gated_signal(a, 0, t.sig_stage);
}

template <typename T, int OP, int VEC, int U>
__global__ void k_staged_reduce(const GatedArgs a)
{
    using A = typename Cvt<T>::A;
    const GatedTargets gt = gated_targets(a);
    if (!gated_wait(a, 0, gt.stage) ||
        !gated_wait(a, 2, gt.prev_gather)) {
        return;
    }
    const uint64_t cnt = (a.sl_e - a.sl_b) / sizeof(T);
    const uint64_t tid = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
    const uint64_t str = (uint64_t)gridDim.x * blockDim.x;
    const uint64_t nv  = cnt / VEC;
    const int      n   = a.nranks;
    using P            = Pack<T, OP, VEC>;
    /* fully unrolled over kMaxRanks with a uniform guard: a runtime-
     * indexed pointer array would be materialized in scratch memory and
     * re-loaded inside the hot loop (observed in the gfx950 ISA);
     * unrolling keeps the 8 peer base pointers in SGPRs and issues all
     * peer loads back-to-back for xGMI latency overlap. */
    P *out = (P *)a.my_out;
    /* U-deep batching: all U x n peer loads are issued before any
     * reduction consumes them. The per-lane bytes-in-flight must cover
     * the bandwidth-latency product (≈6.3 TB/s x ~600 ns needs ~4 MB
     * chip-wide): U is chosen at launch so U*n*16B stays ~128B/lane
     * without spilling the pack registers (U=4 at n<=2, 2 at n<=4,
     * 1 at n>=5 where the peer unroll already provides the depth). */
    uint64_t i = tid;
    if (U > 1) {
        for (; i + (U - 1) * str < nv; i += U * str) {
            P x[U][kMaxRanks];
#pragma unroll
            for (int s = 0; s < kMaxRanks; s++) {
                if (s < n) {
                    const P *src =
                        (const P *)((const uint8_t *)a.peer_in[s] +
                                    a.sl_b);
#pragma unroll
                    for (int u = 0; u < U; u++) {
                        x[u][s] = src[i + (uint64_t)u * str];
                    }
                }
            }
            A r[U][VEC];
#pragma unroll
            for (int u = 0; u < U; u++) {
                vload<T, OP, VEC, A>(x[u][0], r[u]);
            }
#pragma unroll
            for (int s = 1; s < kMaxRanks; s++) {
                if (s < n) {
#pragma unroll
                    for (int u = 0; u < U; u++) {
                        vaccum<T, OP, VEC, A>(r[u], x[u][s]);
                    }
                }
            }
            P o[U];
#pragma unroll
            for (int u = 0; u < U; u++) {
                vstore<T, OP, VEC, A>(o[u], r[u], a.alpha);
            }
            if (a.zc_write) {
                /* write my reduced slice into every rank's dst */
#pragma unroll
                for (int s = 0; s < kMaxRanks; s++) {
                    if (s < n) {
#pragma unroll
                        for (int u = 0; u < U; u++) {
                            ((P *)a.peer_out[s])[i + (uint64_t)u * str] =
                                o[u];
                        }
                    }
                }
            } else {
#pragma unroll
                for (int u = 0; u < U; u++) {
                    out[i + (uint64_t)u * str] = o[u];
                }
            }
        }
    }
    for (; i < nv; i += str) {
        A r[VEC];
        {
            P acc = ((const P *)((const uint8_t *)a.peer_in[0] +
                                 a.sl_b))[i];
            vload<T, OP, VEC, A>(acc, r);
        }
#pragma unroll
        for (int s = 1; s < kMaxRanks; s++) {
            if (s < n) {
                P x = ((const P *)((const uint8_t *)a.peer_in[s] +
                                   a.sl_b))[i];
                vaccum<T, OP, VEC, A>(r, x);
            }
        }
        P o;
        vstore<T, OP, VEC, A>(o, r, a.alpha);
        if (a.zc_write) {
#pragma unroll
            for (int s = 0; s < kMaxRanks; s++) {
                if (s < n) {
                    ((P *)a.peer_out[s])[i] = o;
                }
            }
        } else {
            out[i] = o;
        }
    }
    for (uint64_t t = nv * VEC + tid; t < cnt; t += str) {
        A r = Cvt<T>::load(
            ((const T *)((const uint8_t *)a.peer_in[0] + a.sl_b))[t]);
#pragma unroll
        for (int s = 1; s < kMaxRanks; s++) {
            if (s < n) {
                r = red<A, OP>(
                    r, Cvt<T>::load(((const T *)((const uint8_t *)
                                                     a.peer_in[s] +
                                                 a.sl_b))[t]));
            }
        }
        T ov = Cvt<T>::store(apply_alpha<A>(r, a.alpha));
        if (a.zc_write) {
#pragma unroll
            for (int s = 0; s < kMaxRanks; s++) {
                if (s < n) {
                    ((T *)a.peer_out[s])[t] = ov;
                }
            }
        } else {
            ((T *)out)[t] = ov;
        }
    }
    gated_signal(a, 1, gt.sig_reduce);
}

__global__ void k_staged_gather(const GatedArgs a)
{
    const GatedTargets t = gated_targets(a);
    if (!gated_wait(a, a.gw_phase, t.gather_wait)) {
        return;
    }
    /* partition the grid across sources: every peer link streams
     * concurrently instead of one-after-another (7 x 153 GB/s at once) */
    const int grp = gridDim.x / a.nranks > 0 ? gridDim.x / a.nranks : 1;
    const int src = (int)blockIdx.x / grp;
    const int bid = (int)blockIdx.x % grp;
    if (src < a.nranks) {
        uint64_t b = a.slice_b[src], e = a.slice_e[src];
        if (e > b) {
            d_copy_bytes_part((uint8_t *)a.dst + b,
                              (const uint8_t *)a.peer_out[src], e - b,
                              (uint64_t)bid, (uint64_t)grp);
        }
    }
    gated_signal(a, 2, t.sig_gather);
}

static inline int gated_grid(const GatedArgs &a)
{
    int b = a.nblocks ? a.nblocks : kGatedBlocks;
    return b > kGatedMaxBlocks ? kGatedMaxBlocks : b;
}

/* Copy-engine collectives (reference alltoallv_ce.c:50,218-226 role):
 * data moves by hipMemcpyAsync on SDMA engines, kernels only gate.
 * k_gated_wait_only orders the memcpys after every rank's entry signal;
 * k_gated_done publishes my-copies-complete and waits the team. */
__global__ void k_gated_wait_only(const GatedArgs a)
{
    if (!gated_wait(a, a.gw_phase, a.t_gather_wait)) {
        return;
    }
    if (a.done_host && blockIdx.x == 0 && threadIdx.x == 0) {
        /* completion point: the team's final phase is confirmed, so
         * every peer has stopped reading this rank's buffers */
        __hip_atomic_store(a.done_host, a.done_seq, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_SYSTEM);
    }
}

__global__ void k_gated_done(const GatedArgs a)
{
    gated_signal(a, 2, a.t_sig_gather, /*pub_done=*/false);
    if (!gated_wait(a, 2, a.t_gather_wait)) {
        return;
    }
    if (a.done_host && blockIdx.x == 0 && threadIdx.x == 0) {
        __hip_atomic_store(a.done_host, a.done_seq, __ATOMIC_RELEASE,
                           __HIP_MEMORY_SCOPE_SYSTEM);
    }
}

ucc_status_t gated_wait_only(const GatedArgs &a, hipStream_t s)
{
    /* pure wait: one block is enough (it adds no signal counts) */
    hipLaunchKernelGGL(k_gated_wait_only, dim3(1), dim3(256), 0, s, a);
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

ucc_status_t gated_done(const GatedArgs &a, hipStream_t s)
{
    /* must launch the full gated grid: signal arithmetic is counts*B */
    hipLaunchKernelGGL(k_gated_done, dim3(gated_grid(a)), dim3(256),
                       0, s, a);
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

/* ----------------------------------------------------------- launchers */
static bool aligned16(const void *p) { return (((uintptr_t)p) & 15) == 0; }

template <typename T, int OP>
static ucc_status_t launch_reduce(const ReduceArgs &a, hipStream_t s)
{
    constexpr int VEC = VecOf<T>::value;
    bool          vec = aligned16(a.dst);
    for (int i = 0; i < a.n_srcs; i++) {
        vec = vec && aligned16(a.srcs[i]);
    }
    int  threads = 256;
    long work    = (long)(a.count / (vec ? VEC : 1));
    int  blocks  = (int)std::min<long>((work + threads - 1) / threads, 2048);
    if (blocks < 1) {
        blocks = 1;
    }
    if (vec) {
        hipLaunchKernelGGL((k_reduce<T, OP, VEC>), dim3(blocks),
                           dim3(threads), 0, s, a);
    } else {
        hipLaunchKernelGGL((k_reduce_scalar<T, OP>), dim3(blocks),
                           dim3(threads), 0, s, a);
    }
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

template <typename T, int OP>
static ucc_status_t launch_fused(const FusedArgs &a, hipStream_t s)
{
    constexpr int VEC = VecOf<T>::value;
    hipLaunchKernelGGL((k_fused_allreduce<T, OP, VEC>), dim3(a.nblocks),
                       dim3(256), 0, s, a);
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

template <typename T, int OP>
static ucc_status_t launch_fused_graph(const GraphFusedArgs &a,
                                       hipStream_t s)
{
    constexpr int VEC = VecOf<T>::value;
    hipLaunchKernelGGL((k_fused_allreduce_graph<T, OP, VEC>),
                       dim3(a.nblocks), dim3(256), 0, s, a);
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

#define UCC_DT_CASE_FLOAT(DT, T, LFN)                                        \
    case DT:                                                                 \
        switch (op) {                                                        \
        case UCC_OP_SUM: return LFN<T, 0>(a, s);                             \
        case UCC_OP_AVG: return LFN<T, 12>(a, s);                            \
        case UCC_OP_PROD: return LFN<T, 1>(a, s);                            \
        case UCC_OP_MAX: return LFN<T, 2>(a, s);                             \
        case UCC_OP_MIN: return LFN<T, 3>(a, s);                             \
        default: return UCC_ERR_NOT_SUPPORTED;                               \
        }

#define UCC_DT_CASE_INT(DT, T, LFN)                                          \
    case DT:                                                                 \
        switch (op) {                                                        \
        case UCC_OP_SUM: return LFN<T, 0>(a, s);                             \
        case UCC_OP_PROD: return LFN<T, 1>(a, s);                            \
        case UCC_OP_MAX: return LFN<T, 2>(a, s);                             \
        case UCC_OP_MIN: return LFN<T, 3>(a, s);                             \
        case UCC_OP_LAND: return LFN<T, 4>(a, s);                            \
        case UCC_OP_LOR: return LFN<T, 5>(a, s);                             \
        case UCC_OP_LXOR: return LFN<T, 6>(a, s);                            \
        case UCC_OP_BAND: return LFN<T, 7>(a, s);                            \
        case UCC_OP_BOR: return LFN<T, 8>(a, s);                             \
        case UCC_OP_BXOR: return LFN<T, 9>(a, s);                            \
        default: return UCC_ERR_NOT_SUPPORTED;                               \
        }

ucc_status_t reduce(const ReduceArgs &a_in, hipStream_t s)
{
    /* complex SUM/AVG == elementwise float SUM/AVG on 2x the elements */
    ReduceArgs a = a_in;
    if ((a.dt == UCC_DT_FLOAT32_COMPLEX || a.dt == UCC_DT_FLOAT64_COMPLEX) &&
        (a.op == UCC_OP_SUM || a.op == UCC_OP_AVG ||
         (int)a.op == 12)) {
        a.dt = a.dt == UCC_DT_FLOAT32_COMPLEX ? UCC_DT_FLOAT32
                                              : UCC_DT_FLOAT64;
        a.count *= 2;
    }
    ucc_reduction_op_t op = a.op;
    switch (a.dt) {
        UCC_DT_CASE_FLOAT(UCC_DT_BFLOAT16, bf16_t, launch_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT16, fp16_t, launch_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT32, float, launch_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT64, double, launch_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E4M3, e4m3_t, launch_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E5M2, e5m2_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT8, int8_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT8, uint8_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT16, int16_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT16, uint16_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT32, int32_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT32, uint32_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT64, int64_t, launch_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT64, uint64_t, launch_reduce)
    default: return UCC_ERR_NOT_SUPPORTED;
    }
}

ucc_status_t fused_allreduce(const FusedArgs &a_in, hipStream_t s)
{
    /* complex SUM/AVG == elementwise float SUM/AVG on 2x the elements */
    FusedArgs a = a_in;
    if ((a.dt == UCC_DT_FLOAT32_COMPLEX || a.dt == UCC_DT_FLOAT64_COMPLEX) &&
        (a.op == UCC_OP_SUM || a.op == UCC_OP_AVG ||
         (int)a.op == 12)) {
        a.dt = a.dt == UCC_DT_FLOAT32_COMPLEX ? UCC_DT_FLOAT32
                                              : UCC_DT_FLOAT64;
        a.count *= 2;
    }
    ucc_reduction_op_t op = a.op;
    switch (a.dt) {
        UCC_DT_CASE_FLOAT(UCC_DT_BFLOAT16, bf16_t, launch_fused)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT16, fp16_t, launch_fused)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT32, float, launch_fused)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT64, double, launch_fused)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E4M3, e4m3_t, launch_fused)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E5M2, e5m2_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_INT8, int8_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_UINT8, uint8_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_INT16, int16_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_UINT16, uint16_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_INT32, int32_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_UINT32, uint32_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_INT64, int64_t, launch_fused)
        UCC_DT_CASE_INT(UCC_DT_UINT64, uint64_t, launch_fused)
    default: return UCC_ERR_NOT_SUPPORTED;
    }
}

ucc_status_t fused_allreduce_graph(const GraphFusedArgs &a_in, hipStream_t s)
{
    /* complex SUM/AVG == elementwise float SUM/AVG on 2x the elements */
    GraphFusedArgs a = a_in;
    if ((a.dt == UCC_DT_FLOAT32_COMPLEX || a.dt == UCC_DT_FLOAT64_COMPLEX) &&
        (a.op == UCC_OP_SUM || a.op == UCC_OP_AVG ||
         (int)a.op == 12)) {
        a.dt = a.dt == UCC_DT_FLOAT32_COMPLEX ? UCC_DT_FLOAT32
                                              : UCC_DT_FLOAT64;
        a.count *= 2;
    }
    ucc_reduction_op_t op = a.op;
    switch (a.dt) {
        UCC_DT_CASE_FLOAT(UCC_DT_BFLOAT16, bf16_t, launch_fused_graph)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT16, fp16_t, launch_fused_graph)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT32, float, launch_fused_graph)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT64, double, launch_fused_graph)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E4M3, e4m3_t, launch_fused_graph)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E5M2, e5m2_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_INT8, int8_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_UINT8, uint8_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_INT16, int16_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_UINT16, uint16_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_INT32, int32_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_UINT32, uint32_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_INT64, int64_t, launch_fused_graph)
        UCC_DT_CASE_INT(UCC_DT_UINT64, uint64_t, launch_fused_graph)
    default: return UCC_ERR_NOT_SUPPORTED;
    }
}

ucc_status_t gather_copy(const GatherArgs &a, hipStream_t s)
{
    uint64_t total = 0;
    for (int i = 0; i < a.n; i++) {
        total += a.lens[i];
    }
    int threads = 256;
    int blocks =
        (int)std::min<uint64_t>((total / 16 + threads - 1) / threads, 2048);
    if (blocks < 1) {
        blocks = 1;
    }
    hipLaunchKernelGGL(k_gather_copy, dim3(blocks), dim3(threads), 0, s, a);
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

ucc_status_t staged_stage(const GatedArgs &a, hipStream_t s)
{
    hipLaunchKernelGGL(k_staged_stage, dim3(gated_grid(a)), dim3(256), 0,
                       s, a);
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

template <typename T, int OP>
static ucc_status_t launch_staged_reduce(const GatedArgs &a, hipStream_t s)
{
    constexpr int VEC = VecOf<T>::value;
    /* batching depth by rank count (bytes-in-flight per lane vs pack
     * register budget — see the kernel comment); UCC_EC_REDUCE_DEPTH
     * overrides for tuning (0 = auto) */
    static int ov = [] {
        const char *e = getenv("UCC_EC_REDUCE_DEPTH");
        return e ? atoi(e) : 0;
    }();
    static int thr = [] {
        const char *e = getenv("UCC_EC_REDUCE_THREADS");
        int         v = e ? atoi(e) : 256;
        return (v == 128 || v == 512 || v == 1024) ? v : 256;
    }();
    int depth = ov ? ov : (a.nranks <= 2 ? 4 : a.nranks <= 4 ? 2 : 1);
    /* the protocol counts BLOCKS, so the grid is fixed by the ledger;
     * threads-per-block is a free tuning axis (the wait/signal logic
     * only needs >=64 threads) */
    int blocks = gated_grid(a), threads = thr;
    switch (depth) {
    case 8:
        hipLaunchKernelGGL((k_staged_reduce<T, OP, VEC, 8>),
                           dim3(blocks), dim3(threads), 0, s, a);
        break;
    case 4:
        hipLaunchKernelGGL((k_staged_reduce<T, OP, VEC, 4>),
                           dim3(blocks), dim3(threads), 0, s, a);
        break;
    case 2:
        hipLaunchKernelGGL((k_staged_reduce<T, OP, VEC, 2>),
                           dim3(blocks), dim3(threads), 0, s, a);
        break;
    default:
        hipLaunchKernelGGL((k_staged_reduce<T, OP, VEC, 1>),
                           dim3(blocks), dim3(threads), 0, s, a);
    }
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

ucc_status_t staged_reduce(const GatedArgs &a_in, hipStream_t s)
{
    /* complex SUM/AVG == elementwise float SUM/AVG on 2x the elements */
    GatedArgs a = a_in;
    if ((a.dt == UCC_DT_FLOAT32_COMPLEX || a.dt == UCC_DT_FLOAT64_COMPLEX) &&
        (a.op == UCC_OP_SUM || a.op == UCC_OP_AVG ||
         (int)a.op == 12)) {
        a.dt = a.dt == UCC_DT_FLOAT32_COMPLEX ? UCC_DT_FLOAT32
                                              : UCC_DT_FLOAT64;
    }
    ucc_reduction_op_t op = a.op;
    switch (a.dt) {
        UCC_DT_CASE_FLOAT(UCC_DT_BFLOAT16, bf16_t, launch_staged_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT16, fp16_t, launch_staged_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT32, float, launch_staged_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT64, double, launch_staged_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E4M3, e4m3_t, launch_staged_reduce)
        UCC_DT_CASE_FLOAT(UCC_DT_FLOAT8_E5M2, e5m2_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT8, int8_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT8, uint8_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT16, int16_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT16, uint16_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT32, int32_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT32, uint32_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_INT64, int64_t, launch_staged_reduce)
        UCC_DT_CASE_INT(UCC_DT_UINT64, uint64_t, launch_staged_reduce)
    default: return UCC_ERR_NOT_SUPPORTED;
    }
}

ucc_status_t staged_gather(const GatedArgs &a, hipStream_t s)
{
    hipLaunchKernelGGL(k_staged_gather, dim3(gated_grid(a)), dim3(256), 0,
                       s, a);
    return hipGetLastError() == hipSuccess ? UCC_OK : UCC_ERR_NO_RESOURCE;
}

bool dt_supported(ucc_datatype_t dt)
{
    switch (dt) {
    case UCC_DT_BFLOAT16:
    case UCC_DT_FLOAT16:
    case UCC_DT_FLOAT32:
    case UCC_DT_FLOAT64:
    case UCC_DT_FLOAT8_E4M3:
    case UCC_DT_FLOAT8_E5M2:
    case UCC_DT_INT8:
    case UCC_DT_UINT8:
    case UCC_DT_INT16:
    case UCC_DT_UINT16:
    case UCC_DT_INT32:
    case UCC_DT_UINT32:
    case UCC_DT_INT64:
    case UCC_DT_UINT64: return true;
    default: return false;
    }
}

bool op_supported(ucc_datatype_t dt, ucc_reduction_op_t op)
{
    if (dt == UCC_DT_FLOAT32_COMPLEX || dt == UCC_DT_FLOAT64_COMPLEX) {
        return op == UCC_OP_SUM || op == UCC_OP_AVG;
    }
    bool is_float = dt == UCC_DT_BFLOAT16 || dt == UCC_DT_FLOAT16 ||
                    dt == UCC_DT_FLOAT32 || dt == UCC_DT_FLOAT64 ||
                    dt == UCC_DT_FLOAT8_E4M3 || dt == UCC_DT_FLOAT8_E5M2;
    switch (op) {
    case UCC_OP_SUM:
    case UCC_OP_PROD:
    case UCC_OP_MAX:
    case UCC_OP_MIN: return dt_supported(dt);
    case UCC_OP_AVG: return is_float;
    case UCC_OP_LAND:
    case UCC_OP_LOR:
    case UCC_OP_LXOR:
    case UCC_OP_BAND:
    case UCC_OP_BOR:
    case UCC_OP_BXOR: return dt_supported(dt) && !is_float;
    default: return false;
    }
}

} // namespace ec_hip
} // namespace ucc
