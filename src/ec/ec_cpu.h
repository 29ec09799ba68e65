/* Host execution component: n-source reduction + copy for every predefined
 * dtype x op, with the AVG alpha-scale hook. Parity: reference
 * components/ec/cpu/ec_cpu_reduce.c (incl. software bf16) — re-written as
 * C++ templates. */
#ifndef UCC_AMD_EC_CPU_H_
#define UCC_AMD_EC_CPU_H_

#include <cstddef>
#include "../api/ucc.h"

namespace ucc {
namespace ec_cpu {

/* dst[i] = alpha * op(srcs[0][i], ..., srcs[n_srcs-1][i]) */
ucc_status_t reduce(void *dst, const void *const *srcs, int n_srcs,
                    size_t count, ucc_datatype_t dt, ucc_reduction_op_t op,
                    double alpha = 1.0);

/* Strided variant: src1 + k*stride for k in [0, n_src2); used by
 * reduce-scatter style loops. */
ucc_status_t reduce_strided(void *dst, const void *src1, const void *src2,
                            size_t stride_bytes, int n_src2, size_t count,
                            ucc_datatype_t dt, ucc_reduction_op_t op,
                            double alpha = 1.0);

/* host<->host copy */
void copy(void *dst, const void *src, size_t bytes);

/* scalar conversion helpers shared with tests */
float    bf16_to_float(uint16_t v);
uint16_t float_to_bf16(float f);
float    fp16_to_float(uint16_t v);
uint16_t float_to_fp16(float f);
float    fp8e4m3_to_float(uint8_t v);
uint8_t  float_to_fp8e4m3(float f);
float    fp8e5m2_to_float(uint8_t v);
uint8_t  float_to_fp8e5m2(float f);

} // namespace ec_cpu
} // namespace ucc

#endif
