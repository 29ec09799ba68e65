"""Datatype / op / memory-type ids matching src/api/ucc.h enums."""

INT8 = 0
UINT8 = 1
INT16 = 2
UINT16 = 3
INT32 = 4
UINT32 = 5
INT64 = 6
UINT64 = 7
INT128 = 8
UINT128 = 9
FLOAT16 = 10
BFLOAT16 = 11
FLOAT32 = 12
FLOAT64 = 13
FLOAT128 = 14
FLOAT32_COMPLEX = 15
FLOAT64_COMPLEX = 16
FLOAT128_COMPLEX = 17
FLOAT8_E4M3 = 18
FLOAT8_E5M2 = 19

OP_SUM = 0
OP_PROD = 1
OP_MAX = 2
OP_MIN = 3
OP_LAND = 4
OP_LOR = 5
OP_LXOR = 6
OP_BAND = 7
OP_BOR = 8
OP_BXOR = 9
OP_MAXLOC = 10
OP_MINLOC = 11
OP_AVG = 12

MEM_HOST = 0
MEM_CUDA = 1  # device memory (HIP)

DT_SIZE = {
    INT8: 1, UINT8: 1, INT16: 2, UINT16: 2, INT32: 4, UINT32: 4,
    INT64: 8, UINT64: 8, INT128: 16, UINT128: 16, FLOAT16: 2,
    BFLOAT16: 2, FLOAT32: 4, FLOAT64: 8, FLOAT128: 16,
    FLOAT32_COMPLEX: 8, FLOAT64_COMPLEX: 16, FLOAT128_COMPLEX: 32,
    FLOAT8_E4M3: 1, FLOAT8_E5M2: 1,
}


def from_numpy(np_dtype):
    import numpy as np

    m = {
        np.dtype(np.int8): INT8, np.dtype(np.uint8): UINT8,
        np.dtype(np.int16): INT16, np.dtype(np.uint16): UINT16,
        np.dtype(np.int32): INT32, np.dtype(np.uint32): UINT32,
        np.dtype(np.int64): INT64, np.dtype(np.uint64): UINT64,
        np.dtype(np.float16): FLOAT16, np.dtype(np.float32): FLOAT32,
        np.dtype(np.float64): FLOAT64,
        np.dtype(np.complex64): FLOAT32_COMPLEX,
        np.dtype(np.complex128): FLOAT64_COMPLEX,
    }
    return m[np.dtype(np_dtype)]


def from_torch(t_dtype):
    import torch

    m = {
        torch.int8: INT8, torch.uint8: UINT8, torch.int16: INT16,
        torch.int32: INT32, torch.int64: INT64, torch.float16: FLOAT16,
        torch.bfloat16: BFLOAT16, torch.float32: FLOAT32,
        torch.float64: FLOAT64,
    }
    if hasattr(torch, "float8_e4m3fn"):
        m[torch.float8_e4m3fn] = FLOAT8_E4M3
        m[torch.float8_e5m2] = FLOAT8_E5M2
    return m[t_dtype]
