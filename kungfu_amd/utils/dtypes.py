"""dtype/op code mappings between torch/numpy and the native extensions."""
import numpy as np

# must match kf::DType in csrc/core/common.hpp
_CORE_DTYPES = {
    "uint8": 0,
    "int8": 1,
    "int16": 2,
    "int32": 3,
    "int64": 4,
    "uint16": 5,
    "uint32": 6,
    "uint64": 7,
    "float16": 8,
    "bfloat16": 9,
    "float32": 10,
    "float64": 11,
}

# must match kf::ReduceOp
_CORE_OPS = {"sum": 0, "min": 1, "max": 2, "prod": 3}

# must match csrc/hip dtype codes: 0 f32, 1 bf16, 2 f16
_HIP_DTYPES = {"float32": 0, "bfloat16": 1, "float16": 2}


def _dtype_name(t):
    if isinstance(t, str):
        return t
    if isinstance(t, np.dtype):
        return t.name
    # torch dtype
    s = str(t)
    return s.split(".")[-1]


def core_dtype(t):
    name = _dtype_name(t)
    if name not in _CORE_DTYPES:
        raise TypeError("unsupported dtype for kungfu core: %s" % name)
    return _CORE_DTYPES[name]


def core_op(op):
    if op not in _CORE_OPS:
        raise ValueError("unsupported reduce op: %s" % op)
    return _CORE_OPS[op]


def hip_dtype(t):
    name = _dtype_name(t)
    if name not in _HIP_DTYPES:
        raise TypeError("unsupported dtype for kungfu HIP kernels: %s" % name)
    return _HIP_DTYPES[name]
