"""ctypes DLPack ABI (vendor-neutral; reference:
tritonclient/utils/_dlpack.py:57-115).

Used to (a) consume tensors from any framework into the HIP data plane
and (b) export HIP-shm regions as zero-copy tensors (kDLROCM device).
"""

import ctypes

DLPACK_VERSION = (0, 8)


class DLDeviceType:
    kDLCPU = 1
    kDLCUDA = 2
    kDLCUDAHost = 3
    kDLOpenCL = 4
    kDLVulkan = 7
    kDLMetal = 8
    kDLVPI = 9
    kDLROCM = 10
    kDLROCMHost = 11
    kDLCUDAManaged = 13
    kDLOneAPI = 14


class DLDevice(ctypes.Structure):
    _fields_ = [
        ("device_type", ctypes.c_int),
        ("device_id", ctypes.c_int),
    ]


class DLDataTypeCode:
    kDLInt = 0
    kDLUInt = 1
    kDLFloat = 2
    kDLOpaqueHandle = 3
    kDLBfloat = 4
    kDLComplex = 5
    kDLBool = 6


class DLDataType(ctypes.Structure):
    _fields_ = [
        ("type_code", ctypes.c_uint8),
        ("bits", ctypes.c_uint8),
        ("lanes", ctypes.c_uint16),
    ]


class DLTensor(ctypes.Structure):
    _fields_ = [
        ("data", ctypes.c_void_p),
        ("device", DLDevice),
        ("ndim", ctypes.c_int),
        ("dtype", DLDataType),
        ("shape", ctypes.POINTER(ctypes.c_int64)),
        ("strides", ctypes.POINTER(ctypes.c_int64)),
        ("byte_offset", ctypes.c_uint64),
    ]


class DLManagedTensor(ctypes.Structure):
    pass


DLManagedTensorDeleter = ctypes.CFUNCTYPE(None, ctypes.POINTER(DLManagedTensor))

DLManagedTensor._fields_ = [
    ("dl_tensor", DLTensor),
    ("manager_ctx", ctypes.c_void_p),
    ("deleter", DLManagedTensorDeleter),
]

_c_str_dltensor = b"dltensor"
_c_str_used_dltensor = b"used_dltensor"

ctypes.pythonapi.PyCapsule_New.restype = ctypes.py_object
ctypes.pythonapi.PyCapsule_New.argtypes = [
    ctypes.c_void_p,
    ctypes.c_char_p,
    ctypes.c_void_p,
]
ctypes.pythonapi.PyCapsule_GetPointer.restype = ctypes.c_void_p
ctypes.pythonapi.PyCapsule_GetPointer.argtypes = [
    ctypes.py_object,
    ctypes.c_char_p,
]
ctypes.pythonapi.PyCapsule_IsValid.restype = ctypes.c_int
ctypes.pythonapi.PyCapsule_IsValid.argtypes = [ctypes.py_object, ctypes.c_char_p]
ctypes.pythonapi.PyCapsule_SetName.restype = ctypes.c_int
ctypes.pythonapi.PyCapsule_SetName.argtypes = [ctypes.py_object, ctypes.c_char_p]


# Triton datatype string -> DLDataType
_TRITON_TO_DLPACK = {
    "BOOL": (DLDataTypeCode.kDLBool, 8),
    "INT8": (DLDataTypeCode.kDLInt, 8),
    "INT16": (DLDataTypeCode.kDLInt, 16),
    "INT32": (DLDataTypeCode.kDLInt, 32),
    "INT64": (DLDataTypeCode.kDLInt, 64),
    "UINT8": (DLDataTypeCode.kDLUInt, 8),
    "UINT16": (DLDataTypeCode.kDLUInt, 16),
    "UINT32": (DLDataTypeCode.kDLUInt, 32),
    "UINT64": (DLDataTypeCode.kDLUInt, 64),
    "FP16": (DLDataTypeCode.kDLFloat, 16),
    "FP32": (DLDataTypeCode.kDLFloat, 32),
    "FP64": (DLDataTypeCode.kDLFloat, 64),
    "BF16": (DLDataTypeCode.kDLBfloat, 16),
}


def triton_to_dlpack_dtype(dtype):
    entry = _TRITON_TO_DLPACK.get(dtype)
    if entry is None:
        raise ValueError(f"unsupported datatype for DLPack: {dtype}")
    code, bits = entry
    return DLDataType(type_code=code, bits=bits, lanes=1)


def is_contiguous_data(ndim, shape, strides):
    """True if strides describe C-contiguous layout (strides may be NULL)."""
    if not strides:
        return True
    expected = 1
    for i in reversed(range(ndim)):
        if shape[i] != 1 and strides[i] != expected:
            return False
        expected *= shape[i]
    return True


def get_byte_size(dtype, shape):
    n = 1
    for d in shape:
        n *= d
    return n * dtype.bits * dtype.lanes // 8


def get_dlpack_capsule(dlpack_obj, stream=None):
    """Consume an object's __dlpack__ and return the capsule."""
    if hasattr(dlpack_obj, "__dlpack__"):
        try:
            return dlpack_obj.__dlpack__(stream=stream)
        except TypeError:
            return dlpack_obj.__dlpack__()
    raise ValueError("object does not support the DLPack protocol")


def get_managed_tensor(capsule):
    """Capsule -> DLManagedTensor struct view."""
    ptr = ctypes.pythonapi.PyCapsule_GetPointer(capsule, _c_str_dltensor)
    return ctypes.cast(ptr, ctypes.POINTER(DLManagedTensor)).contents
