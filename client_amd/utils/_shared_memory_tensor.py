"""Zero-copy DLPack export of shared-memory regions (reference:
tritonclient/utils/_shared_memory_tensor.py:34-88), with kDLROCM for
HIP-shm regions so torch-ROCm wraps them without any copy."""

import ctypes

from ._dlpack import (
    DLDevice,
    DLManagedTensor,
    DLManagedTensorDeleter,
    DLTensor,
    _c_str_dltensor,
    triton_to_dlpack_dtype,
)


# The DLManagedTensor struct, its shape array and the ctypes deleter
# trampoline must outlive the consumer's tensor: the consumer (e.g.
# torch) keeps the DLManagedTensor* and invokes ->deleter when ITS
# tensor dies, which can be long after the SharedMemoryTensor is gone.
# Entries are keyed by the struct address and removed by the deleter.
_EXPORT_KEEPALIVE = {}


class SharedMemoryTensor:
    def __init__(self, datatype, shape, base_addr, byte_offset, device_type,
                 device_id):
        self._datatype = datatype
        self._shape = shape
        self._base_addr = base_addr
        self._byte_offset = byte_offset
        self._device_type = device_type
        self._device_id = device_id

    def __dlpack__(self, stream=None):
        dl_dtype = triton_to_dlpack_dtype(self._datatype)
        ndim = len(self._shape)
        shape_arr = (ctypes.c_int64 * ndim)(*self._shape)
        managed = DLManagedTensor()
        managed.dl_tensor = DLTensor(
            data=ctypes.c_void_p(self._base_addr),
            device=DLDevice(self._device_type, self._device_id),
            ndim=ndim,
            dtype=dl_dtype,
            shape=ctypes.cast(shape_arr, ctypes.POINTER(ctypes.c_int64)),
            strides=None,
            byte_offset=self._byte_offset,
        )
        managed.manager_ctx = None
        box = ctypes.pointer(managed)
        key = ctypes.addressof(managed)

        def _deleter(handle, _key=key):
            _EXPORT_KEEPALIVE.pop(_key, None)

        deleter = DLManagedTensorDeleter(_deleter)
        managed.deleter = deleter
        _EXPORT_KEEPALIVE[key] = (managed, shape_arr, deleter, box)
        capsule = ctypes.pythonapi.PyCapsule_New(
            ctypes.cast(box, ctypes.c_void_p), _c_str_dltensor, None
        )
        return capsule

    def __dlpack_device__(self):
        return (self._device_type, self._device_id)
