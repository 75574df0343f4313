"""System (POSIX) shared-memory utilities.

API-compatible with ``tritonclient.utils.shared_memory`` (reference:
tritonclient/utils/shared_memory/__init__.py): create/set/get/destroy
plus the global key -> (refcount, needs_unlink) registry that makes
repeated create/destroy of the same key safe (reference :36, :78-112,
:225-251).
"""

import mmap
import os

import numpy as np

from .. import serialize_byte_tensor

_key_mapping = {}


class SharedMemoryException(Exception):
    def __init__(self, err):
        self.err_ = err
        super().__init__(str(err))

    def __str__(self):
        return self.err_ if isinstance(self.err_, str) else str(self.err_)


class SharedMemoryRegion:
    def __init__(self, triton_shm_name, shm_key):
        self._triton_shm_name = triton_shm_name
        self._shm_key = shm_key
        self._mpsm_handle = None
        self._byte_size = 0
        self._fd = -1


def _shm_path(shm_key):
    return "/dev/shm/" + shm_key.lstrip("/")


def create_shared_memory_region(triton_shm_name, shm_key, byte_size,
                                create_only=False):
    """Create (or open) a POSIX shm region; returns the region handle."""
    if byte_size <= 0:
        raise SharedMemoryException(
            "unable to create the shared memory region: byte_size must be "
            "positive"
        )
    shm_handle = SharedMemoryRegion(triton_shm_name, shm_key)
    path = _shm_path(shm_key)
    exists = os.path.exists(path)
    if create_only and exists:
        raise SharedMemoryException(
            f"unable to create the shared memory region '{shm_key}', "
            "already exists"
        )
    try:
        fd = os.open(path, os.O_RDWR | os.O_CREAT, 0o600)
    except OSError as e:
        raise SharedMemoryException(
            f"unable to create the shared memory region '{shm_key}': {e}"
        )
    try:
        cur = os.fstat(fd).st_size
        if cur < byte_size:
            os.ftruncate(fd, byte_size)
        shm_handle._mpsm_handle = mmap.mmap(fd, byte_size)
    except (OSError, ValueError) as e:
        os.close(fd)
        raise SharedMemoryException(
            f"unable to initialize the size of shared memory region "
            f"'{shm_key}': {e}"
        )
    shm_handle._fd = fd
    shm_handle._byte_size = byte_size
    if shm_key in _key_mapping:
        _key_mapping[shm_key]["refcount"] += 1
        _key_mapping[shm_key]["needs_unlink"] |= not exists
    else:
        _key_mapping[shm_key] = {"refcount": 1, "needs_unlink": not exists}
    return shm_handle


def set_shared_memory_region(shm_handle, input_values, offset=0):
    """Copy numpy tensors into the region starting at ``offset``
    (reference :115-163); BYTES tensors are serialized first."""
    if not isinstance(input_values, (list, tuple)):
        raise SharedMemoryException(
            "input_values must be specified as a list/tuple of numpy arrays"
        )
    offset_current = offset
    mem = shm_handle._mpsm_handle
    for input_value in input_values:
        input_value = np.ascontiguousarray(input_value)
        if input_value.dtype == np.object_:
            byte_data = serialize_byte_tensor(input_value).item()
        else:
            byte_data = input_value.tobytes()
        if offset_current + len(byte_data) > shm_handle._byte_size:
            raise SharedMemoryException(
                "unable to set the shared memory region: tensors exceed "
                f"region size ({shm_handle._byte_size} bytes)"
            )
        mem[offset_current : offset_current + len(byte_data)] = byte_data
        offset_current += len(byte_data)


def get_contents_as_numpy(shm_handle, datatype, shape, offset=0):
    """View region contents as a numpy tensor (reference :166-210)."""
    from .. import deserialize_bytes_tensor

    mem = shm_handle._mpsm_handle
    if datatype == np.object_ or datatype == bytes:
        raw = bytes(mem[offset : shm_handle._byte_size])
        result = deserialize_bytes_tensor(raw)[: int(np.prod(shape))]
        return result.reshape(shape)
    dt = np.dtype(datatype)
    count = int(np.prod(shape)) if shape else 1
    nbytes = count * dt.itemsize
    arr = np.frombuffer(mem, dtype=dt, count=count, offset=offset)
    return arr.reshape(shape)


def mapped_shared_memory_regions():
    """List of shm keys currently mapped by this process (reference :213-222)."""
    return list(_key_mapping.keys())


def destroy_shared_memory_region(shm_handle):
    """Unmap; unlink the file once the last local reference is destroyed
    and this process created it (reference :225-251)."""
    key = shm_handle._shm_key
    if key not in _key_mapping:
        raise SharedMemoryException(
            f"unable to destroy shared memory region '{key}': not mapped"
        )
    try:
        shm_handle._mpsm_handle.close()
    except Exception:
        pass
    if shm_handle._fd >= 0:
        try:
            os.close(shm_handle._fd)
        except OSError:
            pass
        shm_handle._fd = -1
    _key_mapping[key]["refcount"] -= 1
    if _key_mapping[key]["refcount"] == 0:
        needs_unlink = _key_mapping[key]["needs_unlink"]
        del _key_mapping[key]
        if needs_unlink:
            try:
                os.unlink(_shm_path(key))
            except FileNotFoundError:
                pass
