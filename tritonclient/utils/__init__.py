from client_amd.utils import *  # noqa: F401,F403
from client_amd.utils import (  # noqa: F401
    InferenceServerException, np_to_triton_dtype, triton_to_np_dtype,
    serialize_byte_tensor, deserialize_bytes_tensor, serialize_bf16_tensor,
    deserialize_bf16_tensor, serialized_byte_size, raise_error,
)
