from .adaptive import PerTensorCompression, RoleAdaptiveCompression, SizeAdaptiveCompression
from .base import CompressionBase, CompressionInfo, CompressionType, NoCompression, TensorRole, WireTensor
from .floating import Float16Compression, ScaledFloat16Compression
from .quantization import BlockwiseQuantization, Quantile8BitQuantization, Uniform8BitQuantization
from .serialization import (
    BASE_COMPRESSION_TYPES,
    combine_from_streaming,
    deserialize_tensor_stream,
    deserialize_torch_tensor,
    serialize_torch_tensor,
    split_for_streaming,
)
