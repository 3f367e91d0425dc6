"""hivemind_amd: an MI355X-native decentralized deep learning framework with the
capabilities and public API surface of learning-at-home/hivemind.

Control plane (DHT, matchmaking, progress tracking, RPC) runs on CPU over
asyncio TCP; the tensor data plane is RCCL over xGMI + hand-written CDNA4 HIP
kernels (hivemind_amd.ops). See SURVEY.md for the full design map.
"""

from .averaging import DecentralizedAverager
from .compression import (
    BlockwiseQuantization,
    CompressionBase,
    CompressionType,
    Float16Compression,
    NoCompression,
    Quantile8BitQuantization,
    ScaledFloat16Compression,
    SizeAdaptiveCompression,
    Uniform8BitQuantization,
)
from .dht import DHT
from .moe import (
    ModuleBackend,
    RemoteExpert,
    RemoteMixtureOfExperts,
    RemoteSwitchMixtureOfExperts,
    Server,
    register_expert_class,
)
from .optim import GradScaler, GradientAverager, Optimizer, ProgressTracker, TrainingAverager, TrainingStateAverager
from .p2p import P2P, PeerID, PeerInfo
from .utils import get_dht_time, get_logger, use_hivemind_log_handler

__version__ = "0.1.0"
