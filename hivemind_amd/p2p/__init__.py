from .peer_id import PeerID, PeerInfo, b58decode, b58encode
from .servicer import RpcMessage, ServicerBase, StubBase
from .transport import P2P, P2PDaemonError, P2PHandlerError, RpcContext, STREAM_CHUNK_SIZE
