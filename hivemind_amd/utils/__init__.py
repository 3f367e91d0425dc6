from .asyncio_utils import (
    EventLoopThread,
    achain,
    aenumerate,
    afirst,
    aiter_with_timeout,
    amap_in_executor,
    as_aiter,
    asingle,
    attach_event_on_finished,
    azip,
    cancel_and_wait,
    enter_asynchronously,
    switch_to_uvloop,
)
from .crypto import PrivateKey, PublicKey
from .logging import get_logger, use_hivemind_log_handler
from .math_utils import get_flatten_greedy_dims, orthogonalize_
from .nested import nested_compare, nested_flatten, nested_map, nested_pack
from .networking import Endpoint, LOCALHOST, choose_ip_address, get_free_port, increase_file_limit, make_endpoint, split_endpoint
from .performance_ema import PerformanceEMA
from .serializer import MSGPackSerializer, SerializerBase
from .tensor_descr import BatchTensorDescriptor, DescriptorBase, TensorDescriptor
from .timed_storage import (
    DHTExpiration,
    MAX_DHT_TIME_DISCREPANCY_SECONDS,
    TimedStorage,
    ValueWithExpiration,
    get_dht_time,
)
