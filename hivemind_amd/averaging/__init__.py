from .allreduce import AllReduceRunner, AveragingMode
from .averager import AveragingError, DecentralizedAverager, compute_schema_hash
from .control import AveragingStage, StepControl
from .group_info import GroupInfo
from .key_manager import GroupKeyManager
from .load_balancing import load_balance_peers
from .matchmaking import Matchmaking, MatchmakingException
from .partition import TensorPartContainer, TensorPartReducer
from .rccl import DistributedAllReduceRunner
