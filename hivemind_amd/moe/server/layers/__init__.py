from .common import FeedforwardBlock, NopExpert, TransformerEncoderLayer, DeterministicDropout, DeterministicDropoutNetwork
from .custom_experts import name_to_block, name_to_input, register_expert_class
from .lr_schedule import get_linear_schedule_with_warmup
from .optim import ClippingWrapper, OptimizerWrapper
