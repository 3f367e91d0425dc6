from .beam_search import MoEBeamSearcher
from .expert import RemoteExpert, create_remote_experts
from .moe import RemoteMixtureOfExperts
from .remote_expert_worker import RemoteExpertWorker
from .switch_moe import RemoteSwitchMixtureOfExperts
