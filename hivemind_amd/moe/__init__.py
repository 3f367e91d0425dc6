from .client import MoEBeamSearcher, RemoteExpert, RemoteMixtureOfExperts, RemoteSwitchMixtureOfExperts
from .expert_uid import ExpertInfo, is_valid_prefix, is_valid_uid, split_uid
from .server import ModuleBackend, Server, background_server, declare_experts, get_experts
from .server.layers import register_expert_class
