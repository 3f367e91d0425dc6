from .checkpoints import CheckpointSaver, load_experts, store_experts
from .connection_handler import ConnectionHandler, ExpertRequest, ExpertResponse
from .dht_handler import DHTHandlerThread, declare_experts, get_expert_infos, get_experts
from .module_backend import ModuleBackend
from .runtime import Runtime
from .server import Server, background_server
from .task_pool import Task, TaskPool
