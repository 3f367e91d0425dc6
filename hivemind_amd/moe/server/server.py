"""MoE expert server.

Parity target: reference ``hivemind/moe/server/server.py:35-411``:
``Server.create`` generates expert UIDs from a pattern (``"ffn.[0:256]"``),
builds layers/optimizers/schedulers, and runs DHT declaration + RPC handlers +
the batching Runtime; ``background_server`` is the context-manager variant.
The reference's N forked ConnectionHandler processes collapse into asyncio
handlers on the DHT loop -- the GPU data path is one process end-to-end.
"""

from __future__ import annotations

import asyncio
import random
import re
import threading
from contextlib import contextmanager
from pathlib import Path
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ...dht import DHT
from ...utils.logging import get_logger
from ...utils.tensor_descr import BatchTensorDescriptor
from ..expert_uid import UID_DELIMITER, is_valid_uid
from .checkpoints import CheckpointSaver, load_experts
from .connection_handler import ConnectionHandler
from .dht_handler import DHTHandlerThread, declare_experts, get_expert_infos
from .layers import get_linear_schedule_with_warmup, name_to_block, name_to_input
from .module_backend import ModuleBackend
from .runtime import Runtime

logger = get_logger(__name__)


class Server(threading.Thread):
    def __init__(
        self,
        dht: DHT,
        module_backends: Dict[str, ModuleBackend],
        *,
        device: Optional[torch.device] = None,
        num_connection_handlers: int = 4,
        update_period: float = 30.0,
        expiration: Optional[float] = None,
        checkpoint_dir: Optional[Path] = None,
        stats_report_interval: Optional[float] = None,
        relay_endpoint: Optional[str] = None,
        start: bool = False,
    ):
        super().__init__(name="moe-server", daemon=True)
        self.dht, self.module_backends = dht, module_backends
        self.device = device
        self.relay_endpoint = relay_endpoint
        self.num_connection_handlers = max(1, num_connection_handlers)
        self.conn_handler = ConnectionHandler(module_backends)
        self.runtime = Runtime(module_backends, device=device, stats_report_interval=stats_report_interval)
        self._handler_loops: list = []
        self._handler_p2ps: list = []
        self.checkpoint_saver = (
            CheckpointSaver(module_backends, checkpoint_dir, update_period) if checkpoint_dir is not None else None
        )
        self._update_period, self._expiration = update_period, expiration
        self.dht_handler_thread: Optional[DHTHandlerThread] = None
        self.ready = threading.Event()
        self._stop_requested = threading.Event()
        if start:
            self.run_in_background(await_ready=True)

    def run(self):
        """Start all components and serve until shutdown (reference server.py:231-263).

        Experts are sharded across ``num_connection_handlers`` listener loops,
        each its own thread + P2P endpoint: tensor (de)serialization for
        different experts runs in parallel (the reference achieves the same
        with N forked ConnectionHandler processes + the p2pd ``balanced`` flag).
        """
        from ...p2p import P2P
        from ...utils.asyncio_utils import EventLoopThread

        # handler 0 lives on the DHT loop (keeps single-loop deployments simple)
        p2p0 = self.dht.replicate_p2p()
        asyncio.run_coroutine_threadsafe(self.conn_handler.add_handlers(p2p0), self.dht.loop).result(15)
        self._handler_p2ps.append(p2p0)
        handler_loops = [self.dht.loop]
        for i in range(1, self.num_connection_handlers):
            loop_thread = EventLoopThread(name=f"moe-handler-{i}")
            loop_thread.start_and_wait()
            # behind NAT (relay_endpoint set): no inbound socket; each handler
            # loop registers with the relay and advertises a relay:// endpoint
            p2p_i = asyncio.run_coroutine_threadsafe(
                P2P.create(listen=self.relay_endpoint is None, relay_endpoint=self.relay_endpoint),
                loop_thread.loop,
            ).result(15)
            asyncio.run_coroutine_threadsafe(self.conn_handler.add_handlers(p2p_i), loop_thread.loop).result(15)
            self._handler_loops.append(loop_thread)
            self._handler_p2ps.append(p2p_i)
            handler_loops.append(loop_thread.loop)

        uid_to_peer = {}
        for idx, uid in enumerate(sorted(self.module_backends.keys())):
            p2p_i = self._handler_p2ps[idx % len(self._handler_p2ps)]
            uid_to_peer[uid] = (p2p_i.peer_id.to_base58(), p2p_i.endpoint)
        self.dht_handler_thread = DHTHandlerThread(
            self.module_backends, self.dht, self._update_period, self._expiration, uid_to_peer=uid_to_peer
        )
        self.dht_handler_thread.start()
        if self.checkpoint_saver is not None:
            self.checkpoint_saver.start()
        self.runtime.start()
        self.runtime.ready.wait()
        self.ready.set()
        self._stop_requested.wait()

    def run_in_background(self, await_ready: bool = True, timeout: Optional[float] = 30.0):
        self.start()
        if await_ready and not self.ready.wait(timeout):
            raise TimeoutError("server didn't start within timeout")

    def shutdown(self):
        self.ready.clear()
        self._stop_requested.set()
        try:
            self.conn_handler.remove_p2p_handlers(self.dht.replicate_p2p())
        except Exception:
            pass
        for loop_thread in self._handler_loops:
            try:
                loop_thread.shutdown()
            except Exception:
                pass
        if self.dht_handler_thread is not None:
            self.dht_handler_thread.shutdown()
        if self.checkpoint_saver is not None:
            self.checkpoint_saver.shutdown()
        self.runtime.shutdown()

    @classmethod
    def create(
        cls,
        *,
        dht: Optional[DHT] = None,
        initial_peers: Sequence[str] = (),
        expert_uids: Optional[Sequence[str]] = None,
        expert_pattern: Optional[str] = None,
        num_experts: Optional[int] = None,
        expert_cls: str = "ffn",
        hidden_dim: int = 1024,
        optim_cls=torch.optim.Adam,
        scheduler: Optional[str] = None,
        num_warmup_steps: Optional[int] = None,
        num_total_steps: Optional[int] = None,
        clip_grad_norm: Optional[float] = None,
        min_batch_size: int = 1,
        max_batch_size: int = 16384,
        num_connection_handlers: int = 4,
        device: Optional[str] = None,
        checkpoint_dir: Optional[Path] = None,
        load_experts_from_dir: bool = False,
        stats_report_interval: Optional[float] = None,
        update_period: float = 30.0,
        expiration: Optional[float] = None,
        relay_endpoint: Optional[str] = None,
        start: bool = False,
        **kwargs,
    ) -> "Server":
        """Build a server hosting `num_experts` experts (reference server.py:88-229)."""
        if dht is None:
            dht = DHT(initial_peers=list(initial_peers), start=True)
        assert expert_cls in name_to_block, f"unknown expert class {expert_cls}"

        if expert_uids is None:
            assert num_experts is not None and expert_pattern is not None, (
                "supply either expert_uids or (num_experts + expert_pattern)"
            )
            expert_uids = _generate_uids(num_experts, expert_pattern, dht)

        device = device if device is not None else ("cuda" if torch.cuda.is_available() else "cpu")
        sample_input = name_to_input[expert_cls](4, hidden_dim)
        if isinstance(sample_input, tuple):
            args_schema = tuple(BatchTensorDescriptor.from_tensor(arg) for arg in sample_input)
        else:
            args_schema = (BatchTensorDescriptor.from_tensor(sample_input),)

        module_backends = {}
        for uid in expert_uids:
            expert = name_to_block[expert_cls](hidden_dim)
            optimizer = optim_cls(expert.parameters()) if optim_cls is not None else None
            if clip_grad_norm is not None and optimizer is not None:
                from .layers.optim import ClippingWrapper

                optimizer = ClippingWrapper(optimizer, clip_grad_norm)
            if scheduler == "linear" and optimizer is not None:
                sched = get_linear_schedule_with_warmup(optimizer, num_warmup_steps or 0, num_total_steps or 10**9)
            else:
                sched = None
            module_backends[uid] = ModuleBackend(
                name=uid,
                module=expert,
                optimizer=optimizer,
                scheduler=sched,
                args_schema=args_schema,
                min_batch_size=min_batch_size,
                max_batch_size=max_batch_size,
            )
        if checkpoint_dir is not None and load_experts_from_dir:
            load_experts(module_backends, checkpoint_dir)
        return cls(
            dht,
            module_backends,
            device=torch.device(device),
            num_connection_handlers=num_connection_handlers,
            update_period=update_period,
            expiration=expiration,
            checkpoint_dir=checkpoint_dir,
            stats_report_interval=stats_report_interval,
            relay_endpoint=relay_endpoint,
            start=start,
        )


def _generate_uids(num_experts: int, expert_pattern: str, dht: Optional[DHT] = None, attempts_per_expert: int = 10) -> List[str]:
    """Sample unique expert uids from a pattern like ``"ffn.[0:256].[0:256]"``
    (reference server.py:351-399); checks the DHT for collisions."""
    remaining_attempts = num_experts * attempts_per_expert
    found_uids, attempted_uids = [], set()

    def _random_uid():
        uid = []
        for block in expert_pattern.split(UID_DELIMITER):
            match = re.fullmatch(r"\[(\d+):(\d+)\]", block)
            if match:
                start, stop = map(int, match.groups())
                uid.append(str(random.randint(start, stop - 1)))
            else:
                uid.append(block)
        return UID_DELIMITER.join(uid)

    while remaining_attempts > 0 and len(found_uids) < num_experts:
        batch = []
        while len(batch) < min(num_experts - len(found_uids), 32) and remaining_attempts > 0:
            uid = _random_uid()
            remaining_attempts -= 1
            if uid not in attempted_uids:
                attempted_uids.add(uid)
                assert is_valid_uid(uid), f"pattern {expert_pattern} produced invalid uid {uid}"
                batch.append(uid)
        if not batch:
            break
        if dht is not None:
            existing = get_expert_infos(dht, batch)
            batch = [uid for uid, info in zip(batch, existing) if info is None]
        found_uids.extend(batch)
    if len(found_uids) < num_experts:
        logger.warning(f"found only {len(found_uids)} of {num_experts} requested expert uids")
    return found_uids[:num_experts]


@contextmanager
def background_server(*args, shutdown_timeout: float = 5.0, **kwargs):
    """Spin up a server for the duration of a with-block (reference server.py:308)."""
    server = Server.create(*args, start=True, **kwargs)
    try:
        yield server.dht.peer_info
    finally:
        server.shutdown()
