"""DecentralizedAverager: matchmaking + group all-reduce + state sharing.

Parity target: reference ``hivemind/averaging/averager.py:50-821``. Key
behavioral points preserved:

* ``step(gather=..., weight=..., wait=...) -> gathered | StepControl``:
  matchmaking via DHT group keys, then an in-place weighted group average of
  ``averaged_tensors``; retries within the step deadline; a ``gather``
  side-channel distributes small metadata with group assembly.
* a schema hash guards that only compatible averagers group together;
* ``rpc_aggregate_part`` routes into the registered AllReduceRunner for the
  announced ``group_id`` -- including the race where a peer's first part
  arrives before BEGIN_ALLREDUCE is processed locally
  (reference averager.py:585-589: ``_pending_groups_registered``);
* state sharing: ``load_state_from_peers`` downloads (metadata, tensors) from
  the best donor found under ``{prefix}.all_averagers``.

Architecture difference (MI355X-native): the averager is NOT a forked process
with shared-memory tensors. One process per GPU owns its tensors directly;
averager coroutines run on the DHT's event-loop thread; the data plane is
either the RCCL bucketed all-reduce over xGMI (same-node groups,
averaging/rccl.py) or the RPC butterfly (averaging/allreduce.py) for
WAN/CPU peers.
"""

from __future__ import annotations

import asyncio
import concurrent.futures
import contextlib
import hashlib
import os
import random
import threading
import weakref
from dataclasses import dataclass, field
from typing import Any, AsyncIterator, Dict, List, Optional, Sequence, Tuple, Union

import torch

from ..compression import (
    CompressionBase,
    CompressionInfo,
    NoCompression,
    WireTensor,
    deserialize_torch_tensor,
    serialize_torch_tensor,
    split_for_streaming,
    combine_from_streaming,
)
from ..dht import DHT
from ..p2p import P2P, PeerID, RpcContext, RpcMessage, ServicerBase
from ..utils.asyncio_utils import achain, aiter_with_timeout, anext_impl, as_aiter, enter_asynchronously
from ..utils.logging import get_logger
from ..utils.serializer import MSGPackSerializer
from ..utils.timed_storage import DHTExpiration, ValueWithExpiration, get_dht_time
from .allreduce import AllReduceRunner, AveragingData, AveragingMode, DataCode
from .control import AveragingStage, StepControl
from .group_info import GroupInfo
from .key_manager import GroupKeyManager
from .load_balancing import load_balance_peers
from .matchmaking import JoinRequest, Matchmaking, MatchmakingException, MessageFromLeader
from .partition import DEFAULT_PART_SIZE_BYTES
from .rccl import (
    DistributedAllReduceRunner,
    distributed_world_info,
    get_process_group_for_ranks,
    group_matches_world,
    group_world_ranks,
    release_collective_ticket,
)

logger = get_logger(__name__)


class AveragingError(Exception):
    pass


@dataclass
class GroupMetadata:
    """Per-peer metadata distributed with group assembly (gather side channel)."""

    bandwidths: list
    modes: list
    user_gathered: dict
    dist_infos: list
    weights: list
    rccl_ticket: Optional[int] = None  # launch-order slot for the RCCL plane


class _null_actx:
    async def __aenter__(self):
        return None

    async def __aexit__(self, *args):
        return False


@dataclass
class DownloadRequest(RpcMessage):
    pass


@dataclass
class DownloadData(RpcMessage):
    metadata: bytes = b""
    tensor: Optional[WireTensor] = None


class DecentralizedAverager(ServicerBase):
    # all averager subclasses (Training/Gradient/PowerSGD) speak one wire
    # protocol: a plain client averager can download state from any of them
    _servicer_name = "DecentralizedAverager"

    """Averages a fixed-schema list of tensors with dynamically matched groups of peers."""

    _matchmaking: Matchmaking
    _allreduce_runner_class = AllReduceRunner  # override point for fault-injection tests

    def __init__(
        self,
        averaged_tensors: Sequence[torch.Tensor],
        dht: DHT,
        *,
        start: bool,
        prefix: str,
        target_group_size: Optional[int] = None,
        min_group_size: int = 2,
        initial_group_bits: str = "",
        min_matchmaking_time: float = 5.0,
        request_timeout: float = 3.0,
        averaging_alpha: float = 1.0,
        part_size_bytes: int = DEFAULT_PART_SIZE_BYTES,
        allreduce_timeout: Optional[float] = None,
        next_chunk_timeout: Optional[float] = None,
        sender_timeout: Optional[float] = None,
        reducer_timeout: Optional[float] = None,
        compression: CompressionBase = NoCompression(),
        state_compression: CompressionBase = NoCompression(),
        tensor_infos: Optional[Sequence[CompressionInfo]] = None,
        bandwidth: Optional[float] = None,
        client_mode: bool = False,
        auxiliary: bool = False,
        allow_state_sharing: Optional[bool] = None,
        declare_state_period: float = 30.0,
        allreduce_wire_dtype: Optional[torch.dtype] = None,
        allreduce_codec: Optional[str] = None,
        use_rccl_when_available: bool = True,
        shutdown_timeout: float = 5.0,
    ):
        assert "." not in prefix, "prefix must not contain '.'"
        self.dht = dht
        self.prefix = prefix
        self._p2p: P2P = dht.replicate_p2p()
        self.peer_id: PeerID = self._p2p.peer_id
        self.mode = (
            AveragingMode.AUX if auxiliary else (AveragingMode.CLIENT if client_mode else AveragingMode.NODE)
        )
        self.client_mode, self.auxiliary = client_mode, auxiliary
        self.bandwidth = bandwidth
        self.min_matchmaking_time = min_matchmaking_time
        self.request_timeout = request_timeout
        self.averaging_alpha = averaging_alpha
        self.part_size_bytes = part_size_bytes
        self.allreduce_timeout = allreduce_timeout or float("inf")
        self.sender_timeout = sender_timeout if sender_timeout is not None else (next_chunk_timeout or 10.0)
        self.reducer_timeout = reducer_timeout if reducer_timeout is not None else (next_chunk_timeout or 15.0)
        self.compression, self.state_compression = compression, state_compression
        self.allreduce_wire_dtype = allreduce_wire_dtype
        self.allreduce_codec = allreduce_codec
        self.use_rccl_when_available = use_rccl_when_available
        self.shutdown_timeout = shutdown_timeout
        self.declare_state_period = declare_state_period
        self._allow_state_sharing = bool(allow_state_sharing if allow_state_sharing is not None else not (client_mode or auxiliary))
        self._state_sharing_priority: Optional[float] = None

        self._averaged_tensors = tuple(averaged_tensors)
        self.lock_averaged_tensors = threading.Lock()
        for tensor in self._averaged_tensors:
            assert tensor.grad_fn is None, "averaged_tensors must be either leaves or detached"
        self.total_size = sum(t.numel() for t in self._averaged_tensors)
        self.schema_hash = compute_schema_hash(self._averaged_tensors)
        self.tensor_infos = tensor_infos

        self.matchmaking_kwargs = dict(
            prefix=prefix,
            target_group_size=target_group_size,
            min_group_size=min_group_size,
            min_matchmaking_time=min_matchmaking_time,
            request_timeout=request_timeout,
        )
        self._initial_group_bits = initial_group_bits
        self._inflight_futures: set = set()
        self._running_groups: Dict[bytes, asyncio.Future] = {}
        self.last_data_plane: Optional[str] = None  # "rccl" | "rpc" after a round
        # Communicators are created LAZILY per matched rank-set via
        # rccl.get_process_group_for_ranks (use_local_synchronization=True):
        # only group members rendezvous, so ranks that never average (or only
        # average in other groups) cannot hang everyone in a collective
        # new_group(). Cross-averager interleaving on a shared communicator is
        # prevented by the CollectiveSequencer tickets + process collective
        # lock (rccl.py).
        self._state_updated = asyncio.Event()
        self._declare_state_task: Optional[asyncio.Task] = None
        self._ready = concurrent.futures.Future()
        self._alive = False
        if start:
            self.run_in_background(await_ready=True)

    # ------------------------------------------------------------- lifecycle

    @property
    def _loop(self):
        return self.dht.loop

    def _keep_future(self, future):
        """Strong-reference fire-and-forget futures until done: asyncio tasks
        are weakly referenced and can be GC'd mid-await otherwise."""
        self._inflight_futures.add(future)
        future.add_done_callback(self._inflight_futures.discard)
        return future

    def run_in_background(self, await_ready: bool = True, timeout: Optional[float] = 30.0):
        self._alive = True
        self._keep_future(asyncio.run_coroutine_threadsafe(self._startup(), self._loop))
        if await_ready:
            self._ready.result(timeout)

    async def _startup(self):
        try:
            self.key_manager = GroupKeyManager(
                self.dht,
                self.prefix,
                initial_group_bits=self._initial_group_bits,
                target_group_size=self.matchmaking_kwargs["target_group_size"],
                p2p=self._p2p,
            )
            self._matchmaking = Matchmaking(
                self._p2p,
                self.schema_hash,
                self.dht,
                self.key_manager,
                client_mode=self.client_mode,
                servicer_namespace=self.prefix,
                servicer_type=type(self),
                **self.matchmaking_kwargs,
            )
            if not self.client_mode:
                await self.add_p2p_handlers(self._p2p, namespace=self.prefix)
                if self._allow_state_sharing:
                    self._declare_state_task = asyncio.create_task(self._declare_state_periodically())
            self._ready.set_result(None)
        except Exception as e:
            logger.exception("averager failed to start")
            self._ready.set_exception(e)

    @property
    def is_alive(self) -> bool:
        return self._alive

    @property
    def allow_state_sharing(self) -> bool:
        return self._allow_state_sharing

    @allow_state_sharing.setter
    def allow_state_sharing(self, value: bool):
        self._allow_state_sharing = value

    def shutdown(self):
        if not self._alive:
            return
        self._alive = False
        try:
            future = asyncio.run_coroutine_threadsafe(self._shutdown_async(), self._loop)
            future.result(self.shutdown_timeout)
        except Exception:
            pass

    async def _shutdown_async(self):
        if self._declare_state_task is not None:
            self._declare_state_task.cancel()
        with contextlib.suppress(Exception):
            self.remove_p2p_handlers(self._p2p, namespace=self.prefix)

    def __del__(self):
        try:
            if self._alive:
                self.shutdown()
        except Exception:
            pass

    # ------------------------------------------------------------ public API

    @contextlib.contextmanager
    def get_tensors(self):
        """Access the averaged tensors under lock; modifications are allowed
        (reference averager.py:564-572)."""
        with self.lock_averaged_tensors:
            yield self._averaged_tensors

    def step(
        self,
        gather: Optional[Any] = None,
        scheduled_time: Optional[DHTExpiration] = None,
        weight: Optional[float] = None,
        timeout: Optional[float] = None,
        allow_retries: bool = True,
        require_trigger: bool = False,
        wait: bool = True,
        rccl_ticket: Optional[int] = None,
    ) -> Union[Optional[Dict[PeerID, Any]], StepControl]:
        """Look for a group and average tensors with it (reference averager.py:367-419).

        :returns: on success, {peer_id: gathered_value} for all group peers.
        """
        if self.mode == AveragingMode.AUX and weight is not None:
            logger.warning("aux peers averaging weight is always 0")
        deadline = get_dht_time() + timeout if timeout is not None else float("inf")
        scheduled_time = scheduled_time if scheduled_time is not None else get_dht_time() + self.min_matchmaking_time
        weight = weight if weight is not None else float(self.mode != AveragingMode.AUX)
        gather_binary = MSGPackSerializer.dumps(self._gather_metadata(gather, weight))
        control = StepControl(
            scheduled_time=scheduled_time,
            deadline=deadline,
            allow_retries=allow_retries,
            weight=weight,
            gather_binary=gather_binary,
        )
        control.rccl_ticket = rccl_ticket
        if not require_trigger:
            control.allow_allreduce()
        logger.debug(f"{self.prefix}@{self.peer_id}: step() created control {id(control):#x} trig={control.triggered}")
        self._keep_future(asyncio.run_coroutine_threadsafe(self._step(control), self._loop))
        return control.result(timeout) if wait else control

    async def _step(self, step: StepControl):
        try:
            while not step.done():
                try:
                    self._pending_groups_registered = asyncio.Event()
                    step.stage = AveragingStage.LOOKING_FOR_GROUP
                    logger.debug(f"{self.prefix}@{self.peer_id}: looking for group (deadline in {step.deadline - get_dht_time():.1f}s)")
                    group_info = await self._matchmaking.look_for_group(step)
                    logger.debug(f"{self.prefix}@{self.peer_id}: matchmaking returned {group_info}")
                    if group_info is None:
                        if step.allow_retries and get_dht_time() < step.deadline:
                            continue
                        raise AveragingError("Averaging step failed: could not find a group")

                    if step.done():
                        break  # cancelled while matchmaking
                    if not step.triggered:
                        step.stage = AveragingStage.AWAITING_TRIGGER
                        logger.debug(f"{self.prefix}@{self.peer_id}: awaiting trigger on {id(step):#x}")
                        try:
                            await step.wait_for_trigger()
                        except (asyncio.CancelledError, concurrent.futures.CancelledError):
                            break  # step was cancelled: do not run the round
                    if step.done() or step.cancelled():
                        break
                    logger.debug(f"{self.prefix}@{self.peer_id}: trigger received")
                    step.stage = AveragingStage.RUNNING_ALLREDUCE
                    gathered = await asyncio.wait_for(
                        self._aggregate_with_group(group_info, step.weight, ticket=step.rccl_ticket),
                        timeout=self.allreduce_timeout,
                    )
                    step.set_result(gathered)
                except (
                    AveragingError,
                    MatchmakingException,
                    AssertionError,
                    asyncio.InvalidStateError,
                    asyncio.TimeoutError,
                ) as e:
                    if not step.allow_retries or get_dht_time() >= step.deadline:
                        logger.warning(f"averager step failed: {e}")
                        step.set_exception(e if isinstance(e, Exception) else AveragingError(str(e)))
                    else:
                        logger.debug(f"averager step attempt failed: {e!r}; retrying")
        except asyncio.CancelledError:
            step.cancel()
            raise
        except BaseException as e:
            step.set_exception(e if isinstance(e, Exception) else AveragingError(repr(e)))
        finally:
            # free the launch-order slot whether the round ran, failed, or was
            # abandoned -- otherwise every later RCCL round on this rank stalls
            release_collective_ticket(step.rccl_ticket)
            step.rccl_ticket = None

    def _parse_group_metadata(self, group_info: GroupInfo, weight: float) -> "GroupMetadata":
        """Unpack per-peer gathered metadata: [bandwidth, mode, user_gather, dist_info, weight]."""
        bandwidths, modes, user_gathered_raw, dist_infos, weights = [], [], [], [], []
        for raw in group_info.gathered:
            meta = MSGPackSerializer.loads(raw) if raw else {}
            bandwidths.append(meta.get("bandwidth"))
            modes.append(AveragingMode(meta.get("mode", AveragingMode.NODE.value)))
            user_gathered_raw.append(meta.get("gather"))
            dist_infos.append(tuple(meta["dist"]) if meta.get("dist") else None)
            weights.append(float(meta.get("weight", 1.0)))
        my_index = group_info.peer_ids.index(self.peer_id)
        weights[my_index] = weight  # trust our own latest weight
        return GroupMetadata(
            bandwidths=bandwidths,
            modes=modes,
            user_gathered=dict(zip(group_info.peer_ids, user_gathered_raw)),
            dist_infos=dist_infos,
            weights=weights,
        )

    def _rccl_group_ranks(self, meta: "GroupMetadata") -> Optional[list]:
        """World ranks of the matched group if the whole group lives in our
        torch.distributed world (any subset -- Moshpit subgroups of the node
        included), else None -> RPC butterfly."""
        if not self.use_rccl_when_available or not all(m == AveragingMode.NODE for m in meta.modes):
            return None
        return group_world_ranks(meta.dist_infos)

    async def _average_tensors_with_group(
        self,
        tensors: Sequence[torch.Tensor],
        group_info: GroupInfo,
        weight: float,
        meta: "GroupMetadata",
        group_id: Optional[bytes] = None,
        take_lock: bool = True,
    ):
        """Average an arbitrary tensor list with the group, picking the data plane:
        bucketed RCCL-over-xGMI when the group is exactly our torch.distributed
        world, else the RPC butterfly. Subclasses (PowerSGD) call this several
        times per round with distinct group_id suffixes."""
        lock_ctx = enter_asynchronously(self.lock_averaged_tensors) if take_lock else _null_actx()
        rccl_ranks = self._rccl_group_ranks(meta)
        if rccl_ranks is not None:
            try:
                self.last_data_plane = "rccl"
                # full node and Moshpit subgroups alike: cached member-only communicator
                process_group = get_process_group_for_ranks(rccl_ranks)
                async with lock_ctx:
                    await asyncio.get_event_loop().run_in_executor(
                        None,
                        DistributedAllReduceRunner(
                            tensors,
                            weight,
                            process_group=process_group,
                            wire_dtype=self.allreduce_wire_dtype,
                            codec=self.allreduce_codec,
                            averaging_alpha=self.averaging_alpha,
                            ticket=meta.rccl_ticket,
                        ).run,
                    )
                return
            except RuntimeError as e:
                # communicator creation / collective failure: the matched group
                # is still valid, so fall through to the RPC butterfly rather
                # than failing the whole round. NOTE: this branch is a
                # same-decision-on-every-rank situation only for communicator
                # SETUP failures; a mid-collective failure already poisons the
                # communicator on every member, so they all land here.
                logger.warning(f"RCCL data plane failed ({e!r}); falling back to the RPC butterfly")
                self.last_data_plane = "rpc"
        self.last_data_plane = "rpc"
        download_bandwidths = [
            (0.0 if mode == AveragingMode.CLIENT else bw) for mode, bw in zip(meta.modes, meta.bandwidths)
        ]
        total_size = sum(t.numel() for t in tensors)
        peer_fractions = await asyncio.get_event_loop().run_in_executor(
            None, load_balance_peers, total_size, download_bandwidths, self.part_size_bytes
        )
        async with lock_ctx:
            await self._run_allreduce_inplace_(
                tensors, group_info, group_id=group_id, peer_fractions=peer_fractions, weight=weight, modes=meta.modes
            )

    async def _aggregate_with_group(
        self, group_info: GroupInfo, weight: float, ticket: Optional[int] = None
    ) -> Dict[PeerID, Any]:
        """Run the data plane for one assembled group (reference averager.py:514-562)."""
        logger.debug(f"{self.prefix}@{self.peer_id}: entering aggregation, group={group_info.group_id.hex()[:8]}")
        try:
            meta = self._parse_group_metadata(group_info, weight)
            meta.rccl_ticket = ticket
            await self._aggregate_tensors_with_group(group_info, weight, meta)
            self._state_updated.set()
            return meta.user_gathered
        except BaseException as e:
            if isinstance(e, asyncio.CancelledError):
                raise
            logger.debug(f"aggregation failed: {e!r}")
            raise AveragingError(f"aggregation stage failed: {e!r}") from e

    async def _aggregate_tensors_with_group(self, group_info: GroupInfo, weight: float, meta: "GroupMetadata"):
        """Default round: average the averager's own tensors. Override point for
        multi-phase schemes (PowerSGD runs two chained rounds)."""
        await self._average_tensors_with_group(self._averaged_tensors, group_info, weight, meta)

    async def _run_allreduce_inplace_(
        self,
        tensors: Sequence[torch.Tensor],
        group_info: GroupInfo,
        group_id: Optional[bytes] = None,
        weight: float = 1.0,
        peer_fractions: Optional[Tuple[float, ...]] = None,
        modes: Optional[Sequence[AveragingMode]] = None,
        **kwargs,
    ):
        """Run one allreduce round on the RPC butterfly (reference averager.py:537-562)."""
        group_id = group_id if group_id is not None else group_info.group_id
        runner = self._allreduce_runner_class(
            p2p=self._p2p,
            servicer_type=type(self),
            namespace=self.prefix,
            group_id=group_id,
            tensors=tensors,
            ordered_peer_ids=group_info.peer_ids,
            peer_fractions=peer_fractions,
            weight=weight,
            modes=modes,
            compression=self.compression,
            part_size_bytes=self.part_size_bytes,
            sender_timeout=self.sender_timeout,
            reducer_timeout=self.reducer_timeout,
            **kwargs,
        )
        self._register_allreduce_group(group_id, runner)
        try:
            iter_results = runner.run()
            index = 0
            async for delta in aiter_with_timeout(iter_results, self.reducer_timeout):
                if runner.modes[runner.ordered_peer_ids.index(self.peer_id)] != AveragingMode.AUX:
                    from ..ops import apply_delta_

                    # fused in-place delta apply (HIP kernel on GPU)
                    apply_delta_(tensors[index].detach(), delta.to(tensors[index].device), self.averaging_alpha)
                index += 1
            self._state_updated.set()
        finally:
            self._unregister_allreduce_group(group_id)

    def _register_allreduce_group(self, group_id: bytes, runner: AllReduceRunner):
        future = self._running_groups.get(group_id)
        if future is None:
            self._running_groups[group_id] = future = asyncio.Future()
        if not future.done():
            future.set_result(runner)

    def _unregister_allreduce_group(self, group_id: bytes):
        self._running_groups.pop(group_id, None)

    # ----------------------------------------------------- servicer handlers

    async def rpc_join_group(self, request: JoinRequest, context: RpcContext) -> AsyncIterator[MessageFromLeader]:
        """Matchmaking entry (delegates to the Matchmaking state machine)."""
        async for message in self._matchmaking.rpc_join_group_impl(request, context):
            yield message

    async def rpc_aggregate_part(
        self, stream: AsyncIterator[AveragingData], context: RpcContext
    ) -> AsyncIterator[AveragingData]:
        """Route an incoming sender stream into the runner registered for its group.

        Handles the BEGIN_ALLREDUCE race: a peer's first part may arrive before
        we have registered the group locally (reference averager.py:581-598)."""
        first_message = await anext_impl(stream.__aiter__() if hasattr(stream, "__aiter__") else stream)
        group_id = first_message.group_id
        future = self._running_groups.get(group_id)
        if future is None:
            self._running_groups[group_id] = future = asyncio.Future()
        try:
            runner: AllReduceRunner = await asyncio.wait_for(asyncio.shield(future), timeout=self.request_timeout * 3)
        except asyncio.TimeoutError:
            yield AveragingData(code=DataCode.ERROR, group_id=group_id)
            return

        async def recombined_stream():
            yield first_message
            async for msg in stream:
                yield msg

        async for message in runner.rpc_aggregate_part(recombined_stream(), context):
            yield message

    # --------------------------------------------------------- state sharing

    async def _declare_state_periodically(self):
        """Advertise ourselves as a state donor under {prefix}.all_averagers
        (reference averager.py:600-626)."""
        while True:
            try:
                if self._allow_state_sharing:
                    await asyncio.wrap_future(
                        self.dht.store(
                            key=f"{self.prefix}.all_averagers",
                            subkey=self.peer_id.to_base58(),
                            value=self.state_sharing_priority,
                            expiration_time=get_dht_time() + self.declare_state_period,
                            return_future=True,
                        )
                    )
                await asyncio.sleep(self.declare_state_period / 2)
            except asyncio.CancelledError:
                break
            except Exception as e:
                logger.debug(f"declare_state failed: {e!r}")
                await asyncio.sleep(self.declare_state_period / 2)

    @property
    def state_sharing_priority(self) -> float:
        """Donors with higher priority are tried first. Assignable (reference
        test_load_state_priority); subclasses override the getter
        (TrainingStateAverager uses local epoch)."""
        if self._state_sharing_priority is not None:
            return float(self._state_sharing_priority)
        return float(self._allow_state_sharing)

    @state_sharing_priority.setter
    def state_sharing_priority(self, value: float):
        self._state_sharing_priority = float(value)

    async def _get_current_state_from_host_process(self) -> Tuple[Any, Sequence[torch.Tensor]]:
        """Hook: collect the state to share. Default: metadata=None + averaged tensors."""
        return await asyncio.get_event_loop().run_in_executor(None, self.get_current_state)

    def get_current_state(self) -> Tuple[Any, Sequence[torch.Tensor]]:
        """Override to share optimizer state etc. (reference averager.py:769-809)."""
        with self.get_tensors() as tensors:
            return None, [t.detach().cpu().clone() for t in tensors]

    def load_state(self, metadata: Any, tensors: Sequence[torch.Tensor]):
        """Override to restore downloaded state; default replaces averaged tensors."""
        with self.get_tensors() as local:
            if len(local) == len(tensors):
                for mine, new in zip(local, tensors):
                    mine.detach().copy_(new.to(mine.device, mine.dtype))

    def _debug_state(self) -> str:
        return f"{self.prefix}@{self.peer_id}"

    async def rpc_download_state(self, request: DownloadRequest, context: RpcContext) -> AsyncIterator[DownloadData]:
        """Stream (metadata, tensors) to a joining peer (reference averager.py:628-666)."""
        if not self._allow_state_sharing:
            return
        metadata, tensors = await self._get_current_state_from_host_process()
        metadata_blob = MSGPackSerializer.dumps(metadata)
        yield DownloadData(metadata=metadata_blob)
        for tensor in tensors:
            serialized = await asyncio.get_event_loop().run_in_executor(
                None, lambda t=tensor: self.state_compression.compress(t, CompressionInfo.from_tensor(t))
            )
            for part in split_for_streaming(serialized):
                yield DownloadData(tensor=part)

    def load_state_from_peers(self, wait: bool = True, timeout: Optional[float] = None, apply: bool = False):
        """Download the latest state from the best donor (reference averager.py:668-736).

        Returns (metadata, tensors) without modifying local state unless
        ``apply=True`` (reference semantics: the base class only downloads;
        TrainingStateAverager overrides the default to install the state)."""
        future = asyncio.run_coroutine_threadsafe(self._load_state_from_peers(timeout, apply), self._loop)
        return future.result(timeout) if wait else future

    async def _load_state_from_peers(self, timeout: Optional[float] = None, apply: bool = False) -> Optional[Tuple[Any, Sequence[torch.Tensor]]]:
        key = f"{self.prefix}.all_averagers"
        result = await asyncio.wrap_future(self.dht.get(key, latest=True, return_future=True))
        if result is None or not isinstance(result.value, dict):
            logger.info(f"no state donors found under {key}")
            return None
        # sort donors by priority (highest first), skip ourselves
        donors = []
        for donor_b58, entry in result.value.items():
            try:
                donor = PeerID.from_base58(donor_b58)
            except Exception:
                continue
            if donor == self.peer_id:
                continue
            try:
                priority = float(entry.value)
            except (TypeError, ValueError):
                priority = 0.0
            donors.append((priority, random.random(), donor))
        donors.sort(reverse=True)
        for _, _, donor in donors:
            try:
                stub = type(self).get_stub(self._p2p, donor, namespace=self.prefix)
                stream = stub.rpc_download_state(DownloadRequest())
                metadata = None
                tensor_parts: List[WireTensor] = []
                tensors: List[torch.Tensor] = []
                async for payload in aiter_with_timeout(stream, self.reducer_timeout):
                    message = DownloadData.loads(payload)
                    if message.metadata:
                        metadata = MSGPackSerializer.loads(message.metadata)
                    if message.tensor is not None:
                        if message.tensor.chunks and tensor_parts:
                            tensors.append(deserialize_torch_tensor(combine_from_streaming(tensor_parts)))
                            tensor_parts = []
                        tensor_parts.append(message.tensor)
                if tensor_parts:
                    tensors.append(deserialize_torch_tensor(combine_from_streaming(tensor_parts)))
                logger.info(f"downloaded state from {donor}: {len(tensors)} tensors")
                if apply:
                    await asyncio.get_event_loop().run_in_executor(None, self.load_state, metadata, tensors)
                return metadata, tensors
            except asyncio.CancelledError:
                raise
            except Exception as e:
                logger.warning(f"failed to download state from {donor}: {e!r}")
                continue
        return None

    # ------------------------------------------------------------- group key

    def get_group_bits(self) -> str:
        return self.key_manager.group_bits

    def set_group_bits(self, group_bits: str):
        self.key_manager.group_bits = group_bits

    # --------------------------------------------------------------- helpers

    def _gather_metadata(self, user_gather: Any, weight: float) -> dict:
        return {
            "bandwidth": 0.0 if self.client_mode else self.bandwidth,
            "mode": self.mode.value,
            "gather": user_gather,
            "dist": distributed_world_info(),
            "weight": weight,
        }


def compute_schema_hash(tensors: Sequence[torch.Tensor]) -> bytes:
    """Hash of tensor shapes/dtypes/devices classes (reference averager.py:812-821)."""
    schema_dicts = [
        {
            "shape": list(tensor.shape),
            "dtype": str(tensor.dtype),
        }
        for tensor in tensors
    ]
    return hashlib.sha256(MSGPackSerializer.dumps(schema_dicts)).digest()
