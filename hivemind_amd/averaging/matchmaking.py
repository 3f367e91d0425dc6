"""Decentralized matchmaking: leader election for averaging groups.

Parity target: reference ``hivemind/averaging/matchmaking.py:24-550``. The
protocol is preserved:

* every peer declares ``(peer_id -> looking_for_group)`` under the current
  group key with an expiration time;
* peers try to *follow* the declared peer with the nearest expiration earlier
  than their own (ties by peer id); the would-be leader accepts followers via
  a long-lived ``rpc_join_group`` stream;
* the leader assembles the group when ``target_group_size`` is reached or its
  own declared expiration arrives with >= 2 members: it freezes an ordered
  peer list, draws a random ``group_id`` and broadcasts BEGIN_ALLREDUCE with
  everyone's gathered metadata;
* races resolve via explicit codes (GROUP_IS_FULL, NOT_LOOKING_FOR_GROUP,
  GROUP_DISBANDED with an optional suggested leader, ...). The
  ``request_timeout < averaging_expiration`` invariant breaks deadlock cycles
  (reference matchmaking.py:29-35).
"""

from __future__ import annotations

import asyncio
import contextlib
import os
import random
from dataclasses import dataclass, field
from typing import AsyncIterator, Dict, List, Optional, Set, Tuple

from ..dht import DHT
from ..p2p import P2P, PeerID, RpcContext, RpcMessage, ServicerBase
from ..utils.asyncio_utils import anext_impl
from ..utils.logging import get_logger
from ..utils.timed_storage import DHTExpiration, get_dht_time
from .control import StepControl
from .group_info import GroupInfo
from .key_manager import GroupKey, GroupKeyManager

logger = get_logger(__name__)


class MatchmakingException(Exception):
    pass


# message codes (reference proto/averaging.proto:5-28)
class Code:
    ACCEPTED = 0
    BEGIN_ALLREDUCE = 1
    GROUP_DISBANDED = 2
    REJECTED = 3
    GROUP_IS_FULL = 4
    NOT_LOOKING_FOR_GROUP = 5
    BAD_SCHEMA_HASH = 6
    BAD_GROUP_KEY = 7
    DUPLICATE_PEER_ID = 8
    NOT_DECLARED = 9
    PROTOCOL_VIOLATION = 10
    CANCELLED = 11
    BAD_EXPIRATION_TIME = 12


@dataclass
class JoinRequest(RpcMessage):
    peer_id: bytes = b""
    schema_hash: bytes = b""
    expiration: float = 0.0
    gather: bytes = b""
    client_mode: bool = False
    group_key: str = ""


@dataclass
class MessageFromLeader(RpcMessage):
    code: int = Code.REJECTED
    group_id: bytes = b""
    ordered_peer_ids: List[bytes] = field(default_factory=list)
    gathered: List[bytes] = field(default_factory=list)
    suggested_leader: bytes = b""


class Matchmaking:
    """One matchmaking agent per averager; shares the averager's servicer."""

    def __init__(
        self,
        p2p: P2P,
        schema_hash: bytes,
        dht: DHT,
        key_manager: GroupKeyManager,
        *,
        prefix: str,
        target_group_size: Optional[int],
        min_group_size: int,
        min_matchmaking_time: float,
        request_timeout: float,
        client_mode: bool = False,
        servicer_namespace: Optional[str] = None,
        servicer_type=None,
    ):
        self.p2p, self.schema_hash, self.dht = p2p, schema_hash, dht
        self.key_manager = key_manager
        self.peer_id = p2p.peer_id
        self.prefix = prefix
        self.target_group_size, self.min_group_size = target_group_size, min_group_size
        self.min_matchmaking_time = min_matchmaking_time
        self.request_timeout = request_timeout
        self.client_mode = client_mode
        self.servicer_namespace = servicer_namespace
        self.servicer_type = servicer_type if servicer_type is not None else MatchmakingServicer

        self.lock_looking_for_group = asyncio.Lock()
        self.lock_request_join_group = asyncio.Lock()
        self.follower_was_discarded = asyncio.Event()
        self.was_accepted_to_group = asyncio.Event()
        self.assembled_group: asyncio.Future = asyncio.Future()

        self.current_leader: Optional[PeerID] = None
        self.current_followers: Dict[PeerID, JoinRequest] = {}
        self.potential_leaders = PotentialLeaders(self.peer_id, min_matchmaking_time)
        self.step_control: Optional[StepControl] = None
        self._servicer: Optional["MatchmakingServicer"] = None

    @property
    def is_looking_for_group(self) -> bool:
        return self.lock_looking_for_group.locked()

    def _make_rejection(self, code: int, suggested_leader: Optional[PeerID] = None) -> MessageFromLeader:
        return MessageFromLeader(
            code=code, suggested_leader=suggested_leader.to_bytes() if suggested_leader else b""
        )

    async def look_for_group(self, step: StepControl) -> Optional[GroupInfo]:
        """Main entry: find peers and assemble a group (reference matchmaking.py:112-176)."""
        async with self.lock_looking_for_group:
            self.step_control = step
            request_leaders_task = asyncio.create_task(self._request_join_potential_leaders(step))
            try:
                return await asyncio.wait_for(
                    asyncio.shield(self.assembled_group), timeout=max(0.0, step.deadline - get_dht_time())
                )
            except asyncio.TimeoutError:
                return None
            except asyncio.CancelledError:
                raise
            except (MatchmakingException, Exception) as e:
                if not isinstance(e, MatchmakingException):
                    logger.exception("matchmaking failed")
                return None
            finally:
                await self._cancel_task(request_leaders_task)
                self.step_control = None
                # disband any remaining followers
                if self.current_followers:
                    for follower in list(self.current_followers):
                        pass  # their streams will observe GROUP_DISBANDED via assembled_group state
                self.was_accepted_to_group.clear()
                self.current_leader = None
                if self.assembled_group.done():
                    self.assembled_group = asyncio.Future()

    @staticmethod
    async def _cancel_task(task: asyncio.Task):
        task.cancel()
        with contextlib.suppress(asyncio.CancelledError):
            await task

    async def _request_join_potential_leaders(self, step: StepControl) -> Optional[GroupInfo]:
        """Cycle through potential leaders, trying to join each (reference matchmaking.py:134-176)."""
        async with self.potential_leaders.begin_search(step, self.key_manager, declare=not self.client_mode):
            while True:
                try:
                    next_leader = await self.potential_leaders.pop_next_leader()
                    group = await self._request_join_group(next_leader)
                    if group is not None:
                        if not self.assembled_group.done():
                            self.assembled_group.set_result(group)
                        return group
                except asyncio.TimeoutError:
                    # our declared expiration has arrived: if we have followers, lead; else redeclare
                    if len(self.current_followers) + 1 >= self.min_group_size and not self.assembled_group.done():
                        return await self._leader_assemble_group()
                    elif not step.allow_retries or get_dht_time() >= step.deadline:
                        if not self.assembled_group.done():
                            self.assembled_group.set_exception(MatchmakingException("matchmaking deadline"))
                        return None
                    # else: continue searching
                except (asyncio.CancelledError, concurrent_futures_CancelledError()):
                    raise
                except Exception as e:
                    logger.debug(f"matchmaking leader-search error: {e!r}")
                    await asyncio.sleep(self.request_timeout / 4)

    async def _request_join_group(self, leader: PeerID) -> Optional[GroupInfo]:
        """Join a specific leader; wait for ACCEPTED then BEGIN_ALLREDUCE
        (reference matchmaking.py:178-260)."""
        stream = None
        try:
            async with self.lock_request_join_group:
                if self.assembled_group.done():
                    return self.assembled_group.result()
                request = JoinRequest(
                    peer_id=self.peer_id.to_bytes(),
                    schema_hash=self.schema_hash,
                    expiration=self.potential_leaders.declared_expiration_time,
                    gather=self.step_control.gather_binary if self.step_control else b"",
                    client_mode=self.client_mode,
                    group_key=self.key_manager.current_key,
                )
                stub = self.servicer_type.get_stub(self.p2p, leader, namespace=self.servicer_namespace)
                stream = stub.rpc_join_group(request).__aiter__()
                message = MessageFromLeader.loads(
                    await asyncio.wait_for(anext_impl(stream), timeout=self.request_timeout)
                )
                if message.code != Code.ACCEPTED:
                    code = message.code
                    logger.debug(f"{self.peer_id}: leader {leader} rejected us with code {code}")
                    if code == Code.GROUP_DISBANDED and message.suggested_leader:
                        suggested = PeerID(message.suggested_leader)
                        if suggested != self.peer_id:
                            self.potential_leaders.suggest_leader(suggested)
                    return None
                self.current_leader = leader
                self.was_accepted_to_group.set()
                if len(self.current_followers) > 0:
                    await self._disband_group(suggested_leader=leader)

            # outside the lock: wait for the leader to begin all-reduce
            time_to_expiration = max(0.0, self.potential_leaders.declared_expiration_time - get_dht_time())
            message = MessageFromLeader.loads(
                await asyncio.wait_for(anext_impl(stream), time_to_expiration + self.request_timeout)
            )
            if message.code == Code.BEGIN_ALLREDUCE:
                return self._follower_assemble_group(leader, message)
            if message.code in (Code.GROUP_DISBANDED, Code.CANCELLED):
                if message.suggested_leader:
                    suggested = PeerID(message.suggested_leader)
                    if suggested != self.peer_id:
                        self.potential_leaders.suggest_leader(suggested)
            return None
        except asyncio.TimeoutError:
            return None
        except asyncio.CancelledError:
            raise
        except Exception as e:
            logger.debug(f"request_join_group({leader}) failed: {e!r}")
            return None
        finally:
            self.was_accepted_to_group.clear()
            self.current_leader = None
            if stream is not None:
                with contextlib.suppress(Exception):
                    await stream.aclose()

    # ------------------------------------------------------------ leader side

    async def rpc_join_group_impl(self, request: JoinRequest, context: RpcContext) -> AsyncIterator[MessageFromLeader]:
        """Accept or reject a follower (reference matchmaking.py:262-332)."""
        try:
            reason_to_reject = self._check_reasons_to_reject(request, context)
            if reason_to_reject is not None:
                yield reason_to_reject
                return
            follower_id = PeerID(request.peer_id)
            self.current_followers[follower_id] = request
            yield MessageFromLeader(code=Code.ACCEPTED)

            if (
                self.target_group_size is not None
                and len(self.current_followers) + 1 >= self.target_group_size
                and not self.assembled_group.done()
            ):
                # the group is full: assemble immediately
                await self._leader_assemble_group()

            # wait until the group is assembled or disbanded
            timeout = max(0.0, self.potential_leaders.declared_expiration_time - get_dht_time()) + self.request_timeout
            try:
                group_info = await asyncio.wait_for(asyncio.shield(self.assembled_group), timeout)
                if follower_id in group_info:
                    yield MessageFromLeader(
                        code=Code.BEGIN_ALLREDUCE,
                        group_id=group_info.group_id,
                        ordered_peer_ids=[p.to_bytes() for p in group_info.peer_ids],
                        gathered=list(group_info.gathered),
                    )
                    return
            except (asyncio.TimeoutError, asyncio.CancelledError, Exception):
                pass
            if self.was_accepted_to_group.is_set() and self.current_leader is not None:
                yield MessageFromLeader(
                    code=Code.GROUP_DISBANDED, suggested_leader=self.current_leader.to_bytes()
                )
            else:
                yield MessageFromLeader(code=Code.GROUP_DISBANDED)
        finally:
            self.current_followers.pop(PeerID(request.peer_id), None)
            self.follower_was_discarded.set()

    def _check_reasons_to_reject(self, request: JoinRequest, context: RpcContext) -> Optional[MessageFromLeader]:
        if not self.is_looking_for_group or self.assembled_group.done():
            return MessageFromLeader(code=Code.NOT_LOOKING_FOR_GROUP)
        if request.schema_hash != self.schema_hash:
            return MessageFromLeader(code=Code.BAD_SCHEMA_HASH)
        if request.group_key != self.key_manager.current_key:
            return MessageFromLeader(code=Code.BAD_GROUP_KEY)
        if self.potential_leaders.declared_group_key is None:
            return MessageFromLeader(code=Code.NOT_DECLARED)
        if self.potential_leaders.declared_expiration_time > (request.expiration or float("inf")):
            return MessageFromLeader(code=Code.BAD_EXPIRATION_TIME)
        peer_id = PeerID(request.peer_id)
        if self.current_leader is not None:
            return MessageFromLeader(code=Code.GROUP_DISBANDED, suggested_leader=self.current_leader.to_bytes())
        if peer_id == self.peer_id or peer_id in self.current_followers:
            return MessageFromLeader(code=Code.DUPLICATE_PEER_ID)
        if self.target_group_size is not None and len(self.current_followers) + 1 >= self.target_group_size:
            return MessageFromLeader(code=Code.GROUP_IS_FULL)
        return None

    async def _leader_assemble_group(self) -> GroupInfo:
        """Freeze the group, draw group_id, broadcast (reference matchmaking.py:371-394)."""
        assert self.lock_looking_for_group.locked()
        assert not self.assembled_group.done()
        group_id = os.urandom(16)
        ordered_peer_ids = [self.peer_id] + list(self.current_followers)
        random.shuffle(ordered_peer_ids)
        gathered = tuple(
            self.step_control.gather_binary
            if peer_id == self.peer_id
            else self.current_followers[peer_id].gather
            for peer_id in ordered_peer_ids
        )
        group_info = GroupInfo(group_id, tuple(ordered_peer_ids), gathered)
        logger.debug(f"{self.peer_id}: assembled group of {len(ordered_peer_ids)} peers")
        await self.key_manager.update_key_on_group_assembled(group_info)
        self.assembled_group.set_result(group_info)
        return group_info

    def _follower_assemble_group(self, leader: PeerID, msg: MessageFromLeader) -> GroupInfo:
        """Accept the group info from our leader (reference matchmaking.py:396-406)."""
        group_id = msg.group_id
        ordered_peer_ids = tuple(PeerID(item) for item in msg.ordered_peer_ids)
        assert self.peer_id in ordered_peer_ids, "peer is not a part of the assembled group"
        gathered = tuple(msg.gathered)
        group_info = GroupInfo(group_id, ordered_peer_ids, gathered)
        if not self.assembled_group.done():
            self.assembled_group.set_result(group_info)
        return group_info

    async def _disband_group(self, suggested_leader: Optional[PeerID] = None):
        """We joined someone else; our own followers must be turned away."""
        self.current_followers.clear()  # their streams time out and send GROUP_DISBANDED


def concurrent_futures_CancelledError():
    import concurrent.futures

    return concurrent.futures.CancelledError


class MatchmakingServicer(ServicerBase):
    """Servicer wrapper so several averagers can coexist under namespaces."""

    def __init__(self, matchmaking: Optional[Matchmaking] = None):
        self.matchmaking = matchmaking

    async def rpc_join_group(self, request: JoinRequest, context: RpcContext) -> AsyncIterator[MessageFromLeader]:
        assert self.matchmaking is not None
        async for message in self.matchmaking.rpc_join_group_impl(request, context):
            yield message


class PotentialLeaders:
    """Tracks candidate leaders sorted by declared expiration (reference matchmaking.py:414-546)."""

    def __init__(self, peer_id: PeerID, min_matchmaking_time: float):
        self._background_tasks: set = set()
        self.peer_id = peer_id
        self.min_matchmaking_time = min_matchmaking_time
        self.running = asyncio.Event()
        self.update_triggered = asyncio.Event()
        self.update_finished = asyncio.Event()
        self.declared_expiration = asyncio.Event()
        self.lock_search = asyncio.Lock()
        self.leader_queue: Dict[PeerID, DHTExpiration] = {}
        self.past_attempts: Set[Tuple[PeerID, DHTExpiration]] = set()
        self.declared_expiration_time = float("inf")
        self.declared_group_key: Optional[GroupKey] = None
        self.max_assured_time = float("-inf")
        self.search_end_time = float("inf")
        self._suggested: List[PeerID] = []

    @contextlib.asynccontextmanager
    async def begin_search(self, step: StepControl, key_manager: GroupKeyManager, declare: bool = True):
        async with self.lock_search:
            self.running.set()
            self.search_end_time = step.deadline if step.deadline is not None else float("inf")
            update_queue_task = asyncio.create_task(self._update_queue_periodically(key_manager))
            declare_task = asyncio.create_task(self._declare_averager_periodically(step, key_manager)) if declare else None
            try:
                yield self
            finally:
                update_queue_task.cancel()
                if declare_task is not None:
                    declare_task.cancel()
                for task in (update_queue_task, declare_task):
                    if task is not None:
                        with contextlib.suppress(asyncio.CancelledError):
                            await task
                self.running.clear()
                self.update_triggered.clear()
                self.update_finished.clear()
                if declare and self.declared_group_key is not None:
                    prev_key, prev_expiration = self.declared_group_key, self.declared_expiration_time
                    self.declared_group_key, self.declared_expiration_time = None, float("inf")
                    self.leader_queue.clear()
                    self.past_attempts.clear()
                    _undeclare = asyncio.create_task(
                        key_manager.declare_averager(prev_key, self.peer_id, prev_expiration, looking_for_group=False)
                    )
                    self._background_tasks.add(_undeclare)  # keep alive: tasks are weakly referenced
                    _undeclare.add_done_callback(self._background_tasks.discard)

    def suggest_leader(self, peer_id: PeerID):
        self._suggested.append(peer_id)

    async def pop_next_leader(self) -> PeerID:
        """Next leader to try: earliest-expiring declared peer earlier than us
        (raises asyncio.TimeoutError when our own expiration arrives)."""
        assert self.running.is_set()
        while True:
            if self._suggested:
                return self._suggested.pop(0)
            maybe_next_leader, entry = None, None
            for peer, expiration in self.leader_queue.items():
                if (peer, expiration) in self.past_attempts:
                    continue
                if (expiration, peer.to_bytes()) < (self.declared_expiration_time, self.peer_id.to_bytes()):
                    if entry is None or (expiration, peer.to_bytes()) < entry:
                        maybe_next_leader, entry = peer, (expiration, peer.to_bytes())
            if maybe_next_leader is not None:
                self.past_attempts.add((maybe_next_leader, entry[0]))
                return maybe_next_leader
            # no candidates: trigger a queue update and wait for it or our expiration
            self.update_triggered.set()
            self.update_finished.clear()
            timeout = None
            if self.declared_expiration_time != float("inf"):
                timeout = max(0.0, self.declared_expiration_time - get_dht_time())
            done, pending = await asyncio.wait(
                [
                    asyncio.ensure_future(self.update_finished.wait()),
                    asyncio.ensure_future(self.declared_expiration.wait()),
                ],
                timeout=timeout,
                return_when=asyncio.FIRST_COMPLETED,
            )
            for fut in pending:
                fut.cancel()
            if not done and timeout is not None:
                raise asyncio.TimeoutError("reached our declared expiration: time to lead or retry")
            self.declared_expiration.clear()

    async def _update_queue_periodically(self, key_manager: GroupKeyManager):
        while self.running.is_set():
            self.update_triggered.clear()
            new_peers = await key_manager.get_averagers(key_manager.current_key, only_active=True)
            self.max_assured_time = max(self.max_assured_time, get_dht_time() + self.min_matchmaking_time)
            self.leader_queue.clear()
            for peer, expiration_time in new_peers:
                if peer == self.peer_id or (peer, expiration_time) in self.past_attempts:
                    continue
                self.leader_queue[peer] = expiration_time
                self.max_assured_time = max(self.max_assured_time, expiration_time)
            self.update_finished.set()
            await asyncio.wait(
                [asyncio.ensure_future(self.update_triggered.wait())],
                timeout=max(0.1, self.min_matchmaking_time / 2),
            )

    async def _declare_averager_periodically(self, step: StepControl, key_manager: GroupKeyManager):
        try:
            while self.running.is_set():
                new_expiration_time = min(
                    max(get_dht_time() + self.min_matchmaking_time, step.scheduled_time), self.search_end_time
                )
                self.declared_group_key = group_key = key_manager.current_key
                self.declared_expiration_time = new_expiration_time
                self.declared_expiration.set()
                await key_manager.declare_averager(group_key, self.peer_id, new_expiration_time, looking_for_group=True)
                await asyncio.sleep(max(0.0, self.declared_expiration_time - get_dht_time()))
                if self.running.is_set() and len(self.leader_queue) == 0:
                    await key_manager.update_key_on_not_enough_peers()
        except asyncio.CancelledError:
            pass
        finally:
            if self.declared_group_key is not None:
                prev_key, prev_expiration = self.declared_group_key, self.declared_expiration_time
                with contextlib.suppress(Exception):
                    _undeclare = asyncio.create_task(
                        key_manager.declare_averager(prev_key, self.peer_id, prev_expiration, looking_for_group=False)
                    )
                    self._background_tasks.add(_undeclare)  # keep alive: tasks are weakly referenced
                    _undeclare.add_done_callback(self._background_tasks.discard)
