"""ProgressTracker: swarm-wide sample accounting and epoch synchronization.

Parity target: reference ``hivemind/optim/progress_tracker.py:31-363``:
each peer periodically publishes signed ``LocalTrainingProgress`` (epoch,
samples accumulated, samples/s, client flag) under ``{prefix}_progress``; a
fetcher aggregates everyone's records into ``GlobalTrainingProgress``
(current global epoch, total samples toward the target batch, ETA of the next
epoch); ``ready_to_update_epoch`` fires when the target batch is reached
globally or the swarm has advanced beyond our epoch. The refresh period
adapts to the expected drift (more peers -> less frequent refresh per peer).
"""

from __future__ import annotations

import asyncio
import contextlib
import math
import random
import threading
from dataclasses import dataclass, field
from typing import Dict, Optional

import pydantic

from ..dht import DHT
from ..dht.crypto import SignatureValidator
from ..dht.schema import BytesWithPublicKey, SchemaValidator
from ..p2p import PeerID
from ..utils.logging import get_logger
from ..utils.performance_ema import PerformanceEMA
from ..utils.timed_storage import DHTExpiration, ValueWithExpiration, get_dht_time

logger = get_logger(__name__)


class LocalTrainingProgress(pydantic.BaseModel):
    peer_id: bytes
    epoch: int
    samples_accumulated: int
    samples_per_second: float
    time: float
    client_mode: bool

    @pydantic.field_validator("epoch", "samples_accumulated")
    @classmethod
    def _nonneg(cls, v):
        assert v >= 0
        return v


class TrainingProgressSchema(pydantic.BaseModel):
    progress: Dict[BytesWithPublicKey, Optional[LocalTrainingProgress]]


@dataclass(frozen=True)
class GlobalTrainingProgress:
    global_epoch: int
    samples_accumulated: int
    target_batch_size: int
    num_peers: int
    num_clients: int
    eta_next_epoch: float
    next_fetch_time: float


class ProgressTracker(threading.Thread):
    """Publishes local and aggregates global training progress."""

    def __init__(
        self,
        dht: DHT,
        prefix: str,
        target_batch_size: int,
        *,
        client_mode: bool = False,
        expected_drift_peers: float = 3.0,
        expected_drift_rate: float = 0.2,
        performance_ema_alpha: float = 0.1,
        metadata_expiration: float = 60.0,
        status_loglevel: int = 10,
        min_refresh_period: float = 0.5,
        max_refresh_period: float = 10.0,
        default_refresh_period: float = 3.0,
        private_key=None,
        daemon: bool = True,
        start: bool = True,
    ):
        super().__init__(name=f"{self.__class__.__name__}({prefix})", daemon=daemon)
        self.dht, self.prefix, self.client_mode = dht, prefix, client_mode
        self.training_progress_key = f"{prefix}_progress"
        self.target_batch_size = target_batch_size
        self.expected_drift_peers, self.expected_drift_rate = expected_drift_peers, expected_drift_rate
        self.status_loglevel = status_loglevel
        self.min_refresh_period, self.max_refresh_period = min_refresh_period, max_refresh_period
        self.default_refresh_period = default_refresh_period
        self.metadata_expiration = metadata_expiration
        self.performance_ema = PerformanceEMA(alpha=performance_ema_alpha)

        signature_validator = SignatureValidator(private_key)
        self._local_public_key = signature_validator.local_public_key
        dht.add_validators([SchemaValidator(TrainingProgressSchema, prefix=prefix), signature_validator])

        self.local_progress = self._get_local_progress(local_epoch=0, samples_accumulated=0)
        self.global_progress = GlobalTrainingProgress(
            0, 0, target_batch_size, 1, int(client_mode), float("inf"), get_dht_time() + default_refresh_period
        )
        self.lock_global_progress = threading.Lock()
        self.global_state_updated = threading.Event()
        self.should_report_progress = threading.Event()
        self.fetched_global_progress_this_epoch = threading.Event()
        self.shutdown_triggered = threading.Event()
        self.shutdown_complete = threading.Event()
        if start:
            self.start()

    @property
    def global_epoch(self) -> int:
        return self.global_progress.global_epoch

    @property
    def ready_to_update_epoch(self) -> bool:
        """True if we should begin the next epoch (reference optimizer.py:433-436)."""
        return (
            self.global_epoch > self.local_progress.epoch
            or self.global_progress.samples_accumulated >= self.target_batch_size
            or get_dht_time() >= self.global_progress.eta_next_epoch
        )

    @property
    def estimated_next_update_time(self) -> DHTExpiration:
        if self.ready_to_update_epoch:
            return get_dht_time()
        return self.global_progress.eta_next_epoch

    def _get_local_progress(self, local_epoch: int, samples_accumulated: int) -> LocalTrainingProgress:
        return LocalTrainingProgress(
            peer_id=self.dht.peer_id.to_bytes(),
            epoch=local_epoch,
            samples_accumulated=samples_accumulated,
            samples_per_second=self.performance_ema.samples_per_second,
            time=get_dht_time(),
            client_mode=self.client_mode,
        )

    def report_local_progress(self, local_epoch: int, samples_accumulated: int, update_global_samples: bool = True):
        """Update the local progress and wake the reporter thread
        (reference progress_tracker.py:153-180)."""
        extra_samples = samples_accumulated - self.local_progress.samples_accumulated
        if update_global_samples and local_epoch == self.local_progress.epoch == self.global_progress.global_epoch and extra_samples > 0:
            with self.lock_global_progress:
                # optimistic local bump; the fetcher recomputes the true total
                self.global_progress = GlobalTrainingProgress(
                    self.global_progress.global_epoch,
                    self.global_progress.samples_accumulated + extra_samples,
                    self.global_progress.target_batch_size,
                    self.global_progress.num_peers,
                    self.global_progress.num_clients,
                    self.global_progress.eta_next_epoch,
                    self.global_progress.next_fetch_time,
                )
        if extra_samples > 0:
            self.performance_ema.update(task_size=extra_samples)
        self.local_progress = self._get_local_progress(local_epoch, samples_accumulated)
        self.should_report_progress.set()

    def update_epoch(self, new_epoch: int) -> None:
        """Switch to a new epoch and reset local sample counters."""
        self.local_progress = self._get_local_progress(new_epoch, samples_accumulated=0)
        with self.lock_global_progress:
            if new_epoch > self.global_progress.global_epoch:
                self.global_progress = GlobalTrainingProgress(
                    new_epoch,
                    0,
                    self.target_batch_size,
                    self.global_progress.num_peers,
                    self.global_progress.num_clients,
                    float("inf"),
                    get_dht_time(),
                )
        self.fetched_global_progress_this_epoch.clear()
        self.should_report_progress.set()

    # -------------------------------------------------------------- threads

    def run(self):
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        try:
            loop.run_until_complete(self._run_async())
        finally:
            with contextlib.suppress(Exception):
                loop.close()
            self.shutdown_complete.set()

    async def _run_async(self):
        reporter = asyncio.ensure_future(self._progress_reporter())
        fetcher = asyncio.ensure_future(self._progress_fetcher())
        while not self.shutdown_triggered.is_set():
            await asyncio.sleep(0.1)
        reporter.cancel()
        fetcher.cancel()
        for task in (reporter, fetcher):
            with contextlib.suppress(asyncio.CancelledError):
                await task

    async def _progress_reporter(self):
        """Publish local progress whenever it changes (reference progress_tracker.py:195-233)."""
        last_report_time = -float("inf")
        # announce ourselves immediately: peers must count us BEFORE our first
        # step (waiting for the first report made fresh swarms invisible to
        # each other until training had already begun)
        self.should_report_progress.set()
        while not self.shutdown_triggered.is_set():
            wait_timeout = max(0.0, last_report_time + self.metadata_expiration / 2 - get_dht_time())
            await asyncio.get_event_loop().run_in_executor(
                None, lambda: self.should_report_progress.wait(wait_timeout if wait_timeout > 0 else None)
            )
            if self.shutdown_triggered.is_set():
                break
            self.should_report_progress.clear()
            last_report_time = get_dht_time()
            local_progress = self.local_progress
            try:
                await asyncio.wrap_future(
                    self.dht.store(
                        key=self.training_progress_key,
                        subkey=self._local_public_key,
                        value=local_progress.model_dump(),
                        expiration_time=get_dht_time() + self.metadata_expiration,
                        return_future=True,
                    )
                )
            except Exception as e:
                logger.debug(f"progress report failed: {e!r}")

    async def _progress_fetcher(self):
        """Periodically aggregate the swarm's progress (reference progress_tracker.py:235-273)."""
        while not self.shutdown_triggered.is_set():
            with self.lock_global_progress:
                next_fetch = self.global_progress.next_fetch_time
            await asyncio.sleep(max(0.0, min(next_fetch - get_dht_time(), self.max_refresh_period)))
            if self.shutdown_triggered.is_set():
                break
            try:
                result = await asyncio.wrap_future(
                    self.dht.get(self.training_progress_key, latest=True, return_future=True)
                )
                with self.lock_global_progress:
                    self.global_progress = self._parse_swarm_progress_data(result)
                    self.fetched_global_progress_this_epoch.set()
                    self.global_state_updated.set()
            except Exception as e:
                logger.debug(f"progress fetch failed: {e!r}")
                await asyncio.sleep(self.default_refresh_period)

    def _parse_swarm_progress_data(self, metadata: Optional[ValueWithExpiration]) -> GlobalTrainingProgress:
        """Aggregate per-peer records into global progress (reference progress_tracker.py:275-331)."""
        current_time = get_dht_time()
        local = self.local_progress
        if metadata is None or not isinstance(metadata.value, dict) or len(metadata.value) == 0:
            return GlobalTrainingProgress(
                local.epoch,
                local.samples_accumulated,
                self.target_batch_size,
                num_peers=1,
                num_clients=int(self.client_mode),
                eta_next_epoch=current_time
                + max(0, self.target_batch_size - local.samples_accumulated) / max(local.samples_per_second, 1e-9),
                next_fetch_time=current_time + self.default_refresh_period,
            )
        valid_peer_entries = []
        for _subkey, entry in metadata.value.items():
            if entry.value is None:
                continue
            try:
                valid_peer_entries.append(LocalTrainingProgress.model_validate(entry.value))
            except Exception:
                continue
        num_peers = len(valid_peer_entries)
        num_clients = sum(1 for p in valid_peer_entries if p.client_mode)
        global_epoch = local.epoch
        for p in valid_peer_entries:
            if not p.client_mode:
                global_epoch = max(global_epoch, p.epoch)
        total_samples_accumulated = 0
        total_samples_per_second = 1e-9
        estimated_current_samples = 0.0
        for p in valid_peer_entries:
            total_samples_per_second += p.samples_per_second
            if p.epoch == global_epoch:
                total_samples_accumulated += p.samples_accumulated
                estimated_current_samples += (
                    p.samples_accumulated + max(0.0, current_time - p.time) * p.samples_per_second
                )
        estimated_time_to_next_epoch = max(0.0, self.target_batch_size - estimated_current_samples) / total_samples_per_second

        expected_max_peers = max(num_peers + self.expected_drift_peers, num_peers * (1 + self.expected_drift_rate))
        time_to_next_fetch = float(
            max(
                self.min_refresh_period,
                min(self.max_refresh_period, estimated_time_to_next_epoch * num_peers / max(expected_max_peers, 1e-9)),
            )
        )
        return GlobalTrainingProgress(
            global_epoch,
            total_samples_accumulated,
            target_batch_size=self.target_batch_size,
            num_peers=num_peers,
            num_clients=num_clients,
            eta_next_epoch=current_time + estimated_time_to_next_epoch,
            next_fetch_time=current_time + time_to_next_fetch,
        )

    def shutdown(self, timeout: Optional[float] = None):
        """Stop the tracker and deregister from the DHT."""
        self.shutdown_triggered.set()
        self.should_report_progress.set()
        self.global_state_updated.set()
        self.shutdown_complete.wait(timeout)
        if not self.dht.is_alive:
            return
        try:
            self.dht.store(
                key=self.training_progress_key,
                subkey=self._local_public_key,
                value=None,
                expiration_time=get_dht_time() + self.metadata_expiration,
                return_future=True,
            )
        except Exception:
            pass

    def __enter__(self):
        return self

    def __exit__(self, *args):
        self.shutdown()
