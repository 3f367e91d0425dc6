"""TrainingStateAverager: averages model params (+ optimizer stats), runs the
(optionally delayed) optimizer step, and serves full-state downloads.

Parity target: reference ``hivemind/optim/state_averager.py:37-740``:

* averaged tensors = optimized fp32 params + selected optimizer statistics +
  extra tensors; a schema-compatible averager groups only matching peers;
* ``step(optimizer_step=..., averaging_round=..., increment_epoch=...,
  delay_optimizer_step=..., delay_averaging=...)`` queues work, optionally on
  a background executor so compute overlaps communication (the reference's
  "delayed parameter updates" / DPU mode);
* ``get_current_state``/``load_state_from_peers`` move the complete training
  state (epoch, params, optimizer state dict) between peers.

MI355X-native differences: the "offloaded" optimizer keeps its fp32 master
copy in HBM on the same GPU by default (288 GB leaves room; host offload
remains available with ``offload_device=torch.device('cpu')``), and delayed
steps are plain executor tasks whose GPU work lands on the compute stream --
the RCCL averaging itself runs on a side stream (averaging/rccl.py).
"""

from __future__ import annotations

import threading
import time
from concurrent.futures import ThreadPoolExecutor
from itertools import chain
from typing import Any, Callable, Dict, Iterable, List, Optional, Sequence, Tuple, Union

import torch

from ..averaging import DecentralizedAverager
from ..averaging.control import StepControl
from ..dht import DHT
from ..utils.logging import get_logger
from ..utils.nested import nested_flatten, nested_pack
from ..utils.timed_storage import DHTExpiration, get_dht_time

logger = get_logger(__name__)

Parameters = Iterable[torch.Tensor]
ParamGroups = Iterable[Dict[str, Any]]
TorchOptimizer = torch.optim.Optimizer
OptimizerFactory = Callable[[ParamGroups], TorchOptimizer]
SchedulerFactory = Callable[[TorchOptimizer], Any]


class TrainingStateAverager(DecentralizedAverager):
    def __init__(
        self,
        *,
        dht: DHT,
        optimizer: Union[TorchOptimizer, OptimizerFactory],
        params: Optional[Union[Parameters, ParamGroups]] = None,
        scheduler: Optional[Union[Any, SchedulerFactory]] = None,
        initialize_optimizer: Optional[bool] = None,
        offload_optimizer: bool = False,
        offload_device: Optional[torch.device] = None,
        custom_gradients: bool = False,
        reuse_tensors: Optional[bool] = None,
        delta_rule_averaging: bool = False,
        performance_ema_alpha: float = 0.1,
        average_opt_statistics: Sequence[str] = (),
        extra_tensors: Sequence[torch.Tensor] = (),
        status_loglevel: int = 20,
        **kwargs,
    ):
        average_opt_statistics = tuple(average_opt_statistics)
        assert custom_gradients or not offload_optimizer or True
        self.status_loglevel = status_loglevel
        self.offload_optimizer, self.custom_gradients = offload_optimizer, custom_gradients
        self.delta_rule_averaging = delta_rule_averaging
        # in one-process-per-GPU mode the averager can average the live tensors
        # directly (no IPC copies); delta-rule needs its own buffers
        self.reuse_tensors = reuse_tensors if reuse_tensors is not None else not delta_rule_averaging
        self.opt_keys_for_averaging = average_opt_statistics

        param_groups, main_parameters, parameter_names = self._check_params(optimizer, params)
        self.main_parameters, self.parameter_names = main_parameters, parameter_names
        self._averaged_parameters: List[torch.Tensor] = []
        self.optimizer, self.scheduler = self._init_components(
            param_groups, optimizer, scheduler, initialize_optimizer, offload_device
        )
        self.local_epoch = 0
        self.delta_buffers: Optional[List[torch.Tensor]] = None

        self.step_executor = ThreadPoolExecutor(max_workers=1, thread_name_prefix="state_averager")
        self.finished_optimizer_step = threading.Event()
        self.finished_averaging_round = threading.Event()
        self.finished_optimizer_step.set()
        self.finished_averaging_round.set()
        self.pending_updates: set = set()
        self._step_lock = threading.Lock()

        self.extra_tensors = tuple(extra_tensors)
        averaged_tensors = self._make_averaged_tensors()
        super().__init__(averaged_tensors=averaged_tensors, dht=dht, **kwargs)

    # ---------------------------------------------------------------- set-up

    @staticmethod
    def _check_params(optimizer, params) -> Tuple[ParamGroups, List[torch.Tensor], List[str]]:
        if params is None:
            assert hasattr(optimizer, "param_groups"), "optimizer instance or explicit params required"
            param_groups = optimizer.param_groups
        else:
            params = list(params)
            if params and isinstance(params[0], dict):
                param_groups = params
            else:
                param_groups = [{"params": params}]
        main_parameters = []
        for group in param_groups:
            main_parameters.extend(group["params"])
        parameter_names = [f"param_{i}" for i in range(len(main_parameters))]
        return param_groups, main_parameters, parameter_names

    def _init_components(self, param_groups, optimizer_or_factory, scheduler_or_factory, initialize_optimizer, offload_device):
        if self.offload_optimizer:
            device = offload_device if offload_device is not None else self.main_parameters[0].device
            offloaded_param_groups = []
            offloaded_params = []
            for group in param_groups:
                new_group = {k: v for k, v in group.items() if k != "params"}
                new_group["params"] = [
                    torch.nn.Parameter(p.detach().to(device=device, dtype=torch.float32, copy=True), requires_grad=True)
                    for p in group["params"]
                ]
                offloaded_params.extend(new_group["params"])
                offloaded_param_groups.append(new_group)
            self.optimized_parameters = offloaded_params
            opt_param_groups = offloaded_param_groups
        else:
            self.optimized_parameters = list(self.main_parameters)
            opt_param_groups = param_groups

        if isinstance(optimizer_or_factory, TorchOptimizer):
            optimizer = optimizer_or_factory
            assert not self.offload_optimizer, "offload_optimizer requires an optimizer factory, not an instance"
        else:
            optimizer = optimizer_or_factory(opt_param_groups)

        if initialize_optimizer is None:
            initialize_optimizer = not any(isinstance(v, torch.Tensor) for v in nested_flatten(optimizer.state_dict()))
        if initialize_optimizer:
            initialize_optimizer_state_(optimizer)

        if scheduler_or_factory is None or not callable(scheduler_or_factory):
            scheduler = scheduler_or_factory
        else:
            scheduler = scheduler_or_factory(optimizer)
        return optimizer, scheduler

    def _local_tensors(self) -> List[torch.Tensor]:
        """Everything that gets averaged, in schema order: params, opt stats, extras."""
        tensors = [p.detach() for p in self.optimized_parameters]
        for stat_name in self.opt_keys_for_averaging:
            for param in self.optimized_parameters:
                state = self.optimizer.state.get(param, {})
                assert stat_name in state, f"optimizer state has no '{stat_name}' for some parameter"
                tensors.append(state[stat_name])
        tensors.extend(self.extra_tensors)
        return tensors

    def _make_averaged_tensors(self) -> List[torch.Tensor]:
        local = self._local_tensors()
        if self.reuse_tensors:
            return local
        return [t.detach().clone() for t in local]

    # ------------------------------------------------------------------ step

    def schedule_step(self, scheduled_time: Optional[DHTExpiration] = None, **kwargs) -> StepControl:
        """Begin matchmaking for a parameter-averaging round ahead of time; the
        all-reduce runs when the control is triggered (reference optimizer.py:569-592)."""
        kwargs.pop("require_trigger", None)
        kwargs.pop("wait", None)
        return DecentralizedAverager.step(self, scheduled_time=scheduled_time, wait=False, require_trigger=True, **kwargs)

    def step(
        self,
        wait_for_delayed_updates: Optional[bool] = None,
        apply_delayed_updates: bool = True,
        increment_epoch: bool = False,
        optimizer_step: bool = False,
        zero_grad: bool = False,
        averaging_round: bool = False,
        delay_optimizer_step: bool = False,
        delay_averaging: Optional[bool] = None,
        averaging_control: Optional[StepControl] = None,
        wait_for_trigger: Optional[Callable[[], Any]] = None,
        grad_scaler: Optional[Any] = None,
        averaging_opts: Optional[Dict[str, Any]] = None,
        timeout: Optional[float] = None,
    ):
        """Run requested operations, inline or delayed (reference state_averager.py:329-476)."""
        if delay_averaging is None:
            delay_averaging = delay_optimizer_step
        if wait_for_delayed_updates is None:
            wait_for_delayed_updates = optimizer_step or zero_grad or averaging_round
        assert not delay_optimizer_step or delay_averaging or not averaging_round, (
            "a delayed optimizer step cannot be combined with a non-delayed averaging round"
        )
        if delay_optimizer_step:
            assert self.offload_optimizer, "delayed optimizer steps require offload_optimizer=True"

        # wait for previously delayed work
        if wait_for_delayed_updates:
            for future in list(self.pending_updates):
                try:
                    future.result(timeout)
                except Exception:
                    pass

        if apply_delayed_updates:
            self._apply_optimizer_parameters_()

        output = None
        if increment_epoch:
            self.local_epoch += 1

        if averaging_round:
            # reserve the RCCL launch slot HERE, in the foreground: every rank
            # schedules its rounds in the same program order, so tickets give a
            # consistent cross-rank launch order even when the round itself
            # runs on the background executor
            from ..averaging.rccl import issue_collective_ticket, release_collective_ticket

            ticket = issue_collective_ticket()
            if averaging_control is not None:
                averaging_control.rccl_ticket = ticket
                if averaging_control.done():  # round already over: don't stall later rounds
                    release_collective_ticket(ticket)
            else:
                averaging_opts = dict(averaging_opts or {})
                averaging_opts["rccl_ticket"] = ticket

        if optimizer_step or zero_grad or averaging_round:
            task = lambda: self._do(
                wait_for_trigger, optimizer_step, zero_grad, averaging_round, averaging_control, grad_scaler,
                **(averaging_opts or {}),
            )
            if delay_optimizer_step or (delay_averaging and averaging_round):
                self.finished_optimizer_step.clear()
                if averaging_round:
                    self.finished_averaging_round.clear()
                future = self.step_executor.submit(task)
                self.pending_updates.add(future)
                future.add_done_callback(lambda f: self.pending_updates.discard(f))
            else:
                output = task()

        if increment_epoch and self.scheduler is not None:
            self._update_scheduler()
        return output

    def _do(
        self,
        wait_for_trigger: Optional[Callable[[], Any]],
        optimizer_step: bool,
        zero_grad: bool,
        averaging_round: bool,
        averaging_control: Optional[StepControl],
        grad_scaler: Optional[Any],
        **averaging_opts,
    ):
        """The actual sequence: [trigger] -> optimizer step -> zero grad ->
        averaging round -> apply results (reference state_averager.py:478-574)."""
        gathered = None
        try:
            if wait_for_trigger is not None:
                wait_for_trigger()
            if optimizer_step:
                with self.lock_averaged_tensors if self.reuse_tensors else _null_lock():
                    if not self.custom_gradients:
                        self._load_gradients_into_optimizer_()
                    if grad_scaler is not None:
                        with grad_scaler.running_global_step():
                            assert grad_scaler.step(self.optimizer)
                    else:
                        self.optimizer.step()
                self.finished_optimizer_step.set()
            if zero_grad:
                self.optimizer.zero_grad(set_to_none=False)
                if self.offload_optimizer:
                    for param in self.main_parameters:
                        # snapshot the ref: the training thread may concurrently
                        # set param.grad = None (its own zero_grad); zeroing a
                        # tensor whose last reference vanished mid-call segfaults
                        grad = param.grad
                        if grad is not None:
                            grad.zero_()
            if averaging_round:
                if self.delta_rule_averaging:
                    with torch.no_grad():
                        self.delta_buffers = [t.detach().clone() for t in self._local_tensors()]
                if not self.reuse_tensors:
                    self._load_local_tensors_into_averager_()
                try:
                    if averaging_control is None:
                        gathered = super().step(**averaging_opts)
                    else:
                        averaging_control.allow_allreduce()
                        gathered = averaging_control.result(averaging_opts.get("timeout"))
                    self._apply_averaging_results_()
                except BaseException as e:
                    logger.warning(f"averaging round failed: {e!r}")
                self.finished_averaging_round.set()
            if optimizer_step or averaging_round:
                self._apply_optimizer_parameters_()
            return gathered
        except Exception:
            logger.exception("state averager step failed")
            self.finished_optimizer_step.set()
            self.finished_averaging_round.set()
            raise

    @torch.no_grad()
    def _load_gradients_into_optimizer_(self):
        """Copy model gradients into the (possibly offloaded) optimizer params."""
        if self.offload_optimizer:
            for main_param, opt_param in zip(self.main_parameters, self.optimized_parameters):
                grad = main_param.grad  # ref snapshot (see _do zero_grad note)
                if grad is not None:
                    if opt_param.grad is None:
                        opt_param.grad = torch.zeros_like(opt_param)
                    opt_param.grad.copy_(grad.to(opt_param.device, opt_param.dtype), non_blocking=True)

    @torch.no_grad()
    def _apply_optimizer_parameters_(self):
        """Copy optimized params back into the model (reference state_averager.py:589-595)."""
        if self.offload_optimizer:
            for main_param, opt_param in zip(self.main_parameters, self.optimized_parameters):
                # write through .data: with delayed (DPU) steps this runs while
                # the NEXT microbatch's autograd graph may still reference the
                # parameter -- .data bypasses the version counter, so backward
                # proceeds with slightly stale weights instead of crashing
                # (reference state_averager applies delayed updates the same way)
                main_param.data.copy_(opt_param.detach().to(main_param.device, main_param.dtype), non_blocking=True)

    @torch.no_grad()
    def _load_local_tensors_into_averager_(self):
        with self.get_tensors() as averaged:
            for local, avg in zip(self._local_tensors(), averaged):
                avg.copy_(local.to(avg.device, avg.dtype), non_blocking=True)

    @torch.no_grad()
    def _apply_averaging_results_(self):
        """Pull averaged values back into the live training state."""
        if self.reuse_tensors and not self.delta_rule_averaging:
            return  # averaging already happened in place
        with self.get_tensors() as averaged:
            local = self._local_tensors()
            if self.delta_rule_averaging and self.delta_buffers is not None:
                # new_local = local + (averaged - old_local): tolerates local
                # progress made while the (delayed) round was running
                for loc, avg, old in zip(local, averaged, self.delta_buffers):
                    loc.data.add_(avg.to(loc.device, loc.dtype) - old.to(loc.device, loc.dtype))
                self.delta_buffers = None
            else:
                # .data: see _apply_optimizer_parameters_ -- delayed rounds may
                # overlap an in-flight autograd graph
                for loc, avg in zip(local, averaged):
                    loc.data.copy_(avg.to(loc.device, loc.dtype), non_blocking=True)

    def _update_scheduler(self):
        """Advance the LR scheduler to the current epoch (epoch-based schedule)."""
        if self.scheduler is not None:
            while getattr(self.scheduler, "_step_count", 0) <= self.local_epoch:
                self.scheduler.step()

    # ------------------------------------------------------------ state sync

    @property
    def state_sharing_priority(self) -> float:
        return float(self.local_epoch) if self.allow_state_sharing else float("-inf")

    def get_current_state(self) -> Tuple[Any, Sequence[torch.Tensor]]:
        """(metadata, tensors) for newcomers (reference state_averager.py:627-656)."""
        with torch.no_grad():
            optimized_parameters = [p.detach().cpu() for p in self.optimized_parameters]
            parameter_infos = [
                {"shape": list(p.shape), "dtype": str(p.dtype)} for p in optimized_parameters
            ]
            extra_tensors = [t.detach().cpu() for t in self.extra_tensors]
            optimizer_metadata, optimizer_tensors = dump_optimizer_state(self.optimizer)
        metadata = dict(epoch=self.local_epoch, group_bits=self.get_group_bits(), optimizer_metadata=optimizer_metadata)
        all_tensors = list(chain(optimized_parameters, extra_tensors, optimizer_tensors))
        return metadata, all_tensors

    def load_state_from_peers(self, wait: bool = True, timeout=None, apply: bool = True):
        """Training peers install the downloaded state by default (the base
        averager only returns it -- reference TrainingStateAverager)."""
        return super().load_state_from_peers(wait=wait, timeout=timeout, apply=apply)

    def load_state(self, metadata: Any, tensors: Sequence[torch.Tensor]):
        """Restore a downloaded state (reference state_averager.py:658-704)."""
        if metadata is None or not isinstance(metadata, dict):
            logger.warning("donor state has no metadata; ignoring")
            return
        num_params = len(self.optimized_parameters)
        num_extras = len(self.extra_tensors)
        params = tensors[:num_params]
        extras = tensors[num_params : num_params + num_extras]
        opt_tensors = tensors[num_params + num_extras :]
        with torch.no_grad():
            for local, loaded in zip(self.optimized_parameters, params):
                local.detach().copy_(loaded.to(local.device, local.dtype))
            for local, loaded in zip(self.extra_tensors, extras):
                local.detach().copy_(loaded.to(local.device, local.dtype))
            try:
                load_optimizer_state(self.optimizer, metadata.get("optimizer_metadata"), opt_tensors)
            except Exception as e:
                logger.warning(f"failed to restore optimizer state: {e!r}")
            self._apply_optimizer_parameters_()
            if not self.reuse_tensors:
                self._load_local_tensors_into_averager_()
        self.local_epoch = int(metadata.get("epoch", self.local_epoch))
        self._update_scheduler()

    def shutdown(self):
        self.step_executor.shutdown(wait=False)
        super().shutdown()


def initialize_optimizer_state_(optimizer: TorchOptimizer):
    """Run a zero-gradient step so optimizer statistics exist
    (reference state_averager.py:707-716)."""
    flat_params = [param for group in optimizer.param_groups for param in group["params"]]
    old_grads = []
    for param in flat_params:
        old_grads.append(param.grad)
        param.grad = torch.zeros_like(param)
    optimizer.step()
    for param, old_grad in zip(flat_params, old_grads):
        param.grad = old_grad


def dump_optimizer_state(optimizer: TorchOptimizer) -> Tuple[list, List[torch.Tensor]]:
    """Split an optimizer state dict into (msgpack-safe metadata, tensor list)
    (reference state_averager.py:718-729)."""
    with torch.no_grad():
        flat_metadata, flat_tensors = [], []
        for elem in nested_flatten(optimizer.state_dict()):
            if isinstance(elem, torch.Tensor):
                flat_metadata.append(dict(type="tensor", index=len(flat_tensors)))
                flat_tensors.append(elem.detach().cpu())
            else:
                flat_metadata.append(dict(type="value", value=elem))
        return flat_metadata, flat_tensors


def load_optimizer_state(optimizer: TorchOptimizer, flat_metadata: list, flat_tensors: Sequence[torch.Tensor]):
    """Inverse of dump_optimizer_state (reference state_averager.py:731-740)."""
    if flat_metadata is None:
        return
    flat_optimizer_state = []
    for elem in flat_metadata:
        if elem.get("type") == "tensor" and isinstance(elem.get("index"), int):
            flat_optimizer_state.append(flat_tensors[elem["index"]])
        elif elem.get("type") == "value" and "value" in elem:
            flat_optimizer_state.append(elem["value"])
    return optimizer.load_state_dict(nested_pack(flat_optimizer_state, structure=optimizer.state_dict()))


class _null_lock:
    def __enter__(self):
        return self

    def __exit__(self, *args):
        return False
