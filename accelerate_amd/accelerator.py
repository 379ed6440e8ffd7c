"""The public orchestrator (reference: accelerator.py).

`Accelerator` keeps the reference's API contract — ``prepare()``,
``backward()``, ``accumulate()``, ``clip_grad_norm_()``, ``gather()``,
``save_state()/load_state()`` — while routing everything MI355X-native:

- models wrap into our RCCL/xGMI bucketed reducer
  (parallel/ddp.py) instead of torch DDP;
- fp16 uses our GradScaler + CDNA4 unscale/non-finite kernel;
- clip_grad_norm_ uses the CDNA4 multi-tensor L2-norm kernel;
- bf16 runs through torch.autocast on ROCm (HIP autocast dispatch).
"""

import contextlib
import functools
import os
import shutil
import warnings
from collections import OrderedDict
from contextlib import contextmanager
from typing import Any, Callable, List, Optional, Union

import torch
import torch.utils.hooks as hooks

from .checkpointing import load_accelerator_state, load_custom_state, save_accelerator_state, save_custom_state
from .data_loader import DataLoaderDispatcher, prepare_data_loader
from .logging import get_logger
from .optimizer import AcceleratedOptimizer
from .parallel.ddp import DistributedDataParallelEngine
from .scheduler import AcceleratedScheduler
from .state import AcceleratorState, GradientState, PartialState
from .tracking import GeneralTracker, filter_trackers
from .utils.dataclasses import (
    AutocastKwargs,
    DistributedDataParallelKwargs,
    DistributedType,
    FP8RecipeKwargs,
    FullyShardedDataParallelPlugin,
    GradientAccumulationPlugin,
    GradScalerKwargs,
    InitProcessGroupKwargs,
    ProfileKwargs,
    ProjectConfiguration,
    RNGType,
)
from .utils.operations import (
    broadcast,
    broadcast_object_list,
    convert_outputs_to_fp32,
    gather,
    gather_object,
    pad_across_processes,
    recursively_apply,
    reduce,
    send_to_device,
)
from .utils.environment import parse_flag_from_env
from .utils.other import extract_model_from_parallel, wait_for_everyone

logger = get_logger(__name__)


class Accelerator:
    """(reference: accelerator.py:184)"""

    def __init__(
        self,
        device_placement: bool = True,
        split_batches: bool = False,
        mixed_precision: Union[str, None] = None,
        gradient_accumulation_steps: int = 1,
        cpu: bool = False,
        dataloader_config=None,
        log_with=None,
        project_dir: Optional[str] = None,
        project_config: Optional[ProjectConfiguration] = None,
        gradient_accumulation_plugin: Optional[GradientAccumulationPlugin] = None,
        step_scheduler_with_optimizer: bool = True,
        kwargs_handlers: Optional[List[Any]] = None,
        fsdp_plugin: Optional[FullyShardedDataParallelPlugin] = None,
        rng_types: Optional[List[Union[str, RNGType]]] = None,
        parallelism_config=None,
        dynamo_plugin=None,
        dynamo_backend: Union[str, None] = None,
        dispatch_batches: Optional[bool] = None,
        even_batches: bool = True,
        use_seedable_sampler: bool = False,
        non_blocking: bool = True,
    ):
        self.trackers = []
        if project_config is not None:
            self.project_configuration = project_config
        else:
            self.project_configuration = ProjectConfiguration(project_dir=project_dir)
        if project_dir is not None and self.project_configuration.project_dir is None:
            self.project_configuration.set_directories(project_dir)

        # kwargs handlers
        self.ddp_handler = None
        self.scaler_handler = None
        self.init_handler = None
        self.autocast_handler = None
        self.fp8_recipe_handler = None
        self.profile_handler = None
        if kwargs_handlers is not None:
            for handler in kwargs_handlers:
                if isinstance(handler, DistributedDataParallelKwargs):
                    self.ddp_handler = handler
                elif isinstance(handler, GradScalerKwargs):
                    self.scaler_handler = handler
                elif isinstance(handler, InitProcessGroupKwargs):
                    self.init_handler = handler
                elif isinstance(handler, AutocastKwargs):
                    self.autocast_handler = handler
                elif isinstance(handler, FP8RecipeKwargs):
                    self.fp8_recipe_handler = handler
                elif isinstance(handler, ProfileKwargs):
                    self.profile_handler = handler

        kwargs = self.init_handler.to_kwargs() if self.init_handler is not None else {}
        self.state = AcceleratorState(
            mixed_precision=mixed_precision,
            cpu=cpu,
            fsdp_plugin=fsdp_plugin,
            _from_accelerator=True,
            **kwargs,
        )

        self.device_placement = device_placement
        self.split_batches = split_batches
        self.dispatch_batches = dispatch_batches
        self.even_batches = even_batches
        self.use_seedable_sampler = use_seedable_sampler
        self.non_blocking = non_blocking
        self.use_stateful_dataloader = False
        self.data_seed = None
        # multi-dimensional parallelism descriptor (reference accelerator.py:301);
        # with no explicit config the PARALLELISM_CONFIG_* env plane set by the
        # launcher takes over (reference: parallelism_config.py:274-341)
        if parallelism_config is None and parse_flag_from_env("ACCELERATE_USE_PARALLELISM_CONFIG", False):
            from .parallelism_config import ParallelismConfig

            parallelism_config = ParallelismConfig()
        self.parallelism_config = parallelism_config
        if parallelism_config is not None and self.state.distributed_type != DistributedType.NO:
            parallelism_config.validate(self.num_processes)
        # torch.compile plane (reference: dataclasses.py:1033 TorchDynamoPlugin)
        from .utils.dataclasses import TorchDynamoPlugin

        if dynamo_plugin is None:
            dynamo_plugin = TorchDynamoPlugin(backend=dynamo_backend) if dynamo_backend else TorchDynamoPlugin()
        self.dynamo_plugin = dynamo_plugin
        if dataloader_config is not None:  # bundled options take precedence
            self.split_batches = dataloader_config.split_batches
            self.dispatch_batches = dataloader_config.dispatch_batches
            self.even_batches = dataloader_config.even_batches
            self.use_seedable_sampler = dataloader_config.use_seedable_sampler
            self.non_blocking = dataloader_config.non_blocking
            self.use_stateful_dataloader = getattr(dataloader_config, "use_stateful_dataloader", False)
            self.data_seed = getattr(dataloader_config, "data_seed", None)
        self.step_scheduler_with_optimizer = step_scheduler_with_optimizer

        # mixed precision
        self.scaler = None
        self.native_amp = False
        if self.state.mixed_precision == "fp16":
            self.native_amp = True
            scaler_kwargs = self.scaler_handler.to_kwargs() if self.scaler_handler is not None else {}
            if self.device.type == "cuda":
                from .ops.grad_scaler import GradScaler

                self.scaler = GradScaler(**scaler_kwargs)
            else:
                self.scaler = torch.amp.GradScaler("cpu", **scaler_kwargs)
        elif self.state.mixed_precision in ("bf16", "fp8"):
            self.native_amp = True

        # gradient accumulation
        if gradient_accumulation_plugin is None:
            ga_steps = int(os.environ.get("ACCELERATE_GRADIENT_ACCUMULATION_STEPS", gradient_accumulation_steps))
            gradient_accumulation_plugin = GradientAccumulationPlugin(num_steps=ga_steps)
        elif gradient_accumulation_steps != 1:
            raise ValueError("Pass either gradient_accumulation_steps or gradient_accumulation_plugin, not both.")
        self.gradient_state = GradientState(gradient_accumulation_plugin=gradient_accumulation_plugin)

        self.log_with = filter_trackers(log_with, self.logging_dir)
        self.rng_types = rng_types
        if self.rng_types is None:
            self.rng_types = ["generator"]

        # internal bookkeeping
        self._optimizers = []
        self._models = []
        self._schedulers = []
        self._dataloaders = []
        self._custom_objects = []
        self._save_model_state_pre_hook = OrderedDict()
        self._load_model_state_pre_hook = OrderedDict()
        self.step = 0
        self.flag_tensor = None

    # ------------------------------------------------------------------
    # properties
    # ------------------------------------------------------------------

    @property
    def distributed_type(self):
        return self.state.distributed_type

    @property
    def num_processes(self):
        return self.state.num_processes

    @property
    def process_index(self):
        return self.state.process_index

    @property
    def local_process_index(self):
        return self.state.local_process_index

    @property
    def device(self):
        return self.state.device

    @property
    def project_dir(self):
        return self.project_configuration.project_dir

    @property
    def logging_dir(self):
        return self.project_configuration.logging_dir

    @property
    def save_iteration(self):
        return self.project_configuration.iteration

    @property
    def is_main_process(self):
        return self.state.is_main_process

    @property
    def is_local_main_process(self):
        return self.state.is_local_main_process

    @property
    def is_last_process(self):
        return self.process_index == self.num_processes - 1

    @property
    def use_distributed(self):
        return self.state.use_distributed

    @property
    def mixed_precision(self):
        return self.state.mixed_precision

    @property
    def sync_gradients(self):
        return self.gradient_state.sync_gradients

    @sync_gradients.setter
    def sync_gradients(self, sync_gradients):
        self.gradient_state.sync_gradients = sync_gradients

    @property
    def gradient_accumulation_steps(self):
        return self.gradient_state.num_steps

    @gradient_accumulation_steps.setter
    def gradient_accumulation_steps(self, gradient_accumulation_steps):
        self.gradient_state.plugin_kwargs.update({"num_steps": gradient_accumulation_steps})

    # rank-gated decorators --------------------------------------------------

    def on_main_process(self, function: Callable[..., Any] = None):
        if function is None:
            return functools.partial(self.on_main_process)

        def _inner(*args, **kwargs):
            return PartialState().on_main_process(function)(*args, **kwargs)

        return _inner

    def on_local_main_process(self, function: Callable[..., Any] = None):
        if function is None:
            return functools.partial(self.on_local_main_process)

        def _inner(*args, **kwargs):
            return PartialState().on_local_main_process(function)(*args, **kwargs)

        return _inner

    def on_last_process(self, function: Callable[..., Any]):
        def _inner(*args, **kwargs):
            return PartialState().on_last_process(function)(*args, **kwargs)

        return _inner

    def on_process(self, function: Callable[..., Any] = None, process_index: int = None):
        if function is None:
            return functools.partial(self.on_process, process_index=process_index)

        def _inner(*args, **kwargs):
            return PartialState().on_process(function, process_index)(*args, **kwargs)

        return _inner

    def on_local_process(self, function: Callable[..., Any] = None, local_process_index: int = None):
        if function is None:
            return functools.partial(self.on_local_process, local_process_index=local_process_index)

        def _inner(*args, **kwargs):
            return PartialState().on_local_process(function, local_process_index)(*args, **kwargs)

        return _inner

    @contextmanager
    def main_process_first(self):
        with self.state.main_process_first():
            yield

    @contextmanager
    def local_main_process_first(self):
        with self.state.local_main_process_first():
            yield

    @contextmanager
    def split_between_processes(self, inputs, apply_padding: bool = False):
        with PartialState().split_between_processes(inputs, apply_padding=apply_padding) as result:
            yield result

    # ------------------------------------------------------------------
    # no_sync / accumulate
    # ------------------------------------------------------------------

    @contextmanager
    def no_sync(self, model):
        """Disable gradient sync inside the context (reference: accelerator.py:1132)."""
        from .parallel.fsdp import ShardedModel

        context = contextlib.nullcontext
        if isinstance(model, (DistributedDataParallelEngine, ShardedModel)):
            context = model.no_sync
        with context():
            yield

    def _do_sync(self):
        if self.gradient_state.sync_with_dataloader and self.gradient_state.end_of_dataloader:
            self.step = 0
            self.gradient_state._set_sync_gradients(True)
        else:
            self.step += 1
            self.gradient_state._set_sync_gradients((self.step % self.gradient_state.num_steps) == 0)

    @contextmanager
    def accumulate(self, *models):
        """Gradient-accumulation window (reference: accelerator.py:1255)."""
        self._do_sync()
        allow_gradient_sync = self.sync_gradients or (
            self.use_distributed and self.gradient_state.plugin_kwargs.get("sync_each_batch", False)
        )
        with contextlib.ExitStack() as cm_stack:
            for m in models:
                cm_stack.enter_context(
                    contextlib.nullcontext() if allow_gradient_sync else self.no_sync(m)
                )
            yield

    @contextmanager
    def join_uneven_inputs(self, joinables, even_batches=None):
        """Training on UNEVEN per-rank inputs (reference: accelerator.py:1300
        wrapping torch.distributed.algorithms.Join).

        Protocol (our reducer's equivalent of the Join algorithm): while the
        context is active every engine forward opens a liveness collective
        round ``[n_live, n_live_syncing]``; a rank that exhausts its data
        reaches the context exit and enters a drain loop that keeps
        answering those rounds with zeros and SHADOWS each live step's
        collectives (buffer broadcast + zero-contribution bucket
        all-reduces, in the recorded launch order) until every rank has
        joined — so uneven iteration counts cannot desynchronize or
        deadlock the reducer. ``even_batches`` may additionally be
        overridden on prepared dataloaders for the scope of the context.
        """
        if even_batches is None:
            even_batches = self.even_batches
        engines = []
        for j in joinables:
            if isinstance(j, DistributedDataParallelEngine):
                engines.append(j)
            else:
                for m in self._models:
                    if isinstance(m, DistributedDataParallelEngine) and m.module is j:
                        engines.append(m)
        iterable_dl_seen = False
        dl_even_batches_values = []
        for dl_idx, dl in enumerate(self._dataloaders):
            if isinstance(dl, DataLoaderDispatcher):
                iterable_dl_seen = True
                continue
            if hasattr(dl, "batch_sampler") and hasattr(dl.batch_sampler, "even_batches"):
                dl_even_batches_values.append((dl_idx, dl.batch_sampler.even_batches))
                dl.batch_sampler.even_batches = even_batches
        if iterable_dl_seen:
            warnings.warn("Overriding even_batches is only supported for map-style datasets; ignored for dispatchers.")
        if self.use_distributed:
            for e in engines:
                e._join_active = True
        try:
            yield
        finally:
            if self.use_distributed:
                for e in engines:
                    # this rank is done: shadow the stragglers' steps
                    e.join_drain()
                    e._join_active = False
            for dl_idx, value in dl_even_batches_values:
                self._dataloaders[dl_idx].batch_sampler.even_batches = value

    # ------------------------------------------------------------------
    # prepare
    # ------------------------------------------------------------------

    def _prepare_one(self, obj, first_pass=False, device_placement=None):
        # First pass of preparation: DataLoader, model, optimizer
        # Second pass: scheduler (needs prepared optimizers)
        if first_pass:
            if isinstance(obj, torch.utils.data.DataLoader):
                return self.prepare_data_loader(obj, device_placement=device_placement)
            elif isinstance(obj, torch.nn.Module):
                return self.prepare_model(obj, device_placement=device_placement)
            elif isinstance(obj, torch.optim.Optimizer):
                return self.prepare_optimizer(obj, device_placement=device_placement)
        elif isinstance(obj, torch.optim.lr_scheduler.LRScheduler) or (
            hasattr(obj, "optimizer") and hasattr(obj, "step") and not isinstance(obj, AcceleratedOptimizer)
        ):
            return self.prepare_scheduler(obj)
        return obj

    def prepare(self, *args, device_placement=None):
        """Prepare all objects for the current distributed world
        (reference: accelerator.py:1414)."""
        if device_placement is None:
            device_placement = [None for _ in args]
        elif len(device_placement) != len(args):
            raise ValueError(f"`device_placement` should be a list with {len(args)} elements (got {len(device_placement)}).")

        for obj in args:
            if (
                isinstance(obj, torch.nn.Module)
                and self.verify_device_map(obj)
                and self.distributed_type != DistributedType.NO
            ):
                raise ValueError(
                    "You can't train a model that has been loaded with `device_map='auto'` in any distributed mode."
                    " Please rerun your script specifying `--num_processes=1` or by launching with `python {{myscript.py}}`."
                )

        # multi-dimensional worlds (TP/CP x DP) route through the
        # ParallelismConfig groups (reference: accelerator.py:1531-1560
        # _prepare_tp/_prepare_cp): shard each model by its tp_plan, register
        # the sequence-parallel context, then fall through so DP wrapping and
        # dataloader sharding happen on the dp dimension only.
        pc = self.parallelism_config
        if pc is not None and (pc.tp_size > 1 or pc.cp_size > 1):
            if not pc._groups:
                pc.build_groups()
            if pc.tp_size > 1:
                from .parallel.tp import apply_tp_plan

                args = tuple(
                    apply_tp_plan(obj, group=pc._groups["tp"]) if isinstance(obj, torch.nn.Module) else obj
                    for obj in args
                )
                # re-point already-constructed optimizers at the sharded
                # params (reference: accelerator.py:1647-1654 optimizer remap)
                swap = {}
                for obj in args:
                    if isinstance(obj, torch.nn.Module):
                        swap.update(getattr(obj, "_tp_param_swap", {}))
                for obj in args:
                    if isinstance(obj, torch.optim.Optimizer):
                        if len(obj.state) > 0:
                            raise RuntimeError(
                                "prepare() with tensor parallelism must run before the "
                                "optimizer takes its first step (param swap would orphan state)"
                            )
                        for group in obj.param_groups:
                            group["params"] = [swap.get(id(p), p) for p in group["params"]]
                        # moments now live on tp SHARDS: checkpointing must
                        # write one optimizer file per rank (like FSDP), not
                        # main-rank-only (which would load rank 0's shard
                        # moments everywhere)
                        obj._sharded = True
            # cp activation is scoped to `maybe_context_parallel` (the
            # per-step ctx manager, reference accelerator.py:4111) so
            # unprepared/eval models are never caught by the ambient context

        if self.distributed_type == DistributedType.FSDP:
            from .parallel.fsdp import fsdp_prepare

            return fsdp_prepare(self, args, device_placement)

        result = tuple(
            self._prepare_one(obj, first_pass=True, device_placement=d) for obj, d in zip(args, device_placement)
        )
        result = tuple(self._prepare_one(obj, device_placement=d) for obj, d in zip(result, device_placement))
        if len(result) == 1:
            return result[0]
        return result

    def prepare_model(self, model: torch.nn.Module, device_placement: bool = None, evaluation_mode: bool = False):
        """(reference: accelerator.py:1769)"""
        if getattr(model, "_is_accelerate_prepared", False):
            # double-wrap protection (reference test_accelerator.py:469)
            if model not in self._models:
                self._models.append(model)
            return model
        if device_placement is None:
            device_placement = self.device_placement

        # mixed-precision forward wrap: autocast + fp32 outputs
        if self.native_amp:
            model._original_forward = model.forward
            make_ctx = functools.partial(
                get_mixed_precision_context_manager,
                self.native_amp,
                self.state.mixed_precision,
                self.device,
                self.autocast_handler,
            )
            new_forward = autocast_context_wrap(make_ctx, model.forward)
            model.forward = convert_outputs_to_fp32(new_forward)

        if self.mixed_precision == "fp8":
            from .ops.fp8 import convert_linears_to_fp8

            pc_fp8 = self.parallelism_config
            if pc_fp8 is not None and pc_fp8.tp_size > 1:
                logger.warning(
                    "fp8 conversion targets plain nn.Linear modules: tensor-parallel-sharded "
                    "projections stay in bf16 (fp8 column/row-parallel linears are not yet "
                    "implemented). Expect fp8 gains only on unsharded layers."
                )
            model = convert_linears_to_fp8(model, recipe=self.fp8_recipe_handler)

        if device_placement and not self.verify_device_map(model):
            model = model.to(self.device)

        pc = self.parallelism_config
        if (
            pc is not None
            and pc.dp_shard_size > 1
            and getattr(self.state, "fsdp_plugin", None) is None
        ):
            # mirror the reference's requirement (state.py:995-1007):
            # dp_shard is the FSDP dimension — silently replicating here
            # would "work" while delivering none of the promised sharding
            raise ValueError(
                "ParallelismConfig.dp_shard_size > 1 requires an FSDP plugin "
                "(fsdp_plugin=FullyShardedDataParallelPlugin(...) or the FSDP_* env plane); "
                "use dp_replicate_size for replicated data parallelism."
            )
        multi_dim = pc is not None and (pc.tp_size > 1 or pc.cp_size > 1) and pc._groups
        if not evaluation_mode and self.use_distributed and self.distributed_type in (
            DistributedType.MULTI_GPU,
            DistributedType.MULTI_CPU,
        ):
            ddp_kwargs = self.ddp_handler.to_dict() if self.ddp_handler is not None else {}
            if multi_dim:
                # gradient averaging happens on the dp x cp domain; tp ranks
                # hold DIFFERENT shards that must not be averaged (cp ranks
                # replicate params over different sequence shards)
                if pc.dp_size * pc.cp_size > 1:
                    model = DistributedDataParallelEngine(model, process_group=pc._groups["grad"], **ddp_kwargs)
            else:
                model = DistributedDataParallelEngine(model, **ddp_kwargs)
        # torch.compile LAST, over the wrapped model (reference:
        # accelerator.py:2062-2066); regional compilation compiles each
        # repeated block once (utils/other.py compile_regions)
        if getattr(self, "dynamo_plugin", None) is not None and self.dynamo_plugin.enabled:
            if self.dynamo_plugin.use_regional_compilation:
                from .utils.other import compile_regions

                model = compile_regions(model, **self.dynamo_plugin.compile_kwargs())
            else:
                model = torch.compile(model, **self.dynamo_plugin.compile_kwargs())
        model._is_accelerate_prepared = True
        self._models.append(model)
        return model

    def prepare_data_loader(self, data_loader, device_placement=None, slice_fn_for_dispatch=None):
        # Ensure we can't double wrap a DataLoader due to `find_batch_size`
        if getattr(data_loader, "_is_accelerate_prepared", False):
            if data_loader not in self._dataloaders:
                self._dataloaders.append(data_loader)
            return data_loader
        if device_placement is None:
            device_placement = self.device_placement
        # Under TP/CP the batch is sharded over the DP dimension only: every
        # rank in the same tp/cp group must see the SAME batch (reference:
        # data_loader.py:1129-1165 process_index // (tp*cp) remap).
        dl_num, dl_idx = self.num_processes, self.process_index
        pc = self.parallelism_config
        if pc is not None and (pc.tp_size > 1 or pc.cp_size > 1):
            dl_num = pc.dp_size
            dl_idx = self.data_parallel_rank
        prepared = prepare_data_loader(
            data_loader,
            self.device,
            num_processes=dl_num,
            process_index=dl_idx,
            split_batches=self.split_batches,
            put_on_device=device_placement,
            rng_types=self.rng_types.copy() if self.rng_types else None,
            dispatch_batches=self.dispatch_batches,
            even_batches=self.even_batches,
            slice_fn_for_dispatch=slice_fn_for_dispatch,
            use_seedable_sampler=self.use_seedable_sampler,
            data_seed=self.data_seed,
            non_blocking=self.non_blocking,
        )
        prepared._is_accelerate_prepared = True
        self._dataloaders.append(prepared)
        return prepared

    def prepare_optimizer(self, optimizer: torch.optim.Optimizer, device_placement=None):
        if getattr(optimizer, "_is_accelerate_prepared", False):
            if optimizer not in self._optimizers:
                self._optimizers.append(optimizer)
            return optimizer
        if device_placement is None:
            device_placement = self.device_placement
        optimizer = AcceleratedOptimizer(optimizer, device_placement=device_placement, scaler=self.scaler)
        optimizer._is_accelerate_prepared = True
        self._optimizers.append(optimizer)
        return optimizer

    def prepare_scheduler(self, scheduler):
        if getattr(scheduler, "_is_accelerate_prepared", False):
            if scheduler not in self._schedulers:
                self._schedulers.append(scheduler)
            return scheduler
        # find the optimizer this scheduler drives
        optimizer = self._optimizers
        for opt in self._optimizers:
            if getattr(scheduler, "optimizer", None) == opt.optimizer:
                optimizer = opt
                break
        pc = self.parallelism_config
        shards = pc.dp_size if pc is not None and (pc.tp_size > 1 or pc.cp_size > 1) else None
        scheduler = AcceleratedScheduler(
            scheduler,
            optimizer,
            step_with_optimizer=self.step_scheduler_with_optimizer,
            split_batches=self.split_batches,
            num_batch_shards=shards,
        )
        scheduler._is_accelerate_prepared = True
        self._schedulers.append(scheduler)
        return scheduler

    # ------------------------------------------------------------------
    # train step primitives
    # ------------------------------------------------------------------

    def backward(self, loss, **kwargs):
        """(reference: accelerator.py:2818). Scales for grad accumulation and
        fp16; on sync steps finalizes the reducer (bucketed RCCL all-reduce)."""
        if self.gradient_state.num_steps > 1:
            loss = loss / self.gradient_state.num_steps
        if self.scaler is not None:
            scaled = self.scaler.scale(loss) if hasattr(self.scaler, "scale") else loss
            scaled.backward(**kwargs)
        else:
            loss.backward(**kwargs)
        # deterministic reduction epilogue: we own backward(), so no autograd
        # engine callbacks are needed (reference delegates to DDP's C++ hooks)
        from .parallel.fsdp import ShardedModel

        for model in self._models:
            if isinstance(model, DistributedDataParallelEngine):
                model.finalize()
            elif isinstance(model, ShardedModel):
                model.finalize_backward()

    def unscale_gradients(self, optimizer=None):
        """(reference: accelerator.py:2935)"""
        if self.native_amp and self.mixed_precision == "fp16":
            if optimizer is None:
                optimizer = self._optimizers
            elif not isinstance(optimizer, (tuple, list)):
                optimizer = [optimizer]
            for opt in optimizer:
                while isinstance(opt, AcceleratedOptimizer):
                    opt = opt.optimizer
                self.scaler.unscale_(opt)

    def clip_grad_norm_(self, parameters, max_norm, norm_type=2):
        """(reference: accelerator.py:2946). Runs the CDNA4 multi-tensor
        L2-norm kernel; under DDP grads are already reduced and identical
        across ranks so no extra collective is required."""
        self.unscale_gradients()
        from .parallel.fsdp import ShardedModel

        parameters = list(parameters) if not isinstance(parameters, torch.Tensor) else [parameters]
        for model in self._models:
            if isinstance(model, ShardedModel):
                # sharded world: the clip must happen on the gradient SHARDS
                # with a cross-rank norm reduction (reference: FSDP
                # model.clip_grad_norm_, accelerator.py:2977-3007)
                if any(id(p) in model._unit_param_ids for p in parameters):
                    return model.clip_grad_norm_(max_norm, norm_type)
        pc = self.parallelism_config
        if pc is not None and pc.tp_size > 1 and any(
            getattr(p, "_tp_sharded", False) for p in parameters
        ):
            # tensor parallelism: the global norm sums the SHARDED params'
            # contributions over the tp group (each shard counted once)
            # plus the replicated params' contribution counted ONCE; using
            # the local norm would clip replicas by different coefficients
            # and diverge them (reference: DTensor-aware clip)
            grads = [(p, p.grad) for p in parameters if p.grad is not None]
            if not grads:
                return torch.tensor(0.0)
            tp_group = pc._groups.get("tp")
            dev = grads[0][1].device
            if norm_type == float("inf"):
                local = max(g.detach().abs().max() for _, g in grads)
                total = local.clone().to(dev)
                torch.distributed.all_reduce(total, op=torch.distributed.ReduceOp.MAX, group=tp_group)
            else:
                shard_sq = sum(
                    g.detach().float().abs().pow(norm_type).sum()
                    for p, g in grads
                    if getattr(p, "_tp_sharded", False)
                )
                rep_sq = sum(
                    g.detach().float().abs().pow(norm_type).sum()
                    for p, g in grads
                    if not getattr(p, "_tp_sharded", False)
                )
                t = torch.as_tensor(shard_sq, dtype=torch.float32, device=dev).clone()
                torch.distributed.all_reduce(t, group=tp_group)
                total = (t + torch.as_tensor(rep_sq, dtype=torch.float32, device=dev)).pow(
                    1.0 / norm_type
                )
            coef = (max_norm / (total + 1e-6)).clamp(max=1.0)
            for _, g in grads:
                g.detach().mul_(coef.to(g.dtype))
            return total
        from .ops.clip_grad import clip_grad_norm_ as _clip

        return _clip(parameters, max_norm, norm_type=norm_type)

    def clip_grad_value_(self, parameters, clip_value):
        self.unscale_gradients()
        torch.nn.utils.clip_grad_value_(parameters, clip_value)

    # cooperative cross-rank breakpoint (reference: accelerator.py:2852-2909)

    def set_trigger(self):
        self.flag_tensor = torch.tensor(1, device=self.device)

    def check_trigger(self):
        if self.flag_tensor is None:
            self.flag_tensor = torch.tensor(0, device=self.device)
        flag_tensor = reduce(self.flag_tensor, reduction="sum")
        if flag_tensor.item() >= 1:
            self.flag_tensor = torch.tensor(0, device=self.device)
            return True
        return False

    # ------------------------------------------------------------------
    # collectives for metrics
    # ------------------------------------------------------------------

    def gather(self, tensor):
        return gather(tensor)

    def gather_for_metrics(self, input_data, use_gather_object: bool = False):
        """(reference: accelerator.py:3068-3140) — drops the duplicate tail
        samples introduced by even_batches padding."""
        try:
            recursively_apply(lambda x: x, input_data, error_on_other_type=True)
            all_tensors = True
        except TypeError:
            all_tensors = False

        use_gather_object = use_gather_object or not all_tensors
        if use_gather_object:
            data = gather_object(input_data)
        else:
            data = self.gather(input_data)

        try:
            if self.gradient_state.end_of_dataloader:
                # at the end of a dataloader, `gather_for_metrics` regresses to `gather` unless the dataset has a remainder
                if self.gradient_state.remainder == -1:
                    return data
                elif self.gradient_state.remainder > 0:
                    # Last batch needs to be truncated on distributed systems as it contains additional samples
                    def _adjust_samples(tensor):
                        return tensor[: self.gradient_state.remainder]

                    if use_gather_object:
                        return _adjust_samples(data)
                    else:
                        return recursively_apply(_adjust_samples, data)
                else:
                    return data
            else:
                return data
        except Exception:
            return data

    def reduce(self, tensor, reduction="sum", scale=1.0):
        return reduce(tensor, reduction, scale)

    def pad_across_processes(self, tensor, dim=0, pad_index=0, pad_first=False):
        return pad_across_processes(tensor, dim=dim, pad_index=pad_index, pad_first=pad_first)

    # ------------------------------------------------------------------
    # misc
    # ------------------------------------------------------------------

    def unwrap_model(self, model, keep_fp32_wrapper: bool = True):
        return extract_model_from_parallel(model, keep_fp32_wrapper)

    def wait_for_everyone(self):
        wait_for_everyone()

    def print(self, *args, **kwargs):
        self.state.print(*args, **kwargs)

    @property
    def optimizer_step_was_skipped(self) -> bool:
        """True if any prepared optimizer skipped its last step (fp16 inf/nan
        or accumulation gating) — reference accelerator.py property."""
        return any(getattr(opt, "step_was_skipped", False) for opt in self._optimizers)

    @property
    def multi_device(self) -> bool:
        return self.use_distributed and self.num_processes > 1

    @property
    def is_fsdp2(self) -> bool:
        """Our sharded engine IS the FSDP2-equivalent (single implementation,
        no FSDP1/FSDP2 split) — true whenever FSDP is active."""
        return self.distributed_type == DistributedType.FSDP

    @property
    def fp8_backend(self):
        """'NATIVE' when our CDNA4 fp8 stack is active (the reference returns
        TE/AO/MSAMP — all replaced by ops/fp8.py here)."""
        return "NATIVE" if self.mixed_precision == "fp8" else None

    @property
    def should_save_model(self) -> bool:
        """Whether THIS rank must write model weights in save_model
        (sharded state dicts write per-rank; full dicts write on main)."""
        plugin = getattr(self.state, "fsdp_plugin", None)
        if plugin is not None and getattr(plugin, "state_dict_type", "") == "SHARDED_STATE_DICT":
            return True
        return self.is_main_process

    # -- parallelism-dimension ranks (reference accelerator.py properties) --
    def _pc_coord(self, dim: str) -> int:
        pc = self.parallelism_config
        if pc is None:
            return 0
        return pc.coords(self.process_index)[dim]

    @property
    def data_parallel_rank(self) -> int:
        pc = self.parallelism_config
        if pc is None:
            return self.process_index
        c = pc.coords(self.process_index)
        return c["dp_replicate"] * pc.dp_shard_size + c["dp_shard"]

    @property
    def data_parallel_shard_rank(self) -> int:
        return self._pc_coord("dp_shard")

    @property
    def tensor_parallel_rank(self) -> int:
        return self._pc_coord("tp")

    @property
    def context_parallel_rank(self) -> int:
        return self._pc_coord("cp")

    @property
    def torch_device_mesh(self):
        """The per-dimension process groups built from parallelism_config
        (our mesh equivalent; None until build_groups has run)."""
        pc = self.parallelism_config
        return pc._groups if pc is not None and pc._groups else None

    def skip_first_batches(self, dataloader, num_batches: int = 0):
        """Mid-epoch resume helper (reference accelerator.py:3568)."""
        from .data_loader import skip_first_batches

        return skip_first_batches(dataloader, num_batches)

    def trigger_sync_in_backward(self, model):
        """Force gradient sync on the NEXT backward even inside a no_sync /
        accumulation window (reference accelerator.py:1210-1227) — used to
        flush accumulated grads early, e.g. before an optimizer swap."""
        from .parallel.ddp import DistributedDataParallelEngine

        unwrapped = model
        if isinstance(model, DistributedDataParallelEngine):
            model.require_backward_grad_sync = True
            return
        for m in self._models:
            if isinstance(m, DistributedDataParallelEngine) and (m is model or m.module is unwrapped):
                m.require_backward_grad_sync = True
                return

    @contextmanager
    def maybe_context_parallel(self, buffers=None, buffer_seq_dims=None):
        """Shard the given buffers along their sequence dim for this step when
        a context-parallel group is active (reference accelerator.py:4111
        maybe_context_parallel); no-op otherwise. Buffers are replaced
        IN PLACE in the caller's list."""
        pc = self.parallelism_config
        group = None
        if pc is not None and pc.cp_size > 1 and pc._groups:
            group = pc._groups.get("cp")
        if group is None:
            yield
            return
        if buffers:
            from .parallel.cp import shard_sequence

            dims = buffer_seq_dims or [1] * len(buffers)
            for i, (buf, dim) in enumerate(zip(buffers, dims)):
                buffers[i] = shard_sequence(buf, group=group, dim=dim)
        # activate the sequence-parallel collective pattern for the scope of
        # this step: models route attention through ops.attention.dispatch
        from .ops.attention import set_sequence_parallel

        set_sequence_parallel(pc.cp_impl, group)
        try:
            yield
        finally:
            set_sequence_parallel(None)

    @contextmanager
    def autocast(self, autocast_handler: AutocastKwargs = None):
        """bf16/fp16 autocast context over the ROCm HIP autocast dispatcher
        (reference: accelerator.py:4178)."""
        if autocast_handler is None:
            autocast_handler = self.autocast_handler
        ctx = get_mixed_precision_context_manager(self.native_amp, self.state.mixed_precision, self.device, autocast_handler)
        with ctx:
            yield

    @contextmanager
    def profile(self, profile_handler: Optional[ProfileKwargs] = None):
        """torch.profiler over kineto/roctracer (reference: accelerator.py:4203)."""
        profile_handler = profile_handler or self.profile_handler or ProfileKwargs()
        with profile_handler.build() as profiler:
            yield profiler
        if profile_handler.output_trace_dir is None:
            return
        os.makedirs(profile_handler.output_trace_dir, exist_ok=True)
        profiler.export_chrome_trace(
            os.path.join(profile_handler.output_trace_dir, f"profile_{self.process_index}.json")
        )
        self.wait_for_everyone()

    def free_memory(self, *objects):
        from .utils.memory import release_memory

        self._schedulers = []
        self._optimizers = []
        self._models = []
        self._dataloaders = []
        self.step = 0
        return release_memory(*objects)

    def clear(self, *objects):
        return self.free_memory(*objects)

    def verify_device_map(self, model: torch.nn.Module) -> bool:
        """Checks if the model was dispatched with a non-trivial device_map."""
        for m in model.modules():
            if hasattr(m, "_hf_hook") and getattr(m._hf_hook, "execution_device", None) is not None:
                return True
        return getattr(model, "hf_device_map", None) is not None and len(getattr(model, "hf_device_map", {})) > 1

    # ------------------------------------------------------------------
    # tracking
    # ------------------------------------------------------------------

    def init_trackers(self, project_name: str, config: Optional[dict] = None, init_kwargs: Optional[dict] = None):
        init_kwargs = init_kwargs or {}
        self.trackers = []
        for tracker in self.log_with:
            if issubclass(type(tracker), GeneralTracker):
                self.trackers.append(tracker)
            else:
                tracker_init = tracker
                if getattr(tracker_init, "requires_logging_directory", False):
                    self.trackers.append(
                        tracker_init(project_name, self.logging_dir, **init_kwargs.get(str(tracker_init.name), {}))
                    )
                else:
                    self.trackers.append(tracker_init(project_name, **init_kwargs.get(str(tracker_init.name), {})))
        if config is not None:
            for tracker in self.trackers:
                tracker.store_init_configuration(config)

    def get_tracker(self, name: str, unwrap: bool = False):
        if len(self.trackers) > 0:
            for tracker in self.trackers:
                if tracker.name == name:
                    return tracker.tracker if unwrap else tracker
            raise ValueError(f"{name} is not an available tracker stored inside the `Accelerator`.")
        from .tracking import GeneralTracker as _GT

        return _GT(_blank=True)

    def log(self, values: dict, step: Optional[int] = None, log_kwargs: Optional[dict] = None):
        log_kwargs = log_kwargs or {}
        for tracker in self.trackers:
            tracker.log(values, step=step, **log_kwargs.get(str(tracker.name), {}))

    def end_training(self):
        for tracker in self.trackers:
            tracker.finish()
        self.state.destroy_process_group()

    # ------------------------------------------------------------------
    # checkpointing
    # ------------------------------------------------------------------

    def save(self, obj, f, safe_serialization=False):
        from .checkpointing import save as _save

        _save(obj, f, save_on_each_node=self.project_configuration.save_on_each_node, safe_serialization=safe_serialization)

    def register_save_state_pre_hook(self, hook: Callable[..., None]) -> hooks.RemovableHandle:
        handle = hooks.RemovableHandle(self._save_model_state_pre_hook)
        self._save_model_state_pre_hook[handle.id] = hook
        return handle

    def register_load_state_pre_hook(self, hook: Callable[..., None]) -> hooks.RemovableHandle:
        handle = hooks.RemovableHandle(self._load_model_state_pre_hook)
        self._load_model_state_pre_hook[handle.id] = hook
        return handle

    def save_state(self, output_dir: str = None, safe_serialization: bool = True, **save_model_func_kwargs):
        """(reference: accelerator.py:3584)"""
        if self.project_configuration.automatic_checkpoint_naming:
            output_dir = os.path.join(self.project_dir, "checkpoints")
        os.makedirs(output_dir, exist_ok=True)
        if self.project_configuration.automatic_checkpoint_naming:
            folders = [os.path.join(output_dir, folder) for folder in os.listdir(output_dir)]
            if (
                self.project_configuration.total_limit is not None
                and (len(folders) + 1 > self.project_configuration.total_limit)
                and self.is_main_process
            ):

                def _inner(folder):
                    return list(map(int, [s for s in folder.replace(output_dir, "").split("_") if s.isdigit()]))[0]

                folders.sort(key=_inner)
                logger.warning(
                    f"Deleting {len(folders) + 1 - self.project_configuration.total_limit} checkpoints to make room for new checkpoint."
                )
                for folder in folders[: len(folders) + 1 - self.project_configuration.total_limit]:
                    shutil.rmtree(folder)
            output_dir = os.path.join(output_dir, f"checkpoint_{self.save_iteration}")
            if os.path.exists(output_dir):
                raise ValueError(
                    f"Checkpoint directory {output_dir} ({self.save_iteration}) already exists. Please manually override `self.save_iteration` with what iteration to start with."
                )
            self.wait_for_everyone()
        os.makedirs(output_dir, exist_ok=True)
        logger.info(f"Saving current state to {output_dir}")

        # SHARDED_STATE_DICT: every rank writes its own shard file with zero
        # cross-rank communication (reference rationale: fsdp_utils.py:107-118
        # — the DCP-timeout fix at 2800+ GPUs); FULL dicts all-gather per unit
        # and rank 0 writes one file.
        from .parallel.fsdp import ShardedModel

        plugin = getattr(self.state, "fsdp_plugin", None)
        sharded_dicts = plugin is not None and getattr(plugin, "state_dict_type", "") == "sharded_state_dict"
        weights = []
        for i, m in enumerate(self._models):
            if sharded_dicts and isinstance(m, ShardedModel):
                suffix = f"_{i}" if i > 0 else ""
                shard_file = os.path.join(output_dir, f"model_fsdp{suffix}_rank{self.process_index}.bin")
                torch.save(m.sharded_state_dict(), shard_file)
                weights.append({"__fsdp_sharded__": True})  # placeholder: skip full save
            else:
                weights.append(self.get_state_dict(m, unwrap=False))
        # Save the samplers of the dataloaders
        dataloaders = self._dataloaders

        for hook in self._save_model_state_pre_hook.values():
            hook(self._models, weights, output_dir)

        save_location = save_accelerator_state(
            output_dir,
            weights,
            self._optimizers,
            self._schedulers,
            dataloaders,
            self.state.process_index,
            self.step,
            scaler=self.scaler,
            save_on_each_node=self.project_configuration.save_on_each_node,
            safe_serialization=safe_serialization,
        )
        for i, obj in enumerate(self._custom_objects):
            save_custom_state(obj, output_dir, i, save_on_each_node=self.project_configuration.save_on_each_node)
        self.project_configuration.iteration += 1
        # every file durable before ANY rank proceeds: a non-main rank that
        # immediately calls load_state (or reads the dir) must not race the
        # main rank's writes
        self.wait_for_everyone()
        return save_location

    def load_state(self, input_dir: str = None, load_kwargs=None, **load_model_func_kwargs):
        """(reference: accelerator.py:3750)"""
        if input_dir is not None:
            input_dir = os.path.expanduser(input_dir)
            if not os.path.isdir(input_dir):
                raise ValueError(f"Tried to find {input_dir} but folder does not exist")
        elif self.project_configuration.automatic_checkpoint_naming:
            # Pick up from automatic checkpoint naming
            input_dir = os.path.join(self.project_dir, "checkpoints")
            folders = [os.path.join(input_dir, folder) for folder in os.listdir(input_dir)]

            def _inner(folder):
                return list(map(int, [s for s in folder.replace(input_dir, "").split("_") if s.isdigit()]))[0]

            folders.sort(key=_inner)
            input_dir = folders[len(folders) - 1]
        else:
            raise ValueError("No input_dir provided and automatic checkpoint naming is disabled.")
        logger.info(f"Loading states from {input_dir}")

        models = self._models
        for hook in self._load_model_state_pre_hook.values():
            hook(models, input_dir)

        # sharded checkpoints load per-rank with no communication; those
        # models are excluded from the full-dict loading below
        from .parallel.fsdp import ShardedModel

        skip_models = set()
        for i, m in enumerate(models):
            suffix = f"_{i}" if i > 0 else ""
            shard_file = os.path.join(input_dir, f"model_fsdp{suffix}_rank{self.process_index}.bin")
            if isinstance(m, ShardedModel) and os.path.exists(shard_file):
                m.load_sharded_state_dict(torch.load(shard_file, weights_only=False))
                skip_models.add(i)
        # placeholders keep indices aligned with the on-disk _{i} suffixes
        models = [None if i in skip_models else m for i, m in enumerate(models)]

        map_location = load_model_func_kwargs.pop("map_location", None)
        if map_location is None:
            if self.num_processes > 1 and self.distributed_type == DistributedType.MULTI_GPU:
                map_location = "on_device"
            else:
                map_location = "cpu"

        override_attributes = load_accelerator_state(
            input_dir,
            models,
            self._optimizers,
            self._schedulers,
            self._dataloaders,
            self.state.process_index,
            scaler=self.scaler,
            map_location=map_location,
            load_model_func_kwargs=load_model_func_kwargs,
            load_kwargs=load_kwargs,
        )
        if "step" in override_attributes:
            self.step = override_attributes["step"]
        custom_checkpoints = [
            f for f in os.listdir(input_dir) if "custom_checkpoint" in f and os.path.splitext(f)[0].split("_")[-1].isdigit()
        ]
        if len(custom_checkpoints) != len(self._custom_objects):
            err = f"Number of custom checkpoints in folder {input_dir} does not match the number of registered objects:"
            err += f"\n\tFound checkpoints: {len(custom_checkpoints)}"
            err += f"\n\tRegistered objects: {len(self._custom_objects)}\n"
            err += "Please make sure to only load checkpoints from folders that were created with the same set of registered objects."
            raise RuntimeError(err)
        else:
            logger.info(f"Loading in {len(custom_checkpoints)} custom states")
            for index, obj in enumerate(self._custom_objects):
                load_custom_state(obj, input_dir, index)

    def register_for_checkpointing(self, *objects):
        """(reference: accelerator.py register_for_checkpointing)"""
        invalid_objects = []
        for obj in objects:
            if not hasattr(obj, "state_dict") or not hasattr(obj, "load_state_dict"):
                invalid_objects.append(obj)
        if len(invalid_objects) > 0:
            err = "All `objects` must include a `state_dict` and `load_state_dict` function to be stored. The following inputs are invalid:"
            for index, obj in enumerate(invalid_objects):
                err += f"\n\t- Item at index {index}, `{type(obj).__name__}`"
            raise ValueError(err)
        self._custom_objects.extend(objects)

    def get_state_dict(self, model, unwrap=True):
        """Full state-dict of a (possibly wrapped) model (reference: accelerator.py:3384)."""
        if unwrap:
            model = self.unwrap_model(model)
        from .parallel.fsdp import ShardedModel, gather_full_state_dict

        if isinstance(model, ShardedModel):
            return gather_full_state_dict(model)
        return model.state_dict()

    def save_model(
        self,
        model: torch.nn.Module,
        save_directory: str,
        max_shard_size: Union[int, str] = "10GB",
        safe_serialization: bool = True,
    ):
        """Save a model's weights for inference, sharded with an index
        (reference: accelerator.py:3439)."""
        if os.path.isfile(save_directory):
            logger.error(f"Provided path ({save_directory}) should be a directory, not a file")
            return
        os.makedirs(save_directory, exist_ok=True)
        state_dict = self.get_state_dict(model)
        from .utils.modeling import save_model_weights

        save_model_weights(
            state_dict,
            save_directory,
            max_shard_size=max_shard_size,
            safe_serialization=safe_serialization,
            is_main_process=self.is_main_process,
        )

    # ------------------------------------------------------------------
    # hipGraph step capture (MI355X: launch-bound small-batch loops)
    # ------------------------------------------------------------------

    def capture_step(self, step_fn: Callable, *example_args, warmup: int = 3):
        """Capture ``step_fn(*example_args)`` into a hipGraph and return a
        callable replaying it. The caller owns keeping input buffers static
        (copy new data into the example tensors before replay). Eliminates
        per-kernel launch latency in launch-bound inner loops — the MI355X
        counterpart of the reference's torch.compile integration
        (reference: dataclasses.py:1033 TorchDynamoPlugin)."""
        if self.device.type != "cuda":
            return step_fn
        # Free stale autograd graphs: an AccumulateGrad node kept alive from a
        # pre-capture iteration and pinned to the default stream segfaults
        # ROCm's hipGraph capture_end (observed on MI355X, torch 2.10+rocm7.0).
        import gc

        gc.collect()
        torch.cuda.synchronize()
        side_stream = torch.cuda.Stream()
        side_stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side_stream):
            for _ in range(warmup):
                step_fn(*example_args)
        torch.cuda.current_stream().wait_stream(side_stream)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            out = step_fn(*example_args)

        def replay():
            graph.replay()
            return out

        replay._graph = graph
        return replay

    def __deepcopy__(self, memo):
        return self


def autocast_context_wrap(make_ctx, model_forward):
    """Wrap a forward so each call enters a FRESH autocast context."""

    @functools.wraps(model_forward)
    def wrapped(*args, **kwargs):
        with make_ctx():
            return model_forward(*args, **kwargs)

    wrapped.__wrapped__ = model_forward
    return wrapped


def get_mixed_precision_context_manager(native_amp, mixed_precision, device, autocast_kwargs=None):
    """(reference: modeling.py:2066)"""
    if not native_amp or mixed_precision not in ("fp16", "bf16", "fp8"):
        return contextlib.nullcontext()
    kwargs = autocast_kwargs.to_kwargs() if autocast_kwargs is not None else {}
    dtype = torch.bfloat16 if mixed_precision in ("bf16", "fp8") else torch.float16
    device_type = "cuda" if device.type == "cuda" else "cpu"
    if device_type == "cpu":
        dtype = torch.bfloat16
    return torch.autocast(device_type=device_type, dtype=dtype, **kwargs)
