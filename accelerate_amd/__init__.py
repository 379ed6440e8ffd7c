"""accelerate_amd — an MI355X-native training-loop framework with the
capabilities of huggingface/accelerate, built from scratch for CDNA4:
PyTorch-ROCm at the tensor/autograd level; our own RCCL/xGMI gradient
reducer, CDNA4 HIP kernel pack (fused AdamW, clip, scaler), device-map
dispatch sized for 288 GB HBM3E; no CUDA shims, no Triton, no plugin
dispatch to third-party engines.
"""

__version__ = "0.1.0"

from .accelerator import Accelerator
from .big_modeling import (
    attach_layerwise_casting_hooks,
    cpu_offload,
    cpu_offload_with_hook,
    disk_offload,
    dispatch_model,
    init_empty_weights,
    init_on_device,
    load_checkpoint_and_dispatch,
)
from .data_loader import prepare_data_loader, skip_first_batches
from .launchers import debug_launcher, notebook_launcher
from .inference import prepare_pipeline, prepare_pippy
from .local_sgd import LocalSGD
from .parallelism_config import ParallelismConfig
from .state import AcceleratorState, GradientState, PartialState
from .utils.imports import is_rich_available  # noqa: F401

if is_rich_available():  # reference: pretty tracebacks when rich is present
    from .utils import rich  # noqa: F401
from .utils.random_utils import synchronize_rng_states  # noqa: F401
from .utils.dataclasses import (
    AutocastKwargs,
    DeepSpeedPlugin,
    DataLoaderConfiguration,
    DDPCommunicationHookType,
    DistributedDataParallelKwargs,
    DistributedType,
    FP8RecipeKwargs,
    FullyShardedDataParallelPlugin,
    GradientAccumulationPlugin,
    GradScalerKwargs,
    InitProcessGroupKwargs,
    ProfileKwargs,
    ProjectConfiguration,
)
from .utils.modeling import infer_auto_device_map, load_checkpoint_in_model
from .utils.memory import find_executable_batch_size
from .utils.operations import send_to_device
from .utils.random_utils import set_seed
