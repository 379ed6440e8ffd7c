"""Framework-wide constants.

Checkpoint file-name contract mirrors the reference layout
(reference: utils/constants.py:20-33,48) so users migrating from
huggingface/accelerate find the same on-disk structure.
"""

MODEL_NAME = "pytorch_model"
SAFE_MODEL_NAME = "model"
OPTIMIZER_NAME = "optimizer"
SCHEDULER_NAME = "scheduler"
SAMPLER_NAME = "sampler"
SCALER_NAME = "scaler"
RNG_STATE_NAME = "random_states"
CUSTOM_STATE_NAME = "custom_checkpoint"
PROFILE_PATTERN_NAME = "profile_{suffix}.json"

WEIGHTS_NAME = f"{MODEL_NAME}.bin"
WEIGHTS_PATTERN_NAME = "pytorch_model{suffix}.bin"
WEIGHTS_INDEX_NAME = f"{WEIGHTS_NAME}.index.json"
SAFE_WEIGHTS_NAME = f"{SAFE_MODEL_NAME}.safetensors"
SAFE_WEIGHTS_PATTERN_NAME = "model{suffix}.safetensors"
SAFE_WEIGHTS_INDEX_NAME = f"{SAFE_WEIGHTS_NAME}.index.json"

SAGEMAKER_PYTORCH_VERSION = None  # not supported: AWS-specific, out of scope

# Target architecture. This framework is MI355X (gfx950)-only.
GFX_ARCH = "gfx950"

# Number of point-to-point xGMI links per MI355X GPU and per-link BW (GB/s).
# Ring collectives are per-link bound; bucket sizing derives from this.
XGMI_LINKS_PER_GPU = 7
XGMI_LINK_GBPS = 153

# Default gradient-bucket size for the DDP reducer, in MiB. Chosen for
# 7-link xGMI: a ring all-reduce of S bytes moves 2(n-1)/n*S over one link
# path; ~64 MiB buckets keep each RCCL call in the bandwidth-bound regime
# (>4 MiB/rank-chunk at n=8) while still giving backward/comm overlap.
# (reference DDP default is 25 MiB, dataclasses.py:185 — sized for NVLink.)
DDP_BUCKET_CAP_MB = 64

TORCH_LAUNCH_PARAMS = [
    "nnodes", "nproc_per_node", "rdzv_backend", "rdzv_endpoint", "rdzv_id",
    "rdzv_conf", "standalone", "max_restarts", "monitor_interval",
    "start_method", "role", "module", "no_python", "run_path", "log_dir",
    "redirects", "tee", "node_rank", "master_addr", "master_port",
]

ELASTIC_LOG_LINE_PREFIX_TEMPLATE_PYTORCH_VERSION = "2.2.0"
