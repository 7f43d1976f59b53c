"""HBM-aware replica sizing for MI355X (288 GB HBM3E per GPU).

The reference schedules opaque containers with zero GPU awareness
(SURVEY.md §2.3 'no GPU awareness'); this module gives the operator a
model-size -> memory-footprint -> fits/GPUs-needed calculator, used at
admission: a job that declares its model (by name or parameter count via
annotations) and requests amd.com/gpu is rejected when the per-GPU
training state cannot fit.

Footprint model (DP training, the managed workload's layout):
  bf16 params + bf16 grads            = 4 bytes/param
  fp32 master + AdamW m + v           = 12 bytes/param
  activations                         ~ micro_batch * seq * per-token cost
All DP state is replicated per GPU (flat store, parallel/flat.py), so the
per-GPU footprint is independent of the DP degree.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from .constants import CRD_GROUP, MI355X_HBM_BYTES

MODEL_PARAMS_ANNOTATION = f"{CRD_GROUP}/model-params"
MODEL_NAME_ANNOTATION = f"{CRD_GROUP}/model"
# model sharded over N devices (pp x tp): per-GPU state = total / N
MODEL_SHARDS_ANNOTATION = f"{CRD_GROUP}/model-shards"

# ~8% HBM headroom: allocator fragmentation, RCCL buffers, HIP runtime.
_HEADROOM = 0.92


@dataclass
class MemoryEstimate:
    params_bytes: int
    grads_bytes: int
    optimizer_bytes: int
    activation_bytes: int

    @property
    def total_bytes(self) -> int:
        return (self.params_bytes + self.grads_bytes
                + self.optimizer_bytes + self.activation_bytes)

    @property
    def total_gb(self) -> float:
        return self.total_bytes / (1 << 30)


def estimate_training_bytes(n_params: int, micro_batch: int = 1,
                            seq_len: int = 4096,
                            hidden_size: Optional[int] = None,
                            num_layers: Optional[int] = None,
                            checkpoint_activations: bool = False) -> MemoryEstimate:
    """DP-training footprint per GPU for a bf16 model with fused fp32 AdamW.

    Activation estimate: ~66 KB per token per (4096-hidden) layer measured
    on the Llama stack (embeddings, norms, attention and MLP intermediates
    incl. the SwiGLU expansion), scaled linearly in hidden size; with
    activation checkpointing only the per-layer boundary tensors stay.
    """
    params = n_params * 2
    grads = n_params * 2
    optimizer = n_params * 12  # fp32 master + m + v
    if hidden_size is None or num_layers is None:
        # derive a rough transformer shape from the parameter count
        # (n ~ 12 * L * H^2 with L ~ H / 128)
        hidden_size = hidden_size or int(round(
            (n_params / 12 * 128) ** (1 / 3) / 128) * 128) or 4096
        num_layers = num_layers or max(hidden_size // 128, 1)
    per_token_per_layer = 66_000 * hidden_size / 4096
    if checkpoint_activations:
        per_token_per_layer = 4 * hidden_size  # boundary tensors only
    activations = int(micro_batch * seq_len * per_token_per_layer
                      * num_layers)
    # logits + CE workspace for LM heads (vocab ~ 128k worst case)
    activations += micro_batch * seq_len * 128_256 * 2 * 2
    return MemoryEstimate(params, grads, optimizer, activations)


def fits_per_gpu(estimate: MemoryEstimate, gpus_per_pod: int = 1,
                 hbm_bytes: int = MI355X_HBM_BYTES) -> bool:
    if gpus_per_pod <= 0:
        return False
    return estimate.total_bytes <= hbm_bytes * gpus_per_pod * _HEADROOM


def min_gpus_for(estimate: MemoryEstimate,
                 hbm_bytes: int = MI355X_HBM_BYTES) -> int:
    """Minimum GPUs per pod so the DP state fits (model parallelism is a
    workload concern; this bounds plain DP replication)."""
    usable = hbm_bytes * _HEADROOM
    return max(1, -(-estimate.total_bytes // int(usable)))


# Parameter counts for the models the managed workload ships
# (models/config.py); annotation "elasticdeeplearning.ai/model" selects one.
KNOWN_MODELS = {
    "llama3-8b": 8_030_000_000,
    "llama3-70b": 70_600_000_000,
    "moe-8x7b": 46_700_000_000,
    "llama-1b": 1_100_000_000,
}


def declared_params(job) -> Optional[int]:
    """Model size from job annotations (count wins over name)."""
    ann = job.annotations
    if MODEL_PARAMS_ANNOTATION in ann:
        try:
            return int(ann[MODEL_PARAMS_ANNOTATION])
        except ValueError:
            return None
    name = ann.get(MODEL_NAME_ANNOTATION)
    if name:
        return KNOWN_MODELS.get(name)
    return None


def declared_shards(job) -> int:
    """How many ways the model is sharded across devices (the launcher's
    pp x tp degree, or ep for MoE), from the model-shards annotation;
    1 = unsharded/DP. Admission divides the state uniformly — a slight
    underestimate for EP (dense params replicate across the plane), which
    the sizing headroom absorbs."""
    try:
        return max(1, int(job.annotations.get(MODEL_SHARDS_ANNOTATION, 1)))
    except (TypeError, ValueError):
        return 1
