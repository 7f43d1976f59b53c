"""HBM-aware sizing calculator + admission integration."""
from trainingjob_operator_amd.api import constants as C, sizing
from trainingjob_operator_amd.api.defaults import set_defaults
from trainingjob_operator_amd.api.types import AITrainingJob
from trainingjob_operator_amd.api.validation import validate


def test_llama8b_fits_one_gpu():
    est = sizing.estimate_training_bytes(
        8_030_000_000, micro_batch=2, seq_len=4096,
        hidden_size=4096, num_layers=32)
    # measured on hardware: ~150-190 GB peak at this config
    assert 130 < est.total_gb < 270
    assert sizing.fits_per_gpu(est, gpus_per_pod=1)
    assert sizing.min_gpus_for(est) == 1


def test_huge_model_needs_many_gpus():
    est = sizing.estimate_training_bytes(70_000_000_000,
                                         hidden_size=8192, num_layers=80)
    # 70B: 16 bytes/param state alone = 1.12 TB
    assert not sizing.fits_per_gpu(est, 1)
    assert sizing.min_gpus_for(est) >= 4


def test_checkpointing_shrinks_activations():
    a = sizing.estimate_training_bytes(8e9, micro_batch=4, seq_len=8192,
                                       hidden_size=4096, num_layers=32,
                                       checkpoint_activations=False)
    b = sizing.estimate_training_bytes(8e9, micro_batch=4, seq_len=8192,
                                       hidden_size=4096, num_layers=32,
                                       checkpoint_activations=True)
    # the lm-head logits term is not checkpointable, so the ratio is bounded
    assert b.activation_bytes < a.activation_bytes / 2


def _job(params_ann=None, model_ann=None, gpus=1):
    ann = {}
    if params_ann:
        ann[sizing.MODEL_PARAMS_ANNOTATION] = str(params_ann)
    if model_ann:
        ann[sizing.MODEL_NAME_ANNOTATION] = model_ann
    return set_defaults(AITrainingJob.from_dict({
        "metadata": {"name": "j", "namespace": "d", "annotations": ann},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 2,
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "resources": {"limits": {C.AMD_GPU_RESOURCE: gpus}},
            }]}}}}},
    }))


def test_admission_accepts_declared_8b():
    assert validate(_job(model_ann="llama3-8b")) == []


def test_admission_rejects_oversized_model():
    errs = validate(_job(params_ann=70_000_000_000, gpus=1))
    assert len(errs) == 1
    assert "needs ~" in errs[0] and "request at least" in errs[0]
    # granting enough GPUs clears it
    assert validate(_job(params_ann=70_000_000_000, gpus=8)) == []


def test_no_declaration_no_check():
    assert validate(_job()) == []


def test_estimate_derives_shape_when_unspecified():
    est = sizing.estimate_training_bytes(8_000_000_000)
    # derived hidden/layers must land in a sane transformer regime
    assert est.total_bytes > 8_000_000_000 * 16  # at least the state bytes
    assert sizing.min_gpus_for(est) >= 1


def test_declared_params_precedence():
    j = _job(params_ann=123, model_ann="llama3-8b")
    assert sizing.declared_params(j) == 123  # count wins over name
    assert sizing.declared_params(_job(model_ann="nope")) is None
    assert sizing.declared_params(_job()) is None


def test_sharded_model_admission():
    """model-shards annotation: a pp x tp sharded 70B fits 8 x 1-GPU pods;
    under-provisioned shard counts are rejected."""
    j = _job(model_ann="llama3-70b", gpus=1)
    j.annotations[sizing.MODEL_SHARDS_ANNOTATION] = "8"
    j.spec.replica_specs["trainer"].replicas = 8
    assert validate(j) == []
    # unsharded 70B on 1-GPU pods is rejected
    assert validate(_job(model_ann="llama3-70b", gpus=1)) != []
    # declaring more shards than provided GPUs is rejected
    j.spec.replica_specs["trainer"].replicas = 4
    errs = validate(j)
    assert len(errs) == 1 and "model-shards=8" in errs[0]
