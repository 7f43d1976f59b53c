"""Example manifests must stay schema-valid (parsed through the same types +
validation the controller applies)."""
import glob
import os

import yaml

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.crd import crd_manifest
from trainingjob_operator_amd.api.defaults import set_defaults
from trainingjob_operator_amd.api.types import AITrainingJob
from trainingjob_operator_amd.api.validation import validate

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_all_examples_valid():
    paths = sorted(glob.glob(os.path.join(ROOT, "manifests", "examples",
                                          "*.yaml")))
    assert len(paths) >= 4
    for path in paths:
        doc = yaml.safe_load(open(path))
        assert doc["apiVersion"] == C.API_VERSION, path
        assert doc["kind"] == C.CRD_KIND, path
        job = set_defaults(AITrainingJob.from_dict(doc))
        assert validate(job) == [], f"{path}: {validate(job)}"


def test_llama_example_requests_gpus():
    from trainingjob_operator_amd.api.validation import gpus_requested
    doc = yaml.safe_load(open(os.path.join(
        ROOT, "manifests", "examples", "llama3-8b-ddp.yaml")))
    job = set_defaults(AITrainingJob.from_dict(doc))
    spec = job.spec.replica_specs["trainer"]
    assert spec.replicas == 8
    assert gpus_requested(spec) == 1


def test_crd_manifest_matches_committed_yaml():
    committed = yaml.safe_load(open(os.path.join(ROOT, "manifests",
                                                 "crd.yaml")))
    assert committed == crd_manifest()


def test_crd_has_status_subresource_and_shortname():
    crd = crd_manifest()
    v = crd["spec"]["versions"][0]
    assert v["subresources"] == {"status": {}}
    assert crd["spec"]["names"]["shortNames"] == ["aitj"]


def test_validate_job_cli():
    import glob
    import subprocess
    import sys
    paths = sorted(glob.glob(os.path.join(
        ROOT, "manifests", "examples", "*.yaml")))
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "scripts", "validate_job.py")]
        + paths, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert r.stdout.count("OK") == len(paths)
    # an invalid spec is rejected with a non-zero exit
    import tempfile
    with tempfile.NamedTemporaryFile("w", suffix=".yaml",
                                     delete=False) as f:
        f.write("apiVersion: elasticdeeplearning.ai/v1\n"
                "kind: AITrainingJob\n"
                "metadata: {name: Bad_Name}\n"
                "spec: {replicaSpecs: {t: {template: {spec: "
                "{containers: [{name: main}]}}}}}\n")
        bad = f.name
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "scripts", "validate_job.py"),
         bad], capture_output=True, text=True, timeout=300)
    os.unlink(bad)
    assert r.returncode != 0
