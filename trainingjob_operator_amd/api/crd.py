"""CRD manifest for aitrainingjobs.elasticdeeplearning.ai.

The reference self-registers a v1beta1 CRD with NO OpenAPI schema
(reference: pkg/controller/controller.go:210-234); this is the modern
apiextensions/v1 equivalent with a real structural schema and the same
group/kind/plural/shortname (register.go:27-33).
"""
from __future__ import annotations

from . import constants as C

_ENDING_POLICY = {"type": "string", "enum": ["All", "Rank0", "Any", "None"]}


def _replica_spec_schema() -> dict:
    return {
        "type": "object",
        "properties": {
            "minReplicas": {"type": "integer", "minimum": 0},
            "maxReplicas": {"type": "integer", "minimum": 0},
            "replicas": {"type": "integer", "minimum": 0},
            "restartLimit": {"type": "integer", "minimum": 0},
            "template": {"type": "object",
                         "x-kubernetes-preserve-unknown-fields": True},
            "restartPolicy": {
                "type": "string",
                "enum": ["Always", "OnFailure", "OnNodeFail", "Never",
                         "ExitCode", "OnNodeFailWithExitCode"],
            },
            "restartScope": {"type": "string",
                             "enum": ["All", "Replica", "Pod"]},
            "failPolicy": _ENDING_POLICY,
            "completePolicy": _ENDING_POLICY,
            "edlPolicy": {"type": "string",
                          "enum": ["Auto", "Manual", "Never"]},
        },
    }


def crd_manifest() -> dict:
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": C.CRD_NAME},
        "spec": {
            "group": C.CRD_GROUP,
            "names": {
                "kind": C.CRD_KIND,
                "listKind": C.CRD_KIND_LIST,
                "plural": C.CRD_PLURAL,
                "singular": C.CRD_SINGULAR,
                "shortNames": [C.CRD_SHORT_NAME],
            },
            "scope": "Namespaced",
            "versions": [{
                "name": C.CRD_VERSION,
                "served": True,
                "storage": True,
                "subresources": {"status": {}},
                "additionalPrinterColumns": [
                    {"name": "Phase", "type": "string",
                     "jsonPath": ".status.phase"},
                    {"name": "Age", "type": "date",
                     "jsonPath": ".metadata.creationTimestamp"},
                ],
                "schema": {"openAPIV3Schema": {
                    "type": "object",
                    "properties": {
                        "spec": {
                            "type": "object",
                            "required": ["replicaSpecs"],
                            "properties": {
                                "restartingExitCode": {"type": "string"},
                                "frameworkType": {"type": "string"},
                                "faultTolerant": {"type": "boolean"},
                                "priority": {"type": "string"},
                                "schedulerName": {"type": "string"},
                                "timeLimit": {"type": "integer"},
                                "cleanPodPolicy": {
                                    "type": "string",
                                    "enum": ["All", "None"]},
                                "failPolicy": _ENDING_POLICY,
                                "completePolicy": _ENDING_POLICY,
                                "replicaSpecs": {
                                    "type": "object",
                                    "additionalProperties":
                                        _replica_spec_schema(),
                                },
                            },
                        },
                        "status": {
                            "type": "object",
                            "x-kubernetes-preserve-unknown-fields": True,
                        },
                    },
                }},
            }],
        },
    }
