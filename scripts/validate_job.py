#!/usr/bin/env python3
"""Validate AITrainingJob manifests offline (the admission checks the
controller applies: schema, policies, elastic ranges, HBM sizing).

    python scripts/validate_job.py manifests/examples/*.yaml
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import yaml

from trainingjob_operator_amd.api.defaults import set_defaults
from trainingjob_operator_amd.api.types import AITrainingJob
from trainingjob_operator_amd.api.validation import validate


def main(paths) -> int:
    if not paths:
        print(__doc__)
        return 2
    rc = 0
    for path in paths:
        for doc in yaml.safe_load_all(open(path)):
            if not doc:
                continue
            if doc.get("kind") != "AITrainingJob":
                print(f"{path}: skipping kind={doc.get('kind')}")
                continue
            job = set_defaults(AITrainingJob.from_dict(doc))
            errors = validate(job)
            if errors:
                rc = 1
                print(f"{path}: INVALID ({job.name})")
                for e in errors:
                    print(f"  - {e}")
            else:
                roles = {rt: rs.replicas
                         for rt, rs in job.spec.replica_specs.items()}
                print(f"{path}: OK ({job.name}, replicas={roles})")
    return rc


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
