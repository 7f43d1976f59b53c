#!/usr/bin/env python3
"""Convert a training checkpoint between parallel topologies.

    python scripts/reshard_ckpt.py --model llama3-8b \
        --in-dir /ckpt --src pp=4 --out-dir /ckpt-tp --dst tp=2

Layouts: full | pp=N | tp=N | ep=N | zero=N — directory conventions match the
launcher (full -> DIR/, pp -> DIR/stage{i}/, tp -> DIR/tp{r}/,
ep -> DIR/ep{r}/ for MoE expert shards). Parameters and
AdamW moments are both converted; the target streams load directly via
the launcher's normal resume path.
"""
import argparse
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

from trainingjob_operator_amd.launcher.reshard import reshard  # noqa: E402


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model", required=True)
    ap.add_argument("--in-dir", required=True)
    ap.add_argument("--out-dir", required=True)
    ap.add_argument("--src", required=True, help="full | pp=N | tp=N | ep=N | zero=N")
    ap.add_argument("--dst", required=True, help="full | pp=N | tp=N | ep=N | zero=N")
    a = ap.parse_args(argv)
    for path in reshard(a.model, a.in_dir, a.out_dir, a.src, a.dst):
        print(path)
    return 0


if __name__ == "__main__":
    sys.exit(main())
