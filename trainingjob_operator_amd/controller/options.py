"""Operator options/flags (reference: cmd/app/options/options.go:12-72).

Flag names kept where they map onto the reference's
(--namespace, --resync-period, --thread-num, creating-failure knobs).
"""
from __future__ import annotations

import argparse
from dataclasses import dataclass

from ..policy.engine import CreatingFailurePolicy


@dataclass
class OperatorOptions:
    master: str = ""
    kubeconfig: str = ""
    run_in_cluster: bool = False
    thread_num: int = 1
    namespace: str = ""              # "" == all namespaces
    resync_period: float = 10.0
    creating_restart_period: float = 5.0
    creating_duration_period: float = 15 * 60.0
    enable_creating_failed: bool = False
    gc_period: float = 600.0
    leader_elect: bool = True
    lease_namespace: str = "kube-system"
    lease_name: str = "trainingjob-operator"
    lease_duration: float = 15.0
    renew_deadline: float = 10.0
    retry_period: float = 3.0
    metrics_port: int = 0            # 0 == disabled
    # EdlPolicy=Auto control loop (new semantics; the reference declared
    # the fields but never read them — SURVEY.md §C15)
    elastic_unschedulable_grace: float = 60.0
    elastic_scaleup_interval: float = 300.0

    def creating_failure_policy(self) -> CreatingFailurePolicy:
        return CreatingFailurePolicy(
            creating_restart_seconds=self.creating_restart_period,
            creating_duration_seconds=self.creating_duration_period,
            enable_creating_failed=self.enable_creating_failed,
        )

    @classmethod
    def from_yaml(cls, path: str) -> "OperatorOptions":
        """Optional YAML config (SURVEY.md §5 'Config / flag system'):
        keys are the dataclass field names; unknown keys are rejected."""
        import yaml
        with open(path) as f:
            data = yaml.safe_load(f) or {}
        fields = {f.name for f in cls.__dataclass_fields__.values()}
        unknown = set(data) - fields
        if unknown:
            raise ValueError(f"{path}: unknown option(s) {sorted(unknown)}")
        return cls(**data)

    @classmethod
    def add_flags(cls, ap: argparse.ArgumentParser) -> None:
        d = cls()
        ap.add_argument("--config", default="",
                        help="YAML config file (field-named keys); "
                             "explicitly-passed flags override its values")
        ap.add_argument("--master", default=d.master,
                        help="API server URL (default: in-cluster or proxy)")
        ap.add_argument("--kubeconfig", default=d.kubeconfig)
        ap.add_argument("--run-in-cluster", action="store_true")
        ap.add_argument("--thread-num", type=int, default=d.thread_num)
        ap.add_argument("--namespace", default=d.namespace)
        ap.add_argument("--resync-period", type=float,
                        default=d.resync_period)
        ap.add_argument("--creating-restart-period", type=float,
                        default=d.creating_restart_period)
        ap.add_argument("--creating-duration-period", type=float,
                        default=d.creating_duration_period)
        ap.add_argument("--enable-creating-failed", action="store_true")
        ap.add_argument("--gc-period", type=float, default=d.gc_period)
        ap.add_argument("--leader-elect", dest="leader_elect",
                        action="store_true", default=d.leader_elect)
        ap.add_argument("--no-leader-elect", dest="leader_elect",
                        action="store_false")
        ap.add_argument("--metrics-port", type=int, default=d.metrics_port)
        ap.add_argument("--elastic-unschedulable-grace", type=float,
                        default=d.elastic_unschedulable_grace,
                        help="seconds a pod may sit unschedulable before an "
                             "EdlPolicy=Auto job scales down toward "
                             "minReplicas")
        ap.add_argument("--elastic-scaleup-interval", type=float,
                        default=d.elastic_scaleup_interval,
                        help="seconds between EdlPolicy=Auto +1 scale-up "
                             "probes toward maxReplicas")

    @classmethod
    def from_args(cls, args: argparse.Namespace) -> "OperatorOptions":
        if getattr(args, "config", ""):
            base = cls.from_yaml(args.config)
            # explicit CLI values override the file's (argparse gives us
            # the default when a flag was not passed; detect overrides by
            # comparing to the dataclass defaults)
            d = cls()
            for f in cls.__dataclass_fields__.values():
                cli = getattr(args, f.name, None)
                if cli is not None and cli != getattr(d, f.name):
                    setattr(base, f.name, cli)
            return base
        return cls(
            master=args.master, kubeconfig=args.kubeconfig,
            run_in_cluster=args.run_in_cluster, thread_num=args.thread_num,
            namespace=args.namespace, resync_period=args.resync_period,
            creating_restart_period=args.creating_restart_period,
            creating_duration_period=args.creating_duration_period,
            enable_creating_failed=args.enable_creating_failed,
            gc_period=args.gc_period, leader_elect=args.leader_elect,
            metrics_port=args.metrics_port,
            elastic_unschedulable_grace=args.elastic_unschedulable_grace,
            elastic_scaleup_interval=args.elastic_scaleup_interval,
        )
