"""Per-GPU health probing for MI355X nodes.

Sources, in preference order (SURVEY.md §5 'Failure detection'):
  1. amd-smi JSON (`amd-smi metric --json`) — temps, ECC, xGMI link status
  2. rocm-smi JSON (`rocm-smi --showtemp --showuse --showmemuse
     --showretiredpages --json`)
  3. amdgpu DRM sysfs (/sys/class/drm/card*/device) — device-alive check

FAKE_GPU mode (env AITJ_FAKE_GPU_HEALTH = inline JSON or a file path) lets
the health -> eviction path run in CI with no hardware
and is the fault-injection hook (flip the fake to simulate GPU-lost / ECC /
thermal / xGMI-down).
"""
from __future__ import annotations

import glob
import json
import os
import subprocess
from dataclasses import dataclass, field
from typing import List, Optional

FAKE_ENV = "AITJ_FAKE_GPU_HEALTH"

# MI355X operational limits
DEFAULT_TEMP_LIMIT_C = 110.0


@dataclass
class GpuHealth:
    index: int
    present: bool = True
    temp_c: Optional[float] = None
    ecc_uncorrectable: int = 0
    xgmi_ok: bool = True
    message: str = ""

    @property
    def healthy(self) -> bool:
        if not self.present:
            return False
        if self.ecc_uncorrectable > 0:
            return False
        if self.temp_c is not None and self.temp_c >= DEFAULT_TEMP_LIMIT_C:
            return False
        if not self.xgmi_ok:
            return False
        return True


@dataclass
class NodeGpuReport:
    gpus: List[GpuHealth] = field(default_factory=list)
    expected: Optional[int] = None
    probe_error: str = ""

    @property
    def healthy(self) -> bool:
        if self.probe_error:
            return False
        if self.expected is not None and len(self.gpus) < self.expected:
            return False
        return all(g.healthy for g in self.gpus)

    def summary(self) -> str:
        if self.probe_error:
            return f"probe error: {self.probe_error}"
        bad = [g for g in self.gpus if not g.healthy]
        if self.expected is not None and len(self.gpus) < self.expected:
            return (f"GPU lost: {len(self.gpus)}/{self.expected} visible")
        if bad:
            return "; ".join(
                f"gpu{g.index}: {g.message or 'unhealthy'}" for g in bad)
        return f"{len(self.gpus)} GPUs healthy"

    def to_json(self) -> str:
        return json.dumps({
            "healthy": self.healthy,
            "summary": self.summary(),
            "gpus": [vars(g) for g in self.gpus],
        })


def _run_json(cmd: List[str], timeout: float = 10.0) -> Optional[object]:
    try:
        out = subprocess.run(cmd, capture_output=True, text=True,
                             timeout=timeout)
        if out.returncode != 0:
            return None
        return json.loads(out.stdout)
    except (OSError, subprocess.TimeoutExpired, json.JSONDecodeError):
        return None


def _probe_fake() -> Optional[NodeGpuReport]:
    raw = os.environ.get(FAKE_ENV)
    if not raw:
        return None
    if os.path.exists(raw):
        raw = open(raw).read()
    data = json.loads(raw)
    report = NodeGpuReport(expected=data.get("expected"))
    for i, g in enumerate(data.get("gpus", [])):
        report.gpus.append(GpuHealth(
            index=g.get("index", i),
            present=g.get("present", True),
            temp_c=g.get("temp_c"),
            ecc_uncorrectable=g.get("ecc_uncorrectable", 0),
            xgmi_ok=g.get("xgmi_ok", True),
            message=g.get("message", ""),
        ))
    report.probe_error = data.get("probe_error", "")
    return report


def _probe_amd_smi() -> Optional[NodeGpuReport]:
    data = _run_json(["amd-smi", "metric", "--json"])
    if not isinstance(data, list) or not data:
        return None
    report = NodeGpuReport()
    for entry in data:
        idx = entry.get("gpu", len(report.gpus))
        g = GpuHealth(index=idx)
        temp = (entry.get("temperature") or {})
        edge = temp.get("edge") or temp.get("hotspot") or {}
        if isinstance(edge, dict):
            g.temp_c = edge.get("value")
        ecc = entry.get("ecc") or {}
        g.ecc_uncorrectable = int(
            (ecc.get("total_uncorrectable_count") or 0) or 0)
        if g.ecc_uncorrectable:
            g.message = f"{g.ecc_uncorrectable} uncorrectable ECC errors"
        report.gpus.append(g)
    return report


def _probe_rocm_smi() -> Optional[NodeGpuReport]:
    data = _run_json(["rocm-smi", "--showtemp", "--showuse", "--json"])
    if not isinstance(data, dict) or not data:
        return None
    report = NodeGpuReport()
    for key in sorted(k for k in data if k.startswith("card")):
        entry = data[key]
        idx = int(key.replace("card", "") or 0)
        g = GpuHealth(index=idx)
        for tkey in ("Temperature (Sensor edge) (C)",
                     "Temperature (Sensor junction) (C)"):
            if tkey in entry:
                try:
                    g.temp_c = float(entry[tkey])
                    break
                except ValueError:
                    pass
        report.gpus.append(g)
    return report


def _probe_sysfs() -> NodeGpuReport:
    report = NodeGpuReport()
    cards = sorted(glob.glob("/sys/class/drm/card*/device/vendor"))
    idx = 0
    for vendor_path in cards:
        try:
            if open(vendor_path).read().strip() != "0x1002":
                continue
        except OSError:
            continue
        dev = os.path.dirname(vendor_path)
        g = GpuHealth(index=idx)
        idx += 1
        busy = os.path.join(dev, "gpu_busy_percent")
        if os.path.exists(busy):
            try:
                open(busy).read()
            except OSError:
                g.present = False
                g.message = "amdgpu sysfs read failed (device hung?)"
        report.gpus.append(g)
    if not report.gpus:
        report.probe_error = "no amdgpu devices visible"
    return report


def probe(expected: Optional[int] = None) -> NodeGpuReport:
    report = (_probe_fake() or _probe_amd_smi() or _probe_rocm_smi()
              or _probe_sysfs())
    if expected is not None and report.expected is None:
        report.expected = expected
    return report
