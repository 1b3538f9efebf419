"""ROCm profiling utilities for the trial runner.

The reference times trials with ``timeit`` only.  On MI355X the
trial runner can additionally source:

- per-kernel time from ``rocprofv3 --kernel-trace --stats`` (kernel names let
  it verify the HIP fast-path actually ran);
- HBM high-water / utilization from ``rocm-smi`` and
  ``torch.cuda.memory_stats`` (feeding the OOM-infeasibility signal with a
  measured number instead of exception-string matching,
  reference FSDP.py:92-98).
"""

from __future__ import annotations

import csv
import io
import json
import os
import subprocess
from typing import Dict, List, Optional


def device_memory_stats(device: int = 0) -> Dict[str, int]:
    """Allocator high-water marks for one visible device (bytes)."""
    import torch

    if not torch.cuda.is_available():
        return {"allocated_peak": 0, "reserved_peak": 0, "total": 0}
    free, total = torch.cuda.mem_get_info(device)
    return {
        "allocated_peak": torch.cuda.max_memory_allocated(device),
        "reserved_peak": torch.cuda.max_memory_reserved(device),
        "free": free,
        "total": total,
    }


def rocm_smi_sample() -> List[Dict[str, str]]:
    """One rocm-smi sample per GPU: VRAM use and utilization.  Returns []
    when rocm-smi is unavailable (CPU container)."""
    try:
        out = subprocess.run(
            ["rocm-smi", "--showmeminfo", "vram", "--showuse", "--json"],
            capture_output=True,
            text=True,
            timeout=30,
        )
        if out.returncode != 0:
            return []
        data = json.loads(out.stdout)
        return [dict(v, card=k) for k, v in data.items() if k.startswith("card")]
    except Exception:
        return []


def rocprof_stats(
    cmd: List[str],
    workdir: Optional[str] = None,
    timeout: float = 1800.0,
) -> Optional[List[Dict[str, str]]]:
    """Run ``cmd`` under ``rocprofv3 --kernel-trace --stats`` and return the
    kernel-stats rows (name, calls, total/avg duration) sorted by total time.

    Per pool policy, counter collection (--pmc) is never combined with trace
    domains; this helper collects kernel timing only.
    """
    workdir = workdir or os.environ.get("TMPDIR", "/tmp")
    outdir = os.path.join(workdir, f"rocprof_{os.getpid()}")
    full = [
        "rocprofv3",
        "--kernel-trace",
        "--stats",
        "-d",
        outdir,
        "--output-format",
        "csv",
        "--",
        *cmd,
    ]
    try:
        # own process group: on timeout kill the WHOLE tree (rocprofv3's
        # traced cell spawns gang workers; killing only the wrapper would
        # leak them on the box)
        proc = subprocess.Popen(
            full,
            stdout=subprocess.DEVNULL,
            stderr=subprocess.DEVNULL,
            cwd=workdir,
            start_new_session=True,
        )
        try:
            proc.wait(timeout=timeout)
        except subprocess.TimeoutExpired:
            import signal

            os.killpg(os.getpgid(proc.pid), signal.SIGKILL)
            proc.wait(timeout=30)
            return None
    except Exception:
        return None
    rows: List[Dict[str, str]] = []
    for root, _dirs, files in os.walk(outdir):
        for f in files:
            if f.endswith("kernel_stats.csv"):
                with open(os.path.join(root, f), newline="") as fh:
                    rows.extend(csv.DictReader(fh))
    if not rows:
        return None

    def total_ns(r: Dict[str, str]) -> float:
        for k in ("TotalDurationNs", "DurationNs", "TOTAL_DURATION_NS"):
            if k in r:
                try:
                    return float(r[k])
                except ValueError:
                    pass
        return 0.0

    return sorted(rows, key=total_ns, reverse=True)


def summarize_kernel_stats(rows: List[Dict[str, str]], top: int = 20) -> str:
    """Human-readable top-k kernel table for profiles/ artifacts."""
    buf = io.StringIO()
    for r in rows[:top]:
        name = r.get("Name") or r.get("KernelName") or "?"
        buf.write(f"{name}\t{r}\n")
    return buf.getvalue()
