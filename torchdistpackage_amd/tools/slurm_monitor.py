"""SLURM job babysitter: submit, poll, auto-resubmit until COMPLETED.

Reference parity: /root/reference/tools/slurm_job_monitor.py:16-122 — the
job-level "poor-man's elastic" recovery tier: in-process
failures kill the job; this watchdog resubmits it.

Pure subprocess/slurm; no GPU or torch dependency.
"""

from __future__ import annotations

import subprocess
import time
from typing import Optional


def _run(cmd) -> str:
    return subprocess.check_output(cmd, text=True, shell=isinstance(cmd, str))


def submit_job(sbatch_script: str, extra_args: Optional[list] = None) -> str:
    """sbatch the script; returns the job id."""
    cmd = ["sbatch"] + (extra_args or []) + [sbatch_script]
    out = _run(cmd)
    # "Submitted batch job 123456"
    return out.strip().split()[-1]


def job_state(job_id: str) -> str:
    """Parse sacct fixed-width output for the job's primary state."""
    try:
        out = _run(["sacct", "-j", job_id, "--format=JobID,State", "-n", "-P"])
    except (OSError, subprocess.CalledProcessError):
        return "UNKNOWN"
    for line in out.splitlines():
        parts = line.split("|")
        if len(parts) >= 2 and parts[0].strip() == job_id:
            return parts[1].strip().split()[0]
    return "PENDING"


_FAILED = {"FAILED", "TIMEOUT", "NODE_FAIL", "PREEMPTED", "OUT_OF_MEMORY",
           "CANCELLED"}


def monitor_job(sbatch_script: str, poll_s: float = 30.0,
                max_resubmits: int = 100, extra_args: Optional[list] = None,
                verbose: bool = True) -> bool:
    """Submit and babysit a job: on failure states, resubmit (up to
    ``max_resubmits``); returns True when COMPLETED."""
    job_id = submit_job(sbatch_script, extra_args)
    resubmits = 0
    if verbose:
        print(f"[slurm_monitor] submitted {job_id}")
    while True:
        time.sleep(poll_s)
        state = job_state(job_id)
        if verbose:
            print(f"[slurm_monitor] job {job_id}: {state}")
        if state == "COMPLETED":
            return True
        if state in _FAILED:
            if resubmits >= max_resubmits:
                print(f"[slurm_monitor] giving up after {resubmits} resubmits")
                return False
            resubmits += 1
            job_id = submit_job(sbatch_script, extra_args)
            if verbose:
                print(f"[slurm_monitor] resubmitted as {job_id} "
                      f"({resubmits}/{max_resubmits})")
