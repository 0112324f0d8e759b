"""RCCL-backend GPU tests at world_size=1.

A 1-GPU "nccl" (= RCCL on ROCm) process group with MSBN_FORCE_SYNC=1 runs the
exact multi-GPU code path: `all_gather_into_tensor` of packed BN stats, the
backward stat `all_reduce`, and the C++ reducer's bucket all-reduce on
ProcessGroupNCCL's dedicated comm stream — on real hardware, in the driver's
GPU test tier.  Each case runs in a subprocess: hipGraph capture segfaults
inside the pytest host process on ROCm (see test_gpu_fused.py), and a
dedicated process also keeps the NCCL communicator lifecycle clean.
"""

import os
import socket
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_case(case, timeout=420):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(_free_port())
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "nccl_world1_check.py"),
         case],
        capture_output=True, text=True, timeout=timeout, cwd=REPO, env=env,
    )
    assert r.returncode == 0 and f"CASE {case} OK" in r.stdout, (
        f"case {case} failed\nstdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    )


def test_nccl_world1_ddp_syncbn():
    """Full DDP+SyncBN train steps over RCCL; forced-sync == local path."""
    _run_case("ddp")


def test_nccl_world1_whole_step_hipgraph():
    """Whole train step (fwd+bwd+opt) captured in a hipGraph WITH the RCCL
    collectives inside; replays track the eager clone."""
    _run_case("graph")


def test_nccl_world1_input_h2d_and_backward_stats():
    """CPU inputs moved to GPU on a side stream in DDP.forward; per-param
    grad-ready timestamps surfaced via _get_ddp_logging_data."""
    _run_case("h2d")


def test_nccl_world1_allocator_stress():
    """Async bucket all-reduces stay correct under caching-allocator churn:
    30 DDP steps with random temporaries in flight track a plain clone."""
    _run_case("stress")
