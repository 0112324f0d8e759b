"""The driver depends on bench.py's CLI + JSON contract; pin it."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _run_bench(extra):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py")] + extra,
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr
    line = r.stdout.strip().splitlines()[-1]
    return json.loads(line)


def test_bench_json_contract_cpu():
    out = _run_bench(["--gpus", "1", "--steps", "2", "--warmup", "1",
                      "--batch-size", "2", "--model", "resnet18",
                      "--dtype", "fp32", "--memory-format", "contiguous"])
    assert REQUIRED_KEYS.issubset(out.keys()), REQUIRED_KEYS - out.keys()
    assert out["n_gpus"] == 1
    assert out["steps"] == 2 and out["warmup"] == 1
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["config"]["global_batch"] == 2
    assert out["config"]["parallelism"] == "dp1"


def test_bench_distributed_launch_cpu():
    """torchrun-launched bench on CPU/gloo world 2 (the driver's launch
    shape), value aggregated over ranks."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", "--no-python", sys.executable,
         os.path.join(REPO, "bench.py"), "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--batch-size", "2", "--model", "resnet18",
         "--dtype", "fp32", "--memory-format", "contiguous"],
        capture_output=True, text=True, timeout=600, env=env, cwd=REPO,
    )
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["config"]["global_batch"] == 4


def test_bench_multirank_cpu_gloo(tmp_path):
    """bench.py end-to-end under torch.distributed.run with 2 CPU/gloo ranks
    (VERDICT r01 next-round #1 done-criterion): the distributed branch —
    msbn DDP wrap, barriers, MAX-over-ranks timing, single JSON line."""
    import json
    import subprocess
    import sys

    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "1", "--warmup", "0",
         "--batch-size", "2", "--model", "resnet18", "--dtype", "fp32",
         "--memory-format", "contiguous"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env,
    )
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2 and d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 4
    assert d["value"] > 0
