"""msbn.run — single/multi-node process launcher (torchrun-equivalent core).

Spawns one worker process per GPU with the standard env contract
(SURVEY.md §5.6 / run.py:187-232 of the stock launcher):

  MASTER_ADDR, MASTER_PORT, RANK, LOCAL_RANK, WORLD_SIZE, LOCAL_WORLD_SIZE,
  GROUP_RANK, NODE_RANK, TORCHELASTIC_RESTART_COUNT, TORCH_NCCL_ASYNC_ERROR_HANDLING=1

plus optional ``--local-rank=<r>`` argv injection (legacy ``msbn.launch``
contract, README.md:15-19).  On any worker failure the whole group is torn
down and restarted (up to ``--max-restarts``), mirroring the elastic agent's
restart-on-failure loop (SURVEY.md §2.2 "run.py:261-277").

Usage:
    python -m msbn.run --nproc-per-node=8 train.py ARGS...
"""

import argparse
import os
import signal
import socket
import subprocess
import sys
import time
import uuid
from typing import List, Optional


def _free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _device_count() -> int:
    try:
        import torch

        n = torch.cuda.device_count()
        return n if n > 0 else 1
    except Exception:
        return 1


def parse_args(argv: Optional[List[str]] = None, use_env_default: bool = True):
    p = argparse.ArgumentParser(
        description="msbn distributed launcher (one process per GPU)"
    )
    p.add_argument(
        "--nproc-per-node", "--nproc_per_node", type=str, default="1",
        help="number of workers on this node ('auto' = GPU count)",
    )
    p.add_argument("--nnodes", type=int, default=1)
    p.add_argument("--node-rank", "--node_rank", type=int, default=0)
    p.add_argument(
        "--master-addr", "--master_addr", type=str, default="127.0.0.1"
    )
    p.add_argument("--master-port", "--master_port", type=int, default=None)
    p.add_argument("--max-restarts", "--max_restarts", type=int, default=0)
    p.add_argument("--standalone", action="store_true",
                   help="single-node shortcut (torchrun compat): implies "
                        "127.0.0.1 rendezvous on a free port")
    p.add_argument("--local-addr", "--local_addr", type=str, default=None,
                   help="accepted for torchrun compatibility")
    p.add_argument("--rdzv-backend", "--rdzv_backend", type=str, default=None,
                   help="accepted for torchrun compatibility (static only)")
    p.add_argument("--rdzv-endpoint", "--rdzv_endpoint", type=str, default=None,
                   help="host:port (torchrun compat); overrides master addr/port")
    p.add_argument(
        "--use-env", "--use_env", action="store_true", default=use_env_default,
        help="pass LOCAL_RANK via env only (no --local-rank argv)",
    )
    p.add_argument(
        "--no-python", "--no_python", action="store_true",
        help="run training_script directly instead of `python training_script`",
    )
    p.add_argument("--module", "-m", action="store_true",
                   help="run the script as a python module (python -m script)")
    p.add_argument("--run-id", "--run_id", type=str, default=None)
    p.add_argument("training_script", type=str)
    p.add_argument("training_script_args", nargs=argparse.REMAINDER)
    return p.parse_args(argv)


def _worker_env(args, local_rank: int, nproc: int, port: int, restart: int):
    env = dict(os.environ)
    world_size = nproc * args.nnodes
    rank = args.node_rank * nproc + local_rank
    env.update(
        MASTER_ADDR=args.master_addr,
        MASTER_PORT=str(port),
        RANK=str(rank),
        LOCAL_RANK=str(local_rank),
        WORLD_SIZE=str(world_size),
        LOCAL_WORLD_SIZE=str(nproc),
        GROUP_RANK=str(args.node_rank),
        NODE_RANK=str(args.node_rank),
        TORCHELASTIC_RESTART_COUNT=str(restart),
        TORCHELASTIC_RUN_ID=args.run_id or uuid.uuid4().hex[:8],
        OMP_NUM_THREADS=env.get("OMP_NUM_THREADS", "1"),
    )
    env.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
    if getattr(args, "_agent_store_hosted", False):
        # workers attach to the launcher-hosted TCPStore instead of rank 0
        # hosting one (stock rendezvous.py:160 contract)
        env["TORCHELASTIC_USE_AGENT_STORE"] = str(True)
    return env


def _spawn_group(args, nproc: int, port: int, restart: int):
    procs = []
    for local_rank in range(nproc):
        cmd: List[str] = []
        if not args.no_python:
            cmd = [sys.executable, "-u"]
            if args.module:
                cmd.append("-m")
        cmd.append(args.training_script)
        script_args = list(args.training_script_args)
        if not args.use_env:
            # msbn.launch (legacy shim) injects the underscore spelling the
            # reference README's argparse registers (README.md:15-19) —
            # argparse does NOT alias the two spellings
            flag = (
                "--local_rank"
                if getattr(args, "legacy_underscore_flag", False)
                else "--local-rank"
            )
            script_args = [f"{flag}={local_rank}"] + script_args
        cmd.extend(script_args)
        env = _worker_env(args, local_rank, nproc, port, restart)
        procs.append(subprocess.Popen(cmd, env=env))
    return procs


def _kill_group(procs):
    for p in procs:
        if p.poll() is None:
            p.send_signal(signal.SIGTERM)
    deadline = time.time() + 10
    for p in procs:
        try:
            p.wait(timeout=max(0.1, deadline - time.time()))
        except subprocess.TimeoutExpired:
            p.kill()


def _host_agent_store(addr: str, port: int):
    """Host the rendezvous TCPStore in the LAUNCHER (stock elastic-agent
    behavior, local_elastic_agent.py + TORCHELASTIC_USE_AGENT_STORE): the
    store outlives worker restarts, so re-spawned groups rendezvous on the
    same port with no rank-0 hosting race."""
    try:
        from torch.distributed import TCPStore

        return TCPStore(addr, port, is_master=True, multi_tenant=True)
    except Exception:
        return None  # workers fall back to rank-0-hosted store


def run(args) -> int:
    if getattr(args, "rdzv_endpoint", None):
        host, _, port = args.rdzv_endpoint.partition(":")
        if host:
            args.master_addr = host
        if port:
            args.master_port = int(port)
    if getattr(args, "standalone", False):
        args.master_addr = "127.0.0.1"
        args.master_port = None
    if args.nnodes > 1 and args.master_port is None:
        # Every node must agree on MASTER_PORT; a per-node _free_port() would
        # disagree and rendezvous could never form.  Match the stock
        # launchers' fixed default.
        args.master_port = 29500
    auto_nproc = str(args.nproc_per_node) in ("auto", "gpu")
    nproc = _device_count() if auto_nproc else int(args.nproc_per_node)
    if args.run_id is None:
        args.run_id = uuid.uuid4().hex[:8]
    restarts = 0
    # one port + one agent-hosted store for the whole (restartable) job
    fixed_port = args.master_port or _free_port()
    agent_store = None
    if args.nnodes == 1 or args.node_rank == 0:
        agent_store = _host_agent_store(args.master_addr, fixed_port)
    args._agent_store_hosted = agent_store is not None  # ref keeps the store alive
    while True:
        port = fixed_port
        if auto_nproc and restarts > 0:
            # elastic-style scale-down: if a device died with the failed
            # group, respawn with the surviving count
            new_nproc = _device_count()
            if new_nproc != nproc:
                print(
                    f"[msbn.run] device count changed {nproc} -> {new_nproc}; "
                    "re-forming group at the new size",
                    file=sys.stderr,
                )
                nproc = new_nproc
        procs = _spawn_group(args, nproc, port, restarts)
        failed_rc = None
        live = list(procs)
        while live and failed_rc is None:
            time.sleep(0.2)
            for p in list(live):
                rc = p.poll()
                if rc is None:
                    continue
                live.remove(p)
                if rc != 0:
                    failed_rc = rc
                    break
        if failed_rc is None:
            return 0
        _kill_group(procs)
        if restarts >= args.max_restarts:
            print(
                f"[msbn.run] worker failed with exit code {failed_rc}; "
                f"no restarts left ({restarts}/{args.max_restarts})",
                file=sys.stderr,
            )
            return failed_rc
        restarts += 1
        print(
            f"[msbn.run] worker failed (rc={failed_rc}); restarting group "
            f"({restarts}/{args.max_restarts})",
            file=sys.stderr,
        )


def main(argv: Optional[List[str]] = None) -> int:
    return run(parse_args(argv))


if __name__ == "__main__":
    sys.exit(main())
