"""End-to-end launcher test: the reference command line (README.md:98-100)
through msbn.launch / msbn.run on CPU/gloo (BASELINE.json config 1)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import argparse, os, sys
sys.path.insert(0, {repo!r})
import torch, torch.distributed as dist
import msbn

# Step 1 (README.md:15-19): the --local_rank contract
parser = argparse.ArgumentParser()
parser.add_argument('--local_rank', '--local-rank', type=int,
                    default=int(os.environ.get('LOCAL_RANK', 0)),
                    dest='local_rank')
parser.add_argument('--ngpu', type=int, default=2)
args = parser.parse_args()

# Step 2 (README.md:26-36): init_process_group env:// (gloo here: no GPU)
dist.init_process_group('gloo', init_method='env://',
                        world_size=args.ngpu, rank=args.local_rank)

# Steps 3-5: convert + DDP + sampler
torch.manual_seed(0)
net = msbn.nn.SyncBatchNorm.convert_sync_batchnorm(msbn.models.SimpleCNN(width=8))
net = msbn.parallel.DistributedDataParallel(net)
ds = msbn.data.SyntheticImageDataset(length=16, shape=(3, 16, 16), num_classes=10)
sampler = msbn.data.DistributedSampler(ds, num_replicas=args.ngpu,
                                       rank=args.local_rank)
loader = torch.utils.data.DataLoader(ds, batch_size=4, sampler=sampler,
                                     drop_last=True, num_workers=0)
opt = torch.optim.SGD(net.parameters(), lr=0.01)
loss_fn = torch.nn.CrossEntropyLoss()
for epoch in range(2):
    sampler.set_epoch(epoch)
    for x, y in loader:
        opt.zero_grad()
        loss = loss_fn(net(x), y)
        loss.backward()
        opt.step()
if args.local_rank == 0:
    print('LAUNCH_OK', loss.item())
"""


def _run_launcher(module, extra_args, tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER.format(repo=REPO))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    cmd = [
        sys.executable, "-m", module, "--nproc_per_node=2",
        "--master-addr", "127.0.0.1",
    ] + extra_args + [str(script), "--ngpu=2"]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=180,
                       env=env, cwd=REPO)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "LAUNCH_OK" in r.stdout


def test_msbn_launch_argv_contract(tmp_path):
    """legacy launch: --local-rank=<r> injected on argv."""
    _run_launcher("msbn.launch", [], tmp_path)


def test_msbn_run_env_contract(tmp_path):
    """torchrun-style: LOCAL_RANK via env only."""
    _run_launcher("msbn.run", [], tmp_path)


def test_restart_on_failure(tmp_path):
    script = tmp_path / "flaky.py"
    marker = tmp_path / "marker"
    script.write_text(
        "import os, sys\n"
        f"m = {str(marker)!r}\n"
        "if not os.path.exists(m):\n"
        "    open(m, 'w').close()\n"
        "    if os.environ['RANK'] == '1':\n"
        "        sys.exit(3)\n"
        "print('SECOND_TRY_OK')\n"
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, "-m", "msbn.run", "--nproc_per_node=2",
         "--max-restarts=1", str(script)],
        capture_output=True, text=True, timeout=120, env=env, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr
    assert "SECOND_TRY_OK" in r.stdout


def test_example_script_end_to_end(tmp_path):
    """The shipped reference-recipe example runs under msbn.launch (CPU/gloo)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    ck = tmp_path / "ex.pt"
    r = subprocess.run(
        [sys.executable, "-m", "msbn.launch", "--nproc_per_node=2",
         os.path.join(REPO, "examples", "distributed_train.py"),
         "--ngpu", "2", "--epochs", "1", "--batch-size", "8",
         "--ckpt", str(ck)],
        capture_output=True, text=True, timeout=300, env=env, cwd=REPO,
    )
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "TRAIN_OK" in r.stdout
    assert ck.exists()


def test_multi_node_env_contract(tmp_path):
    """Two msbn.run agents on one host emulate a 2-node x 2-proc job:
    global RANK/WORLD_SIZE must come out right on every worker."""
    script = tmp_path / "check.py"
    script.write_text(
        "import os, sys\n"
        "import torch.distributed as dist\n"
        "dist.init_process_group('gloo', init_method='env://')\n"
        "t = __import__('torch').tensor([dist.get_rank()])\n"
        "dist.all_reduce(t)\n"
        "assert dist.get_world_size() == 4\n"
        "assert t.item() == 6, t.item()\n"  # 0+1+2+3
        "if dist.get_rank() == 0:\n"
        "    print('MULTINODE_OK')\n"
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    port = 29541
    agents = []
    for node in range(2):
        agents.append(subprocess.Popen(
            [sys.executable, "-m", "msbn.run", "--nproc_per_node=2",
             "--nnodes=2", f"--node-rank={node}",
             "--master-addr", "127.0.0.1", "--master-port", str(port),
             str(script)],
            env=env, cwd=REPO, stdout=subprocess.PIPE, text=True,
        ))
    outs = []
    for a in agents:
        out, _ = a.communicate(timeout=180)
        outs.append(out)
        assert a.returncode == 0, out
    assert any("MULTINODE_OK" in o for o in outs)


def test_standalone_flag(tmp_path):
    script = tmp_path / "s.py"
    script.write_text(
        "import os\n"
        "assert os.environ['MASTER_ADDR'] == '127.0.0.1'\n"
        "print('STANDALONE_OK', os.environ['RANK'])\n"
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, "-m", "msbn.run", "--standalone",
         "--nproc_per_node=2", str(script)],
        capture_output=True, text=True, timeout=120, env=env, cwd=REPO,
    )
    assert r.returncode == 0, r.stderr
    assert "STANDALONE_OK 0" in r.stdout and "STANDALONE_OK 1" in r.stdout


def test_stock_torch_distributed_launch_interop(tmp_path):
    """The reference's exact launcher (`python -m torch.distributed.launch`,
    README.md:98-100) drives msbn workers unchanged — env/argv contracts
    are byte-compatible."""
    script = tmp_path / "worker.py"
    script.write_text(WORKER.format(repo=REPO))
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.launch",
         "--nproc_per_node=2", "--master_addr", "127.0.0.1",
         "--master_port", "29561", str(script), "--ngpu=2"],
        capture_output=True, text=True, timeout=180, env=env, cwd=REPO,
    )
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert "LAUNCH_OK" in r.stdout
