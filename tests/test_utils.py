"""utils coverage: logger fields, master_print, env helpers, comm log."""

import io

import torch

import msbn
from msbn.utils.logging import CommLog, master_print


def test_master_print_rank0_only(capsys, monkeypatch):
    import torch.distributed as dist

    if dist.is_available() and dist.is_initialized():
        # a prior test holds a world-1 group: rank 0 -> prints
        master_print("hello")
        assert "hello" in capsys.readouterr().out
        return
    monkeypatch.setenv("RANK", "0")
    master_print("hello")
    assert "hello" in capsys.readouterr().out
    monkeypatch.setenv("RANK", "3")
    master_print("nope")
    assert "nope" not in capsys.readouterr().out


def test_comm_log_ring():
    log = CommLog(capacity=2)
    log.enabled = True
    log.record("all_reduce", 100)
    log.record("all_gather", 200)
    log.record("broadcast", 300)
    assert len(log.buf) == 2
    buf = io.StringIO()
    log.dump(buf)
    out = buf.getvalue()
    assert "broadcast" in out and "all_reduce" not in out


def test_add_local_rank_arg(monkeypatch):
    import argparse

    from msbn.utils.env import add_local_rank_arg, get_local_rank

    p = argparse.ArgumentParser()
    add_local_rank_arg(p)
    a = p.parse_args(["--local-rank=3"])
    assert a.local_rank == 3
    a2 = p.parse_args(["--local_rank", "5"])
    assert a2.local_rank == 5
    monkeypatch.setenv("LOCAL_RANK", "7")
    assert get_local_rank() == 7


def test_ddp_logging_data(tmp_path):
    import torch.distributed as dist

    if not dist.is_initialized():
        dist.init_process_group(
            "gloo", init_method=f"file://{tmp_path}/pgx", rank=0, world_size=1
        )
    net = msbn.parallel.DistributedDataParallel(torch.nn.Linear(4, 2))
    net(torch.randn(2, 4)).sum().backward()
    d = net._get_ddp_logging_data()
    assert d["ints_map"]["world_size"] == 1
    assert d["ints_map"]["num_parameter_tensors"] == 2
    assert d["ints_map"]["forward_count"] == 1
    assert d["strs_map"]["module_name"] == "Linear"


def test_init_distributed_helper(monkeypatch, tmp_path):
    """msbn.utils.env.init_distributed wires env:// like README step 2."""
    import subprocess
    import sys
    import os as _os

    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    script = tmp_path / "e.py"
    script.write_text(
        "import sys; sys.path.insert(0, %r)\n"
        "import os\n"
        "from msbn.utils.env import init_distributed\n"
        "os.environ['MASTER_PORT'] = '29551'\n"
        "rank, world = init_distributed(local_rank=0, world_size=1)\n"
        "assert (rank, world) == (0, 1)\n"
        "import torch.distributed as dist\n"
        "assert dist.get_backend() == 'gloo'\n"
        "print('ENVINIT_OK')\n" % repo
    )
    r = subprocess.run([sys.executable, str(script)], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stderr
    assert "ENVINIT_OK" in r.stdout


def test_combined_view_adjacent_and_not():
    from msbn.nn.functions import _combined_view

    base = torch.arange(8, dtype=torch.float32)
    a, b = base[:4], base[4:]
    v, copied = _combined_view(a, b, 4)
    assert not copied
    v.add_(1)  # in-place through the view reaches the originals
    assert a[0].item() == 1.0 and b[0].item() == 5.0
    # non-adjacent -> cat fallback
    c, d = torch.zeros(4), torch.ones(4)
    v2, copied2 = _combined_view(c, d, 4)
    assert copied2 and v2.shape == (8,)


def test_dtype_sorted_grouping():
    """Buffer/state broadcasts must group by dtype (one flat chunk per dtype,
    not one per consecutive run: BN models interleave fp32 stats with int64
    counters)."""
    from msbn.parallel.distributed import DistributedDataParallel as DDP

    ts = [torch.zeros(2), torch.zeros(2, dtype=torch.long), torch.zeros(2),
          torch.zeros(2, dtype=torch.long), torch.zeros(2)]
    out = DDP._dtype_sorted(ts)
    kinds = [t.dtype for t in out]
    # all fp32 first (first-seen dtype), then all int64 — 2 runs total
    runs = 1 + sum(1 for a, b in zip(kinds, kinds[1:]) if a != b)
    assert runs == 2, kinds
