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
