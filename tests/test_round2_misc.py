"""Small unit tests for round-2 surfaces that the integration tests only
exercise implicitly."""

import os
import subprocess
import sys

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_convert_to_torch_batchnorm_roundtrip():
    """The benchmarks' --stock converter: msbn model -> torch BN modules with
    identical parameters, stats, and forward outputs."""
    import msbn
    from msbn.models import convert_to_torch_batchnorm

    torch.manual_seed(3)
    m1 = msbn.models.SimpleCNN(width=8)
    torch.manual_seed(3)
    m2 = convert_to_torch_batchnorm(msbn.models.SimpleCNN(width=8))
    for mod in m2.modules():
        assert not isinstance(mod, msbn.nn.batchnorm._NormBase)
    n_bn = sum(isinstance(m, torch.nn.modules.batchnorm._BatchNorm)
               for m in m2.modules())
    assert n_bn > 0
    x = torch.randn(4, 3, 8, 8)
    m1.eval(), m2.eval()
    torch.testing.assert_close(m1(x), m2(x), atol=1e-6, rtol=1e-6)
    m1.train(), m2.train()
    torch.testing.assert_close(m1(x), m2(x), atol=1e-5, rtol=1e-5)
    # running stats updated identically
    bn1 = [m for m in m1.modules()
           if isinstance(m, msbn.nn.batchnorm._NormBase)][0]
    bn2 = [m for m in m2.modules()
           if isinstance(m, torch.nn.modules.batchnorm._BatchNorm)][0]
    torch.testing.assert_close(bn1.running_mean, bn2.running_mean,
                               atol=1e-6, rtol=1e-6)


def test_gradbucket_surface():
    """GradBucket carrier exposes the stock dist.GradBucket API."""
    from msbn.parallel import GradBucket

    flat = torch.arange(6.0)
    views = [flat[:2], flat[2:]]
    b = GradBucket(flat, views, 3, True)
    assert b.buffer() is flat
    assert b.index() == 3 and b.is_last()
    assert len(b.gradients()) == 2
    b.set_buffer(torch.ones(6))
    assert torch.equal(flat, torch.ones(6))


def test_agent_store_env_in_workers(tmp_path):
    """msbn.run hosts the rendezvous TCPStore and advertises it to workers
    (TORCHELASTIC_USE_AGENT_STORE), so restarts reuse the same port."""
    script = tmp_path / "peek.py"
    script.write_text(
        "import os\n"
        "assert os.environ.get('TORCHELASTIC_USE_AGENT_STORE') == 'True', "
        "os.environ.get('TORCHELASTIC_USE_AGENT_STORE')\n"
        "import torch.distributed as dist\n"
        "dist.init_process_group('gloo', init_method='env://')\n"
        "dist.barrier()\n"
        "print('STORE_OK', os.environ['MASTER_PORT'])\n"
        "dist.destroy_process_group()\n"
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    r = subprocess.run(
        [sys.executable, "-m", "msbn.run", "--nproc_per_node=2", str(script)],
        capture_output=True, text=True, timeout=120, env=env, cwd=REPO,
    )
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    ports = {l.split()[1] for l in r.stdout.splitlines()
             if l.startswith("STORE_OK")}
    assert len(ports) == 1  # both workers used the agent-hosted store port
