"""Integration tests (reference model: test/integration — fork the real
CLI/launch paths on localhost)."""
import json
import os
import subprocess
import sys

import pytest

from tests.parallel_util import REPO


def _env():
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    return env


def test_bench_contract_torchrun_np2():
    """The driver contract: torch.distributed.run launches bench.py and rank
    0 prints one valid JSON line with the whole-job metric."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29581", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1"],
        env=_env(), capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    line = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(line) == 1, r.stdout
    res = json.loads(line[0])
    assert res["n_gpus"] == 2
    assert res["config"]["parallelism"] == "dp2"
    assert res["value"] > 0
    assert res["data"] == "synthetic"


def test_mnist_example_np2():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bin", "hvdrun"), "-np", "2",
         sys.executable, os.path.join(REPO, "examples", "pytorch_mnist.py"),
         "--epochs", "1", "--samples", "512"],
        env=_env(), capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "Final averaged loss" in r.stdout, r.stdout


def test_synthetic_benchmark_example_np2():
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bin", "hvdrun"), "-np", "2",
         sys.executable,
         os.path.join(REPO, "examples", "pytorch_synthetic_benchmark.py"),
         "--no-cuda", "--batch-size", "2", "--num-warmup-batches", "1",
         "--num-batches-per-iter", "1", "--num-iters", "2"],
        env=_env(), capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "Total img/sec" in r.stdout, r.stdout


def test_bert_cpu_np2():
    """BERT-base fwd/bwd with the distributed optimizer on CPU."""
    from tests.parallel_util import run_workers
    run_workers(2, """
        from horovod_amd.models.bert import BertConfig, BertForPretraining
        torch.manual_seed(0)
        cfg = BertConfig(vocab_size=1000, hidden=64, layers=2, heads=4,
                         intermediate=128, max_seq=64)
        model = BertForPretraining(cfg)
        opt = hvd.DistributedOptimizer(
            torch.optim.Adam(model.parameters(), lr=1e-4),
            named_parameters=model.named_parameters())
        hvd.broadcast_parameters(model.state_dict(), root_rank=0)
        torch.manual_seed(rank)
        ids = torch.randint(0, 1000, (2, 16))
        tgt = torch.randint(0, 1000, (2, 16))
        for _ in range(2):
            opt.zero_grad()
            mlm, nsp = model(ids)
            loss = torch.nn.functional.cross_entropy(
                mlm.flatten(0, 1), tgt.flatten())
            loss.backward()
            opt.step()
        flat = torch.cat([p.detach().flatten() for p in model.parameters()])
        g = hvd.allgather(flat[:500].unsqueeze(0), name="bert_p")
        assert torch.allclose(g[0], g[1], atol=1e-6)
    """, timeout=300)


def test_elastic_cli_example(tmp_path):
    """`hvdrun --host-discovery-script` end-to-end with the shipped elastic
    MNIST example (reference: test_elastic_torch.py integration model)."""
    script = tmp_path / "discover.sh"
    script.write_text("#!/bin/sh\necho '127.0.0.1:2'\n")
    script.chmod(0o755)
    r = subprocess.run(
        [sys.executable, "-m", "horovod_amd.runner.launch",
         "--host-discovery-script", str(script), "--min-np", "1",
         "--max-np", "2",
         sys.executable, os.path.join(REPO, "examples",
                                      "pytorch_elastic_mnist.py"),
         "--epochs", "1", "--samples", "256"],
        env=_env(), capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stdout + r.stderr


def test_imagenet_example_np2(tmp_path):
    """The canonical ImageNet ResNet-50 example (synthetic mode) runs under
    the launcher, saves a checkpoint and resumes from it."""
    import subprocess
    import sys
    env = _env()
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bin", "hvdrun"), "-np", "2",
         sys.executable,
         os.path.join(REPO, "examples", "pytorch_imagenet_resnet50.py"),
         "--batch-size", "2", "--steps-per-epoch", "2", "--epochs", "1",
         "--checkpoint-format",
         str(tmp_path / "ck-{epoch}.pt")],
        env=env, capture_output=True, text=True, timeout=600, cwd=str(tmp_path))
    assert r.returncode == 0, r.stdout + r.stderr
    assert "img/s total" in r.stdout, r.stdout
    assert (tmp_path / "ck-0.pt").exists()
