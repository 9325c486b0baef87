"""The example entry points run end-to-end on CPU (reference
example/image-classification scripts are the user-facing surface)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=420):
    env = dict(os.environ)
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR", "MASTER_PORT"):
        env.pop(k, None)
    r = subprocess.run([sys.executable] + args, capture_output=True,
                       timeout=timeout, env=env, cwd=ROOT)
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    return r.stdout.decode() + r.stderr.decode()


@pytest.mark.timeout(600)
def test_train_mnist_cpu():
    out = _run(["examples/train_mnist.py", "--num-epochs", "1",
                "--batch-size", "64"])
    assert "Validation-accuracy" in out


@pytest.mark.timeout(600)
def test_train_cifar10_cpu():
    out = _run(["examples/train_cifar10.py", "--num-epochs", "1",
                "--batch-size", "64", "--network", "lenet"])
    assert "Epoch[0]" in out


@pytest.mark.timeout(600)
def test_benchmark_score_cpu_graceful():
    # CPU run: falls back / reports rather than crashing
    out = _run(["examples/benchmark_score.py", "--networks", "lenet",
                "--batch-sizes", "1", "--gpus", ""])
    assert "lenet" in out
