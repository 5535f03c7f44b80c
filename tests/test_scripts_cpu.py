"""The two-script reference surface runs end to end on CPU (SURVEY.md §2a:
components 2/4 — serial and DDP trainers with the reference CLI/behavior)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(cmd, extra_env, timeout=300):
    env = dict(os.environ)
    env.update(extra_env)
    return subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                          text=True, timeout=timeout)


def test_cifar_example_serial_cpu(tmp_path):
    r = _run([sys.executable, "cifar_example.py"], {
        "MI355X_SYNTHETIC": "1",
        "MI355X_STEPS": "5",
        "MI355X_EPOCHS": "1",
        "MI355X_BATCH": "8",
        "MI355X_CKPT": str(tmp_path / "net.pth"),
    })
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Finished Training" in r.stdout
    assert "Accuracy of the network" in r.stdout
    assert (tmp_path / "net.pth").exists()  # MI355X_CKPT honored
    assert not os.path.exists(os.path.join(REPO, "cifar_net.pth"))


def test_cifar_example_ddp_2proc_cpu(tmp_path):
    r = _run([sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
              "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
              "--master-port", "29613", "cifar_example_ddp.py"], {
        "MI355X_SYNTHETIC": "1",
        "MI355X_STEPS": "4",
        "MI355X_EPOCHS": "1",
        "MI355X_BATCH": "8",
        "MI355X_CKPT": str(tmp_path / "net_ddp.pth"),
    }, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Accuracy of the network" in r.stdout
    assert (tmp_path / "net_ddp.pth").exists()  # MI355X_CKPT honored
