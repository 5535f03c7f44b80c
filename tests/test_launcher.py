"""Launcher tests: env contract + child-failure handling."""

import subprocess
import sys
import textwrap


def _write(tmp_path, name, body):
    p = tmp_path / name
    p.write_text(textwrap.dedent(body))
    return str(p)


def test_env_contract_and_success(tmp_path):
    script = _write(tmp_path, "ok.py", """
        import os, sys
        rank = int(os.environ["RANK"])
        world = int(os.environ["WORLD_SIZE"])
        local = int(os.environ["LOCAL_RANK"])
        assert os.environ["MASTER_ADDR"] == "127.0.0.1"
        assert "MASTER_PORT" in os.environ
        assert rank == local and 0 <= rank < world == 3
        sys.exit(0)
    """)
    r = subprocess.run([sys.executable, "-m", "mi355x.launcher",
                        "--nproc-per-node", "3", "--master-port", "29701",
                        script], timeout=60)
    assert r.returncode == 0


def test_child_failure_kills_job(tmp_path):
    script = _write(tmp_path, "fail.py", """
        import os, sys, time
        if int(os.environ["RANK"]) == 1:
            sys.exit(7)   # one rank dies...
        time.sleep(30)    # ...the others would hang without the launcher
    """)
    r = subprocess.run([sys.executable, "-m", "mi355x.launcher",
                        "--nproc-per-node", "2", "--master-port", "29703",
                        script], timeout=60)
    assert r.returncode == 7  # propagated, well before the 30s sleep
