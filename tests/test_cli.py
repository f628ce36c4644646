"""Reference-parity CLI drivers (SURVEY.md §2.1 main/__main__ rows): the
module entry and the root model.py shim must run end-to-end on CPU (the
reference's only working path — SURVEY.md §3.2) and write the loguru-style
rotating log the reference writes (model.py:160)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(cmd, cwd):
    # the package is in-tree (not pip-installed into site-packages, so the
    # round-end native-code check sees the in-tree .so) — put the repo on
    # PYTHONPATH like any user launching from a checkout
    env = dict(os.environ, PYTHONPATH=REPO)
    return subprocess.run(cmd, capture_output=True, text=True, timeout=600,
                          cwd=cwd, env=env)


def test_module_cli_cpu(tmp_path):
    r = _run([sys.executable, "-m", "tree_attention_torch_amd",
              "--seq-len", "512", "--num-heads", "4", "--head-dim", "64"],
             cwd=str(tmp_path))  # run from tmp so the log lands there
    assert r.returncode == 0, r.stderr[-2000:]
    assert (tmp_path / "tree_attention_log.log").exists()


def test_model_py_shim_cpu(tmp_path):
    env = dict(os.environ, PYTHONPATH=REPO)
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "model.py"),
         "--seq-len", "256", "--num-heads", "2", "--head-dim", "64"],
        capture_output=True, text=True, timeout=600, cwd=str(tmp_path),
        env=env)
    assert r.returncode == 0, r.stderr[-2000:]
