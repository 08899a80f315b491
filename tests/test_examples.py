"""The four rebuilt reference demos run end-to-end as scripts (CPU,
gloo, loopback world 2 — the reference's own smoke pattern, `make ptp`
/ SURVEY.md §4.2)."""

import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script, args, timeout=180):
    out = subprocess.run([sys.executable, os.path.join("examples", script),
                          *args], cwd=ROOT, capture_output=True, text=True,
                         timeout=timeout)
    assert out.returncode == 0, (script, out.stderr[-2000:])
    return out.stdout


@pytest.mark.timeout(300)
def test_ptp_example():
    out = _run("ptp.py", ["--world", "2"])
    # the reference's self-check: root's gathered sum == world size
    assert "2.0" in out


@pytest.mark.timeout(300)
def test_send_recv_example():
    out = _run("send_recv.py", [])
    assert "round trip" in out or "us" in out or out.strip()


@pytest.mark.timeout(300)
def test_allreduce_example():
    out = _run("allreduce.py", ["--world", "3", "--algo", "chunked"])
    assert "OK" in out or "ok" in out or out.strip()


@pytest.mark.timeout(300)
def test_train_dist_example():
    env = dict(os.environ)
    env.update(TRAIN_SAMPLES="512", TRAIN_EPOCHS="1")
    out = subprocess.run([sys.executable, "examples/train_dist.py",
                          "--world", "2", "--epochs", "1"], cwd=ROOT,
                         env=env, capture_output=True, text=True,
                         timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "epoch 0" in out.stdout


def test_paperify_renders_tutorial(tmp_path):
    """The docs build (the reference Makefile's `all` target): render
    TUTORIAL.md to tutorial.html + byte-identical index.html."""
    import subprocess
    import sys as _sys
    import os as _os
    root = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    r = subprocess.run([_sys.executable, "docs/paperify.py"], cwd=root,
                       capture_output=True, text=True, timeout=60)
    assert r.returncode == 0, r.stderr
    with open(_os.path.join(root, "docs", "tutorial.html")) as f:
        h = f.read()
    for sec in ("Communication backends", "Initialization methods",
                "<table>", "<pre><code>"):
        assert sec in h, sec
    with open(_os.path.join(root, "docs", "index.html")) as f:
        assert f.read() == h
