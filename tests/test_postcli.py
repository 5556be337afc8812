"""Operator CLI surface (postcli.py) — CPU-safe paths only.

The reference operator tool is spacemeshos/post's postcli (reached from
the node docs); these tests pin the subcommand surface and the no-GPU
behavior (clean nonzero exit, not a traceback).
"""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLI = os.path.join(REPO, "postcli.py")


def run(*argv):
    return subprocess.run([sys.executable, CLI, *argv], cwd=REPO,
                          capture_output=True, text=True, timeout=120)


def test_help_lists_subcommands():
    r = run("--help")
    assert r.returncode == 0
    for sub in ("providers", "benchmark", "init", "prove", "verify"):
        assert sub in r.stdout


def test_providers_without_gpu_exits_nonzero_cleanly():
    import torch
    if torch.cuda.is_available():
        import pytest
        pytest.skip("GPU present: providers is nonempty here")
    r = run("providers")
    assert r.returncode == 1
    assert "no MI355X providers" in r.stderr
    assert "Traceback" not in r.stderr


def test_subcommand_helps():
    for sub in ("init", "prove", "verify", "benchmark"):
        r = run(sub, "--help")
        assert r.returncode == 0, r.stderr


def test_bad_arguments_exit_2():
    r = run("init", "--no-such-flag")
    assert r.returncode == 2
    r = run("frobnicate")
    assert r.returncode == 2
