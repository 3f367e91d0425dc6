"""CLI smoke tests (reference tests/test_cli_scripts.py, test_start_server.py)."""

import os
import re
import subprocess
import sys
import time

import pytest


def test_run_dht_starts_and_peers_connect():
    proc = subprocess.Popen(
        [sys.executable, "-m", "hivemind_amd.hivemind_cli.run_dht", "--refresh_period", "1"],
        stderr=subprocess.PIPE,
        text=True,
        encoding="utf-8",
    )
    try:
        endpoint = None
        deadline = time.monotonic() + 30
        first_lines = []
        while time.monotonic() < deadline:
            line = proc.stderr.readline()
            first_lines.append(line)
            match = re.search(r"Running a DHT instance at (\S+)", line)
            if match:
                endpoint = match.group(1)
                break
        assert endpoint is not None, f"no readiness line in: {first_lines}"

        # a second peer joins via --initial_peers
        proc2 = subprocess.Popen(
            [sys.executable, "-m", "hivemind_amd.hivemind_cli.run_dht", "--initial_peers", endpoint,
             "--refresh_period", "1"],
            stderr=subprocess.PIPE,
            text=True,
            encoding="utf-8",
        )
        try:
            saw_peer = False
            deadline = time.monotonic() + 30
            while time.monotonic() < deadline:
                line = proc2.stderr.readline()
                if re.search(r"DHT status: [1-9]\d* known peers", line):
                    saw_peer = True
                    break
            assert saw_peer, "second peer never saw the first"
        finally:
            proc2.terminate()
            proc2.wait(10)
    finally:
        proc.terminate()
        proc.wait(10)


def test_run_server_starts():
    proc = subprocess.Popen(
        [sys.executable, "-m", "hivemind_amd.hivemind_cli.run_server",
         "--num_experts", "2", "--expert_pattern", "clitest.[0:16]",
         "--expert_cls", "ffn", "--hidden_dim", "16", "--device", "cpu"],
        stderr=subprocess.PIPE,
        text=True,
        encoding="utf-8",
    )
    try:
        ready = False
        deadline = time.monotonic() + 45
        lines = []
        while time.monotonic() < deadline:
            line = proc.stderr.readline()
            lines.append(line)
            if "Server started with 2 experts" in line:
                ready = True
                break
        assert ready, f"server never became ready: {lines[-5:]}"
    finally:
        proc.terminate()
        proc.wait(10)


def test_config_yaml_defaults_and_precedence(tmp_path):
    """--config config.yml provides defaults; explicit CLI flags win
    (reference run_server.py:21-22 configargparse semantics)."""
    import argparse

    from hivemind_amd.utils.config import parse_args_with_config

    cfg = tmp_path / "config.yml"
    cfg.write_text("hidden_dim: 2048\nexpert_cls: transformer\n")

    def build():
        p = argparse.ArgumentParser()
        p.add_argument("--hidden_dim", type=int, default=1024)
        p.add_argument("--expert_cls", type=str, default="ffn")
        return p

    args = parse_args_with_config(build(), ["--config", str(cfg)])
    assert args.hidden_dim == 2048 and args.expert_cls == "transformer"
    args = parse_args_with_config(build(), ["--config", str(cfg), "--hidden_dim", "4096"])
    assert args.hidden_dim == 4096 and args.expert_cls == "transformer"
    args = parse_args_with_config(build(), [])
    assert args.hidden_dim == 1024
    # unknown keys are rejected loudly
    bad = tmp_path / "bad.yml"
    bad.write_text("no_such_flag: 1\n")
    import pytest as _pytest

    with _pytest.raises(ValueError):
        parse_args_with_config(build(), ["--config", str(bad)])


def test_albert_trainer_lamb_recipe_smoke(tmp_path):
    """The collaborative-ALBERT example runs the reference recipe: LAMB with
    gradient clipping through hivemind_amd.Optimizer (reference
    examples/albert/run_trainer.py:266). 1-peer swarm, tiny shapes, 2 epochs."""
    import subprocess
    import sys

    proc = subprocess.run(
        [
            sys.executable, "examples/albert/run_trainer.py",
            "--optimizer", "lamb", "--clip_grad_norm", "1.0",
            "--batch_size", "4", "--seq_len", "32", "--target_batch_size", "8",
            "--max_epochs", "2", "--state_path", str(tmp_path / "state.pt"),
            "--backup_every_epochs", "0",
        ],
        capture_output=True, text=True, timeout=420,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    assert proc.returncode == 0, f"trainer failed:\n{proc.stderr[-2000:]}"
    assert "epoch 2" in proc.stderr or "epoch 2" in proc.stdout, proc.stderr[-1500:]
