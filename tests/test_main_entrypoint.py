"""End-to-end smoke of the training entrypoint (reference main.py:11-79
analog): train a tiny Pendulum config on CPU, save a checkpoint, then
restore it in a second invocation and keep training — exercising the
argparse overlay, the Worker/Chief wiring, the eval loop, and the
save/restore path exactly as a user would drive them."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_main(tmp_path, extra, cfg):
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    cmd = [sys.executable, os.path.join(REPO, "main.py"),
           "--config", str(cfg_path)] + extra
    env = dict(os.environ, CUDA_VISIBLE_DEVICES="", PYTHONPATH=REPO)
    return subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                          text=True, timeout=300)


@pytest.mark.timeout(600)
def test_main_train_save_restore_roundtrip(tmp_path):
    cfg = {"GAME": "Pendulum-v1", "NUM_ENVS": 4, "MAX_EPOCH_STEPS": 16,
           "HIDDEN_SIZES": [16], "EPOCH_MAX": 8, "SEED": 7,
           "USE_GRAPHS": False, "MINIBATCH_SIZE": 0}
    ckpt = tmp_path / "ck.pt"
    r = _run_main(tmp_path, ["--rounds", "2", "--save", str(ckpt),
                             "--eval-episodes", "1"], cfg)
    assert r.returncode == 0, r.stderr[-2000:]
    assert ckpt.exists()
    assert "eval episode reward" in r.stdout

    r2 = _run_main(tmp_path, ["--rounds", "1", "--restore", str(ckpt),
                              "--eval-episodes", "0"], cfg)
    assert r2.returncode == 0, r2.stderr[-2000:]


@pytest.mark.timeout(600)
def test_main_two_rank_torchrun(tmp_path):
    """The distributed launch exactly as documented in README: torchrun
    with 2 ranks over gloo (CPU), both ranks training in lockstep."""
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cfg = {"GAME": "Pendulum-v1", "NUM_ENVS": 4, "MAX_EPOCH_STEPS": 16,
           "HIDDEN_SIZES": [16], "EPOCH_MAX": 8, "SEED": 7,
           "USE_GRAPHS": False, "MINIBATCH_SIZE": 0}
    cfg_path = tmp_path / "cfg.json"
    cfg_path.write_text(json.dumps(cfg))
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
           "--master-port", str(port),
           os.path.join(REPO, "main.py"), "--config", str(cfg_path),
           "--rounds", "2", "--eval-episodes", "0"]
    env = dict(os.environ, CUDA_VISIBLE_DEVICES="", PYTHONPATH=REPO)
    r = subprocess.run(cmd, cwd=REPO, env=env, capture_output=True,
                       text=True, timeout=420)
    assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
