"""End-to-end main_zero.py CLI plumbing test (BASELINE config #1):
tiny GPT-2 ZeRO-1 on CPU/gloo at world_size=2, launched exactly as torchrun
would (RANK/WORLD_SIZE/MASTER_* env), including checkpoint + --resume."""

import os
import socket
import subprocess
import sys

import pytest
import torch
import yaml

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _write_cfgs(tmpdir, total_steps):
    model_cfg = {
        "tiny": {
            "embedding_dim": 64,
            "vocab_size": 256,
            "num_head": 4,
            "block_size": 32,
            "dropout": 0.0,
            "N": 2,
            "alibi_attn": True,
        }
    }
    cfg = {
        "training": {
            "max_epochs": 4,
            "batch_size": 4,
            "peak_learning_rate": 1e-3,
            "warmup_steps": 2,
            "total_steps": total_steps,
            "decay_steps": 10,
            "end_learning_rate": 1e-4,
            "weight_decay": 0.1,
            "gradient_accumulation_steps": 1,
            "evaluation_frequency": 3,
            "maximum_evaluation_steps": 2,
            "train_context": 32,
            "seed": 7,
        },
        "model": {"size": "tiny", "warm_init": False, "warm_init_dir": ""},
        "data": {
            "corpus": "synthetic",
            "max_context": 32,
            "train_samples": 64,
            "checkpoint_directory": os.path.join(tmpdir, "ckpt"),
            "index_path_train": "",
            "index_path_validation": "",
            "wandb_project": None,
            "steps_per_epoch": 16,
        },
        "distributed": {"bucket_mb": 1, "overlap_comm": False},
    }
    mpath = os.path.join(tmpdir, "model_config.yaml")
    cpath = os.path.join(tmpdir, "config.yaml")
    with open(mpath, "w") as f:
        yaml.safe_dump(model_cfg, f)
    with open(cpath, "w") as f:
        yaml.safe_dump(cfg, f)
    return cpath, mpath


def _launch(world, cpath, mpath, port, resume=False):
    procs = []
    for rank in range(world):
        env = dict(
            os.environ,
            RANK=str(rank),
            LOCAL_RANK=str(rank),
            WORLD_SIZE=str(world),
            MASTER_ADDR="127.0.0.1",
            MASTER_PORT=str(port),
        )
        cmd = [sys.executable, os.path.join(REPO, "main_zero.py"),
               "--cfg", cpath, "--model-cfg", mpath]
        if resume:
            cmd.append("--resume")
        procs.append(subprocess.Popen(cmd, env=env, cwd=REPO,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=240)
        outs.append(out.decode())
    return procs, outs


@pytest.mark.timeout(600)
def test_main_zero_cli_world2_train_checkpoint_resume(tmp_path):
    if torch.cuda.is_available():
        pytest.skip("CPU/gloo plumbing test")
    tmpdir = str(tmp_path)
    cpath, mpath = _write_cfgs(tmpdir, total_steps=5)
    procs, outs = _launch(2, cpath, mpath, _free_port())
    for p, out in zip(procs, outs):
        assert p.returncode == 0, out[-2000:]
    ckdir = os.path.join(tmpdir, "ckpt", "tiny")
    files = os.listdir(ckdir)
    assert any(f.startswith("params_") for f in files), files
    assert any(f.startswith("optimizer_") for f in files), files

    # resume continues past the checkpointed step to the new total
    cpath2, mpath2 = _write_cfgs(tmpdir, total_steps=7)
    procs, outs = _launch(2, cpath2, mpath2, _free_port(), resume=True)
    for p, out in zip(procs, outs):
        assert p.returncode == 0, out[-2000:]
    assert any("resumed from step" in o for o in outs)
    assert any("training done at step 7" in o for o in outs)
