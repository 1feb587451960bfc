"""Trace-file contract + reference-compatible entry points (CPU)."""

import os
import subprocess
import sys

import torch

from util_dist import init_env, run_world

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _trace_worker(rank, world, port, outdir):
    init_env(rank, world, port)
    from eventgrad_amd.config import (DataConfig, OptimConfig, RunConfig,
                                      TriggerConfig)
    from eventgrad_amd.train.trainer import Trainer

    cfg = RunConfig(
        mode="event", model="mlp", epochs=1, device="cpu",
        trace=True, trace_dir=outdir,
        trigger=TriggerConfig(adaptive=True, horizon=1.01,
                              initial_comm_passes=2),
        data=DataConfig(dataset="synthetic-mnist", batch_size=32,
                        synthetic_train_samples=256,
                        synthetic_test_samples=32),
        optim=OptimConfig(lr=0.01), eval_at_end=False)
    Trainer(cfg).train()
    torch.distributed.destroy_process_group()


def test_trace_files_written(tmp_path):
    run_world(_trace_worker, 2, str(tmp_path))
    for rank in range(2):
        send = os.path.join(tmp_path, f"send{rank}.txt")
        recv = os.path.join(tmp_path, f"recv{rank}.txt")
        train = os.path.join(tmp_path, f"train{rank}.txt")
        assert os.path.exists(send) and os.path.exists(recv)
        assert os.path.exists(train)
        send_lines = open(send).read().strip().splitlines()
        # 256/2 shard / 32 batch = 4 passes
        assert len(send_lines) == 4
        # per tensor: "<norm>,  <thres>,  <fired01>,  " -> 3 fields x 4 tensors
        first = [f.strip() for f in send_lines[0].split(",") if f.strip()]
        assert len(first) == 3 * 4
        fired_flags = first[2::3]
        assert all(f == "1" for f in fired_flags)  # warmup pass fires all
        recv_first = [f.strip() for f in
                      open(recv).read().strip().splitlines()[0].split(",")
                      if f.strip()]
        assert len(recv_first) == 2 * 2 * 4  # (flag, norm) x 2 nbrs x 4 tensors


def test_app_entry_point_serial():
    """Reference positional-argv contract, single process (serial guard)."""
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    code = (
        "import sys; sys.argv = ['x', '0', '1', '1.01'];"
        "from eventgrad_amd.apps._compat import run;"
        "from eventgrad_amd.config import preset;"
        "import eventgrad_amd.apps._compat as C;"
        "cfg = preset('dmnist-event');"
        "cfg.epochs = 1; cfg.data.dataset = 'synthetic-mnist';"
        "cfg.data.synthetic_train_samples = 128;"
        "cfg.data.synthetic_test_samples = 32; cfg.eval_at_end = False;"
        "import eventgrad_amd.config as cc;"
        "cc.preset = lambda name: cfg;"
        "C.preset = cc.preset;"
        "raise SystemExit(run('dmnist-event'))"
    )
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Training time" in r.stdout


def test_cli_json_out(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = ROOT
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = os.path.join(tmp_path, "m.json")
    r = subprocess.run(
        [sys.executable, "-m", "eventgrad_amd.train", "--mode", "serial",
         "--model", "mlp", "--dataset", "synthetic-mnist", "--epochs", "1",
         "--batch-size", "32", "--synthetic-samples", "128",
         "--no-eval", "--json-out", out],
        env=env, capture_output=True, text=True, timeout=300,
        cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-2000:]
    import json
    m = json.load(open(out))
    assert m["total_passes"] > 0


def test_values_file_format(tmp_path):
    """values{rank}.txt: the cent binary's per-epoch '<epoch>, <loss>'
    lines (cent.cpp:80-92,124)."""
    from eventgrad_amd.train.trace import Tracer
    tr = Tracer(0, str(tmp_path))
    tr.train_line(1, 50.0, 2.25)
    tr.train_line(2, 75.0, 1.5)
    tr.close()
    lines = open(tmp_path / "values0.txt").read().splitlines()
    assert lines == ["1, 2.25", "2, 1.5"]
