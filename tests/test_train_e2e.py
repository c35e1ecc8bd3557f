"""End-to-end train.py on CPU: tiny synthetic run through the real CLI."""
import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_train_cli_one_epoch(tmp_path):
    extra = {
        "data.name": "synthetic", "data.img_h": 64, "data.img_w": 64,
        "mpi.num_bins_coarse": 4, "data.per_gpu_batch_size": 2,
        "data.visible_point_count": 16, "training.amp_dtype": "fp32",
        "data.synthetic_length": 4, "training.epochs": 1,
        "training.eval_interval": 1000000, "data.num_workers": 0,
        "training.checkpoint_interval": 2,
    }
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "train.py"),
         "--config_path", os.path.join(ROOT, "configs", "params_default.yaml"),
         "--workspace", str(tmp_path), "--version", "t1",
         "--extra_config", json.dumps(extra)],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]

    ws = tmp_path / "t1"
    assert (ws / "params.yaml").exists()
    assert (ws / "training.log").exists()
    assert (ws / "checkpoint_latest.pth").exists()
    # JSONL scalars written by the fallback summary writer
    assert (ws / "scalars.jsonl").exists()


def test_video_cli_writes_frames(tmp_path):
    extra = {
        "data.name": "synthetic", "data.img_h": 48, "data.img_w": 64,
        "mpi.num_bins_coarse": 4, "data.per_gpu_batch_size": 1,
        "data.visible_point_count": 8, "training.amp_dtype": "fp32",
    }
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "visualizations", "image_to_video.py"),
         "--output_dir", str(tmp_path / "vid"), "--num_frames", "3",
         "--extra_config", json.dumps(extra)],
        cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-3000:]
    frames = list((tmp_path / "vid" / "frames").glob("*.png"))
    assert len(frames) == 3


def test_bench_distributed_torchrun_cpu():
    """The driver's exact bench invocation shape (torchrun, one rank per
    GPU) — exercised with gloo on CPU, world_size 2."""
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29611", os.path.join(ROOT, "bench.py"),
         "--gpus", "2", "--steps", "1", "--warmup", "1", "--batch", "1",
         "--height", "128", "--width", "192", "--planes", "4",
         "--dtype", "fp32"],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["n_gpus"] == 2
    assert out["config"]["parallelism"] == "dp2"
    assert out["value"] > 0


def test_train_cli_two_ranks_gloo(tmp_path):
    """train.py under torchrun with 2 CPU ranks: DistributedSampler,
    rank-0 logging/checkpointing, barrier, grad engine."""
    extra = {
        "data.name": "synthetic", "data.img_h": 64, "data.img_w": 64,
        "mpi.num_bins_coarse": 4, "data.per_gpu_batch_size": 2,
        "data.visible_point_count": 8, "training.amp_dtype": "fp32",
        "data.synthetic_length": 8, "training.epochs": 1,
        "training.eval_interval": 1000000, "data.num_workers": 0,
        "training.checkpoint_interval": 2,
    }
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29613", os.path.join(ROOT, "train.py"),
         "--config_path", os.path.join(ROOT, "configs", "params_default.yaml"),
         "--workspace", str(tmp_path), "--version", "t2",
         "--extra_config", json.dumps(extra)],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=900)
    assert r.returncode == 0, (r.stdout[-1200:], r.stderr[-1500:])
    ws = tmp_path / "t2"
    assert (ws / "checkpoint_latest.pth").exists()
    assert (ws / "training.log").exists()


def test_checkpoint_dir_contract_train_to_video(tmp_path):
    """The reference's checkpoint-dir contract (params.yaml next to
    checkpoint.pth, consumed by the inference CLI; ref
    image_to_video.py:272-278): train.py writes it, image_to_video.py
    renders from it."""
    extra = {
        "data.name": "synthetic", "data.img_h": 64, "data.img_w": 64,
        "mpi.num_bins_coarse": 4, "data.per_gpu_batch_size": 2,
        "data.visible_point_count": 8, "training.amp_dtype": "fp32",
        "data.synthetic_length": 4, "training.epochs": 1,
        "training.eval_interval": 1000000, "data.num_workers": 0,
        "training.checkpoint_interval": 2,
    }
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "train.py"),
         "--config_path", os.path.join(ROOT, "configs", "params_default.yaml"),
         "--workspace", str(tmp_path), "--version", "v1",
         "--extra_config", json.dumps(extra)],
        cwd=ROOT, env=env, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    ckpt = tmp_path / "v1" / "checkpoint_latest.pth"
    assert ckpt.exists() and (tmp_path / "v1" / "params.yaml").exists()

    r2 = subprocess.run(
        [sys.executable, os.path.join(ROOT, "visualizations", "image_to_video.py"),
         "--checkpoint_path", str(ckpt),
         "--output_dir", str(tmp_path / "vid"), "--num_frames", "2"],
        cwd=ROOT, capture_output=True, text=True, timeout=600)
    assert r2.returncode == 0, r2.stderr[-2000:]
    assert len(list((tmp_path / "vid" / "frames").glob("*.png"))) == 2
