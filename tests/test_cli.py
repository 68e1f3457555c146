"""End-to-end CLI tests: train.py and infer.py as subprocesses on the
shipped CPU-plumbing config (BASELINE config 1 class: small synthetic
data, CPU only)."""

import json
import subprocess
import sys
from pathlib import Path

import pytest
import yaml

REPO = Path(__file__).resolve().parent.parent


@pytest.fixture(scope="module")
def cli_workspace(tmp_path_factory):
    ws = tmp_path_factory.mktemp("cli")
    from esr_amd.data import make_synthetic_dataset
    datalist = make_synthetic_dataset(ws / "data", num_sequences=2,
                                      resolution=(64, 64), num_events=50_000,
                                      seed=21)
    # derive a tiny run config from the shipped CPU-plumbing config
    cfg = yaml.safe_load((REPO / "configs" / "train_cpu_plumbing.yml")
                         .read_text())
    cfg["trainer"]["output_path"] = str(ws / "runs")
    cfg["trainer"]["iteration_based_train"]["iterations"] = 2
    cfg["trainer"]["iteration_based_train"]["save_period"] = 1
    cfg["trainer"]["iteration_based_train"]["valid_step"] = 100
    for dl in ("train_dataloader", "valid_dataloader"):
        cfg[dl]["path_to_datalist_txt"] = datalist
        cfg[dl]["num_workers"] = 0
    cfg_path = ws / "cfg.yml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    return ws, cfg_path, datalist


@pytest.mark.timeout(600)
def test_train_cli(cli_workspace):
    ws, cfg_path, _ = cli_workspace
    out = subprocess.run(
        [sys.executable, "train.py", "-c", str(cfg_path), "-id", "cli0"],
        capture_output=True, text=True, timeout=500, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-3000:]
    ckpts = list((ws / "runs" / "cli0" / "models")
                 .glob("checkpoint-iteration*.pth"))
    assert ckpts, "train.py produced no checkpoint"
    assert (ws / "runs" / "cli0" / "config.yml").exists()


@pytest.mark.timeout(600)
def test_infer_cli(cli_workspace):
    ws, cfg_path, datalist = cli_workspace
    ckpts = sorted((ws / "runs" / "cli0" / "models")
                   .glob("checkpoint-iteration*.pth"))
    if not ckpts:  # ordering safety: train first
        test_train_cli(cli_workspace)
        ckpts = sorted((ws / "runs" / "cli0" / "models")
                       .glob("checkpoint-iteration*.pth"))
    from esr_amd.data import read_datalist
    seq = read_datalist(str(datalist))[0]
    out = subprocess.run(
        [sys.executable, "infer.py", "--model_path", str(ckpts[-1]),
         "--data_path", seq, "--output_path", str(ws / "eval"),
         "--device", "cpu", "--scale", "2", "--ori_scale", "down4",
         "--window", "256", "--sliding_window", "128",
         "--seql", "4", "--seqn", "3", "--need_gt_events", "--no_images",
         "--num_workers", "0"],
        capture_output=True, text=True, timeout=500, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-3000:]
    mean = yaml.safe_load((ws / "eval" / "mean_results.yml").read_text())
    assert "esr_rmse" in mean["mean"]
    assert mean["mean"]["bicubic_rmse"] > 0


@pytest.mark.timeout(300)
def test_make_synth_data_cli(tmp_path):
    out = subprocess.run(
        [sys.executable, "tools/make_synth_data.py", "--out",
         str(tmp_path / "d"), "--sequences", "2", "--resolution", "64",
         "--events", "20000"],
        capture_output=True, text=True, timeout=250, cwd=str(REPO))
    assert out.returncode == 0, out.stderr[-2000:]
    assert (tmp_path / "d" / "train_datalist.txt").exists()
    assert (tmp_path / "d" / "valid_datalist.txt").exists()
