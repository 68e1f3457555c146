"""GPU trainer tests: the hipGraph-captured training step through the real
Trainer + dataloader path, and atomics determinism characterization."""

import math
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu


def _train_config(datalist, out_dir, iterations=2, hip_graphs=True):
    ds = {
        "scale": 2, "ori_scale": "down4", "time_bins": 1,
        "need_gt_frame": False, "need_gt_events": True,
        "mode": "events", "window": 1024, "sliding_window": 512,
        "data_augment": {"enabled": False, "augment": [], "augment_prob": []},
        "hot_filter": {"enabled": False},
        "sequence": {"sequence_length": 3, "seqn": 3, "step_size": None,
                     "pause": {"enabled": False,
                               "proba_pause_when_running": 0.05,
                               "proba_pause_when_paused": 0.9}},
    }
    dl = {"use_ddp": False, "path_to_datalist_txt": str(datalist),
          "batch_size": 2, "shuffle": False, "num_workers": 0,
          "pin_memory": True, "drop_last": True, "dataset": ds}
    return {
        "experiment": "gpu-test", "SEQN": 3, "precision": "bf16",
        "model": {"name": "ESRNet",
                  "args": {"inch": 2, "basech": 8, "num_frame": 3,
                           "upsampler": "pixelshuffle"}},
        "optimizer": {"name": "Adam", "args": {"lr": 1e-3}},
        "lr_scheduler": {"name": "ExponentialLR", "args": {"gamma": 0.95}},
        "trainer": {
            "output_path": str(out_dir),
            "hip_graphs": hip_graphs,
            "epoch_based_train": {"enabled": False},
            "iteration_based_train": {
                "enabled": True, "iterations": iterations, "save_period": 1,
                "train_log_step": 1, "valid_log_step": 1, "valid_step": 100,
                "lr_change_rate": 1000},
            "monitor": "off", "tensorboard": False,
            "vis": {"enabled": False},
        },
        "train_dataloader": dl,
        "valid_dataloader": None,
    }


@pytest.fixture(scope="module")
def gpu_synth_datalist(tmp_path_factory):
    from esr_amd.data import make_synthetic_dataset
    root = tmp_path_factory.mktemp("gpusynth")
    return make_synthetic_dataset(root, num_sequences=1,
                                  resolution=(64, 64), num_events=60_000,
                                  seed=11)


def _run_training(tmp_path, datalist, hip_graphs, run_id):
    from esr_amd.config import ConfigParser
    from esr_amd.engine import build_training
    from esr_amd.utils.logging import setup_logging
    cfg = _train_config(datalist, tmp_path / f"out_{run_id}",
                        hip_graphs=hip_graphs)
    parser = ConfigParser(cfg, run_id=run_id)
    logger = setup_logging(f"gpu-test-{run_id}", None)
    trainer = build_training(parser, torch.device("cuda:0"), logger)
    trainer.train()
    return parser, trainer


def test_trainer_hipgraph_path(tmp_path, gpu_synth_datalist):
    parser, trainer = _run_training(tmp_path, gpu_synth_datalist,
                                    hip_graphs=True, run_id="g1")
    assert trainer.use_graphs, "graph path fell back to eager"
    assert trainer._graph_step is not None
    avg = trainer.train_metrics.avg("train_loss")
    assert math.isfinite(avg) and avg > 0
    ckpts = list(Path(parser.save_dir).glob("checkpoint-iteration*.pth"))
    assert ckpts


def test_trainer_eager_vs_graph_losses_close(tmp_path, gpu_synth_datalist):
    """Same data, same seed: the first-step loss of the graphed trainer must
    match the eager trainer to bf16 tolerance."""
    torch.manual_seed(7)
    _, t_eager = _run_training(tmp_path, gpu_synth_datalist,
                               hip_graphs=False, run_id="e1")
    torch.manual_seed(7)
    _, t_graph = _run_training(tmp_path, gpu_synth_datalist,
                               hip_graphs=True, run_id="g2")  # reuses find db
    a = t_eager.train_metrics.avg("train_mse_loss")
    b = t_graph.train_metrics.avg("train_mse_loss")
    assert abs(a - b) / max(abs(a), 1e-6) < 0.1, (a, b)


def test_dcn_backward_replay_determinism():
    """col2im uses fp32 atomicAdd: characterize run-to-run variation.
    Identical inputs on identical kernels must agree to fp32 atomic
    reorder tolerance (determinism tests for atomics)."""
    from esr_amd.ops.native import get_ext
    ext = get_ext()
    assert ext is not None
    g = torch.Generator().manual_seed(0)
    B, C, H, W, dg = 2, 16, 32, 32, 4
    input = torch.randn(B, C, H, W, generator=g).cuda()
    offset = (torch.randn(B, dg * 18, H, W, generator=g) * 2).cuda()
    mask = torch.rand(B, dg * 9, H, W, generator=g).cuda()
    weight = (torch.randn(8, C, 3, 3, generator=g) * 0.2).cuda()
    gout = torch.randn(B, 8, H, W, generator=g).cuda()
    g1 = ext.deform_conv2d_backward(input, offset, mask, weight, gout,
                                    1, 1, 1, 1, 1, 1, dg)[0]
    g2 = ext.deform_conv2d_backward(input, offset, mask, weight, gout,
                                    1, 1, 1, 1, 1, 1, dg)[0]
    # fp32 atomics reorder rounding only: tight but not bitwise
    assert torch.allclose(g1, g2, atol=1e-5, rtol=1e-5)


def test_graph_capture_preserves_weights():
    """capture()'s warmup steps must not leave garbage-gradient Adam updates
    in the model: params bit-identical before/after capture, optimizer state
    fresh (advisor finding r1)."""
    from esr_amd.engine.graph_runner import GraphedBPTTStep, flatten_grads
    from esr_amd.models import build_model

    device = torch.device("cuda:0")
    torch.manual_seed(0)
    model = build_model("ESRNet", inch=2, basech=8, num_frame=3,
                        upsampler="pixelshuffle").to(device)
    params = [p for p in model.parameters() if p.requires_grad]
    before = [p.detach().clone() for p in params]
    flat = flatten_grads(params, device)
    opt = torch.optim.Adam(params, lr=1e-3, capturable=True)
    runner = GraphedBPTTStep(
        model, opt, flat, n_windows=2, inp_shape=(2, 4, 2, 32, 32),
        gt_shape=(2, 2, 2, 32, 32), device=device,
        amp_dtype=torch.bfloat16, sequence=True, seqn=3)
    runner.capture()
    for p, b in zip(params, before):
        assert torch.equal(p.detach(), b), "capture() mutated model weights"
    for state in opt.state.values():
        for v in state.values():
            if torch.is_tensor(v):
                assert not v.any(), "capture() left non-fresh optimizer state"
    # and the captured graph still trains: one replay changes the weights
    x = torch.randn(2, 4, 2, 32, 32, device=device)
    g = torch.randn(2, 2, 2, 32, 32, device=device)
    runner.run([x], [g])
    torch.cuda.synchronize()
    assert any(not torch.equal(p.detach(), b) for p, b in zip(params, before))
