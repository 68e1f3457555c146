"""Property/parity tests for benchmark-defining details: windowing
formulas, scale-pair edge cases, random-mode redistribution, config
overrides, scalar logging."""

import json

import numpy as np
import pytest
import torch

from esr_amd.config import ConfigParser, set_by_path, get_by_path
from esr_amd.data import write_synthetic_store, EventSRDataset
from esr_amd.data.dataset import resolve_scale_pair
from esr_amd.ops import redistribute_stack
from esr_amd.utils.logging import MetricWriter


@pytest.fixture(scope="module")
def store64(tmp_path_factory):
    p = tmp_path_factory.mktemp("evs2") / "s.evs"
    write_synthetic_store(p, resolution=(64, 64), num_events=50_000, seed=4)
    return p


def _cfg(**over):
    cfg = {"scale": 2, "ori_scale": "down4", "time_bins": 1,
           "need_gt_frame": False, "need_gt_events": True,
           "mode": "events", "window": 500, "sliding_window": 100,
           "data_augment": {"enabled": False, "augment": [],
                            "augment_prob": []},
           "hot_filter": {"enabled": False},
           "sequence": {"sequence_length": 2, "seqn": 2, "step_size": None,
                        "pause": {"enabled": False,
                                  "proba_pause_when_running": 0,
                                  "proba_pause_when_paused": 0}}}
    cfg.update(over)
    return cfg


def test_window_index_formula(store64):
    """Window i covers [(window-sliding)*i, +window), clamped — the
    reference's exact formula (ESR:dataloader/h5dataset.py:196-215)."""
    ds = EventSRDataset(store64, _cfg())
    n = ds.store.num_events("down4")
    assert len(ds) == int(n / (500 - 100))
    for i in [0, 1, len(ds) - 1]:
        idx0, idx1 = ds.event_indices[i]
        assert idx0 == 400 * i
        assert idx1 == min(idx0 + 500, n - 1)


def test_dataset_length_cap(store64):
    ds = EventSRDataset(store64, _cfg(dataset_length=3))
    assert len(ds) == 3


def test_scale1_identity_pair():
    inp_p, inp_r, gt_p, gt_r, down = resolve_scale_pair(
        (128, 128), "down2", 1, need_gt_events=True)
    assert inp_p == gt_p == "down2"
    assert inp_r == gt_r == [64, 64]


def test_real_world_pair():
    inp_p, inp_r, gt_p, gt_r, down = resolve_scale_pair(
        (512, 512), "down8", 2, need_gt_events=False, real_world_test=True)
    assert inp_p == "down8_real"
    assert inp_r == [64, 64]
    assert gt_r == [128, 128]
    with pytest.raises(ValueError):
        resolve_scale_pair((512, 512), "down4", 2, need_gt_events=False,
                           real_world_test=True)


def test_redistribute_random_mode_bounds():
    g = torch.Generator().manual_seed(2)
    stack = torch.randint(0, 5, (1, 8, 4, 4), generator=g).float()
    gen = torch.Generator().manual_seed(9)
    cloud = redistribute_stack(stack, mode="random", generator=gen)
    ev = cloud[0]
    ev = ev[ev.abs().sum(1) > 0]
    # every timestamp stays inside its bin's open-left interval
    c = ((ev[:, 2] - 1e-6) * 8).long().clamp(0, 7)
    lo = c.float() / 8 + 1 / (100 * 8)
    hi = (c.float() + 1) / 8
    assert (ev[:, 2] >= lo - 1e-5).all() and (ev[:, 2] <= hi + 1e-5).all()
    # reproducible under the same generator seed
    cloud2 = redistribute_stack(stack, mode="random",
                                generator=torch.Generator().manual_seed(9))
    assert torch.allclose(cloud, cloud2)


def test_config_path_overrides(tmp_path):
    cfg = {"a": {"b": {"c": 1}}, "trainer": {}}
    set_by_path(cfg, "a;b;c", 5)
    assert get_by_path(cfg, "a;b;c") == 5
    set_by_path(cfg, "x;y", "new")
    assert cfg["x"]["y"] == "new"
    parser = ConfigParser(cfg, run_id="cfg0", output_path=str(tmp_path),
                          make_dirs=True)
    assert (tmp_path / "cfg0" / "config.yml").exists()
    assert parser["a"]["b"]["c"] == 5


def test_metric_writer_jsonl(tmp_path):
    w = MetricWriter(tmp_path, enabled=True)
    w.set_step(0)
    w.add_scalar("loss", 1.5)
    w.set_step(1)
    w.add_scalar("loss", 1.0)
    w.close()
    recs = [json.loads(ln) for ln in
            (tmp_path / "scalars.jsonl").read_text().splitlines()]
    losses = [r for r in recs if r["tag"] == "loss"]
    assert len(losses) == 2 and losses[1]["step"] == 1
    assert any(r["tag"] == "steps_per_sec" for r in recs)


def test_count_conservation_under_augmentation(store64):
    """Flips/polarity permute events; total count must be conserved."""
    cfg = _cfg()
    cfg["data_augment"] = {"enabled": True,
                           "augment": ["Horizontal", "Vertical", "Polarity"],
                           "augment_prob": [1.0, 1.0, 1.0]}
    a = EventSRDataset(store64, cfg).__getitem__(0, seed=5)
    b = EventSRDataset(store64, _cfg()).__getitem__(0, seed=5)
    assert a["inp_cnt"].sum() == b["inp_cnt"].sum()
    assert a["gt_cnt"].sum() == b["gt_cnt"].sum()


def test_store_events_dtype_roundtrip(tmp_path):
    from esr_amd.data.store import EventStoreWriter, EventStore
    xs = np.array([0, 65535], dtype=np.uint16)
    ys = np.array([1, 2])
    ts = np.array([0.5, 1.5])
    ps = np.array([1, -1])
    with EventStoreWriter(tmp_path / "t.evs", (4, 4)) as w:
        w.add_group("ori", xs, ys, ts, ps)
    s = EventStore(tmp_path / "t.evs")
    ev = s.events("ori", 0, 2)
    assert ev[0, 1] == 65535 and ev[3, 1] == -1 and ev[2, 1] == 1.5


def test_ssim_psnr_sanity():
    from esr_amd.loss import ssim, psnr, rmse, mse
    g = torch.Generator().manual_seed(1)
    a = torch.rand(2, 16, 16, generator=g)
    assert abs(ssim(a, a) - 1.0) < 1e-6           # identical -> 1
    noisy = a + 0.2 * torch.randn(2, 16, 16, generator=g)
    noisier = a + 0.6 * torch.randn(2, 16, 16, generator=g)
    assert ssim(a, noisy) > ssim(a, noisier)      # monotone in noise
    assert psnr(noisy, a) > psnr(noisier, a)
    assert abs(rmse(noisy, a) ** 2 - mse(noisy, a)) < 1e-5
    assert psnr(a, a) == float("inf")


def test_lpips_deterministic_and_discriminative():
    from esr_amd.loss import PerceptualLoss
    g = torch.Generator().manual_seed(2)
    a = torch.rand(1, 2, 32, 32, generator=g)
    b = torch.rand(1, 2, 32, 32, generator=g)
    p1 = PerceptualLoss(net="alex")
    p2 = PerceptualLoss(net="alex")          # same seed -> same projection
    assert abs(p1(a, b).item() - p2(a, b).item()) < 1e-6
    assert p1(a, a).item() < 1e-6
    assert p1(a, b).item() > p1(a, a).item()


def test_lpips_reference_heads_and_rng_isolation():
    """The bundled v0.1 linear heads load by default and constructing the
    metric neither reseeds nor advances the global torch RNG."""
    from esr_amd.loss.lpips import LPIPS
    torch.manual_seed(77)
    before = torch.rand(4)
    torch.manual_seed(77)
    m = LPIPS("alex")
    assert torch.equal(torch.rand(4), before)
    assert m.heads_pretrained and not m.backbone_pretrained
    # head channel widths match the alex slice dims (the reference v0.1 file)
    assert [lin.weight.shape[1] for lin in m.lins] == [64, 192, 384, 256, 256]
    assert all((lin.weight >= 0).all() for lin in m.lins)  # trained heads are nonneg


def test_lpips_backbone_mapping_roundtrip():
    from esr_amd.loss.lpips import LPIPS, _map_torchvision_backbone
    g = torch.Generator().manual_seed(0)
    shapes = [(64, 3, 11, 11), (192, 64, 5, 5), (384, 192, 3, 3),
              (256, 384, 3, 3), (256, 256, 3, 3)]
    sd = {}
    for shp, idx in zip(shapes, [0, 3, 6, 8, 10]):  # torchvision alexnet layout
        sd[f"features.{idx}.weight"] = torch.randn(*shp, generator=g)
        sd[f"features.{idx}.bias"] = torch.randn(shp[0], generator=g)
    m = LPIPS("alex")
    m.features.load_state_dict(_map_torchvision_backbone("alex", sd))
    assert torch.equal(m.features.slice1[0].weight, sd["features.0.weight"])
    assert torch.equal(m.features.slice5[0].bias, sd["features.10.bias"])
    m2 = LPIPS("alex")
    assert not m2.backbone_pretrained
