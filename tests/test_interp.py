"""Frame-interpolation (Super-SloMo-class) tests."""

import torch

from esr_amd.models.interp import FrameInterpolator, backwarp, upsample_frames


def test_backwarp_identity_and_shift():
    img = torch.rand(1, 1, 16, 16)
    assert torch.allclose(backwarp(img, torch.zeros(1, 2, 16, 16)), img,
                          atol=1e-5)
    flow = torch.zeros(1, 2, 16, 16)
    flow[:, 0] = 1.0  # sample from x+1
    out = backwarp(img, flow)
    assert torch.allclose(out[..., :, :-1], img[..., :, 1:], atol=1e-4)


def test_interpolator_shapes_and_recursion():
    m = FrameInterpolator(base=8)
    out = m(torch.rand(2, 1, 32, 32), torch.rand(2, 1, 32, 32), 0.5)
    assert out.shape == (2, 1, 32, 32)
    up = upsample_frames(m, torch.rand(3, 1, 32, 32), 4)
    assert up.shape[0] == 9  # (3-1)*4 + 1


def test_interpolator_trainable():
    m = FrameInterpolator(base=8)
    I0, I1 = torch.rand(1, 1, 32, 32), torch.rand(1, 1, 32, 32)
    loss = (m(I0, I1, 0.5) - 0.5 * (I0 + I1)).abs().mean()
    loss.backward()
    assert any(p.grad is not None and p.grad.abs().sum() > 0
               for p in m.parameters())


def test_self_training_reduces_loss(tmp_path):
    """tools/train_interp.py: the interpolator LEARNS on the synthetic
    motion triplets (the reference relies on a downloaded Super-SloMo
    checkpoint; this makes the generation pipeline self-contained)."""
    import sys
    from pathlib import Path
    sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tools"))
    import train_interp
    old_argv = sys.argv
    sys.argv = ["train_interp.py", "--steps", "120", "--batch", "4",
                "--res", "64", "--out", str(tmp_path / "interp.pth"),
                "--device", "cpu", "--log-every", "1000"]
    try:
        first, last = train_interp.main()
    finally:
        sys.argv = old_argv
    assert last < first * 0.95, f"no learning: {first} -> {last}"
    ckpt = torch.load(tmp_path / "interp.pth", map_location="cpu")
    assert ckpt["model"]["name"] == "FrameInterpolator"
    from esr_amd.models.interp import FrameInterpolator
    m = FrameInterpolator()
    m.load_state_dict(ckpt["model"]["states"])
