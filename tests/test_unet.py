"""U-Net model family tests."""

import torch

from esr_amd.models import build_model


def _kwargs(**over):
    kw = dict(base_num_channels=8, num_encoders=3, num_residual_blocks=2,
              num_output_channels=1, skip_type="sum", norm=None,
              use_upsample_conv=True, num_bins=5,
              recurrent_block_type="convlstm", kernel_size=5)
    kw.update(over)
    return kw


def test_unet_recurrent():
    m = build_model("UNetRecurrent", **_kwargs())
    x = torch.ones(2, 5, 16, 16)
    y = m(x)
    assert y.shape == (2, 1, 16, 16)
    assert m.states[0] is not None
    y2 = m(x)
    assert not torch.allclose(y, y2)
    m.reset_states()


def test_sr_unet_recurrent_2x():
    m = build_model("SRUNetRecurrent",
                    **_kwargs(num_output_channels=5,
                              recurrent_block_type="convgru"))
    x = torch.ones(2, 5, 8, 8)
    y = m(x)
    assert y.shape == (2, 5, 16, 16)  # 2x SR output


def test_sr_unet_concat_skip():
    m = build_model("SRUNetRecurrent",
                    **_kwargs(skip_type="concat", num_output_channels=2))
    y = m(torch.ones(1, 5, 8, 8))
    assert y.shape == (1, 2, 16, 16)


def test_multires_unet():
    m = build_model("MultiResUNet", **_kwargs(num_output_channels=2))
    preds = m(torch.ones(1, 5, 32, 32))
    assert len(preds) == 3
    assert preds[-1].shape == (1, 2, 32, 32)
    assert preds[0].shape == (1, 2, 8, 8)


def test_unet_backward():
    m = build_model("SRUNetRecurrent", **_kwargs(num_output_channels=5))
    loss = 0
    for _ in range(2):
        loss = loss + m(torch.randn(1, 5, 8, 8)).square().mean()
    loss.backward()
    assert all(torch.isfinite(p.grad).all()
               for p in m.parameters() if p.grad is not None)
