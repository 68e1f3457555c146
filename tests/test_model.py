"""ESRNet model-family tests."""

import torch

from esr_amd.models import build_model, list_models
from esr_amd.ops.convgru import ConvGRUCell


def test_registry():
    assert "ESRNet" in list_models()
    assert "DeepRecurrNet" in list_models()   # reference-config alias


def test_forward_shapes_and_padcrop():
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
    for H, W in [(32, 32), (36, 44), (40, 40)]:
        y = m(torch.randn(1, 3, 2, H, W))
        assert y.shape == (1, 2, H, W)
        m.reset_states()


def test_bptt_state_persistence_and_reset():
    torch.manual_seed(0)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
    x = torch.rand(2, 3, 2, 32, 32)  # nonneg counts keep the tail ReLU live
    m.reset_states()
    y1 = m(x)
    s1 = m.time_propagate.state
    assert s1 is not None
    y2 = m(x)           # state carried -> state (and output) evolve
    s2 = m.time_propagate.state
    assert not torch.allclose(s1, s2)
    assert not torch.allclose(y1, y2)
    m.reset_states()
    y3 = m(x)
    assert torch.allclose(y1, y3, atol=1e-6)


def test_bptt_backward_through_sequence():
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
    m.reset_states()
    loss = 0
    for _ in range(3):
        loss = loss + (m(torch.randn(1, 3, 2, 32, 32)) ** 2).mean()
    loss.backward()
    grads = [p.grad for p in m.parameters() if p.grad is not None]
    assert len(grads) > 0
    assert all(torch.isfinite(g).all() for g in grads)


def test_convgru_cell_math():
    torch.manual_seed(0)
    cell = ConvGRUCell(4, 4, 3)
    x = torch.randn(2, 4, 8, 8)
    h = cell(x, None)
    assert h.shape == (2, 4, 8, 8)
    h2 = cell(x, h)
    assert not torch.allclose(h, h2)
    # manual recompute of the gate math
    import torch.nn.functional as F
    xh = torch.cat([x, h], 1)
    ur = cell.ur_gate(xh)
    u, r = torch.sigmoid(ur[:, :4]), torch.sigmoid(ur[:, 4:])
    o = torch.tanh(cell.out_gate(torch.cat([x, h * r], 1)))
    ref = h * (1 - u) + o * u
    assert torch.allclose(h2, ref, atol=1e-6)


def test_pixelshuffle_upsampler_variant():
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3,
                    upsampler="pixelshuffle")
    y = m(torch.randn(1, 3, 2, 32, 32))
    assert y.shape == (1, 2, 32, 32)


def test_ablation_flags():
    for kwargs in [dict(has_ltc=False), dict(has_gtc=False),
                   dict(has_dcnatten=False), dict(has_scaleaggre=False)]:
        m = build_model("ESRNet", inch=2, basech=8, num_frame=3, **kwargs)
        y = m(torch.randn(1, 3, 2, 16, 16))
        assert y.shape == (1, 2, 16, 16)


def test_num_frame_5():
    m = build_model("ESRNet", inch=2, basech=8, num_frame=5)
    y = m(torch.randn(1, 5, 2, 16, 16))
    assert y.shape == (1, 2, 16, 16)


def test_forward_sequence_matches_per_window():
    """forward_sequence (shared encoder) must equal per-window forward()
    exactly, including gradient totals."""
    torch.manual_seed(3)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
    frames = torch.rand(2, 5, 2, 32, 32)

    m.reset_states()
    outs_seq = m.forward_sequence(frames, seqn=3)
    loss_seq = sum((o ** 2).mean() for o in outs_seq)
    loss_seq.backward()
    grads_seq = [p.grad.clone() for p in m.parameters() if p.grad is not None]
    for p in m.parameters():
        p.grad = None

    m.reset_states()
    outs_ref = [m(frames[:, w:w + 3]) for w in range(3)]
    loss_ref = sum((o ** 2).mean() for o in outs_ref)
    loss_ref.backward()
    grads_ref = [p.grad.clone() for p in m.parameters() if p.grad is not None]

    for a, b in zip(outs_seq, outs_ref):
        assert torch.allclose(a, b, atol=1e-6), (a - b).abs().max().item()
    assert abs(loss_seq.item() - loss_ref.item()) < 1e-6
    for ga, gb in zip(grads_seq, grads_ref):
        assert torch.allclose(ga, gb, atol=1e-5), (ga - gb).abs().max().item()


def test_gtc_frozen_stateless():
    torch.manual_seed(4)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3, gtc_frozen=True)
    x = torch.rand(1, 3, 2, 16, 16)
    m.reset_states()
    y1 = m(x)
    assert m.time_propagate.state is None     # frozen: never persists
    y2 = m(x)
    assert torch.allclose(y1, y2, atol=1e-6)  # stateless => identical


def test_convlstm_recurrent_type():
    torch.manual_seed(5)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3,
                    recurrent_block_type="convlstm")
    y = m(torch.rand(1, 3, 2, 16, 16))
    assert y.shape == (1, 2, 16, 16)
    state = m.time_propagate.state
    assert isinstance(state, tuple) and len(state) == 2  # (h, c)
    m(torch.rand(1, 3, 2, 16, 16))
    m.detach_states()
    assert not m.time_propagate.state[0].requires_grad


def test_detach_states_truncates_bptt():
    torch.manual_seed(6)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
    m.reset_states()
    m(torch.rand(1, 3, 2, 16, 16))
    m.detach_states()
    loss = (m(torch.rand(1, 3, 2, 16, 16)) ** 2).mean()
    loss.backward()   # must not error about freed graphs from window 1
    assert all(torch.isfinite(p.grad).all()
               for p in m.parameters() if p.grad is not None)
