"""Streaming (serving) inference tests."""

import torch

from esr_amd.engine.streaming import StreamingESR
from esr_amd.models import build_model


def _window(n=256, H=32, W=32, seed=0):
    g = torch.Generator().manual_seed(seed)
    xs = (torch.rand(n, generator=g) * W).floor()
    ys = (torch.rand(n, generator=g) * H).floor()
    ts = torch.sort(torch.rand(n, generator=g)).values
    ps = torch.randint(0, 2, (n,), generator=g).float() * 2 - 1
    return torch.stack([xs, ys, ts, ps])


def test_streaming_cpu_warmup_and_outputs():
    torch.manual_seed(0)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
    s = StreamingESR(m, (32, 32), scale=2, seqn=3, device="cpu",
                     use_graphs=False, amp_dtype=None)
    assert s.push(_window(seed=1)) is None
    assert s.push(_window(seed=2)) is None
    out = s.push(_window(seed=3))
    assert out is not None and out.shape == (2, 64, 64)
    out2 = s.push(_window(seed=4))
    assert not torch.allclose(out, out2)
    s.reset()
    assert s.push(_window(seed=1)) is None


def test_streaming_state_continuity_cpu():
    """Streaming over windows w1..w4 must equal manual sliding-window calls
    with persistent model state."""
    torch.manual_seed(1)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3)
    windows = [_window(seed=i) for i in range(4)]

    s = StreamingESR(m, (32, 32), scale=2, seqn=3, device="cpu",
                     use_graphs=False, amp_dtype=None)
    outs = [s.push(w) for w in windows]

    m.reset_states()
    from esr_amd.ops import events_to_channels
    frames = []
    for w in windows:
        frames.append(events_to_channels(w[0].floor() * 2, w[1].floor() * 2,
                                         w[3], (64, 64)))
    ref3 = m(torch.stack(frames[0:3])[None])[0]
    ref4 = m(torch.stack(frames[1:4])[None])[0]
    assert torch.allclose(outs[2], ref3, atol=1e-5)
    assert torch.allclose(outs[3], ref4, atol=1e-5)


import pytest  # noqa: E402


@pytest.mark.gpu
def test_streaming_graphed_matches_eager_gpu():
    torch.manual_seed(2)
    m = build_model("ESRNet", inch=2, basech=8, num_frame=3).cuda()
    windows = [_window(seed=i) for i in range(6)]

    s_eager = StreamingESR(m, (32, 32), scale=2, seqn=3, device="cuda:0",
                           use_graphs=False, amp_dtype=None)
    eager = [s_eager.push(w) for w in windows]

    m.reset_states()
    s_graph = StreamingESR(m, (32, 32), scale=2, seqn=3, device="cuda:0",
                           use_graphs=True, amp_dtype=None)
    graphed = [s_graph.push(w) for w in windows]
    assert s_graph._graph is not None, "graph was not captured"

    for i in range(2, 6):
        assert torch.allclose(eager[i], graphed[i], atol=1e-4), \
            (i, (eager[i] - graphed[i]).abs().max().item())


def test_push_events_roundtrip():
    """Serving-side count->event output: the emitted HR event stream
    splats back to the predicted count map."""
    from esr_amd.engine.streaming import StreamingESR
    from esr_amd.models import build_model
    from esr_amd.ops.events import events_to_channels
    torch.manual_seed(0)
    model = build_model("ESRNet", inch=2, basech=4, num_frame=3)
    srv = StreamingESR(model, lr_resolution=(16, 16), scale=2,
                       device="cpu", use_graphs=False)
    out_ev = None
    for i in range(4):
        g = torch.Generator().manual_seed(i)
        n = 200
        ev = torch.stack([
            torch.randint(0, 16, (n,), generator=g).float(),
            torch.randint(0, 16, (n,), generator=g).float(),
            torch.sort(torch.rand(n, generator=g)).values,
            torch.randint(0, 2, (n,), generator=g).float() * 2 - 1])
        out_ev = srv.push_events(ev)
    assert out_ev is not None and out_ev.shape[1] == 4
    t = out_ev[:, 2]
    assert (t[1:] >= t[:-1]).all()
    # splat back == the (rounded, clamped) predicted count map
    srv2 = StreamingESR(model, lr_resolution=(16, 16), scale=2,
                        device="cpu", use_graphs=False)
    model.reset_states()
    cnt = None
    for i in range(4):
        g = torch.Generator().manual_seed(i)
        n = 200
        ev = torch.stack([
            torch.randint(0, 16, (n,), generator=g).float(),
            torch.randint(0, 16, (n,), generator=g).float(),
            torch.sort(torch.rand(n, generator=g)).values,
            torch.randint(0, 2, (n,), generator=g).float() * 2 - 1])
        cnt = srv2.push(ev)
    want = cnt.float().round().clamp(min=0)
    back = events_to_channels(out_ev[:, 0], out_ev[:, 1], out_ev[:, 3],
                              want.shape[-2:])
    assert torch.equal(back, want)
