"""tools/rosbag_to_evs.py: pure-python ROS1 bag -> EVS conversion
(parity target: ESR:generate_dataset/tools/rosbag_to_h5.py, which needs
the ROS stack).  The fixture synthesizes a spec-conformant v2.0 bag with
a bz2 chunk of dvs_msgs/EventArray messages."""

import bz2
import struct
import sys
from pathlib import Path

import numpy as np
import pytest

sys.path.insert(0, str(Path(__file__).resolve().parent.parent / "tools"))

import rosbag_to_evs  # noqa: E402


def _field(name: str, value: bytes) -> bytes:
    payload = name.encode() + b"=" + value
    return struct.pack("<I", len(payload)) + payload


def _record(fields: dict, data: bytes) -> bytes:
    header = b"".join(_field(k, v) for k, v in fields.items())
    return struct.pack("<I", len(header)) + header + \
        struct.pack("<I", len(data)) + data


def _event_array_msg(xs, ys, secs, nsecs, ps, h=64, w=64) -> bytes:
    out = [struct.pack("<I", 0),            # header.seq
           struct.pack("<II", 1, 2),        # header.stamp
           struct.pack("<I", 3) + b"dvs",   # header.frame_id
           struct.pack("<III", h, w, len(xs))]
    for x, y, s, ns, p in zip(xs, ys, secs, nsecs, ps):
        out.append(struct.pack("<HHIIB", x, y, s, ns, 1 if p > 0 else 0))
    return b"".join(out)


def _make_bag(path, msgs, compression=b"bz2", topic=b"/dvs/events"):
    conn = _record(
        {"op": b"\x07", "conn": struct.pack("<I", 0), "topic": topic},
        _field("type", b"dvs_msgs/EventArray") +
        _field("md5sum", b"0" * 32) +
        _field("message_definition", b""))
    inner = conn + b"".join(
        _record({"op": b"\x02", "conn": struct.pack("<I", 0),
                 "time": struct.pack("<II", 1, 0)}, m) for m in msgs)
    payload = bz2.compress(inner) if compression == b"bz2" else inner
    chunk = _record({"op": b"\x05", "compression": compression,
                     "size": struct.pack("<I", len(inner))}, payload)
    baghdr = _record({"op": b"\x03", "index_pos": struct.pack("<Q", 0),
                      "conn_count": struct.pack("<I", 1),
                      "chunk_count": struct.pack("<I", 1)}, b" " * 64)
    Path(path).write_bytes(b"#ROSBAG V2.0\n" + baghdr + chunk)


@pytest.fixture
def bag(tmp_path):
    rng = np.random.default_rng(3)
    msgs = []
    all_ev = []
    for i in range(3):
        n = 50
        xs = rng.integers(0, 64, n)
        ys = rng.integers(0, 64, n)
        secs = np.full(n, 100 + i)
        nsecs = np.sort(rng.integers(0, 10 ** 9, n))
        ps = rng.integers(0, 2, n)
        msgs.append(_event_array_msg(xs, ys, secs, nsecs, ps))
        all_ev.append((xs, ys, secs + nsecs * 1e-9, ps))
    p = tmp_path / "ev.bag"
    _make_bag(p, msgs)
    return p, all_ev


def test_read_bag_events(bag):
    path, all_ev = bag
    H, W, xs, ys, ts, ps = rosbag_to_evs.read_bag_events(str(path))
    assert (H, W) == (64, 64)
    n_total = sum(len(e[0]) for e in all_ev)
    assert len(xs) == n_total
    assert (np.diff(ts) >= 0).all()
    want_x = np.concatenate([e[0] for e in all_ev])
    assert np.array_equal(np.sort(xs), np.sort(want_x.astype(np.float64)))
    assert set(np.unique(ps)) <= {-1.0, 1.0}


def test_convert_to_store_and_load(bag, tmp_path):
    path, all_ev = bag
    out, n = rosbag_to_evs.convert(str(path), str(tmp_path / "seq.evs"))
    assert n == sum(len(e[0]) for e in all_ev)
    from esr_amd.data.store import EventStore
    st = EventStore(out)
    assert st.num_events("ori") == n
    assert st.num_events("down2") == len(range(0, n, 4))
    ev = st.events("ori", 0, n)           # [4, n]: x, y, t, p
    assert ev[2].min() == 0.0             # sequence-relative time
    assert (np.diff(ev[2]) >= 0).all()
    # down2 coordinates halved
    ev2 = st.events("down2", 0, st.num_events("down2"))
    assert ev2[0].max() <= 32


def test_uncompressed_and_topic_filter(bag, tmp_path):
    path, all_ev = bag
    # uncompressed chunk variant + wrong-topic filter
    msgs = [_event_array_msg([1], [2], [5], [0], [1])]
    p2 = tmp_path / "plain.bag"
    _make_bag(p2, msgs, compression=b"none", topic=b"/other")
    H, W, xs, ys, ts, ps = rosbag_to_evs.read_bag_events(str(p2))
    assert len(xs) == 1 and xs[0] == 1 and ys[0] == 2
    with pytest.raises(ValueError):
        rosbag_to_evs.read_bag_events(str(p2), topic="/dvs/events")
