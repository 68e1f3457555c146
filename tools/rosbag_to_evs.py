#!/usr/bin/env python3
"""Convert a ROS1 bag of dvs_msgs/EventArray messages to an EVS store.

Parity: ESR:generate_dataset/tools/rosbag_to_h5.py:1-191 — that script
needs the ROS stack (`import rosbag`); this one parses the ROS1 bag v2.0
container format directly in pure python (no ROS install), so DVS
recordings convert in this environment too.

Supported: bag format 2.0, chunk compression none/bz2 (lz4 if the `lz4`
module is importable), message type dvs_msgs/EventArray (the DVS/DAVIS
event camera driver topic).  Derived down-scale groups are written the
same way as the reference's packager (ori/down2/.../down16).

  python tools/rosbag_to_evs.py events.bag out.evs --topic /dvs/events
"""

import argparse
import bz2
import struct
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from esr_amd.data.store import EventStoreWriter  # noqa: E402

NAMES = {1: "ori", 2: "down2", 4: "down4", 8: "down8", 16: "down16"}

# ROS1 bag record opcodes
OP_MSG = 0x02
OP_BAGHDR = 0x03
OP_INDEX = 0x04
OP_CHUNK = 0x05
OP_CHUNKINFO = 0x06
OP_CONNECTION = 0x07


def _parse_header(buf):
    """Record header: sequence of (len u32, 'name=value') fields."""
    fields = {}
    off = 0
    while off < len(buf):
        (flen,) = struct.unpack_from("<I", buf, off)
        off += 4
        field = buf[off:off + flen]
        off += flen
        name, _, value = field.partition(b"=")
        fields[name.decode()] = value
    return fields


def _iter_records(buf, off=0):
    """Yield (header_fields, data_bytes) records from a buffer."""
    end = len(buf)
    while off < end:
        (hlen,) = struct.unpack_from("<I", buf, off)
        off += 4
        header = _parse_header(buf[off:off + hlen])
        off += hlen
        (dlen,) = struct.unpack_from("<I", buf, off)
        off += 4
        data = buf[off:off + dlen]
        off += dlen
        yield header, data


def _decompress(kind, data):
    if kind in (b"none", "none", None):
        return data
    if kind in (b"bz2", "bz2"):
        return bz2.decompress(data)
    if kind in (b"lz4", "lz4"):
        try:
            import lz4.frame
        except ImportError as e:
            raise RuntimeError("bag uses lz4 chunks but the lz4 module "
                               "is not installed") from e
        return lz4.frame.decompress(data)
    raise ValueError(f"unknown chunk compression {kind!r}")


def _decode_event_array(data):
    """Deserialize one dvs_msgs/EventArray message.

    Layout (ROS serialization):
      std_msgs/Header: seq u32, stamp (secs u32, nsecs u32),
                       frame_id (len u32 + bytes)
      height u32, width u32,
      events: n u32, then n x { x u16, y u16, ts (secs u32, nsecs u32),
                                polarity u8 }
    """
    off = 4 + 8  # seq + stamp
    (slen,) = struct.unpack_from("<I", data, off)
    off += 4 + slen
    height, width, n = struct.unpack_from("<III", data, off)
    off += 12
    ev = np.frombuffer(data, dtype=np.dtype([
        ("x", "<u2"), ("y", "<u2"), ("secs", "<u4"), ("nsecs", "<u4"),
        ("p", "u1")]), count=n, offset=off)
    ts = ev["secs"].astype(np.float64) + ev["nsecs"].astype(np.float64) * 1e-9
    ps = np.where(ev["p"] > 0, 1.0, -1.0)
    return (height, width, ev["x"].astype(np.float64),
            ev["y"].astype(np.float64), ts, ps)


def read_bag_events(path, topic=None, msg_type=b"dvs_msgs/EventArray"):
    """Return (height, width, xs, ys, ts, ps) concatenated over the bag."""
    raw = Path(path).read_bytes()
    magic, _, rest = raw.partition(b"\n")
    if not magic.startswith(b"#ROSBAG V2.0"):
        raise ValueError(f"not a ROS1 v2.0 bag: {magic[:20]!r}")

    conns = {}          # conn id -> (topic, type)
    chunks_out = []
    xs, ys, ts, ps = [], [], [], []
    H = W = None

    def handle(header, data):
        nonlocal H, W
        op = header["op"][0]
        if op == OP_CONNECTION:
            cid = struct.unpack("<I", header["conn"])[0]
            sub = _parse_header(data)
            conns[cid] = (header.get("topic", b"").decode(),
                          sub.get("type", b""))
        elif op == OP_MSG:
            cid = struct.unpack("<I", header["conn"])[0]
            ctopic, ctype = conns.get(cid, ("", b""))
            if ctype != msg_type:
                return
            if topic is not None and ctopic != topic:
                return
            h, w, x, y, t, p = _decode_event_array(data)
            H, W = h, w
            xs.append(x); ys.append(y); ts.append(t); ps.append(p)

    for header, data in _iter_records(raw, off=len(magic) + 1):
        op = header["op"][0]
        if op == OP_CHUNK:
            payload = _decompress(header.get("compression", b"none"), data)
            for h2, d2 in _iter_records(payload):
                handle(h2, d2)
        elif op in (OP_CONNECTION, OP_MSG):  # unchunked bags
            handle(header, data)
        # bag header / index / chunk-info records: skipped

    if not ts:
        raise ValueError(f"no {msg_type.decode()} messages"
                         + (f" on topic {topic}" if topic else ""))
    xs = np.concatenate(xs); ys = np.concatenate(ys)
    ts = np.concatenate(ts); ps = np.concatenate(ps)
    order = np.argsort(ts, kind="stable")
    return H, W, xs[order], ys[order], ts[order], ps[order]


def convert(bag_path, out_path, topic=None, levels=(1, 2, 4, 8, 16)):
    H, W, xs, ys, ts, ps = read_bag_events(bag_path, topic)
    t0 = ts[0]
    ts = ts - t0                      # bag time -> sequence-relative time
    with EventStoreWriter(out_path, (H, W)) as w:
        for lvl in levels:
            sub = slice(None, None, lvl * lvl)   # 1/k^2 count thinning
            w.add_group(NAMES[lvl], np.floor(xs[sub] / lvl),
                        np.floor(ys[sub] / lvl), ts[sub], ps[sub])
    return out_path, len(ts)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("bag_path")
    p.add_argument("out_path")
    p.add_argument("--topic", default=None,
                   help="restrict to one topic (default: any EventArray)")
    p.add_argument("--levels", type=int, nargs="+", default=[1, 2, 4, 8, 16])
    args = p.parse_args()
    out, n = convert(args.bag_path, args.out_path, args.topic,
                     tuple(args.levels))
    print(f"wrote {out}: {n} events")


if __name__ == "__main__":
    main()
