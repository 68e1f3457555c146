"""Small-tool coverage: profile summarizer + baseline-config arm specs."""

import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_summarize_profile(tmp_path):
    csv = tmp_path / "stats.csv"
    csv.write_text(
        '"Name","Calls","TotalDurationNs","AverageNs","Percentage"\n'
        '"conv2d_fwd_mfma_kernel<3,1,0,2>","10","5000000","500000","50"\n'
        '"igemm_wrw_gtcx35_nhwc_bf16","5","3000000","600000","30"\n'
        '"batched_transpose_64x64","20","1000000","50000","10"\n'
        '"Cijk_Ailk_Bjlk_BBS","4","1000000","250000","10"\n')
    out = subprocess.run(
        [sys.executable, str(REPO / "tools" / "summarize_profile.py"),
         str(csv), "4"], capture_output=True, text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert "native-conv" in out.stdout
    assert "miopen-conv" in out.stdout
    assert "total 10.0 ms" in out.stdout


def test_baseline_config_arms_spec():
    sys.path.insert(0, str(REPO / "tools"))
    import bench_baseline_configs as bbc
    assert set(bbc.ARMS) == {"config3_4x_64to256", "config4_longbptt_seq16",
                             "config5_dvs_fp16"}
    for name, arm in bbc.ARMS.items():
        flags = arm["flags"]
        assert "--metric-suffix" in flags
        assert any("{steps}" in f for f in flags)
    # config 5 is the fp16 DVS-native arm
    f5 = bbc.ARMS["config5_dvs_fp16"]["flags"]
    assert "fp16" in f5 and "180" in f5 and "240" in f5


def test_txt_to_evs_cli(tmp_path):
    import numpy as np
    rng = np.random.default_rng(0)
    n = 500
    txt = tmp_path / "ev.txt"
    data = np.stack([np.sort(rng.uniform(0, 1, n)),
                     rng.integers(0, 64, n),
                     rng.integers(0, 48, n),
                     rng.integers(0, 2, n)], axis=1)
    np.savetxt(txt, data)
    out = subprocess.run(
        [sys.executable, str(REPO / "tools" / "txt_to_evs.py"), str(txt),
         str(tmp_path / "seq.evs"), "--height", "48", "--width", "64"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    sys.path.insert(0, str(REPO))
    from esr_amd.data.store import EventStore
    st = EventStore(tmp_path / "seq.evs")
    assert st.num_events("ori") == n
    ev = st.events("down2", 0, st.num_events("down2"))
    assert ev[0].max() < 32 and ev[1].max() < 24
