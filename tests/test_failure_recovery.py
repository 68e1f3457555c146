"""E2E failure-detection + relaunch-from-checkpoint drill (gloo, CPU).

The reference hangs forever when a rank dies (SURVEY §5,
ESR:train_ours_cnt_seq.py:339).  Here the Trainer arms the Watchdog in
distributed runs: when rank 1 crashes mid-training, rank 0's watchdog
detects the stale heartbeat and aborts (exit 134) instead of hanging on
the next collective — the launcher can then relaunch the job with
--resume, which this test does, asserting the run continues from the
saved iteration counter and completes.
"""

import os
from pathlib import Path

import pytest
import torch.multiprocessing as mp

WORLD = 2


def _tiny_cfg(ws, datalist, iterations):
    ds = {
        "scale": 2, "ori_scale": "down2", "time_bins": 1,
        "need_gt_frame": False, "need_gt_events": True,
        "mode": "events", "window": 256, "sliding_window": 128,
        "data_augment": {"enabled": False, "augment": [], "augment_prob": []},
        "hot_filter": {"enabled": False},
        "sequence": {"sequence_length": 3, "seqn": 3, "step_size": None,
                     "pause": {"enabled": False,
                               "proba_pause_when_running": 0.0,
                               "proba_pause_when_paused": 0.0}},
    }
    dl = {"use_ddp": True, "path_to_datalist_txt": str(datalist),
          "batch_size": 1, "shuffle": False, "num_workers": 0,
          "pin_memory": False, "drop_last": True, "dataset": ds}
    return {
        "experiment": "wd-e2e", "SEQN": 3,
        "model": {"name": "ESRNet",
                  "args": {"inch": 2, "basech": 4, "num_frame": 3,
                           "upsampler": "pixelshuffle"}},
        "optimizer": {"name": "Adam", "args": {"lr": 1e-3}},
        "lr_scheduler": {"name": "ExponentialLR", "args": {"gamma": 0.95}},
        "trainer": {
            "output_path": str(ws / "runs"),
            "watchdog": {"enabled": True, "timeout": 3.0, "interval": 0.3},
            "epoch_based_train": {"enabled": False},
            "iteration_based_train": {
                "enabled": True, "iterations": iterations, "save_period": 2,
                "train_log_step": 100, "valid_log_step": 100,
                "valid_step": 10_000, "lr_change_rate": 10_000},
            "monitor": "off", "tensorboard": False,
            "vis": {"enabled": False},
        },
        "train_dataloader": dl,
        "valid_dataloader": None,
    }


def _train_child(rank, port, ws, datalist, crash_at, resume, run_id, q):
    os.environ.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
                       "RANK": str(rank), "WORLD_SIZE": str(WORLD),
                       "LOCAL_RANK": str(rank)})
    import torch
    from esr_amd.config import ConfigParser
    from esr_amd.engine import build_training
    from esr_amd.parallel import init_distributed

    init_distributed()
    torch.manual_seed(7 + rank)
    iterations = 50 if crash_at is not None else 6
    cfg = _tiny_cfg(Path(ws), datalist, iterations)
    cp = ConfigParser(cfg, run_id=run_id, make_dirs=rank == 0)
    import logging
    logger = logging.getLogger(f"wd-e2e-{rank}")
    device = torch.device("cpu")
    trainer = build_training(cp, device, logger, resume=resume)

    if crash_at is not None and rank == 1:
        # simulate a SILENT rank (the RCCL failure mode: the process is
        # alive but wedged, collectives hang instead of erroring — gloo
        # would error fast on a dead peer, bypassing the watchdog): stop
        # heartbeating and disable self-abort, keep running
        orig = trainer.bptt_step
        calls = {"n": 0}

        def silent_step(*a, **kw):
            calls["n"] += 1
            if calls["n"] > crash_at and trainer.watchdog is not None:
                trainer.watchdog.beat = lambda: None
                trainer.watchdog.on_failure = lambda ranks: None
            return orig(*a, **kw)
        trainer.bptt_step = silent_step

    if resume is not None:
        q.put(("start_iteration", rank, trainer.start_iteration))
    trainer.train()
    q.put(("done", rank, trainer.start_iteration))


@pytest.mark.timeout(600)
def test_rank_death_detected_and_resumable(tmp_path):
    from esr_amd.data import make_synthetic_dataset
    datalist = make_synthetic_dataset(tmp_path / "data", num_sequences=1,
                                      resolution=(32, 32), num_events=4_000,
                                      seed=5)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()

    # phase 1: rank 1 goes silent after 3 steps; rank 0 must NOT hang —
    # its watchdog aborts with 134 once the heartbeat goes stale
    procs = [ctx.Process(target=_train_child,
                         args=(r, 29611, str(tmp_path), datalist, 3, None,
                               "wd0", q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    procs[0].join(timeout=240)
    assert procs[0].exitcode == 134, \
        f"rank 0 should watchdog-abort, got {procs[0].exitcode}"
    # the silent rank then dies on its next collective (peer gone) or is
    # cleaned up here; either way it must not survive as a zombie trainer
    procs[1].join(timeout=60)
    if procs[1].exitcode is None:
        procs[1].terminate()
        procs[1].join(timeout=30)

    ckpts = sorted((tmp_path / "runs" / "wd0" / "models")
                   .glob("checkpoint-iteration*.pth"))
    assert ckpts, "no checkpoint was saved before the crash"
    latest = str(ckpts[-1])

    # phase 2: relaunch from the checkpoint; training resumes past the
    # saved iteration and completes cleanly on both ranks
    procs = [ctx.Process(target=_train_child,
                         args=(r, 29613, str(tmp_path), datalist, None,
                               latest, "wd1", q))
             for r in range(WORLD)]
    for p in procs:
        p.start()
    msgs = [q.get(timeout=240) for _ in range(2 * WORLD)]
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    starts = [m[2] for m in msgs if m[0] == "start_iteration"]
    assert len(starts) == WORLD
    assert all(s > 0 for s in starts), "resume did not restore the counter"
    assert len([m for m in msgs if m[0] == "done"]) == WORLD
