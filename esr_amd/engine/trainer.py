"""Training runtime: DDP BPTT trainer.

Parity with the reference trainer (ESR:train_ours_cnt_seq.py:88-739):
iteration- and epoch-based modes, truncated BPTT over window sequences with
persistent ConvGRU state reset per sequence, middle-frame MSE supervision,
loss all-reduce for logging, monitor/early-stop, periodic + best
checkpointing, lr stepping every `lr_change_rate` iterations while
lr >= 1e-4, resume.

MI355X deltas:
  * optional bf16 autocast for forward+loss (the reference is fp32-only);
  * the per-iteration dist.barrier() is dropped (DDP's all-reduce is the
    sync point; the barrier is one more latency-bound xGMI collective);
  * non_blocking H2D copies from pinned buffers.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

from ..config import build_lr_scheduler, build_optimizer
from ..parallel import get_rank, reduce_tensor
from ..utils import MetricTracker, MetricWriter
from .checkpoint import Resumer, save_checkpoint

__all__ = ["Trainer"]


class Trainer:
    def __init__(self, config_parser, train_dataloader, valid_dataloader,
                 model, loss_fns, optimizer, lr_scheduler, logger, device,
                 resume: str | None = None, reset: bool = False):
        self.config_parser = config_parser
        self.cfg = config_parser.config
        self.train_dataloader = train_dataloader
        self.valid_dataloader = valid_dataloader
        self.model = model
        self.loss_fns = loss_fns
        self.optimizer = optimizer
        self.lr_scheduler = lr_scheduler
        self.logger = logger
        self.device = device
        self.amp_dtype = {"bf16": torch.bfloat16, "fp16": torch.float16,
                          None: None, "fp32": None}[
                              self.cfg.get("precision", None)]

        tcfg = self.cfg["trainer"]
        self.monitor = tcfg.get("monitor", "off")
        self.checkpoint_dir = config_parser.save_dir
        self.do_validation = valid_dataloader is not None

        it_cfg = tcfg.get("iteration_based_train", {"enabled": False})
        ep_cfg = tcfg.get("epoch_based_train", {"enabled": False})
        if it_cfg.get("enabled") == ep_cfg.get("enabled"):
            raise ValueError("exactly one training mode must be enabled")
        if it_cfg.get("enabled"):
            self.training_mode = "iteration_based_train"
            self.iterations = int(it_cfg["iterations"])
            self.save_period = it_cfg["save_period"]
            self.train_log_step = it_cfg["train_log_step"]
            self.valid_log_step = it_cfg["valid_log_step"]
            self.valid_step = it_cfg["valid_step"]
            self.lr_change_rate = it_cfg["lr_change_rate"]
        else:
            self.training_mode = "epoch_based_train"
            self.epochs = ep_cfg["epochs"]
            self.save_period = ep_cfg["save_period"]
            self.train_log_step = max(len(train_dataloader)
                                      // ep_cfg["train_log_step"], 1)
            self.valid_log_step = max(
                (len(valid_dataloader) if valid_dataloader else 1)
                // ep_cfg["valid_log_step"], 1)
            self.valid_step = ep_cfg["valid_step"]
        self.start_epoch = 1

        if self.monitor == "off":
            self.mnt_mode = "off"
            self.mnt_best = 0
        else:
            self.mnt_mode, self.mnt_metric = self.monitor.split()
            assert self.mnt_mode in ("min", "max")
            self.mnt_best = math.inf if self.mnt_mode == "min" else -math.inf
        self.early_stop = tcfg.get("early_stop", math.inf)
        self.not_improved_count = 0

        self.writer = MetricWriter(config_parser.log_dir,
                                   enabled=get_rank() == 0)
        self.train_metrics = MetricTracker(
            ["train_mse_loss", "train_loss"],
            writer=self.writer if get_rank() == 0 else None)
        self.valid_metrics = MetricTracker(["valid_mse_loss", "valid_loss"])

        self.mid_idx = (train_dataloader.seqn - 1) // 2
        self.start_iteration = 0

        # hipGraph-captured training step (GPU, fixed shapes):
        # built lazily on the first batch
        self.use_graphs = (tcfg.get("hip_graphs", False)
                           and self.device.type == "cuda")
        self._graph_step = None

        # failure detection (the reference hangs forever on a
        # dead rank); heartbeats in the training loop, monitor thread
        # aborts so the launcher can relaunch from the last checkpoint
        self._wd_cfg = tcfg.get("watchdog", {})
        self.watchdog = None

        if resume is not None:
            self._resume_checkpoint(resume, reset)

    # ---------------- core step ----------------

    def _unwrapped(self):
        return self.model.module if hasattr(self.model, "module") else self.model

    def _autocast(self):
        if self.amp_dtype is not None and self.device.type == "cuda":
            return torch.autocast("cuda", dtype=self.amp_dtype)
        import contextlib
        return contextlib.nullcontext()

    def _get_graph_step(self, inputs_seq):
        """Build (or fetch) the captured BPTT step for this batch shape."""
        from ..parallel import get_world_size
        from .graph_runner import GraphedBPTTStep, flatten_grads
        shared = isinstance(inputs_seq, dict)
        if shared:
            inp_shape = tuple(inputs_seq["frames"].shape)
            gt_shape = tuple(inputs_seq["gt_mids"].shape)
            n_windows = gt_shape[1]
        else:
            inp_shape = tuple(inputs_seq[0]["inp_scaled_cnt"].shape)
            gt_shape = tuple(inputs_seq[0]["gt_cnt"][:, self.mid_idx].shape)
            n_windows = len(inputs_seq)
        key = (shared, n_windows, inp_shape, gt_shape)
        if self._graph_step is not None and self._graph_step[0] == key:
            return self._graph_step[1]
        params = [p for p in self.model.parameters() if p.requires_grad]
        flat = flatten_grads(params, self.device)
        runner = GraphedBPTTStep(
            self.model, self.optimizer, flat, n_windows,
            inp_shape, gt_shape, self.device,
            amp_dtype=self.amp_dtype, world_size=get_world_size(),
            sequence=shared, seqn=self.train_dataloader.seqn).capture()
        self._graph_step = (key, runner)
        return runner

    def graphed_bptt_step(self, inputs_seq):
        runner = self._get_graph_step(inputs_seq)
        if isinstance(inputs_seq, dict):
            loss, mse = runner.run([inputs_seq["frames"]],
                                   [inputs_seq["gt_mids"]])
        else:
            loss, mse = runner.run(
                [w["inp_scaled_cnt"] for w in inputs_seq],
                [w["gt_cnt"][:, self.mid_idx] for w in inputs_seq])
        return loss.detach().clone(), mse.detach().clone(), None

    def bptt_step(self, inputs_seq, train: bool = True):
        """One optimizer step over a window sequence: loss summed over the
        seqn-sliding windows, single backward through persistent GRU state
        (parity: ESR:train_ours_cnt_seq.py:210-235)."""
        if train and self.use_graphs:
            try:
                return self.graphed_bptt_step(inputs_seq)
            except Exception as e:
                self.logger.warning(
                    f"hipGraph step failed ({type(e).__name__}: {e}); "
                    f"falling back to eager")
                self.use_graphs = False
        if train:
            self.optimizer.zero_grad(set_to_none=True)
        self._unwrapped().reset_states()
        loss = 0
        mse_loss = None
        if isinstance(inputs_seq, dict):
            # shared-encoder sequence batch (collate='shared'); bypasses the
            # DDP wrapper's forward, so it requires either single process or
            # the graphed step's own all-reduce
            if train and hasattr(self.model, "module"):
                raise RuntimeError(
                    "collate='shared' training under DDP requires "
                    "trainer.hip_graphs (the graphed step all-reduces the "
                    "flat gradient itself)")
            frames = inputs_seq["frames"].to(self.device, non_blocking=True)
            gts = inputs_seq["gt_mids"].to(self.device, non_blocking=True)
            with self._autocast():
                preds = self._unwrapped().forward_sequence(
                    frames, inputs_seq["seqn"])
            for w, pred in enumerate(preds):
                mse_loss = self.loss_fns["mse"](pred.float(), gts[:, w].float())
                loss = loss + mse_loss
            pred = preds[-1]
        else:
            for inputs in inputs_seq:
                inp = inputs["inp_scaled_cnt"].to(self.device, non_blocking=True)
                gt = inputs["gt_cnt"][:, self.mid_idx].to(self.device,
                                                          non_blocking=True)
                with self._autocast():
                    pred = self.model(inp)
                    if pred.shape[-2:] != gt.shape[-2:]:
                        pred = F.interpolate(pred, size=gt.shape[-2:],
                                             mode="bicubic", align_corners=False)
                    mse_loss = self.loss_fns["mse"](pred.float(), gt.float())
                loss = loss + mse_loss
        if train:
            loss.backward()
            self._sync_grads_if_needed()
            self.optimizer.step()
        return loss.detach(), mse_loss.detach(), pred.detach()

    def _sync_grads_if_needed(self):
        """Manual gradient all-reduce for the eager path when the model is
        not DDP-wrapped (hip_graphs mode skips wrap_ddp; if graph capture
        fails at runtime the eager fallback must still synchronize ranks —
        advisor finding r1)."""
        from ..parallel import get_world_size
        world = get_world_size()
        if world <= 1 or hasattr(self.model, "module"):
            return
        import torch.distributed as dist
        grads = [p.grad for p in self.model.parameters()
                 if p.requires_grad and p.grad is not None]
        if not grads:
            return
        flat = torch._utils._flatten_dense_tensors(grads)
        dist.all_reduce(flat)
        flat.div_(world)
        for g, s in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
            g.copy_(s)

    # ---------------- loops ----------------

    def train(self):
        self._start_watchdog()
        try:
            if self.training_mode == "iteration_based_train":
                self.iteration_based_training()
            else:
                self.epoch_based_training()
        finally:
            if self.watchdog is not None:
                self.watchdog.stop()

    def _start_watchdog(self):
        import torch.distributed as dist
        if not (dist.is_available() and dist.is_initialized()
                and dist.get_world_size() > 1):
            return
        if not self._wd_cfg.get("enabled", True):
            return
        from ..parallel import Watchdog
        self.watchdog = Watchdog(
            timeout=float(self._wd_cfg.get("timeout", 120.0)),
            interval=float(self._wd_cfg.get("interval", 5.0))).start()
        if get_rank() == 0:
            self.logger.info(
                f"Watchdog armed: timeout {self.watchdog.timeout}s, "
                f"interval {self.watchdog.interval}s")

    def _beat(self):
        if self.watchdog is not None:
            self.watchdog.beat()

    def iteration_based_training(self):
        self.model.train()
        self.train_metrics.reset()
        valid_stamp = 1
        epoch = 0
        done = False
        while not done:
            if hasattr(self.train_dataloader, "set_epoch"):
                # epoch-keyed item seeds + sharded-sampler shuffle
                self.train_dataloader.set_epoch(epoch)
            elif self.train_dataloader.dist_sampler is not None:
                self.train_dataloader.dist_sampler.set_epoch(epoch)
            for idx, inputs_seq in enumerate(self.train_dataloader):
                iter_idx = idx + len(self.train_dataloader) * epoch \
                    + self.start_iteration
                best = False
                loss, mse_loss, pred = self.bptt_step(inputs_seq, train=True)
                self._beat()

                reduced_mse = reduce_tensor(mse_loss)
                reduced_loss = reduce_tensor(loss)

                if get_rank() == 0:
                    self.writer.set_step(iter_idx)
                    self.train_metrics.update("train_mse_loss", reduced_mse.item())
                    self.train_metrics.update("train_loss", reduced_loss.item())
                    lr = self.lr_scheduler.get_last_lr()[0]
                    self.writer.add_scalar("learning_rate", lr)
                    if iter_idx % self.train_log_step == 0:
                        self.logger.info(
                            f"Train epoch {epoch + 1} iter {iter_idx}/"
                            f"{self.iterations} mse {reduced_mse.item():.4e} "
                            f"loss {reduced_loss.item():.4e} lr {lr:.3e}")
                    self._maybe_visualize(iter_idx, inputs_seq, pred)

                if self.do_validation and iter_idx % self.valid_step == 0 \
                        and iter_idx != 0:
                    with torch.no_grad():
                        val_log = self._valid(valid_stamp)
                    if get_rank() == 0:
                        for k, v in val_log.items():
                            self.writer.add_scalar(f"stamp_{k}", v,
                                                   global_step=valid_stamp)
                        self.logger.info(f"Valid stamp {valid_stamp}: {val_log}")
                    stop, best = self.eval_model_performance(val_log)
                    if stop:
                        done = True
                        break
                    valid_stamp += 1
                    self.model.train()

                if get_rank() == 0 and (
                        (iter_idx % self.save_period == 0 and iter_idx != 0)
                        or best):
                    self._save(iter_idx, best)

                if self.lr_scheduler is not None \
                        and iter_idx % self.lr_change_rate == 0 \
                        and iter_idx != 0 \
                        and self.lr_scheduler.get_last_lr()[0] >= 1e-4:
                    self.lr_scheduler.step()

                if iter_idx + 1 >= self.iterations:
                    if get_rank() == 0:
                        self.logger.info("Training completes!")
                    done = True
                    break
            epoch += 1

    def epoch_based_training(self):
        for epoch in range(self.start_epoch, self.epochs + 1):
            if hasattr(self.train_dataloader, "set_epoch"):
                self.train_dataloader.set_epoch(epoch)
            elif self.train_dataloader.dist_sampler is not None:
                self.train_dataloader.dist_sampler.set_epoch(epoch)
            self.model.train()
            self.train_metrics.reset()
            for idx, inputs_seq in enumerate(self.train_dataloader):
                loss, mse_loss, _ = self.bptt_step(inputs_seq, train=True)
                self._beat()
                reduced_mse = reduce_tensor(mse_loss)
                reduced_loss = reduce_tensor(loss)
                if get_rank() == 0:
                    step = (epoch - 1) * len(self.train_dataloader) + idx
                    self.writer.set_step(step)
                    self.train_metrics.update("train_mse_loss", reduced_mse.item())
                    self.train_metrics.update("train_loss", reduced_loss.item())
                    if idx % self.train_log_step == 0:
                        self.logger.info(
                            f"Train epoch {epoch} [{idx}/{len(self.train_dataloader)}]"
                            f" mse {reduced_mse.item():.4e}")
            log = self.train_metrics.result()
            if self.do_validation and epoch % self.valid_step == 0:
                with torch.no_grad():
                    log.update(self._valid(epoch))
            if self.lr_scheduler is not None:
                self.lr_scheduler.step()
            stop, best = self.eval_model_performance(log)
            if get_rank() == 0 and (epoch % self.save_period == 0 or best):
                self._save(epoch, best)
            if stop:
                break
        if get_rank() == 0:
            self.logger.info("Training completes!")

    def _valid(self, stamp):
        self.model.eval()
        self.valid_metrics.reset()
        for batch_idx, inputs_seq in enumerate(self.valid_dataloader):
            loss, mse_loss, _ = self.bptt_step(inputs_seq, train=False)
            reduced_mse = reduce_tensor(mse_loss)
            reduced_loss = reduce_tensor(loss)
            self.valid_metrics.update("valid_mse_loss", reduced_mse.item())
            self.valid_metrics.update("valid_loss", reduced_loss.item())
            if get_rank() == 0 and batch_idx % self.valid_log_step == 0:
                self.logger.info(
                    f"Valid stamp {stamp} [{batch_idx}/{len(self.valid_dataloader)}]"
                    f" mse {reduced_mse.item():.4e}")
        return self.valid_metrics.result()

    def _maybe_visualize(self, iter_idx, inputs_seq, pred):
        """Render input/scaled/pred/gt count maps to the writer every
        train_img_writer_num iterations (parity:
        ESR:train_ours_cnt_seq.py:258-293).  No-ops unless a TensorBoard
        sink is active."""
        vis_cfg = self.cfg["trainer"].get("vis", {"enabled": False})
        if not vis_cfg.get("enabled", False) or pred is None:
            return
        if self.writer is None or getattr(self.writer, "_tb", None) is None:
            return
        step = vis_cfg.get("train_img_writer_num", 20)
        if iter_idx % step != 0:
            return
        if isinstance(inputs_seq, dict):
            return  # shared-sequence batches carry no per-window dicts
        from ..utils.vis import EventVisualizer
        vis = EventVisualizer()
        inputs = inputs_seq[-1]
        hwc = lambda t: t.cpu().float().numpy().transpose(1, 2, 0)  # noqa: E731
        self.writer.add_image(
            "train_inp_events_cnt",
            vis.plot_event_cnt(hwc(inputs["inp_cnt"][0, self.mid_idx])),
            global_step=iter_idx)
        self.writer.add_image(
            "train_inp_scaled_events_cnt",
            vis.plot_event_cnt(hwc(inputs["inp_scaled_cnt"][0, self.mid_idx])),
            global_step=iter_idx)
        self.writer.add_image(
            "train_esr_events_cnt",
            vis.plot_event_cnt(hwc(pred[0].round())), global_step=iter_idx)
        self.writer.add_image(
            "train_gt_events_cnt",
            vis.plot_event_cnt(hwc(inputs["gt_cnt"][0, self.mid_idx])),
            global_step=iter_idx)

    # ---------------- monitoring / checkpoints ----------------

    def eval_model_performance(self, log):
        """Monitor-metric improvement check + early stop (parity:
        ESR:train_ours_cnt_seq.py:383-424)."""
        best = False
        stop = False
        if self.mnt_mode != "off":
            if self.mnt_metric not in log:
                if get_rank() == 0:
                    self.logger.warning(
                        f"metric '{self.mnt_metric}' not found; skipping")
                return False, False
            val = log[self.mnt_metric]
            improved = (self.mnt_mode == "min" and val <= self.mnt_best) or \
                       (self.mnt_mode == "max" and val >= self.mnt_best)
            if improved:
                self.mnt_best = val
                self.not_improved_count = 0
                best = True
            else:
                self.not_improved_count += 1
            if self.not_improved_count > self.early_stop:
                if get_rank() == 0:
                    self.logger.info(
                        f"No improvement for {self.early_stop} stamps; stopping.")
                stop = True
        return stop, best

    def _save(self, idx, best):
        if self.checkpoint_dir is None:
            return
        key = "epoch" if self.training_mode == "epoch_based_train" else "iteration"
        trainer_state = {"training_mode": self.training_mode, key: idx,
                         "monitor_best": self.mnt_best}
        fn = self.checkpoint_dir / f"checkpoint-{key}{idx}.pth"
        save_checkpoint(fn, self.cfg, self.model, self.optimizer,
                        self.lr_scheduler, trainer_state)
        self.logger.info(f"Saved checkpoint: {fn}")
        if best:
            bfn = self.checkpoint_dir / f"model_best_until_{key}{idx}.pth"
            save_checkpoint(bfn, self.cfg, self.model, self.optimizer,
                            self.lr_scheduler, trainer_state)
            self.logger.info(f"Saved current best: {bfn}")

    def _resume_checkpoint(self, path, reset):
        resumer = Resumer(path, self.logger, self.cfg)
        tstate = resumer.resume_trainer()
        same_mode = tstate.get("training_mode") == self.training_mode
        if not reset and same_mode:
            if self.training_mode == "epoch_based_train":
                self.start_epoch = tstate["epoch"] + 1
            else:
                self.start_iteration = tstate["iteration"] + 1
            self.mnt_best = tstate["monitor_best"]
        resumer.resume_model(self.model)
        resumer.resume_optimizer(self.optimizer)
        resumer.resume_lr_scheduler(self.lr_scheduler)
        if get_rank() == 0:
            self.logger.info(f"Resumed from {path}")


def build_training(config_parser, device, logger, resume=None, reset=False):
    """Construct loaders, model, optimizer, scheduler, Trainer from a config
    (parity: ESR:train_ours_cnt_seq.py:742-807, registry-based)."""
    import torch.nn as nn

    from ..data import SequenceDataLoader
    from ..models import build_model
    from ..parallel import wrap_ddp

    cfg = config_parser.config
    train_loader = SequenceDataLoader(cfg["train_dataloader"])
    valid_loader = SequenceDataLoader(cfg["valid_dataloader"]) \
        if cfg.get("valid_dataloader") else None

    model = build_model(cfg["model"]["name"], **cfg["model"]["args"]).to(device)
    use_graphs = cfg["trainer"].get("hip_graphs", False) \
        and device.type == "cuda"
    if use_graphs:
        # graphed mode does its own single-bucket RCCL all-reduce inside the
        # captured step; DDP's hook-driven reducer is not used
        if cfg.get("sync_bn", False):
            raise ValueError("hip_graphs is incompatible with sync_bn")
    else:
        model = wrap_ddp(model, device=device if device.type == "cuda" else None,
                         sync_bn=cfg.get("sync_bn", False))

    loss_fns = {"mse": nn.MSELoss(), "l1": nn.L1Loss()}
    params = [p for p in model.parameters() if p.requires_grad]
    opt_kwargs = dict(cfg["optimizer"]["args"])
    if use_graphs and cfg["optimizer"]["name"] in ("Adam", "AdamW"):
        opt_kwargs["capturable"] = True
    optimizer = build_optimizer(cfg["optimizer"]["name"], params,
                                **opt_kwargs)
    lr_scheduler = build_lr_scheduler(cfg["lr_scheduler"]["name"], optimizer,
                                      **cfg["lr_scheduler"]["args"])
    return Trainer(config_parser, train_loader, valid_loader, model, loss_fns,
                   optimizer, lr_scheduler, logger, device,
                   resume=resume, reset=reset)
