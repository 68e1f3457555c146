"""hipGraph-captured BPTT training step.

The flagship model is launch-latency-bound on MI355X (1.8 M params,
~130 kernels per window); capturing the whole BPTT optimizer step (all
sliding windows forward + one backward + gradient all-reduce + Adam) in a
hipGraph and replaying it cuts the step time ~2x (see profiles/README.md).

Used by the Trainer when ``trainer.hip_graphs`` is enabled (GPU, fixed
shapes) and by bench.py.  In distributed mode the model must NOT be
DDP-wrapped — gradients live in one flat buffer that is all-reduced with a
single RCCL call inside the graph (xGMI collectives on this model's few-MB
gradients are latency-bound, so one fused call is the right shape —
SURVEY §2.4).
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn.functional as F

__all__ = ["GraphedBPTTStep", "flatten_grads"]


def flatten_grads(params, device, dtype=torch.float32):
    """Point every param's .grad into one flat buffer; returns the buffer."""
    total = sum(p.numel() for p in params)
    flat = torch.zeros(total, device=device, dtype=dtype)
    off = 0
    for p in params:
        p.grad = flat[off:off + p.numel()].view_as(p)
        off += p.numel()
    return flat


class GraphedBPTTStep:
    """Captures loss = sum_w MSE(model(inp_w), gt_w); backward; (all-reduce);
    optimizer.step() into a replayable hipGraph with static I/O buffers."""

    def __init__(self, model, optimizer, flat_grad, n_windows,
                 inp_shape, gt_shape, device, amp_dtype=None,
                 world_size: int = 1, warmup: int = 3,
                 sequence: bool = False, seqn: int = 3):
        self.model = model
        self.optimizer = optimizer
        self.flat_grad = flat_grad
        self.device = device
        self.amp_dtype = amp_dtype
        self.world = world_size
        self.sequence = sequence
        self.seqn = seqn
        if sequence:
            # inp_shape = [B, L, C, H, W]; gt_shape = [B, n_windows, C, H, W]
            self.static_in = [torch.zeros(inp_shape, device=device)]
            self.static_gt = [torch.zeros(gt_shape, device=device)]
        else:
            self.static_in = [torch.zeros(inp_shape, device=device)
                              for _ in range(n_windows)]
            self.static_gt = [torch.zeros(gt_shape, device=device)
                              for _ in range(n_windows)]
        self.graph = None
        self.static_loss = None
        self._warmup = warmup

    def _autocast(self):
        import contextlib
        if self.amp_dtype is None:
            return contextlib.nullcontext()
        return torch.autocast("cuda", dtype=self.amp_dtype,
                              cache_enabled=False)

    def _body(self):
        self.flat_grad.zero_()
        inner = self.model.module if hasattr(self.model, "module") else self.model
        inner.reset_states()
        loss = 0
        mse = None
        if self.sequence:
            with self._autocast():
                preds = inner.forward_sequence(self.static_in[0], self.seqn)
            gts = self.static_gt[0]
            for w, pred in enumerate(preds):
                mse = F.mse_loss(pred.float(), gts[:, w])
                loss = loss + mse
        else:
            for inp, gt in zip(self.static_in, self.static_gt):
                with self._autocast():
                    pred = self.model(inp)
                mse = F.mse_loss(pred.float(), gt)
                loss = loss + mse
        loss.backward()
        if self.world > 1:
            dist.all_reduce(self.flat_grad)
            self.flat_grad.div_(self.world)
        self.optimizer.step()
        return loss, mse

    def _snapshot_state(self):
        params = [p for g in self.optimizer.param_groups for p in g["params"]]
        return [p.detach().clone() for p in params], params

    def _restore_state(self, snapshot, params):
        """Undo the warmup steps: params back to their pre-warmup values and
        the optimizer state zeroed in place (= never stepped).  In-place so
        the state tensors the capture records keep their addresses."""
        with torch.no_grad():
            for p, s in zip(params, snapshot):
                p.copy_(s)
            for state in self.optimizer.state.values():
                for v in state.values():
                    if torch.is_tensor(v):
                        v.zero_()

    def capture(self):
        # warmup on a side stream initializes cuDNN/MIOpen plans, autograd
        # graph allocations and the capturable optimizer's state TENSORS —
        # but the real optimizer.step()s it runs would perturb the model
        # with garbage-gradient updates (advisor finding r1), so snapshot
        # first and roll back before recording the graph.
        snapshot, params = self._snapshot_state()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(self._warmup):
                self._body()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self._restore_state(snapshot, params)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.static_loss, self.static_mse = self._body()
        torch.cuda.synchronize()
        return self

    def run(self, window_inputs, window_gts):
        """Copy inputs into the static buffers (H2D if needed) and replay."""
        for si, sg, inp, gt in zip(self.static_in, self.static_gt,
                                   window_inputs, window_gts):
            si.copy_(inp, non_blocking=True)
            sg.copy_(gt, non_blocking=True)
        self.graph.replay()
        return self.static_loss, self.static_mse
