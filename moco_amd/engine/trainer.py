"""Training engine: the per-step hot loop and the per-epoch driver.

Reference parity: train() at main_moco.py:290-335 (meters, print cadence,
acc1/acc5 on the contrastive logits).  MI355X-native differences:
- augmentation runs on the GPU (ops.two_crop_augment), the loader ships raw
  tensors only;
- compute dtype is bf16 via autocast with fp32 master weights (the MFMA
  path); CPU runs fp32;
- activations/filters use channels_last (NHWC) — the layout the conv kernels
  and MIOpen's fused paths want on gfx950;
- the optimizer step is one fused multi-tensor HIP launch (ops.FusedSGD);
- per-phase timing uses CUDA events, not host clocks.
"""

import time
from typing import Optional

import torch

from ..builder import moco_loss
from ..ops import two_crop_augment
from ..utils import AverageMeter, ProgressMeter, accuracy


class MoCoStepper:
    """Owns one full training step: aug -> forward -> loss -> backward ->
    fused optimizer.  Used by main_moco.py, bench.py and the smoke test."""

    def __init__(self, model, optimizer, device: torch.device,
                 aug_plus: bool = False, out_size: int = 224,
                 dtype: torch.dtype = torch.bfloat16,
                 channels_last: bool = True,
                 aug_generator: Optional[torch.Generator] = None,
                 use_graph: Optional[bool] = None):
        self.model = model
        self.optimizer = optimizer
        self.device = device
        self.aug_plus = aug_plus
        self.out_size = out_size
        self.is_cuda = device.type == "cuda"
        if self.is_cuda:
            # MIOpen kernel search: without this, untuned conv configs fall
            # back to naive_conv_* kernels (measured: 84% of step time on
            # MI355X).  Shapes are static per run, so the find cost is a
            # one-time warmup expense.
            torch.backends.cudnn.benchmark = True
        self.dtype = dtype if self.is_cuda else torch.float32
        self.channels_last = channels_last and self.is_cuda
        self.aug_generator = aug_generator
        self._timing = False
        self._phase_events = None
        # hipGraph capture of the whole step (fwd+bwd+opt+enqueue): kills
        # the per-step launch/dispatch host cost.  World-1 only (RCCL
        # collectives stay eager) and off while phase-timing.
        if use_graph is None:
            import os
            use_graph = (self.is_cuda
                         and os.environ.get("MOCO_AMD_NO_GRAPH", "0") != "1"
                         and not (torch.distributed.is_available()
                                  and torch.distributed.is_initialized()))
        self.use_graph = bool(use_graph) and self.is_cuda
        self._graph = None
        self._graph_key = None
        self._gwarm = 0

    def augment(self, raw: torch.Tensor):
        raw = raw.to(self.device, non_blocking=True)
        return two_crop_augment(
            raw, out_size=self.out_size, aug_plus=self.aug_plus,
            out_dtype=self.dtype if self.dtype != torch.float32 else None,
            channels_last=self.channels_last,
            generator=self.aug_generator)

    def step(self, im_q: torch.Tensor, im_k: torch.Tensor):
        """One optimization step on pre-augmented crops.
        Returns (loss, logits, labels)."""
        raw_model = getattr(self.model, "module", self.model)
        if getattr(raw_model, "graph_safe", False):
            # leaving graph mode (e.g. a phase-timing step): back to the
            # eager enqueue-in-forward path
            raw_model.flush_queue()
            raw_model.graph_safe = False
        autocast_on = self.is_cuda and self.dtype != torch.float32
        self._mark("fwd")
        with torch.autocast("cuda", dtype=self.dtype, enabled=autocast_on):
            logits, labels = self.model(im_q, im_k)
        self._mark("loss")
        loss = moco_loss(logits)
        self.optimizer.zero_grad(set_to_none=True)
        self._mark("bwd")
        loss.backward()
        self._mark("opt")
        self.optimizer.step()
        self._mark("end")
        return loss, logits, labels

    def step_from_raw(self, raw: torch.Tensor):
        self._phase_events = [] if self._timing else None
        self._mark("aug")
        im_q, im_k = self.augment(raw)
        if self.use_graph and not self._timing:
            return self.step_graphed(im_q, im_k)
        return self.step(im_q, im_k)

    # -- hipGraph whole-step capture ----------------------------------------
    def _inner_step(self, im_q, im_k):
        """The capturable step body: fwd -> loss -> bwd -> deferred
        enqueue.  Grads are None on entry (set_to_none=True), so the
        captured backward ASSIGNS each grad buffer — replays overwrite
        them in place with no per-step zero or accumulate kernels.  The
        fused SGD runs eagerly after each replay (one launch; grad
        addresses are pool-stable, and lr changes need no re-capture)."""
        with torch.autocast("cuda", dtype=self.dtype,
                            enabled=self.dtype != torch.float32):
            logits, labels = self.model(im_q, im_k)
        loss = moco_loss(logits)
        loss.backward()
        raw_model = getattr(self.model, "module", self.model)
        raw_model.flush_queue()
        return loss, logits, labels

    def step_graphed(self, im_q: torch.Tensor, im_k: torch.Tensor):
        key = im_q.shape
        if self._graph_key != key:
            self._graph = None
            self._gwarm = 0
            self._graph_key = key
        raw_model = getattr(self.model, "module", self.model)
        raw_model.graph_safe = True
        if self._graph is None:
            if self._gwarm == 0:
                self._gq = im_q.clone()
                self._gk = im_k.clone()
                self._gstream = torch.cuda.Stream()
            else:
                self._gq.copy_(im_q)
                self._gk.copy_(im_k)
            if self._gwarm < 3:
                # warmup on the side stream that will also host the capture
                # (AccumulateGrad nodes pin the stream they were created
                # on; capturing on a different stream trips cross-stream
                # event syncs) — also runs the MIOpen finds and builds the
                # EMA chunk tables at stable addresses
                self.optimizer.zero_grad(set_to_none=True)
                s = self._gstream
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    out = self._inner_step(self._gq, self._gk)
                torch.cuda.current_stream().wait_stream(s)
                self.optimizer.step()
                self._gwarm += 1
                return out
            self.optimizer.zero_grad(set_to_none=True)
            try:
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g, stream=self._gstream):
                    self._gout = self._inner_step(self._gq, self._gk)
            except Exception as e:  # capture executes nothing: state is
                # untouched on device — fall back to eager permanently
                # rather than killing the run
                import warnings
                warnings.warn(f"hipGraph capture failed ({e!r}); "
                              "continuing eager")
                self.use_graph = False
                raw_model._pending_keys = None
                raw_model.graph_safe = False
                torch.cuda.synchronize()
                return self.step(im_q, im_k)
            self._graph = g
            # capture itself executes nothing: replay once to take the step
            g.replay()
            self.optimizer.step()
            return self._gout
        self._gq.copy_(im_q)
        self._gk.copy_(im_k)
        self._graph.replay()
        self.optimizer.step()
        return self._gout

    # -- optional per-phase GPU timing (HIP events) ------------
    def enable_timing(self, on: bool = True):
        self._timing = on and self.is_cuda

    def _mark(self, name: str):
        if self._timing and self._phase_events is not None:
            ev = torch.cuda.Event(enable_timing=True)
            ev.record()
            self._phase_events.append((name, ev))

    def phase_times_ms(self):
        """Call after a synchronize: ms per phase of the last step."""
        if not self._phase_events:
            return {}
        out = {}
        evs = self._phase_events
        for (name, e0), (_, e1) in zip(evs[:-1], evs[1:]):
            out[name] = e0.elapsed_time(e1)
        return out


def train_epoch(train_loader, stepper: MoCoStepper, epoch: int, args,
                metrics=None):
    """One pretraining epoch (reference train(), main_moco.py:290-335)."""
    batch_time = AverageMeter("Time", ":6.3f")
    data_time = AverageMeter("Data", ":6.3f")
    losses = AverageMeter("Loss", ":.4e")
    top1 = AverageMeter("Acc@1", ":6.2f")
    top5 = AverageMeter("Acc@5", ":6.2f")
    progress = ProgressMeter(len(train_loader),
                             [batch_time, data_time, losses, top1, top5],
                             prefix="Epoch: [{}]".format(epoch))

    stepper.model.train()
    total_images = 0
    # meter reads (.item()) force a device sync; under the hipGraph
    # stepper that would serialize every replay, so accumulate the window
    # sums ON DEVICE and sync only at the print cadence.  Meter `val`
    # becomes the window mean rather than the single last batch — the
    # running averages are identical to the reference's.
    win = {"loss": None, "acc1": None, "acc5": None, "n": 0, "bs": 0}
    end = time.time()
    for i, (raw, _) in enumerate(train_loader):
        data_time.update(time.time() - end)

        loss, output, target = stepper.step_from_raw(raw)

        acc1, acc5 = accuracy(output.detach(), target, topk=(1, 5))
        vals = (loss.detach().float(), acc1[0].detach(), acc5[0].detach())
        if win["n"] == 0:
            win["loss"], win["acc1"], win["acc5"] = \
                [v.clone() for v in vals]
        else:
            win["loss"] += vals[0]
            win["acc1"] += vals[1]
            win["acc5"] += vals[2]
        win["n"] += 1
        win["bs"] = raw.size(0)

        batch_time.update(time.time() - end)
        end = time.time()
        total_images += raw.size(0)

        if i % args.print_freq == 0:
            n = win["n"]
            bs = raw.size(0)
            losses.update(win["loss"].item() / n, bs * n)
            top1.update(win["acc1"].item() / n, bs * n)
            top5.update(win["acc5"].item() / n, bs * n)
            win["n"] = 0
            progress.display(i)
            if metrics is not None:
                metrics.log(epoch=epoch, step=i, loss=round(losses.val, 5),
                            acc1=round(top1.val, 3), acc5=round(top5.val, 3),
                            img_s=round(raw.size(0) /
                                        max(batch_time.val, 1e-9), 2),
                            data_s=round(data_time.val, 4))
    if win["n"]:
        n = win["n"]
        losses.update(win["loss"].item() / n, win["bs"] * n)
        top1.update(win["acc1"].item() / n, win["bs"] * n)
        top5.update(win["acc5"].item() / n, win["bs"] * n)
    if metrics is not None:
        metrics.log(epoch=epoch, summary=True, loss_avg=round(losses.avg, 5),
                    acc1_avg=round(top1.avg, 3),
                    img_s_avg=round(total_images /
                                    max(batch_time.sum, 1e-9), 2))
    return losses.avg
