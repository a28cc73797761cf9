"""PipelineEngine: 1F1B microbatch training over RCCL/xGMI.

The native core that replaces DeepSpeed's PipelineEngine as the reference
drives it — ``engine.train_batch(data_iter)`` runs ``gradient_accumulation_
steps`` microbatches through a fill-drain 1F1B schedule, exchanges
activations/gradients with neighbour stages, all-reduces DP gradients,
clips, steps AdamW + LR, and returns the mean loss on every rank
(trainer_base_ds_mp.py:354).

Schedule (per stage ``s`` of ``P``, ``M`` microbatches):
  warmup   = min(P - 1 - s, M) forwards,
  steady   = M - warmup 1F1B pairs (forward + oldest-outstanding backward),
  cooldown = warmup backwards.
Only the first and last stages touch the data iterator (README.md:64-67 —
the property that enables the reference's placeholder-dataset trick; here
middle stages simply do not need a dataset at all).

Precision: bf16 activations/params, fp32 gradient accumulation (hooks into
the optimizer's flat fp32 buffer), fp32 master AdamW.  fp16 with dynamic
loss scaling is supported for reference parity (conf/...yaml:137-143).
"""

from __future__ import annotations

import logging
import os
import threading
import time
from collections import deque
from typing import Iterator, Optional

import torch
import torch.distributed as dist

from .config import TrainConfig, torch_dtype
from .optim import MixedPrecisionAdamW, WarmupDecayLR
from .p2p import PipeP2P
from .pipeline_module import PipelineModule
from .topology import ProcessGrid
from .utils.timers import DeviceTimers

logger = logging.getLogger(__name__)


class DynamicLossScaler:
    """fp16 dynamic loss scaling (init 2^12, window 1000, hysteresis 2,
    min 1 — conf/...yaml:137-143)."""

    def __init__(self, init_scale=2.0**12, scale_window=1000, hysteresis=2, min_scale=1.0):
        self.scale = init_scale
        self.scale_window = scale_window
        self.hysteresis = hysteresis
        self.min_scale = min_scale
        self._good_steps = 0
        self._hyst = hysteresis

    def update(self, found_inf: bool) -> None:
        if found_inf:
            self._hyst -= 1
            if self._hyst <= 0:
                self.scale = max(self.scale / 2.0, self.min_scale)
                self._hyst = self.hysteresis
            self._good_steps = 0
        else:
            self._good_steps += 1
            if self._good_steps >= self.scale_window:
                self.scale *= 2.0
                self._good_steps = 0


class _Watchdog:
    """Deadlock watchdog (the reference's only backstop is
    the 7200s NCCL timeout).  While armed, if a train_batch exceeds the
    timeout the thread dumps this rank's schedule position to stderr so a
    hung pipeline is diagnosable per rank instead of dying silently.
    Enable with LPP_WATCHDOG_S=<seconds> (or TrainConfig.watchdog_timeout_s)."""

    def __init__(self, engine: "PipelineEngine", timeout_s: float):
        self.engine = engine
        self.timeout_s = timeout_s
        self._deadline = None
        self._fired = False
        self._stop = False
        self._cv = threading.Condition()
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    def arm(self):
        with self._cv:
            self._deadline = time.time() + self.timeout_s
            self._fired = False
            self._cv.notify()

    def disarm(self):
        with self._cv:
            self._deadline = None

    def stop(self):
        with self._cv:
            self._stop = True
            self._deadline = None
            self._cv.notify()
        self._thread.join(timeout=2.0)

    def _run(self):
        while True:
            with self._cv:
                if self._stop:
                    return
                if self._deadline is None:
                    self._cv.wait(timeout=1.0)
                    continue
                remaining = self._deadline - time.time()
            if remaining <= 0:
                if not self._fired:
                    self._fired = True
                    e = self.engine
                    logger.error(
                        "[watchdog] rank %d stage %d: train_batch stuck >%.0fs at %r "
                        "(step %d, %d pending microbatches)",
                        e.grid.rank, e.grid.stage_id, self.timeout_s,
                        e.schedule_position, e.global_steps, len(e._pending_dbg),
                    )
                time.sleep(5.0)
            else:
                time.sleep(min(remaining, 5.0))


class PipelineEngine:
    def __init__(self, module: PipelineModule, config: TrainConfig, grid: ProcessGrid,
                 device: Optional[torch.device] = None):
        self.module = module
        self.config = config
        self.grid = grid
        self.dtype = torch_dtype(config.dtype)
        if device is None:
            device = torch.device("cuda", torch.cuda.current_device()) \
                if torch.cuda.is_available() else torch.device("cpu")
        self.device = device

        self.module.to(device)
        if self.dtype != torch.float32:
            self.module.to(self.dtype)

        self.micro_batch_size = config.micro_batch_size
        self.micro_batches = config.gradient_accumulation_steps
        self.seq_len = config.seq_len
        hidden = config.model.hidden_size
        act_shape = (self.micro_batch_size, self.seq_len, hidden)
        comm_dtype = self.dtype if self.dtype != torch.float32 else torch.float32
        self.p2p = PipeP2P(grid, act_shape, comm_dtype, device)

        opt_cfg = config.optimizer
        self.zero1 = bool(getattr(config, "zero_stage", 0) == 1 and grid.dp_degree > 1)
        self.optimizer = MixedPrecisionAdamW(
            self.module.parameters(),
            lr=opt_cfg.lr,
            betas=tuple(opt_cfg.betas),
            eps=opt_cfg.eps,
            weight_decay=opt_cfg.weight_decay,
            shard_group=grid.dp_group if self.zero1 else None,
            shard_rank=grid.dp_id if self.zero1 else 0,
            shard_world=grid.dp_degree if self.zero1 else 1,
        )
        warmup = opt_cfg.warmup_steps or max(1, int(opt_cfg.warmup_proportion * opt_cfg.total_num_steps))
        self.lr_scheduler = WarmupDecayLR(
            self.optimizer, warmup, opt_cfg.total_num_steps, opt_cfg.lr
        )
        self.loss_scaler = DynamicLossScaler() if config.dtype == "fp16" else None
        self.global_steps = 0
        self.skipped_steps = 0
        self._step_time = 0.0
        # observability — device-true section timers
        # (hipEvent pairs on GPU; round-1's host timers bracketed enqueue
        # only and misattributed all async work to "backward")
        self.device_timers = DeviceTimers(device)
        self.schedule_position = "idle"
        self._pending_dbg = ()

        # ---- DP gradient-bucket overlap plan (reference overlap_comm: True,
        # conf/...yaml:154-159).  Buckets are built over the contiguous flat
        # fp32 grad buffer in REVERSE parameter order — the order the final
        # backward retires them — so each bucket's all-reduce can launch as
        # soon as its last gradient lands, overlapping the remaining
        # backward compute on the comm stream.
        self._dp_overlap = bool(
            config.overlap_allreduce and grid.dp_degree > 1 and not self.zero1
        )
        self._buckets = []           # dicts: start,end,count,remaining,launched
        self._bucket_of = {}         # id(param) -> bucket index
        self._ar_handles = []
        self._grad_finalizing = False
        if self._dp_overlap:
            bucket_elems = max(1, config.allreduce_bucket_mb * 1024 * 1024 // 4)
            offs, off = {}, 0
            for p in self.optimizer.params:
                offs[id(p)] = off
                off += p.numel()
            cur, cur_elems = [], 0
            for p in reversed(self.optimizer.params):
                cur.append(p)
                cur_elems += p.numel()
                if cur_elems >= bucket_elems:
                    self._push_bucket(cur, offs)
                    cur, cur_elems = [], 0
            if cur:
                self._push_bucket(cur, offs)
            self.optimizer.on_accumulate = self._on_grad_accumulated
            # The fused-wgrad linear path accumulates into main_grad
            # directly (no p.grad, so the autograd post-accumulate hook
            # never fires for those weights) — it notifies through this
            # per-param stamp instead (ops/linear.py backward).
            for p in self.optimizer.params:
                p._on_accumulate = self._on_grad_accumulated
        # Initialize every communicator at a controlled point (RCCL comm
        # creation is collective; doing it lazily inside the overlapped
        # schedule would interleave group inits across ranks).
        self.p2p.warmup_channels()
        if dist.is_initialized():
            warm = torch.zeros(1, device=device)
            if grid.dp_degree > 1:
                dist.all_reduce(warm, group=grid.dp_group)
            if grid.num_stages > 1:
                dist.all_reduce(warm, group=grid.pipe_group)

        try:
            wd_s = float(os.environ.get("LPP_WATCHDOG_S", "") or
                         getattr(config, "watchdog_timeout_s", 0) or 0)
        except ValueError:
            logger.warning("unparseable LPP_WATCHDOG_S=%r; watchdog from config",
                           os.environ.get("LPP_WATCHDOG_S"))
            wd_s = float(getattr(config, "watchdog_timeout_s", 0) or 0)
        self.watchdog = _Watchdog(self, wd_s) if wd_s > 0 else None

    # ------------------------------------------------------------------
    def _push_bucket(self, params, offs) -> None:
        idx = len(self._buckets)
        start = min(offs[id(p)] for p in params)
        end = max(offs[id(p)] + p.numel() for p in params)
        self._buckets.append(
            {"start": start, "end": end, "count": len(params),
             "remaining": len(params), "launched": False}
        )
        for p in params:
            self._bucket_of[id(p)] = idx

    def _on_grad_accumulated(self, p) -> None:
        """Optimizer post-accumulate observer: during the FINAL microbatch's
        backward, launch a bucket's DP all-reduce the moment its last
        gradient is final (runs on the autograd thread, async_op — the
        collective rides the comm stream while backward keeps computing)."""
        if not self._grad_finalizing:
            return
        bi = self._bucket_of.get(id(p))
        if bi is None:  # unknown tensor identity (re-wrapped save) — boundary
            return      # fallback covers its bucket
        b = self._buckets[bi]
        b["remaining"] -= 1
        if b["remaining"] == 0 and not b["launched"]:
            b["launched"] = True
            chunk = self.optimizer.flat_grads.narrow(0, b["start"], b["end"] - b["start"])
            self._ar_handles.append(
                dist.all_reduce(chunk, op=dist.ReduceOp.SUM, group=self.grid.dp_group,
                                async_op=True)
            )

    @property
    def timers(self) -> dict:
        base = {k: 0.0 for k in ("forward", "backward", "p2p", "allreduce", "optimizer")}
        base.update(self.device_timers.totals_nosync())
        return base

    @property
    def is_first_stage(self) -> bool:
        return self.grid.is_first_stage()

    @property
    def is_last_stage(self) -> bool:
        return self.grid.is_last_stage()

    def _next_batch(self, data_iter: Iterator):
        batch = next(data_iter)
        out = {}
        for k, v in batch.items():
            if torch.is_tensor(v):
                out[k] = v.to(self.device, non_blocking=True)
            else:
                out[k] = v
        return out

    # -- microbatch fwd/bwd ------------------------------------------------
    def _forward_step(self, recv_act: Optional[torch.Tensor], data_iter):
        """Returns (input_tensor_for_grad, backward_handle, loss_detached)."""
        if self.is_first_stage:
            batch = self._next_batch(data_iter)
            x = batch["input_ids"]
            inp = None
            first_batch = batch
        else:
            x = recv_act
            x.requires_grad_(True)
            inp = x
            first_batch = None

        out = self.module(x)

        if self.is_last_stage:
            if self.is_first_stage:
                labels = first_batch["labels"]
            else:
                labels = self._next_batch(data_iter)["labels"]
            loss = self.module.loss_fn(out, labels)
            scaled = loss / self.micro_batches
            if self.loss_scaler is not None:
                scaled = scaled * self.loss_scaler.scale
            return inp, scaled, loss.detach()
        return inp, out, None

    def _backward_step(self, inp, handle, recv_grad) -> Optional[torch.Tensor]:
        if self.is_last_stage:
            handle.backward()
        else:
            torch.autograd.backward(handle, grad_tensors=recv_grad)
        if inp is None:
            return None
        g = inp.grad
        inp.grad = None
        return g

    # -- the 1F1B schedule --------------------------------------------------
    def train_batch(self, data_iter: Iterator) -> torch.Tensor:
        """One optimizer step == ``micro_batches`` microbatches, 1F1B.

        With ``p2p_overlap`` (default) every network receive is PRE-POSTED
        one use ahead on its dedicated channel (PipeP2P docstring) and
        waited only at the point of use, so the xGMI transfer of the next
        microbatch's activation rides under this microbatch's backward, and
        sends are fire-and-forget (drained at the step boundary).  With the
        flag off, each receive is posted-and-waited at its use site and
        sends are waited immediately — the strictly serial reference
        ordering, kept as the A/B baseline."""
        t0 = time.time()
        if self.watchdog:
            self.watchdog.arm()
        self.module.train()
        timers = self.device_timers
        M = self.micro_batches
        P = self.grid.num_stages
        s = self.grid.stage_id
        warmup = min(P - 1 - s, M)
        remaining = M - warmup
        overlap = bool(self.config.p2p_overlap)

        pending = deque()  # (input_tensor, backward_handle)
        losses = []
        sends = []  # in-flight sends (overlap mode)

        first, last = self.is_first_stage, self.is_last_stage
        acts_needed = 0 if first else M     # network activation recvs
        grads_needed = 0 if last else M     # network gradient recvs
        acts_posted = grads_posted = 0
        pend_act = pend_grad = None
        backwards_done = 0

        def post_act():
            nonlocal acts_posted
            acts_posted += 1
            return self.p2p.irecv_forward()

        def post_grad():
            nonlocal grads_posted
            grads_posted += 1
            return self.p2p.irecv_backward()

        def do_send(pending_send):
            # Sends are fire-and-forget in BOTH modes (drained at the step
            # boundary): waiting a send in-line would rendezvous against the
            # peer's recv-post order and can deadlock on gloo.  The overlap
            # flag gates only the pre-posting of receives.
            if pending_send is not None:
                sends.append(pending_send)

        if overlap and acts_posted < acts_needed:
            pend_act = post_act()

        def take_act():
            nonlocal pend_act
            if first:
                return None
            with timers.section("p2p"):
                p = pend_act if pend_act is not None else post_act()
                x = p.wait()
            pend_act = post_act() if (overlap and acts_posted < acts_needed) else None
            return x

        def take_grad():
            nonlocal pend_grad
            if last:
                return None
            with timers.section("p2p"):
                p = pend_grad if pend_grad is not None else post_grad()
                g = p.wait()
            pend_grad = post_grad() if (overlap and grads_posted < grads_needed) else None
            return g

        def run_backward(recv_grad):
            nonlocal backwards_done
            b_inp, b_handle = pending.popleft()
            if backwards_done + 1 == M and self._dp_overlap and dist.is_initialized():
                self._grad_finalizing = True
            with timers.section("backward"):
                g = self._backward_step(b_inp, b_handle, recv_grad)
            backwards_done += 1
            self._grad_finalizing = False
            if g is not None:
                with timers.section("p2p"):
                    do_send(self.p2p.isend_backward(g))

        # ---- warmup forwards
        for wi in range(warmup):
            self.schedule_position = f"warmup fwd {wi + 1}/{warmup}"
            x = take_act()
            with timers.section("forward"):
                inp, handle, loss = self._forward_step(x, data_iter)
            if loss is not None:
                losses.append(loss)
            if not last:
                with timers.section("p2p"):
                    do_send(self.p2p.isend_forward(handle))
            pending.append((inp, handle))
            self._pending_dbg = tuple(range(len(pending)))

        if overlap and grads_posted < grads_needed:
            pend_grad = post_grad()

        # ---- steady 1F1B
        for i in range(remaining):
            self.schedule_position = f"steady 1F1B {i + 1}/{remaining}"
            x = take_act()
            with timers.section("forward"):
                inp, handle, loss = self._forward_step(x, data_iter)
            if loss is not None:
                losses.append(loss)
            if not last:
                with timers.section("p2p"):
                    do_send(self.p2p.isend_forward(handle))
            pending.append((inp, handle))
            run_backward(take_grad())

        # ---- cooldown backwards
        for ci in range(warmup):
            self.schedule_position = f"cooldown bwd {ci + 1}/{warmup}"
            run_backward(take_grad())

        # drain outstanding sends before touching the buffers' storages
        with timers.section("p2p"):
            for sp in sends:
                sp.wait()

        # ---- boundary: DP all-reduce, clip, step
        self.schedule_position = "optimizer"
        self._optimizer_step()

        loss_out = self._reduce_loss(losses)
        self._step_time = time.time() - t0
        self.schedule_position = "idle"
        if self.watchdog:
            self.watchdog.disarm()
        return loss_out

    @torch.no_grad()
    def eval_batch(self, data_iter: Iterator, micro_batches: Optional[int] = None) -> torch.Tensor:
        """Forward-only pipeline pass over ``micro_batches`` microbatches;
        returns the mean loss on every rank.  Fills the gap the reference
        leaves open (its config names an evaluator class that does not
        exist and train() never calls one — SURVEY.md §2.9)."""
        self.module.eval()
        M = micro_batches or self.micro_batches
        losses = []
        for _ in range(M):
            x = self.p2p.recv_forward()
            if self.is_first_stage:
                batch = self._next_batch(data_iter)
                x = batch["input_ids"]
            out = self.module(x)
            self.p2p.send_forward(out if not self.is_last_stage else None)
            if self.is_last_stage:
                if self.is_first_stage:
                    labels = batch["labels"]
                else:
                    labels = self._next_batch(data_iter)["labels"]
                losses.append(self.module.loss_fn(out, labels).detach())
        self.module.train()
        return self._reduce_loss(losses)

    # ------------------------------------------------------------------
    def _reduce_loss(self, losses) -> torch.Tensor:
        if self.is_last_stage:
            loss = torch.stack(losses).mean() if losses else torch.zeros((), device=self.device)
            loss = loss.float()
            if self.grid.dp_degree > 1 and dist.is_initialized():
                # SUM+div (AVG is NCCL-only; gloo path must work for CPU tests)
                dist.all_reduce(loss, op=dist.ReduceOp.SUM, group=self.grid.dp_group)
                loss /= self.grid.dp_degree
        else:
            loss = torch.zeros((), dtype=torch.float32, device=self.device)
        if self.grid.num_stages > 1 and dist.is_initialized():
            src = self.grid.stage_to_rank(self.grid.num_stages - 1)
            dist.broadcast(loss, src=src, group=self.grid.pipe_group)
        return loss

    def _allreduce_gradients(self) -> None:
        if self.grid.dp_degree <= 1 or not dist.is_initialized():
            return
        flat = self.optimizer.flat_grads
        if self.zero1:
            # ZeRO-1: each DP rank only needs ITS shard of the summed grads.
            opt = self.optimizer
            if dist.get_backend(self.grid.dp_group) == "nccl":
                shard = torch.empty_like(opt.grad_shard)
                dist.reduce_scatter_tensor(shard, flat, op=dist.ReduceOp.SUM,
                                           group=self.grid.dp_group)
                opt.grad_shard.copy_(shard)
            else:  # gloo: no reduce_scatter; all-reduce then slice locally
                dist.all_reduce(flat, op=dist.ReduceOp.SUM, group=self.grid.dp_group)
            opt.grad_shard.div_(self.grid.dp_degree)
            return
        if self._dp_overlap:
            # buckets not retired by the final backward's hooks (shouldn't
            # happen, but a param skipped in the graph would strand one)
            for b in self._buckets:
                if not b["launched"]:
                    b["launched"] = True
                    chunk = flat.narrow(0, b["start"], b["end"] - b["start"])
                    self._ar_handles.append(
                        dist.all_reduce(chunk, op=dist.ReduceOp.SUM,
                                        group=self.grid.dp_group, async_op=True)
                    )
            for h in self._ar_handles:
                h.wait()
            self._ar_handles.clear()
            for b in self._buckets:
                b["remaining"] = b["count"]
                b["launched"] = False
        else:
            bucket_elems = max(1, self.config.allreduce_bucket_mb * 1024 * 1024 // 4)
            handles = []
            for off in range(0, flat.numel(), bucket_elems):
                chunk = flat.narrow(0, off, min(bucket_elems, flat.numel() - off))
                handles.append(
                    dist.all_reduce(chunk, op=dist.ReduceOp.SUM, group=self.grid.dp_group,
                                    async_op=True)
                )
            for h in handles:
                h.wait()
        flat.div_(self.grid.dp_degree)

    def _optimizer_step(self) -> None:
        timers = self.device_timers
        with timers.section("allreduce"):
            self._allreduce_gradients()
        with timers.section("optimizer"):
            inv_scale = 1.0
            if self.loss_scaler is not None:
                inv_scale = 1.0 / self.loss_scaler.scale

            sq = self.optimizer.grad_sq_sum()
            if dist.is_initialized():
                if self.zero1:
                    # shards tile (dp x stage): sum across the whole world
                    dist.all_reduce(sq, op=dist.ReduceOp.SUM)
                elif self.grid.num_stages > 1:
                    dist.all_reduce(sq, op=dist.ReduceOp.SUM, group=self.grid.pipe_group)
            global_norm = (sq.float().sqrt() * inv_scale).item()

            if self.loss_scaler is not None:
                found_inf = not (global_norm == global_norm and global_norm != float("inf"))
                self.loss_scaler.update(found_inf)
                if found_inf:
                    self.optimizer.zero_grad()
                    self.skipped_steps += 1
                    self.global_steps += 1
                    return

            clip = self.config.optimizer.max_grad_norm
            coef = inv_scale
            if clip > 0 and global_norm > clip:
                coef *= clip / (global_norm + 1e-6)
            self.optimizer.step(grad_scale=coef)
            self.optimizer.zero_grad()
            self.lr_scheduler.step()
            self.global_steps += 1
            # weights changed: drop the cached W^T dgrad operands
            # (lazily rebuilt by the first backward of the next step)
            from .ops.linear import invalidate_weight_transposes

            invalidate_weight_transposes(self.module)

    # ------------------------------------------------------------------
    @property
    def last_step_time(self) -> float:
        return self._step_time

    def timer_summary(self, reset: bool = True) -> dict:
        """Device-true accumulated section times (seconds) since last reset.
        On GPU these are hipEvent pairs recorded on the compute stream —
        forward/backward are real kernel time, p2p/allreduce are the stalls
        the compute stream suffered waiting on the comm streams.  Folding
        synchronizes the device."""
        base = {k: 0.0 for k in ("forward", "backward", "p2p", "allreduce", "optimizer")}
        base.update(self.device_timers.summary(reset=reset))
        return base

    def get_lr(self) -> float:
        return self.optimizer.lr

    # checkpoint integration lives in lpp_amd.checkpoint
    def state_dict_local(self) -> dict:
        return {
            "optimizer": self.optimizer.state_dict(),
            "lr_scheduler": self.lr_scheduler.state_dict(),
            "global_steps": self.global_steps,
            "skipped_steps": self.skipped_steps,
            "loss_scale": self.loss_scaler.scale if self.loss_scaler else None,
        }

    def load_state_dict_local(self, sd: dict) -> None:
        self.optimizer.load_state_dict(sd["optimizer"])
        self.lr_scheduler.load_state_dict(sd["lr_scheduler"])
        self.global_steps = sd.get("global_steps", 0)
        self.skipped_steps = sd.get("skipped_steps", 0)
        if self.loss_scaler is not None and sd.get("loss_scale"):
            self.loss_scaler.scale = sd["loss_scale"]
