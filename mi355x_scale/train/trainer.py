"""Trainer: the DDP training loop (Lightning-free).

Covers the pl.Trainer flag set the reference drives
(``deep_learning/2.distributed-data-loading-petastorm.py:395-410``):
``strategy="ddp"/"auto"``, ``max_epochs``, ``limit_train_batches``,
``val_check_interval``, ``num_sanity_val_steps=0``, ``limit_val_batches``,
``reload_dataloaders_every_n_epochs=1``, ``use_distributed_sampler=False``
(sharding lives in the reader), ``enable_checkpointing``,
``default_root_dir``, plus a logger.

MI355X-first choices:
  * DDP over RCCL/xGMI with ``gradient_as_bucket_view`` and a bucket size
    tuned for the 7×153 GB/s point-to-point links (``bucket_cap_mb``).
  * bf16 autocast compute (BASELINE dtype), fp32 master weights.
  * epoch length imposed externally on an infinite reader — exactly the
    contract the reference relies on (``:218-220,387-388``), so ranks
    always run the same number of steps and collectives never mismatch.
"""
from __future__ import annotations

import time
from typing import Optional

import torch
import torch.distributed as dist
from torch.nn.parallel import DistributedDataParallel as DDP

from ..parallel.comm import DistContext, init_distributed, barrier
from .module import DataModule, ScaleModule
from .checkpoint import CheckpointManager


class Trainer:
    def __init__(
        self,
        accelerator: str = "auto",
        strategy: str = "auto",
        devices: int = 1,
        num_nodes: int = 1,
        max_epochs: int = 1,
        limit_train_batches: Optional[int] = None,
        limit_val_batches: Optional[int] = None,
        val_check_interval: Optional[int] = None,
        num_sanity_val_steps: int = 0,
        reload_dataloaders_every_n_epochs: int = 1,
        use_distributed_sampler: bool = False,
        enable_checkpointing: bool = True,
        default_root_dir: str = "./checkpoints",
        logger=None,
        precision: str = "bf16-mixed",
        bucket_cap_mb: int = 32,
        log_every_n_steps: int = 50,
        use_hipgraph: str = "auto",
        resume_from: str = None,
    ):
        self.strategy = strategy
        self.max_epochs = max_epochs
        self.limit_train_batches = limit_train_batches
        self.limit_val_batches = limit_val_batches
        self.val_check_interval = val_check_interval
        self.num_sanity_val_steps = num_sanity_val_steps
        self.reload_dataloaders_every_n_epochs = reload_dataloaders_every_n_epochs
        self.enable_checkpointing = enable_checkpointing
        self.default_root_dir = default_root_dir
        self.logger = logger
        self.precision = precision
        self.bucket_cap_mb = bucket_cap_mb
        self.log_every_n_steps = log_every_n_steps
        # "auto": capture the train step as a hipGraph on GPU (single
        # replay per step; DP becomes one flat RCCL all-reduce between
        # replays). "off"/False disables (eager DDP path).
        self.use_hipgraph = use_hipgraph
        self.resume_from = resume_from
        self.ctx: Optional[DistContext] = None
        self.checkpoint_callback = None
        self._sync_accum = {}

    # -- module.log plumbing ------------------------------------------------
    def _on_module_log(self, key: str, value: float, sync_dist: bool) -> None:
        self._sync_accum.setdefault(key, []).append((value, sync_dist))

    def _flush_logs(self, step: int) -> dict:
        out = {}
        for key, vals in self._sync_accum.items():
            mean = sum(v for v, _ in vals) / len(vals)
            if any(s for _, s in vals) and dist.is_initialized():
                # tensor device follows the BACKEND (gloo reduces on host
                # even when compute is on GPU)
                t = torch.tensor([mean], device=self.ctx.device
                                 if dist.get_backend() == "nccl" else "cpu")
                dist.all_reduce(t, op=dist.ReduceOp.SUM)
                mean = (t / dist.get_world_size()).item()
            out[key] = mean
            if self.logger is not None and self.ctx.is_main:
                self.logger.log_metric(key, mean, step)
        self._sync_accum.clear()
        return out

    # -- main entry ----------------------------------------------------------
    def fit(self, model: ScaleModule, datamodule: DataModule):
        self.ctx = init_distributed()
        ctx = self.ctx
        device = ctx.device
        model.trainer = self
        model.to(device)
        if getattr(model, "channels_last", False) and device.type == "cuda":
            model.to(memory_format=torch.channels_last)

        graph_on = (device.type == "cuda"
                    and self.use_hipgraph in ("auto", "on", True))
        use_ddp = not graph_on and ((self.strategy == "ddp") or (
            self.strategy == "auto" and ctx.world_size > 1
        ))
        wrapped = None
        if use_ddp and ctx.world_size > 1:
            kwargs = dict(
                bucket_cap_mb=self.bucket_cap_mb,
                gradient_as_bucket_view=True,
            )
            if device.type == "cuda":
                kwargs["device_ids"] = [device.index]
            # DDP wraps a shim whose forward IS training_step, so gradient
            # bucket hooks fire on the training forward pass.
            wrapped = DDP(_TrainStepShim(model), **kwargs)

        if graph_on:
            # fused flat Adam: one HIP kernel per step; its flat grad
            # buffer is the graphed step's single all-reduce target.
            # Hyperparameters come from the module's own
            # configure_optimizers() when it builds an Adam — graph mode
            # swaps the implementation, not the optimizer settings.
            from .flat_adam import FlatAdam
            hp = {"lr": getattr(model, "lr", 1e-3),
                  "betas": (0.9, 0.999), "eps": 1e-8, "weight_decay": 0.0}
            try:
                configured = model.configure_optimizers()
            except Exception:
                configured = None
            if configured is not None and getattr(
                    configured, "param_groups", None):
                g0 = configured.param_groups[0]
                for k in hp:
                    if k in g0:
                        hp[k] = g0[k]
                if not isinstance(configured, torch.optim.Adam):
                    import warnings
                    warnings.warn(
                        "graph mode replaces the configured "
                        f"{type(configured).__name__} with FlatAdam "
                        "(Adam math); pass use_hipgraph='off' to keep it")
                del configured
            optimizer = FlatAdam(model.parameters(),
                                 bf16_params=device.type == "cuda", **hp)
        else:
            optimizer = model.configure_optimizers()
        start_epoch = 0
        if self.resume_from:
            state = CheckpointManager.load(self.resume_from, model,
                                           optimizer,
                                           map_location=device)
            start_epoch = state["epoch"] + 1
            model.global_step = state["step"]
        if self.enable_checkpointing:
            self.checkpoint_callback = CheckpointManager(
                self.default_root_dir, monitor="val_loss", rank=ctx.rank)

        datamodule.setup("fit")
        amp_dtype = (torch.bfloat16
                     if self.precision.startswith("bf16") else None)
        autocast_on = amp_dtype is not None and device.type == "cuda"

        graphed = None
        step = model.global_step
        for epoch in range(start_epoch, self.max_epochs):
            model.current_epoch = epoch
            (wrapped or model).train()
            loader = datamodule.train_dataloader()
            t_epoch = time.time()
            n_batches = 0
            it = iter(loader)
            limit = self.limit_train_batches or float("inf")
            while n_batches < limit:
                try:
                    batch = next(it)
                except StopIteration:
                    break
                batch = _move(batch, device)
                if graph_on and graphed is None:
                    from .graphstep import GraphedTrainStep
                    graphed = GraphedTrainStep(
                        model, optimizer, batch,
                        autocast_dtype=amp_dtype if autocast_on else None,
                        world_size=ctx.world_size)
                if graphed is not None:
                    loss = graphed.step(batch)
                    if (step + 1) % self.log_every_n_steps == 0:
                        model.log("train_loss", float(loss.item()))
                else:
                    with torch.autocast(device_type="cuda",
                                        dtype=amp_dtype,
                                        enabled=autocast_on):
                        if wrapped is not None:
                            loss = wrapped(batch, n_batches)
                        else:
                            loss = model.training_step(batch, n_batches)
                    optimizer.zero_grad(set_to_none=True)
                    loss.backward()
                    optimizer.step()
                n_batches += 1
                step += 1
                model.global_step = step
                if step % self.log_every_n_steps == 0:
                    self._flush_logs(step)
                if (self.val_check_interval
                        and step % self.val_check_interval == 0):
                    self._run_validation(wrapped, model, datamodule, device,
                                         amp_dtype, autocast_on, step)
                    (wrapped or model).train()
            if hasattr(loader, "close"):
                loader.close()
            metrics = self._flush_logs(step)
            val_metrics = self._run_validation(
                wrapped, model, datamodule, device, amp_dtype, autocast_on,
                step)
            metrics.update(val_metrics)
            if self.checkpoint_callback is not None:
                self.checkpoint_callback.save(model, optimizer, epoch, step,
                                              metrics)
            if ctx.is_main:
                dt = time.time() - t_epoch
                print(f"[epoch {epoch}] {n_batches} steps in {dt:.1f}s "
                      f"metrics={metrics}")
            barrier()
        datamodule.teardown("fit")
        return model

    @torch.no_grad()
    def _broadcast_buffers(self, model) -> None:
        """Rank-0 → all broadcast of module buffers (BN running stats).

        DDP broadcasts buffers every forward; the graphed step does not
        touch them cross-rank, so running_mean/var drift per rank (train
        forwards use batch stats — training is unaffected) and eval
        metrics would disagree. One coalesced broadcast before each
        validation pass restores DDP's eval semantics.
        """
        if not (dist.is_initialized() and dist.get_world_size() > 1):
            return
        bufs = [b for b in model.buffers() if b.numel()]
        if not bufs:
            return
        # coalesce per dtype to avoid one collective per BN layer
        by_dtype: dict = {}
        for b in bufs:
            by_dtype.setdefault(b.dtype, []).append(b)
        on_host = dist.get_backend() == "gloo"
        for dt, group in by_dtype.items():
            flat = torch.cat([b.detach().reshape(-1) for b in group])
            if on_host:
                flat = flat.cpu()
            dist.broadcast(flat, src=0)
            flat = flat.to(group[0].device)
            off = 0
            for b in group:
                b.copy_(flat[off:off + b.numel()].view_as(b))
                off += b.numel()

    @torch.no_grad()
    def _run_validation(self, wrapped, model, datamodule, device, amp_dtype,
                        autocast_on, step) -> dict:
        loader = datamodule.val_dataloader()
        if loader is None:
            return {}
        self._broadcast_buffers(model)
        (wrapped or model).eval()
        limit = self.limit_val_batches or float("inf")
        n = 0
        it = iter(loader)
        while n < limit:
            try:
                batch = next(it)
            except StopIteration:
                break
            batch = _move(batch, device)
            with torch.autocast(device_type="cuda", dtype=amp_dtype,
                                enabled=autocast_on):
                model.validation_step(batch, n)
            n += 1
        if hasattr(loader, "close"):
            loader.close()
        return self._flush_logs(step)


class _TrainStepShim(torch.nn.Module):
    """forward(batch, idx) == module.training_step(batch, idx) — lets DDP
    hook the training forward pass while owning the module's parameters."""

    def __init__(self, module: ScaleModule):
        super().__init__()
        self.module = module

    def forward(self, batch, batch_idx):
        return self.module.training_step(batch, batch_idx)


def _move(batch, device):
    if torch.is_tensor(batch):
        return batch.to(device, non_blocking=True)
    if isinstance(batch, dict):
        return {k: _move(v, device) for k, v in batch.items()}
    if isinstance(batch, (list, tuple)):
        return type(batch)(_move(v, device) for v in batch)
    return batch
