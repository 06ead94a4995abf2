# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""PyTorch model serving + the MLRun-style training interface.

Parity targets (reference): frameworks/pytorch/model_server.py:29
PyTorchModelServer; frameworks/pytorch/mlrun_interface.py (the Horovod
`_setup`/`_train` path :555-718) re-designed as apply_mlrun/train over
torch.distributed with RCCL — see mlrun_amd/parallel/ddp.py for the
gradient-bucketing engine.
"""

import os
import typing

import torch

from ..serving.v2_serving import V2ModelServer
from ..utils import logger


class PyTorchModelServer(V2ModelServer):
    """Serve a torch module: TorchScript file, state_dict + model class,
    or a live module instance."""

    def load(self):
        device = self.get_param(
            "device", "cuda:0" if torch.cuda.is_available() else "cpu")
        dtype = self.get_param("dtype", "bfloat16")
        self._device = torch.device(device)
        self._dtype = getattr(torch, dtype)
        if isinstance(self.model, torch.nn.Module):
            pass
        elif self.model_path:
            model_file, extra = self.get_model(".pt")
            model_class = self.get_param("model_class")
            if model_class is not None:
                kwargs = self.get_param("init_kwargs", {}) or {}
                module = model_class(**kwargs) if callable(model_class) \
                    else None
                state = torch.load(model_file, map_location="cpu",
                                   weights_only=True)
                module.load_state_dict(state)
                self.model = module
            else:
                self.model = torch.jit.load(model_file, map_location="cpu")
        else:
            raise ValueError(f"model {self.name}: no model or model_path")
        self.model = self.model.to(self._device, dtype=self._dtype).eval()

    @torch.inference_mode()
    def predict(self, request: dict):
        inputs = torch.as_tensor(request["inputs"]).to(self._device,
                                                       dtype=self._dtype)
        outputs = self.model(inputs)
        if isinstance(outputs, (tuple, list)):
            outputs = outputs[0]
        return outputs.float().cpu().tolist()


class TrainingCallback:
    """Epoch-hook protocol (reference frameworks/pytorch/callbacks
    base): subclass and override; train() drives the hooks."""

    def on_train_begin(self, interface):
        pass

    def on_epoch_end(self, interface, epoch: int, results: dict):
        pass

    def on_train_end(self, interface, history: dict):
        pass


class MLRunLoggingCallback(TrainingCallback):
    """Per-epoch child-iteration results + final model logging
    (reference mlrun_logging_callback.py:30) — rank-0 gated."""

    def __init__(self, model_key: str = "model", log_model: bool = True):
        self.model_key = model_key
        self.log_model = log_model

    def on_epoch_end(self, interface, epoch, results):
        if interface.context is not None and interface.rank == 0:
            for key, value in results.items():
                interface.context.log_result(f"epoch_{epoch}_{key}",
                                             value)

    def on_train_end(self, interface, history):
        if self.log_model:
            interface.log_model(
                self.model_key,
                metrics={k: v[-1] for k, v in history.items() if v})


class CheckpointCallback(TrainingCallback):
    """Save a model checkpoint artifact every N epochs (reference
    ModelHandler.save cadence; restartable-runs story)."""

    def __init__(self, every: int = 1, key: str = "checkpoint"):
        self.every = max(int(every), 1)
        self.key = key

    def on_epoch_end(self, interface, epoch, results):
        if (epoch + 1) % self.every == 0:
            interface.log_model(f"{self.key}-epoch-{epoch}")


class EarlyStoppingCallback(TrainingCallback):
    """Stop when a monitored metric fails to improve for `patience`
    epochs (reference callbacks; train() checks `should_stop`)."""

    def __init__(self, monitor: str = "loss", patience: int = 3,
                 mode: str = "min"):
        self.monitor = monitor
        self.patience = patience
        self.mode = mode
        self.best = None
        self.bad_epochs = 0
        self.should_stop = False

    def on_epoch_end(self, interface, epoch, results):
        value = results.get(self.monitor)
        if value is None:
            return
        better = self.best is None or (
            value < self.best if self.mode == "min" else
            value > self.best)
        if better:
            self.best = value
            self.bad_epochs = 0
        else:
            self.bad_epochs += 1
            if self.bad_epochs >= self.patience:
                self.should_stop = True


class MLRunTorchInterface:
    """Training-loop instrumentation: epoch loop + metric logging +
    distributed data parallel over RCCL (the Horovod-equivalent)."""

    def __init__(self, model: torch.nn.Module, context=None):
        self.model = model
        self.context = context
        self.rank = int(os.environ.get("RANK", "0"))
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self._ddp = None

    def setup_distributed(self, bucket_cap_mb: int = None):
        """Wrap the model for data-parallel training (bucketed bf16
        all-reduce over xGMI; parameters broadcast from rank 0)."""
        if self.world_size <= 1:
            return self.model
        from ..parallel.ddp import DistributedModel

        self._ddp = DistributedModel(self.model, bucket_cap_mb=bucket_cap_mb)
        return self._ddp

    def _shard_loader(self, loader):
        """Shard a DataLoader across ranks with a DistributedSampler
        (reference mlrun_interface.py:617-631 _add_distributed_sampler);
        non-DataLoader iterables are returned unchanged (the caller
        already shards, e.g. synthetic per-rank batches)."""
        if self.world_size <= 1 or \
                not isinstance(loader, torch.utils.data.DataLoader) or \
                isinstance(loader.sampler,
                           torch.utils.data.distributed.DistributedSampler):
            return loader
        sampler = torch.utils.data.distributed.DistributedSampler(
            loader.dataset, num_replicas=self.world_size, rank=self.rank)
        return torch.utils.data.DataLoader(
            loader.dataset, batch_size=loader.batch_size,
            sampler=sampler, num_workers=loader.num_workers,
            collate_fn=loader.collate_fn, drop_last=loader.drop_last)

    def train(self, train_loader, loss_fn, optimizer, epochs: int = 1,
              validation_loader=None, metric_fns: list = None,
              scheduler=None, use_amp: bool = False,
              auto_log_model: bool = True, model_key: str = "model",
              callbacks: list = None):
        """The epoch loop (reference mlrun_interface.py:657 _train):
        auto DDP (if apply_mlrun set it up), per-rank shards via
        DistributedSampler, allreduce-mean epoch metrics, rank-0
        logging, callbacks (epoch hooks / checkpoints / early stop),
        and final model artifact logging."""
        model = self._ddp or self.model
        train_loader = self._shard_loader(train_loader)
        if validation_loader is not None:
            validation_loader = self._shard_loader(validation_loader)
        metric_fns = metric_fns or []
        callbacks = callbacks or []
        for callback in callbacks:
            callback.on_train_begin(self)
        history: typing.Dict[str, list] = {}
        for epoch in range(epochs):
            model.train()
            total_loss, steps = 0.0, 0
            for batch in train_loader:
                x, y = batch
                optimizer.zero_grad(set_to_none=True)
                out = model(x)
                loss = loss_fn(out, y)
                loss.backward()
                if self._ddp is not None:
                    self._ddp.finalize_backward()
                optimizer.step()
                total_loss += float(loss.detach())
                steps += 1
            if scheduler is not None:
                scheduler.step()
            epoch_loss = self._metric_average(total_loss / max(steps, 1))
            history.setdefault("loss", []).append(epoch_loss)
            results = {"loss": epoch_loss, "epoch": epoch}
            if validation_loader is not None:
                results.update(self.evaluate(validation_loader, loss_fn,
                                             metric_fns))
            if self.context is not None and self.rank == 0:
                self.context.log_results(results)
            logger.info("epoch done", **results)
            stop = False
            for callback in callbacks:
                callback.on_epoch_end(self, epoch, results)
                stop = stop or getattr(callback, "should_stop", False)
            if stop:
                logger.info("early stopping", epoch=epoch)
                break
        for callback in callbacks:
            callback.on_train_end(self, history)
        if auto_log_model and self.context is not None:
            self.log_model(model_key,
                           metrics={k: v[-1] for k, v in history.items()})
        return history

    @torch.no_grad()
    def evaluate(self, loader, loss_fn, metric_fns=None) -> dict:
        model = self._ddp or self.model
        model.eval()
        total, steps = 0.0, 0
        metric_totals = [0.0] * len(metric_fns or [])
        for x, y in loader:
            out = model(x)
            total += float(loss_fn(out, y))
            for i, fn in enumerate(metric_fns or []):
                metric_totals[i] += float(fn(out, y))
            steps += 1
        results = {"validation_loss":
                   self._metric_average(total / max(steps, 1))}
        for i, fn in enumerate(metric_fns or []):
            name = getattr(fn, "__name__", f"metric_{i}")
            results[name] = self._metric_average(
                metric_totals[i] / max(steps, 1))
        return results

    def _metric_average(self, value: float) -> float:
        """All-reduce-mean a scalar metric across ranks (reference
        mlrun_interface.py:860 _metric_average over hvd.allreduce)."""
        if self.world_size <= 1:
            return value
        import torch.distributed as dist

        if not dist.is_initialized():
            return value
        tensor = torch.tensor([value], dtype=torch.float64)
        if dist.get_backend() == "nccl":
            tensor = tensor.cuda()
        dist.all_reduce(tensor, op=dist.ReduceOp.SUM)
        return float(tensor.item()) / self.world_size

    def log_model(self, key="model", **kwargs):
        """Rank-0-only model artifact (reference is_logging_worker gate,
        execution.py:1040)."""
        if self.context is None or self.rank != 0:
            return None
        import io

        buf = io.BytesIO()
        torch.save(self.model.state_dict(), buf)
        return self.context.log_model(key, body=buf.getvalue(),
                                      framework="pytorch", **kwargs)


def apply_mlrun(model: torch.nn.Module, context=None, auto_ddp=True,
                bucket_cap_mb: int = None) -> MLRunTorchInterface:
    """Attach MLRun training instrumentation to a torch module
    (reference frameworks/pytorch mlrun_interface.py:555 _setup):
    when launched under WORLD_SIZE > 1, this auto-initializes the
    RCCL/gloo process group, pins the local GPU, broadcasts rank-0
    parameters and wraps the model in the bucketed-allreduce
    DistributedModel — the Horovod-equivalent, own engine."""
    interface = MLRunTorchInterface(model, context=context)
    if auto_ddp and interface.world_size > 1:
        import torch.distributed as dist

        from ..parallel.ddp import init_process_group

        if not dist.is_initialized():
            init_process_group()
        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK",
                                            interface.rank))
            torch.cuda.set_device(local_rank)
            model.cuda(local_rank)
        interface.setup_distributed(bucket_cap_mb=bucket_cap_mb)
    return interface
