# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Trainable Llama (fine-tuning path — baseline config 4).

Training is compute-bound: the GEMMs ride hipBLASLt (torch.matmul) and
attention rides SDPA, which is the idiomatic MI355X split (hand-write
the memory-bound fused serving ops, use the tuned GEMM library for
dense training math).  What the framework owns here is the
distributed layer: DistributedModel's bucketed bf16 all-reduce over
RCCL/xGMI overlapped with backward (mlrun_amd/parallel/ddp.py), the
MpiRuntime gang launcher, and checkpointing into model artifacts.

288 GB HBM sizing: Llama-3-8B bf16 params (16 GB) + grads (16 GB) +
Adam moments fp32 (64 GB) + activations fits a single MI355X with
room; DP=8 shards nothing and still fits.
"""


import torch
import torch.nn as nn
import torch.nn.functional as F

from .llama import LlamaConfig


class RMSNorm(nn.Module):
    def __init__(self, hidden: int, eps: float = 1e-5):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden))
        self.eps = eps

    def forward(self, x):
        dtype = x.dtype
        x = x.float()
        var = x.pow(2).mean(dim=-1, keepdim=True)
        return (x * torch.rsqrt(var + self.eps)).to(dtype) * self.weight


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.hq = cfg.num_heads
        self.hkv = cfg.num_kv_heads
        self.d = cfg.head_dim
        h = cfg.hidden_size
        self.wqkv = nn.Linear(h, (self.hq + 2 * self.hkv) * self.d,
                              bias=False)
        self.wo = nn.Linear(self.hq * self.d, h, bias=False)

    def forward(self, x, cos, sin):
        B, S, _ = x.shape
        qkv = self.wqkv(x)
        q, k, v = qkv.split([self.hq * self.d, self.hkv * self.d,
                             self.hkv * self.d], dim=-1)
        q = q.view(B, S, self.hq, self.d).transpose(1, 2)
        k = k.view(B, S, self.hkv, self.d).transpose(1, 2)
        v = v.view(B, S, self.hkv, self.d).transpose(1, 2)
        q = _apply_rope(q, cos, sin)
        k = _apply_rope(k, cos, sin)
        out = F.scaled_dot_product_attention(q, k, v, is_causal=True,
                                             enable_gqa=True)
        out = out.transpose(1, 2).reshape(B, S, self.hq * self.d)
        return self.wo(out)


def _apply_rope(x, cos, sin):
    # x: [B, H, S, D]; cos/sin: [S, D/2]
    half = x.shape[-1] // 2
    x1, x2 = x[..., :half], x[..., half:]
    cos = cos.view(1, 1, -1, half)
    sin = sin.view(1, 1, -1, half)
    return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.wgu = nn.Linear(cfg.hidden_size, 2 * cfg.intermediate_size,
                             bias=False)
        self.wdown = nn.Linear(cfg.intermediate_size, cfg.hidden_size,
                               bias=False)
        self.inter = cfg.intermediate_size

    def forward(self, x):
        gu = self.wgu(x)
        return self.wdown(F.silu(gu[..., :self.inter]) *
                          gu[..., self.inter:])


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.attn_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.attn = LlamaAttention(cfg)
        self.ffn_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.mlp = LlamaMLP(cfg)

    def forward(self, x, cos, sin):
        x = x + self.attn(self.attn_norm(x), cos, sin)
        x = x + self.mlp(self.ffn_norm(x))
        return x


class LlamaForCausalLM(nn.Module):
    """Causal-LM Llama for fine-tuning (bf16 master weights)."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        self.embed = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.blocks = nn.ModuleList(
            LlamaBlock(cfg) for _ in range(cfg.num_layers))
        self.final_norm = RMSNorm(cfg.hidden_size, cfg.rms_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size,
                                 bias=False)
        half = cfg.head_dim // 2
        inv_freq = 1.0 / (cfg.rope_theta ** (
            torch.arange(0, half, dtype=torch.float32) / half))
        self.register_buffer("inv_freq", inv_freq, persistent=False)
        self.apply(self._init)

    @staticmethod
    def _init(module):
        if isinstance(module, (nn.Linear, nn.Embedding)):
            nn.init.normal_(module.weight, std=0.02)

    def _rope_tables(self, S, device, dtype):
        pos = torch.arange(S, device=device, dtype=torch.float32)
        angles = torch.outer(pos, self.inv_freq.to(device))
        return angles.cos().to(dtype), angles.sin().to(dtype)

    def forward(self, tokens, labels=None):
        B, S = tokens.shape
        x = self.embed(tokens)
        cos, sin = self._rope_tables(S, tokens.device, x.dtype)
        for block in self.blocks:
            x = block(x, cos, sin)
        x = self.final_norm(x)
        logits = self.lm_head(x)
        if labels is None:
            return logits
        loss = F.cross_entropy(
            logits[:, :-1].reshape(-1, self.cfg.vocab_size).float(),
            labels[:, 1:].reshape(-1))
        return logits, loss


class LlamaTrainer:
    """The config-4 training loop: DDP over RCCL/xGMI, synthetic data,
    tokens/s accounting, artifact checkpointing."""

    def __init__(self, cfg: LlamaConfig, device=None, lr: float = 1e-4,
                 bucket_cap_mb: int = None, context=None,
                 grad_accum_steps: int = 1, warmup_steps: int = 0,
                 total_steps: int = 0, min_lr_ratio: float = 0.1,
                 grad_clip: float = 1.0):
        self.cfg = cfg
        self.grad_accum_steps = max(int(grad_accum_steps), 1)
        self.warmup_steps = warmup_steps
        self.total_steps = total_steps
        self.min_lr_ratio = min_lr_ratio
        self.grad_clip = grad_clip
        self.base_lr = lr
        self.step_count = 0
        self._accum = 0
        self.device = torch.device(
            device or ("cuda:0" if torch.cuda.is_available() else "cpu"))
        dtype = torch.bfloat16 if self.device.type == "cuda" \
            else torch.float32
        self.model = LlamaForCausalLM(cfg).to(self.device, dtype=dtype)
        self.context = context
        import torch.distributed as dist

        self.world_size = dist.get_world_size() if dist.is_initialized() \
            else 1
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        if self.world_size > 1:
            from ..parallel.ddp import DistributedModel

            self.ddp = DistributedModel(self.model,
                                        bucket_cap_mb=bucket_cap_mb)
        else:
            self.ddp = None
        self.optimizer = torch.optim.AdamW(self.model.parameters(), lr=lr,
                                           betas=(0.9, 0.95),
                                           weight_decay=0.1, foreach=True)

    def current_lr(self) -> float:
        """Linear warmup + cosine decay to min_lr_ratio (standard
        llama fine-tune schedule; constant lr when total_steps=0)."""
        import math

        step = self.step_count
        if self.warmup_steps and step < self.warmup_steps:
            return self.base_lr * (step + 1) / self.warmup_steps
        if not self.total_steps:
            return self.base_lr
        progress = min(1.0, (step - self.warmup_steps) /
                       max(1, self.total_steps - self.warmup_steps))
        floor = self.base_lr * self.min_lr_ratio
        return floor + 0.5 * (self.base_lr - floor) * (
            1.0 + math.cos(math.pi * progress))

    def train_step(self, tokens: torch.Tensor) -> float:
        """One micro-step: forward, backward (+overlapped all-reduce);
        the optimizer steps every ``grad_accum_steps`` micro-steps
        with the scheduled lr."""
        tokens = tokens.to(self.device)
        model = self.ddp or self.model
        if self._accum == 0:
            self.optimizer.zero_grad(set_to_none=True)
        if self.ddp is not None:
            # reduce only on the accumulation boundary (no_sync analog)
            self.ddp.require_backward_grad_sync = \
                self._accum + 1 >= self.grad_accum_steps
        _, loss = model(tokens, labels=tokens)
        (loss / self.grad_accum_steps).backward()
        self._accum += 1
        if self._accum >= self.grad_accum_steps:
            self._accum = 0
            if self.ddp is not None:
                self.ddp.finalize_backward()
            if self.grad_clip:
                torch.nn.utils.clip_grad_norm_(self.model.parameters(),
                                               self.grad_clip)
            lr = self.current_lr()
            for group in self.optimizer.param_groups:
                group["lr"] = lr
            self.optimizer.step()
            self.step_count += 1
        return float(loss.detach())

    def save_checkpoint(self, key: str = "model"):
        """Log the model as an artifact (rank 0 only) in the standard
        model_spec.yaml layout."""
        if self.rank != 0 or self.context is None:
            return None
        import io

        buf = io.BytesIO()
        torch.save(self.model.state_dict(), buf)
        return self.context.log_model(
            key, body=buf.getvalue(), framework="pytorch",
            parameters={"config": self.cfg.name,
                        "num_layers": self.cfg.num_layers})

    def load_checkpoint(self, model_uri: str):
        from ..artifacts import get_model

        model_file, spec, extra = get_model(model_uri)
        state = torch.load(model_file, map_location=self.device,
                           weights_only=True)
        self.model.load_state_dict(state)
        return spec


def export_decode_state(model: "LlamaForCausalLM") -> dict:
    """Convert a trained LlamaForCausalLM state into the serving
    engine's weight layout (LlamaWeights.state_dict format) — the
    train -> checkpoint -> deploy bridge."""
    state = model.state_dict()
    out = {
        "embed": state["embed.weight"],
        "final_norm": state["final_norm.weight"],
        "lm_head": state["lm_head.weight"],
    }
    n_layers = model.cfg.num_layers
    for i in range(n_layers):
        prefix = f"blocks.{i}."
        out[f"layers.{i}.attn_norm"] = state[prefix + "attn_norm.weight"]
        out[f"layers.{i}.wqkv"] = state[prefix + "attn.wqkv.weight"]
        out[f"layers.{i}.wo"] = state[prefix + "attn.wo.weight"]
        out[f"layers.{i}.ffn_norm"] = state[prefix + "ffn_norm.weight"]
        out[f"layers.{i}.wgu"] = state[prefix + "mlp.wgu.weight"]
        out[f"layers.{i}.wdown"] = state[prefix + "mlp.wdown.weight"]
    return {k: v.to(torch.bfloat16) for k, v in out.items()}
