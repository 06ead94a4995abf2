# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Llama-family generative engine, decode-optimized for MI355X.

Design (MI355X-first, not a port — the reference has no model code):
- weights live as raw bf16 [N, K] tensors resident in HBM (288 GB:
  an 8B model + caches is ~6% of one GPU)
- the decode step runs entirely on hand-written CDNA4 kernels
  (mlrun_amd/ops: MFMA skinny GEMM, fused add+RMSNorm, RoPE, GQA
  decode attention, SwiGLU) with every buffer preallocated and the
  whole token step captured in ONE hipGraph (torch.cuda.CUDAGraph is
  hipGraph on ROCm) — per-token CPU cost is a single graph replay
- prefill batches through hipBLASLt (torch.matmul) + SDPA: it is
  compute-bound and library GEMMs are the right tool there
  (guide: hand-write the fused hot ops, use hipBLASLt for plain GEMMs)
- tensor parallelism (Llama-70B): row/col sharded projections with
  one RCCL all-reduce after attn-out and after mlp-down, over xGMI
"""

import math
import typing
from dataclasses import dataclass

import torch

from .. import ops
from ..utils import logger


@dataclass
class LlamaConfig:
    name: str = "llama-3-8b"
    hidden_size: int = 4096
    intermediate_size: int = 14336
    num_layers: int = 32
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    vocab_size: int = 128256
    rope_theta: float = 500000.0
    rms_eps: float = 1e-5
    max_seq_len: int = 2048

    @classmethod
    def llama3_8b(cls, **over):
        return cls(**{**dict(name="llama-3-8b"), **over})

    @classmethod
    def llama3_70b(cls, **over):
        return cls(**{**dict(
            name="llama-3-70b", hidden_size=8192, intermediate_size=28672,
            num_layers=80, num_heads=64, num_kv_heads=8), **over})

    @classmethod
    def tiny(cls, **over):
        """Small config for CPU tests."""
        return cls(**{**dict(
            name="llama-tiny", hidden_size=256, intermediate_size=512,
            num_layers=2, num_heads=2, num_kv_heads=2, head_dim=128,
            vocab_size=1024, max_seq_len=256), **over})


class LlamaWeights:
    """Per-layer raw bf16 weight tensors ([N, K] row-major, matching
    the skinny-GEMM W layout).  TP sharding slices head/intermediate
    dims; each rank holds 1/tp of qkv+o+mlp weights."""

    def __init__(self, cfg: LlamaConfig, device, tp_rank=0, tp_size=1,
                 seed=1234):
        self.cfg = cfg
        self.device = device
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        hq = cfg.num_heads // tp_size
        hkv = max(cfg.num_kv_heads // tp_size, 1)
        inter = cfg.intermediate_size // tp_size
        d, h = cfg.head_dim, cfg.hidden_size
        gen = torch.Generator(device="cpu").manual_seed(seed)

        def w(n, k, std=0.02):
            t = torch.empty(n, k, dtype=torch.bfloat16, device=device)
            t.normal_(0.0, std)
            return t

        # random-init on device (no network for checkpoints; see
        # BASELINE.md — synthetic data / random weights)
        torch.manual_seed(seed + tp_rank)
        self.embed = w(cfg.vocab_size, h)
        self.layers = []
        for _ in range(cfg.num_layers):
            self.layers.append({
                "attn_norm": torch.ones(h, dtype=torch.bfloat16,
                                        device=device),
                "wqkv": w((hq + 2 * hkv) * d, h),
                "wo": w(h, hq * d),
                "ffn_norm": torch.ones(h, dtype=torch.bfloat16,
                                       device=device),
                # fused gate|up: one [2*inter, h] GEMM per mlp
                "wgu": w(2 * inter, h),
                "wdown": w(h, inter),
            })
        self.final_norm = torch.ones(h, dtype=torch.bfloat16, device=device)
        self.lm_head = w(cfg.vocab_size, h)
        self.hq, self.hkv, self.inter = hq, hkv, inter

    def load_state_dict(self, state: dict):
        """Load a checkpoint saved by state_dict() — or a TRAINING
        checkpoint (LlamaForCausalLM.state_dict(), keys like
        "embed.weight"/"blocks.N..."), converted in place: artifacts
        logged by LlamaTrainer.save_checkpoint serve directly."""
        if "embed" not in state and "embed.weight" in state:
            state = _training_state_to_decode(state)
        self.embed.copy_(state["embed"])
        for i, layer in enumerate(self.layers):
            for key in layer:
                layer[key].copy_(state[f"layers.{i}.{key}"])
        self.final_norm.copy_(state["final_norm"])
        self.lm_head.copy_(state["lm_head"])

    def state_dict(self) -> dict:
        out = {"embed": self.embed, "final_norm": self.final_norm,
               "lm_head": self.lm_head}
        for i, layer in enumerate(self.layers):
            for key, value in layer.items():
                out[f"layers.{i}.{key}"] = value
        return out


def _training_state_to_decode(state: dict) -> dict:
    """Map a LlamaForCausalLM (training) state dict onto the decode
    weight layout (same mapping as llama_train.export_decode_state,
    duplicated here so serving never imports the training stack)."""
    out = {
        "embed": state["embed.weight"],
        "final_norm": state["final_norm.weight"],
        "lm_head": state["lm_head.weight"],
    }
    i = 0
    while f"blocks.{i}.attn_norm.weight" in state:
        prefix = f"blocks.{i}."
        out[f"layers.{i}.attn_norm"] = state[prefix + "attn_norm.weight"]
        out[f"layers.{i}.wqkv"] = state[prefix + "attn.wqkv.weight"]
        out[f"layers.{i}.wo"] = state[prefix + "attn.wo.weight"]
        out[f"layers.{i}.ffn_norm"] = state[prefix + "ffn_norm.weight"]
        out[f"layers.{i}.wgu"] = state[prefix + "mlp.wgu.weight"]
        out[f"layers.{i}.wdown"] = state[prefix + "mlp.wdown.weight"]
        i += 1
    return {k: v.to(torch.bfloat16) for k, v in out.items()}


class LlamaDecodeEngine:
    """Batched generation engine: prefill (hipBLASLt+SDPA) + hipGraph-
    captured decode step on the custom kernel chain."""

    def __init__(self, cfg: LlamaConfig, batch_size: int, device=None,
                 tp_group=None, tp_rank=0, tp_size=1, use_graph=True,
                 seed=1234, weights: "LlamaWeights" = None,
                 weight_dtype: str = "bf16", kv_dtype: str = "bf16",
                 temperature: float = 0.0, top_k: int = 0):
        self.cfg = cfg
        self.B = batch_size
        self.device = torch.device(
            device or ("cuda:0" if torch.cuda.is_available() else "cpu"))
        self.on_gpu = self.device.type == "cuda"
        self.tp_group = tp_group
        self.tp_rank = tp_rank
        self.tp_size = tp_size
        self.use_graph = use_graph and self.on_gpu
        # sampling: 0 = greedy argmax; otherwise Gumbel-max over
        # logits/T (graph-safe: torch captures RNG ops so every
        # replay draws fresh noise), optionally top-k restricted.
        # temperature < 0 enables PER-SLOT temperatures via buf_temp
        # (one graph serves mixed greedy/sampled requests: greedy
        # slots get a tiny T so the scaled logit gap swamps the noise)
        self.temperature = float(temperature)
        self.top_k = int(top_k)
        self.per_slot_temp = self.temperature < 0
        self.buf_temp = torch.full((batch_size, 1), 1e-4,
                                   dtype=torch.float32,
                                   device=self.device) \
            if self.per_slot_temp else None
        # weights are read-only at serving time: replica engines on the
        # same GPU share one copy (16 GB for 8B) and keep private
        # KV caches / buffers / streams
        self.weights = weights if weights is not None else \
            LlamaWeights(cfg, self.device, tp_rank, tp_size, seed)
        w = self.weights
        d, h = cfg.head_dim, cfg.hidden_size
        B, smax = batch_size, cfg.max_seq_len
        bf16 = dict(dtype=torch.bfloat16, device=self.device)

        # per-layer KV caches (per-rank kv heads): [L, B, Hkv, Smax, D].
        # kv_dtype="fp8": OCP e4m3 bytes + per-row f32 scales — halves
        # KV memory AND decode-attention HBM traffic (dequant happens
        # while staging chunks through LDS).
        self.kv_dtype = kv_dtype
        if kv_dtype == "fp8":
            u8 = dict(dtype=torch.uint8, device=self.device)
            self.k_cache = torch.zeros(cfg.num_layers, B, w.hkv, smax, d,
                                       **u8)
            self.v_cache = torch.zeros(cfg.num_layers, B, w.hkv, smax, d,
                                       **u8)
            f32 = dict(dtype=torch.float32, device=self.device)
            self.k_scale = torch.ones(cfg.num_layers, B, w.hkv, smax,
                                      **f32)
            self.v_scale = torch.ones(cfg.num_layers, B, w.hkv, smax,
                                      **f32)
        else:
            self.k_cache = torch.zeros(cfg.num_layers, B, w.hkv, smax, d,
                                       **bf16)
            self.v_cache = torch.zeros(cfg.num_layers, B, w.hkv, smax, d,
                                       **bf16)
            self.k_scale = self.v_scale = None
        self.cache_lens = torch.zeros(B, dtype=torch.int32,
                                      device=self.device)
        self.buf_positions = torch.zeros(B, dtype=torch.int32,
                                         device=self.device)
        self.cos_sin = ops.build_rope_cos_sin(smax, d, cfg.rope_theta,
                                              self.device)

        # preallocated decode buffers (graph-capture requirement)
        qkv_n = (w.hq + 2 * w.hkv) * d
        # split-K scratch: ksplit * B * N f32 for the largest case
        gemm_shapes = [(qkv_n, h), (h, w.hq * d), (2 * w.inter, h),
                       (h, w.inter), (cfg.vocab_size, h)]
        scratch = max(ops.pick_gemm_plan(B, n, k)[0] * n
                      for n, k in gemm_shapes)
        self.buf_tokens = torch.zeros(B, dtype=torch.int64,
                                      device=self.device)
        self.buf_hidden = torch.zeros(B, h, **bf16)
        self.buf_residual = torch.zeros(B, h, **bf16)
        self.buf_qkv = torch.zeros(B, qkv_n, **bf16)
        self.buf_attn_out = torch.zeros(B, w.hq * d, **bf16)
        self.buf_proj = torch.zeros(B, h, **bf16)
        self.buf_gu = torch.zeros(B, 2 * w.inter, **bf16)
        self.buf_act = torch.zeros(B, w.inter, **bf16)
        self.buf_down = torch.zeros(B, h, **bf16)
        self.buf_logits = torch.zeros(B, cfg.vocab_size, **bf16)
        # flat f32 split-K scratch (shared across all decode GEMMs)
        self.buf_c32 = torch.empty(B * scratch, dtype=torch.float32,
                                   device=self.device)
        self.scale = 1.0 / math.sqrt(d)
        self._capture_stream_ctx = None
        # opt-in fp8-weight decode: per-row e4m3 packs keyed by the
        # bf16 master tensor (prefill stays bf16/hipBLASLt; ~7.5 GB
        # extra for the packs on 8B — 288 GB HBM absorbs it)
        self.weight_dtype = weight_dtype
        self._fp8_packs = {}
        if weight_dtype == "fp8w" and self.on_gpu and tp_size == 1:
            max_k = max(h, w.inter, w.hq * d)
            self.buf_a8 = torch.empty(B * max_k, dtype=torch.uint8,
                                      device=self.device)
            self.buf_a_scale = torch.empty(B, dtype=torch.float32,
                                           device=self.device)
            # hidden quantized as a sidecar of the rmsnorm kernels —
            # qkv/gate-up/lm_head consume it without a quant launch
            self.buf_h8 = torch.empty(B, h, dtype=torch.uint8,
                                      device=self.device)
            self.buf_h8_scale = torch.empty(B, dtype=torch.float32,
                                            device=self.device)
            self.rebuild_fp8_packs()
        # fill-based (ns4 at B=32): the graph bakes ONE nsplit for all
        # context lengths, and decode mostly runs at S << max_seq_len —
        # measured (scripts/bench_attn_longctx.py): chunk-aware ns16
        # wins >=10% only at S>=1024 but costs 2.4% on the headline
        # (S~160).  Long-context deployments: pass seq_len to
        # pick_attn_nsplit / set MLRUN_ATTN_NSPLIT.
        import os as _os
        _ns = _os.environ.get("MLRUN_ATTN_NSPLIT")
        self.attn_nsplit = int(_ns) if _ns else \
            ops.pick_attn_nsplit(B, w.hkv)
        self.buf_attn_ws = torch.empty(
            B * w.hq * self.attn_nsplit * (d + 2), dtype=torch.float32,
            device=self.device) if self.on_gpu else None
        self._graph = None
        self._ksplits = {}

    # ------------------------------------------------------------ gemm
    def _plan(self, w_):
        key = (w_.shape[0], w_.shape[1])
        if key not in self._ksplits:
            self._ksplits[key] = ops.pick_gemm_plan(self.B, *key)
        return self._ksplits[key]

    def _gemm(self, a, w_, out, a8=None, a_scale=None):
        """skinny GEMM into a preallocated bf16 out + f32 scratch.
        Uses the fp8-weight pack when this engine runs in fp8w mode;
        a8/a_scale skip the quant launch (rmsnorm sidecar)."""
        pack = self._fp8_packs.get(id(w_))
        if pack is not None:
            M, K = a.shape
            if a8 is None:
                a8 = self.buf_a8.narrow(0, 0, M * K).view(M, K)
                ops.quant_fp8_rows(a, a8, self.buf_a_scale)
                a_scale = self.buf_a_scale
            ksplit, _ = self._plan(w_)
            return ops.skinny_gemm_fp8(a8, a_scale, pack[0],
                                       pack[1], out=out,
                                       c_f32=self.buf_c32, ksplit=ksplit)
        ksplit, variant = self._plan(w_)
        return ops.skinny_gemm(a, w_, out=out, c_f32=self.buf_c32,
                               ksplit=ksplit, variant=variant)

    def _gemm_slabs(self, a, w_):
        """split-K GEMM emitting f32 slabs (consumer folds them);
        returns the ksplit used.  GPU-only."""
        from mlrun_amd import _hip_ops

        ksplit, variant = self._plan(w_)
        if ksplit <= 1:
            return 0  # caller must use the bf16 path
        _hip_ops.skinny_gemm_slabs(self.buf_c32, a, w_, ksplit, variant)
        return ksplit

    def _maybe_allreduce(self, t):
        if self.tp_size > 1:
            from ..parallel.tp import all_reduce_tensor

            all_reduce_tensor(t, group=self.tp_group)
        return t

    # ---------------------------------------------------------- decode
    def _decode_step_body(self):
        """One token step for all B sequences.  Reads buf_tokens,
        leaves next tokens in buf_tokens (greedy).  Entirely on-device:
        capturable as one hipGraph."""
        cfg, w = self.cfg, self.weights
        d = cfg.head_dim
        B = self.B
        # embedding (full embed table on every rank)
        torch.index_select(w.embed, 0, self.buf_tokens,
                           out=self.buf_residual)
        fp8 = self.weight_dtype == "fp8w" and self.on_gpu and \
            self.tp_size == 1
        h8 = self.buf_h8 if fp8 else None
        h8s = self.buf_h8_scale if fp8 else None
        ops.rmsnorm(self.buf_residual, w.layers[0]["attn_norm"],
                    eps=cfg.rms_eps, out=self.buf_hidden,
                    out8=h8, out_scale=h8s)
        # incoming token sits at position cache_lens; bump the length
        # once up front so attention covers it in every layer.  Clamp
        # to the cache end: under continuous batching idle/finished
        # slots keep stepping until reused — they must park on the
        # last row instead of running off the cache (graph-safe ops)
        torch.clamp(self.cache_lens, max=self.cfg.max_seq_len - 1,
                    out=self.buf_positions)
        self.cache_lens.add_(1).clamp_(max=self.cfg.max_seq_len)
        positions = self.buf_positions
        # slab fusion (GPU, TP=1): split-K consumers fold the f32 slabs
        # directly.  MEASURED: fusing into the 16-row norm kernels is a
        # net loss (consumer grid = B rows is parallelism-starved vs the
        # 2048-block reduce_cast), so only the rope/KV consumer (grid
        # B x heads) fuses by default; override with MLRUN_SLAB_MODE.
        import os as _os

        mode = _os.environ.get("MLRUN_SLAB_MODE", "rope")
        on = self.on_gpu and self.tp_size == 1 and mode != "none"
        use_slabs = on and self.weight_dtype == "bf16" and \
            self.kv_dtype == "bf16"  # qkv slab (bf16 caches only)
        use_norm_slabs = on and mode == "all" 
        for li, layer in enumerate(w.layers):
            # qkv projection -> rope -> caches
            done = 0
            if use_slabs:
                done = self._gemm_slabs(self.buf_hidden, layer["wqkv"])
                if done:
                    from mlrun_amd import _hip_ops

                    _hip_ops.rope_kv_slab(
                        self.buf_qkv, self.k_cache[li], self.v_cache[li],
                        self.buf_c32, positions, self.cos_sin, w.hq, done)
            if not done:
                self._gemm(self.buf_hidden, layer["wqkv"], self.buf_qkv,
                           a8=h8, a_scale=h8s)
                ops.rope_kv_fused(self.buf_qkv, self.k_cache[li],
                                  self.v_cache[li], positions,
                                  self.cos_sin, w.hq,
                                  k_scale=self._kscale(li),
                                  v_scale=self._vscale(li))
            q = self.buf_qkv[:, :w.hq * d].view(B, w.hq, d)
            attn_view = self.buf_attn_out.view(B, w.hq, d)
            ops.attn_decode(q, self.k_cache[li], self.v_cache[li],
                            self.cache_lens, self.scale, out=attn_view,
                            partial_ws=self.buf_attn_ws,
                            nsplit=self.attn_nsplit,
                            k_scale=self._kscale(li),
                            v_scale=self._vscale(li))
            # attn-out projection -> residual+norm
            done = 0
            if use_norm_slabs:
                done = self._gemm_slabs(self.buf_attn_out, layer["wo"])
                if done:
                    from mlrun_amd import _hip_ops

                    _hip_ops.fused_add_rmsnorm_slab(
                        self.buf_hidden, self.buf_residual, self.buf_c32,
                        layer["ffn_norm"], done, cfg.rms_eps)
            if not done:
                self._gemm(self.buf_attn_out, layer["wo"], self.buf_proj)
                self._maybe_allreduce(self.buf_proj)
                ops.fused_add_rmsnorm(self.buf_proj, layer["ffn_norm"],
                                      residual=self.buf_residual,
                                      eps=cfg.rms_eps, out=self.buf_hidden,
                                      out8=h8, out_scale=h8s)
            # mlp (fused gate|up projection)
            done = 0
            if use_norm_slabs:
                done = self._gemm_slabs(self.buf_hidden, layer["wgu"])
                if done:
                    from mlrun_amd import _hip_ops

                    _hip_ops.swiglu_slab(self.buf_act, self.buf_c32, done)
            if not done:
                self._gemm(self.buf_hidden, layer["wgu"], self.buf_gu,
                           a8=h8, a_scale=h8s)
                ops.swiglu_fused(self.buf_gu, out=self.buf_act)
            next_norm = w.layers[li + 1]["attn_norm"] \
                if li + 1 < cfg.num_layers else w.final_norm
            done = 0
            if use_norm_slabs:
                done = self._gemm_slabs(self.buf_act, layer["wdown"])
                if done:
                    from mlrun_amd import _hip_ops

                    _hip_ops.fused_add_rmsnorm_slab(
                        self.buf_hidden, self.buf_residual, self.buf_c32,
                        next_norm, done, cfg.rms_eps)
            if not done:
                self._gemm(self.buf_act, layer["wdown"], self.buf_down)
                self._maybe_allreduce(self.buf_down)
                ops.fused_add_rmsnorm(self.buf_down, next_norm,
                                      residual=self.buf_residual,
                                      eps=cfg.rms_eps, out=self.buf_hidden,
                                      out8=h8, out_scale=h8s)
        self._gemm(self.buf_hidden, w.lm_head, self.buf_logits,
                   a8=h8, a_scale=h8s)
        self._select_tokens(self.buf_logits, out=self.buf_tokens)

    def _select_tokens(self, logits: torch.Tensor,
                       out: torch.Tensor = None) -> torch.Tensor:
        """Greedy argmax, or temperature/top-k sampling via the
        Gumbel-max trick: argmax(logits/T + G), G = -log(-log(U)).
        One argmax either way — capture-safe and branch-free on
        device."""
        if self.per_slot_temp and logits.shape[0] == self.B:
            scores = logits.float() / self.buf_temp
            u = torch.rand_like(scores).clamp_(1e-9, 1.0 - 1e-9)
            gumbel = -torch.log(-torch.log(u))
            if out is None:
                return (scores + gumbel).argmax(dim=-1)
            torch.argmax(scores + gumbel, dim=-1, out=out)
            return out
        if self.temperature <= 0.0:
            if out is None:
                return logits.argmax(dim=-1)
            torch.argmax(logits, dim=-1, out=out)
            return out
        scores = logits.float() / self.temperature
        if self.top_k > 0 and self.top_k < scores.shape[-1]:
            kth = torch.topk(scores, self.top_k, dim=-1
                             ).values[..., -1:]
            scores = torch.where(scores < kth,
                                 torch.full_like(scores, float("-inf")),
                                 scores)
        u = torch.rand_like(scores).clamp_(1e-9, 1.0 - 1e-9)
        gumbel = -torch.log(-torch.log(u))
        if out is None:
            return (scores + gumbel).argmax(dim=-1)
        torch.argmax(scores + gumbel, dim=-1, out=out)
        return out

    _capture_lock = None

    def capture_graph(self):
        """Capture the decode step as one hipGraph (3 warmup runs on a
        side stream per torch graph discipline).  Captures are
        serialized process-wide: concurrent stream captures on one
        device corrupt each other."""
        if not self.use_graph or self._graph is not None:
            return
        import threading

        cls = type(self)
        if cls._capture_lock is None:
            cls._capture_lock = threading.Lock()
        with cls._capture_lock:
            self._capture_graph_locked()

    def _capture_graph_locked(self):
        if self._graph is not None:
            return
        import torch.distributed as dist

        lens_backup = self.cache_lens.clone()
        tokens_backup = self.buf_tokens.clone()
        if self.tp_size > 1 and dist.is_available() and \
                dist.is_initialized():
            # warmup runs REAL collectives — align ranks first so the
            # three eager steps pair up even if ranks arrive staggered
            dist.barrier(group=self.tp_group)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(3):
                self._decode_step_body()
        torch.cuda.current_stream().wait_stream(side)
        self.cache_lens.copy_(lens_backup)
        self.buf_tokens.copy_(tokens_backup)
        graph = torch.cuda.CUDAGraph()
        # RCCL collectives ARE capturable (the all-reduce kernel is
        # recorded into the graph and pairs up across ranks at replay),
        # but the NCCL watchdog thread polls events during capture —
        # "global" capture mode would see those as stray captures, so
        # distributed captures use thread_local error mode (the decode
        # body launches everything from this thread).
        err_mode = "thread_local" if dist.is_available() and \
            dist.is_initialized() else "global"
        with torch.cuda.graph(graph, capture_error_mode=err_mode):
            self._decode_step_body()
        # stream capture records without executing, but restore anyway
        # in case a backend executed eagerly during capture
        self.cache_lens.copy_(lens_backup)
        self.buf_tokens.copy_(tokens_backup)
        self._graph = graph
        logger.info("decode hipGraph captured", batch=self.B,
                    layers=self.cfg.num_layers)

    def decode_step(self):
        if self._graph is not None:
            self._graph.replay()
        else:
            self._decode_step_body()

    # --------------------------------------------------------- prefill
    def rebuild_fp8_packs(self):
        """(Re)quantize the fp8 weight packs from the CURRENT weights.
        Must be called again after load_state_dict — otherwise serving
        would run on packs quantized from the discarded init weights."""
        if self.weight_dtype != "fp8w" or not (self.on_gpu and
                                               self.tp_size == 1):
            return
        w = self.weights
        self._fp8_packs = {}
        for layer in w.layers:
            for key in ("wqkv", "wo", "wgu", "wdown"):
                self._fp8_packs[id(layer[key])] = \
                    ops.quantize_fp8_weight(layer[key])
        self._fp8_packs[id(w.lm_head)] = \
            ops.quantize_fp8_weight(w.lm_head)

    def _kscale(self, li):
        return None if self.k_scale is None else self.k_scale[li]

    def _vscale(self, li):
        return None if self.v_scale is None else self.v_scale[li]

    @torch.no_grad()
    def prefill(self, tokens: torch.Tensor) -> torch.Tensor:
        """Process prompts [B, S]; fill KV caches; return last-token
        logits [B, vocab].  hipBLASLt GEMMs + SDPA."""
        cfg, w = self.cfg, self.weights
        B, S = tokens.shape
        d = cfg.head_dim
        tokens = tokens.to(self.device)
        x = w.embed[tokens.reshape(-1)]  # [B*S, H]
        residual = x.contiguous()
        positions = torch.arange(S, dtype=torch.int32, device=self.device
                                 ).repeat(B)
        hidden = ops.rmsnorm(residual, w.layers[0]["attn_norm"],
                             eps=cfg.rms_eps)
        for li, layer in enumerate(w.layers):
            qkv = hidden @ layer["wqkv"].t()
            q = qkv[:, :w.hq * d].reshape(B * S, w.hq, d).contiguous()
            k = qkv[:, w.hq * d:(w.hq + w.hkv) * d].reshape(
                B * S, w.hkv, d).contiguous()
            v = qkv[:, (w.hq + w.hkv) * d:].reshape(B * S, w.hkv, d)
            ops.rope_inplace(q, positions, self.cos_sin)
            ops.rope_inplace(k, positions, self.cos_sin)
            kc = k.view(B, S, w.hkv, d).transpose(1, 2).contiguous()
            vc = v.view(B, S, w.hkv, d).transpose(1, 2).contiguous()
            if self.kv_dtype == "fp8":
                k8, ks = ops.quantize_kv_rows(kc)
                v8, vs = ops.quantize_kv_rows(vc)
                self.k_cache[li, :, :, :S] = k8
                self.v_cache[li, :, :, :S] = v8
                self.k_scale[li, :, :, :S] = ks
                self.v_scale[li, :, :, :S] = vs
            else:
                self.k_cache[li, :, :, :S] = kc
                self.v_cache[li, :, :, :S] = vc
            qh = q.view(B, S, w.hq, d).transpose(1, 2)
            attn = torch.nn.functional.scaled_dot_product_attention(
                qh, kc, vc, is_causal=True, enable_gqa=True)
            attn = attn.transpose(1, 2).reshape(B * S, w.hq * d)
            proj = attn @ layer["wo"].t()
            self._maybe_allreduce(proj)
            hidden = ops.fused_add_rmsnorm(proj, layer["ffn_norm"],
                                           residual=residual,
                                           eps=cfg.rms_eps)
            gu = hidden @ layer["wgu"].t()
            act = ops.swiglu_fused(gu.contiguous())
            down = act @ layer["wdown"].t()
            self._maybe_allreduce(down)
            next_norm = w.layers[li + 1]["attn_norm"] \
                if li + 1 < cfg.num_layers else w.final_norm
            hidden = ops.fused_add_rmsnorm(down, next_norm,
                                           residual=residual,
                                           eps=cfg.rms_eps)
        self.cache_lens.fill_(S)
        last = hidden.view(B, S, -1)[:, -1]
        logits = last @ w.lm_head.t()
        return logits

    @torch.no_grad()
    def prefill_slots(self, tokens: torch.Tensor,
                      slots: torch.Tensor) -> torch.Tensor:
        """Continuous-batching admission: prefill B' <= B prompts and
        scatter their KV/state into cache rows ``slots``, leaving the
        other slots' caches and lengths untouched.  Returns last-token
        logits [B'].  (Runs the same prefill math on a temporary
        B'-sized view; decode then advances ALL slots each step.)"""
        cfg, w = self.cfg, self.weights
        Bp, S = tokens.shape
        slots = torch.as_tensor(slots, dtype=torch.long,
                                device=self.device)
        assert Bp == slots.numel() and Bp <= self.B
        assert S < cfg.max_seq_len
        d = cfg.head_dim
        tokens = tokens.to(self.device)
        x = w.embed[tokens.reshape(-1)]
        residual = x.contiguous()
        positions = torch.arange(S, dtype=torch.int32,
                                 device=self.device).repeat(Bp)
        hidden = ops.rmsnorm(residual, w.layers[0]["attn_norm"],
                             eps=cfg.rms_eps)
        for li, layer in enumerate(w.layers):
            qkv = hidden @ layer["wqkv"].t()
            q = qkv[:, :w.hq * d].reshape(Bp * S, w.hq, d).contiguous()
            k = qkv[:, w.hq * d:(w.hq + w.hkv) * d].reshape(
                Bp * S, w.hkv, d).contiguous()
            v = qkv[:, (w.hq + w.hkv) * d:].reshape(Bp * S, w.hkv, d)
            ops.rope_inplace(q, positions, self.cos_sin)
            ops.rope_inplace(k, positions, self.cos_sin)
            kc = k.view(Bp, S, w.hkv, d).transpose(1, 2).contiguous()
            vc = v.view(Bp, S, w.hkv, d).transpose(1, 2).contiguous()
            if self.kv_dtype == "fp8":
                k8, ks = ops.quantize_kv_rows(kc)
                v8, vs = ops.quantize_kv_rows(vc)
                self.k_cache[li, slots, :, :S] = k8
                self.v_cache[li, slots, :, :S] = v8
                self.k_scale[li, slots, :, :S] = ks
                self.v_scale[li, slots, :, :S] = vs
            else:
                self.k_cache[li, slots, :, :S] = kc
                self.v_cache[li, slots, :, :S] = vc
            qh = q.view(Bp, S, w.hq, d).transpose(1, 2)
            attn = torch.nn.functional.scaled_dot_product_attention(
                qh, kc, vc, is_causal=True, enable_gqa=True)
            attn = attn.transpose(1, 2).reshape(Bp * S, w.hq * d)
            proj = attn @ layer["wo"].t()
            self._maybe_allreduce(proj)
            hidden = ops.fused_add_rmsnorm(proj, layer["ffn_norm"],
                                           residual=residual,
                                           eps=cfg.rms_eps)
            gu = hidden @ layer["wgu"].t()
            act = ops.swiglu_fused(gu.contiguous())
            down = act @ layer["wdown"].t()
            self._maybe_allreduce(down)
            next_norm = w.layers[li + 1]["attn_norm"] \
                if li + 1 < cfg.num_layers else w.final_norm
            hidden = ops.fused_add_rmsnorm(down, next_norm,
                                           residual=residual,
                                           eps=cfg.rms_eps)
        self.cache_lens[slots] = S
        last = hidden.view(Bp, S, -1)[:, -1]
        logits = last @ w.lm_head.t()
        if self.per_slot_temp:
            scores = logits.float() / self.buf_temp[slots]
            u = torch.rand_like(scores).clamp_(1e-9, 1.0 - 1e-9)
            self.buf_tokens[slots] = (
                scores - torch.log(-torch.log(u))).argmax(dim=-1)
        else:
            self.buf_tokens[slots] = self._select_tokens(logits)
        return logits

    # -------------------------------------------------------- generate
    @torch.no_grad()
    def generate(self, tokens: torch.Tensor, max_new_tokens: int = 32
                 ) -> torch.Tensor:
        """Greedy generation.  tokens [B, S] -> [B, max_new_tokens]."""
        B, S = tokens.shape
        assert B == self.B, f"engine built for batch {self.B}, got {B}"
        assert S + max_new_tokens <= self.cfg.max_seq_len
        logits = self.prefill(tokens)
        next_tokens = self._select_tokens(logits)
        self.buf_tokens.copy_(next_tokens)
        generated = [next_tokens.clone()]
        if self.use_graph and self._graph is None:
            self.capture_graph()
        for _ in range(max_new_tokens - 1):
            self.decode_step()
            generated.append(self.buf_tokens.clone())
        return torch.stack(generated, dim=1)

    def reset(self):
        self.cache_lens.zero_()


class LlamaServer:
    """V2ModelServer-protocol step serving LlamaDecodeEngine inside a
    serving graph (the north-star config: gen-AI serving graph with a
    Llama V2ModelServer — BASELINE.json).

    Request: {"inputs": [[tok, ...], ...], "max_tokens": G}
    Response outputs: [[generated tokens], ...]

    Serving structure:
    - ``replicas`` engine replicas per GPU, each on its OWN HIP stream
      with private KV caches/buffers but SHARED weights — decode is
      latency-bound (PMC wait/busy ~7), so concurrent replica streams
      raise aggregate HBM utilization; requests must arrive
      concurrently (or via the batcher) to overlap
    - a task queue feeds the replica workers; events are chunked to
      the engine batch size
    - ``batch_window_ms``: small concurrent requests coalesce into one
      engine pass (dynamic batching)
    """

    def __init__(self, context=None, name=None, model_path=None,
                 config=None, batch_size=16, max_new_tokens=32,
                 device=None, use_graph=True, batch_window_ms=0,
                 replicas=1, weight_dtype="bf16", kv_dtype="bf16",
                 scheduling="batch", stop_token: int = None,
                 temperature: float = 0.0, top_k: int = 0,
                 **class_args):
        import queue as queue_mod
        import threading

        from ..serving.v2_serving import V2ModelServer  # noqa: F401

        self.name = name
        self.context = context
        self.model_path = model_path
        self.ready = False
        self.error = ""
        self.protocol = "v2"
        self.model_spec = None
        self._model_logger = None
        self._params = class_args
        self.cfg_name = config or "llama-3-8b"
        self.batch_size = batch_size
        self.max_new_tokens = max_new_tokens
        self.device = device
        self.use_graph = use_graph
        self.weight_dtype = weight_dtype
        self.kv_dtype = kv_dtype
        self.scheduling = scheduling  # "batch" | "continuous"
        self.stop_token = stop_token  # finish a request at this token
        self.temperature = float(temperature)
        self.top_k = int(top_k)
        self.replicas = max(int(replicas), 1)
        self.engines: typing.List[LlamaDecodeEngine] = []
        self.batch_window_ms = batch_window_ms
        self._tasks: "queue_mod.Queue" = queue_mod.Queue()
        self._workers: typing.List[threading.Thread] = []
        self._pending = []          # [(prompt, max_new, future)]
        self._pending_cv = threading.Condition()
        self._batcher = None
        self.engine_calls = 0

    @property
    def engine(self):
        return self.engines[0] if self.engines else None

    def post_init(self, mode="sync"):
        stream = getattr(self.context, "stream", None) if self.context \
            else None
        if stream is not None:
            from ..serving.v2_serving import _ModelLogPusher

            self._model_logger = _ModelLogPusher(self, stream)
        if mode == "sync":
            self.load()
            self.ready = True

    def _build_config(self) -> LlamaConfig:
        if self.cfg_name in ("llama-3-8b", "8b"):
            cfg = LlamaConfig.llama3_8b()
        elif self.cfg_name in ("llama-3-70b", "70b"):
            cfg = LlamaConfig.llama3_70b()
        elif self.cfg_name == "tiny":
            cfg = LlamaConfig.tiny()
        elif isinstance(self.cfg_name, LlamaConfig):
            cfg = self.cfg_name
        else:
            raise ValueError(f"unknown llama config {self.cfg_name}")
        for key, value in self._params.items():
            if hasattr(cfg, key):
                setattr(cfg, key, value)
        return cfg

    def load(self):
        import threading

        cfg = self._build_config()
        if not torch.cuda.is_available():
            self.replicas = 1
        first = LlamaDecodeEngine(cfg, self.batch_size, device=self.device,
                                  use_graph=self.use_graph,
                                  weight_dtype=self.weight_dtype,
                                  kv_dtype=self.kv_dtype,
                                  temperature=self.temperature,
                                  top_k=self.top_k)
        self.engines = [first]
        for _ in range(self.replicas - 1):
            replica = LlamaDecodeEngine(
                cfg, self.batch_size, device=self.device,
                use_graph=self.use_graph, weights=first.weights,
                weight_dtype="bf16", kv_dtype=self.kv_dtype,
                temperature=self.temperature, top_k=self.top_k)
            replica.weight_dtype = first.weight_dtype
            replica._fp8_packs = first._fp8_packs  # shared packs
            if first.weight_dtype == "fp8w" and replica.on_gpu:
                import torch as _torch

                replica.buf_a8 = _torch.empty_like(first.buf_a8)
                replica.buf_a_scale = _torch.empty_like(
                    first.buf_a_scale)
                replica.buf_h8 = _torch.empty_like(first.buf_h8)
                replica.buf_h8_scale = _torch.empty_like(
                    first.buf_h8_scale)
            self.engines.append(replica)
        # load a checkpoint artifact if given (model_spec.yaml layout)
        if self.model_path:
            from ..artifacts import get_model

            model_file, spec, extra = get_model(self.model_path)
            state = torch.load(model_file, map_location=first.device,
                               weights_only=True)
            first.weights.load_state_dict(state)
            first.rebuild_fp8_packs()  # packs quantized pre-load are
            self.model_spec = spec     # stale (fp8w + model_path)
            for replica in self.engines[1:]:
                replica._fp8_packs = first._fp8_packs
        for idx, engine in enumerate(self.engines):
            engine._serve_stream = torch.cuda.Stream() \
                if engine.on_gpu else None
            if engine.use_graph:
                # capture sequentially before workers start (concurrent
                # captures on one device conflict); cache state is
                # restored by capture_graph
                if engine._serve_stream is not None:
                    with torch.cuda.stream(engine._serve_stream):
                        engine.capture_graph()
                    engine._serve_stream.synchronize()
                else:
                    engine.capture_graph()
            target = self._continuous_loop \
                if self.scheduling == "continuous" else self._worker_loop
            worker = threading.Thread(target=target,
                                      args=(engine,), daemon=True,
                                      name=f"llama-worker-{self.name}-{idx}")
            worker.start()
            self._workers.append(worker)

    def shutdown(self):
        """Stop the worker/batcher threads (one sentinel per engine).
        In-flight requests finish first; daemon threads would
        otherwise live until process exit (test hygiene)."""
        self._stopping = True
        with self._pending_cv:
            self._pending_cv.notify_all()
        for _ in self.engines:
            self._tasks.put(None)
        for worker in self._workers:
            worker.join(timeout=10)
        self._workers = []
        if self._batcher is not None:
            self._batcher.join(timeout=10)
            self._batcher = None

    def _worker_loop(self, engine: LlamaDecodeEngine):
        while True:
            item = self._tasks.get()
            if item is None:
                return
            prompts, max_new, future = item
            try:
                if engine._serve_stream is not None:
                    with torch.cuda.stream(engine._serve_stream):
                        result = self._generate_on(engine, prompts,
                                                   max_new)
                    engine._serve_stream.synchronize()
                else:
                    result = self._generate_on(engine, prompts, max_new)
                future.set_result(result)
            except Exception as exc:
                if not future.done():
                    future.set_exception(exc)

    def _stream_generate(self, inputs, max_new, event, req_temp=None):
        """Token streaming (continuous scheduling only): yields one
        json line per generated token as the slots produce them."""
        import concurrent.futures
        import json as _json
        import queue as queue_mod

        streams = []
        for prompt in inputs:
            future = concurrent.futures.Future()
            stream_q: "queue_mod.Queue" = queue_mod.Queue()
            self._tasks.put((prompt, max_new, future, stream_q,
                             req_temp))
            streams.append(stream_q)

        def gen():
            done = 0
            indexes = {id(q): i for i, q in enumerate(streams)}
            active = list(streams)
            while active:
                for stream_q in list(active):
                    try:
                        token = stream_q.get(timeout=0.05)
                    except queue_mod.Empty:
                        continue
                    if token is None:
                        active.remove(stream_q)
                        continue
                    yield _json.dumps(
                        {"index": indexes[id(stream_q)],
                         "token": token}) + "\n"

        return gen()

    def _continuous_loop(self, engine: LlamaDecodeEngine):
        """Token-level continuous batching: the engine decodes ALL
        slots every step; finished slots free up and new prompts are
        admitted at token boundaries via ``prefill_slots`` — no
        generate-pass barrier (vLLM-style scheduling on the fixed-B
        hipGraph).  Each request is one queue item (prompt, max_new,
        future)."""
        import contextlib
        import queue as queue_mod

        B = self.batch_size
        ring_len = engine.cfg.max_seq_len
        out_ring = torch.empty(B, ring_len, dtype=torch.int64,
                               device=engine.device)
        slots: list = [None] * B
        step = 0

        def stream_ctx():
            return torch.cuda.stream(engine._serve_stream) \
                if engine._serve_stream is not None else \
                contextlib.nullcontext()

        if engine.use_graph and engine._graph is None:
            with stream_ctx():
                engine.capture_graph()
        while True:
            free = [i for i in range(B) if slots[i] is None]
            have_active = len(free) < B
            taken = []
            if free:
                try:
                    item = self._tasks.get(block=not have_active)
                    if item is None:
                        return  # shutdown sentinel
                    taken.append(item)
                except queue_mod.Empty:
                    pass
                while len(taken) < len(free):
                    try:
                        item = self._tasks.get_nowait()
                        if item is None:
                            return
                        taken.append(item)
                    except queue_mod.Empty:
                        break
            if taken:
                self.engine_calls += 1
                prompts, admit_slots = [], []
                for item, slot in zip(taken, free):
                    prompt, max_new, future = item[:3]
                    stream_q = item[3] if len(item) > 3 else None
                    temp = item[4] if len(item) > 4 else None
                    if engine.per_slot_temp:
                        engine.buf_temp[slot] = max(
                            float(temp or 0.0), 1e-4)
                    max_new = max(1, min(max_new,
                                         engine.cfg.max_seq_len - 2))
                    keep = engine.cfg.max_seq_len - max_new - 1
                    prompt = list(prompt)[-keep:] or [0]
                    prompts.append(prompt)
                    admit_slots.append(slot)
                    slots[slot] = {"future": future, "max_new": max_new,
                                   "start": step, "produced": 1,
                                   "stream": stream_q}
                # prefill per EXACT prompt length: left-padding a
                # mixed-length admission group would make a request's
                # output depend on co-arriving traffic (pad tokens are
                # attended) — measured by the scheduler property test
                by_len: dict = {}
                for prompt, slot in zip(prompts, admit_slots):
                    by_len.setdefault(len(prompt), []).append(
                        (prompt, slot))
                try:
                    for length, group in by_len.items():
                        tokens = torch.tensor(
                            [p for p, _ in group], dtype=torch.int64)
                        slot_ids = torch.tensor(
                            [sl for _, sl in group], dtype=torch.long)
                        with stream_ctx():
                            engine.prefill_slots(tokens, slot_ids)
                            out_ring[slot_ids, step % ring_len] = \
                                engine.buf_tokens[slot_ids]
                except Exception as exc:  # admission failed: fail the
                    for slot in admit_slots:  # requests, free the slots
                        state = slots[slot]
                        if state is None:
                            continue
                        if not state["future"].done():
                            state["future"].set_exception(exc)
                        if state["stream"]:
                            state["stream"].put(None)
                        slots[slot] = None
                    logger.error("continuous admission failed",
                                 error=str(exc))
                    continue
                done_at_admit = [
                    slot for slot in admit_slots
                    if slots[slot]["max_new"] <= 1]
                if self.stop_token is not None or done_at_admit:
                    # the ADMISSION token can already be the stop
                    # token, and max_tokens=1 requests finish here
                    if engine._serve_stream is not None:
                        engine._serve_stream.synchronize()
                    first = engine.buf_tokens.cpu()
                    for slot in admit_slots:
                        state = slots[slot]
                        token = int(first[slot])
                        if token != self.stop_token and \
                                state["max_new"] > 1:
                            continue
                        if state["stream"]:
                            state["stream"].put(token)
                            state["stream"].put(None)
                            state["_first_sent"] = True
                        state["future"].set_result([token])
                        slots[slot] = None
            active = [i for i in range(B) if slots[i] is not None]
            if not active:
                continue
            stop = self.stop_token
            streaming = [i for i in active if slots[i]["stream"]]
            if streaming:
                # streamers need the admission token on the host too
                for i in streaming:
                    state = slots[i]
                    if state["produced"] == 1 and \
                            not state.get("_first_sent"):
                        if engine._serve_stream is not None:
                            engine._serve_stream.synchronize()
                        state["stream"].put(int(
                            out_ring[i, state["start"] % ring_len]))
                        state["_first_sent"] = True
            with stream_ctx():
                engine.decode_step()
                step += 1
                out_ring[:, step % ring_len].copy_(engine.buf_tokens)
            host_tokens = None
            if streaming or stop is not None:
                if engine._serve_stream is not None:
                    engine._serve_stream.synchronize()
                host_tokens = engine.buf_tokens.cpu()
                for i in streaming:
                    slots[i]["stream"].put(int(host_tokens[i]))
            finished = []
            for i in active:
                state = slots[i]
                state["produced"] += 1
                hit_stop = stop is not None and \
                    int(host_tokens[i]) == stop
                if state["produced"] >= state["max_new"] or hit_stop:
                    state["max_new"] = state["produced"]  # actual len
                    finished.append(i)
            if finished:
                if engine._serve_stream is not None:
                    engine._serve_stream.synchronize()
                for i in finished:
                    state = slots[i]
                    cols = torch.tensor(
                        [(state["start"] + j) % ring_len
                         for j in range(state["max_new"])],
                        dtype=torch.long)
                    result = out_ring[i, cols.to(out_ring.device)]
                    state["future"].set_result(result.cpu().tolist())
                    if state["stream"]:
                        state["stream"].put(None)  # end-of-stream
                    slots[i] = None

    def do_event(self, event):
        import concurrent.futures
        import time as _time

        start = _time.perf_counter()
        body = event.body if isinstance(event.body, dict) else {}
        path = event.path or ""
        if path.endswith("/ready"):
            event.body = {"name": self.name, "ready": self.ready}
            return event
        inputs = body.get("inputs")
        if inputs is None:
            raise ValueError('expected {"inputs": [[token ids], ...]}')
        max_new = int(body.get("max_tokens", self.max_new_tokens))
        req_temp = body.get("temperature")
        if self.scheduling == "continuous" and body.get("stream"):
            event.body = self._stream_generate(inputs, max_new, event,
                                               req_temp)
            return event
        if self.scheduling == "continuous":
            futures = []
            for prompt in inputs:
                future = concurrent.futures.Future()
                self._tasks.put((prompt, max_new, future, None,
                                 req_temp))
                futures.append(future)
            outputs = [f.result(timeout=600) for f in futures]
        elif self.batch_window_ms and len(inputs) < self.batch_size:
            outputs = self._batched_submit(inputs, max_new)
        else:
            futures = []
            for chunk_start in range(0, len(inputs), self.batch_size):
                chunk = inputs[chunk_start:chunk_start + self.batch_size]
                future = concurrent.futures.Future()
                self._tasks.put((chunk, max_new, future))
                futures.append(future)
            outputs = []
            for future in futures:
                outputs.extend(future.result(timeout=600))
        event.body = {"id": event.id, "model_name": self.name,
                      "outputs": outputs}
        if "retrieval" in body:  # RAG upstream step metadata
            event.body["retrieval"] = body["retrieval"]
        if self._model_logger:
            self._model_logger.push(start, {"inputs": [len(inputs)]},
                                    event.body)
        return event

    def _batched_submit(self, inputs: list, max_new: int) -> list:
        """Queue small requests; the batcher gathers up to batch_size
        prompts within batch_window_ms and submits ONE engine task."""
        import concurrent.futures
        import threading

        futures = []
        with self._pending_cv:
            if self._batcher is None:
                self._batcher = threading.Thread(
                    target=self._batch_loop, daemon=True,
                    name=f"llama-batcher-{self.name}")
                self._batcher.start()
            for prompt in inputs:
                future = concurrent.futures.Future()
                self._pending.append((prompt, max_new, future))
                futures.append(future)
            self._pending_cv.notify()
        return [f.result(timeout=600) for f in futures]

    def _batch_loop(self):
        import concurrent.futures
        import time as _time

        while not getattr(self, "_stopping", False):
            with self._pending_cv:
                while not self._pending:
                    self._pending_cv.wait()
                    if getattr(self, "_stopping", False):
                        return
            _time.sleep(self.batch_window_ms / 1000.0)
            with self._pending_cv:
                batch, self._pending = \
                    self._pending[:self.batch_size], \
                    self._pending[self.batch_size:]
            if not batch:
                continue
            prompts = [b[0] for b in batch]
            max_new = max(b[1] for b in batch)
            task_future = concurrent.futures.Future()
            self._tasks.put((prompts, max_new, task_future))
            try:
                results = task_future.result(timeout=600)
                for (prompt, want, future), result in zip(batch, results):
                    future.set_result(result[:want])
            except Exception as exc:
                for _, _, future in batch:
                    if not future.done():
                        future.set_exception(exc)

    def _generate_on(self, engine: LlamaDecodeEngine, prompts: list,
                     max_new: int) -> list:
        """Batch generation with per-LENGTH prefill groups: padding a
        mixed-length batch would let pad tokens be attended, making a
        prompt's output depend on its co-batch (see the continuous
        scheduler property test)."""
        self.engine_calls += 1
        n = len(prompts)
        by_len: dict = {}
        for i, prompt in enumerate(prompts):
            by_len.setdefault(len(prompt), []).append(i)
        engine.reset()
        for length, idxs in by_len.items():
            tokens = torch.tensor([prompts[i] for i in idxs],
                                  dtype=torch.int64)
            engine.prefill_slots(tokens, torch.tensor(idxs,
                                                      dtype=torch.long))
        if engine.use_graph and engine._graph is None:
            engine.capture_graph()
        generated = [engine.buf_tokens.clone()]
        for _ in range(max_new - 1):
            engine.decode_step()
            generated.append(engine.buf_tokens.clone())
        out = torch.stack(generated, dim=1)
        rows = out[:n].cpu().tolist()
        if self.stop_token is not None:
            trimmed = []
            for row in rows:
                if self.stop_token in row:
                    row = row[:row.index(self.stop_token) + 1]
                trimmed.append(row)
            return trimmed
        return rows

    def logged_results(self, request, response, op):
        return request.get("inputs"), None

    def stats(self):
        return {}
