"""Training loop for the real-run side of perf-vs-real validation.

Megatron-semantics mixed precision:
* bf16 parameters
* fp32 main_grad buffers (post-accumulate-grad hooks, matching
  use_fp32_accum_grad in the strategy schema)
* fp32 master weights + Adam moments (12 B/param optimizer state)
* DP: flat fp32 grad all_reduce over RCCL, optionally bucket-overlapped
  with backward

The memory/time structure of one step is exactly what PerfLLM predicts
with zero_state=0 for the same strategy config.
"""

from __future__ import annotations

import os
import time
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

from ..core.config import ModelConfig
from .model import LlamaForTraining


@dataclass
class TrainConfig:
    seq_len: int = 4096
    micro_batch_size: int = 1
    micro_batch_num: int = 1
    lr: float = 1e-4
    adam_betas = (0.9, 0.95)
    adam_eps: float = 1e-8
    grad_clip: float = 1.0
    overlap_grad_reduce: bool = True
    bucket_bytes: int = 100 * 1024**2


class MixedPrecisionAdam:
    """fp32 master + moments over bf16 params with fp32 main grads."""

    def __init__(self, params, cfg: TrainConfig):
        self.params = [p for p in params if p.requires_grad]
        self.cfg = cfg
        self.masters = [p.detach().float().clone() for p in self.params]
        self.m = [torch.zeros_like(w) for w in self.masters]
        self.v = [torch.zeros_like(w) for w in self.masters]
        self.t = 0
        for p in self.params:
            p.main_grad = torch.zeros(p.shape, dtype=torch.float32,
                                      device=p.device)

    def zero_grad(self):
        for p in self.params:
            p.main_grad.zero_()

    @torch.no_grad()
    def step(self):
        """In-place Adam: no fp32 temporaries beyond one per-param denom
        (Megatron's fused optimizer behaves the same; the foreach_div path
        would transiently allocate 2x the full fp32 state = +64 GiB on an
        8B model)."""
        self.t += 1
        b1, b2 = self.cfg.adam_betas
        grads = [p.main_grad for p in self.params]
        # global grad-norm clip (Megatron clip_grad)
        norm = torch.norm(torch.stack([g.norm(2) for g in grads]), 2)
        scale = self.cfg.grad_clip / (norm + 1e-6)
        if scale < 1.0:
            torch._foreach_mul_(grads, scale)
        torch._foreach_mul_(self.m, b1)
        torch._foreach_add_(self.m, grads, alpha=1 - b1)
        torch._foreach_mul_(self.v, b2)
        torch._foreach_addcmul_(self.v, grads, grads, value=1 - b2)
        bc1 = 1 - b1 ** self.t
        bc2 = 1 - b2 ** self.t
        # fold bias corrections: m/(sqrt(v)/sqrt(bc2)+eps)/bc1
        #   = sqrt(bc2)/bc1 * m/(sqrt(v)+eps*sqrt(bc2))  (exact)
        sqrt_bc2 = bc2 ** 0.5
        step_size = self.cfg.lr * sqrt_bc2 / bc1
        eps2 = self.cfg.adam_eps * sqrt_bc2
        for w, m, v, p in zip(self.masters, self.m, self.v, self.params):
            denom = v.sqrt().add_(eps2)
            w.addcdiv_(m, denom, value=-step_size)
            p.data.copy_(w)


class DataParallelGradReducer:
    """Bucketed fp32 main_grad all_reduce, overlapped with backward."""

    def __init__(self, params, overlap: bool, bucket_bytes: int):
        self.params = [p for p in params if p.requires_grad]
        self.overlap = overlap and dist.is_initialized() and dist.get_world_size() > 1
        self.enabled = dist.is_initialized() and dist.get_world_size() > 1
        self.bucket_bytes = bucket_bytes
        self.handles = []
        # Megatron no_sync semantics: only the LAST microbatch's backward
        # triggers the bucketed all_reduce
        self.reduce_this_pass = True
        if self.overlap:
            # reverse order (grads become ready back-to-front)
            buckets, cur, cur_bytes = [], [], 0
            for p in reversed(self.params):
                cur.append(p)
                cur_bytes += p.numel() * 4
                if cur_bytes >= bucket_bytes:
                    buckets.append(cur)
                    cur, cur_bytes = [], 0
            if cur:
                buckets.append(cur)
            self._pending = {}
            for bi, bucket in enumerate(buckets):
                remaining = {id(p) for p in bucket}
                for p in bucket:
                    p.register_post_accumulate_grad_hook(
                        self._make_hook(bi, bucket, remaining))
            self._buckets = buckets

    def _make_hook(self, bi, bucket, remaining):
        def hook(p):
            if not getattr(p, "_fused_wgrad", False) or not p.is_cuda:
                p.main_grad.add_(p.grad.float())
            p.grad = None
            remaining.discard(id(p))
            if not remaining:
                if self.reduce_this_pass:
                    ws = dist.get_world_size()
                    for q in bucket:
                        q.main_grad.div_(ws)
                        self.handles.append(
                            dist.all_reduce(q.main_grad, async_op=True))
                # rearm for the next backward pass
                remaining.update(id(q) for q in bucket)
        return hook

    def finalize(self):
        if self.overlap:
            for h in self.handles:
                h.wait()
            self.handles.clear()
        elif self.enabled:
            ws = dist.get_world_size()
            for p in self.params:
                p.main_grad.div_(ws)
                dist.all_reduce(p.main_grad)


def accumulate_main_grads(params):
    for p in params:
        if p.grad is not None:
            if not getattr(p, "_fused_wgrad", False) or not p.is_cuda:
                p.main_grad.add_(p.grad.float())
            p.grad = None


def build_trainer(model_cfg: ModelConfig, cfg: TrainConfig, device="cuda",
                  tp_size: int = 1):
    torch.manual_seed(1234)
    # Megatron-style vocab padding (keeps CE vocab a GPU-friendly multiple
    # and makes GEMM shape keys match the calibration tables)
    model_cfg.maybe_pad_vocab_size(tp_size)
    model = LlamaForTraining(model_cfg, cfg.seq_len, device=device)
    opt = MixedPrecisionAdam(model.parameters(), cfg)
    reducer = DataParallelGradReducer(list(model.parameters()),
                                      cfg.overlap_grad_reduce, cfg.bucket_bytes)
    return model, opt, reducer


def train_step(model, opt, reducer, tokens, labels, micro_batch_num=1):
    """One optimizer step = micro_batch_num fwd+bwd + grad reduce + adam."""
    opt.zero_grad()
    total_loss = 0.0
    for mb in range(micro_batch_num):
        reducer.reduce_this_pass = mb == micro_batch_num - 1
        loss = model(tokens[mb], labels[mb])
        loss.backward()
        if not reducer.overlap:
            accumulate_main_grads(opt.params)
        total_loss += loss.item()
    reducer.finalize()
    opt.step()
    return total_loss / micro_batch_num


def make_synthetic_batch(vocab_size, micro_batch_num, micro_batch_size,
                         seq_len, device, seed=0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    toks = torch.randint(0, vocab_size,
                         (micro_batch_num, micro_batch_size, seq_len),
                         generator=g)
    labels = torch.randint(0, vocab_size,
                           (micro_batch_num, micro_batch_size, seq_len),
                           generator=g)
    return toks.to(device), labels.to(device)
