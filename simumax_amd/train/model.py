"""Megatron-style Llama training model used as the real-run side of the
perf-vs-real validation (the reference validates against Megatron-LM runs:
tools/b200/run_megatron_perf_real_pipeline.py; here the trainer is in-repo
and ROCm-native).

Uses the gfx950 HIP kernels for the fused hot ops (RMSNorm, RoPE, SwiGLU,
flash attention, fused CE); plain linear layers go through hipBLASLt via
torch.matmul. Activation-save behavior deliberately matches the
simulator's per-leaf cache accounting (simumax_amd/ops/dense.py docstring).
"""

from __future__ import annotations


import torch
import torch.nn as nn  # noqa: F401 (Embedding)

from ..core.config import ModelConfig
from ..kernels import ops as K


class GQAAttention(nn.Module):
    """Fused-qkv GQA attention (Llama/Qwen/Mixtral family). With tp > 1
    the heads are split over the TP group: column-parallel qkv (local
    head sections), row-parallel out projection (Megatron layout)."""

    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16, device=None,
                 tp_group=None, tp_size=1, sp=False, cp_group=None,
                 cp_size=1, cp_rank=0, cp_comm_type="a2a",
                 cp_sharding="contiguous", fp8=False):
        super().__init__()
        h = cfg.hidden_size
        assert cfg.head_num % tp_size == 0 and cfg.kv_head_num % tp_size == 0
        self.heads = cfg.head_num // tp_size
        self.kv_heads = cfg.kv_head_num // tp_size
        if cp_size > 1 and cp_comm_type == "a2a":
            # Ulysses CP: heads scattered over cp inside attention
            # (all_gather mode keeps heads whole — no divisibility rule)
            assert self.heads % cp_size == 0 and self.kv_heads % cp_size == 0
            assert cp_sharding == "contiguous", \
                "a2a reassembles shards in rank order (contiguous only)"
        self.head_size = cfg.head_size
        self.tp_group = tp_group
        self.cp_group = cp_group
        self.cp_rank = cp_rank
        self.cp_comm_type = cp_comm_type
        self.cp_size_ = cp_size
        self.cp_sharding = cp_sharding
        self.sp = sp
        qkv_out = (self.heads + 2 * self.kv_heads) * cfg.head_size
        Lin = K.FusedLinear
        if fp8:
            from ..kernels.fp8 import Fp8Linear as Lin
        self.qkv_proj = Lin(h, qkv_out, dtype=dtype, device=device)
        self.out_proj = Lin(self.heads * cfg.head_size, h,
                            dtype=dtype, device=device)
        if tp_size > 1:
            self.qkv_proj.weight._is_tp_shard = True
            self.out_proj.weight._is_tp_shard = True

    def forward(self, y, rope_cs, pos):
        from .tp import copy_to_tp, gather_seq, reduce_from_tp, scatter_seq

        if self.sp:
            y = gather_seq(y, self.tp_group)   # [B, S/tp, H] -> [B, S, H]
            qkv = self.qkv_proj(y)
        else:
            qkv = self.qkv_proj(copy_to_tp(y, self.tp_group))
        B, S, _ = y.shape
        d = self.head_size
        q, k, v = qkv.split(
            [self.heads * d, self.kv_heads * d, self.kv_heads * d], dim=-1)
        q = K.apply_rope(q.reshape(B * S, self.heads, d), rope_cs, pos)
        k = K.apply_rope(k.reshape(B * S, self.kv_heads, d), rope_cs, pos)
        q = q.view(B, S, self.heads, d)
        k = k.view(B, S, self.kv_heads, d)
        v = v.reshape(B, S, self.kv_heads, d)
        if self.cp_group is not None and self.cp_comm_type == "ring":
            # ring attention: K/V blocks circulate over p2p, online-LSE
            # accumulation — one remote block resident at a time
            from .cp import ring_attention_pos

            ctx = ring_attention_pos(q, k, v, self.cp_group, self.cp_rank,
                                     pos[:S].long(),
                                     self.cp_sharding == "zigzag")
        elif self.cp_group is not None and self.cp_comm_type == "all_gather":
            # kv all_gather: q stays seq-sharded; K/V are gathered to the
            # full sequence and attention runs with a positional causal
            # mask (math SDP — the a2a mode is the flash-kernel path)
            from .cp import cp_allgather_kv, cp_positions, masked_sdp

            k_full = cp_allgather_kv(k, self.cp_group)
            v_full = cp_allgather_kv(v, self.cp_group)
            cp = self.cp_size_
            kpos = torch.cat([cp_positions(S * cp, cp, r,
                                           self.cp_sharding == "zigzag",
                                           q.device)
                              for r in range(cp)]).long()
            ctx = masked_sdp(q, k_full, v_full, pos[:S].long(), kpos)
        elif self.cp_group is not None:
            # a2a: scatter heads / gather sequence (Ulysses), flash on the
            # full sequence, inverse a2a on the context
            from .cp import cp_post_attention, cp_pre_attention

            q = cp_pre_attention(q, self.cp_group)
            k = cp_pre_attention(k, self.cp_group)
            v = cp_pre_attention(v, self.cp_group)
            ctx = K.flash_attention(q, k, v, causal=True)
            ctx = cp_post_attention(ctx, self.cp_group)
        else:
            ctx = K.flash_attention(q, k, v, causal=True)
        out = self.out_proj(ctx.reshape(B, S, self.heads * d))
        if self.sp:
            return scatter_seq(out, self.tp_group)
        return reduce_from_tp(out, self.tp_group)


class MLAAttention(nn.Module):
    """DeepSeek multi-head latent attention: low-rank q/kv projections,
    RoPE only on the positional sub-dims, asymmetric-head flash SDP
    (Dqk = qk_head_dim + qk_pos_emb_head_dim, Dv = v_head_dim). Mirrors
    the simulator's MLAAttention op graph (ops/dense.py MLAAttention)."""

    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16, device=None,
                 cp_group=None, cp_rank=0, cp_comm_type="a2a", cp_size=1,
                 cp_sharding="contiguous"):
        super().__init__()
        h = cfg.hidden_size
        self.heads = cfg.head_num
        self.cp_group = cp_group
        self.cp_rank = cp_rank
        self.cp_comm_type = cp_comm_type
        self.cp_size_ = cp_size
        self.cp_sharding = cp_sharding
        self.dn = cfg.qk_head_dim            # nope part (128)
        self.dp = cfg.qk_pos_emb_head_dim    # rope part (64)
        self.dv = cfg.v_head_dim
        self.kv_lora = cfg.kv_lora_rank
        qk_total = self.dn + self.dp
        self.q_lora = bool(cfg.q_lora_rank)
        if self.q_lora:
            # DeepSeek-V2+: low-rank q projection
            self.q_down = K.FusedLinear(h, cfg.q_lora_rank, dtype=dtype,
                                        device=device)
            self.q_norm = K.RMSNorm(cfg.q_lora_rank, dtype=dtype,
                                    device=device)
            self.q_up = K.FusedLinear(cfg.q_lora_rank, self.heads * qk_total,
                                      dtype=dtype, device=device)
        else:
            # DeepSeek-V2-Lite: direct q projection (simulator q_proj
            # branch, ops/dense.py MLAAttention)
            self.q_proj = K.FusedLinear(h, self.heads * qk_total,
                                        dtype=dtype, device=device)
        self.kv_down = K.FusedLinear(h, self.kv_lora + self.dp, dtype=dtype,
                                     device=device)
        self.kv_norm = K.RMSNorm(self.kv_lora, dtype=dtype, device=device)
        self.kv_up = K.FusedLinear(self.kv_lora,
                                   self.heads * (self.dn + self.dv),
                                   dtype=dtype, device=device)
        self.out_proj = K.FusedLinear(self.heads * self.dv, h, dtype=dtype,
                                      device=device)

    def forward(self, y, rope_cs, pos):
        B, S, _ = y.shape
        H, dn, dp, dv = self.heads, self.dn, self.dp, self.dv
        if self.q_lora:
            q = self.q_up(self.q_norm(self.q_down(y))).view(B * S, H, dn + dp)
        else:
            q = self.q_proj(y).view(B * S, H, dn + dp)
        q_pe = K.apply_rope(q[..., dn:].contiguous(), rope_cs, pos)
        kv = self.kv_down(y)
        k_pe = K.apply_rope(
            kv[..., self.kv_lora:].reshape(B * S, 1, dp).contiguous(),
            rope_cs, pos)
        kvu = self.kv_up(self.kv_norm(kv[..., :self.kv_lora]))             .view(B * S, H, dn + dv)
        qf = torch.cat([q[..., :dn], q_pe], dim=-1).view(B, S, H, dn + dp)
        kf = torch.cat([kvu[..., :dn], k_pe.expand(B * S, H, dp)], dim=-1)             .view(B, S, H, dn + dp)
        v = kvu[..., dn:].reshape(B, S, H, dv).contiguous()
        if self.cp_group is not None and self.cp_comm_type == "ring":
            from .cp import ring_attention_pos

            ctx = ring_attention_pos(qf, kf, v, self.cp_group, self.cp_rank,
                                     pos[:S].long(),
                                     self.cp_sharding == "zigzag")
        elif self.cp_group is not None:
            # kv all_gather (a2a head-scatter is incompatible with MLA's
            # per-token k_pe shared across heads)
            from .cp import cp_allgather_kv, cp_positions, masked_sdp

            kf_full = cp_allgather_kv(kf, self.cp_group)
            v_full = cp_allgather_kv(v, self.cp_group)
            cp = self.cp_size_
            kpos = torch.cat([cp_positions(S * cp, cp, r,
                                           self.cp_sharding == "zigzag",
                                           qf.device)
                              for r in range(cp)]).long()
            ctx = masked_sdp(qf, kf_full, v_full, pos[:S].long(), kpos)
        else:
            ctx = K.flash_attention(qf, kf, v, causal=True)
        return self.out_proj(ctx.reshape(B, S, H * dv))


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, dtype=torch.bfloat16, device=None,
                 layer_idx=0, ep_group=None, ep_size=1, tp_group=None,
                 tp_size=1, sp=False, cp_group=None, cp_size=1, cp_rank=0,
                 cp_comm_type="a2a", cp_sharding="contiguous", fp8=False):
        super().__init__()
        h = cfg.hidden_size
        self.tp_group = tp_group
        self.sp = sp
        self.input_norm = K.RMSNorm(h, dtype=dtype, device=device)
        if getattr(cfg, "attention_type", "gqa") == "mla":
            assert tp_size == 1, "MLA requires tp_size == 1 (simulator parity)"
            assert cp_size == 1 or cp_comm_type in ("all_gather", "ring"), \
                "MLA CP needs all_gather or ring (a2a head-scatter clashes " \
                "with the shared k_pe)"
            self.attention = MLAAttention(cfg, dtype=dtype, device=device,
                                          cp_group=cp_group, cp_rank=cp_rank,
                                          cp_comm_type=cp_comm_type,
                                          cp_size=cp_size,
                                          cp_sharding=cp_sharding)
        else:
            self.attention = GQAAttention(cfg, dtype=dtype, device=device,
                                          tp_group=tp_group, tp_size=tp_size,
                                          sp=sp, cp_group=cp_group,
                                          cp_size=cp_size, cp_rank=cp_rank,
                                          cp_comm_type=cp_comm_type,
                                          cp_sharding=cp_sharding, fp8=fp8)
        if sp:
            # SP norms see only the local seq shard: their weight grads
            # are partial sums and need a tp all_reduce (reducer handles
            # params flagged this way)
            self.input_norm.weight._needs_tp_grad_reduce = True
        self.pre_mlp_norm = K.RMSNorm(h, dtype=dtype, device=device)
        if sp:
            self.pre_mlp_norm.weight._needs_tp_grad_reduce = True
        assert cfg.use_swiglu
        self.use_moe = (cfg.model_type == "moe"
                        and layer_idx >= (cfg.dense_layers or 0))
        if self.use_moe:
            from .moe import MoEMLP

            assert tp_size == 1 or sp, \
                "MoE with tp requires sequence parallel (tp x ep)"
            self.moe_mlp = MoEMLP(cfg, dtype=dtype, device=device,
                                  ep_group=ep_group, ep_size=ep_size)
            if tp_size > 1 and sp:
                # the router (and any shared-expert fcs) see only this tp
                # rank's seq shard: their weight grads are partial sums
                # and need the tp all_reduce (like the SP norms)
                self.moe_mlp.router.weight._needs_tp_grad_reduce = True
                if getattr(self.moe_mlp, "shared", False):
                    self.moe_mlp.shared_fc1.weight._needs_tp_grad_reduce = True
                    self.moe_mlp.shared_fc2.weight._needs_tp_grad_reduce = True
        else:
            # Megatron MLP split: gate and up each sharded I/tp; fc2 row-
            # parallel over I/tp with one fwd all_reduce
            assert cfg.intermediate_size % tp_size == 0
            i_local = cfg.intermediate_size // tp_size
            Lin = K.FusedLinear
            if fp8:
                from ..kernels.fp8 import Fp8Linear as Lin
            self.fc1 = Lin(h, 2 * i_local, dtype=dtype, device=device)
            self.fc2 = Lin(i_local, h, dtype=dtype, device=device)
            if tp_size > 1:
                self.fc1.weight._is_tp_shard = True
                self.fc2.weight._is_tp_shard = True

    def forward(self, x, rope_cs, pos):
        from .tp import copy_to_tp, gather_seq, reduce_from_tp, scatter_seq

        # x: [B, S, H] (seq-sharded under SP)
        res = x
        y = self.input_norm(x)
        x = res + self.attention(y, rope_cs, pos)
        res = x
        y = self.pre_mlp_norm(x)
        if self.use_moe:
            y = self.moe_mlp(y)
        elif self.sp:
            y = self.fc2(K.swiglu(self.fc1(gather_seq(y, self.tp_group))))
            y = scatter_seq(y, self.tp_group)
        else:
            y = self.fc2(K.swiglu(self.fc1(copy_to_tp(y, self.tp_group))))
            y = reduce_from_tp(y, self.tp_group)
        return res + y


class LlamaForTraining(nn.Module):
    def __init__(self, cfg: ModelConfig, seq_len: int, dtype=torch.bfloat16,
                 rope_base=500000.0, device=None, ep_group=None, ep_size=1,
                 tp_group=None, tp_size=1, tp_rank=0, sp=False,
                 cp_group=None, cp_rank=0, cp_size=1, cp_comm_type="a2a",
                 cp_sharding="contiguous", fp8=False, recompute_layers=0):
        super().__init__()
        self.cfg = cfg
        self.seq_len = seq_len          # FULL sequence (rope cache size)
        self.cp_rank = cp_rank
        self.cp_size = cp_size
        self.cp_sharding = cp_sharding
        self.recompute_layers = recompute_layers
        self.tp_group = tp_group
        self.tp_size = tp_size
        self.sp = sp and tp_size > 1
        assert cfg.vocab_size % tp_size == 0
        self.vocab_local = cfg.vocab_size // tp_size
        self.vocab_start = tp_rank * self.vocab_local
        self.embedding = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                      dtype=dtype, device=device)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg, dtype, device, layer_idx=i,
                               ep_group=ep_group, ep_size=ep_size,
                               tp_group=tp_group, tp_size=tp_size,
                               sp=sp and tp_size > 1, cp_group=cp_group,
                               cp_size=cp_size, cp_rank=cp_rank,
                               cp_comm_type=cp_comm_type,
                               cp_sharding=cp_sharding, fp8=fp8)
             for i in range(cfg.layer_num)])
        self.final_norm = K.RMSNorm(cfg.hidden_size, dtype=dtype, device=device)
        if self.sp:
            self.final_norm.weight._needs_tp_grad_reduce = True
        self.lm_head = K.FusedLinear(cfg.hidden_size, self.vocab_local,
                                     dtype=dtype, device=device)
        if tp_size > 1:
            self.lm_head.weight._is_tp_shard = True
        rope_dim = (cfg.qk_pos_emb_head_dim
                    if getattr(cfg, "attention_type", "gqa") == "mla"
                    else cfg.head_size)
        cs = K.build_rope_cache(seq_len, rope_dim, base=rope_base,
                                device=device or "cpu")
        self.register_buffer("rope_cs", cs, persistent=False)

    def forward(self, tokens, labels):
        # tokens/labels: [B, S] int64 (S = the LOCAL seq slice under CP)
        B, S = tokens.shape
        if self.cp_size > 1:
            from .cp import cp_positions

            pos = cp_positions(S * self.cp_size, self.cp_size,
                               self.cp_rank,
                               self.cp_sharding == "zigzag",
                               tokens.device).repeat(B)
        else:
            pos = (torch.arange(S, device=tokens.device,
                                dtype=torch.int32).repeat(B))
        x = self.embedding(tokens)
        if self.sp:
            from .tp import slice_seq

            x = slice_seq(x, self.tp_group)
        for i, layer in enumerate(self.layers):
            if self.training and i < self.recompute_layers:
                # full-block activation recompute: only the block input is
                # held; the forward reruns during backward (simulator
                # full_block semantics, models/llm.py apply_recompute)
                import torch.utils.checkpoint as ckpt

                x = ckpt.checkpoint(layer, x, self.rope_cs, pos,
                                    use_reentrant=False)
            else:
                x = layer(x, self.rope_cs, pos)
        x = self.final_norm(x)
        if self.tp_size > 1:
            from .tp import copy_to_tp, gather_seq, vocab_parallel_ce

            if self.sp:
                logits = self.lm_head(gather_seq(x, self.tp_group))
            else:
                logits = self.lm_head(copy_to_tp(x, self.tp_group))
            loss = vocab_parallel_ce(logits.reshape(B * S, -1),
                                     labels.reshape(-1), self.tp_group,
                                     self.vocab_start)
        else:
            logits = self.lm_head(x)
            loss = K.fused_cross_entropy(
                logits.reshape(B * S, -1), labels.reshape(-1))
        return loss.mean()

    def num_params(self):
        return sum(p.numel() for p in self.parameters())
