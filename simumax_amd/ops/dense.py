"""Dense transformer operator library (L3).

Parity target: simumax/core/transformer/dense_module.py (Embedding,
LinearCol, LinearRow, LayerNorm, CoreAttention, MLACoreAttention,
RotaryEmbedding, Swiglu, Gelu, ParallelCE, Float8Quantizer,
Attention, MLAAttention, MLP) — re-derived for MI355X:

* GEMM-shaped leaves emit the exact shape-key strings the HIP/hipBLASLt
  calibration harness measures (module LinearBase).
* Memory-bound leaves (norm/rope/swiglu/ce) are priced by bandwidth-table
  keys matching the shipped CDNA4 HIP kernels.
* Every collective is a CommEvent priced by the RCCL-over-xGMI model.

Activation-cache convention (drives the <1% peak-memory target): each leaf
caches exactly the tensors the matching autograd op in the in-repo
Megatron-ROCm reference trainer (simumax_amd/train) saves:
  - linears: their (sharded, under SP) input
  - fused RMSNorm: input + rstd(fp32)
  - RoPE: nothing (linear in x; cos/sin tables are persistent)
  - flash SDP: q,k,v + softmax LSE (output O is cached by the out-proj as
    its input — counted once)
  - swiglu: its input (both halves of fc1 output)
  - CE: logits + labels
"""

from __future__ import annotations

from ..core.module import CommEvent, LinearBase, MetaModule
from ..core.records import InputOutputInfo
from ..core.tensor import TensorSize

FP32 = 4
LSE_BYTES = 4  # softmax log-sum-exp, fp32 per (b, head, s)


def _state_div(strategy, is_expert=False):
    """ZeRO-1 shards optimizer state over the (e)dp group."""
    if strategy.zero_state >= 1:
        return strategy.edp_size if is_expert else strategy.dp_size * strategy.cp_size
    return 1


class ParamMixin:
    """Weight/grad/optimizer-state accounting for modules that own params."""

    def add_param(self, info, numel, is_expert=False):
        s = self.strategy
        numel = int(numel)
        weight = numel * self.element_size
        grad = numel * (4 if s.use_fp32_accum_grad else self.element_size)
        # Megatron mixed-precision Adam: fp32 master + exp_avg + exp_avg_sq
        state = numel * 12 / _state_div(s, is_expert)
        if s.dtype == "fp32":
            state = numel * 8 / _state_div(s, is_expert)  # no separate master
        if is_expert:
            info.moe_weight_bytes += weight
            info.moe_grad_bytes += grad
            info.moe_state_bytes += state
        else:
            info.dense_weight_bytes += weight
            info.dense_grad_bytes += grad
            info.dense_state_bytes += state


# ==========================================================================
# leaves
# ==========================================================================
class Embedding(MetaModule, ParamMixin):
    """Vocab-parallel embedding. TP: all_reduce after lookup; SP:
    reduce_scatter (output sequence-sharded). Reference:
    dense_module.py:18-194."""

    def __init__(self, vocab_size, hidden_size, strategy, system, name="embedding"):
        super().__init__(strategy, system, name)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size

    def create_output_info(self, input_info):
        ids = input_info.tensors[0]  # [B, S/cp] int64
        b, s = ids.shape[0], ids.shape[1]
        if self.strategy.enable_sequence_parallel:
            s //= self.strategy.tp_size
        return InputOutputInfo([TensorSize([b, s, self.hidden_size], self.strategy.dtype)])

    def _leaf_model_info(self, info):
        self.add_param(info, self.vocab_size // self.strategy.tp_size * self.hidden_size)

    def _leaf_act_info(self, info):
        # bwd scatter-add needs the token ids
        info.activation_mem_cache = self.input_info.tensors[0].mem_bytes()
        # autograd materializes a DENSE bf16 weight grad, converted to fp32
        # for main_grad accumulation (2+4 B/elem transient)
        local_numel = self.vocab_size // self.strategy.tp_size * self.hidden_size
        info.bwd_peak_mem_no_cache = local_numel * (self.element_size + 4)

    def _leaf_compute_info(self, info):
        out_full = self.output_info.first.mem_bytes()
        if self.strategy.enable_sequence_parallel:
            out_full *= self.strategy.tp_size
        info.fwd_accessed_mem = out_full + self.input_info.tensors[0].mem_bytes()
        # bwd: scatter-add of grad into weight grad buffer
        local_w = self.vocab_size // self.strategy.tp_size * self.hidden_size
        info.bwd_grad_w_accessed_mem = out_full + (
            local_w * self.grad_element_size
        )
        if self.strategy.use_fused_grad_accumulation:
            # post-accumulate hook: p.grad.float() temp (r2+w4) then
            # main_grad.add_ (r4+r4+w4) = 18 B/elem, per microbatch
            info.bwd_grad_w_extra_mem = local_w * 18

    def _leaf_intra_net_info(self):
        tp = self.strategy.tp_size
        if tp <= 1:
            return
        full = self.output_info.first.mem_bytes()
        if self.strategy.enable_sequence_parallel:
            full *= tp
            self.add_comm("fwd", "reduce_scatter", full, tp, "tp")
            self.add_comm("bwd_w", "all_gather", full, tp, "tp")
        else:
            self.add_comm("fwd", "all_reduce", full, tp, "tp")


class LinearCol(LinearBase, ParamMixin):
    """Column-parallel GEMM (input full H, output sharded N/tp).
    SP: all_gather fwd + reduce_scatter bwd_act + all_gather bwd_w.
    TP (no SP): all_reduce in bwd_act. Reference: dense_module.py:195-510."""

    fwd_op = "matmul"
    bwd_act_op = "matmul"
    bwd_w_op = "matmul"

    def __init__(self, input_size, output_size, strategy, system, name="linear_col",
                 with_tp_comm=True, is_expert=False, sp_gather=None):
        # input_size/output_size are the LOCAL gemm K and N (already /tp)
        super().__init__(input_size, output_size, strategy, system, name)
        self.with_tp_comm = with_tp_comm
        self.is_expert = is_expert
        tp = strategy.etp_size if is_expert else strategy.tp_size
        self.tp = tp if with_tp_comm else 1
        self.sp = (strategy.enable_sequence_parallel and self.tp > 1) if sp_gather is None else sp_gather
        if strategy.fp8:
            self.fwd_op = self.bwd_act_op = self.bwd_w_op = "fp8_matmul"
            self.fwd_extra_op = "fp8_quant"
            self.bwd_act_extra_op = "fp8_quant"
            self.bwd_w_extra_op = "fp8_quant"

    @property
    def micro_input_tensor(self):
        t = self.input_info.tensors[0]
        if self.sp:
            t = t.scale_dim(-2 if t.ndim == 3 else 0, self.tp, 1)
        return t

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        assert t.shape[-1] == self.input_size, (
            f"{self.full_name}: input {t.shape} vs K {self.input_size}"
        )
        s_dim = t.ndim - 2
        shape = list(t.shape)
        if self.sp:
            shape[s_dim] *= self.tp
        shape[-1] = self.output_size
        return InputOutputInfo([TensorSize(shape, self.strategy.dtype)])

    def _leaf_model_info(self, info):
        self.add_param(info, self.input_size * self.output_size, self.is_expert)
        if self.strategy.fp8:
            # per-step weight-quant cache: wq + column-major copy, 1 B/elem
            # each (kernels/fp8.py Fp8Linear._weight_quant)
            nw = self.input_size * self.output_size
            info.cache_bytes += 2 * nw

    def _leaf_act_info(self, info):
        # caches the (sharded under SP) input; the gathered copy is transient
        info.activation_mem_cache = self.input_info.first.mem_bytes()
        if self.strategy.fp8:
            # the fp8 path saves the 1 B/elem transposed fp8 input instead
            # of the bf16 input
            info.activation_mem_cache //= 2
        # bwd transient: the freshly allocated placeholder wgrad that
        # AccumulateGrad steals (freed by the post-accumulate hook)
        info.bwd_peak_mem_no_cache = (
            self.input_size * self.output_size * self.element_size)
        if self.sp:
            gathered = self.input_info.first.mem_bytes() * self.tp
            info.fwd_peak_mem_no_cache = max(info.fwd_peak_mem_no_cache, gathered)
            info.bwd_peak_mem_no_cache += gathered

    def _leaf_compute_info(self, info):
        k = self.get_gemm_bmnk("fwd")
        flops = 2 * k["B"] * k["M"] * k["K"] * k["N"]
        info.fwd_flops = flops
        info.bwd_grad_act_flops = flops
        info.bwd_grad_w_flops = flops
        e = self.element_size
        in_b = k["B"] * k["M"] * k["K"] * e
        w_b = k["K"] * k["N"] * e
        out_b = k["B"] * k["M"] * k["N"] * e
        info.fwd_accessed_mem = in_b + w_b + out_b
        info.bwd_grad_act_accessed_mem = out_b + w_b + in_b
        info.bwd_grad_w_accessed_mem = out_b + in_b + k["K"] * k["N"] * self.grad_element_size
        if self.strategy.fp8:
            # trainer fp8 path (kernels/fp8.py): fused cast_transpose of
            # the input (fwd) and of dy (bwd) — read bf16, write BOTH fp8
            # images = 2x the bf16 bytes each; wgrad lands in a bf16
            # p.grad that the hook adds into the fp32 main_grad
            # (2+4+4 B/elem); the per-step weight quant amortizes over
            # the microbatches
            nw = k["K"] * k["N"]
            mbc = max(1, self.strategy.micro_batch_num)
            info.fwd_extra_mem += 2 * in_b + 2 * w_b // mbc
            info.bwd_grad_act_extra_mem += 2 * out_b
            info.bwd_grad_w_extra_mem += 10 * nw

    def _leaf_intra_net_info(self):
        if self.tp <= 1:
            return
        stage = "etp" if self.is_expert else "tp"
        full_in = self.micro_input_tensor.mem_bytes()
        if self.sp:
            self.add_comm("fwd", "all_gather", full_in, self.tp, stage)
            self.add_comm("bwd_act", "reduce_scatter", full_in, self.tp, stage)
            # wgrad re-gathers the input (Megatron sequence_parallel),
            # overlapped with dgrad GEMM on a separate stream
            self.add_comm("bwd_w", "all_gather", full_in, self.tp, stage,
                          overlap=self.strategy.overlap_grad_reduce)
        else:
            self.add_comm("bwd_act", "all_reduce", full_in, self.tp, stage)


class LinearRow(LinearBase, ParamMixin):
    """Row-parallel GEMM (input sharded K/tp, output full N then reduced).
    SP: reduce_scatter fwd + all_gather bwd_act; TP: all_reduce fwd.
    Reference: dense_module.py:511-783."""

    fwd_op = "matmul"
    bwd_act_op = "matmul"
    bwd_w_op = "matmul"

    def __init__(self, input_size, output_size, strategy, system, name="linear_row",
                 with_tp_comm=True, is_expert=False, sp_scatter=None):
        super().__init__(input_size, output_size, strategy, system, name)
        self.with_tp_comm = with_tp_comm
        self.is_expert = is_expert
        tp = strategy.etp_size if is_expert else strategy.tp_size
        self.tp = tp if with_tp_comm else 1
        self.sp = (strategy.enable_sequence_parallel and self.tp > 1) if sp_scatter is None else sp_scatter
        if strategy.fp8:
            self.fwd_op = self.bwd_act_op = self.bwd_w_op = "fp8_matmul"
            self.fwd_extra_op = "fp8_quant"
            self.bwd_act_extra_op = "fp8_quant"
            self.bwd_w_extra_op = "fp8_quant"

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        assert t.shape[-1] == self.input_size
        shape = list(t.shape)
        shape[-1] = self.output_size
        if self.sp:
            shape[t.ndim - 2] //= self.tp
        return InputOutputInfo([TensorSize(shape, self.strategy.dtype)])

    def _leaf_model_info(self, info):
        self.add_param(info, self.input_size * self.output_size, self.is_expert)
        if self.strategy.fp8:
            nw = self.input_size * self.output_size
            info.cache_bytes += 2 * nw

    def _leaf_act_info(self, info):
        info.activation_mem_cache = self.input_info.first.mem_bytes()
        if self.strategy.fp8:
            info.activation_mem_cache //= 2
        info.bwd_peak_mem_no_cache = (
            self.input_size * self.output_size * self.element_size)
        # full (pre-scatter) output is transient under SP
        if self.sp:
            full_out = self.output_info.first.mem_bytes() * self.tp
            info.fwd_peak_mem_no_cache = full_out

    def _leaf_compute_info(self, info):
        k = self.get_gemm_bmnk("fwd")
        flops = 2 * k["B"] * k["M"] * k["K"] * k["N"]
        info.fwd_flops = flops
        info.bwd_grad_act_flops = flops
        info.bwd_grad_w_flops = flops
        e = self.element_size
        in_b = k["B"] * k["M"] * k["K"] * e
        w_b = k["K"] * k["N"] * e
        out_b = k["B"] * k["M"] * k["N"] * e
        info.fwd_accessed_mem = in_b + w_b + out_b
        info.bwd_grad_act_accessed_mem = out_b + w_b + in_b
        info.bwd_grad_w_accessed_mem = out_b + in_b + k["K"] * k["N"] * self.grad_element_size
        if self.strategy.fp8:
            # trainer fp8 path (kernels/fp8.py): fused cast_transpose of
            # the input (fwd) and of dy (bwd) — read bf16, write BOTH fp8
            # images = 2x the bf16 bytes each; wgrad lands in a bf16
            # p.grad that the hook adds into the fp32 main_grad
            # (2+4+4 B/elem); the per-step weight quant amortizes over
            # the microbatches
            nw = k["K"] * k["N"]
            mbc = max(1, self.strategy.micro_batch_num)
            info.fwd_extra_mem += 2 * in_b + 2 * w_b // mbc
            info.bwd_grad_act_extra_mem += 2 * out_b
            info.bwd_grad_w_extra_mem += 10 * nw

    def _leaf_intra_net_info(self):
        if self.tp <= 1:
            return
        stage = "etp" if self.is_expert else "tp"
        t = self.input_info.tensors[0]
        full_out_numel = 1
        for i, d in enumerate(t.shape[:-1]):
            full_out_numel *= d
        full_out = full_out_numel * self.output_size * self.element_size
        if self.sp:
            self.add_comm("fwd", "reduce_scatter", full_out, self.tp, stage)
            self.add_comm("bwd_act", "all_gather", full_out, self.tp, stage)
        else:
            self.add_comm("fwd", "all_reduce", full_out, self.tp, stage)


class LayerNorm(MetaModule, ParamMixin):
    """RMSNorm/LayerNorm. Fused = the shipped CDNA4 HIP kernel (one pass
    read-in/write-out); unfused = 2 extra passes. Reference:
    dense_module.py:784-995."""

    # measured stream efficiencies of the shipped CDNA4 kernels
    fwd_mem_op = "rmsnorm_fwd"
    bwd_act_mem_op = "rmsnorm_bwd"

    def __init__(self, hidden_size, strategy, system, name="norm", norm_type="rms"):
        super().__init__(strategy, system, name)
        self.hidden_size = hidden_size
        self.norm_type = norm_type

    def _leaf_model_info(self, info):
        numel = self.hidden_size * (2 if self.norm_type == "layernorm" else 1)
        self.add_param(info, numel)

    def _leaf_act_info(self, info):
        t = self.input_info.first
        rows = t.numel() // t.shape[-1]
        info.activation_mem_cache = t.mem_bytes() + rows * FP32  # input + rstd

    def _leaf_compute_info(self, info):
        t = self.input_info.first
        n = t.numel()
        b = t.mem_bytes()
        info.fwd_flops = 4 * n
        info.bwd_grad_act_flops = 8 * n
        passes = 1 if self.strategy.use_fused_norm else 2
        info.fwd_accessed_mem = 2 * b * passes
        # bwd reads dout + input, writes din (+weight-grad reduction)
        info.bwd_grad_act_accessed_mem = 3 * b * passes
        info.bwd_grad_w_accessed_mem = 0
        if self.strategy.use_fused_grad_accumulation:
            # post-accumulate hook cast+add of the norm weight grad
            info.bwd_grad_w_extra_mem = self.hidden_size * 18


class RotaryEmbedding(MetaModule):
    """RoPE on q,k. Linear in x: bwd needs only cos/sin tables, so no cache.
    Fused CDNA4 kernel reads+writes q,k once. Reference:
    dense_module.py:1806-1873."""

    fwd_mem_op = "rope"
    bwd_act_mem_op = "rope"

    def __init__(self, strategy, system, name="rope"):
        super().__init__(strategy, system, name)

    def _leaf_compute_info(self, info):
        b = self.input_info.total_bytes()
        n = sum(t.numel() for t in self.input_info.tensors)
        info.fwd_flops = 3 * n
        info.bwd_grad_act_flops = 3 * n
        info.fwd_accessed_mem = 2 * b
        info.bwd_grad_act_accessed_mem = 2 * b


class CoreAttention(MetaModule):
    """Scaled-dot-product attention (flash by default). Inputs q,k,v
    (possibly GQA). CP handled via a2a head-scatter/seq-gather (Ulysses) or
    kv all_gather. Reference: dense_module.py:1061-1605.

    Op keys sdp_fwd/sdp_bwd index the per-shape efficiency table measured by
    the CDNA4 flash-attention HIP kernel harness."""

    fwd_op = "sdp_fwd"
    bwd_act_op = "sdp_bwd"

    def __init__(self, head_num, kv_head_num, qk_head_dim, v_head_dim,
                 strategy, system, name="core_attn", qkv_contiguous=True):
        super().__init__(strategy, system, name)
        # local (post-TP, post-CP-a2a) head counts
        self.head_num = head_num
        self.kv_head_num = kv_head_num
        self.qk_head_dim = qk_head_dim
        self.v_head_dim = v_head_dim
        self.qkv_contiguous = qkv_contiguous
        assert strategy.use_flash_sdp or strategy.use_math_sdp
        self.use_flash = strategy.use_flash_sdp
        self.cp = strategy.cp_size
        self.cp_a2a = strategy.cp_comm_type == "a2a"
        self.cp_mode = strategy.cp_comm_type if self.cp > 1 else None

    # ---- geometry ------------------------------------------------------
    @property
    def _bsd(self):
        q = self.input_info.tensors[0]  # [B, S_local, Hq*Dq]
        b, s = q.shape[0], q.shape[1]
        return b, s

    @property
    def full_seq(self):
        b, s = self._bsd
        return s * self.cp

    @property
    def q_seq(self):
        """query rows per rank: a2a gathers the full sequence (with H/cp
        heads); all_gather/ring keep q seq-sharded with full heads."""
        b, s = self._bsd
        return s * self.cp if self.cp_a2a else s

    @property
    def sdp_head_num(self):
        """heads seen by one rank's SDP kernel (after CP a2a head-scatter)"""
        if self.cp > 1 and self.cp_a2a:
            assert self.head_num % self.cp == 0
            return self.head_num // self.cp
        return self.head_num

    @property
    def sdp_kv_head_num(self):
        if self.cp > 1 and self.cp_a2a:
            return max(1, self.kv_head_num // self.cp)
        return self.kv_head_num

    def get_input_shapes_desc(self, stage):
        if stage not in ("fwd", "bwd_grad_act"):
            return ""
        b, _ = self._bsd
        return (
            f"batch={b}, seq_len={self.full_seq}, head_num={self.sdp_head_num}, "
            f"kv_head_num={self.sdp_kv_head_num}, qk_head_dim={self.qk_head_dim}, "
            f"v_head_dim={self.v_head_dim}, qkv_contiguous={self.qkv_contiguous}"
        )

    def create_output_info(self, input_info):
        b, s = input_info.tensors[0].shape[0], input_info.tensors[0].shape[1]
        return InputOutputInfo(
            [TensorSize([b, s, self.head_num * self.v_head_dim], self.strategy.dtype)]
        )

    def _sdp_bytes(self):
        """Kernel-visible tensor bytes per rank: q/o/lse span q_seq rows,
        k/v span the full (gathered or circulated) sequence."""
        b, _ = self._bsd
        s = self.full_seq
        e = self.element_size
        q = b * self.q_seq * self.sdp_head_num * self.qk_head_dim * e
        k = b * s * self.sdp_kv_head_num * self.qk_head_dim * e
        v = b * s * self.sdp_kv_head_num * self.v_head_dim * e
        o = b * self.q_seq * self.sdp_head_num * self.v_head_dim * e
        lse = b * self.q_seq * self.sdp_head_num * LSE_BYTES
        return q, k, v, o, lse

    def _leaf_act_info(self, info):
        q, k, v, o, lse = self._sdp_bytes()
        if self.cp > 1 and self.cp_a2a:
            # Ulysses a2a transient buffers (reference parity:
            # dense_module.py:1259-1338): async_cp posts q/k/v a2a together
            # (input + output buffers all live), sync_cp moves one at a time
            if self.strategy.cp_a2a_mode == "async_cp":
                info.fwd_peak_mem_no_cache = q + k + v  # post-a2a copies
                info.bwd_peak_mem_no_cache = q + k + v + o
            else:
                info.fwd_peak_mem_no_cache = 2 * max(q, k, v)
                info.bwd_peak_mem_no_cache = 2 * max(q, k, v)
            if self.strategy.te_cp_a2a_saves_pre_posta2a_output:
                # bwd re-uses the saved pre-PostA2A O: only dO moves back
                info.bwd_peak_mem_no_cache = max(
                    0.0, info.bwd_peak_mem_no_cache - o)
        if self.use_flash:
            info.activation_mem_cache = q + k + v + lse
            if (self.cp > 1 and self.cp_a2a
                    and self.strategy.te_cp_a2a_saves_pre_posta2a_output):
                info.activation_mem_cache += o
        else:
            b, _ = self._bsd
            s = self.full_seq
            scores = b * self.sdp_head_num * self.q_seq * s * self.element_size
            info.activation_mem_cache = q + k + v + scores
            info.fwd_peak_mem_no_cache = scores
        if self.cp_mode == "ring":
            # blockwise ring (flash assumption): each rank caches only its
            # OWN K/V shard — blocks re-circulate in backward (the 2x bwd
            # p2p is priced) — and holds at most two in-flight blocks
            info.activation_mem_cache -= (k + v) * (self.cp - 1) / self.cp
            info.fwd_peak_mem_no_cache = max(
                info.fwd_peak_mem_no_cache, 2 * (k + v) / self.cp)

    def _leaf_compute_info(self, info):
        b, _ = self._bsd
        s = self.full_seq
        r = self.strategy.attention_sparse_ratio
        if self.cp_mode in ("all_gather", "ring"):
            if self.strategy.cp_sharding == "zigzag":
                # zigzag chunk pairing balances the causal load exactly:
                # every rank computes full_causal/cp
                sparse = 1.0 - r
            else:
                # contiguous shards: the WORST (last) rank sees a causal
                # share of 1 - r/cp of the keys; the step is bounded by it
                sparse = 1.0 - r / self.cp
        else:
            sparse = 1.0 - r
        qk = 2 * b * self.sdp_head_num * self.q_seq * s * self.qk_head_dim
        pv = 2 * b * self.sdp_head_num * self.q_seq * s * self.v_head_dim
        info.fwd_flops = (qk + pv) * sparse
        extra = 1 if self.use_flash else 0  # flash bwd recomputes QK^T
        info.bwd_grad_act_flops = (2 * qk + 2 * pv + extra * qk) * sparse
        q, k, v, o, lse = self._sdp_bytes()
        if self.use_flash:
            info.fwd_accessed_mem = q + k + v + o + lse
            info.bwd_grad_act_accessed_mem = 2 * (q + k + v) + 2 * o + lse
            if self.qkv_contiguous:
                # trainer parity: q/k/v are strided views of the fused qkv
                # GEMM output and are materialized contiguous before the
                # kernel (fwd); backward of the split cats dq/dk/dv into
                # one dqkv buffer. Both are separate copy kernels.
                info.fwd_extra_mem = 2 * (q + k + v)
                info.bwd_grad_act_extra_mem = 2 * (q + k + v)
        else:
            scores = b * self.sdp_head_num * self.q_seq * s * self.element_size
            info.fwd_accessed_mem = q + k + v + o + 4 * scores
            info.bwd_grad_act_accessed_mem = 2 * (q + k + v + o) + 6 * scores

    def _leaf_intra_net_info(self):
        if self.cp <= 1:
            return
        q, k, v, o, _ = self._sdp_bytes()
        if self.cp_a2a:
            # Ulysses: scatter heads / gather sequence — q,k,v pre + o post
            for t in (q, k, v):
                self.add_comm("fwd", "all2all", t, self.cp, "cp")
                self.add_comm("bwd_act", "all2all", t, self.cp, "cp")
            self.add_comm("fwd", "all2all", o, self.cp, "cp")
            self.add_comm("bwd_act", "all2all", o, self.cp, "cp")
        elif self.strategy.cp_comm_type == "ring":
            # ring attention (extension — absent in the reference): cp-1
            # p2p hops of one K/V block each way; backward re-circulates
            # the blocks and accumulates dK/dV over the reverse ring. On
            # xGMI each hop is an independent point-to-point link, so the
            # hops are priced individually (per-hop latency counts).
            blk = (k + v) / self.cp
            for _ in range(self.cp - 1):
                self.add_comm("fwd", "p2p", blk, 2, "cp")
                self.add_comm("bwd_act", "p2p", 2 * blk, 2, "cp")
        else:
            # kv all_gather mode
            self.add_comm("fwd", "all_gather", k + v, self.cp, "cp")
            self.add_comm("bwd_act", "all_gather", k + v, self.cp, "cp")
            self.add_comm("bwd_act", "reduce_scatter", k + v, self.cp, "cp")


class MLACoreAttention(CoreAttention):
    """MLA-shaped SDP (asymmetric 192/128 head dims). Reference:
    dense_module.py:1606-1805."""

    def __init__(self, head_num, qk_head_dim, v_head_dim, strategy, system,
                 name="mla_core_attn"):
        super().__init__(head_num, head_num, qk_head_dim, v_head_dim,
                         strategy, system, name, qkv_contiguous=False)

    def _leaf_compute_info(self, info):
        super()._leaf_compute_info(info)
        # trainer parity: q/k are materialized by concatenating the nope
        # and RoPE sub-dims (fwd) and split back in backward
        q, k, v, o, _ = self._sdp_bytes()
        info.fwd_extra_mem = 2 * (q + k) + 2 * v  # cats + v contiguous
        info.bwd_grad_act_extra_mem = 2 * (q + k)


class Swiglu(MetaModule):
    """Fused SwiGLU: y = silu(x1) * x2 over the fc1 output's two halves.
    Caches its input (fc2 caches y itself). Reference:
    dense_module.py:1874-2096."""

    fwd_mem_op = "swiglu"
    bwd_act_mem_op = "swiglu_bwd"

    def __init__(self, strategy, system, name="swiglu", weighted=False):
        super().__init__(strategy, system, name)
        self.weighted = weighted  # MoE dispatch_probs fused multiply

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        return InputOutputInfo([t.scale_dim(-1, 1, 2)])

    def _leaf_act_info(self, info):
        info.activation_mem_cache = self.input_info.first.mem_bytes()

    def _leaf_compute_info(self, info):
        in_b = self.input_info.first.mem_bytes()
        out_b = self.output_info.first.mem_bytes()
        n = self.input_info.first.numel()
        info.fwd_flops = 4 * n
        info.bwd_grad_act_flops = 6 * n
        passes = 1 if self.strategy.use_fused_swiglu else 2
        info.fwd_accessed_mem = (in_b + out_b) * passes
        info.bwd_grad_act_accessed_mem = (in_b + out_b + in_b) * passes


class Gelu(MetaModule):
    def create_output_info(self, input_info):
        return input_info.clone()

    def _leaf_act_info(self, info):
        info.activation_mem_cache = self.input_info.first.mem_bytes()

    def _leaf_compute_info(self, info):
        b = self.input_info.first.mem_bytes()
        n = self.input_info.first.numel()
        info.fwd_flops = 8 * n
        info.bwd_grad_act_flops = 10 * n
        info.fwd_accessed_mem = 2 * b
        info.bwd_grad_act_accessed_mem = 3 * b


class ParallelCE(MetaModule):
    """Vocab-parallel cross entropy over [B,S,V/tp] logits.
    2 (fused) or 3 TP all_reduces of [B,S] fp32. Bandwidth keys
    ce / ce_fusion index the fused-CE CDNA4 kernel's measured efficiency.
    Reference: dense_module.py:2097-2364."""

    def __init__(self, strategy, system, name="parallel_ce"):
        super().__init__(strategy, system, name)

    @property
    def _bw_key(self):
        return "ce_fusion" if self.strategy.cross_entropy_loss_fusion else "ce"

    def create_output_info(self, input_info):
        t = input_info.tensors[0]
        b, s = t.shape[0], t.shape[1]
        return InputOutputInfo([TensorSize([b, s], "fp32")])

    def _leaf_act_info(self, info):
        t = self.input_info.first
        b, s = t.shape[0], t.shape[1]
        # caches logits (for dlogits) + labels + per-token stats
        info.activation_mem_cache = t.mem_bytes() + b * s * (8 + 2 * FP32)
        if not self.strategy.cross_entropy_loss_fusion:
            # unfused keeps fp32 softmax copy transiently
            info.fwd_peak_mem_no_cache = t.numel() * FP32
        # bwd materializes dlogits (same size as logits)
        info.bwd_peak_mem_no_cache = t.mem_bytes()

    def _leaf_compute_info(self, info):
        t = self.input_info.first
        b = t.mem_bytes()
        n = t.numel()
        info.fwd_flops = 5 * n
        info.bwd_grad_act_flops = 3 * n
        if self.strategy.cross_entropy_loss_fusion:
            info.fwd_accessed_mem = b  # one fused read pass, O(BS) writes
            info.bwd_grad_act_accessed_mem = 2 * b
        else:
            info.fwd_accessed_mem = 2 * b + n * FP32
            info.bwd_grad_act_accessed_mem = 2 * b + n * FP32

    def _comp_leaf_cost_info(self):
        # memory-bound op priced by its dedicated bandwidth key
        sysc = self.system
        comp = self._compute_info
        ci = self._cost_info
        ci.fwd_compute_time = sysc.compute_end2end_time(
            sysc.compute_op_accuracy_time("default", comp.fwd_flops),
            sysc.compute_mem_access_time(self._bw_key, comp.fwd_accessed_mem),
        )
        ci.bwd_grad_act_time = sysc.compute_end2end_time(
            sysc.compute_op_accuracy_time("default", comp.bwd_grad_act_flops),
            sysc.compute_mem_access_time(self._bw_key, comp.bwd_grad_act_accessed_mem),
        )
        ci.bwd_grad_w_time = 0.0
        self._price_comm()
        ci.recompute_compute_time = ci.fwd_compute_time if self.enable_recompute else 0.0

    def _leaf_intra_net_info(self):
        tp = self.strategy.tp_size
        if tp <= 1:
            return
        t = self.input_info.first
        b, s = t.shape[0], t.shape[1]
        bs_fp32 = b * s * FP32
        n_reduces = 2 if self.strategy.cross_entropy_loss_fusion else 3
        for _ in range(n_reduces):
            self.add_comm("fwd", "all_reduce", bs_fp32, tp, "tp")


class Float8Quantizer(MetaModule):
    """Quantize-to-fp8 leaf (amax reduce + cast). Reference:
    dense_module.py:2365-2453."""

    def create_output_info(self, input_info):
        return InputOutputInfo([t.to("fp8") for t in input_info.tensors])

    def _leaf_act_info(self, info):
        # fp8 copy (row+col major) kept for bwd GEMMs
        info.activation_mem_cache = 2 * sum(t.numel() for t in self.input_info.tensors)

    def _leaf_compute_info(self, info):
        in_b = self.input_info.total_bytes()
        out_b = sum(t.numel() for t in self.input_info.tensors)
        info.fwd_accessed_mem = in_b + out_b
        info.bwd_grad_act_accessed_mem = in_b + out_b


class QuantizedColLinear(MetaModule):
    """Float8Quantizer -> LinearCol with fp8 op keys (reference parity:
    dense_module.py:2365-2453). The quantizer's fp8 copies are the extra
    activation the fp8 path caches."""

    def __init__(self, input_size, output_size, strategy, system,
                 name="quant_linear_col", **kw):
        super().__init__(strategy, system, name)
        assert strategy.fp8, "QuantizedColLinear requires strategy.fp8"
        self.quantizer = Float8Quantizer(strategy, system, "quantize")
        self.linear = LinearCol(input_size, output_size, strategy, system,
                                "linear", **kw)

    def forward(self, input_info):
        q = self.quantizer(input_info, self.path_debug_context)
        # the GEMM consumes the fp8 tensor but keys/caches track bf16 I/O
        return self.linear(InputOutputInfo([t.to(self.strategy.dtype)
                                            for t in q.tensors]),
                           self.path_debug_context)


class QuantizedRowLinear(MetaModule):
    def __init__(self, input_size, output_size, strategy, system,
                 name="quant_linear_row", **kw):
        super().__init__(strategy, system, name)
        assert strategy.fp8, "QuantizedRowLinear requires strategy.fp8"
        self.quantizer = Float8Quantizer(strategy, system, "quantize")
        self.linear = LinearRow(input_size, output_size, strategy, system,
                                "linear", **kw)

    def forward(self, input_info):
        q = self.quantizer(input_info, self.path_debug_context)
        return self.linear(InputOutputInfo([t.to(self.strategy.dtype)
                                            for t in q.tensors]),
                           self.path_debug_context)


class Add(MetaModule):
    """Residual add: 2 reads + 1 write, nothing cached (linear)."""

    def create_output_info(self, input_info):
        return InputOutputInfo([input_info.tensors[0].clone()])

    def _leaf_compute_info(self, info):
        b = self.input_info.tensors[0].mem_bytes()
        info.fwd_flops = self.input_info.tensors[0].numel()
        info.fwd_accessed_mem = 3 * b
        # bwd of add passes grads through, but the residual FAN-IN (the
        # skip tensor is consumed twice) makes autograd accumulate two
        # gradient paths: one elementwise add of the grad tensor
        info.bwd_grad_act_extra_mem = 3 * b


# ==========================================================================
# composites
# ==========================================================================
class Attention(MetaModule):
    """GQA attention: qkv LinearCol -> RoPE -> CoreAttention -> out LinearRow.
    Reference: dense_module.py:2454-2568."""

    def __init__(self, model_cfg, strategy, system, name="attention"):
        super().__init__(strategy, system, name)
        m = model_cfg
        tp = strategy.tp_size
        assert m.head_num % tp == 0, f"head_num {m.head_num} % tp {tp}"
        self.heads_local = m.head_num // tp
        self.kv_heads_local = max(1, m.kv_head_num // tp)
        self.head_size = m.head_size
        qkv_out = (self.heads_local + 2 * self.kv_heads_local) * m.head_size
        self.qkv_proj = LinearCol(m.hidden_size, qkv_out, strategy, system, "qkv_proj")
        self.rope = RotaryEmbedding(strategy, system)
        self.core_attn = CoreAttention(
            self.heads_local, self.kv_heads_local, m.head_size, m.head_size,
            strategy, system,
        )
        self.out_proj = LinearRow(self.heads_local * m.head_size, m.hidden_size,
                                  strategy, system, "out_proj")

    def forward(self, input_info):
        qkv = self.qkv_proj(input_info, self.path_debug_context)
        t = qkv.tensors[0]
        b, s = t.shape[0], t.shape[1]
        q = TensorSize([b, s, self.heads_local * self.head_size], t.dtype)
        k = TensorSize([b, s, self.kv_heads_local * self.head_size], t.dtype)
        v = TensorSize([b, s, self.kv_heads_local * self.head_size], t.dtype)
        self.rope(InputOutputInfo([q, k]), self.path_debug_context)
        ctx = self.core_attn(InputOutputInfo([q, k, v]), self.path_debug_context)
        return self.out_proj(ctx, self.path_debug_context)

    def apply_recompute(self):
        cfg = self.strategy.parse_attention_recompute()
        if cfg.recompute_qkv:
            self.qkv_proj.set_recompute()
            self.rope.set_recompute()
        if cfg.recompute_core_attn:
            self.core_attn.set_recompute()
        if cfg.recompute_out_proj:
            self.out_proj.set_recompute()


class MLAAttention(MetaModule):
    """DeepSeek MLA attention: q down/up (+norm), kv down/up (+norm), RoPE,
    asymmetric-head SDP, out proj. Asserts tp==1 (reference parity:
    dense_module.py:2583). Reference: dense_module.py:2569-2887."""

    def __init__(self, model_cfg, strategy, system, name="mla_attention"):
        super().__init__(strategy, system, name)
        assert strategy.tp_size == 1, "MLA attention requires tp_size == 1"
        m = model_cfg
        self.m = m
        h = m.hidden_size
        qk_total = m.qk_head_dim + m.qk_pos_emb_head_dim
        if m.q_lora_rank:
            self.q_down = LinearCol(h, m.q_lora_rank, strategy, system, "q_down",
                                    with_tp_comm=False)
            self.q_norm = LayerNorm(m.q_lora_rank, strategy, system, "q_norm")
            self.q_up = LinearCol(m.q_lora_rank, m.head_num * qk_total, strategy,
                                  system, "q_up", with_tp_comm=False)
        else:
            self.q_proj = LinearCol(h, m.head_num * qk_total, strategy, system,
                                    "q_proj", with_tp_comm=False)
        self.kv_down = LinearCol(h, m.kv_lora_rank + m.qk_pos_emb_head_dim,
                                 strategy, system, "kv_down", with_tp_comm=False)
        self.kv_norm = LayerNorm(m.kv_lora_rank, strategy, system, "kv_norm")
        self.kv_up = LinearCol(m.kv_lora_rank,
                               m.head_num * (m.qk_head_dim + m.v_head_dim),
                               strategy, system, "kv_up", with_tp_comm=False)
        self.rope = RotaryEmbedding(strategy, system)
        self.core_attn = MLACoreAttention(m.head_num, qk_total, m.v_head_dim,
                                          strategy, system)
        self.out_proj = LinearRow(m.head_num * m.v_head_dim, h, strategy, system,
                                  "out_proj", with_tp_comm=False)

    def forward(self, input_info):
        m = self.m
        dbg = self.path_debug_context
        t = input_info.tensors[0]
        b, s = t.shape[0], t.shape[1]
        if m.q_lora_rank:
            qd = self.q_down(input_info, dbg)
            qn = self.q_norm(qd, dbg)
            q = self.q_up(qn, dbg)
        else:
            q = self.q_proj(input_info, dbg)
        kvd = self.kv_down(input_info, dbg)
        kv_c = InputOutputInfo([TensorSize([b, s, m.kv_lora_rank], t.dtype)])
        k_pe = TensorSize([b, s, m.qk_pos_emb_head_dim], t.dtype)
        kvn = self.kv_norm(kv_c, dbg)
        kv = self.kv_up(kvn, dbg)
        q_pe = TensorSize([b, s, m.head_num * m.qk_pos_emb_head_dim], t.dtype)
        self.rope(InputOutputInfo([q_pe, k_pe]), dbg)
        qk_total = m.qk_head_dim + m.qk_pos_emb_head_dim
        qf = TensorSize([b, s, m.head_num * qk_total], t.dtype)
        kf = TensorSize([b, s, m.head_num * qk_total], t.dtype)
        vf = TensorSize([b, s, m.head_num * m.v_head_dim], t.dtype)
        ctx = self.core_attn(InputOutputInfo([qf, kf, vf]), dbg)
        return self.out_proj(ctx, dbg)

    def apply_recompute(self):
        cfg = self.strategy.parse_attention_recompute()
        if cfg.recompute_core_attn:
            self.core_attn.set_recompute()
        if cfg.recompute_qkv:
            for name in ("q_down", "q_norm", "q_up", "q_proj", "kv_down",
                         "kv_norm", "kv_up", "rope"):
                mod = getattr(self, name, None)
                if mod is not None:
                    mod.set_recompute()
        if self.strategy.mla_rms_recompute or (
            self.strategy.megatron_recompute
            and "mla_up_proj" in self.strategy.megatron_recompute_module_set
        ):
            self.q_norm_recompute_tail()
        if cfg.recompute_out_proj:
            self.out_proj.set_recompute()

    def q_norm_recompute_tail(self):
        for name in ("q_norm", "kv_norm", "q_up", "kv_up"):
            mod = getattr(self, name, None)
            if mod is not None:
                mod.set_recompute(True)
                mod.is_variance_node = True


class MLP(MetaModule):
    """fc1 LinearCol -> Swiglu/Gelu -> fc2 LinearRow. Reference:
    dense_module.py:2888-2988."""

    def __init__(self, hidden_size, ffn_size, strategy, system, name="mlp",
                 use_swiglu=True, is_expert=False, local_ffn_divide=None):
        super().__init__(strategy, system, name)
        tp = strategy.etp_size if is_expert else strategy.tp_size
        div = local_ffn_divide or tp
        assert ffn_size % div == 0
        ffn_local = ffn_size // div
        fc1_out = (2 * ffn_local) if use_swiglu else ffn_local
        self.fc1 = LinearCol(hidden_size, fc1_out, strategy, system, "fc1",
                             is_expert=is_expert)
        self.act = (Swiglu(strategy, system) if use_swiglu else Gelu(strategy, system))
        self.fc2 = LinearRow(ffn_local, hidden_size, strategy, system, "fc2",
                             is_expert=is_expert)

    def forward(self, input_info):
        dbg = self.path_debug_context
        h = self.fc1(input_info, dbg)
        a = self.act(h, dbg)
        return self.fc2(a, dbg)

    def apply_recompute(self):
        cfg = self.strategy.parse_mlp_recompute()
        if cfg.recompute_fc1:
            self.fc1.set_recompute()
        if cfg.recompute_act:
            self.act.set_recompute()
        if cfg.recompute_fc2:
            self.fc2.set_recompute()
