"""Config layer + cost oracle (L0).

Three JSON-backed configs with the same schema as the reference
(simumax/core/config.py:209-1228 — StrategyConfig / SystemConfig /
ModelConfig, docs/{strategy,system,model}.md), with the cost model
re-derived for MI355X:

* compute: roofline over per-shape measured efficiency tables
  (MFMA dense peaks: 2.5 PF bf16 / 5 PF fp8; HBM3E 8 TB/s).
* network: alpha-beta per collective over RCCL; the 8-GPU xGMI mesh is a
  fully-connected 7-link topology, so intra-node bandwidth scales with
  (comm_num-1)/7 (`FC8` mode) — RCCL builds (comm_num-1) rotated rings so a
  ring collective over n GPUs uses exactly the links between participants.
"""

from __future__ import annotations

import copy
import json
import math
import re
import warnings
from collections import OrderedDict
from dataclasses import dataclass, field, fields
from typing import Any, Dict, List, Optional

from .consts import NET_OPS, SIMU_DEBUG


# --------------------------------------------------------------------------
# base
# --------------------------------------------------------------------------
@dataclass
class Config:
    @staticmethod
    def read_json_file(path: str) -> dict:
        with open(path, "r", encoding="utf-8") as f:
            return json.load(f)

    @classmethod
    def init_from_dict(cls, d: Dict[str, Any]):
        names = {f.name for f in fields(cls)}
        unknown = set(d) - names
        if unknown:
            warnings.warn(f"{cls.__name__}: ignoring unknown config keys {sorted(unknown)}")
        return cls(**{k: v for k, v in d.items() if k in names})

    @classmethod
    def init_from_config_file(cls, path: str):
        return cls.init_from_dict(cls.read_json_file(path))

    def to_dict(self) -> dict:
        out = {}
        for f in fields(self):
            v = getattr(self, f.name)
            if isinstance(v, Config):
                v = v.to_dict()
            out[f.name] = v
        # include simple public properties for report parity
        for name in dir(type(self)):
            if name.startswith("_") or name in out:
                continue
            attr = getattr(type(self), name, None)
            if isinstance(attr, property):
                try:
                    out[name] = getattr(self, name)
                except Exception:  # derived props can be invalid mid-setup
                    pass
        return {k: v for k, v in out.items() if not callable(v)}

    def sanity_check(self):
        pass


# --------------------------------------------------------------------------
# strategy
# --------------------------------------------------------------------------
VALID_RECOMPUTE_GRANULARITY = [
    "full_block",
    "attn_only",
    "mlp_only",
    "sdp_only",
    "selective_recompute",
]
VALID_MEGATRON_RECOMPUTE_MODULES = [
    "core_attn",
    "layernorm",
    "mla_up_proj",
    "moe_act",
    "mlp",
    "moe",
]
VALID_CP_A2A_MODES = ["async_cp", "sync_cp"]


@dataclass
class AttentionRecomputeConfig(Config):
    recompute_qkv: bool = False
    recompute_core_attn: bool = False
    recompute_out_proj: bool = False
    recompute_norm: bool = False


@dataclass
class MLPRecomputeConfig(Config):
    recompute_fc1: bool = False
    recompute_act: bool = False
    recompute_fc2: bool = False
    recompute_norm: bool = False


@dataclass
class StrategyConfig(Config):
    """All parallelism + runtime knobs. Field names are the JSON schema and
    match the reference (simumax/core/config.py:209-373)."""

    seq_len: Optional[int] = None
    micro_batch_size: Optional[int] = None
    micro_batch_num: Optional[int] = None
    dtype: Optional[str] = "bf16"
    fp8: bool = False

    # distributed strategy
    world_size: int = 8
    tp_size: int = 1
    cp_size: int = 1
    pp_size: int = 1
    ep_size: int = 1
    etp_size: int = 1
    cp_comm_type: str = "a2a"
    cp_a2a_mode: str = "async_cp"
    cp_sharding: str = "contiguous"   # | "zigzag" (balanced causal load)
    order_of_paralielism: str = "tp-cp-ep-dp-pp"  # (sic) reference spelling
    moe_dispatcher_policy: str = "all2all"
    num_layers_in_first_pipeline_stage: Optional[int] = None
    num_layers_in_last_pipeline_stage: Optional[int] = None
    account_for_embedding_in_pipeline_split: bool = False
    account_for_loss_in_pipeline_split: bool = False

    # memory optimization
    grad_reduce_in_bf16: bool = False
    cache_groupgemm_col_fp8_inputs: bool = False
    offload_groupgemm_col_inputs: bool = False

    attn_recompute: bool = False
    mla_rms_recompute: bool = False
    mlp_recompute: bool = False
    mlp_rms_recompute: bool = False

    enable_sequence_parallel: bool = True
    interleaving_size: int = 1
    microbatch_group_size_per_vp_stage: Optional[int] = None
    pp_comm_async: bool = True
    enable_straggler_model: bool = True
    zero_state: int = 1

    attention_sparse_ratio: float = 0.0  # 0.5 ~ causal flash attention
    enable_dropout: bool = False
    use_fp32_accum_grad: bool = True
    use_accm_weight: bool = True

    # recompute family
    enable_recompute: bool = True
    recompute_granularity: Optional[str] = None
    recompute_layer_num: int = 0
    recompute_variance: bool = False
    megatron_recompute: bool = False
    megatron_recompute_modules: Optional[List[str]] = None

    # fused kernels (MI355X: these gate which HIP kernel's efficiency entry
    # prices the op — fused RMSNorm/Swiglu/CE are the shipped CDNA4 kernels)
    use_flash_sdp: bool = True
    use_math_sdp: bool = False
    use_fused_norm: bool = True
    use_fused_swiglu: bool = True
    use_fused_grad_accumulation: bool = True
    cross_entropy_loss_fusion: bool = False
    overlap_grad_reduce: bool = True

    # Kept for schema parity with reference configs (TE version gates);
    # on ROCm these drive the same allocator-behaviour switches, re-measured
    # against Megatron-ROCm rather than TransformerEngine.
    te_version: Optional[str] = None
    te_dummy_wgrad_min_version: str = "2.3.0"
    te_cp_a2a_save_pre_posta2a_min_version: str = "2.8.0"
    te_grouped_linear_dummy_wgrad_min_version: str = "2.10.0"

    # per-group network tier selection
    tp_net: Optional[str] = "auto"
    cp_net: Optional[str] = "auto"
    pp_net: Optional[str] = "auto"
    dp_net: Optional[str] = "auto"
    ep_net: Optional[str] = "auto"
    etp_net: Optional[str] = "auto"
    edp_net: Optional[str] = "auto"

    dispatch_probs: bool = False  # fold probs into swiglu after grouped gemm1

    mem_factor: float = 0.94

    # ---- parsing helpers ------------------------------------------------
    @classmethod
    def init_from_format_strings(cls, s: str) -> "StrategyConfig":
        """Parse compact strings like 'seq4096.mbs1.mbc8.gbs64 tp2.pp2'."""
        pats = {
            "seq_len": (r"seq(\d+)", 4096),
            "micro_batch_size": (r"mbs(\d+)", 1),
            "micro_batch_num": (r"mbc(\d+)", 1),
            "tp_size": (r"tp(\d+)", 1),
            "cp_size": (r"cp(\d+)", 1),
            "ep_size": (r"ep(\d+)", 1),
            "pp_size": (r"pp(\d+)", 1),
            "world_size": (r"world_size:(\d+)", 8),
        }
        kwargs = {}
        for name, (pat, default) in pats.items():
            m = re.search(pat, s)
            kwargs[name] = int(m.group(1)) if m else default
        gbs = re.search(r"gbs(\d+)", s)
        cfg = cls(**kwargs)
        if gbs:
            cfg.reset_global_batch_size(int(gbs.group(1)))
        return cfg

    def reset_global_batch_size(self, gbs: int):
        per_dp = gbs // self.dp_size
        assert per_dp * self.dp_size == gbs, f"gbs {gbs} not divisible by dp {self.dp_size}"
        assert per_dp % self.micro_batch_size == 0
        self.micro_batch_num = per_dp // self.micro_batch_size

    # ---- derived --------------------------------------------------------
    @property
    def shard_size(self) -> int:
        return self.pp_size * self.tp_size * self.cp_size

    @property
    def dp_size(self) -> int:
        assert self.world_size % self.shard_size == 0, (
            f"world {self.world_size} % shard {self.shard_size} != 0"
        )
        return self.world_size // self.shard_size

    @property
    def global_batch_size(self) -> int:
        return self.micro_batch_size * self.micro_batch_num * self.dp_size

    @property
    def edp_size(self) -> int:
        return self.world_size // (self.ep_size * self.etp_size * self.pp_size)

    @property
    def parallelism(self) -> str:
        sp = f"sp{self.tp_size}." if self.enable_sequence_parallel else ""
        return (
            f"seq{self.seq_len}.mbs{self.micro_batch_size}.mbc{self.micro_batch_num}"
            f".gbs{self.global_batch_size} tp{self.tp_size}.{sp}cp{self.cp_size}"
            f".ep{self.ep_size}.pp{self.pp_size}.dp{self.dp_size}"
            f".etp{self.etp_size}.edp{self.edp_size}, world_size:{self.world_size}"
        )

    @property
    def megatron_recompute_module_set(self):
        return set(self.megatron_recompute_modules or [])

    @property
    def is_megatron_selective_recompute(self):
        return self.megatron_recompute and bool(self.megatron_recompute_module_set)

    @property
    def use_variance_tail_model(self):
        return self.recompute_variance or (
            self.is_megatron_selective_recompute
            and bool(self.megatron_recompute_module_set & {"layernorm", "mla_up_proj", "moe_act"})
        )

    @staticmethod
    def _version_tuple(v: Optional[str]):
        if not v:
            return None
        parts = re.findall(r"\d+", str(v))
        if not parts:
            return None
        nums = [int(p) for p in parts[:3]]
        while len(nums) < 3:
            nums.append(0)
        return tuple(nums)

    def _ver_ge(self, min_v):
        cur = self._version_tuple(self.te_version)
        mv = self._version_tuple(min_v)
        return cur is not None and mv is not None and cur >= mv

    @property
    def te_dummy_wgrad_memory_enabled(self):
        return self._ver_ge(self.te_dummy_wgrad_min_version)

    @property
    def te_grouped_linear_dummy_wgrad_memory_enabled(self):
        return self._ver_ge(self.te_grouped_linear_dummy_wgrad_min_version)

    @property
    def te_cp_a2a_saves_pre_posta2a_output(self):
        return self._ver_ge(self.te_cp_a2a_save_pre_posta2a_min_version)

    # ---- recompute parsing ---------------------------------------------
    def parse_attention_recompute(self) -> AttentionRecomputeConfig:
        cfg = AttentionRecomputeConfig()
        if not self.enable_recompute:
            return cfg
        g = self.recompute_granularity
        if g == "full_block" or g == "attn_only":
            cfg.recompute_qkv = True
            cfg.recompute_core_attn = True
            cfg.recompute_out_proj = True
            cfg.recompute_norm = True
        elif g == "sdp_only":
            cfg.recompute_core_attn = True
        elif g == "selective_recompute":
            cfg.recompute_core_attn = True
        elif self.megatron_recompute:
            mods = self.megatron_recompute_module_set
            cfg.recompute_core_attn = "core_attn" in mods
            cfg.recompute_norm = "layernorm" in mods
        if self.attn_recompute:
            cfg.recompute_qkv = True
            cfg.recompute_core_attn = True
            cfg.recompute_out_proj = True
        return cfg

    def parse_mlp_recompute(self) -> MLPRecomputeConfig:
        cfg = MLPRecomputeConfig()
        if not self.enable_recompute:
            return cfg
        g = self.recompute_granularity
        if g == "full_block" or g == "mlp_only":
            cfg.recompute_fc1 = True
            cfg.recompute_act = True
            cfg.recompute_fc2 = True
            cfg.recompute_norm = True
        elif g == "selective_recompute":
            cfg.recompute_act = True
        elif self.megatron_recompute:
            mods = self.megatron_recompute_module_set
            cfg.recompute_act = bool(mods & {"moe_act"})
            cfg.recompute_norm = "layernorm" in mods
            if mods & {"mlp", "moe"}:
                cfg.recompute_fc1 = True
                cfg.recompute_act = True
                cfg.recompute_fc2 = True
        if self.mlp_recompute:
            cfg.recompute_fc1 = True
            cfg.recompute_act = True
            cfg.recompute_fc2 = True
        return cfg

    # ---- checks ---------------------------------------------------------
    def sanity_check(self):
        assert self.seq_len and self.micro_batch_size and self.micro_batch_num, (
            "seq_len/micro_batch_size/micro_batch_num must be set"
        )
        assert self.dtype in ("bf16", "fp16", "fp32"), f"bad dtype {self.dtype}"
        assert self.world_size % self.shard_size == 0
        assert self.world_size % (self.ep_size * self.etp_size * self.pp_size) == 0, (
            "world_size must be divisible by ep*etp*pp"
        )
        assert self.zero_state in (0, 1, 2, 3)
        if self.zero_state in (2, 3):
            warnings.warn("zero_state 2/3 are modeled as zero-1 (unsupported, reference parity)")
        if self.recompute_granularity is not None:
            assert self.recompute_granularity in VALID_RECOMPUTE_GRANULARITY, (
                f"{self.recompute_granularity} not in {VALID_RECOMPUTE_GRANULARITY}"
            )
        for m in self.megatron_recompute_module_set:
            assert m in VALID_MEGATRON_RECOMPUTE_MODULES, m
        assert self.cp_comm_type in ("a2a", "all_gather", "ring")
        assert self.cp_a2a_mode in VALID_CP_A2A_MODES
        assert self.cp_sharding in ("contiguous", "zigzag")
        if self.cp_sharding == "zigzag":
            assert self.cp_comm_type != "a2a", \
                "a2a reassembles shards in rank order (contiguous only)"
        if self.enable_sequence_parallel:
            assert self.seq_len % (self.tp_size * self.cp_size) == 0
        if self.interleaving_size > 1:
            assert self.pp_size > 1, "VPP requires pp_size > 1"
        assert not (self.use_math_sdp and self.use_flash_sdp), "pick one sdp impl"


# --------------------------------------------------------------------------
# system
# --------------------------------------------------------------------------
@dataclass
class BandwidthConfig:
    gbps: float = 0.0
    efficient_factor: float = 1.0
    latency_us: float = 0.0
    # host-launch-bound chains whose length scales with a unit count
    # (e.g. per-local-expert GEMM-loop launches in MoE routing)
    per_unit_us: float = 0.0
    fixed_latency: Optional[float] = None
    fixed_latency_us_by_comm_num: Optional[Dict[str, float]] = None


@dataclass
class CompOpConfig:
    tflops: float = 0.0
    efficient_factor: float = 1.0
    accurate_efficient_factor: Optional[Dict[str, float]] = None


@dataclass
class NetOpConfig:
    scale: float = 1.0
    offset: float = 0.0
    efficient_factor: Optional[float] = None
    latency_us: Optional[float] = None
    fixed_latency_us: Optional[float] = None
    fixed_latency_us_by_comm_num: Optional[Dict[str, float]] = None
    # per-comm_num efficiency override (MI355X extension: RCCL efficiency on
    # the xGMI mesh is a strong function of participant count)
    efficient_factor_by_comm_num: Optional[Dict[str, float]] = None
    dp_fixed_bw: Optional[Dict[str, float]] = None


@dataclass
class AcceleratorConfig:
    backend: str = "rocm"
    mem_gbs: float = 288
    op: Dict[str, CompOpConfig] = None
    bandwidth: Dict[str, BandwidthConfig] = None
    mode: str = "roofline"
    # checkpointed-recompute rerun time as a fraction of a normal forward
    # (measured: the rerun's consumers read just-produced activations from
    # L2/L3 and the no-grad pass skips autograd recording); optional,
    # schema-compatible extension ignored by the reference
    recompute_factor: float = 1.0


@dataclass
class NetworkConfig:
    processor_usage: float = 0.0
    bandwidth: BandwidthConfig = None
    op: Dict[str, NetOpConfig] = None


@dataclass
class SystemConfig(Config):
    """Machine model + the three cost primitives (reference parity:
    simumax/core/config.py:746-1038)."""

    sys_name: str = "null"
    num_per_node: int = 8
    accelerator: AcceleratorConfig = None
    networks: Dict[str, NetworkConfig] = None
    FC8: bool = False
    intra_with_pcie: bool = False
    real_comm_bw: dict = field(default_factory=OrderedDict)
    miss_efficiency: dict = field(default_factory=OrderedDict)
    hit_efficiency: dict = field(default_factory=OrderedDict)

    @classmethod
    def init_from_dict(cls, d: Dict[str, Any]):
        d = copy.deepcopy(d)
        acc = d.pop("accelerator")
        networks = d.pop("networks")
        intra_with_pcie = networks.pop("intra_with_pcie", False)
        accelerator = AcceleratorConfig(
            backend=acc.get("backend", "rocm"),
            mem_gbs=acc["mem_gbs"],
            op={k: CompOpConfig(**v) for k, v in acc["op"].items()},
            bandwidth={k: BandwidthConfig(**v) for k, v in acc["bandwidth"].items()},
            mode=acc.get("mode", "roofline"),
            recompute_factor=acc.get("recompute_factor", 1.0),
        )
        nets = {
            name: NetworkConfig(
                processor_usage=net.get("processor_usage", 0.0),
                bandwidth=BandwidthConfig(**net["bandwidth"]),
                op={k: NetOpConfig(**v) for k, v in net["op"].items()},
            )
            for name, net in networks.items()
        }
        return cls(
            sys_name=d.pop("sys_name", "null"),
            num_per_node=d.pop("num_per_node", 8),
            accelerator=accelerator,
            networks=nets,
            FC8=d.pop("FC8", False),
            intra_with_pcie=intra_with_pcie,
        )

    # ---- recorders (observability: which shapes lack calibration) -------
    def record_miss_efficiency(self, op_name, flops, shape_desc, use_eff):
        if shape_desc:
            self.miss_efficiency.setdefault(op_name, {})[f"shape={shape_desc}"] = {
                "flops": flops,
                "use_eff": use_eff,
            }

    def record_hit_efficiency(self, op_name, flops, shape_desc, eff):
        self.hit_efficiency.setdefault(op_name, {})[shape_desc] = (flops, eff)

    def record_net_bw(self, op_name, net, comm_num, comm_stage, base_bw, real_bw,
                      eff_factor, total_time_us, comm_size, latency):
        self.real_comm_bw.setdefault(op_name, {})[str(comm_stage).lower()] = {
            "net": net, "base_bw": base_bw, "real_bw": real_bw,
            "eff_factor": eff_factor, "comm_num": comm_num,
            "comm_size": comm_size, "total_time": total_time_us,
            "latency": latency, "FC8": self.FC8,
        }

    def reset_record_info(self):
        self.miss_efficiency.clear()
        self.hit_efficiency.clear()
        self.real_comm_bw.clear()

    # ---- cost primitives -------------------------------------------------
    def compute_op_accuracy_time(self, op_name, flops, shape_desc="", reture_detail=False):
        """FLOP time in ms via per-shape efficiency table (exact string key)."""
        if flops == 0:
            if reture_detail:
                return dict(op_name=op_name, tflops=None, efficient_factor=None,
                            compute_only_time=0.0)
            return 0.0
        op = self.accelerator.op.get(op_name)
        if op is None:
            warnings.warn(f"op {op_name} not in system config, using default")
            op = self.accelerator.op["default"]
            self.record_miss_efficiency(op_name, flops, shape_desc, None)
        aef = op.accurate_efficient_factor
        if aef is not None and shape_desc in aef:
            eff = aef[shape_desc]
            self.record_hit_efficiency(op_name, flops, shape_desc, eff)
        else:
            eff = op.efficient_factor
            self.record_miss_efficiency(op_name, flops, shape_desc, eff)
        t = flops / (op.tflops * 1e12 * eff) * 1e3
        if reture_detail:
            return dict(op_name=op_name, tflops=op.tflops, efficient_factor=eff,
                        compute_only_time=t)
        return t

    def compute_mem_access_time(self, op_name, mem_bytes, reture_detail=False,
                                units=0):
        """HBM3E access time in ms via bandwidth table (8 TB/s peak, measured
        stream efficiency from the HIP harness). `units` scales the
        per_unit_us launch-latency term (per-local-expert chains)."""
        op = self.accelerator.bandwidth.get(op_name) or self.accelerator.bandwidth["default"]
        t = mem_bytes / (op.gbps * 1024**3 * op.efficient_factor) * 1e3
        t += (op.latency_us + getattr(op, "per_unit_us", 0.0) * units) / 1e3
        if mem_bytes == 0:
            t = 0.0
        if reture_detail:
            return dict(gbps=op.gbps, efficient_factor=op.efficient_factor,
                        latency_us=op.latency_us, io_time=t)
        return t

    @staticmethod
    def _by_comm_num(values, comm_num, default=None):
        if not values:
            return default
        for key in (str(comm_num), comm_num):
            if key in values:
                return values[key]
        return default

    def compute_net_op_time(self, op_name, size, comm_num, net="", comm_stage="unknown",
                            strategy: StrategyConfig = None):
        """Collective time in ms, alpha-beta + FC8 xGMI scaling.

        actual_size = size*scale + (size*scale/comm_num)*offset, i.e. ring
        all_reduce scale=2/offset=-1, ag/rs/a2a 1/-1, p2p 1/0.
        """
        assert op_name in NET_OPS, f"{op_name} not in {NET_OPS}"
        net_data = self.networks.get(net)
        assert net_data is not None, f"net tier '{net}' missing for {op_name}"
        op = net_data.op.get(op_name)
        assert op is not None, f"{op_name} missing in net '{net}'"
        scale, offset = op.scale, op.offset
        eff = self._by_comm_num(op.efficient_factor_by_comm_num, comm_num, None)
        if eff is None:
            eff = op.efficient_factor
        if eff is None:
            eff = net_data.bandwidth.efficient_factor

        actual_size = size * scale
        actual_size += (actual_size / comm_num) * offset

        is_dense_dp = comm_stage in ("dp", "dp_cp")
        if "pcie" in net and is_dense_dp and op.dp_fixed_bw:
            fixed = self._by_comm_num(op.dp_fixed_bw, comm_num)
            if fixed:
                return actual_size / (fixed * 1024**3) * 1e3

        bw = net_data.bandwidth.gbps
        # xGMI fully-connected mesh: only links to participating peers carry
        # traffic; 8-GPU node has 7 links per GPU.
        if self.FC8 and net == "high_intra_node":
            bw *= (comm_num - 1) / (self.num_per_node - 1)

        if net == "inter_node":
            if op_name == "p2p":
                bw /= self.num_per_node
            if op_name == "all2all":
                stage = str(comm_stage).lower()
                if "ep" in stage or "cp" in stage:
                    k = max(1, math.ceil(comm_num / self.num_per_node))
                    actual_size = (k - 1) / k * actual_size
                    bw /= self.num_per_node  # one NIC per a2a group
            if op_name in ("all_reduce", "all_gather", "reduce_scatter") and strategy is not None:
                if is_dense_dp:
                    mult = strategy.tp_size
                    if comm_stage == "dp":
                        mult *= strategy.cp_size
                    bw /= min(self.num_per_node, mult)
                elif comm_stage == "edp":
                    bw /= min(self.num_per_node, strategy.ep_size * strategy.etp_size)

        base_latency = op.latency_us if op.latency_us is not None else net_data.bandwidth.latency_us
        fixed_latency = self._by_comm_num(
            op.fixed_latency_us_by_comm_num, comm_num, op.fixed_latency_us
        )
        if fixed_latency is None:
            fixed_latency = self._by_comm_num(
                net_data.bandwidth.fixed_latency_us_by_comm_num, comm_num,
                net_data.bandwidth.fixed_latency,
            )
        if fixed_latency is None:
            fixed_latency = 0.0
        latency = base_latency
        if comm_num == 1:
            return 0.0
        if self.num_per_node == 8 and op_name in ("all_reduce", "all_gather",
                                                  "reduce_scatter", "all2all"):
            latency = base_latency * (comm_num + offset) * scale
        t = actual_size / (bw * 1024**3 * eff) * 1e3 + (latency + fixed_latency) / 1e3
        self.record_net_bw(op_name, net, comm_num, comm_stage,
                           net_data.bandwidth.gbps, bw * eff, eff, t * 1e3,
                           actual_size, latency)
        return t

    def compute_end2end_time(self, compute_time, mem_time):
        assert self.accelerator.mode in ("only_compute", "roofline")
        if self.accelerator.mode == "only_compute":
            return compute_time if compute_time else mem_time
        return max(compute_time, mem_time)

    def sanity_check(self):
        assert self.accelerator is not None and self.networks is not None
        assert "default" in self.accelerator.op
        assert "default" in self.accelerator.bandwidth


# --------------------------------------------------------------------------
# model
# --------------------------------------------------------------------------
@dataclass
class ModelConfig(Config):
    """Decoder-only transformer architecture (GQA / MLA / MoE), reference
    parity simumax/core/config.py:1042-1228."""

    hidden_size: int = None
    head_num: int = None
    kv_head_num: int = None
    model_type: Optional[str] = None
    model_name: Optional[str] = None
    head_size: Optional[int] = None
    intermediate_size: Optional[int] = None
    layer_num: Optional[int] = None
    vocab_size: Optional[int] = None
    orig_vocab_size: Optional[int] = None
    use_swiglu: Optional[bool] = None
    expert_num: int = 1
    topk: Optional[int] = None
    attention_type: str = "mha"
    moe_ffn_hidden_size: Optional[int] = None
    moe_shared_expert_intermediate_size: Optional[int] = None
    v_head_dim: Optional[int] = None
    qk_head_dim: Optional[int] = None
    qk_pos_emb_head_dim: Optional[int] = None
    q_lora_rank: Optional[int] = None
    kv_lora_rank: Optional[int] = None
    dense_layers: int = 0
    moe_pad_expert_input_to_capacity: bool = True
    capacity: float = 1
    group_linear_mode: str = "parallel"
    make_vocab_size_divisible_by: int = 128
    padded_vocab_size: bool = True

    def __post_init__(self):
        if self.moe_ffn_hidden_size is None:
            self.moe_ffn_hidden_size = self.intermediate_size
        if self.model_type is None:
            self.model_type = "moe" if self.expert_num > 1 else "dense"
        if self.kv_head_num is None:
            self.kv_head_num = self.head_num
        if self.head_size is None and self.head_num and self.hidden_size:
            self.head_size = self.hidden_size // self.head_num

    def maybe_pad_vocab_size(self, tp_size, log=False):
        """Megatron-style vocab padding to make it divisible by 128*tp."""
        if self.padded_vocab_size:
            if self.orig_vocab_size is None:
                self.orig_vocab_size = self.vocab_size
            multiple = self.make_vocab_size_divisible_by * tp_size
            after = int(math.ceil(self.orig_vocab_size / multiple) * multiple)
            if log:
                print(f" > padded vocab {self.orig_vocab_size} -> {after}")
            self.vocab_size = after

    def set_vocab_size(self, vocab_size):
        self.orig_vocab_size = vocab_size
        self.vocab_size = vocab_size

    # ---- element counts --------------------------------------------------
    @property
    def mlp_elements(self):
        factor = 3 if self.use_swiglu else 2
        return factor * self.hidden_size * self.moe_ffn_hidden_size

    @property
    def base_proj_elements(self):
        if self.attention_type == "mla":
            return self.v_head_dim * self.head_num * self.hidden_size
        return self.hidden_size * self.hidden_size

    @property
    def attn_proj_elements(self):
        return self.base_proj_elements

    @property
    def norm_elements(self):
        return self.hidden_size

    @property
    def qkv_proj_elements(self):
        assert self.head_num is not None
        kv_head_num = self.kv_head_num or self.head_num
        if self.attention_type == "mla":
            if self.q_lora_rank is None:
                elems = self.hidden_size * self.head_num * (self.qk_head_dim + self.qk_pos_emb_head_dim)
            else:
                elems = self.hidden_size * self.q_lora_rank
                elems += self.q_lora_rank * self.head_num * (self.qk_head_dim + self.qk_pos_emb_head_dim)
            elems += self.hidden_size * (self.kv_lora_rank + self.qk_pos_emb_head_dim)
            elems += self.kv_lora_rank * self.head_num * (self.qk_head_dim + self.v_head_dim)
            return elems
        proj = self.head_size * self.head_num + 2 * self.head_size * kv_head_num
        return self.hidden_size * proj

    @property
    def vocab_elements(self):
        return self.vocab_size * self.hidden_size

    @property
    def layer_elements(self):
        return (
            self.qkv_proj_elements
            + 2 * self.norm_elements
            + self.attn_proj_elements
            + self.expert_num * self.mlp_elements
        )

    @property
    def layer_act_elements(self):
        factor = 1
        if self.topk is not None and self.topk > 1:
            factor += self.topk - 1
        return (
            self.qkv_proj_elements
            + 2 * self.norm_elements
            + self.attn_proj_elements
            + factor * self.mlp_elements
        )

    @property
    def param_numel(self):
        return 2 * self.vocab_elements + self.layer_elements * self.layer_num + self.norm_elements

    @property
    def activated_param_numel(self):
        return (
            2 * self.vocab_elements
            + self.layer_act_elements * self.layer_num
            + self.norm_elements
        )

    def flops_per_token(self, context_seq_len, with_attn=True):
        """Theoretical 6ND FLOPs per token (fwd+bwd), Megatron convention."""
        attn_matmul = 3 * 2 * self.layer_num * (self.qkv_proj_elements + self.attn_proj_elements)
        factor = 1
        res = 0
        if self.topk is not None and self.topk > 1:
            factor += self.topk - 1
            res += 3 * 2 * self.layer_num * self.hidden_size * self.expert_num
        if self.moe_shared_expert_intermediate_size is not None:
            factor += self.moe_shared_expert_intermediate_size / self.moe_ffn_hidden_size
        mlp_matmul = 3 * 2 * self.layer_num * self.mlp_elements * factor
        res += attn_matmul + mlp_matmul
        if with_attn:
            attn_sdp = 3 * 2 * self.layer_num * (2 * context_seq_len * self.hidden_size)
            if self.attention_type == "mla":
                attn_sdp = 3 * 2 * self.layer_num * (
                    context_seq_len * (self.qk_head_dim + self.qk_pos_emb_head_dim) * self.head_num
                    + context_seq_len * self.v_head_dim * self.head_num
                )
            res += attn_sdp
        res += 3 * 2 * (self.hidden_size * self.vocab_size)
        return res

    def sanity_check(self):
        assert self.hidden_size and self.head_num and self.layer_num and self.vocab_size
        if self.attention_type == "mla":
            for f_ in ("v_head_dim", "qk_head_dim", "qk_pos_emb_head_dim", "kv_lora_rank"):
                assert getattr(self, f_) is not None, f"MLA requires {f_}"
        if self.model_type == "moe":
            assert self.expert_num > 1 and self.topk
