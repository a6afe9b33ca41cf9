"""Per-layer parallel-strategy types + the searched-plan JSON codec.

The JSON schema here IS the contract between the search engine and the
runtime and is kept compatible with the reference's
(reference: galvatron/utils/strategy_utils.py:14-352, config_utils.py:24-46;
example plan: models/gpt/configs/galvatron_config_llama2-7b_*.json):
per-layer comma-joined arrays `tp_sizes_enc`, `tp_consecutive_flags`,
`cp_sizes_enc`, `dp_types_enc`, `use_sp`, `checkpoint`, plus scalars
`pp_deg`, `global_bsz`, `chunks`, `pp_division`, `pipeline_type`,
`default_dp_type`, `vtp`, `vsp`, `vcp`, `embed_sdp`.
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

DP_TYPE_NAMES = {0: "ddp_or_default", 1: "zero3"}


def str2array(s) -> List[int]:
    if isinstance(s, (list, tuple)):
        return [int(x) for x in s]
    return [int(x) for x in str(s).split(",") if str(x).strip() != ""]


def array2str(a) -> str:
    return ",".join(str(int(x)) for x in a)


def read_json_config(path: str) -> Dict[str, Any]:
    with open(path, "r", encoding="utf-8") as f:
        return json.load(f)


def write_json_config(config: Dict[str, Any], path: str) -> None:
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    with open(path, "w") as f:
        json.dump(config, f, indent=4)


@dataclass(frozen=True)
class LayerStrategy:
    """One decoder layer's parallel configuration.

    Reference: utils/strategy_utils.py LayerwiseStrategy dataclasses.
    dp degree is derived: world_size // (pp * tp_sp * cp * dp) partitioning is
    handled by the comm-group fabric; here we store the explicit degrees.
    """

    pp_deg: int = 1
    tp: int = 1            # tensor-parallel degree (megatron TP, with SP when sequence_parallel)
    sp: int = 1            # ulysses sequence-parallel degree (exclusive with tp>1)
    cp: int = 1            # context-parallel degree (zigzag ring)
    dp: int = 1            # data-parallel degree
    dp_type: str = "ddp"   # ddp | zero2 | zero3
    checkpoint: bool = False
    tp_consecutive: bool = True
    ep: int = 1            # expert parallel degree (MoE layers)

    @property
    def tp_sp(self) -> int:
        """Degree occupied by the tensor/sequence dimension (tp or ulysses sp)."""
        return max(self.tp, self.sp)

    @property
    def use_ulysses(self) -> bool:
        return self.sp > 1

    @property
    def sdp(self) -> int:
        """Group size for ZeRO sharding: dp*cp, and for ulysses layers also
        sp (params are replicated over sp, so the ZeRO domain merges it —
        reference comm_groups.py:310; runtime core/comm_groups.py builds
        the sdp group the same way)."""
        return self.dp * self.cp * (self.sp if self.sp > 1 else 1)

    def degree_product(self) -> int:
        return self.pp_deg * self.tp_sp * self.cp * self.dp


@dataclass
class HybridParallelPlan:
    """Whole-model per-layer plan: the runtime's single source of truth."""

    pp_deg: int = 1
    tp_sizes_enc: List[int] = field(default_factory=list)
    tp_consecutive_flags: List[int] = field(default_factory=list)
    cp_sizes_enc: List[int] = field(default_factory=list)
    dp_types_enc: List[int] = field(default_factory=list)      # 0 default, 1 zero3
    use_sp: List[int] = field(default_factory=list)            # 1 => ulysses sp on this layer
    checkpoint_flags: List[int] = field(default_factory=list)
    pp_division: Optional[List[int]] = None                    # layers per stage
    global_bsz: int = 8
    chunks: int = 1
    pipeline_type: str = "pipedream_flush"
    default_dp_type: str = "ddp"
    vtp: int = 1            # vocab (embedding/head) tp degree
    vsp: int = 0            # vocab ulysses-sp flag
    vcp: int = 1            # vocab cp degree
    embed_sdp: int = 0      # 1 => zero3 for embedding/head
    ep_deg: int = 1         # expert-parallel degree (uniform, MoE layers)

    @property
    def num_layers(self) -> int:
        return len(self.tp_sizes_enc)

    def layer(self, i: int, world_size: int) -> LayerStrategy:
        tp = self.tp_sizes_enc[i]
        sp = tp if self.use_sp[i] else 1
        if self.use_sp[i]:
            tp = 1
        cp = self.cp_sizes_enc[i]
        per_stage = world_size // self.pp_deg
        dp = per_stage // (max(tp, sp) * cp)
        dp_type = "zero3" if self.dp_types_enc[i] == 1 else self.default_dp_type
        return LayerStrategy(
            pp_deg=self.pp_deg, tp=tp, sp=sp, cp=cp, dp=dp, dp_type=dp_type,
            checkpoint=bool(self.checkpoint_flags[i]),
            tp_consecutive=bool(self.tp_consecutive_flags[i]),
            ep=self.ep_deg,
        )

    def vocab_strategy(self, world_size: int) -> LayerStrategy:
        tp = self.vtp
        sp = tp if self.vsp else 1
        if self.vsp:
            tp = 1
        cp = self.vcp
        per_stage = world_size // self.pp_deg
        dp = per_stage // (max(tp, sp) * cp)
        return LayerStrategy(
            pp_deg=self.pp_deg, tp=tp, sp=sp, cp=cp, dp=dp,
            dp_type="zero3" if self.embed_sdp else self.default_dp_type,
            checkpoint=False, tp_consecutive=True,
        )

    # ---------------- JSON codec (the search<->runtime contract) -----------

    @classmethod
    def from_config_dict(cls, cfg: Dict[str, Any]) -> "HybridParallelPlan":
        """reference: utils/config_utils.py:24 config2strategy."""
        tp_sizes = str2array(cfg["tp_sizes_enc"])
        n = len(tp_sizes)
        plan = cls(
            pp_deg=int(cfg["pp_deg"]),
            tp_sizes_enc=tp_sizes,
            tp_consecutive_flags=str2array(cfg.get("tp_consecutive_flags", [1] * n)),
            cp_sizes_enc=str2array(cfg.get("cp_sizes_enc", [1] * n)),
            dp_types_enc=str2array(cfg.get("dp_types_enc", [0] * n)),
            use_sp=str2array(cfg.get("use_sp", [0] * n)),
            checkpoint_flags=str2array(cfg.get("checkpoint", [0] * n)),
            pp_division=str2array(cfg["pp_division"]) if "pp_division" in cfg else None,
            global_bsz=int(cfg.get("global_bsz", 8)),
            chunks=int(cfg.get("chunks", 1)),
            pipeline_type=str(cfg.get("pipeline_type", "pipedream_flush")),
            default_dp_type=str(cfg.get("default_dp_type", "ddp")),
            vtp=int(cfg.get("vtp", 1)),
            vsp=int(cfg.get("vsp", 0)),
            vcp=int(cfg.get("vcp", 1)),
            embed_sdp=int(cfg.get("embed_sdp", 0)),
            ep_deg=int(cfg.get("ep_deg", 1)),
        )
        return plan

    def to_config_dict(self) -> Dict[str, Any]:
        """reference: utils/config_utils.py strategy_list2config."""
        out: Dict[str, Any] = {
            "pp_deg": self.pp_deg,
            "tp_sizes_enc": array2str(self.tp_sizes_enc),
            "tp_consecutive_flags": array2str(self.tp_consecutive_flags),
            "cp_sizes_enc": array2str(self.cp_sizes_enc),
            "dp_types_enc": array2str(self.dp_types_enc),
            "use_sp": array2str(self.use_sp),
            "checkpoint": array2str(self.checkpoint_flags),
            "global_bsz": self.global_bsz,
            "chunks": self.chunks,
            "pp_division": array2str(self.pp_division) if self.pp_division else str(self.num_layers),
            "pipeline_type": self.pipeline_type,
            "default_dp_type": self.default_dp_type,
            "vtp": self.vtp,
            "vsp": self.vsp,
            "ep_deg": self.ep_deg,
            "vcp": self.vcp,
            "embed_sdp": self.embed_sdp,
        }
        return out

    @classmethod
    def load(cls, path: str) -> "HybridParallelPlan":
        return cls.from_config_dict(read_json_config(path))

    def save(self, path: str) -> None:
        write_json_config(self.to_config_dict(), path)

    # ---------------- construction helpers ---------------------------------

    @classmethod
    def uniform(cls, num_layers: int, world_size: int, pp: int = 1, tp: int = 1,
                cp: int = 1, use_sp: bool = False, dp_type: str = "ddp",
                checkpoint: bool = False, chunks: int = 1, global_bsz: int = 8,
                pipeline_type: str = "pipedream_flush", vtp: Optional[int] = None,
                vsp: Optional[bool] = None, ep: int = 1) -> "HybridParallelPlan":
        """GLOBAL-mode plan: same strategy on every layer
        (reference: hybrid_parallel_config.py GLOBAL mode)."""
        assert world_size % (pp * tp * cp) == 0, \
            f"world {world_size} not divisible by pp*tp*cp={pp*tp*cp}"
        zero3 = 1 if dp_type == "zero3" else 0
        return cls(
            pp_deg=pp,
            tp_sizes_enc=[tp] * num_layers,
            tp_consecutive_flags=[1] * num_layers,
            cp_sizes_enc=[cp] * num_layers,
            dp_types_enc=[zero3] * num_layers,
            use_sp=[1 if use_sp else 0] * num_layers,
            checkpoint_flags=[1 if checkpoint else 0] * num_layers,
            pp_division=even_pp_division(num_layers, pp),
            global_bsz=global_bsz, chunks=chunks, pipeline_type=pipeline_type,
            default_dp_type=dp_type if dp_type != "zero3" else "zero2",
            vtp=tp if vtp is None else vtp,
            vsp=int(use_sp if vsp is None else vsp),
            vcp=1, embed_sdp=zero3, ep_deg=ep,
        )

    def validate(self, world_size: int) -> None:
        n = self.num_layers
        for name in ("tp_consecutive_flags", "cp_sizes_enc", "dp_types_enc",
                     "use_sp", "checkpoint_flags"):
            arr = getattr(self, name)
            if len(arr) != n:
                raise ValueError(f"{name} has {len(arr)} entries, expected {n}")
        if world_size % self.pp_deg != 0:
            raise ValueError(f"world {world_size} not divisible by pp {self.pp_deg}")
        per_stage = world_size // self.pp_deg
        for i in range(n):
            tp, cp = self.tp_sizes_enc[i], self.cp_sizes_enc[i]
            if per_stage % (tp * cp) != 0:
                raise ValueError(
                    f"layer {i}: tp*cp={tp*cp} does not divide per-stage size {per_stage}")
        if self.pp_division is not None:
            if sum(self.pp_division) != n or len(self.pp_division) != self.pp_deg:
                raise ValueError(
                    f"pp_division {self.pp_division} inconsistent with {n} layers / pp {self.pp_deg}")


def even_pp_division(num_layers: int, pp: int) -> List[int]:
    """Even layer split across pipeline stages (remainder to early stages).
    reference: search_engine.py:1094 even pp division."""
    base = num_layers // pp
    rem = num_layers % pp
    return [base + (1 if i < rem else 0) for i in range(pp)]
