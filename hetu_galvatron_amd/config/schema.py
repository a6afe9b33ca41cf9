"""Pydantic argument schemas.

Capability parity with the reference's Hydra/Pydantic config plane
(reference: galvatron/core/runtime/args_schema.py:18-386,
galvatron/core/profiler/args_schema.py, galvatron/core/search_engine/args_schema.py).
Rebuilt for MI355X: no hydra/omegaconf dependency — plain YAML + dotted
overrides (see loader.py); defaults sized for 288 GB HBM3E per GPU and an
8-GPU xGMI mesh.
"""
from __future__ import annotations

from typing import List, Optional, Union

from pydantic import BaseModel, Field, model_validator


class ParallelArgs(BaseModel):
    """Global-mode parallelism degrees (used when no searched strategy JSON is given).

    Reference: GalvatronParallelArgs (args_schema.py:18-50).
    """

    pp_deg: int = 1
    global_tp_deg: int = 1
    global_cp_deg: int = 1
    global_ep_deg: int = 1
    sdp: int = 0  # 0: ddp/zero2 default, 1: zero3 default
    default_dp_type: str = "ddp"  # ddp | zero2 | zero3
    global_checkpoint: int = 0
    chunks: int = -1  # number of pipeline microbatches; -1 = auto
    pipeline_type: str = "pipedream_flush"  # gpipe | pipedream_flush
    use_ulysses: bool = False  # SP mode for global tp degree
    sequence_parallel: bool = True  # Megatron-SP on TP layers
    cp_mode: str = "zigzag"  # zigzag ring attention
    # vocab (embedding / lm-head) degrees; 0 => follow decoder layer degrees
    vocab_tp: int = 1
    vocab_sp: int = 0
    vocab_cp: int = 1
    mixed_precision: str = "bf16"  # fp32 | bf16 | fp16
    # fp16 dynamic loss scaling (reference: Megatron-style scaler)
    loss_scale_init: float = 65536.0
    loss_scale_window: int = 1000
    min_loss_scale: float = 1.0
    galvatron_config_path: Optional[str] = None  # searched per-layer plan JSON
    gradient_reduce_in_fp32: bool = False
    overlap_grad_reduce: bool = True
    # reduce-scatter grads every microbatch and accumulate in the shard
    # domain (fp32): costs chunks x reduce traffic, saves the 4 B/param
    # full-size fp32 accumulator — required for 70B-class zero3 fits
    reduce_grads_each_microbatch: bool = False


class ModelArgs(BaseModel):
    """Model architecture fields (reference: GalvatronModelArgs args_schema.py:51-177)."""

    model_name: str = "llama-3-8b"
    model_type: str = "llama"  # llama | gpt | moe-llama | t5 (enc-dec)
    hidden_size: int = 4096
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: Optional[int] = 8  # None => MHA
    ffn_hidden_size: int = 14336
    vocab_size: int = 128256
    max_position_embeddings: int = 8192
    seq_length: int = 4096
    hidden_act: str = "silu"  # silu(swiglu) | gelu | geglu | relu
    normalization: str = "rmsnorm"  # rmsnorm | layernorm
    qk_layernorm: bool = False  # per-head norm on q/k pre-RoPE (Qwen3/Gemma2)
    rotary_interleaved: bool = False  # GPT-J pairwise RoPE (default: NEOX halves)
    init_method_std: float = 0.02  # weight-init normal std (output projs scaled /sqrt(2L))
    norm_epsilon: float = 1e-5
    position_embedding_type: str = "rope"  # rope | learned | relative
    relative_attention_num_buckets: int = 32    # t5 bucketized bias
    relative_attention_max_distance: int = 128
    rope_theta: float = 500000.0
    rope_scaling: Optional[float] = None
    sliding_window: Optional[int] = None  # mistral-style windowed attention
    # None => 1/sqrt(head_dim); t5 uses 1.0 (unscaled scores)
    attention_softmax_scale: Optional[float] = None
    add_bias_linear: bool = False
    add_qkv_bias: bool = False
    tie_word_embeddings: bool = False
    hidden_dropout: float = 0.0
    attention_dropout: float = 0.0
    untie_embeddings_and_output_weights: bool = True
    # MoE fields (mixtral-style; used when model_type == "moe-llama")
    num_experts: int = 0
    moe_router_topk: int = 2
    moe_aux_loss_coeff: float = 0.01
    moe_z_loss_coeff: float = 0.0
    moe_router_score_function: str = "softmax"  # softmax | sigmoid
    moe_token_dispatcher_type: str = "alltoall"  # allgather | alltoall
    moe_expert_capacity_factor: Optional[float] = None
    moe_pad_expert_input_to_capacity: bool = False  # static expert shapes
    moe_router_pre_softmax: bool = False
    moe_grouped_gemm: bool = True
    moe_shared_expert_intermediate_size: int = 0
    moe_ffn_hidden_size: Optional[int] = None
    moe_aux_loss_free: bool = False
    moe_router_bias_update_rate: float = 1e-3
    moe_aux_loss_type: str = "aux_loss"  # aux_loss | seq_aux_loss
    moe_router_load_balancing_type: str = "none"  # none | sinkhorn
    # DeepSeek-style group-limited (node-limited) routing
    moe_router_num_groups: Optional[int] = None
    moe_router_group_topk: Optional[int] = None

    @property
    def kv_heads(self) -> int:
        return self.num_key_value_heads or self.num_attention_heads

    num_decoder_layers: Optional[int] = None  # t5: decoder depth
    encoder_seq_length: Optional[int] = None   # t5: encoder seq len
    kv_channels: Optional[int] = None          # per-head dim override (t5 d_kv)

    @property
    def head_dim(self) -> int:
        if self.kv_channels:
            return self.kv_channels
        return self.hidden_size // self.num_attention_heads


class TrainArgs(BaseModel):
    """Reference: CommonTrainArgs (args_schema.py:195-270)."""

    global_train_batch_size: int = 8
    # "start,increment,ramp_samples" (string or 3-int list)
    rampup_batch_size: Optional[Union[str, List[int]]] = None
    epochs: int = 1
    train_iters: int = 20
    lr: float = 1e-4
    min_lr: float = 1e-5
    adam_weight_decay: float = 0.01
    adam_beta1: float = 0.9
    adam_beta2: float = 0.95
    adam_eps: float = 1e-8
    clip_grad: float = 1.0
    lr_decay_style: str = "cosine"  # constant | linear | cosine | wsd
    lr_warmup_iters: int = 0
    lr_decay_iters: Optional[int] = None
    lr_wsd_decay_iters: int = 0
    seed: int = 1234
    entropy_in_fp32: bool = True
    distributed_backend: str = "nccl"  # nccl(=RCCL on ROCm) | gloo
    local_rank: int = 0
    use_fused_adam: bool = True
    deterministic_mode: bool = False
    eval_interval: int = 0  # run validation every N iters (0 = off)
    eval_iters: int = 10
    lr_warmup_init: float = 0.0  # warmup ramps from this LR (reference)
    # every N iters, verify replicated params are identical across their
    # groups (reference test_mode realtime checks); 0 = off
    check_weight_consistency_interval: int = 0


class DataArgs(BaseModel):
    """Reference: CommonDataArgs (args_schema.py:271-322)."""

    dataset: str = "synthetic"  # synthetic | megatron
    data_path: Optional[List[str]] = None
    split: str = "969,30,1"
    tokenizer_type: str = "null"
    vocab_file: Optional[str] = None
    merge_file: Optional[str] = None
    tokenizer_model: Optional[str] = None
    num_workers: int = 0
    synthetic_dataset_size: int = 1024
    eod_mask_loss: bool = False  # zero the loss on end-of-document tokens
    eod_token_id: Optional[int] = None  # required when eod_mask_loss


class CkptArgs(BaseModel):
    """Reference: CommonCkptArgs (args_schema.py:323-334)."""

    load: Optional[str] = None
    save: Optional[str] = None
    save_interval: int = 0
    load_format: str = "auto"  # auto | hf | dist
    no_load_optim: bool = False
    no_save_optim: bool = False
    load_iteration: int = 0  # 0 = latest (reference load_iteration)
    distributed_checkpoint: bool = False


class LoggingArgs(BaseModel):
    """Reference: LoggingConfig (args_schema.py:335-342)."""

    log_interval: int = 1
    tensorboard_dir: Optional[str] = None
    wandb_project: Optional[str] = None
    wandb_exp_name: Optional[str] = None


class ProfileArgs(BaseModel):
    """Runtime/model profiling switches (reference: GalvatronProfileArgs args_schema.py:178-194)."""

    profile: int = 0
    profile_type: str = "memory"  # memory | computation
    profile_forward: int = 0
    profile_mode: str = "static"  # static | batch | sequence
    profile_fixed_batch_size: int = 8
    profile_batch_size_start: int = 1
    profile_batch_size_end: int = 8
    profile_seq_length_list: Optional[List[int]] = None
    profile_min_layer_num: int = 1
    profile_max_layer_num: int = 2
    profile_dir: str = "configs"


class SearchArgs(BaseModel):
    """Search-engine knobs (reference: search_engine/args_schema.py:12-74)."""

    num_nodes: int = 1
    num_gpus_per_node: int = 8
    memory_constraint: int = 240  # GB per GPU budget (MI355X: 288 GB physical)
    min_bsz: int = 8
    max_bsz: int = 64
    bsz_scale: int = 8
    settle_bsz: int = -1
    settle_chunks: int = -1
    max_tp_deg: int = 8
    max_sp_deg: int = 8
    recommend_min_bsz: int = 0  # 1: start the bsz sweep at world size
    debug_costmodel_coe: float = 1.0  # cost-model calibration multiplier
    max_pp_deg: int = 8
    max_cp_deg: int = 1
    disable_dp: int = 0
    disable_tp: int = 0
    disable_pp: int = 0
    disable_sdp: int = 0
    disable_ckpt: int = 0
    disable_cp: int = 1
    disable_vtp: int = 0
    disable_sp: int = 0  # ulysses
    fine_grained_mode: int = 1
    sp_space: str = "tp+sp"  # tp | sp | tp+sp
    time_profile_mode: str = "static"
    memory_profile_mode: str = "static"
    parallel_search: bool = False
    log_dir: str = "logs"
    output_config_path: Optional[str] = None


class HardwareProfileArgs(BaseModel):
    """Hardware-profiler sweep config (reference: profiler/args_schema.py)."""

    num_nodes: int = 1
    num_gpus_per_node: int = 8
    master_addr: str = "127.0.0.1"
    master_port: int = 29511
    max_tp_size: int = 8
    max_pp_deg: int = 8
    start_mb: int = 1  # message sizes 2^k MB
    end_mb: int = 1024
    warmup_iters: int = 5
    measure_iters: int = 20
    avg_or_min_or_first: str = "first_avg"
    hardware_config_dir: str = "hardware_configs"
    overlap_time_multiply: int = 4


class GalvatronConfig(BaseModel):
    """Root config (reference: GalvatronRuntimeArgs args_schema.py:344-386 + CoreArgs)."""

    parallel: ParallelArgs = Field(default_factory=ParallelArgs)
    model: ModelArgs = Field(default_factory=ModelArgs)
    train: TrainArgs = Field(default_factory=TrainArgs)
    data: DataArgs = Field(default_factory=DataArgs)
    ckpt: CkptArgs = Field(default_factory=CkptArgs)
    logging: LoggingArgs = Field(default_factory=LoggingArgs)
    profile: ProfileArgs = Field(default_factory=ProfileArgs)
    search: SearchArgs = Field(default_factory=SearchArgs)
    profile_hardware: HardwareProfileArgs = Field(default_factory=HardwareProfileArgs)

    @model_validator(mode="after")
    def _validate(self) -> "GalvatronConfig":
        if self.parallel.default_dp_type not in ("ddp", "zero2", "zero3"):
            raise ValueError(f"bad default_dp_type {self.parallel.default_dp_type}")
        if self.parallel.pipeline_type not in ("gpipe", "pipedream_flush"):
            raise ValueError(f"bad pipeline_type {self.parallel.pipeline_type}")
        if self.parallel.mixed_precision not in ("fp32", "bf16", "fp16"):
            raise ValueError(f"bad mixed_precision {self.parallel.mixed_precision}")
        if self.model.hidden_size % self.model.num_attention_heads != 0:
            raise ValueError("hidden_size must be divisible by num_attention_heads")
        return self
