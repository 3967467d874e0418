"""Typed config tree with verl-style dot-path CLI overrides.

Mirrors the dot-path surface of the reference's Hydra tree
(ppo_stream_trainer.yaml + verl ppo_trainer defaults) so that
`actor_rollout_ref.actor.ppo_mini_batch_size=256`-style overrides work, while
staying dependency-free (no hydra requirement).
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from typing import Any, List, Optional

import yaml


# ------------------------------------------------------------------- subtrees


@dataclass
class ModelConfig:
    path: str = "llama3-1b"            # registry name or HF-style dir
    # optional safetensors weights (file or HF model dir) loaded by name
    # into the freshly built model before FSDP sharding
    load_weights: str = ""
    dtype: str = "bfloat16"
    override_config: dict = field(default_factory=dict)
    enable_gradient_checkpointing: bool = True
    use_remove_padding: bool = True
    lora_rank: int = 0
    lora_alpha: float = 16.0
    trust_remote_code: bool = False


@dataclass
class OptimConfig:
    lr: float = 1e-6
    betas: tuple = (0.9, 0.999)
    eps: float = 1e-8
    weight_decay: float = 0.01
    grad_clip: float = 1.0
    lr_warmup_steps: int = 0
    lr_warmup_steps_ratio: float = 0.0
    total_training_steps: int = -1
    warmup_style: str = "constant"


@dataclass
class FSDPConfig:
    param_offload: bool = False
    optimizer_offload: bool = False
    reshard_after_forward: bool = True
    mixed_precision_dtype: str = "bfloat16"
    reduce_dtype: str = "float32"
    wrap_policy_min_params: int = 0     # 0 -> wrap per decoder layer


@dataclass
class ActorConfig:
    strategy: str = "fsdp2"
    ppo_mini_batch_size: int = 256
    ppo_micro_batch_size_per_gpu: Optional[int] = None
    ppo_max_token_len_per_gpu: int = 16384
    use_dynamic_bsz: bool = True
    ppo_epochs: int = 1
    clip_ratio: float = 0.2
    clip_ratio_low: Optional[float] = None
    clip_ratio_high: Optional[float] = None
    clip_ratio_c: float = 3.0
    loss_agg_mode: str = "token-mean"
    entropy_coeff: float = 0.0
    use_kl_loss: bool = False
    kl_loss_coef: float = 0.001
    # truncated importance sampling vs the ROLLOUT policy's logprobs
    # (off-policy correction for streamed/stale rollouts); 0 disables
    tis_imp_ratio_cap: float = 0.0
    kl_loss_type: str = "low_var_kl"
    policy_loss_type: str = "vanilla"
    ulysses_sequence_parallel_size: int = 1
    optim: OptimConfig = field(default_factory=OptimConfig)
    fsdp: FSDPConfig = field(default_factory=FSDPConfig)
    use_fused_kernels: bool = True
    shuffle: bool = False
    checkpoint_contents: List[str] = field(
        default_factory=lambda: ["model", "optimizer", "extra"])


@dataclass
class CriticConfig:
    enable: bool = False
    strategy: str = "fsdp2"
    ppo_mini_batch_size: int = 256
    ppo_micro_batch_size_per_gpu: Optional[int] = None
    ppo_max_token_len_per_gpu: int = 16384
    use_dynamic_bsz: bool = True
    cliprange_value: float = 0.5
    loss_agg_mode: str = "token-mean"
    ulysses_sequence_parallel_size: int = 1
    model: ModelConfig = field(default_factory=ModelConfig)
    optim: OptimConfig = field(default_factory=lambda: OptimConfig(lr=1e-5))
    fsdp: FSDPConfig = field(default_factory=FSDPConfig)


@dataclass
class SamplingConfig:
    temperature: float = 1.0
    top_k: int = -1
    top_p: float = 1.0
    n: int = 1                           # samples per prompt
    do_sample: bool = True


@dataclass
class MultiTurnConfig:
    """Reference parity: verl_stream/workers/config/rollout.py:44-58."""
    enable: bool = False
    max_assistant_turns: int = 2
    max_user_turns: int = 8
    max_tool_response_length: int = 256
    # cap tokens per assistant turn (0 = until EOS / response budget)
    per_turn_max_tokens: int = 0
    # python file + fn implementing the interaction:
    #   fn(prompt_ids: list[int], response_ids: list[int]) ->
    #       (user_ids: list[int] | None, done: bool)
    interaction_path: Optional[str] = None
    interaction_name: str = "generate_turn"


@dataclass
class RolloutConfig:
    name: str = "native"                 # in-process MI355X decoder
    prompt_length: int = 512
    response_length: int = 1024
    dtype: str = "bfloat16"
    gpu_memory_utilization: float = 0.6
    tensor_model_parallel_size: int = 1
    data_parallel_size: int = 1
    pipeline_model_parallel_size: int = 1
    expert_parallel_size: int = 1
    max_num_batched_tokens: int = 8192
    max_running_requests: int = 256
    page_size: int = 16                  # KV tokens per page
    decode_chunk_size: int = 16          # device-resident decode chunk
    # cross-request KV prefix reuse (radix trie).  Off by default in RL
    # training: weights change every step and the cache is flushed on each
    # install; turn on for serving / multi-turn / shared system prompts.
    enable_radix_cache: bool = False
    # multi-turn rollouts (reference: MultiTurnConfig, config/rollout.py:44
    # — an interaction/environment generates the next user turn between
    # assistant turns; only assistant tokens carry loss)
    multi_turn: "MultiTurnConfig" = field(
        default_factory=lambda: MultiTurnConfig())
    # disaggregated split (BASELINE config #4): the LAST num_rollout_ranks
    # ranks of the world serve rollout; 0 = co-located
    num_rollout_ranks: int = 0
    rollout_port_base: int = 30000
    max_local_gen_s: float = 0.0         # scheduler time-box (0 = off)
    # elastic mode: manager HTTP facade port on rank 0 (0 = off) — remote
    # instances join a running job here (reference manager port 5000)
    rollout_manager_port: int = 0
    sampling: SamplingConfig = field(default_factory=SamplingConfig)
    calculate_log_probs: bool = True
    min_stream_batch_size: int = 16
    max_local_gen_s: float = 150.0       # initial adaptive time-box (state.rs:79)
    free_cache_engine: bool = False
    enforce_eager: bool = False          # False -> hipGraph-captured decode step


@dataclass
class RefConfig:
    enable: bool = False
    log_prob_micro_batch_size_per_gpu: Optional[int] = None


@dataclass
class KLCtrlConfig:
    type: str = "fixed"
    kl_coef: float = 0.001


@dataclass
class AlgorithmConfig:
    adv_estimator: str = "grpo"          # grpo | gae
    gamma: float = 1.0
    lam: float = 1.0
    use_kl_in_reward: bool = False
    kl_penalty: str = "kl"
    kl_ctrl: KLCtrlConfig = field(default_factory=KLCtrlConfig)
    norm_adv_by_std_in_grpo: bool = True


@dataclass
class DataConfig:
    train_batch_size: int = 128
    max_prompt_length: int = 512
    max_response_length: int = 1024
    shuffle: bool = True
    seed: int = 1
    train_files: List[str] = field(default_factory=list)
    val_files: List[str] = field(default_factory=list)
    prompt_key: str = "prompt"
    reward_fn_key: str = "data_source"
    filter_overlong_prompts: bool = False
    synthetic: bool = False              # synthetic token prompts (bench / tests)
    synthetic_num_prompts: int = 1024


@dataclass
class TrainerConfig:
    total_epochs: int = 1
    total_training_steps: Optional[int] = None
    project_name: str = "polyrl_amd"
    experiment_name: str = "run"
    logger: List[str] = field(default_factory=lambda: ["console"])
    n_gpus_per_node: int = 8
    nnodes: int = 1
    save_freq: int = -1
    test_freq: int = -1
    critic_warmup: int = 0
    default_local_dir: str = "checkpoints"
    resume_mode: str = "auto"            # auto | disable | resume_path
    resume_from_path: Optional[str] = None
    val_before_train: bool = False
    device: str = "cuda"
    seed: int = 1
    # per-step profiling window (reference: global_profiler.tool=nsys with
    # profile steps, main_stream.py:79-93) — torch.profiler chrome traces
    profile_steps: List[int] = field(default_factory=list)
    profile_dir: str = "profiles/torch"
    # dump per-step rollout samples (uid, response ids, score) as jsonl
    rollout_data_dir: str = ""


@dataclass
class ActorRolloutRefConfig:
    model: ModelConfig = field(default_factory=ModelConfig)
    actor: ActorConfig = field(default_factory=ActorConfig)
    rollout: RolloutConfig = field(default_factory=RolloutConfig)
    ref: RefConfig = field(default_factory=RefConfig)


@dataclass
class SchedulerConfig:
    """In-process rollout scheduler knobs (replaces the Rust
    rollout-manager's config.toml; plumbed into scheduler/manager.py by the
    disagg/elastic coordinators).  The HTTP facade port lives on
    RolloutConfig.rollout_manager_port."""
    max_assigned_batches_per_stats_check: int = 4
    health_check_interval_s: float = 1.0
    max_retries: int = 5
    scheduling_policy: str = "zero_queue_rr"   # | least_loaded


@dataclass
class SandboxFusionConfig:
    """Remote code-execution sandbox (reference:
    reward_model.sandbox_fusion {url, max_concurrent, memory_limit_mb},
    trainer/ppo/reward.py:128-141)."""
    url: Optional[str] = None
    max_concurrent: int = 64
    memory_limit_mb: int = 1024


@dataclass
class CustomRewardFunctionConfig:
    """Custom scoring fn loaded from a python file (reference:
    custom_reward_function {path, name}, reward.py:60-93)."""
    path: Optional[str] = None
    name: str = "compute_score"


@dataclass
class RewardModelConfig:
    """Reward-manager selection (reference: config.reward_model tree,
    reward.py:95-150 — naive | prime | batch | dapo registry)."""
    reward_manager: str = "naive"
    sandbox_fusion: SandboxFusionConfig = field(
        default_factory=SandboxFusionConfig)
    # dapo manager knobs
    overlong_buffer_len: int = 0
    overlong_penalty_factor: float = 1.0


@dataclass
class PPOConfig:
    data: DataConfig = field(default_factory=DataConfig)
    actor_rollout_ref: ActorRolloutRefConfig = field(default_factory=ActorRolloutRefConfig)
    critic: CriticConfig = field(default_factory=CriticConfig)
    algorithm: AlgorithmConfig = field(default_factory=AlgorithmConfig)
    trainer: TrainerConfig = field(default_factory=TrainerConfig)
    scheduler: SchedulerConfig = field(default_factory=SchedulerConfig)
    reward_model: RewardModelConfig = field(default_factory=RewardModelConfig)
    custom_reward_function: CustomRewardFunctionConfig = field(
        default_factory=CustomRewardFunctionConfig)


# ---------------------------------------------------------------- overrides


def _coerce(value: str, current: Any) -> Any:
    if isinstance(current, bool):
        return value.lower() in ("1", "true", "yes")
    if isinstance(current, int) and not isinstance(current, bool):
        return int(value)
    if isinstance(current, float):
        return float(value)
    if isinstance(current, (list, tuple, dict)) or current is None:
        try:
            return yaml.safe_load(value)
        except Exception:
            return value
    return value


def apply_overrides(cfg: Any, overrides: List[str]) -> Any:
    """Apply `a.b.c=value` dot-path overrides in place."""
    for ov in overrides:
        if "=" not in ov:
            raise ValueError(f"override {ov!r} must be key=value")
        path, value = ov.split("=", 1)
        keys = path.split(".")
        obj = cfg
        for k in keys[:-1]:
            obj = getattr(obj, k)
        leaf = keys[-1]
        current = getattr(obj, leaf)
        setattr(obj, leaf, _coerce(value, current))
    return cfg


def to_dict(cfg: Any) -> dict:
    return dataclasses.asdict(cfg)


def load_config(yaml_path: Optional[str] = None,
                overrides: Optional[List[str]] = None) -> PPOConfig:
    cfg = PPOConfig()
    if yaml_path:
        with open(yaml_path) as f:
            data = yaml.safe_load(f) or {}
        _merge_dict_into(cfg, data)
    if overrides:
        apply_overrides(cfg, overrides)
    return cfg


def _merge_dict_into(obj: Any, data: dict):
    for k, v in data.items():
        if not hasattr(obj, k):
            raise KeyError(f"unknown config key {k!r} on {type(obj).__name__}")
        cur = getattr(obj, k)
        if dataclasses.is_dataclass(cur) and isinstance(v, dict):
            _merge_dict_into(cur, v)
        else:
            setattr(obj, k, v)
