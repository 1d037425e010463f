"""User-facing experiment configs.

Reference semantics: realhf/api/quickstart/{model,dataset,entrypoint}.py +
realhf/experiments/common/*.py (CommonExperimentConfig:58, PPOConfig
ppo_exp.py:95, SFTConfig, RWConfig, DPOConfig, GenerationConfig) — the
dataclass surface users drive from the CLI.  Hydra is replaced by a small
dotted-override parser (apps/quickstart.py); the dataclass schema is the
same idea: structured configs translated into (DFG, allocations, worker
state).
"""
import dataclasses
from typing import Union, Any, Dict, Optional

from realhf_amd.api.config import ParallelismConfig
from realhf_amd.api.model import GenerationHyperparameters
from realhf_amd.parallel.ddp import OptimizerConfig


@dataclasses.dataclass
class LoRAConfig:
    """Low-rank adaptation of the attention projections
    (reference: api/quickstart/model.py:56 LoRAConfig + the "lora"
    ModelWrapperAbstraction wrapping c_attn/c_proj with a "squash" op).
    delta_W = (scaling / dim) * B @ A; only A/B train."""

    dim: int = 32
    scaling: float = 32.0
    squash_after_train: bool = False  # merge adapters into base on save


@dataclasses.dataclass
class ModelTrainEvalConfig:
    """One role's model (reference: api/quickstart/model.py:114)."""

    path: Optional[str] = None  # HF checkpoint dir; None => random init
    family: str = "llama"
    is_critic: bool = False
    init_critic_from_actor: bool = False
    dtype: str = "bfloat16"
    gradient_checkpointing: bool = False
    # False | True ("cpu": pinned-host offload) | "dp_shard" (ZeRO-3-style
    # sharding over the DP group, restored by one xGMI all-gather)
    offload: Union[bool, str] = False
    optimizer: OptimizerConfig = dataclasses.field(default_factory=OptimizerConfig)
    parallel: ParallelismConfig = dataclasses.field(default_factory=ParallelismConfig)
    # gen-phase override strategy (enables param realloc between phases)
    gen_parallel: Optional[ParallelismConfig] = None
    lora: Optional[LoRAConfig] = None  # train adapters only (base frozen)


@dataclasses.dataclass
class DatasetConfig:
    path: Optional[str] = None
    valid_path: Optional[str] = None  # eval split (reference: valid_path)
    type_: str = "prompt"  # prompt | prompt_answer | rw_paired | synthetic_prompt
    max_prompt_len: int = 256
    max_seqlen: int = 1024
    train_bs_n_seqs: int = 128
    args: Dict[str, Any] = dataclasses.field(default_factory=dict)


@dataclasses.dataclass
class ExperimentSaveEvalControl:
    """reference: system_api.py:157 — save/eval by epochs, steps or
    wall-clock seconds, whichever fires first."""

    total_train_epochs: int = 1
    save_freq_epochs: Optional[int] = None
    save_freq_steps: Optional[int] = None
    save_freq_secs: Optional[float] = None
    eval_freq_epochs: Optional[int] = None
    eval_freq_steps: Optional[int] = None
    eval_freq_secs: Optional[float] = None
    benchmark_steps: Optional[int] = None  # early-exit for throughput runs


@dataclasses.dataclass
class PPOHyperparameters:
    gen: GenerationHyperparameters = dataclasses.field(
        default_factory=lambda: GenerationHyperparameters(max_new_tokens=256)
    )
    ppo_n_minibatches: int = 4
    kl_ctl: float = 0.1
    use_adaptive_kl_ctl: bool = False
    eps_clip: float = 0.2
    value_eps_clip: float = 0.2
    max_reward_clip: float = 20.0
    reward_output_scaling: float = 1.0
    reward_output_bias: float = 0.0
    discount: float = 1.0
    gae_lambda: float = 1.0
    adv_norm: bool = True
    value_norm: bool = True
    value_norm_type: str = "exp"  # exp | ma (reference ppo_exp.py:62-68)
    value_norm_beta: float = 0.99995
    value_norm_eps: float = 1e-5
    early_stop_imp_ratio: Optional[float] = None


@dataclasses.dataclass
class SlurmConfig:
    """Cluster options for mode=slurm (reference: scheduler/slurm/utils.py
    SlurmLaunchInfo fields + cluster spec)."""

    partition: Optional[str] = None
    account: Optional[str] = None
    time_limit: Optional[str] = None
    container_image: Optional[str] = None
    container_mounts: Optional[str] = None
    gpus_per_node: int = 8
    mem_per_node: Optional[str] = None


@dataclasses.dataclass
class CommonExperimentConfig:
    experiment_name: str = "exp"
    trial_name: str = "trial"
    mode: str = "local"  # local | slurm
    slurm: SlurmConfig = dataclasses.field(default_factory=SlurmConfig)
    n_gpus: int = 1
    seed: int = 1
    allocation_mode: str = "global"  # global | manual | heuristic | search | d8t1p1-style
    exp_ctrl: ExperimentSaveEvalControl = dataclasses.field(
        default_factory=ExperimentSaveEvalControl
    )
    dataset: DatasetConfig = dataclasses.field(default_factory=DatasetConfig)
    tokenizer_path: Optional[str] = None
    recover_mode: str = "disabled"  # disabled | auto | resume
    # gang restarts on worker failure (pairs with recover_mode=auto)
    max_restarts: int = 0


@dataclasses.dataclass
class SFTConfig(CommonExperimentConfig):
    model: ModelTrainEvalConfig = dataclasses.field(
        default_factory=ModelTrainEvalConfig
    )


@dataclasses.dataclass
class RWConfig(CommonExperimentConfig):
    model: ModelTrainEvalConfig = dataclasses.field(
        default_factory=lambda: ModelTrainEvalConfig(is_critic=True)
    )


@dataclasses.dataclass
class DPOConfig(CommonExperimentConfig):
    actor: ModelTrainEvalConfig = dataclasses.field(
        default_factory=ModelTrainEvalConfig
    )
    ref: ModelTrainEvalConfig = dataclasses.field(
        default_factory=ModelTrainEvalConfig
    )
    beta: float = 0.1


@dataclasses.dataclass
class PPOConfig(CommonExperimentConfig):
    actor: ModelTrainEvalConfig = dataclasses.field(
        default_factory=ModelTrainEvalConfig
    )
    critic: ModelTrainEvalConfig = dataclasses.field(
        default_factory=lambda: ModelTrainEvalConfig(is_critic=True)
    )
    ref: ModelTrainEvalConfig = dataclasses.field(
        default_factory=lambda: ModelTrainEvalConfig(offload=True)
    )
    rew: ModelTrainEvalConfig = dataclasses.field(
        default_factory=lambda: ModelTrainEvalConfig(is_critic=True, offload=True)
    )
    ppo: PPOHyperparameters = dataclasses.field(default_factory=PPOHyperparameters)


@dataclasses.dataclass
class GRPOConfig(CommonExperimentConfig):
    actor: ModelTrainEvalConfig = dataclasses.field(
        default_factory=ModelTrainEvalConfig
    )
    ref: ModelTrainEvalConfig = dataclasses.field(
        default_factory=lambda: ModelTrainEvalConfig(offload=True)
    )
    rew: ModelTrainEvalConfig = dataclasses.field(
        default_factory=lambda: ModelTrainEvalConfig(is_critic=True, offload=True)
    )
    ppo: PPOHyperparameters = dataclasses.field(default_factory=PPOHyperparameters)
    group_size: int = 4
    kl_in_loss_coef: float = 0.0


@dataclasses.dataclass
class GenerationConfig(CommonExperimentConfig):
    model: ModelTrainEvalConfig = dataclasses.field(
        default_factory=ModelTrainEvalConfig
    )
    gen: GenerationHyperparameters = dataclasses.field(
        default_factory=GenerationHyperparameters
    )
    # jsonl dump of generations under LOG_ROOT (reference gen_exp.py:49);
    # None disables
    output_file: Optional[str] = "output.jsonl"


@dataclasses.dataclass
class ProfileConfig(CommonExperimentConfig):
    """Time interfaces across parallel strategies with mock data
    (reference: ProfileConfig, experiments/benchmark/profile_exp.py:61)."""

    model: ModelTrainEvalConfig = dataclasses.field(
        default_factory=ModelTrainEvalConfig
    )
    interfaces: str = "inference,train_step"  # comma list (+generate)
    strategies: str = "d1"  # ';'-separated d{dp}t{tp}p{pp} specs
    n_seqs: int = 16
    seq_len: int = 512
    gen_tokens: int = 128
    n_steps: int = 3
    warmup: int = 1


EXPERIMENT_TYPES = {
    "sft": SFTConfig,
    "rw": RWConfig,
    "dpo": DPOConfig,
    "ppo": PPOConfig,
    "grpo": GRPOConfig,
    "gen": GenerationConfig,
    "profile": ProfileConfig,
}
