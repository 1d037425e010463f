"""Experiment builder + training loop (the SPMD 'controller').

Reference semantics: realhf/experiments/common/*.py (graph construction
ppo_exp.py:261-377, resolve_replica_ids/resolve_rpc_hooks utils.py:126/143)
+ the master worker's epoch/save/eval loop (master_worker.py:1273-1499),
re-done as a deterministic SPMD program (see runtime/executor.py).
"""
import dataclasses
import os
import pickle
import time
from typing import Dict, List

import torch
import torch.distributed as dist

import realhf_amd.interfaces  # noqa: F401 — register interfaces
import realhf_amd.runtime.engine  # noqa: F401 — register backends
from realhf_amd.api import datasets as ds_impl  # noqa: F401 — register datasets
from realhf_amd.api.config import (
    Abstraction,
    ModelInterfaceType,
    ModelName,
    ParallelismConfig,
    parse_parallelism,
)
from realhf_amd.api.data import PackedDataLoader, make_dataset
from realhf_amd.api.dfg import DFG, MFCDef, OffloadHook, ParamReallocHook, build_graph
from realhf_amd.api.experiment import (
    CommonExperimentConfig,
    DPOConfig,
    GenerationConfig,
    GRPOConfig,
    PPOConfig,
    RWConfig,
    SFTConfig,
)
from realhf_amd.api.model import (
    FinetuneSpec,
    Model,
    make_backend,
    make_interface,
)
from realhf_amd.base import constants, logging, seeding
from realhf_amd.base.topology import ParallelGrid, PipeDataTensorTopology
from realhf_amd.models import hf as hf_reg
from realhf_amd.models.real_model import ReaLModel
from realhf_amd.parallel.realloc import ParallelStrategy
from realhf_amd.runtime.executor import DFGExecutor, MFCAllocation

logger = logging.getLogger("trainer")


def _strategy_for(par: ParallelismConfig, world: int) -> ParallelStrategy:
    ws = par.world_size
    assert ws <= world, (par, world)
    return ParallelStrategy.make(
        par.pipeline_parallel_size, par.data_parallel_size,
        par.tensor_parallel_size, ranks=list(range(ws)),
    )


def _resolve_parallel(cfg: CommonExperimentConfig, model_par: ParallelismConfig,
                      world: int, param_count: int = 0) -> ParallelismConfig:
    mode = cfg.allocation_mode
    if mode == "global":
        return ParallelismConfig(data_parallel_size=world)
    if mode == "heuristic":
        # 288 GB HBM per MI355X: a trainable model needs ~16 bytes/param
        # (bf16 params+grads + fp32 master/m/v); pure DP while that fits,
        # else TP across the node (ZeRO already shards optimizer states
        # over dp, so the bound is params+grads+activations)
        if param_count * 4 < 230e9 or world == 1:
            return ParallelismConfig(data_parallel_size=world)
        tp = min(8, world)
        return ParallelismConfig(
            data_parallel_size=world // tp, tensor_parallel_size=tp
        )
    if mode == "manual":
        return model_par
    return parse_parallelism(mode)


@dataclasses.dataclass
class BuiltExperiment:
    graph: DFG
    allocations: Dict[str, MFCAllocation]
    model_strategies: Dict[ModelName, ParallelStrategy]
    model_cfgs: Dict[ModelName, object]  # ReaLModelConfig
    model_roles: Dict[ModelName, object]  # ModelTrainEvalConfig
    interfaces: Dict[str, object]
    trainable: List[ModelName]


def _apply_search_allocation(cfg, world: int):
    """allocation_mode=search: pick per-MFC strategies with the MCMC
    search engine, then write them back into the model configs (reference:
    apps/main.py experiment._search() -> search_rpc_allocations).
    Handles any experiment type whose roles expose .parallel
    (PPO, GRPO, DPO, ...)."""
    from realhf_amd.search.engine import MFCSpec, search_allocations

    role_cfgs = {
        name: getattr(cfg, name)
        for name in ("actor", "critic", "ref", "rew", "model")
        if hasattr(cfg, name)
    }
    if not role_cfgs:
        logger.warning("allocation_mode=search: no role configs on %s; "
                       "using heuristic", type(cfg).__name__)
        cfg.allocation_mode = "heuristic"
        return

    T = ModelInterfaceType
    tmp = build_experiment(
        dataclasses.replace(cfg, allocation_mode="heuristic"), world
    )
    gen_max = (cfg.ppo.gen.max_new_tokens if hasattr(cfg, "ppo")
               else getattr(getattr(cfg, "gen", None), "max_new_tokens", 0))
    prompt_len = getattr(cfg.dataset, "max_prompt_len", None) or 512
    full_len = prompt_len + gen_max if gen_max else (
        getattr(cfg.dataset, "max_seqlen", None) or 1024)
    specs = {}
    for m in tmp.graph.mfcs:
        rcfg = tmp.model_cfgs[m.model_name]
        p = rcfg.param_count()
        is_gen = m.interface_type == T.GENERATE
        specs[m.name] = MFCSpec(
            name=m.name, role=m.model_name.role,
            interface_type=m.interface_type,
            n_seqs=cfg.dataset.train_bs_n_seqs,
            avg_seqlen=prompt_len if is_gen else full_len,
            gen_tokens=gen_max if is_gen else 0,
            param_bytes=p * 2.0, flops_per_token=2.0 * p,
            n_layers=rcfg.n_layers, hidden_dim=rcfg.hidden_dim,
            n_kv_heads=rcfg.n_kv_heads, head_dim=rcfg.head_dim,
            gradient_checkpointing=getattr(
                tmp.model_roles[m.model_name], "gradient_checkpointing",
                False),
            offload_optimizer=getattr(
                tmp.model_roles[m.model_name].optimizer, "offload", False),
            n_minibatches=getattr(getattr(cfg, "ppo", None),
                                  "ppo_n_minibatches", 1),
        )
    trainable_roles = sorted({n.role for n in tmp.trainable})
    alloc, cost = search_allocations(
        tmp.graph, specs, trainable_roles=trainable_roles, n_gpus=world,
    )
    cfg.allocation_mode = "manual"
    # write back: a role's layout = its TRAIN_STEP mfc if trainable, else
    # its first mfc; a GENERATE mfc with a different layout becomes the
    # gen replica (realloc hooks engage in build_experiment)
    by_role: Dict[str, List] = {}
    for m in tmp.graph.mfcs:
        by_role.setdefault(m.model_name.role, []).append(m)
    for role, ms in by_role.items():
        mc = role_cfgs.get(role) or role_cfgs.get("model")
        if mc is None:
            continue
        train = [m for m in ms if m.interface_type == T.TRAIN_STEP]
        main = train[0] if train else ms[0]
        mc.parallel = alloc[main.name]
        gen = [m for m in ms if m.interface_type == T.GENERATE]
        if gen and alloc[gen[0].name] != alloc[main.name]:
            mc.gen_parallel = alloc[gen[0].name]
    logger.info("search allocation (est %.3fs/step): %s", cost,
                {k: str(v) for k, v in alloc.items()})


def build_experiment(cfg: CommonExperimentConfig, world: int) -> BuiltExperiment:
    T = ModelInterfaceType
    if cfg.allocation_mode == "search":
        _apply_search_allocation(cfg, world)

    def model_cfg_of(mc, name):
        if mc.path:
            rcfg = hf_reg.config_from_hf_path(mc.family, mc.path)
        else:
            fam = hf_reg.get_family(mc.family)
            rcfg = fam.make_test_config()
            rcfg.family = mc.family
        rcfg.is_critic = mc.is_critic
        rcfg.dtype = mc.dtype
        return rcfg

    mfcs: List[MFCDef] = []
    interfaces: Dict[str, object] = {}
    model_strategies: Dict[ModelName, ParallelStrategy] = {}
    model_cfgs: Dict[ModelName, object] = {}
    model_roles: Dict[ModelName, object] = {}
    allocations: Dict[str, MFCAllocation] = {}
    trainable: List[ModelName] = []

    def add_model(role, mc, replica=0, ranks=None):
        name = ModelName(role, replica)
        rcfg = model_cfg_of(mc, name)
        par = _resolve_parallel(cfg, mc.parallel, world, rcfg.param_count())
        if (ranks is not None and par.tensor_parallel_size == 1
                and par.pipeline_parallel_size == 1):
            # asymmetric heuristic: model lives only on a rank subset,
            # pure DP over it (288 GB replication design point).  Models
            # whose resolved layout needs TP/PP keep the full mesh.
            ranks = list(ranks)
            par = ParallelismConfig(data_parallel_size=len(ranks))
            strat = ParallelStrategy.make(1, len(ranks), 1, ranks=ranks)
        else:
            strat = _strategy_for(par, world)
        if rcfg.moe is not None:
            ep = min(rcfg.moe.expert_parallel_size, strat.dp)
            if ep > 1 and strat.dp % ep == 0:
                strat = dataclasses.replace(strat, ep=ep)
        model_strategies[name] = strat
        model_cfgs[name] = rcfg
        model_roles[name] = mc
        return name, par

    def add_mfc(name, model_name, itype, iface_cfg, inp, out, mc, par,
                n_mbs=None, alloc_strategy=None, priority=0):
        mfcs.append(
            MFCDef(
                name=name, model_name=model_name, interface_type=itype,
                interface_impl=iface_cfg, input_keys=tuple(inp),
                output_keys=tuple(out), n_seqs=cfg.dataset.train_bs_n_seqs,
                priority=priority,
            )
        )
        interfaces[name] = make_interface(iface_cfg)
        allocations[name] = MFCAllocation(
            strategy=alloc_strategy or model_strategies[model_name],
            sequence_parallel=par.sequence_parallel,
            gradient_checkpointing=mc.gradient_checkpointing,
            n_mbs=n_mbs,
        )

    if isinstance(cfg, SFTConfig):
        name, par = add_model("default", cfg.model)
        trainable.append(name)
        add_mfc("train", name, T.TRAIN_STEP, Abstraction("sft"),
                ["packed_input_ids", "prompt_mask"], [], cfg.model, par)
    elif isinstance(cfg, RWConfig):
        name, par = add_model("default", cfg.model)
        trainable.append(name)
        add_mfc("train", name, T.TRAIN_STEP, Abstraction("paired_rw"),
                ["packed_input_ids"], [], cfg.model, par)
    elif isinstance(cfg, DPOConfig):
        actor, apar = add_model("actor", cfg.actor)
        ref, rpar = add_model("ref", cfg.ref)
        trainable.append(actor)
        add_mfc("ref_inf", ref, T.INFERENCE, Abstraction("dpo", {"beta": cfg.beta}),
                ["packed_input_ids", "prompt_mask"], ["seqlogp"], cfg.ref, rpar)
        add_mfc("dpo_train", actor, T.TRAIN_STEP,
                Abstraction("dpo", {"beta": cfg.beta}),
                ["packed_input_ids", "prompt_mask", "seqlogp"], [],
                cfg.actor, apar)
        if cfg.ref.offload:
            g = [m for m in mfcs if m.name == "ref_inf"][0]
            g.post_hooks.append(OffloadHook())
    elif isinstance(cfg, GRPOConfig):
        _build_grpo(cfg, world, add_model, add_mfc, mfcs, trainable)
    elif isinstance(cfg, PPOConfig):
        _build_ppo(cfg, world, add_model, add_mfc, mfcs, trainable)
    elif isinstance(cfg, GenerationConfig):
        name, par = add_model("default", cfg.model)
        add_mfc("gen", name, T.GENERATE,
                Abstraction("generation",
                            {"gconfig": dataclasses.asdict(cfg.gen),
                             "output_file": cfg.output_file}),
                ["packed_prompts"], ["packed_input_ids", "prompt_mask"],
                cfg.model, par)
    else:
        raise TypeError(cfg)

    graph = build_graph(mfcs)
    return BuiltExperiment(
        graph=graph, allocations=allocations,
        model_strategies=model_strategies, model_cfgs=model_cfgs,
        model_roles=model_roles, interfaces=interfaces, trainable=trainable,
    )


def _asym_inference_meshes(cfg, world: int):
    """MI355X heuristic (reference counterpart: ppo_exp.py:419-613,
    recalibrated for 288 GB HBM3E): with ≤~30B-param models every role
    fits fully replicated, so all layouts are pure DP and sub-mesh
    execution needs NO parameter reallocation (every rank already holds
    full weights).  The asymmetric lever that remains profitable is
    CONCURRENCY of the independent inference MFCs: critic_inf runs on
    the first half of the node while rew_inf runs on the second half,
    simultaneously (ref_inf keeps the whole node and runs first).  At
    80 GB the reference instead must shard/offload — its level-2/3/4
    split — which on this hardware would only add transfer cost."""
    if world < 2 or cfg.allocation_mode != "heuristic":
        return None
    half = world // 2
    return {
        "first_half": list(range(half)),
        "second_half": list(range(half, world)),
    }


def _build_ppo(cfg: PPOConfig, world, add_model, add_mfc, mfcs, trainable):
    T = ModelInterfaceType
    ppo = cfg.ppo
    asym = _asym_inference_meshes(cfg, world)
    actor, apar = add_model("actor", cfg.actor)
    critic, cpar = add_model("critic", cfg.critic)
    ref, refpar = add_model("ref", cfg.ref)
    # the reward model only ever serves rew_inf: under the asymmetric
    # heuristic it is instantiated on HALF the node only (memory saving +
    # concurrency with critic_inf on the other half)
    rew, rewpar = add_model("rew", cfg.rew,
                            ranks=asym["second_half"] if asym else None)
    trainable += [actor, critic]
    critic_inf_strategy = None
    if (asym and cpar.tensor_parallel_size == 1
            and cpar.pipeline_parallel_size == 1):
        critic_inf_strategy = ParallelStrategy.make(
            1, len(asym["first_half"]), 1, ranks=asym["first_half"]
        )

    gen_cfg = dataclasses.asdict(ppo.gen)
    actor_iface = Abstraction("ppo_actor", {
        "n_minibatches": ppo.ppo_n_minibatches, "gconfig": gen_cfg,
        "kl_ctl": ppo.kl_ctl, "adaptive_kl_ctl": ppo.use_adaptive_kl_ctl,
        "eps_clip": ppo.eps_clip, "max_reward_clip": ppo.max_reward_clip,
        "discount": ppo.discount, "gae_lambda": ppo.gae_lambda,
        "adv_norm": ppo.adv_norm,
        "early_stop_imp_ratio": ppo.early_stop_imp_ratio,
    })
    critic_iface = Abstraction("ppo_critic", {
        "n_minibatches": ppo.ppo_n_minibatches,
        "value_eps_clip": ppo.value_eps_clip, "kl_ctl": ppo.kl_ctl,
        "adaptive_kl_ctl": ppo.use_adaptive_kl_ctl,
        "max_reward_clip": ppo.max_reward_clip, "discount": ppo.discount,
        "gae_lambda": ppo.gae_lambda, "value_norm": ppo.value_norm,
        "value_norm_type": ppo.value_norm_type,
        "value_norm_beta": ppo.value_norm_beta,
        "value_norm_eps": ppo.value_norm_eps,
    })
    rw_iface = Abstraction("paired_rw", {
        "output_scaling": ppo.reward_output_scaling,
        "output_bias": ppo.reward_output_bias,
    })

    # separate generation replica when gen uses a different strategy —
    # the parameter-reallocation flow (reference: resolve_replica_ids +
    # resolve_rpc_hooks, experiments/common/utils.py:126/143)
    gen_model = actor
    gen_par = apar
    if cfg.actor.gen_parallel is not None and cfg.allocation_mode == "manual":
        gen_mc = dataclasses.replace(cfg.actor, parallel=cfg.actor.gen_parallel)
        gen_model, gen_par = add_model("actor", gen_mc, replica=1)
    # logits-mask mode (gen.force_no_logits_mask=False): the sampler's
    # top-k/p removal mask travels actor_gen -> ref_inf/actor_train so
    # re-forward logprobs match the sampled distribution (reference:
    # ppo_interface.py:245-246 emits packed_logits_mask the same way)
    mask_keys = ([] if ppo.gen.force_no_logits_mask
                 else ["packed_logits_mask"])
    add_mfc("actor_gen", gen_model, T.GENERATE, actor_iface,
            ["packed_prompts"],
            ["packed_input_ids", "packed_logprobs", "prompt_mask",
             "seq_no_eos_mask"] + mask_keys, cfg.actor, gen_par)
    if gen_model != actor:
        gen_mfc = mfcs[-1]
        gen_mfc.pre_hooks.append(
            ParamReallocHook(source=actor, target=gen_model)
        )
        gen_mfc.post_hooks.append(OffloadHook())
    # execution order under the asymmetric heuristic: ref_inf first on the
    # WHOLE node, then critic_inf (first half) and rew_inf (second half)
    # CONCURRENTLY — each sub-mesh rank skips the other MFC and proceeds
    add_mfc("ref_inf", ref, T.INFERENCE, actor_iface,
            ["packed_input_ids"] + mask_keys, ["packed_ref_logprobs"],
            cfg.ref, refpar, priority=-1)
    add_mfc("critic_inf", critic, T.INFERENCE, critic_iface,
            ["packed_input_ids"], ["values"], cfg.critic, cpar,
            alloc_strategy=critic_inf_strategy)
    add_mfc("rew_inf", rew, T.INFERENCE, rw_iface,
            ["packed_input_ids"], ["rewards"], cfg.rew, rewpar)
    train_keys = ["packed_input_ids", "packed_logprobs", "packed_ref_logprobs",
                  "rewards", "values", "prompt_mask", "seq_no_eos_mask"]
    # the critic never reads logits — keep the (large) mask off its mesh
    add_mfc("actor_train", actor, T.TRAIN_STEP, actor_iface,
            train_keys + mask_keys, [], cfg.actor, apar)
    add_mfc("critic_train", critic, T.TRAIN_STEP, critic_iface, train_keys, [],
            cfg.critic, cpar)

    for m in mfcs:
        if m.name == "rew_inf" and cfg.rew.offload:
            m.post_hooks.append(OffloadHook(to=_offload_style(cfg.rew)))
        if m.name == "ref_inf" and cfg.ref.offload:
            m.post_hooks.append(OffloadHook(to=_offload_style(cfg.ref)))


def _offload_style(mc) -> str:
    return mc.offload if isinstance(mc.offload, str) else "cpu"


def _build_grpo(cfg: GRPOConfig, world, add_model, add_mfc, mfcs, trainable):
    T = ModelInterfaceType
    ppo = cfg.ppo
    asym = _asym_inference_meshes(cfg, world)
    actor, apar = add_model("actor", cfg.actor)
    # GRPO has no critic: ref_inf and rew_inf are the whole inference
    # level — give each half the node and run them concurrently
    ref, refpar = add_model("ref", cfg.ref,
                            ranks=asym["first_half"] if asym else None)
    rew, rewpar = add_model("rew", cfg.rew,
                            ranks=asym["second_half"] if asym else None)
    trainable.append(actor)
    iface = Abstraction("grpo", {
        "n_minibatches": ppo.ppo_n_minibatches,
        "gconfig": dataclasses.asdict(ppo.gen),
        "kl_ctl": ppo.kl_ctl, "eps_clip": ppo.eps_clip,
        "group_size": cfg.group_size, "kl_in_loss_coef": cfg.kl_in_loss_coef,
    })
    rw_iface = Abstraction("paired_rw", {
        "output_scaling": ppo.reward_output_scaling,
        "output_bias": ppo.reward_output_bias,
    })
    mask_keys = ([] if ppo.gen.force_no_logits_mask
                 else ["packed_logits_mask"])
    add_mfc("actor_gen", actor, T.GENERATE, iface, ["packed_prompts"],
            ["packed_input_ids", "packed_logprobs", "prompt_mask",
             "seq_no_eos_mask"] + mask_keys, cfg.actor, apar)
    add_mfc("rew_inf", rew, T.INFERENCE, rw_iface, ["packed_input_ids"],
            ["rewards"], cfg.rew, rewpar)
    add_mfc("ref_inf", ref, T.INFERENCE, iface,
            ["packed_input_ids"] + mask_keys,
            ["packed_ref_logprobs"], cfg.ref, refpar)
    add_mfc("actor_train", actor, T.TRAIN_STEP, iface,
            ["packed_input_ids", "packed_logprobs", "packed_ref_logprobs",
             "rewards", "prompt_mask"] + mask_keys, [], cfg.actor, apar)
    for m in mfcs:
        if m.name == "rew_inf" and cfg.rew.offload:
            m.post_hooks.append(OffloadHook(to=_offload_style(cfg.rew)))
        if m.name == "ref_inf" and cfg.ref.offload:
            m.post_hooks.append(OffloadHook(to=_offload_style(cfg.ref)))


# ---------------------------------------------------------------------------
class Trainer:
    def __init__(self, cfg: CommonExperimentConfig):
        self.cfg = cfg
        self.rank = dist.get_rank() if dist.is_initialized() else 0
        self.world = dist.get_world_size() if dist.is_initialized() else 1
        seeding.set_random_seed(cfg.seed)
        constants.set_experiment_trial_names(cfg.experiment_name, cfg.trial_name)
        self.built = build_experiment(cfg, self.world)
        self.device = torch.device(
            "cuda", int(os.environ.get("LOCAL_RANK", 0))
        ) if torch.cuda.is_available() else torch.device("cpu")
        self.tokenizer = self._load_tokenizer()
        # The dataloader is built BEFORE the models so FinetuneSpec sees the
        # real dataset length (LR warmup/decay horizon depends on it).
        self.train_dl = self._build_dataloader()
        self.models = self._build_models()
        self.executor = DFGExecutor(
            self.built.graph, self.built.allocations, self.models,
            self.built.interfaces, self.built.model_strategies,
        )
        self.global_step = 0

    def _load_tokenizer(self):
        path = self.cfg.tokenizer_path
        if path is None:
            for mc in self.built.model_roles.values():
                if mc.path:
                    path = mc.path
                    break
        if path is None:
            return None
        try:
            import transformers

            return transformers.AutoTokenizer.from_pretrained(path)
        except Exception as e:  # pragma: no cover
            logger.warning("tokenizer load failed: %s", e)
            return None

    def _build_models(self) -> Dict[ModelName, Model]:
        models = {}
        for name, strat in self.built.model_strategies.items():
            my_coord = None
            for (p, d, t), r in strat.rank_map:
                if r == self.rank:
                    my_coord = (p, d, t)
            # register grid (group creation is collective — all ranks join)
            mc0 = self.built.model_roles[name]
            rcfg0 = self.built.model_cfgs[name]
            ep = strat.ep
            topo = PipeDataTensorTopology(
                num_pp=strat.pp, num_dp=strat.dp, num_tp=strat.tp,
                sequence_parallel=mc0.parallel.sequence_parallel and strat.tp > 1,
                gradient_checkpointing=mc0.gradient_checkpointing,
                ep_size=ep,
            )
            rank_mapping = {
                topo.get_rank(pipe=p, data=d, tensor=t): r
                for (p, d, t), r in strat.rank_map
            }
            if dist.is_initialized():
                grid = ParallelGrid(topo, rank_mapping)
            else:
                from realhf_amd.base.topology import FakeGrid

                grid = FakeGrid(0, topo)
            constants.set_grid(str(name), grid)
            if my_coord is None:
                continue
            p_, d_, t_ = my_coord
            rcfg = self.built.model_cfgs[name]
            mc = self.built.model_roles[name]
            ep_size = getattr(topo, "ep_size", 1)
            with constants.model_scope(str(name)):
                m = ReaLModel(
                    rcfg, device=self.device, dtype=rcfg.torch_dtype,
                    tp_rank=t_, tp_size=strat.tp, pp_rank=p_, pp_size=strat.pp,
                    ep_rank=(d_ % ep_size) if ep_size > 1 else 0,
                    ep_size=ep_size,
                )
                if mc.path:
                    hf_reg.load_from_hf(m, mc.family, mc.path)
                else:
                    m.random_init()
                if getattr(mc, "lora", None) is not None and name in self.built.trainable:
                    lc = mc.lora
                    if isinstance(lc, dict):
                        from realhf_amd.api.experiment import LoRAConfig

                        lc = LoRAConfig(**lc)
                    m.attach_lora(dim=lc.dim, scaling=lc.scaling)
                model = Model(
                    name=name, module=m, tokenizer=self.tokenizer,
                    device=self.device, dtype=rcfg.torch_dtype,
                )
                spec = FinetuneSpec(
                    self.cfg.exp_ctrl.total_train_epochs,
                    len(self.train_dl.dataset),
                    self.cfg.dataset.train_bs_n_seqs,
                )
                if name in self.built.trainable:
                    backend = make_backend(
                        Abstraction("zero1", {"optimizer": dataclasses.asdict(mc.optimizer)})
                    )
                else:
                    backend = make_backend(Abstraction("inference"))
                models[name] = backend.initialize(model, spec)
            if mc.offload and name not in self.built.trainable:
                real = models[name].module.model
                if _offload_style(mc) == "dp_shard":
                    with constants.model_scope(str(name)):
                        real.shard_to_dp()
                else:
                    real.async_offload(non_blocking=False)
        return models

    def _build_dataloader(self, valid: bool = False):
        d = self.cfg.dataset
        path = d.valid_path if valid else d.path
        ds_cfg = Abstraction(d.type_, dict(d.args))
        if d.type_ == "prompt":
            ds_cfg.args.setdefault("path", path)
            ds_cfg.args.setdefault("max_prompt_len", d.max_prompt_len)
        elif d.type_ == "prompt_answer":
            ds_cfg.args.setdefault("path", path)
            ds_cfg.args.setdefault("max_seqlen", d.max_seqlen)
        elif d.type_ == "rw_paired":
            ds_cfg.args.setdefault("path", path)
            ds_cfg.args.setdefault("max_seqlen", d.max_seqlen)
        else:
            ds_cfg.args["path"] = path
        # SPMD: every rank builds the identical dataset/loader (same seed)
        dataset = make_dataset(ds_cfg, seed=self.cfg.seed, dp_rank=0,
                               world_size=1, tokenizer=self.tokenizer)
        return PackedDataLoader(
            dataset, batch_n_seqs=d.train_bs_n_seqs, shuffle=not valid,
            seed=self.cfg.seed,
        )

    def evaluate(self) -> Dict[str, float]:
        """Run each trainable model's interface.evaluate over the eval
        split (reference: master worker eval step, ModelInterfaceType
        .EVALUATE requests to model workers)."""
        if not self.cfg.dataset.valid_path:
            return {}
        eval_dl = self._build_dataloader(valid=True)
        out: Dict[str, float] = {}
        for name in self.built.trainable:
            if name not in self.models:
                continue
            for mfc in self.built.graph.mfcs:
                if (mfc.model_name == name
                        and mfc.interface_type == ModelInterfaceType.TRAIN_STEP):
                    iface = self.built.interfaces[mfc.name]
                    if not hasattr(iface, "evaluate"):
                        break
                    with constants.model_scope(str(name)):
                        try:
                            stats = iface.evaluate(self.models[name], eval_dl)
                        except NotImplementedError:
                            break
                    out.update({f"{name.role}/{k}": v for k, v in stats.items()})
                    break
        if out and self.rank == 0:
            logger.info("eval @ step %d: %s", self.global_step,
                        {k: round(v, 4) for k, v in out.items()})
        return out

    def _fire(self, fc, **kw) -> bool:
        """Evaluate a FrequencyControl SPMD-safely.  Step/epoch triggers
        are deterministic across ranks; a wall-clock trigger is not (save
        and eval run collectives, so ranks must agree) — rank 0 decides
        and broadcasts the flag (reference: the master worker decides
        centrally, master_worker.py:820-838)."""
        if fc is None:
            return False
        fire = fc.check(**kw)
        if fc.freq_sec is not None and dist.is_initialized():
            t = torch.tensor([1 if fire else 0])
            dist.broadcast(t, src=0)
            fire = bool(t.item())
            if fire:  # keep non-zero ranks' clocks aligned with rank 0
                fc._last_time = time.monotonic()
                fc._last_step = fc._last_epoch = 0
        return fire

    # ----------------------------------------------------------------- run
    def run(self):
        cfg = self.cfg
        dl = self.train_dl
        ctrl = cfg.exp_ctrl
        bench_t0 = None
        from realhf_amd.base.timeutil import FrequencyControl

        save_ctl = FrequencyControl(
            freq_epoch=getattr(ctrl, "save_freq_epochs", None),
            freq_step=ctrl.save_freq_steps,
            freq_sec=getattr(ctrl, "save_freq_secs", None))
        eval_ctl = FrequencyControl(
            freq_epoch=getattr(ctrl, "eval_freq_epochs", None),
            freq_step=ctrl.eval_freq_steps,
            freq_sec=getattr(ctrl, "eval_freq_secs", None))
        save_any = (save_ctl.freq_epoch or save_ctl.freq_step
                    or save_ctl.freq_sec)
        eval_any = (eval_ctl.freq_epoch or eval_ctl.freq_step
                    or eval_ctl.freq_sec)
        recover = self._maybe_load_recover()
        start_epoch, start_step = (recover or (0, 0))
        for epoch in range(start_epoch, ctrl.total_train_epochs):
            for i, batch in enumerate(dl):
                if epoch == start_epoch and i < start_step:
                    continue
                t0 = time.time()
                stats = self.executor.run_step(batch)
                dt = time.time() - t0
                self.global_step += 1
                if self.rank == 0:
                    toks = sum(batch.main_seqlens())
                    perf = {k: v for k, v in stats.items()
                            if k.endswith(("/wall_s", "/tflops_per_gpu"))}
                    logger.info(
                        "epoch %d step %d (global %d): %.2fs (%.0f tok/s, "
                        "%.2f samples/s) %s",
                        epoch, i, self.global_step, dt, toks / dt,
                        batch.bs / dt,
                        {k: round(v, 4) for k, v in stats.items()
                         if isinstance(v, float) and k not in perf},
                    )
                    if perf:
                        # reference-style per-MFC perf line
                        # (master_worker.py:1461-1488)
                        logger.info(
                            "perf: %s",
                            " | ".join(
                                f"{m}: {stats.get(m + '/wall_s', 0):.2f}s"
                                + (f" {stats[m + '/tflops_per_gpu']:.0f} TF/s"
                                   if m + "/tflops_per_gpu" in stats else "")
                                for m in sorted({k.split("/")[0]
                                                 for k in perf})
                            ),
                        )
                if save_any and self._fire(save_ctl, steps=1):
                    self.save()
                    self.save_recover_ckpt(epoch, i + 1)
                if eval_any and self._fire(eval_ctl, steps=1):
                    self.evaluate()
                if ctrl.benchmark_steps:
                    if bench_t0 is None:
                        bench_t0 = time.time()
                        bench_step0 = self.global_step
                    elif self.global_step - bench_step0 >= ctrl.benchmark_steps:
                        el = time.time() - bench_t0
                        n = self.global_step - bench_step0
                        if self.rank == 0:
                            logger.info(
                                "benchmark: %d steps, %.3fs/step, %.2f samples/s",
                                n, el / n,
                                n * cfg.dataset.train_bs_n_seqs / el,
                            )
                        self._save_recover_info(epoch, i)
                        return
            self.save_recover_ckpt(epoch + 1, 0)
            if save_any and self._fire(save_ctl, epochs=1):
                self.save()
            if eval_any and self._fire(eval_ctl, epochs=1):
                self.evaluate()
        if save_any:
            self.save()

    def save(self):
        for name in self.built.trainable:
            if name not in self.models:
                continue
            save_dir = os.path.join(
                constants.MODEL_SAVE_ROOT(self.cfg.experiment_name, self.cfg.trial_name),
                name.role, f"globalstep{self.global_step}",
            )
            # find the mfc interface owning this model to save via it
            for mfc in self.built.graph.mfcs:
                if mfc.model_name == name and mfc.interface_type == ModelInterfaceType.TRAIN_STEP:
                    with constants.model_scope(str(name)):
                        self.built.interfaces[mfc.name].save(
                            self.models[name], save_dir
                        )
                    if self.rank == 0:
                        logger.info("saved %s -> %s", name, save_dir)
                    break

    # -------------------------------------------------------------- recover
    def _recover_path(self):
        return os.path.join(
            constants.RECOVER_ROOT(self.cfg.experiment_name, self.cfg.trial_name),
            "recover_info.pkl",
        )

    def _save_recover_info(self, epoch, step):
        if self.cfg.recover_mode == "disabled" or self.rank != 0:
            return
        with open(self._recover_path(), "wb") as f:
            pickle.dump({"epoch": epoch, "step": step,
                         "global_step": self.global_step}, f)

    def _recover_ckpt_path(self):
        return os.path.join(
            constants.RECOVER_ROOT(self.cfg.experiment_name, self.cfg.trial_name),
            f"ckpt_rank{self.rank}.pt",
        )

    def save_recover_ckpt(self, epoch, step):
        """Full resumable state: every trainable model's flat params +
        its optimizer state (fp32 master/m/v shards), per rank.
        Reference: the recover ckpt saved by model workers
        (model_worker.py __save_model for recover + master's
        recover_info)."""
        if self.cfg.recover_mode == "disabled":
            return
        state = {"models": {}, "epoch": epoch, "step": step,
                 "global_step": self.global_step,
                 "interfaces": self._interface_states()}
        for name in self.built.trainable:
            if name not in self.models:
                continue
            eng = self.models[name].module
            m = eng.module
            opt = getattr(eng, "optimizer", None)
            if opt is not None and hasattr(opt, "finish_allgather"):
                opt.finish_allgather()  # deferred AG must land before save
            state["models"][str(name)] = {
                "flat_param": m.flat_param.detach().cpu().clone(),
                "lora_flat": (m.lora_flat.detach().cpu().clone()
                              if getattr(m, "lora_flat", None) is not None
                              else None),
                "optimizer": ({k: (v.cpu() if torch.is_tensor(v) else v)
                               for k, v in opt.state_dict().items()}
                              if opt is not None else None),
            }
        os.makedirs(os.path.dirname(self._recover_ckpt_path()), exist_ok=True)
        torch.save(state, self._recover_ckpt_path())
        self._save_recover_info(epoch, step)

    def _interface_states(self):
        """Algorithm state living on interfaces (value-norm running stats,
        adaptive KL controller value) — without it, elastic restart silently
        resets value normalization to mean 0/std 1 and kl_ctl to init."""
        out = {}
        for mfc_name, iface in self.built.interfaces.items():
            st = {}
            kl = getattr(iface, "_kl_ctl", None)
            if kl is not None:
                st["kl_ctl_value"] = kl.value
            rms = getattr(iface, "_rms", None)
            if rms is not None:
                st["value_norm"] = rms.state_dict()
            if st:
                out[mfc_name] = st
        return out

    def _restore_interface_states(self, states):
        for mfc_name, st in (states or {}).items():
            iface = self.built.interfaces.get(mfc_name)
            if iface is None:
                continue
            kl = getattr(iface, "_kl_ctl", None)
            if kl is not None and "kl_ctl_value" in st:
                kl.value = st["kl_ctl_value"]
            rms = getattr(iface, "_rms", None)
            if rms is not None and "value_norm" in st:
                rms.load_state_dict(st["value_norm"])

    def _maybe_load_recover(self):
        if self.cfg.recover_mode not in ("auto", "resume"):
            return None
        p = self._recover_path()
        if not os.path.exists(p):
            return None
        with open(p, "rb") as f:
            info = pickle.load(f)
        self.global_step = info["global_step"]
        ckpt = self._recover_ckpt_path()
        if os.path.exists(ckpt):
            state = torch.load(ckpt, map_location="cpu", weights_only=False)
            for name in self.built.trainable:
                st = state["models"].get(str(name))
                if st is None or name not in self.models:
                    continue
                eng = self.models[name].module
                m = eng.module
                with torch.no_grad():
                    m.flat_param.copy_(st["flat_param"].to(m.flat_param.device))
                    if st["lora_flat"] is not None:
                        m.lora_flat.copy_(st["lora_flat"].to(m.lora_flat.device))
                opt = getattr(eng, "optimizer", None)
                if opt is not None and st["optimizer"] is not None:
                    opt.load_state_dict(st["optimizer"])
            self._restore_interface_states(state.get("interfaces"))
            logger.info("restored model+optimizer state from %s", ckpt)
        logger.info("recovering from %s", info)
        return (info["epoch"], info["step"])
