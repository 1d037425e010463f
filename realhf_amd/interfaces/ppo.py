"""PPO actor & critic interfaces.

Reference semantics: realhf/impl/model/interface/ppo_interface.py
(PPOActorInterface:110 — generate:180, inference:254, train_step:298;
PPOCriticInterface:639; losses in _ppo_actor_loss_from_model_outputs:28).

Data-key contract (matches the reference DFG):
  actor_gen:    packed_prompts -> packed_input_ids, packed_logprobs,
                prompt_mask, seq_no_eos_mask
  rew_inf:      packed_input_ids -> rewards [bs]
  ref_inf:      packed_input_ids -> packed_ref_logprobs
  critic_inf:   packed_input_ids -> values [total]
  actor_train:  all of the above -> stats
  critic_train: all of the above -> stats
"""
import dataclasses
from typing import Dict, Optional

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import (
    GenerationHyperparameters,
    Model,
    ModelInterface,
    register_interface,
)
from realhf_amd.base import constants, logging
from realhf_amd.interfaces import ppo_math
from realhf_amd.models import moe as moe_mod
from realhf_amd.models.generation import concat_prompt_to_generation_output
from realhf_amd.ops import functional as ops
from realhf_amd.parallel.tp import packed_shifted_logprobs
from realhf_amd.runtime.engine import sample_to_packed
from realhf_amd.utils.functional import (
    apply_logits_mask,
    build_leave_one_indices,
    build_shift_one_indices,
    masked_normalization,
)

logger = logging.getLogger("ppo")


def _short_cu(cu: torch.Tensor) -> torch.Tensor:
    """cu_seqlens over (len-1)-length per-seq arrays."""
    bs = cu.shape[0] - 1
    return cu - torch.arange(bs + 1, device=cu.device, dtype=cu.dtype)


def _warp_logits_like_sampler(logits, cu, mb, temperature: float):
    """Make a re-forward's logits distribution match the sampler's:
    temperature scaling + the recorded top-k/p removal mask on the
    PREDICTING rows (row s..e-2 of each sequence).  In-place on `logits`
    (safe: GEMM backward does not re-read its output).  Reference:
    ppo_interface.py:38-54 applies packed_logits_mask the same way."""
    if temperature != 1.0:
        logits /= max(temperature, 1e-5)
    lm = mb.data.get("packed_logits_mask")
    if lm is None:
        return logits
    rows = build_leave_one_indices(int(cu[-1]), cu)
    sub = logits[rows]                      # advanced-index copy
    apply_logits_mask(sub, lm.bool())       # in-place on the copy
    logits[rows] = sub
    return logits


@torch.no_grad()
def _prepare_ppo_batch(
    sample: SequenceSample,
    kl_ctl_value: float,
    max_reward_clip: float,
    discount: float,
    gae_lambda: float,
    adv_norm: bool,
    value_norm: Optional[ppo_math.ExponentialRunningMeanStd],
):
    """Shared actor/critic prep: KL rewards -> GAE -> (adv norm)."""
    ids, cu, _ = sample_to_packed(sample)
    device = ids.device
    old_logp = sample.data["packed_logprobs"].float()
    ref_logp = sample.data["packed_ref_logprobs"].float()
    values = sample.data["values"].float()
    score = sample.data["rewards"].float()
    prompt_mask = sample.data["prompt_mask"].bool()
    no_eos = (
        sample.data["seq_no_eos_mask"].bool()
        if "seq_no_eos_mask" in sample.keys
        else torch.zeros(sample.bs, dtype=torch.bool, device=device)
    )
    scu = _short_cu(cu)

    kl_rewards, rewards = ppo_math.get_packed_rewards(
        kl_ctl_value, max_reward_clip, score, old_logp, ref_logp, scu, no_eos
    )
    # loss positions: predictions of generated tokens
    shift = build_shift_one_indices(ids.shape[0], cu)
    loss_mask = ~prompt_mask[shift]

    # GAE: values per token [total], bootstrap=0 at sequence end (no_eos
    # sequences are truncations — bootstrap with the last value)
    adv, ret = ops.gae(
        rewards.float(), values, scu.to(torch.int32), no_eos, discount, gae_lambda
    )
    if adv_norm:
        group = constants.dp_group() if constants.has_current() else None
        adv = masked_normalization(adv, loss_mask, all_reduce_group=group)
    return dict(
        cu=cu, scu=scu, old_logp=old_logp, kl_rewards=kl_rewards,
        rewards=rewards, advantages=adv, returns=ret, loss_mask=loss_mask,
        values=values, score=score,
    )


@dataclasses.dataclass
class PPOActorInterface(ModelInterface):
    n_minibatches: int = 4
    gconfig: GenerationHyperparameters = dataclasses.field(
        default_factory=GenerationHyperparameters
    )
    kl_ctl: float = 0.1
    adaptive_kl_ctl: bool = False
    adaptive_kl_target: float = 6.0
    adaptive_kl_horizon: float = 10000.0
    eps_clip: float = 0.2
    max_reward_clip: float = 20.0
    discount: float = 1.0
    gae_lambda: float = 1.0
    adv_norm: bool = True
    early_stop_imp_ratio: Optional[float] = None
    # leave the LAST minibatch's ZeRO param all-gather in flight; it is
    # waited at the model's next engine call (e.g. the critic's AG hides
    # under the following 2.7 s generation phase at dp8)
    defer_final_allgather: bool = False

    def __post_init__(self):
        if isinstance(self.gconfig, dict):
            self.gconfig = GenerationHyperparameters(**self.gconfig)
        if self.adaptive_kl_ctl:
            self._kl_ctl = ppo_math.AdaptiveKLController(
                self.kl_ctl, self.adaptive_kl_target, self.adaptive_kl_horizon
            )
        else:
            self._kl_ctl = ppo_math.FixedKLController(self.kl_ctl)

    # ------------------------------------------------------------ generate
    def generate(self, model: Model, data: SequenceSample, n_mbs=None):
        outs = model.module.generate(
            data, tokenizer=model.tokenizer, gconfig=self.gconfig,
            n_mbs=n_mbs, return_prompt_logprobs=True,
        )
        if outs is None:  # pp mid stage: output lives on the last stage
            return None
        want_mask = not self.gconfig.force_no_logits_mask
        all_ids, all_lp, all_pm, all_noeos, seqlens = [], [], [], [], []
        all_lm = []
        for gen_out, prompts, cu in outs:
            packed, cu_full, pmask = concat_prompt_to_generation_output(
                prompts, cu, gen_out
            )
            bs = cu.shape[0] - 1
            # stitch full-sequence logprobs: per seq [prompt-1 | gen]
            plp = gen_out.prompt_logprobs
            pscu = _short_cu(cu)
            lps = []
            for i in range(bs):
                ps, pe = int(pscu[i]), int(pscu[i + 1])
                gl = int(gen_out.gen_lengths[i])
                lps.append(plp[ps:pe].float())
                lps.append(gen_out.gen_logprobs[i, :gl].float())
                if want_mask and gen_out.logits_mask is not None:
                    # rows align with packed_logprobs (l-1 per seq):
                    # prompt rows allow everything (False = not removed),
                    # gen rows carry the sampler's top-k/p removals
                    # (reference stitches this in
                    # concat_prompt_to_generation_output,
                    # real_llm_generate.py:451)
                    lm = gen_out.logits_mask
                    plen = int(cu[i + 1] - cu[i])
                    all_lm.append(lm.new_zeros(plen - 1, lm.shape[-1]))
                    all_lm.append(lm[i, :gl])
            all_ids.append(packed)
            all_lp.append(torch.cat(lps))
            all_pm.append(pmask)
            all_noeos.append(gen_out.no_eos_mask)
            seqlens += [int(cu_full[i + 1] - cu_full[i]) for i in range(bs)]
        keys = ["packed_input_ids", "packed_logprobs", "prompt_mask",
                "seq_no_eos_mask"]
        seql = {
            "packed_input_ids": [[l] for l in seqlens],
            "packed_logprobs": [[l - 1] for l in seqlens],
            "prompt_mask": [[l] for l in seqlens],
            "seq_no_eos_mask": [[1]] * len(seqlens),
        }
        dat = {
            "packed_input_ids": torch.cat(all_ids),
            "packed_logprobs": torch.cat(all_lp),
            "prompt_mask": torch.cat(all_pm),
            "seq_no_eos_mask": torch.cat(all_noeos),
        }
        if all_lm:
            keys.append("packed_logits_mask")
            seql["packed_logits_mask"] = [[l - 1] for l in seqlens]
            dat["packed_logits_mask"] = torch.cat(all_lm)
        res = SequenceSample(
            keys=tuple(keys), ids=list(data.ids), seqlens=seql, data=dat,
        )
        return res

    # ----------------------------------------------------------- inference
    @torch.no_grad()
    def inference(self, model: Model, data: SequenceSample, n_mbs=None):
        """Used for the REF model: logprobs of the given sequences."""

        def post_hook(logits, mb):
            ids, cu, _ = sample_to_packed(mb)
            _warp_logits_like_sampler(logits, cu, mb, self.gconfig.temperature)
            return packed_shifted_logprobs(logits, cu, ids)

        logp = model.module.forward(data, n_mbs=n_mbs, post_hook=post_hook)
        if logp is None:  # pp mid stage
            return None
        return SequenceSample(
            keys=("packed_ref_logprobs",),
            ids=list(data.ids),
            seqlens={
                "packed_ref_logprobs": [
                    [sum(x) - 1] for x in data.seqlens["packed_input_ids"]
                ]
            },
            data={"packed_ref_logprobs": logp.float()},
        )

    # ---------------------------------------------------------- train_step
    def train_step(self, model: Model, data: SequenceSample, n_mbs=None) -> Dict:
        prep = _prepare_ppo_batch(
            data, self._kl_ctl.value, self.max_reward_clip,
            self.discount, self.gae_lambda, self.adv_norm, None,
        )
        # attach computed tensors for minibatch splitting
        sel = ["packed_input_ids", "prompt_mask"]
        if "packed_logits_mask" in data.keys:
            sel.append("packed_logits_mask")
        train_sample = data.select_keys(sel)
        extra = SequenceSample(
            keys=("old_logp", "advantages", "ppo_loss_mask"),
            ids=list(data.ids),
            seqlens={
                k: [[sum(x) - 1] for x in data.seqlens["packed_input_ids"]]
                for k in ("old_logp", "advantages", "ppo_loss_mask")
            },
            data={
                "old_logp": prep["old_logp"],
                "advantages": prep["advantages"],
                "ppo_loss_mask": prep["loss_mask"].float(),
            },
        )
        train_sample.update_(extra)

        all_stats: Dict[str, float] = {}
        n_mb = min(self.n_minibatches, train_sample.bs)
        mbs = train_sample.split(n_mb) if n_mb > 1 else [train_sample]
        for i_mb, mb in enumerate(mbs):
            stats = model.module.train_batch(
                mb, self._loss_fn, version_steps=model.version.global_step,
                n_mbs=n_mbs,
                defer_allgather=(self.defer_final_allgather
                                 and i_mb == len(mbs) - 1),
            )
            if (
                self.early_stop_imp_ratio is not None
                and stats.get("importance_weight", 1.0) > self.early_stop_imp_ratio
            ):
                logger.warning(
                    "early-stopping PPO minibatches: importance ratio %.2f",
                    stats["importance_weight"],
                )
                break
            for k, v in stats.items():
                all_stats[k] = all_stats.get(k, 0.0) + v / len(mbs)
        model.inc_version()

        with torch.no_grad():
            m = prep["loss_mask"].float()
            n = m.sum().clamp(min=1)
            kl = float((-prep["kl_rewards"] * m).sum() / n / max(self._kl_ctl.value, 1e-8))
        self._kl_ctl.update(kl, n_steps=int(data.bs))
        all_stats["kl"] = kl
        all_stats["kl_ctl"] = self._kl_ctl.value
        all_stats["reward"] = float(prep["score"].mean())
        all_stats["avg_seq_len"] = float(
            sum(sum(x) for x in data.seqlens["packed_input_ids"]) / data.bs
        )
        return all_stats

    def _loss_fn(self, logits, mb: SequenceSample):
        ids, cu, _ = sample_to_packed(mb)
        _warp_logits_like_sampler(logits, cu, mb, self.gconfig.temperature)
        new_logp = packed_shifted_logprobs(logits, cu, ids)
        loss, stats = ppo_math.actor_loss_fn(
            new_logp,
            mb.data["old_logp"],
            mb.data["advantages"],
            self.eps_clip,
            mb.data["ppo_loss_mask"],
        )
        for aux in moe_mod.pop_aux_losses():
            loss = loss + aux
        stats["actor_loss"] = float(loss.detach())
        return loss, stats

    def save(self, model: Model, save_dir: str):
        from realhf_amd.models import hf as hf_reg

        m = model.module.module if hasattr(model.module, "module") else model.module
        hf_reg.save_to_hf(m, m.config.family or "llama", save_dir, model.tokenizer)


@dataclasses.dataclass
class PPOCriticInterface(ModelInterface):
    n_minibatches: int = 4
    kl_ctl: float = 0.1
    adaptive_kl_ctl: bool = False
    adaptive_kl_target: float = 6.0
    adaptive_kl_horizon: float = 10000.0
    value_eps_clip: float = 0.2
    max_reward_clip: float = 20.0
    discount: float = 1.0
    gae_lambda: float = 1.0
    value_norm: bool = True
    value_norm_type: str = "exp"  # exp | ma (reference: rms.py:185)
    value_norm_beta: float = 0.99995
    value_norm_eps: float = 1e-5
    value_loss_type: str = "huber"
    defer_final_allgather: bool = False

    def __post_init__(self):
        self._rms = (ppo_math.make_value_norm(
            self.value_norm_type, self.value_norm_beta, self.value_norm_eps)
            if self.value_norm else None)
        # The critic must adapt its KL coefficient in lockstep with the actor
        # (reference ppo_interface.py:658) — otherwise value targets are
        # computed from a stale kl_ctl while the actor's rewards drift.
        if self.adaptive_kl_ctl:
            self._kl_ctl = ppo_math.AdaptiveKLController(
                self.kl_ctl, self.adaptive_kl_target, self.adaptive_kl_horizon
            )
        else:
            self._kl_ctl = ppo_math.FixedKLController(self.kl_ctl)

    @torch.no_grad()
    def inference(self, model: Model, data: SequenceSample, n_mbs=None):
        """Critic values for each token, denormalized."""

        def post_hook(values, mb):
            return values.squeeze(-1).float()

        vals = model.module.forward(data, n_mbs=n_mbs, post_hook=post_hook)
        if vals is None:  # pp mid stage
            return None
        if self._rms is not None:
            vals = self._rms.denormalize(vals)
        return SequenceSample(
            keys=("values",),
            ids=list(data.ids),
            seqlens={"values": list(data.seqlens["packed_input_ids"])},
            data={"values": vals},
        )

    def train_step(self, model: Model, data: SequenceSample, n_mbs=None) -> Dict:
        prep = _prepare_ppo_batch(
            data, self._kl_ctl.value, self.max_reward_clip,
            self.discount, self.gae_lambda, False, self._rms,
        )
        returns = prep["returns"]
        old_values = prep["values"]
        if self._rms is not None:
            group = constants.dp_group() if constants.has_current() else None
            self._rms.update(returns, prep["loss_mask"], group=group)
            returns_t = self._rms.normalize(returns)
            old_values_t = self._rms.normalize(old_values)
        else:
            returns_t, old_values_t = returns, old_values

        train_sample = data.select_keys(["packed_input_ids"])
        L = [[sum(x) - 1] for x in data.seqlens["packed_input_ids"]]
        Lv = [[sum(x)] for x in data.seqlens["packed_input_ids"]]
        extra = SequenceSample(
            keys=("returns", "old_values", "ppo_loss_mask"),
            ids=list(data.ids),
            seqlens={"returns": L, "old_values": Lv, "ppo_loss_mask": L},
            data={
                "returns": returns_t,
                "old_values": old_values_t,
                "ppo_loss_mask": prep["loss_mask"].float(),
            },
        )
        train_sample.update_(extra)

        all_stats: Dict[str, float] = {}
        n_mb = min(self.n_minibatches, train_sample.bs)
        mbs = train_sample.split(n_mb) if n_mb > 1 else [train_sample]
        for i_mb, mb in enumerate(mbs):
            stats = model.module.train_batch(
                mb, self._loss_fn, version_steps=model.version.global_step,
                n_mbs=n_mbs,
                defer_allgather=(self.defer_final_allgather
                                 and i_mb == len(mbs) - 1),
            )
            for k, v in stats.items():
                all_stats[k] = all_stats.get(k, 0.0) + v / len(mbs)
        model.inc_version()
        with torch.no_grad():
            m = prep["loss_mask"].float()
            n = m.sum().clamp(min=1)
            kl = float(
                (-prep["kl_rewards"] * m).sum() / n / max(self._kl_ctl.value, 1e-8)
            )
        self._kl_ctl.update(kl, n_steps=int(data.bs))
        all_stats["returns_mean"] = float(returns.mean())
        return all_stats

    def _loss_fn(self, values, mb: SequenceSample):
        ids, cu, _ = sample_to_packed(mb)
        from realhf_amd.utils.functional import build_leave_one_indices

        leave = build_leave_one_indices(ids.shape[0], cu)
        new_values = values.squeeze(-1).float()[leave]
        old_values = mb.data["old_values"]
        old_leave = old_values[leave]
        loss, stats = ppo_math.critic_loss_fn(
            new_values, old_leave, mb.data["returns"],
            self.value_eps_clip, mb.data["ppo_loss_mask"], self.value_loss_type,
        )
        stats["critic_loss"] = float(loss.detach())
        return loss, stats

    def save(self, model: Model, save_dir: str):
        from realhf_amd.models import hf as hf_reg

        m = model.module.module if hasattr(model.module, "module") else model.module
        hf_reg.save_to_hf(m, m.config.family or "llama", save_dir, model.tokenizer)


register_interface("ppo_actor", PPOActorInterface)
register_interface("ppo_critic", PPOCriticInterface)
