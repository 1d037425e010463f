"""GRPO interface (reference: examples/new_algorithms/grpo/grpo_exp.py —
a user-code extension in the reference; first-class here for the
Mixtral-EP config).

GRPO = PPO without a critic: each prompt is sampled `group_size` times;
the advantage of a response is its group-normalized reward, applied
uniformly over its generated tokens.
"""
import dataclasses
from typing import Dict

import torch

from realhf_amd.api.data import SequenceSample
from realhf_amd.api.model import Model, register_interface
from realhf_amd.interfaces import ppo_math
from realhf_amd.interfaces.ppo import PPOActorInterface, _short_cu
from realhf_amd.models import moe as moe_mod
from realhf_amd.parallel.tp import packed_shifted_logprobs
from realhf_amd.runtime.engine import sample_to_packed
from realhf_amd.utils.functional import build_shift_one_indices


@dataclasses.dataclass
class GRPOInterface(PPOActorInterface):
    group_size: int = 4
    kl_in_loss_coef: float = 0.0  # GRPO-style direct KL term in the loss

    def generate(self, model: Model, data: SequenceSample, n_mbs=None):
        # replicate each prompt group_size times (fresh ids per replica)
        singles = data.unpack()
        reps = []
        for s in singles:
            for g in range(self.group_size):
                reps.append(
                    SequenceSample(
                        keys=s.keys,
                        ids=[f"{s.ids[0]}@g{g}"],
                        seqlens=dict(s.seqlens),
                        data=dict(s.data),
                        dtypes=dict(s.dtypes),
                        trailing_shapes=dict(s.trailing_shapes),
                    )
                )
        grouped = SequenceSample.gather(reps)
        return super().generate(model, grouped, n_mbs=n_mbs)

    def train_step(self, model: Model, data: SequenceSample, n_mbs=None) -> Dict:
        ids, cu, _ = sample_to_packed(data)
        old_logp = data.data["packed_logprobs"].float()
        ref_logp = data.data["packed_ref_logprobs"].float()
        score = data.data["rewards"].float()  # [n_seqs]
        prompt_mask = data.data["prompt_mask"].bool()
        bs = data.bs

        # group-normalized advantages.  Groups are identified by the id
        # PREFIX ("origid@g{k}" set at generation) — NOT by position: the
        # balanced DP split can cut a group across ranks or reorder, so
        # positional view(bs//g, g) would silently mix groups.  Members
        # co-resident on this shard normalize together; a singleton group
        # (rest of it on another rank) gets advantage 0.
        prefixes = [str(i).split("@g")[0] for i in data.ids]
        uniq = {p: j for j, p in enumerate(dict.fromkeys(prefixes))}
        gidx = torch.tensor([uniq[p] for p in prefixes], device=score.device)
        n_groups = len(uniq)
        cnt = torch.zeros(n_groups, device=score.device).scatter_add_(
            0, gidx, torch.ones_like(score))
        gsum = torch.zeros(n_groups, device=score.device).scatter_add_(
            0, gidx, score)
        gmean = gsum / cnt.clamp(min=1)
        gsq = torch.zeros(n_groups, device=score.device).scatter_add_(
            0, gidx, (score - gmean[gidx]) ** 2)
        # sample std (ddof=1) to match torch.std's default
        gstd = (gsq / (cnt - 1).clamp(min=1)).sqrt()
        adv_seq = (score - gmean[gidx]) / (gstd[gidx] + 1e-5)
        adv_seq = torch.where(cnt[gidx] > 1, adv_seq,
                              torch.zeros_like(adv_seq))

        shift = build_shift_one_indices(ids.shape[0], cu)
        loss_mask = ~prompt_mask[shift]
        scu = _short_cu(cu)
        seg = torch.bucketize(
            torch.arange(old_logp.shape[0], device=ids.device), scu[1:].long(),
            right=True,
        )
        advantages = adv_seq[seg]

        sel = ["packed_input_ids", "prompt_mask"]
        if "packed_logits_mask" in data.keys:
            sel.append("packed_logits_mask")
        train_sample = data.select_keys(sel)
        L = [[sum(x) - 1] for x in data.seqlens["packed_input_ids"]]
        extra = SequenceSample(
            keys=("old_logp", "ref_logp", "advantages", "ppo_loss_mask"),
            ids=list(data.ids),
            seqlens={k: L for k in ("old_logp", "ref_logp", "advantages", "ppo_loss_mask")},
            data={
                "old_logp": old_logp,
                "ref_logp": ref_logp,
                "advantages": advantages,
                "ppo_loss_mask": loss_mask.float(),
            },
        )
        train_sample.update_(extra)

        all_stats: Dict[str, float] = {}
        n_mb = min(self.n_minibatches, train_sample.bs)
        mbs = train_sample.split(n_mb) if n_mb > 1 else [train_sample]
        for mb in mbs:
            stats = model.module.train_batch(
                mb, self._grpo_loss_fn, version_steps=model.version.global_step,
                n_mbs=n_mbs,
            )
            for k, v in stats.items():
                all_stats[k] = all_stats.get(k, 0.0) + v / len(mbs)
        model.inc_version()
        all_stats["reward"] = float(score.mean())
        return all_stats

    def _grpo_loss_fn(self, logits, mb: SequenceSample):
        from realhf_amd.interfaces.ppo import _warp_logits_like_sampler

        ids, cu, _ = sample_to_packed(mb)
        _warp_logits_like_sampler(logits, cu, mb, self.gconfig.temperature)
        new_logp = packed_shifted_logprobs(logits, cu, ids)
        loss, stats = ppo_math.actor_loss_fn(
            new_logp, mb.data["old_logp"], mb.data["advantages"],
            self.eps_clip, mb.data["ppo_loss_mask"],
        )
        if self.kl_in_loss_coef > 0:
            # k3 estimator KL(pi || ref) over generated tokens
            mask = mb.data["ppo_loss_mask"]
            lr = mb.data["ref_logp"] - new_logp
            k3 = (lr.exp() - 1 - lr) * mask
            loss = loss + self.kl_in_loss_coef * k3.sum() / mask.sum().clamp(min=1)
            stats["kl_in_loss"] = float((k3.sum() / mask.sum().clamp(min=1)).detach())
        for aux in moe_mod.pop_aux_losses():
            loss = loss + aux
        stats["actor_loss"] = float(loss.detach())
        return loss, stats


register_interface("grpo", GRPOInterface)
