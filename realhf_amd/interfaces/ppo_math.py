"""PPO math: losses, KL controllers, reward shaping, value normalization.

Reference semantics: realhf/impl/model/utils/ppo_functional.py
(actor_loss_fn:49, critic_loss_fn:135, Adaptive/FixedKLController:21/39,
get_packed_rewards:291) and modules value norm (ExponentialRunningMeanStd).
GAE itself is the HIP kernel / reference scan in ops.functional.gae.
"""
from typing import Optional, Tuple

import torch
import torch.distributed as dist


def actor_loss_fn(
    logprobs: torch.Tensor,
    old_logprobs: torch.Tensor,
    advantages: torch.Tensor,
    eps_clip: float,
    loss_mask: torch.Tensor,
) -> Tuple[torch.Tensor, dict]:
    mask = loss_mask.float()
    n = mask.sum().clamp(min=1)
    ratio = torch.exp((logprobs - old_logprobs) * mask)
    pg1 = -advantages * ratio
    pg2 = -advantages * ratio.clamp(1.0 - eps_clip, 1.0 + eps_clip)
    loss = (torch.max(pg1, pg2) * mask).sum() / n
    with torch.no_grad():
        clip_frac = ((pg2 > pg1).float() * mask).sum() / n
        imp = (ratio.detach() * mask).sum() / n
        approx_kl = (((old_logprobs - logprobs) * mask).sum() / n).detach()
    return loss, {
        "clip_ratio": float(clip_frac),
        "importance_weight": float(imp),
        "approx_kl": float(approx_kl),
    }


def critic_loss_fn(
    value: torch.Tensor,
    old_value: torch.Tensor,
    target_value: torch.Tensor,
    value_eps_clip: float,
    loss_mask: torch.Tensor,
    loss_fn_type: str = "huber",
) -> Tuple[torch.Tensor, dict]:
    mask = loss_mask.float()
    n = mask.sum().clamp(min=1)
    if loss_fn_type == "huber":
        f = lambda v: torch.nn.functional.huber_loss(
            v, target_value.detach(), reduction="none", delta=10.0
        )
    else:
        f = lambda v: (v - target_value.detach()).pow(2) * 0.5
    clipped = old_value + (value - old_value).clamp(-value_eps_clip, value_eps_clip)
    l1, l2 = f(value), f(clipped)
    loss = (torch.max(l1, l2) * mask).sum() / n
    with torch.no_grad():
        clip_frac = ((l2 > l1).float() * mask).sum() / n
    return loss, {"value_clip_ratio": float(clip_frac)}


class FixedKLController:
    def __init__(self, kl_coef: float):
        self.value = kl_coef

    def update(self, current_kl, n_steps):
        pass


class AdaptiveKLController:
    """https://arxiv.org/abs/1909.08593 (reference: ppo_functional.py:21)."""

    def __init__(self, init_kl_coef: float, target_kl: float, horizon: float):
        self.value = init_kl_coef
        self.target = target_kl
        self.horizon = horizon

    def update(self, current_kl: float, n_steps: int):
        err = max(-0.2, min(0.2, current_kl / self.target - 1))
        self.value *= 1 + 0.1 * err * n_steps / self.horizon


@torch.no_grad()
def get_packed_rewards(
    kl_ctl: float,
    clip_reward_value: float,
    score: torch.Tensor,  # [bs] — reward-model score per sequence
    logprobs: torch.Tensor,  # [total - bs]
    ref_logprobs: torch.Tensor,  # [total - bs]
    short_cu_seqlens: torch.Tensor,  # [bs+1] over the (len-1) reward positions
    seq_no_eos_mask: torch.Tensor,  # [bs] — True if the sequence was truncated (no EOS)
) -> Tuple[torch.Tensor, torch.Tensor]:
    """KL-penalty token rewards + clipped score at the last position
    (reference: ppo_functional.py:291).

    Truncated (no-EOS) sequences get NO score at the end: they already
    bootstrap from the last value in GAE, so adding the score would
    double-count reward and bias the policy against emitting EOS
    (reference: ppo_functional.py:306 torch.where(seq_no_eos_mask, 0, score)).
    """
    kl_rewards = -kl_ctl * (logprobs - ref_logprobs)
    rewards = kl_rewards.clone()
    ends = short_cu_seqlens[1:].long() - 1
    sc = score.clamp(-clip_reward_value, clip_reward_value)
    rewards[ends] += torch.where(seq_no_eos_mask.bool(), torch.zeros_like(sc), sc)
    return kl_rewards, rewards


class ExponentialRunningMeanStd:
    """EMA running mean/std for value normalization (reference:
    modules value-norm; beta=0.995)."""

    def __init__(self, beta: float = 0.99995, eps: float = 1e-5, high_precision=True):
        self.beta = beta
        self.eps = eps
        self.mean = 0.0
        self.mean_sq = 0.0
        self.debias = 0.0

    @torch.no_grad()
    def update(self, x: torch.Tensor, mask: Optional[torch.Tensor] = None, group=None):
        xf = x.float()
        if mask is not None:
            m = mask.float()
            cnt = m.sum()
            s = (xf * m).sum()
            sq = (xf * xf * m).sum()
        else:
            cnt = torch.tensor(float(xf.numel()), device=x.device)
            s = xf.sum()
            sq = (xf * xf).sum()
        if group is not None and dist.is_initialized():
            stats = torch.stack([cnt, s, sq])
            dist.all_reduce(stats, group=group)
            cnt, s, sq = stats[0], stats[1], stats[2]
        if float(cnt) < 1:
            return
        bmean = float(s / cnt)
        bmean_sq = float(sq / cnt)
        self.mean = self.beta * self.mean + (1 - self.beta) * bmean
        self.mean_sq = self.beta * self.mean_sq + (1 - self.beta) * bmean_sq
        self.debias = self.beta * self.debias + (1 - self.beta)

    def mean_std(self):
        if self.debias == 0:
            return 0.0, 1.0
        mean = self.mean / self.debias
        var = max(self.mean_sq / self.debias - mean * mean, 0.0)
        return mean, max(var**0.5, self.eps)

    def normalize(self, x: torch.Tensor) -> torch.Tensor:
        mean, std = self.mean_std()
        return (x - mean) / std

    def denormalize(self, x: torch.Tensor) -> torch.Tensor:
        mean, std = self.mean_std()
        return x * std + mean

    def state_dict(self):
        return dict(mean=self.mean, mean_sq=self.mean_sq, debias=self.debias)

    def load_state_dict(self, sd):
        self.mean, self.mean_sq, self.debias = sd["mean"], sd["mean_sq"], sd["debias"]


class MovingAverageRunningMeanStd:
    """Cumulative-average value normalization (reference: modules/rms.py:102
    MovingAverageRunningMeanStd — all history weighted equally, normalize
    clips to +-5).  Selected by value_norm_type="ma"."""

    def __init__(self, eps: float = 1e-5, high_precision=True):
        self.eps = eps
        self.mean = 0.0
        self.mean_sq = 0.0
        self.denom = 0.0

    @torch.no_grad()
    def update(self, x: torch.Tensor, mask=None, group=None):
        xf = x.float()
        if mask is not None:
            m = mask.float()
            cnt = m.sum()
            s = (xf * m).sum()
            sq = (xf * xf * m).sum()
        else:
            cnt = torch.tensor(float(xf.numel()), device=x.device)
            s = xf.sum()
            sq = (xf * xf).sum()
        if group is not None and dist.is_initialized():
            stats = torch.stack([cnt, s, sq])
            dist.all_reduce(stats, group=group)
            cnt, s, sq = stats[0], stats[1], stats[2]
        c = float(cnt)
        if c < 1:
            return
        self.mean = (self.denom * self.mean + float(s)) / (self.denom + c)
        self.mean_sq = (self.denom * self.mean_sq + float(sq)) / (self.denom + c)
        self.denom += c

    def mean_std(self):
        if self.denom == 0:
            return 0.0, 1.0
        var = max(self.mean_sq - self.mean * self.mean, 1e-4)
        return self.mean, var**0.5

    def normalize(self, x: torch.Tensor) -> torch.Tensor:
        mean, std = self.mean_std()
        return ((x - mean) / std).clip(-5, 5)

    def denormalize(self, x: torch.Tensor) -> torch.Tensor:
        mean, std = self.mean_std()
        return x * std + mean

    def state_dict(self):
        return dict(mean=self.mean, mean_sq=self.mean_sq, denom=self.denom)

    def load_state_dict(self, sd):
        self.mean, self.mean_sq, self.denom = sd["mean"], sd["mean_sq"], sd["denom"]


def make_value_norm(value_norm_type: str = "exp", beta: float = 0.99995,
                    eps: float = 1e-5):
    """reference: modules/rms.py:185 (exp | ma)."""
    if value_norm_type == "exp":
        return ExponentialRunningMeanStd(beta=beta, eps=eps)
    if value_norm_type == "ma":
        return MovingAverageRunningMeanStd(eps=eps)
    raise ValueError(f"value_norm_type {value_norm_type!r}")
