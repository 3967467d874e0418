"""PPO / GRPO core algorithms.

Capability parity with the verl core_algos surface the reference consumes
(compute_advantage GAE + GRPO group-norm, apply_kl_penalty,
agg_loss token-mean variants, policy losses {vanilla, gpg, clip_cov},
kl_penalty {kl, abs, mse, low_var_kl, full}, compute_value_loss).
All ops are plain tensor code — the trainer-side hot path is the model
fwd/bwd, not these.
"""
from __future__ import annotations

from collections import defaultdict
from typing import Optional, Tuple

import torch

# --------------------------------------------------------------------- masked


def masked_sum(values: torch.Tensor, mask: torch.Tensor, axis=None) -> torch.Tensor:
    return (values * mask).sum(axis=axis)


def masked_mean(values: torch.Tensor, mask: torch.Tensor, axis=None,
                eps: float = 1e-8) -> torch.Tensor:
    return (values * mask).sum(axis=axis) / (mask.sum(axis=axis) + eps)


def masked_var(values: torch.Tensor, mask: torch.Tensor,
               unbiased: bool = True) -> torch.Tensor:
    mean = masked_mean(values, mask)
    centered = (values - mean) * mask
    var = centered.pow(2).sum() / mask.sum().clamp(min=1)
    if unbiased:
        n = mask.sum()
        var = var * n / (n - 1).clamp(min=1)
    return var


def masked_whiten(values: torch.Tensor, mask: torch.Tensor,
                  shift_mean: bool = True) -> torch.Tensor:
    mean = masked_mean(values, mask)
    var = masked_var(values, mask)
    whitened = (values - mean) * torch.rsqrt(var + 1e-8)
    if not shift_mean:
        whitened = whitened + mean
    return whitened * mask


# ------------------------------------------------------------------ logprobs


def logprobs_from_logits(logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
    """log softmax(logits) gathered at labels.  (bsz, L, V) -> (bsz, L).

    Reference capability: verl logprobs_from_logits.  On
    MI355X the fused HIP kernel in ops/ is used inside the model's
    compute-logprob path; this is the composable fp32-reference form.
    """
    logp = torch.log_softmax(logits.float(), dim=-1)
    return torch.gather(logp, -1, labels.unsqueeze(-1)).squeeze(-1)


def entropy_from_logits(logits: torch.Tensor) -> torch.Tensor:
    """Categorical entropy per position. (..., V) -> (...)."""
    logp = torch.log_softmax(logits.float(), dim=-1)
    return -(logp.exp() * logp).sum(-1)


# ---------------------------------------------------------------- advantages


def compute_gae_advantage_return(
    token_level_rewards: torch.Tensor,   # (bsz, L)
    values: torch.Tensor,                # (bsz, L)
    response_mask: torch.Tensor,         # (bsz, L)
    gamma: float = 1.0,
    lam: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """GAE over the response region; advantages whitened over the mask.

    Matches the semantics of verl's compute_advantage(adv_estimator='gae')
    consumed at stream_ray_trainer.py:490-498.
    """
    with torch.no_grad():
        nextvalues = 0.0
        lastgaelam = 0.0
        advantages_reversed = []
        gen_len = token_level_rewards.shape[-1]
        for t in reversed(range(gen_len)):
            delta = token_level_rewards[:, t] + gamma * nextvalues - values[:, t]
            lastgaelam_ = delta + gamma * lam * lastgaelam
            # skip values and TD-error at padded positions
            m = response_mask[:, t]
            nextvalues = values[:, t] * m + (1 - m) * nextvalues
            lastgaelam = lastgaelam_ * m + (1 - m) * lastgaelam
            advantages_reversed.append(lastgaelam)
        advantages = torch.stack(advantages_reversed[::-1], dim=1)
        returns = advantages + values
        advantages = masked_whiten(advantages, response_mask)
    return advantages, returns


def compute_grpo_outcome_advantage(
    token_level_rewards: torch.Tensor,   # (bsz, L)
    response_mask: torch.Tensor,         # (bsz, L)
    index: "object",                     # np/object array of group uids, len bsz
    epsilon: float = 1e-6,
    norm_adv_by_std_in_grpo: bool = True,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """GRPO outcome advantage: per-uid group mean/std normalization of the
    scalar outcome reward, broadcast over response tokens."""
    scores = token_level_rewards.sum(dim=-1)  # (bsz,)
    id2score = defaultdict(list)
    id2mean, id2std = {}, {}
    bsz = scores.shape[0]
    with torch.no_grad():
        for i in range(bsz):
            id2score[index[i]].append(scores[i])
        for idx in id2score:
            group = torch.stack(id2score[idx])
            if len(group) == 1:
                id2mean[idx] = torch.tensor(0.0, device=scores.device)
                id2std[idx] = torch.tensor(1.0, device=scores.device)
            else:
                id2mean[idx] = group.mean()
                id2std[idx] = group.std()
        adv = scores.clone()
        for i in range(bsz):
            adv[i] = scores[i] - id2mean[index[i]]
            if norm_adv_by_std_in_grpo:
                adv[i] = adv[i] / (id2std[index[i]] + epsilon)
        adv = adv.unsqueeze(-1) * response_mask
    return adv, adv


def compute_rloo_outcome_advantage(
    token_level_rewards: torch.Tensor,
    response_mask: torch.Tensor,
    index,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """RLOO: leave-one-out baseline within each prompt group
    (verl core_algos rloo capability)."""
    scores = token_level_rewards.sum(dim=-1)
    id2score = defaultdict(list)
    bsz = scores.shape[0]
    with torch.no_grad():
        for i in range(bsz):
            id2score[index[i]].append((i, scores[i]))
        adv = scores.clone()
        for idx, pairs in id2score.items():
            k = len(pairs)
            if k == 1:
                adv[pairs[0][0]] = 0.0
                continue
            total = torch.stack([s for _, s in pairs]).sum()
            for i, s in pairs:
                adv[i] = s - (total - s) / (k - 1)
        adv = adv.unsqueeze(-1) * response_mask
    return adv, adv


def compute_reinforce_plus_plus_advantage(
    token_level_rewards: torch.Tensor,
    response_mask: torch.Tensor,
    gamma: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """REINFORCE++: discounted reward-to-go, globally whitened
    (verl core_algos reinforce_plus_plus capability)."""
    with torch.no_grad():
        returns = torch.zeros_like(token_level_rewards)
        run = torch.zeros(token_level_rewards.shape[0],
                          device=token_level_rewards.device)
        for t in reversed(range(token_level_rewards.shape[1])):
            run = token_level_rewards[:, t] + gamma * run
            returns[:, t] = run
        returns = returns * response_mask
        adv = masked_whiten(returns, response_mask) * response_mask
    return adv, returns


def compute_remax_outcome_advantage(
    token_level_rewards: torch.Tensor,
    response_mask: torch.Tensor,
    reward_baselines: torch.Tensor,      # (bsz,) greedy-rollout baseline
) -> Tuple[torch.Tensor, torch.Tensor]:
    """ReMax: outcome reward minus the greedy-decode baseline reward
    (verl core_algos remax capability; the trainer must supply baselines
    from a greedy rollout of the same prompts)."""
    with torch.no_grad():
        scores = token_level_rewards.sum(dim=-1)
        adv = (scores - reward_baselines).unsqueeze(-1) * response_mask
    return adv, adv


# --------------------------------------------------------------- KL penalties


def kl_penalty(logprob: torch.Tensor, ref_logprob: torch.Tensor,
               penalty: str = "kl") -> torch.Tensor:
    """Per-token KL penalty between policy and reference logprobs.

    Variants match the reference's kl_penalty set (stream_dp_actor.py:204-214,
    stream_ray_trainer.py:476-481): kl, abs, mse, low_var_kl, full(unsupported).
    """
    if penalty == "kl":
        return logprob - ref_logprob
    if penalty == "abs":
        return (logprob - ref_logprob).abs()
    if penalty == "mse":
        return 0.5 * (logprob - ref_logprob).square()
    if penalty in ("low_var_kl", "k3"):
        # k3 estimator: exp(r) - r - 1 with r = ref - logprob; non-negative
        kl = ref_logprob - logprob
        kl = torch.clamp(kl, min=-20, max=20)
        ratio = torch.exp(kl)
        return torch.clamp(ratio - kl - 1, min=-10, max=10)
    raise NotImplementedError(f"kl penalty {penalty!r}")


def apply_kl_penalty(token_level_scores: torch.Tensor,
                     old_log_probs: torch.Tensor,
                     ref_log_probs: torch.Tensor,
                     response_mask: torch.Tensor,
                     kl_coef: float,
                     penalty: str = "kl") -> Tuple[torch.Tensor, torch.Tensor]:
    """rewards = scores - kl_coef * KL  (driver-side, stream_ray_trainer.py:465-481)."""
    kld = kl_penalty(old_log_probs, ref_log_probs, penalty) * response_mask
    rewards = token_level_scores - kl_coef * kld
    current_kl = masked_mean(kld, response_mask, axis=-1).mean()
    return rewards, current_kl


# ------------------------------------------------------------- loss agg modes


def agg_loss(loss_mat: torch.Tensor, loss_mask: torch.Tensor,
             loss_agg_mode: str = "token-mean") -> torch.Tensor:
    """Aggregate a (bsz, L) loss matrix under a mask.

    Modes per verl's agg_loss: token-mean, seq-mean-token-sum,
    seq-mean-token-mean, seq-mean-token-sum-norm.
    """
    if loss_agg_mode == "token-mean":
        return masked_mean(loss_mat, loss_mask)
    if loss_agg_mode == "seq-mean-token-sum":
        seq_losses = (loss_mat * loss_mask).sum(dim=-1)
        return seq_losses.mean()
    if loss_agg_mode == "seq-mean-token-mean":
        seq = (loss_mat * loss_mask).sum(dim=-1) / loss_mask.sum(dim=-1).clamp(min=1)
        return seq.mean()
    if loss_agg_mode == "seq-mean-token-sum-norm":
        seq_losses = (loss_mat * loss_mask).sum(dim=-1)
        return seq_losses.sum() / loss_mask.shape[-1]
    raise ValueError(f"unknown loss_agg_mode {loss_agg_mode!r}")


# -------------------------------------------------------------- policy losses


def compute_policy_loss_vanilla(
    old_log_prob: torch.Tensor,
    log_prob: torch.Tensor,
    advantages: torch.Tensor,
    response_mask: torch.Tensor,
    clip_ratio: float = 0.2,
    clip_ratio_low: Optional[float] = None,
    clip_ratio_high: Optional[float] = None,
    clip_ratio_c: float = 3.0,
    loss_agg_mode: str = "token-mean",
    importance_weights: Optional[torch.Tensor] = None,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """PPO clipped surrogate with dual-clip (verl 'vanilla' loss).

    ``importance_weights``: per-token truncated importance-sampling weights
    correcting for rollouts generated on slightly stale weights (the
    reference stream actor's importance-weight capability,
    stream_dp_actor.py:153-224): detached, multiplied into the per-token
    loss before aggregation.

    Returns (pg_loss, pg_clipfrac, ppo_kl, pg_clipfrac_lower).
    """
    cliprange_low = clip_ratio_low if clip_ratio_low is not None else clip_ratio
    cliprange_high = clip_ratio_high if clip_ratio_high is not None else clip_ratio

    negative_approx_kl = log_prob - old_log_prob
    negative_approx_kl = torch.clamp(negative_approx_kl, min=-20, max=20)
    ratio = torch.exp(negative_approx_kl)
    ppo_kl = masked_mean(-negative_approx_kl, response_mask)

    pg_losses1 = -advantages * ratio
    pg_losses2 = -advantages * torch.clamp(ratio, 1 - cliprange_low, 1 + cliprange_high)
    clip_pg_losses1 = torch.maximum(pg_losses1, pg_losses2)
    pg_clipfrac = masked_mean(torch.gt(pg_losses2, pg_losses1).float(), response_mask)

    # dual-clip for strongly negative advantages
    pg_losses3 = -advantages * clip_ratio_c
    clip_pg_losses2 = torch.min(pg_losses3, clip_pg_losses1)
    pg_clipfrac_lower = masked_mean(
        torch.gt(clip_pg_losses1, pg_losses3).float() * (advantages < 0).float(),
        response_mask)
    pg_losses = torch.where(advantages < 0, clip_pg_losses2, clip_pg_losses1)
    if importance_weights is not None:
        pg_losses = pg_losses * importance_weights.detach()
    pg_loss = agg_loss(pg_losses, response_mask, loss_agg_mode)
    return pg_loss, pg_clipfrac, ppo_kl, pg_clipfrac_lower


def compute_policy_loss_gpg(
    old_log_prob: torch.Tensor,
    log_prob: torch.Tensor,
    advantages: torch.Tensor,
    response_mask: torch.Tensor,
    loss_agg_mode: str = "token-mean",
    **_,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """GPG: plain policy-gradient surrogate -adv * logp (no ratio/clip)."""
    pg_losses = -advantages * log_prob
    pg_loss = agg_loss(pg_losses, response_mask, loss_agg_mode)
    z = torch.zeros_like(pg_loss)
    return pg_loss, z, z, z


def compute_policy_loss_clip_cov(
    old_log_prob: torch.Tensor,
    log_prob: torch.Tensor,
    advantages: torch.Tensor,
    response_mask: torch.Tensor,
    clip_ratio: float = 0.2,
    clip_ratio_low: Optional[float] = None,
    clip_ratio_high: Optional[float] = None,
    clip_cov_ratio: float = 0.0002,
    clip_cov_lb: float = 1.0,
    clip_cov_ub: float = 5.0,
    loss_agg_mode: str = "token-mean",
    **_,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Clip-Cov: clip the tokens whose (logp, adv) covariance is extreme."""
    cliprange_low = clip_ratio_low if clip_ratio_low is not None else clip_ratio
    cliprange_high = clip_ratio_high if clip_ratio_high is not None else clip_ratio

    negative_approx_kl = torch.clamp(log_prob - old_log_prob, min=-20, max=20)
    ratio = torch.exp(negative_approx_kl)
    ppo_kl = masked_mean(-negative_approx_kl, response_mask)

    pg_losses1 = -advantages * ratio
    pg_losses2 = -advantages * torch.clamp(ratio, 1 - cliprange_low, 1 + cliprange_high)

    corr = torch.ones_like(advantages)
    with torch.no_grad():
        cov_all = (advantages - masked_mean(advantages, response_mask)) * \
                  (log_prob - masked_mean(log_prob.detach(), response_mask))
        cov_all[response_mask == 0] = -torch.inf
        cov_all[(cov_all < clip_cov_lb) | (cov_all > clip_cov_ub)] = -torch.inf
        n_clip = max(int(clip_cov_ratio * response_mask.sum().item()), 1)
        flat = cov_all.flatten()
        k = min(n_clip, flat.numel())
        top_idx = torch.topk(flat, k=k, largest=True).indices
        valid = flat[top_idx] > -torch.inf
        corr_flat = corr.flatten()
        corr_flat[top_idx[valid]] = 0.0
        corr = corr_flat.view_as(corr)

    pg_clipfrac = masked_mean((corr == 0).float(), response_mask)
    pg_losses = torch.maximum(pg_losses1, pg_losses2) * corr + pg_losses1 * (1 - corr)
    pg_loss = agg_loss(pg_losses, response_mask, loss_agg_mode)
    return pg_loss, pg_clipfrac, ppo_kl, torch.zeros_like(pg_loss)


_POLICY_LOSS_FNS = {
    "vanilla": compute_policy_loss_vanilla,
    "gpg": compute_policy_loss_gpg,
    "clip_cov": compute_policy_loss_clip_cov,
}


def get_policy_loss_fn(name: str = "vanilla"):
    if name not in _POLICY_LOSS_FNS:
        raise NotImplementedError(
            f"policy loss {name!r}; available {list(_POLICY_LOSS_FNS)}")
    return _POLICY_LOSS_FNS[name]


# ---------------------------------------------------------------- value loss


def compute_value_loss(
    vpreds: torch.Tensor,
    returns: torch.Tensor,
    values: torch.Tensor,
    response_mask: torch.Tensor,
    cliprange_value: float = 0.5,
    loss_agg_mode: str = "token-mean",
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Clipped value loss (reference: stream_dp_critic.py:105-113 via verl)."""
    vpredclipped = values + (vpreds - values).clamp(-cliprange_value, cliprange_value)
    vf_losses1 = (vpreds - returns).square()
    vf_losses2 = (vpredclipped - returns).square()
    clipped = torch.gt(vf_losses2, vf_losses1).float()
    vf_loss = 0.5 * agg_loss(torch.maximum(vf_losses1, vf_losses2),
                             response_mask, loss_agg_mode)
    vf_clipfrac = masked_mean(clipped, response_mask)
    return vf_loss, vf_clipfrac


# --------------------------------------------------------- advantage dispatch


def compute_advantage(batch_rewards: torch.Tensor,
                      response_mask: torch.Tensor,
                      adv_estimator: str,
                      values: Optional[torch.Tensor] = None,
                      index=None,
                      gamma: float = 1.0,
                      lam: float = 1.0,
                      norm_adv_by_std_in_grpo: bool = True,
                      reward_baselines: Optional[torch.Tensor] = None,
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    if adv_estimator == "gae":
        assert values is not None
        return compute_gae_advantage_return(batch_rewards, values, response_mask,
                                            gamma=gamma, lam=lam)
    if adv_estimator == "grpo":
        assert index is not None
        return compute_grpo_outcome_advantage(
            batch_rewards, response_mask, index,
            norm_adv_by_std_in_grpo=norm_adv_by_std_in_grpo)
    if adv_estimator == "rloo":
        assert index is not None
        return compute_rloo_outcome_advantage(batch_rewards, response_mask,
                                              index)
    if adv_estimator == "reinforce_plus_plus":
        return compute_reinforce_plus_plus_advantage(batch_rewards,
                                                     response_mask,
                                                     gamma=gamma)
    if adv_estimator == "remax":
        assert reward_baselines is not None, \
            "remax needs greedy-rollout baselines (trainer supplies them)"
        return compute_remax_outcome_advantage(batch_rewards, response_mask,
                                               reward_baselines)
    raise NotImplementedError(f"adv_estimator {adv_estimator!r}")
