import numpy as np
import pytest
import torch

from polyrl_amd.core import algos


def test_logprobs_from_logits_matches_manual():
    torch.manual_seed(0)
    logits = torch.randn(2, 5, 11)
    labels = torch.randint(0, 11, (2, 5))
    lp = algos.logprobs_from_logits(logits, labels)
    ref = torch.distributions.Categorical(logits=logits).log_prob(labels)
    assert torch.allclose(lp, ref, atol=1e-5)


def test_entropy_nonnegative_and_uniform_max():
    logits = torch.zeros(1, 3, 7)
    ent = algos.entropy_from_logits(logits)
    assert torch.allclose(ent, torch.full((1, 3), np.log(7), dtype=torch.float32),
                          atol=1e-5)


def test_gae_gamma_lam_one_is_reward_to_go_minus_value():
    torch.manual_seed(0)
    B, L = 4, 6
    rewards = torch.randn(B, L)
    values = torch.randn(B, L)
    mask = torch.ones(B, L)
    adv, ret = algos.compute_gae_advantage_return(rewards, values, mask,
                                                  gamma=1.0, lam=1.0)
    # returns = reward-to-go (for gamma=lam=1, full mask)
    rtg = torch.flip(torch.cumsum(torch.flip(rewards, [1]), 1), [1])
    assert torch.allclose(ret, rtg, atol=1e-4)
    # advantages whitened: ~0 mean, ~1 std
    assert abs(adv.mean().item()) < 1e-5
    assert abs(adv.std().item() - 1.0) < 0.05


def test_gae_respects_mask_padding():
    B, L = 2, 5
    rewards = torch.zeros(B, L)
    rewards[:, 2] = 1.0  # reward at last valid token
    values = torch.zeros(B, L)
    mask = torch.zeros(B, L)
    mask[:, :3] = 1.0
    adv, ret = algos.compute_gae_advantage_return(rewards, values, mask)
    # all valid positions get return 1 (reward-to-go)
    assert torch.allclose(ret[:, :3], torch.ones(B, 3), atol=1e-5)


def test_grpo_group_normalization():
    B, L = 6, 4
    mask = torch.ones(B, L)
    rewards = torch.zeros(B, L)
    rewards[:, -1] = torch.tensor([1., 0., 1., 1., 0., 0.])
    index = np.array(["a", "a", "a", "b", "b", "b"], dtype=object)
    adv, _ = algos.compute_grpo_outcome_advantage(rewards, mask, index)
    # within each group, mean ~0
    a_adv = adv[:3, 0]
    b_adv = adv[3:, 0]
    assert abs(a_adv.mean().item()) < 1e-5
    assert abs(b_adv.mean().item()) < 1e-5
    # token broadcast
    assert torch.allclose(adv[:, 0], adv[:, -1])


def test_grpo_single_sample_group_passthrough():
    # group of one: mean=0, std=1 (verl convention) -> adv == raw score
    rewards = torch.zeros(1, 3)
    rewards[0, -1] = 5.0
    mask = torch.ones(1, 3)
    adv, _ = algos.compute_grpo_outcome_advantage(rewards, mask,
                                                  np.array(["x"], dtype=object))
    assert torch.allclose(adv, torch.full((1, 3), 5.0))


@pytest.mark.parametrize("penalty", ["kl", "abs", "mse", "low_var_kl"])
def test_kl_penalty_zero_when_equal(penalty):
    lp = torch.randn(3, 4)
    out = algos.kl_penalty(lp, lp.clone(), penalty)
    assert torch.allclose(out, torch.zeros_like(out), atol=1e-6)


def test_low_var_kl_nonnegative():
    lp = torch.randn(10, 8)
    ref = torch.randn(10, 8)
    out = algos.kl_penalty(lp, ref, "low_var_kl")
    assert (out >= -1e-6).all()


def test_policy_loss_vanilla_gradient_direction():
    # positive advantage should push log_prob up (negative loss gradient)
    old_lp = torch.zeros(2, 3)
    lp = torch.zeros(2, 3, requires_grad=True)
    adv = torch.ones(2, 3)
    mask = torch.ones(2, 3)
    loss, clipfrac, ppo_kl, _ = algos.compute_policy_loss_vanilla(
        old_lp, lp, adv, mask)
    loss.backward()
    assert (lp.grad < 0).all()  # increasing lp decreases loss
    assert clipfrac.item() == 0.0


def test_policy_loss_clipping_activates():
    old_lp = torch.zeros(1, 4)
    lp = torch.full((1, 4), 2.0)  # ratio = e^2 >> 1+eps
    adv = torch.ones(1, 4)
    mask = torch.ones(1, 4)
    loss, clipfrac, _, _ = algos.compute_policy_loss_vanilla(
        old_lp, lp, adv, mask, clip_ratio=0.2)
    assert clipfrac.item() == 1.0
    assert torch.allclose(loss, torch.tensor(-1.2), atol=1e-5)


def test_policy_loss_fns_registry():
    assert algos.get_policy_loss_fn("vanilla") is algos.compute_policy_loss_vanilla
    assert algos.get_policy_loss_fn("gpg") is algos.compute_policy_loss_gpg
    assert algos.get_policy_loss_fn("clip_cov") is algos.compute_policy_loss_clip_cov
    with pytest.raises(NotImplementedError):
        algos.get_policy_loss_fn("nope")


def test_value_loss_clip():
    vpreds = torch.tensor([[1.0, 2.0]])
    values = torch.tensor([[0.0, 0.0]])
    returns = torch.tensor([[0.0, 0.0]])
    mask = torch.ones(1, 2)
    loss, clipfrac = algos.compute_value_loss(vpreds, returns, values, mask,
                                              cliprange_value=0.5)
    # clipped preds = 0.5 both; losses max(1,0.25)=1 and max(4,0.25)=4 -> 0.5*2.5
    assert torch.allclose(loss, torch.tensor(0.5 * 2.5), atol=1e-5)


def test_agg_loss_modes():
    loss = torch.tensor([[1.0, 2.0, 0.0], [3.0, 0.0, 0.0]])
    mask = torch.tensor([[1.0, 1.0, 0.0], [1.0, 0.0, 0.0]])
    assert torch.allclose(algos.agg_loss(loss, mask, "token-mean"),
                          torch.tensor(2.0), atol=1e-5)
    assert torch.allclose(algos.agg_loss(loss, mask, "seq-mean-token-sum"),
                          torch.tensor(3.0), atol=1e-5)
    assert torch.allclose(algos.agg_loss(loss, mask, "seq-mean-token-mean"),
                          torch.tensor((1.5 + 3.0) / 2), atol=1e-5)


def test_apply_kl_penalty():
    scores = torch.zeros(2, 3)
    scores[:, -1] = 1.0
    lp = torch.zeros(2, 3)
    ref_lp = torch.zeros(2, 3)
    mask = torch.ones(2, 3)
    rewards, kl = algos.apply_kl_penalty(scores, lp, ref_lp, mask, kl_coef=0.1)
    assert torch.allclose(rewards, scores)
    assert kl.item() == 0.0


def test_rloo_advantage():
    import numpy as np
    import torch
    from polyrl_amd.core import algos
    rewards = torch.zeros(4, 3)
    rewards[:, -1] = torch.tensor([1.0, 3.0, 2.0, 6.0])
    mask = torch.ones(4, 3)
    idx = np.array(["a", "a", "b", "b"], dtype=object)
    adv, ret = algos.compute_rloo_outcome_advantage(rewards, mask, idx)
    # group a: baselines are each other's score
    assert torch.allclose(adv[0], torch.full((3,), 1.0 - 3.0))
    assert torch.allclose(adv[1], torch.full((3,), 3.0 - 1.0))
    assert torch.allclose(adv[2], torch.full((3,), 2.0 - 6.0))
    assert torch.allclose(adv[3], torch.full((3,), 6.0 - 2.0))


def test_reinforce_plus_plus_advantage():
    import torch
    from polyrl_amd.core import algos
    rewards = torch.zeros(2, 4)
    rewards[0, 3] = 1.0
    rewards[1, 3] = -1.0
    mask = torch.ones(2, 4)
    adv, ret = algos.compute_reinforce_plus_plus_advantage(rewards, mask,
                                                           gamma=1.0)
    # reward-to-go is constant over tokens; whitening centers the two rows
    assert torch.allclose(ret[0], torch.ones(4))
    assert torch.allclose(ret[1], -torch.ones(4))
    assert (adv[0] > 0).all() and (adv[1] < 0).all()


def test_remax_advantage():
    import torch
    from polyrl_amd.core import algos
    rewards = torch.zeros(2, 3)
    rewards[:, -1] = torch.tensor([2.0, 0.5])
    mask = torch.ones(2, 3)
    base = torch.tensor([1.0, 1.0])
    adv, _ = algos.compute_remax_outcome_advantage(rewards, mask, base)
    assert torch.allclose(adv[0], torch.full((3,), 1.0))
    assert torch.allclose(adv[1], torch.full((3,), -0.5))


def test_compute_advantage_dispatch_new_estimators():
    import numpy as np
    import torch
    from polyrl_amd.core import algos
    rewards = torch.rand(4, 5)
    mask = torch.ones(4, 5)
    idx = np.array(["a", "a", "b", "b"], dtype=object)
    for est in ("rloo", "reinforce_plus_plus"):
        adv, ret = algos.compute_advantage(rewards, mask, est, index=idx)
        assert adv.shape == rewards.shape


def test_flops_counter_mfu():
    from polyrl_amd.core.metrics import FlopsCounter
    from polyrl_amd.models import get_model_config
    fc = FlopsCounter(get_model_config("llama3-8b"))
    # ~8B params -> per-token param flops ~ 2 * 8e9 (embedding-free)
    assert 1.2e10 < fc.per_token_params_flops < 1.8e10
    f = fc.train_step_flops(65536, 512)
    mfu = fc.mfu(f, seconds=7.0, n_gpus=1)
    assert 0.0 < mfu < 1.0


def test_vanilla_loss_importance_weights():
    """Truncated importance sampling (stream_dp_actor.py:153-224 capability):
    per-token weights scale the clipped surrogate before aggregation."""
    import torch

    from polyrl_amd.core import algos
    torch.manual_seed(0)
    B, L = 4, 6
    old = torch.randn(B, L) * 0.1
    new = old + torch.randn(B, L) * 0.05
    adv = torch.randn(B, L)
    mask = torch.ones(B, L)
    base, *_ = algos.compute_policy_loss_vanilla(old, new, adv, mask)
    iw = torch.full((B, L), 2.0)
    scaled, *_ = algos.compute_policy_loss_vanilla(
        old, new, adv, mask, importance_weights=iw)
    assert torch.allclose(scaled, base * 2.0, atol=1e-6)
    # cap semantics at the caller: exp(old - rollout).clamp(max=cap)
    rollout = old - 10.0                     # wildly off-policy rollout
    w = torch.exp((old - rollout).clamp(-20, 20)).clamp(max=3.0)
    assert torch.all(w == 3.0)
