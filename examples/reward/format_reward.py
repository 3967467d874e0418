"""Example custom reward function (custom_reward_function.path/.name).

Contract (reward.py::FunctionReward): the function receives one
sample as a sliced TensorBatch (tensors: input_ids / responses /
response_mask / attention_mask; non-tensors: data_source, ground_truth,
raw_prompt where present) plus any reward_kwargs from the config, and
returns a float score.  Scores land on the last valid response token
(token-level credit is the KL penalty's job).

    python -m polyrl_amd.trainer.main_stream reward=config \
      custom_reward_function.path=examples/reward/format_reward.py \
      custom_reward_function.name=compute_score ...
"""


def compute_score(sample, length_target: int = 64, **_):
    """+1 for hitting a response-length band, -1 outside it — a tiny
    shaped reward useful for smoke-testing reward plumbing end to end."""
    n = int(sample["response_mask"].sum())
    return 1.0 if 0.5 * length_target <= n <= 2.0 * length_target else -1.0
