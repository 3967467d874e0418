"""Example multi-turn interaction: a calculator "tool" the policy can call.

Contract (rollout.multi_turn.interaction_path/interaction_name, see
trainer/rollout_coordinator.py::load_interaction):

    generate_turn(prompt_ids, response_ids) -> (user_ids_or_None, done)

Both arguments are TOKEN ID lists (the framework is token-in/token-out,
like the engine).  Return the next user/tool turn's token ids and
done=False to continue the dialogue, or (None, True) to finalize.  The
framework inserts the returned tokens with loss_mask=0 (assistant-only
loss) and resubmits the continuation through whichever rollout path is
active (co-located, elastic, or disagg/scheduler).

This example treats a trailing token pair (a, b) of the assistant turn as
a "tool call" and answers with their sum modulo the vocab — stand-in for
detokenize -> run tool -> retokenize in a real deployment (no tokenizer
assets ship in this repo's offline environment).

Launch shape:

    python -m torch.distributed.run --nproc-per-node 8 \
      -m polyrl_amd.trainer.main_stream \
      actor_rollout_ref.rollout.multi_turn.enable=true \
      actor_rollout_ref.rollout.multi_turn.interaction_path=examples/interactions/calc_tool.py \
      actor_rollout_ref.rollout.multi_turn.max_assistant_turns=4 \
      actor_rollout_ref.rollout.multi_turn.per_turn_max_tokens=128 ...
"""

MAX_DIALOGUE_TOKENS = 256


def generate_turn(prompt_ids, response_ids):
    if len(response_ids) >= MAX_DIALOGUE_TOKENS or len(response_ids) < 2:
        return None, True                    # finalize the sample
    a, b = response_ids[-2], response_ids[-1]
    tool_result = [(a + b) % 32000]          # the "tool" output tokens
    return tool_result, False                # continue with loss_mask=0
