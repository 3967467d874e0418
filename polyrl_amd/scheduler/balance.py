"""Adaptive local-generation time-box (reference: rollout-manager/src/
balance.rs:23-213).

The co-located engines generate only for ``max_local_gen_s`` seconds each
iteration before the trainer GPUs context-switch back to training; remote
elastic instances finish the tail.  This state machine adapts that window
from trainer feedback:

  * per-instance-count memo of the best-known gen_s, EMA-updated
    (alpha=0.8 on a throughput plateau, beta=0.2 when the instance count
    changes, balance.rs:23-24,105-191),
  * gradient step: if the trainer bubble (time the trainer waited for
    samples) is smaller than the remote bubble, shrink the local window by
    delta/3 (floor 5 s), else grow it by delta/3 (balance.rs:193-205).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict


ALPHA_PLATEAU = 0.8     # EMA weight on new value when throughput plateaus
BETA_COUNT_CHANGE = 0.2  # EMA weight on new value when instance count changes
MIN_GEN_S = 5.0
DEFAULT_GEN_S = 150.0   # state.rs:79 LoadBalanceState::new(150)


@dataclass
class LoadBalanceState:
    initial_gen_s: float = DEFAULT_GEN_S
    # best-known local gen window per remote-instance count (balance.rs:57-71
    # seeds an 8B-model memo {1:190, 2:160, 3:105, 4:70}; we start empty and
    # learn since the hardware differs)
    optimal_gen_s: Dict[int, float] = field(default_factory=dict)
    current_gen_s: float = 0.0
    last_instance_count: int = -1
    last_step_throughput: float = 0.0
    best_throughput: float = 0.0
    gen_s_at_best: float = 0.0

    def __post_init__(self):
        if self.current_gen_s <= 0:
            self.current_gen_s = self.initial_gen_s

    def update(self, step_time_s: float, trainer_bubble_s: float,
               step_throughput: float, num_instances: int) -> float:
        """One feedback step -> new max_local_gen_s (handlers.rs:867-901)."""
        if num_instances != self.last_instance_count:
            # instance count changed: blend toward that count's memo
            memo = self.optimal_gen_s.get(num_instances)
            if memo is not None:
                self.current_gen_s = (
                    (1 - BETA_COUNT_CHANGE) * memo
                    + BETA_COUNT_CHANGE * self.current_gen_s)
            self.last_instance_count = num_instances
            self.best_throughput = step_throughput
            self.gen_s_at_best = self.current_gen_s
            self.last_step_throughput = step_throughput
            return self.current_gen_s

        # track peak throughput and its window
        if step_throughput > self.best_throughput:
            self.best_throughput = step_throughput
            self.gen_s_at_best = self.current_gen_s
            memo = self.optimal_gen_s.get(num_instances)
            if memo is None:
                self.optimal_gen_s[num_instances] = self.current_gen_s
            else:
                self.optimal_gen_s[num_instances] = (
                    ALPHA_PLATEAU * self.current_gen_s
                    + (1 - ALPHA_PLATEAU) * memo)

        # gradient step on the window (balance.rs:193-205): the remote bubble
        # is the generation time not overlapped with training
        remote_bubble = max(step_time_s - trainer_bubble_s - self.current_gen_s,
                            0.0)
        delta = abs(trainer_bubble_s - remote_bubble)
        if trainer_bubble_s < remote_bubble:
            self.current_gen_s = max(self.current_gen_s - delta / 3.0,
                                     MIN_GEN_S)
        else:
            self.current_gen_s = self.current_gen_s + delta / 3.0
        self.last_step_throughput = step_throughput
        return self.current_gen_s
