"""In-process rollout scheduler — MI355X-native replacement for the
reference's Rust ``rollout-manager`` service (reference: rollout-manager/src/
{state.rs,handlers.rs,balance.rs,instance_manager.rs}).

Same behavioral contract, no HTTP hop for co-located instances:
  * instance registry with health gating and 1 Hz stats sampling,
  * zero-queue round-robin dispatch with per-stats-window admission throttling
    (state.rs:84-147),
  * streamed prompt-group results while the trainer updates (handlers.rs:442-564),
  * token-level continuation of failed/aborted requests on another instance
    (handlers.rs:330-418, utils.rs:140-291),
  * weight-version gating of the active pool (handlers.rs:566-795),
  * adaptive local-generation time-box (balance.rs:93-213).
"""
from .types import GroupRequest, GroupResult, SampleResult
from .balance import LoadBalanceState
from .instances import FakeInstance, InProcessInstance, RolloutInstance
from .manager import RolloutScheduler, SchedulerConfig

__all__ = [
    "GroupRequest", "GroupResult", "SampleResult", "LoadBalanceState",
    "RolloutInstance", "InProcessInstance", "FakeInstance",
    "RolloutScheduler", "SchedulerConfig",
]
