"""Synchronous colocated PPO/GRPO baseline entry (the reference's
main_ppo.py A/B arm): identical trainer, but the stream
size is forced to the full batch so every update waits for the whole
generation (no rollout/update overlap)."""
from __future__ import annotations

import sys

from .main_stream import main as _stream_main


def main(argv=None):
    argv = list(sys.argv[1:] if argv is None else argv)
    # force synchronous behavior unless explicitly overridden
    has_stream = any(a.startswith(
        "actor_rollout_ref.rollout.min_stream_batch_size=") for a in argv)
    if not has_stream:
        tb = next((a.split("=", 1)[1] for a in argv
                   if a.startswith("data.train_batch_size=")), None)
        n = next((a.split("=", 1)[1] for a in argv
                  if a.startswith("actor_rollout_ref.rollout.sampling.n=")),
                 "1")
        if tb is not None:
            argv.append("actor_rollout_ref.rollout.min_stream_batch_size="
                        f"{int(tb) * int(n)}")
    return _stream_main(argv)


if __name__ == "__main__":
    main()
