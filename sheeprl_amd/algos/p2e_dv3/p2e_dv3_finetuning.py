"""Plan2Explore DV3 — finetuning phase.

Parity: sheeprl/algos/p2e_dv3/p2e_dv3_finetuning.py:28 — load the exploration
checkpoint (``checkpoint.exploration_ckpt_path``), keep the world model and
task actor/critic, and continue training on the REAL task reward with the
plain DV3 update; the player switches to the task actor.
"""

from __future__ import annotations

from typing import Any, Dict

from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation


@register_algorithm(name="p2e_dv3_finetuning")
def main(runtime: Runtime, cfg: Any) -> None:
    import sheeprl_amd.algos.dreamer_v3.dreamer_v3 as dv3
    import sheeprl_amd.algos.dreamer_v3.agent as dv3_agent

    ckpt_path = cfg.checkpoint.get("exploration_ckpt_path")
    if not ckpt_path:
        raise ValueError("p2e_dv3_finetuning needs checkpoint.exploration_ckpt_path (the exploration run ckpt)")
    state = runtime.load(ckpt_path)

    orig_build = dv3_agent.build_agent

    def build_with_exploration_state(rt, actions_dim, is_continuous, cfg_, obs_space, *unused):
        return orig_build(
            rt, actions_dim, is_continuous, cfg_, obs_space,
            state.get("world_model"), state.get("actor_task"),
            state.get("critic_task"), state.get("target_critic_task"),
        )

    # the DV3 loop rebuilds the agent from the exploration weights
    import sheeprl_amd.algos.dreamer_v3.dreamer_v3 as dv3_mod

    dv3_mod.build_agent = build_with_exploration_state
    try:
        entry = dv3.main
        fn = entry.__wrapped__ if hasattr(entry, "__wrapped__") else entry
        fn(runtime, cfg)
    finally:
        dv3_mod.build_agent = orig_build


@register_evaluation(algorithms=["p2e_dv3_finetuning"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    from sheeprl_amd.algos.dreamer_v3.evaluate import evaluate as dv3_eval

    return dv3_eval(runtime, cfg, state)
