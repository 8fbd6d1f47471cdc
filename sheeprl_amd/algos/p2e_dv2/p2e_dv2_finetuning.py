"""Plan2Explore DV2 — finetuning (parity: sheeprl/algos/p2e_dv2/
p2e_dv2_finetuning.py): resume the exploration checkpoint's world model and
task actor/critic and continue with the plain DV2 task update."""

from __future__ import annotations

from typing import Any, Dict

from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation


@register_algorithm(name="p2e_dv2_finetuning")
def main(runtime: Runtime, cfg: Any) -> None:
    import sheeprl_amd.algos.dreamer_v2.agent as dv2_agent
    import sheeprl_amd.algos.dreamer_v2.dreamer_v2 as dv2

    ckpt_path = cfg.checkpoint.get("exploration_ckpt_path")
    if not ckpt_path:
        raise ValueError("p2e_dv2_finetuning needs checkpoint.exploration_ckpt_path")
    state = runtime.load(ckpt_path)
    orig_build = dv2_agent.build_agent

    def build_with_state(rt, actions_dim, is_continuous, cfg_, obs_space, *unused):
        return orig_build(
            rt, actions_dim, is_continuous, cfg_, obs_space,
            state.get("world_model"), state.get("actor_task"),
            state.get("critic_task"), state.get("target_critic_task"),
        )

    dv2.build_agent = build_with_state
    try:
        fn = dv2.main.__wrapped__ if hasattr(dv2.main, "__wrapped__") else dv2.main
        fn(runtime, cfg)
    finally:
        dv2.build_agent = orig_build


@register_evaluation(algorithms=["p2e_dv2_finetuning"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    from sheeprl_amd.algos.dreamer_v2.dreamer_v2 import evaluate as dv2_eval

    return dv2_eval(runtime, cfg, state)
