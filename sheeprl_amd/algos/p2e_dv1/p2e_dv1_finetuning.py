"""Plan2Explore DV1 — finetuning (parity: sheeprl/algos/p2e_dv1/
p2e_dv1_finetuning.py): continue with the plain DV1 task update from the
exploration checkpoint's world model + task actor/critic."""

from __future__ import annotations

from typing import Any, Dict

from sheeprl_amd.parallel import Runtime
from sheeprl_amd.utils.registry import register_algorithm, register_evaluation


@register_algorithm(name="p2e_dv1_finetuning")
def main(runtime: Runtime, cfg: Any) -> None:
    import sheeprl_amd.algos.dreamer_v1.dreamer_v1 as dv1

    ckpt_path = cfg.checkpoint.get("exploration_ckpt_path")
    if not ckpt_path:
        raise ValueError("p2e_dv1_finetuning needs checkpoint.exploration_ckpt_path")
    state = runtime.load(ckpt_path)
    orig_build = dv1.build_agent

    def build_with_state(rt, actions_dim, is_continuous, cfg_, obs_space, *unused):
        return orig_build(
            rt, actions_dim, is_continuous, cfg_, obs_space,
            state.get("world_model"), state.get("actor_task"), state.get("critic_task"),
        )

    dv1.build_agent = build_with_state
    try:
        fn = dv1.main.__wrapped__ if hasattr(dv1.main, "__wrapped__") else dv1.main
        fn(runtime, cfg)
    finally:
        dv1.build_agent = orig_build


@register_evaluation(algorithms=["p2e_dv1_finetuning"])
def evaluate(runtime: Runtime, cfg: Any, state: Dict[str, Any]) -> float:
    from sheeprl_amd.algos.dreamer_v1.dreamer_v1 import evaluate as dv1_eval

    fn = dv1_eval.__wrapped__ if hasattr(dv1_eval, "__wrapped__") else dv1_eval
    return fn(runtime, cfg, state)
