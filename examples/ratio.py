"""Replay-ratio scheduler walkthrough (parity: examples/ratio.py in the
reference): shows how `Ratio` converts policy steps into per-rank gradient
repeats for a Hafner-style replay ratio."""

from sheeprl_amd.utils.utils import Ratio

if __name__ == "__main__":
    num_envs = 1
    world_size = 1
    replay_ratio = 0.0625
    per_rank_batch_size = 16
    per_rank_sequence_length = 64
    replayed_steps = world_size * per_rank_batch_size * per_rank_sequence_length
    gradient_steps = 0
    total_policy_steps = 2**10
    r = Ratio(ratio=replay_ratio, pretrain_steps=0)
    policy_steps = num_envs * world_size
    for i in range(0, total_policy_steps, policy_steps):
        if i >= 128:
            per_rank_repeats = r(i / world_size)
            if per_rank_repeats > 0:
                print(
                    f"{per_rank_repeats} repeats/rank ({per_rank_repeats * world_size} global) "
                    f"at global iteration {i}"
                )
            gradient_steps += per_rank_repeats * world_size
    print("Replay ratio   ", replay_ratio)
    print("Hafner ratio   ", replay_ratio * replayed_steps)
    print("Realized ratio ", gradient_steps / total_policy_steps)
