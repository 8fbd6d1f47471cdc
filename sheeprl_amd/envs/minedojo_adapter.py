"""MineDojo adapter (parity surface: sheeprl/envs/minedojo.py:56-307).

Re-implements the reference's MineDojo integration MI355X-side: the simulator
exposes an 8-slot low-level action vector
``[move_fb, move_lr, jump_sneak_sprint, pitch_bin, yaw_bin, functional,
craft_arg, inventory_slot]`` (camera bins 0..24, 12 = centre, one bin = 15
degrees) which the agent sees as a flat ``MultiDiscrete([19, n_craft,
n_items])``:

* head 0 — one of 19 composite controls (no-op, moves, jump/sneak/sprint
  combos, four camera nudges, and the 7 functional verbs);
* head 1 — the craft/smelt argument (consumed only by the craft verb);
* head 2 — the TARGET ITEM of equip/place/destroy, translated to the item's
  current inventory slot.

It also reproduces the reference's stateful behaviours: sticky attack /
sticky jump repetition, pitch limiting, the item-count / max / delta
vectors, the equipment one-hot, and the action masks that feed
``MinedojoActor``'s logit masking (dreamer agents).

The heavy ``minedojo`` package is not in this image: the adapter takes any
object with the simulator API (``make``/``reset``/``step`` + item tables),
so the logic is fully unit-tested against a fake backend
(tests/test_envs_adapters.py) and binds to the real package when present.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env

# functional-verb ids in the simulator's slot 5
FN_NOOP, FN_USE, FN_DROP, FN_ATTACK, FN_CRAFT, FN_EQUIP, FN_PLACE, FN_DESTROY = range(8)
CAM_CENTER = 12  # camera bin for "no rotation"; one bin = 15 degrees
CAM_BIN_DEG = 15.0


def _build_action_table() -> List[np.ndarray]:
    """The 19 composite controls of the flattened action head, expressed as
    (move_fb, move_lr, jump_mode, d_pitch_bins, d_yaw_bins, functional)."""
    specs = [
        (0, 0, 0, 0, 0, FN_NOOP),      # 0 no-op
        (1, 0, 0, 0, 0, FN_NOOP),      # 1 forward
        (2, 0, 0, 0, 0, FN_NOOP),      # 2 back
        (0, 1, 0, 0, 0, FN_NOOP),      # 3 strafe left
        (0, 2, 0, 0, 0, FN_NOOP),      # 4 strafe right
        (1, 0, 1, 0, 0, FN_NOOP),      # 5 jump + forward
        (1, 0, 2, 0, 0, FN_NOOP),      # 6 sneak + forward
        (1, 0, 3, 0, 0, FN_NOOP),      # 7 sprint + forward
        (0, 0, 0, -1, 0, FN_NOOP),     # 8 pitch down
        (0, 0, 0, +1, 0, FN_NOOP),     # 9 pitch up
        (0, 0, 0, 0, -1, FN_NOOP),     # 10 yaw left
        (0, 0, 0, 0, +1, FN_NOOP),     # 11 yaw right
        (0, 0, 0, 0, 0, FN_USE),       # 12
        (0, 0, 0, 0, 0, FN_DROP),      # 13
        (0, 0, 0, 0, 0, FN_ATTACK),    # 14
        (0, 0, 0, 0, 0, FN_CRAFT),     # 15
        (0, 0, 0, 0, 0, FN_EQUIP),     # 16
        (0, 0, 0, 0, 0, FN_PLACE),     # 17
        (0, 0, 0, 0, 0, FN_DESTROY),   # 18
    ]
    table = []
    for fb, lr, jm, dp, dy, fn in specs:
        table.append(np.array([fb, lr, jm, CAM_CENTER + dp, CAM_CENTER + dy, fn, 0, 0], dtype=np.int64))
    return table


ACTION_TABLE = _build_action_table()
N_COMPOSITE = len(ACTION_TABLE)


def _norm_item(name: str) -> str:
    return "_".join(str(name).split(" "))


class MineDojoAdapter(Env):
    def __init__(
        self,
        id: str = "open-ended",
        height: int = 64,
        width: int = 64,
        pitch_limits: Tuple[int, int] = (-60, 60),
        seed: Optional[int] = None,
        sticky_attack: int = 30,
        sticky_jump: int = 10,
        backend: Any = None,        # injectable simulator module (tests)
        **kwargs: Any,
    ) -> None:
        if backend is None:
            try:
                import minedojo  # noqa: PLC0415

                backend = minedojo
            except ImportError as e:  # pragma: no cover - package absent in image
                raise ImportError(
                    "minedojo is not installed; `pip install minedojo` (needs a JDK) "
                    "or pass a backend implementing make()/ALL_ITEMS/ALL_CRAFT_SMELT_ITEMS"
                ) from e
        self._pitch_limits = pitch_limits
        break_speed = kwargs.pop("break_speed_multiplier", 100)
        # fast breaking makes held attacks pointless (reference :76)
        self._sticky_attack = 0 if break_speed > 1 else sticky_attack
        self._sticky_jump = sticky_jump
        self._attack_left = 0
        self._jump_left = 0

        self._items: Sequence[str] = list(getattr(backend, "ALL_ITEMS", None) or backend.sim.ALL_ITEMS)
        self._craft_items: Sequence[str] = list(
            getattr(backend, "ALL_CRAFT_SMELT_ITEMS", None) or backend.sim.ALL_CRAFT_SMELT_ITEMS
        )
        self._item_id = {n: i for i, n in enumerate(self._items)}
        self._n_items = len(self._items)
        start_pos = kwargs.get("start_position")
        if start_pos is not None and not (pitch_limits[0] <= start_pos["pitch"] <= pitch_limits[1]):
            raise ValueError(f"start position pitch {start_pos['pitch']} outside limits {pitch_limits}")

        self._sim = backend.make(
            task_id=id,
            image_size=(height, width),
            world_seed=seed,
            fast_reset=True,
            break_speed_multiplier=break_speed,
            **kwargs,
        )
        self._pos: Dict[str, float] = dict(start_pos or {})
        self._slot_of: Dict[str, List[int]] = {}
        self._slot_names: np.ndarray = np.array([], dtype=object)
        self._inv_max = np.zeros(self._n_items, dtype=np.float32)

        self.action_space = spaces.MultiDiscrete([N_COMPOSITE, len(self._craft_items), self._n_items])
        img_shape = tuple(self._sim.observation_space["rgb"].shape)
        self.observation_space = spaces.Dict({
            "rgb": spaces.Box(0, 255, img_shape, np.uint8),
            "inventory": spaces.Box(0.0, np.inf, (self._n_items,), np.float32),
            "inventory_max": spaces.Box(0.0, np.inf, (self._n_items,), np.float32),
            "inventory_delta": spaces.Box(-np.inf, np.inf, (self._n_items,), np.float32),
            "equipment": spaces.Box(0.0, 1.0, (self._n_items,), np.int32),
            "life_stats": spaces.Box(0.0, 300.0, (3,), np.float32),
            "mask_action_type": spaces.Box(0, 1, (N_COMPOSITE,), np.bool_),
            "mask_equip_place": spaces.Box(0, 1, (self._n_items,), np.bool_),
            "mask_destroy": spaces.Box(0, 1, (self._n_items,), np.bool_),
            "mask_craft_smelt": spaces.Box(0, 1, (len(self._craft_items),), np.bool_),
        })

    # ---- action flattening -------------------------------------------------
    def _lower_action(self, action: np.ndarray) -> np.ndarray:
        """Flat MultiDiscrete triple -> the simulator's 8-slot vector, with
        sticky attack/jump and the craft/target argument routing."""
        low = ACTION_TABLE[int(action[0])].copy()
        if self._sticky_attack:
            if low[5] == FN_ATTACK:
                self._attack_left = self._sticky_attack - 1
            elif low[5] == FN_NOOP and self._attack_left > 0:
                low[5] = FN_ATTACK
                self._attack_left -= 1
            else:
                self._attack_left = 0
        if self._sticky_jump:
            if low[2] == 1:
                self._jump_left = self._sticky_jump - 1
            elif self._jump_left > 0 and low[0] == 0:
                low[2] = 1
                if low[1] == 0:
                    low[0] = 1  # keep moving while the held jump replays
                self._jump_left -= 1
            elif low[2] != 1:
                self._jump_left = 0
        low[6] = int(action[1]) if low[5] == FN_CRAFT else 0
        if low[5] in (FN_EQUIP, FN_PLACE, FN_DESTROY):
            low[7] = self._slot_of[self._items[int(action[2])]][0]
        else:
            low[7] = 0
        return low

    # ---- observation conversion -------------------------------------------
    def _vector_inventory(self, inv: Dict[str, Any]) -> np.ndarray:
        counts = np.zeros(self._n_items, dtype=np.float32)
        self._slot_of = {}
        names = []
        for slot, (name, qty) in enumerate(zip(inv["name"], inv["quantity"])):
            name = _norm_item(name)
            names.append(name)
            self._slot_of.setdefault(name, []).append(slot)
            counts[self._item_id[name]] += 1.0 if name == "air" else float(qty)
        self._slot_names = np.array(names, dtype=object)
        self._inv_max = np.maximum(counts, self._inv_max)
        return counts

    def _vector_delta(self, delta: Dict[str, Any]) -> np.ndarray:
        out = np.zeros(self._n_items, dtype=np.float32)
        for names_k, qty_k, sign in (
            ("inc_name_by_craft", "inc_quantity_by_craft", +1),
            ("dec_name_by_craft", "dec_quantity_by_craft", -1),
            ("inc_name_by_other", "inc_quantity_by_other", +1),
            ("dec_name_by_other", "dec_quantity_by_other", -1),
        ):
            for name, qty in zip(delta[names_k], delta[qty_k]):
                out[self._item_id[_norm_item(name)]] += sign * float(qty)
        return out

    def _convert_masks(self, masks: Dict[str, Any]) -> Dict[str, np.ndarray]:
        equip = np.zeros(self._n_items, dtype=bool)
        destroy = np.zeros(self._n_items, dtype=bool)
        for name, em, dm in zip(self._slot_names, masks["equip"], masks["destroy"]):
            i = self._item_id[name]
            equip[i] = equip[i] or bool(em)
            destroy[i] = destroy[i] or bool(dm)
        fn = np.asarray(masks["action_type"], dtype=bool).copy()
        # equip/place impossible without an equippable item; destroy likewise
        fn[FN_EQUIP] = fn[FN_EQUIP] and bool(equip.any())
        fn[FN_PLACE] = fn[FN_PLACE] and bool(equip.any())
        fn[FN_DESTROY] = fn[FN_DESTROY] and bool(destroy.any())
        # composite mask: movement/camera controls always allowed, then the 7 verbs
        action_type = np.concatenate((np.ones(12, dtype=bool), fn[1:8]))
        return {
            "mask_action_type": action_type,
            "mask_equip_place": equip,
            "mask_destroy": destroy,
            "mask_craft_smelt": np.asarray(masks["craft_smelt"], dtype=bool),
        }

    def _track_pos(self, obs: Dict[str, Any]) -> None:
        loc = obs["location_stats"]
        self._pos = {
            "x": float(loc["pos"][0]), "y": float(loc["pos"][1]), "z": float(loc["pos"][2]),
            "pitch": float(np.asarray(loc["pitch"]).item()),
            "yaw": float(np.asarray(loc["yaw"]).item()),
        }

    def _convert_obs(self, obs: Dict[str, Any]) -> Dict[str, np.ndarray]:
        life = obs["life_stats"]
        return {
            "rgb": np.asarray(obs["rgb"]).copy(),
            "inventory": self._vector_inventory(obs["inventory"]),
            "inventory_max": self._inv_max,
            "inventory_delta": self._vector_delta(obs["delta_inv"]),
            "equipment": self._one_hot(_norm_item(obs["equipment"]["name"][0])),
            "life_stats": np.concatenate(
                (np.atleast_1d(life["life"]), np.atleast_1d(life["food"]), np.atleast_1d(life["oxygen"]))
            ).astype(np.float32),
            **self._convert_masks(obs["masks"]),
        }

    def _one_hot(self, name: str) -> np.ndarray:
        out = np.zeros(self._n_items, dtype=np.int32)
        out[self._item_id[name]] = 1
        return out

    def _info(self, obs: Dict[str, Any]) -> Dict[str, Any]:
        life = obs["life_stats"]
        return {
            "life_stats": {
                "life": float(np.asarray(life["life"]).item()),
                "food": float(np.asarray(life["food"]).item()),
                "oxygen": float(np.asarray(life["oxygen"]).item()),
            },
            "location_stats": copy.deepcopy(self._pos),
            "biomeid": float(np.asarray(obs["location_stats"]["biome_id"]).item()),
        }

    # ---- Env API -----------------------------------------------------------
    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        obs = self._sim.reset()
        self._attack_left = 0
        self._jump_left = 0
        self._inv_max = np.zeros(self._n_items, dtype=np.float32)
        self._track_pos(obs)
        return self._convert_obs(obs), self._info(obs)

    def step(self, action: np.ndarray):
        low = self._lower_action(np.asarray(action).reshape(-1))
        # pitch limiting: cancel a camera nudge that would leave the limits
        next_pitch = self._pos.get("pitch", 0.0) + (low[3] - CAM_CENTER) * CAM_BIN_DEG
        if not (self._pitch_limits[0] <= next_pitch <= self._pitch_limits[1]):
            low[3] = CAM_CENTER
        obs, reward, done, info = self._sim.step(low)
        truncated = bool(info.get("TimeLimit.truncated", False)) and bool(done)
        terminated = bool(done) and not truncated
        self._track_pos(obs)
        out_info = dict(info)
        out_info.update(self._info(obs))
        out_info["action"] = np.asarray(action).tolist()
        return self._convert_obs(obs), float(reward), terminated, truncated, out_info

    def close(self) -> None:
        if hasattr(self._sim, "close"):
            self._sim.close()
