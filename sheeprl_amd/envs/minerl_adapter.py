"""MineRL adapter (parity surface: sheeprl/envs/minerl.py:78-322).

The MineRL simulator takes DICT actions ({"forward": 1, "camera": (p, y),
"craft": item, ...}); the agent sees one flat ``Discrete`` head built at
construction time by enumerating the env's action space:

* index 0 = no-op;
* each binary control gets one index (jump/sneak/sprint auto-add forward);
* the camera gets four ±15-degree nudges (pitch down/up, yaw left/right);
* each Enum control (craft/nearbyCraft/nearbySmelt/place/equip) expands to
  one index per non-"none" value.

Stateful behaviours reproduced from the reference: sticky attack (suppresses
jump while held), sticky jump (auto-forward), pitch limiting with wrapped
yaw tracking, inventory count/max vectors (multihot over all Minecraft items
or the task's own item list), and the mainhand-equipment one-hot with the
"air" fallback for unknown items.

The ``minerl`` package is absent from this image: the env instance and item
table are injectable for the fake-backend unit tests
(tests/test_envs_adapters.py); with minerl installed the adapter builds the
task env itself.
"""

from __future__ import annotations

import copy
from typing import Any, Dict, List, Optional, Sequence, Tuple

import numpy as np

from sheeprl_amd.envs import spaces
from sheeprl_amd.envs.core import Env

CAMERA_DELTA = 15.0
NOOP_ACTION: Dict[str, Any] = {
    "camera": (0.0, 0.0),
    "forward": 0, "back": 0, "left": 0, "right": 0,
    "attack": 0, "sprint": 0, "jump": 0, "sneak": 0,
    "craft": "none", "nearbyCraft": "none", "nearbySmelt": "none",
    "place": "none", "equip": "none",
}


class MineRLAdapter(Env):
    def __init__(
        self,
        id: str = "navigate",
        height: int = 64,
        width: int = 64,
        pitch_limits: Tuple[int, int] = (-60, 60),
        seed: Optional[int] = None,
        sticky_attack: int = 30,
        sticky_jump: int = 10,
        break_speed_multiplier: int = 100,
        multihot_inventory: bool = True,
        sim_env: Any = None,            # injectable simulator instance (tests)
        all_items: Optional[Sequence[str]] = None,
        **kwargs: Any,
    ) -> None:
        if sim_env is None:  # pragma: no cover - package absent in image
            try:
                import minerl  # noqa: F401, PLC0415
                from minerl.herobraine.hero import mc  # noqa: PLC0415
            except ImportError as e:
                raise ImportError(
                    "minerl is not installed; `pip install minerl` (needs a JDK) "
                    "or pass sim_env= / all_items= for an injected backend"
                ) from e
            all_items = list(mc.ALL_ITEMS)
            raise ImportError("minerl task construction requires the task registry; inject sim_env=")
        self._env = sim_env
        self._pitch_limits = pitch_limits
        self._sticky_attack = 0 if break_speed_multiplier > 1 else sticky_attack
        self._sticky_jump = sticky_jump
        self._attack_left = 0
        self._jump_left = 0
        self._pos = {"pitch": 0.0, "yaw": 0.0}
        self._multihot = multihot_inventory
        self._all_items = list(all_items or [])

        # flatten the simulator's dict action space into one Discrete head
        self._action_table: List[Dict[str, Any]] = [{}]
        for name, sp in self._env.action_space.items():
            values = getattr(sp, "values", None)
            if values is not None:  # Enum control: one index per real value
                entries = [{name: v} for v in values if v != "none"]
            elif name == "camera":
                entries = [
                    {name: np.array([-CAMERA_DELTA, 0.0])},
                    {name: np.array([+CAMERA_DELTA, 0.0])},
                    {name: np.array([0.0, -CAMERA_DELTA])},
                    {name: np.array([0.0, +CAMERA_DELTA])},
                ]
            else:  # binary control
                entries = [{name: 1}]
            if name in ("jump", "sneak", "sprint"):
                entries[0]["forward"] = 1
            self._action_table.extend(entries)
        self.action_space = spaces.Discrete(len(self._action_table))

        inv_names = list(self._env.observation_space["inventory"])
        if multihot_inventory:
            self._inv_id = {n: i for i, n in enumerate(self._all_items)}
            self._inv_size = len(self._all_items)
        else:
            self._inv_id = {n: i for i, n in enumerate(inv_names)}
            self._inv_size = len(inv_names)
        obs_spaces: Dict[str, spaces.Space] = {
            "rgb": spaces.Box(0, 255, (3, height, width), np.uint8),
            "life_stats": spaces.Box(0.0, 300.0, (3,), np.float32),
            "inventory": spaces.Box(0.0, np.inf, (self._inv_size,), np.float32),
            "max_inventory": spaces.Box(0.0, np.inf, (self._inv_size,), np.float32),
        }
        obs_sp = self._env.observation_space
        if "compass" in obs_sp:
            obs_spaces["compass"] = spaces.Box(-180.0, 180.0, (1,), np.float32)
        self._has_equipment = "equipped_items" in obs_sp
        if self._has_equipment:
            if multihot_inventory:
                self._equip_id = self._inv_id
                self._equip_size = self._inv_size
            else:
                names = list(obs_sp["equipped_items"]["mainhand"]["type"].values)
                self._equip_id = {n: i for i, n in enumerate(names)}
                self._equip_size = len(names)
            obs_spaces["equipment"] = spaces.Box(0.0, 1.0, (self._equip_size,), np.int32)
        self.observation_space = spaces.Dict(obs_spaces)
        self._max_inv = np.zeros(self._inv_size, dtype=np.float32)

    # ---- actions -----------------------------------------------------------
    def _lower_action(self, action: Any) -> Dict[str, Any]:
        low = copy.deepcopy(NOOP_ACTION)
        low.update(self._action_table[int(np.asarray(action).reshape(-1)[0])])
        if self._sticky_attack:
            if low["attack"]:
                self._attack_left = self._sticky_attack
            if self._attack_left > 0:
                low["attack"] = 1
                low["jump"] = 0  # a held attack suppresses jumping
                self._attack_left -= 1
        if self._sticky_jump:
            if low["jump"]:
                self._jump_left = self._sticky_jump
            if self._jump_left > 0:
                low["jump"] = 1
                low["forward"] = 1  # keep moving while the held jump replays
                self._jump_left -= 1
        return low

    # ---- observations ------------------------------------------------------
    def _vector_inventory(self, inv: Dict[str, Any]) -> Dict[str, np.ndarray]:
        counts = np.zeros(self._inv_size, dtype=np.float32)
        for item, qty in inv.items():
            counts[self._inv_id[item]] += 1.0 if item == "air" else float(np.asarray(qty))
        self._max_inv = np.maximum(counts, self._max_inv)
        return {"inventory": counts, "max_inventory": self._max_inv.copy()}

    def _one_hot_equipment(self, equipped: Dict[str, Any]) -> np.ndarray:
        out = np.zeros(self._equip_size, dtype=np.int32)
        name = equipped["mainhand"]["type"]
        out[self._equip_id.get(name, self._equip_id["air"])] = 1
        return out

    def _convert_obs(self, obs: Dict[str, Any]) -> Dict[str, np.ndarray]:
        life = obs["life_stats"]
        out = {
            "rgb": np.asarray(obs["pov"]).transpose(2, 0, 1).copy(),
            "life_stats": np.array([life["life"], life["food"], life["air"]], dtype=np.float32).reshape(3),
            **self._vector_inventory(obs["inventory"]),
        }
        if self._has_equipment:
            out["equipment"] = self._one_hot_equipment(obs["equipped_items"])
        if "compass" in self.observation_space.keys():
            out["compass"] = np.asarray(obs["compass"]["angle"], dtype=np.float32).reshape(-1)
        return out

    # ---- Env API -----------------------------------------------------------
    def reset(self, *, seed: Optional[int] = None, options: Optional[dict] = None):
        obs = self._env.reset()
        self._max_inv = np.zeros(self._inv_size, dtype=np.float32)
        self._attack_left = 0
        self._jump_left = 0
        self._pos = {"pitch": 0.0, "yaw": 0.0}
        return self._convert_obs(obs), {}

    def step(self, action: Any):
        low = self._lower_action(action)
        cam = np.asarray(low["camera"], dtype=np.float32)
        next_pitch = self._pos["pitch"] + float(cam[0])
        next_yaw = ((self._pos["yaw"] + float(cam[1])) + 180.0) % 360.0 - 180.0
        if not (self._pitch_limits[0] <= next_pitch <= self._pitch_limits[1]):
            low["camera"] = np.array([0.0, float(cam[1])])
            next_pitch = self._pos["pitch"]
        obs, reward, done, info = self._env.step(low)
        self._pos = {"pitch": next_pitch, "yaw": next_yaw}
        return self._convert_obs(obs), float(reward), bool(done), False, dict(info)

    def close(self) -> None:
        if hasattr(self._env, "close"):
            self._env.close()
