"""Importing this package registers every algorithm
(parity: sheeprl/__init__.py:18-47)."""

from sheeprl_amd.algos import ppo  # noqa: F401
from sheeprl_amd.algos import dreamer_v3  # noqa: F401
from sheeprl_amd.algos import sac  # noqa: F401
from sheeprl_amd.algos import a2c  # noqa: F401
from sheeprl_amd.algos import droq  # noqa: F401
from sheeprl_amd.algos import ppo_recurrent  # noqa: F401
from sheeprl_amd.algos import sac_ae  # noqa: F401
from sheeprl_amd.algos import dreamer_v2  # noqa: F401
from sheeprl_amd.algos import dreamer_v1  # noqa: F401
from sheeprl_amd.algos import p2e_dv3  # noqa: F401
from sheeprl_amd.algos import p2e_dv2  # noqa: F401
from sheeprl_amd.algos import p2e_dv1  # noqa: F401
