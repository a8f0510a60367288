from .common import EnvBase, EnvMetaData
from .utils import (
    ExplorationType,
    RandomPolicy,
    check_env_specs,
    exploration_type,
    make_composite_from_td,
    set_exploration_type,
    step_mdp,
    terminated_or_truncated,
)
