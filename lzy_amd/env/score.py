"""Pool scoring functions namespace (reference:
pylzy/lzy/env/provisioning/score.py — imported as ``score`` from
``lzy.api.v1``).  Both functions plug into
``Provisioning.resolve_pool(pools, score=...)``."""
from lzy_amd.env.provisioning import (  # noqa: F401
    maximum_score_function,
    minimum_score_function,
)
