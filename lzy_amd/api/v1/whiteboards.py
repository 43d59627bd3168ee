"""Alias of the reference import path ``lzy.api.v1.whiteboards``."""
from lzy_amd.whiteboards.wb import (  # noqa: F401
    ReadOnlyWhiteboard,
    WritableWhiteboard,
    whiteboard,
    whiteboard_,
)
