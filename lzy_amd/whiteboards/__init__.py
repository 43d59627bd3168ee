from lzy_amd.whiteboards.wb import whiteboard_, whiteboard, is_whiteboard

__all__ = ["whiteboard_", "whiteboard", "is_whiteboard"]
