"""Compatibility namespace: the reference's public import path is
``lzy.api.v1`` — users switching from pylzy keep their import shape
(``from lzy_amd.api.v1 import op, Lzy, ...``)."""
