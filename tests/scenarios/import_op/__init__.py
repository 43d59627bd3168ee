"""An op defined in a sibling local module (reference scenario: import —
exercises local-module availability on workers)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from helper_module import imported_op  # noqa: E402

from lzy_amd import Lzy  # noqa: E402

if __name__ == "__main__":
    with Lzy().workflow("wf", interactive=False):
        print(int(imported_op(6)))
