"""An op failure aborts the workflow and surfaces the original exception
to the client (reference scenario: exec_fail)."""
from lzy_amd import Lzy, op
from lzy_amd.exceptions import LzyExecutionError


@op
def broken(x: int) -> int:
    raise RuntimeError("slot machine jammed")


if __name__ == "__main__":
    try:
        with Lzy().workflow("wf", interactive=False):
            broken(1)
    except LzyExecutionError as e:
        print("workflow failed")
        print(f"cause: {type(e.__cause__).__name__}: {e.__cause__}")
