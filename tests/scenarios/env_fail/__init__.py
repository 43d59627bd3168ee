"""Unsatisfiable provisioning fails the graph up-front with a clear error
(reference scenario: env_fail — broken env aborts before execution)."""
from lzy_amd import Lzy, op
from lzy_amd.exceptions import BadProvisioningError


@op(gpu_count=4096)
def impossible(x: int) -> int:
    return x


if __name__ == "__main__":
    try:
        with Lzy().workflow("wf", interactive=False):
            impossible(1)
    except BadProvisioningError:
        print("provisioning rejected")
