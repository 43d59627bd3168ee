"""An op pinned to a docker image (reference scenario: user_image_cpu —
the op runs inside the user's image).  In-process MI355X semantics: the
container request is validated and WARNED about (there is no container
runtime in the data plane), and the op still executes on the node —
source-compatible, with the difference surfaced loudly."""
import warnings

from lzy_amd.api.v1 import DockerPullPolicy, Lzy, docker_container, op


@docker_container(registry="registry.example.com", image="user/image:1.0",
                  pull_policy=DockerPullPolicy.IF_NOT_EXISTS)
@op
def containered(x: int) -> int:
    return x * 7


if __name__ == "__main__":
    with warnings.catch_warnings(record=True) as caught:
        warnings.simplefilter("always")

        @docker_container(registry="registry.example.com", image="u/i:2")
        @op
        def again(x: int) -> int:
            return x + 1

    assert any("ignored" in str(w.message) for w in caught)
    print("container request warned")
    with Lzy().workflow("wf", interactive=False):
        y = containered(6)
        print(f"result={int(y)}")
