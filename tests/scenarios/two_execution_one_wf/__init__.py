"""Two executions under one workflow name share the cache namespace but
get distinct execution ids (reference scenario: two_execution_one_wf)."""
from lzy_amd import Lzy, op


@op
def ident(x: int) -> int:
    return x


if __name__ == "__main__":
    lzy = Lzy()
    ids = []
    for i in range(2):
        with lzy.workflow("same_name", interactive=False) as wf:
            print(int(ident(i)))
            ids.append(wf.execution_id)
    print(ids[0] != ids[1])
