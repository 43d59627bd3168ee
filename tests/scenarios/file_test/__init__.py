"""File content travels through op boundaries (reference scenario:
file_test)."""
import os

from lzy_amd import File, Lzy, op


@op
def annotate(f: File) -> File:
    out = File.create_tmp(suffix=".txt")
    out.write_text(f.read_text() + " world")
    return out


if __name__ == "__main__":
    src = File.create_tmp(suffix=".txt")
    src.write_text("hello")
    with Lzy().workflow("wf", interactive=False):
        result = annotate(src)
        print(result.read_text())
    os.unlink(src.path)
