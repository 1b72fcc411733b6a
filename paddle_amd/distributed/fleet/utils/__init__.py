"""fleet.utils (reference: python/paddle/distributed/fleet/utils/__init__.py
-- recompute + sequence_parallel_utils re-exports)."""
from ..recompute import recompute, recompute_sequential  # noqa: F401
from .. import sequence_parallel as sequence_parallel_utils  # noqa: F401


class LocalFS:
    """Local filesystem shim (reference: fleet/utils/fs.py:LocalFS)."""

    def ls_dir(self, path):
        import os
        dirs, files = [], []
        for e in os.listdir(path):
            (dirs if os.path.isdir(os.path.join(path, e)) else files).append(e)
        return dirs, files

    def is_exist(self, path):
        import os
        return os.path.exists(path)

    def mkdirs(self, path):
        import os
        os.makedirs(path, exist_ok=True)

    def delete(self, path):
        import os
        import shutil
        if os.path.isdir(path):
            shutil.rmtree(path)
        elif os.path.exists(path):
            os.remove(path)

    def touch(self, path, exist_ok=True):
        open(path, "a").close()

    def mv(self, src, dst, overwrite=False):
        import shutil
        shutil.move(src, dst)


class HDFSClient:
    def __init__(self, *a, **kw):
        raise NotImplementedError("HDFS access requires a hadoop client "
                                  "(not in this image); use LocalFS")
