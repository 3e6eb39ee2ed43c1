"""PathManager-style file IO with async writes.

Reference behavior: libai/utils/file_io.py (iopath-style PathManagerBase +
handlers) and non_blocking_io.py (async writer threads).  This compact
version covers the local filesystem handler, a registry for custom schemes,
and non-blocking checkpoint writes via a background thread pool.
"""

import concurrent.futures
import logging
import os
import shutil

__all__ = ["PathHandler", "NativePathHandler", "PathManager", "NonBlockingWriter"]

logger = logging.getLogger(__name__)


class PathHandler:
    def get_supported_prefixes(self):
        raise NotImplementedError

    def get_local_path(self, path, **kwargs):
        raise NotImplementedError

    def open(self, path, mode="r", **kwargs):
        raise NotImplementedError

    def exists(self, path):
        raise NotImplementedError

    def isfile(self, path):
        raise NotImplementedError

    def isdir(self, path):
        raise NotImplementedError

    def ls(self, path):
        raise NotImplementedError

    def mkdirs(self, path):
        raise NotImplementedError

    def rm(self, path):
        raise NotImplementedError

    def copy(self, src, dst, **kwargs):
        raise NotImplementedError


class NativePathHandler(PathHandler):
    def get_supported_prefixes(self):
        return [""]

    def get_local_path(self, path, **kwargs):
        return path

    def open(self, path, mode="r", **kwargs):
        return open(path, mode, **kwargs)

    def exists(self, path):
        return os.path.exists(path)

    def isfile(self, path):
        return os.path.isfile(path)

    def isdir(self, path):
        return os.path.isdir(path)

    def ls(self, path):
        return sorted(os.listdir(path))

    def mkdirs(self, path):
        os.makedirs(path, exist_ok=True)

    def rm(self, path):
        if os.path.isdir(path):
            shutil.rmtree(path)
        elif os.path.exists(path):
            os.remove(path)

    def copy(self, src, dst, **kwargs):
        shutil.copyfile(src, dst)
        return True


class _PathManager:
    def __init__(self):
        self._native = NativePathHandler()
        self._handlers = {}

    def register_handler(self, handler):
        for prefix in handler.get_supported_prefixes():
            if prefix:
                self._handlers[prefix] = handler

    def _get(self, path):
        for prefix, h in self._handlers.items():
            if path.startswith(prefix):
                return h
        return self._native

    def get_local_path(self, path, **kw):
        return self._get(path).get_local_path(path, **kw)

    def open(self, path, mode="r", **kw):
        return self._get(path).open(path, mode, **kw)

    def exists(self, path):
        return self._get(path).exists(path)

    def isfile(self, path):
        return self._get(path).isfile(path)

    def isdir(self, path):
        return self._get(path).isdir(path)

    def ls(self, path):
        return self._get(path).ls(path)

    def mkdirs(self, path):
        return self._get(path).mkdirs(path)

    def rm(self, path):
        return self._get(path).rm(path)

    def copy(self, src, dst, **kw):
        return self._get(src).copy(src, dst, **kw)


PathManager = _PathManager()


class NonBlockingWriter:
    """Background-thread writes so checkpoint saves don't stall the step loop
    (reference: libai/utils/non_blocking_io.py)."""

    def __init__(self, max_workers=2):
        self._pool = concurrent.futures.ThreadPoolExecutor(max_workers=max_workers)
        self._futures = []

    def submit(self, fn, *args, **kwargs):
        fut = self._pool.submit(fn, *args, **kwargs)
        self._futures.append(fut)
        return fut

    def save_tensor_async(self, obj, path):
        import torch

        return self.submit(torch.save, obj, path)

    def wait(self):
        for f in self._futures:
            exc = f.exception()
            if exc is not None:
                raise exc
        self._futures.clear()

    def close(self):
        self.wait()
        self._pool.shutdown()
