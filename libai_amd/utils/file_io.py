"""PathManager-style file IO with async writes and a URL download cache.

Reference behavior: libai/utils/file_io.py (iopath-style PathManagerBase +
handlers, incl. the HTTP handler) and file_utils.py:281+ (URL cache with md5
verification) and non_blocking_io.py (async writer threads).  This compact
version covers the local filesystem handler, a registry for custom schemes,
the http(s) handler with an on-disk cache, and non-blocking checkpoint
writes via a background thread pool.
"""

import concurrent.futures
import hashlib
import logging
import os
import shutil
import tempfile
import urllib.request

__all__ = ["PathHandler", "NativePathHandler", "HTTPURLHandler", "PathManager",
           "NonBlockingWriter", "cached_path", "check_md5"]

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


def _default_cache_dir():
    return os.environ.get(
        "LIBAI_CACHE",
        os.path.join(os.path.expanduser("~"), ".cache", "libai_amd"),
    )


def check_md5(path, md5):
    """True iff the file's md5 matches (reference file_utils.py check)."""
    h = hashlib.md5()
    with open(path, "rb") as f:
        for chunk in iter(lambda: f.read(1 << 20), b""):
            h.update(chunk)
    return h.hexdigest() == md5


def cached_path(url, cache_dir=None, md5=None, progress=False):
    """Download ``url`` into the cache (keyed by url hash + basename) and
    return the local path; later calls hit the cache.  With ``md5``, a
    cached file failing the check is re-downloaded, and a downloaded file
    failing it raises (reference: file_utils.py:281+ URL cache w/ md5)."""
    cache_dir = cache_dir or _default_cache_dir()
    os.makedirs(cache_dir, exist_ok=True)
    key = hashlib.sha256(url.encode()).hexdigest()[:16]
    base = os.path.basename(url.split("?")[0]) or "download"
    local = os.path.join(cache_dir, f"{key}_{base}")
    if os.path.exists(local):
        if md5 is None or check_md5(local, md5):
            return local
        logger.warning(f"cached {local} fails md5; re-downloading")
        os.remove(local)
    logger.info(f"downloading {url} -> {local}")
    fd, tmp = tempfile.mkstemp(dir=cache_dir)
    os.close(fd)
    try:
        with urllib.request.urlopen(url) as r, open(tmp, "wb") as f:
            shutil.copyfileobj(r, f)
        if md5 is not None and not check_md5(tmp, md5):
            raise IOError(f"md5 mismatch for downloaded {url}")
        os.replace(tmp, local)
    finally:
        if os.path.exists(tmp):
            os.remove(tmp)
    return local


class HTTPURLHandler(PathHandler):
    """http(s):// paths: downloads go through the URL cache; ``open`` opens
    the cached local file (reference libai/utils/file_io.py HTTP handler)."""

    def __init__(self, cache_dir=None):
        self.cache_dir = cache_dir

    def get_supported_prefixes(self):
        return ["http://", "https://"]

    def get_local_path(self, path, md5=None, **kwargs):
        return cached_path(path, cache_dir=self.cache_dir, md5=md5)

    def open(self, path, mode="r", **kwargs):
        assert "w" not in mode and "a" not in mode, "http paths are read-only"
        return open(self.get_local_path(path), mode, **kwargs)

    def exists(self, path):
        try:
            req = urllib.request.Request(path, method="HEAD")
            with urllib.request.urlopen(req, timeout=10) as r:
                return r.status < 400
        except Exception:  # noqa: BLE001
            return False

    def isfile(self, path):
        return self.exists(path)

    def isdir(self, path):
        return False


class _PathManager:
    def __init__(self):
        self._native = NativePathHandler()
        self._handlers = {}
        self.register_handler(HTTPURLHandler())

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
