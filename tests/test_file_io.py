import os

import torch

from libai_amd.utils.file_io import NonBlockingWriter, PathManager


def test_path_manager_local_ops(tmp_path):
    d = str(tmp_path / "sub")
    PathManager.mkdirs(d)
    assert PathManager.isdir(d)
    f = os.path.join(d, "x.txt")
    with PathManager.open(f, "w") as fh:
        fh.write("hello")
    assert PathManager.isfile(f)
    assert PathManager.ls(d) == ["x.txt"]
    f2 = os.path.join(d, "y.txt")
    PathManager.copy(f, f2)
    with PathManager.open(f2) as fh:
        assert fh.read() == "hello"
    PathManager.rm(f2)
    assert not PathManager.exists(f2)


def test_non_blocking_writer(tmp_path):
    w = NonBlockingWriter()
    t = torch.randn(8)
    path = str(tmp_path / "t.pt")
    w.save_tensor_async(t, path)
    w.wait()
    assert torch.equal(torch.load(path, weights_only=False), t)
    w.close()


def test_http_handler_cache_and_md5(tmp_path):
    """HTTPURLHandler + cached_path against a loopback http server."""
    import functools
    import hashlib
    import http.server
    import threading

    from libai_amd.utils.file_io import PathManager, cached_path, check_md5

    serve_dir = tmp_path / "www"
    serve_dir.mkdir()
    payload = b"hello libai_amd" * 100
    (serve_dir / "data.bin").write_bytes(payload)
    md5 = hashlib.md5(payload).hexdigest()

    handler = functools.partial(http.server.SimpleHTTPRequestHandler,
                                directory=str(serve_dir))
    srv = http.server.ThreadingHTTPServer(("127.0.0.1", 0), handler)
    port = srv.server_address[1]
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        url = f"http://127.0.0.1:{port}/data.bin"
        cache = str(tmp_path / "cache")
        local = cached_path(url, cache_dir=cache, md5=md5)
        assert open(local, "rb").read() == payload
        assert check_md5(local, md5)
        # second call: cache hit (same path, no re-download)
        assert cached_path(url, cache_dir=cache, md5=md5) == local
        # corrupt the cache -> re-download restores it
        open(local, "wb").write(b"garbage")
        local2 = cached_path(url, cache_dir=cache, md5=md5)
        assert open(local2, "rb").read() == payload
        # PathManager routes the scheme
        import os
        os.environ["LIBAI_CACHE"] = cache
        try:
            with PathManager.open(url, "rb") as f:
                assert f.read() == payload
            assert PathManager.exists(url)
            assert not PathManager.exists(f"http://127.0.0.1:{port}/nope.bin")
        finally:
            del os.environ["LIBAI_CACHE"]
        # bad md5 on fresh download raises
        import pytest as _pytest
        with _pytest.raises(IOError):
            cached_path(f"http://127.0.0.1:{port}/data.bin?x=1",
                        cache_dir=cache, md5="0" * 32)
    finally:
        srv.shutdown()
