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
