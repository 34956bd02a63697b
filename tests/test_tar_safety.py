"""safe_extractall: path-traversal / symlink-escape rejection for
client-supplied archives (data_store/server.py put_file, commands.py get)."""
import io
import os
import tarfile

import pytest

from kubetorch_amd.utils.tar import safe_extractall


def _tar_with(members):
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tar:
        for ti, data in members:
            tar.addfile(ti, io.BytesIO(data) if data is not None else None)
    buf.seek(0)
    return tarfile.open(fileobj=buf, mode="r:gz")


def _file(name, data=b"x", mode=0o644):
    ti = tarfile.TarInfo(name)
    ti.size = len(data)
    ti.mode = mode
    return ti, data


def test_normal_archive_extracts(tmp_path):
    with _tar_with([_file("a.txt"), _file("sub/b.txt")]) as tar:
        safe_extractall(tar, str(tmp_path))
    assert (tmp_path / "a.txt").read_bytes() == b"x"
    assert (tmp_path / "sub" / "b.txt").exists()


def test_dotdot_traversal_rejected(tmp_path):
    with _tar_with([_file("../evil.txt")]) as tar:
        with pytest.raises(ValueError, match="escapes"):
            safe_extractall(tar, str(tmp_path / "inner"))
    assert not (tmp_path / "evil.txt").exists()


def test_absolute_path_rejected(tmp_path):
    with _tar_with([_file("/tmp/kt_evil_abs.txt")]) as tar:
        with pytest.raises(ValueError, match="escapes"):
            safe_extractall(tar, str(tmp_path))


def test_symlink_escape_rejected(tmp_path):
    ti = tarfile.TarInfo("link")
    ti.type = tarfile.SYMTYPE
    ti.linkname = "../../outside"
    with _tar_with([(ti, None)]) as tar:
        with pytest.raises(ValueError, match="link member escapes"):
            safe_extractall(tar, str(tmp_path / "inner"))


def test_symlink_then_write_through_rejected(tmp_path):
    # classic two-member attack: symlink into parent, then write through it
    ti = tarfile.TarInfo("d")
    ti.type = tarfile.SYMTYPE
    ti.linkname = ".."
    with _tar_with([(ti, None), _file("d/pwned.txt")]) as tar:
        with pytest.raises(ValueError):
            safe_extractall(tar, str(tmp_path / "inner"))
    assert not (tmp_path / "pwned.txt").exists()


def test_device_node_rejected(tmp_path):
    ti = tarfile.TarInfo("dev")
    ti.type = tarfile.CHRTYPE
    with _tar_with([(ti, None)]) as tar:
        with pytest.raises(ValueError, match="unsupported"):
            safe_extractall(tar, str(tmp_path))


def test_setuid_bit_stripped(tmp_path):
    with _tar_with([_file("s.bin", mode=0o4755)]) as tar:
        safe_extractall(tar, str(tmp_path))
    assert (os.stat(tmp_path / "s.bin").st_mode & 0o7777) == 0o755


def test_store_path_guard_separator_aware(tmp_path, monkeypatch):
    """'/data/storeX' must not pass a '/data/store' root (prefix check
    without the separator)."""
    from kubetorch_amd.data_store import server

    monkeypatch.setattr(server, "DATA_ROOT", str(tmp_path / "store"))
    import pytest as _pytest

    assert server._path_for("ns/key").startswith(str(tmp_path / "store"))
    with _pytest.raises(ValueError):
        server._path_for("../storeX/evil")
    with _pytest.raises(ValueError):
        server._path_for("../../etc/passwd")
