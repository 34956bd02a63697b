"""Safe tar extraction for client-supplied archives.

Python 3.12's `extractall(filter="data")` is not available on 3.10, so the
same guarantees are enforced manually: no absolute paths, no `..` escapes,
no symlinks/hardlinks pointing outside the destination, no device nodes.
"""
import os
import tarfile


def _resolves_inside(dest_root, relpath):
    target = os.path.realpath(os.path.join(dest_root, relpath))
    root = os.path.realpath(dest_root)
    return target == root or target.startswith(root + os.sep)


def safe_extractall(tar: tarfile.TarFile, dest: str):
    """Extract `tar` into `dest`, rejecting members that would write or link
    outside `dest` (path traversal / symlink escape from an untrusted peer).
    """
    dest = os.path.abspath(dest)
    members = []
    for m in tar.getmembers():
        # normalize "./a", "a/./b" and collapse inner "a/../b"; anything that
        # still *leads* with ".." after normpath points above dest → reject.
        # (names with dot segments also break tarfile mechanically even when
        # they resolve inside: "0/../y" hits makedirs("0/..") EEXIST)
        name = m.name = os.path.normpath(m.name)
        if os.path.isabs(name) or name == ".." or name.startswith("../") \
                or not _resolves_inside(dest, name):
            raise ValueError(f"tar member escapes destination: {name!r}")
        if m.isfile() and os.path.realpath(
                os.path.join(dest, name)) == os.path.realpath(dest):
            # a REGULAR file named "." would try to open the dest dir itself
            raise ValueError(f"tar file member resolves to destination root: "
                             f"{name!r}")
        if m.issym() or m.islnk():
            link = m.linkname
            base = os.path.dirname(name)
            if os.path.isabs(link) or not _resolves_inside(
                    dest, os.path.join(base, link)):
                raise ValueError(
                    f"tar link member escapes destination: {name!r} -> {link!r}")
        elif not (m.isfile() or m.isdir()):
            # devices/FIFOs have no business in a code/file sync archive
            raise ValueError(f"unsupported tar member type: {name!r}")
        # strip setuid/setgid/sticky; keep rwx bits only
        m.mode &= 0o777
        members.append(m)
    tar.extractall(dest, members=members)  # noqa: S202 - members vetted above
