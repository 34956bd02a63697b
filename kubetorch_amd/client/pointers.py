"""Extract import pointers from a live fn/cls and locate the project root.

A callable ships as (project_root, rel_path, name) — code syncs via the
data store, never pickled. (Reference parity: resources/callables/utils.py:
extract_pointers :53, locate_working_dir :114.)"""
import inspect
import os

PROJECT_MARKERS = (".git", "pyproject.toml", "setup.py", "requirements.txt",
                   ".kt_root")


def locate_working_dir(start_path):
    """Walk up from start_path to the nearest project marker; fall back to
    the starting directory."""
    d = os.path.abspath(start_path)
    if os.path.isfile(d):
        d = os.path.dirname(d)
    cur = d
    while True:
        if any(os.path.exists(os.path.join(cur, m)) for m in PROJECT_MARKERS):
            return cur
        parent = os.path.dirname(cur)
        if parent == cur:
            return d
        cur = parent


def extract_pointers(obj):
    """-> dict(project_root, file_path, rel_path, name). Works for functions
    and classes defined in real files (notebook cells are dumped first)."""
    name = obj.__name__
    try:
        file_path = os.path.abspath(inspect.getfile(obj))
        if not os.path.exists(file_path):
            raise TypeError("source file missing")
        if obj.__module__ in ("__main__",) and file_path.endswith(
                ("<stdin>", "<ipython-input>")):
            raise TypeError("interactive source")
    except (TypeError, OSError):
        # notebook / REPL: dump the cell source to a real file
        # (reference: resources/callables/utils.py notebook support).
        # inspect raises TypeError for builtins and OSError when no source
        # is retrievable (python -c / piped stdin) — the latter cannot be
        # recovered, so say what to do instead of a raw OSError.
        try:
            src = inspect.getsource(obj)
        except OSError as e:
            raise ValueError(
                f"cannot extract source for {name!r}: it was defined in an "
                "interactive/stdin context with no retrievable source. "
                "Define it in a .py file (or a notebook cell) and retry."
            ) from e
        nb_dir = os.path.join(os.getcwd(), ".kt_notebook")
        os.makedirs(nb_dir, exist_ok=True)
        file_path = os.path.join(nb_dir, f"{name}.py")
        with open(file_path, "w") as f:
            f.write(src)
    if file_path.endswith((".pyc", ".pyo")):
        file_path = file_path[:-1]
    root = locate_working_dir(file_path)
    rel = os.path.relpath(file_path, root)
    return {
        "project_root": root,
        "file_path": file_path,
        "rel_path": rel,
        "name": name,
    }
