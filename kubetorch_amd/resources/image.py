"""Image: dockerfile-as-data executed *inside a running pod* by the server's
image-setup interpreter (no image rebuild — the 1-3 s hot loop). Reference
parity: resources/images/image.py."""
import shlex


class Image:
    def __init__(self, image_id=None, name=None):
        self.image_id = image_id
        self.name = name
        self.steps = []  # ordered (kind, payload)

    # -- builders -------------------------------------------------------------
    def pip_install(self, packages, extra_args=""):
        if isinstance(packages, str):
            packages = [packages]
        pkgs = " ".join(shlex.quote(p) for p in packages)
        self.steps.append(("RUN", f"$KT_PIP_INSTALL_CMD {pkgs} {extra_args}".strip()))
        return self

    def run_bash(self, command):
        self.steps.append(("RUN", command))
        return self

    def set_env_vars(self, env: dict):
        for k, v in env.items():
            self.steps.append(("ENV", f"{k}={v}"))
        return self

    def copy(self, src, dest):
        self.steps.append(("COPY", f"{src} {dest}"))
        return self

    def sync_package(self, package_path):
        self.steps.append(("SYNC", package_path))
        return self

    def cmd(self, command):
        self.steps.append(("CMD", command))
        return self

    # -- serialization --------------------------------------------------------
    def contents(self):
        lines = []
        if self.image_id:
            lines.append(f"FROM {self.image_id}")
        for kind, payload in self.steps:
            lines.append(f"{kind} {payload}")
        return "\n".join(lines)

    @classmethod
    def from_dockerfile(cls, text_or_path):
        import os

        text = text_or_path
        if os.path.exists(text_or_path):
            with open(text_or_path) as f:
                text = f.read()
        img = cls()
        for raw in text.splitlines():
            line = raw.strip()
            if not line or line.startswith("#"):
                continue
            kind, _, payload = line.partition(" ")
            kind = kind.upper()
            if kind == "FROM":
                img.image_id = payload.strip()
            elif kind in ("RUN", "ENV", "COPY", "CMD", "SYNC"):
                img.steps.append((kind, payload.strip()))
        return img

    def __repr__(self):
        return f"Image({self.image_id!r}, steps={len(self.steps)})"
