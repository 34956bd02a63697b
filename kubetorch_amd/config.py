"""Layered client config: env (KT_*) > file (~/.ktamd/config.yaml) > defaults.

Reference parity: python_client/kubetorch/config.py:28-383 (KubetorchConfig).
"""
import getpass
import os

import yaml

CONFIG_PATH = os.path.expanduser(os.environ.get("KT_CONFIG_PATH", "~/.ktamd/config.yaml"))

_DEFAULTS = {
    "username": None,          # prefixes service names
    "namespace": "default",
    "install_namespace": "kubetorch",
    "api_url": None,           # controller URL; None -> discover/port-forward
    "stream_logs": True,
    "stream_metrics": False,
    "launch_timeout": 900,
    "image": None,             # default worker image
    "local_mode": False,       # local-process driver instead of k8s
    "workdir": None,
}

_ENV_MAP = {k: "KT_" + k.upper() for k in _DEFAULTS}


class KTConfig:
    def __init__(self, path=CONFIG_PATH):
        self._path = path
        self._file = {}
        self._cluster = None
        if os.path.exists(path):
            try:
                with open(path) as f:
                    self._file = yaml.safe_load(f) or {}
            except Exception:
                self._file = {}

    def get(self, key, default=None):
        env = _ENV_MAP.get(key, "KT_" + key.upper())
        if env in os.environ:
            val = os.environ[env]
            if val.lower() in ("true", "false"):
                return val.lower() == "true"
            return val
        if key in self._file:
            return self._file[key]
        cluster = self.cluster_config()
        if key in cluster:
            return cluster[key]
        return _DEFAULTS.get(key, default)

    def cluster_config(self):
        """Cluster-wide defaults published by the controller (the chart's
        kubetorch-config ConfigMap; reference: service_manager.py:803
        fetching templates/configmaps/kubetorch-config.yaml). Layered
        BELOW env and the user's file, fetched once per process; never
        fetched in local mode or before an api_url is known (no recursive
        discovery)."""
        if self._cluster is not None:
            return self._cluster
        self._cluster = {}
        api = (os.environ.get("KT_API_URL") or self._file.get("api_url"))
        local = os.environ.get("KT_LOCAL_MODE", "").lower() == "true" or \
            self._file.get("local_mode")
        if api and not local:
            try:
                import httpx

                r = httpx.get(api.rstrip("/") + "/controller/config",
                              timeout=5)
                if r.status_code == 200:
                    got = r.json().get("config", {})
                    if isinstance(got, dict):
                        self._cluster = {k: v for k, v in got.items()
                                         if k in _DEFAULTS}
            except Exception:
                pass
        return self._cluster

    def set(self, key, value, persist=False):
        self._file[key] = value
        if persist:
            os.makedirs(os.path.dirname(self._path), exist_ok=True)
            with open(self._path, "w") as f:
                yaml.safe_dump(self._file, f)

    @property
    def username(self):
        u = self.get("username")
        if u:
            return u
        try:
            return getpass.getuser()
        except Exception:
            return "user"

    @property
    def namespace(self):
        return self.get("namespace")

    @property
    def api_url(self):
        return self.get("api_url")

    @property
    def local_mode(self):
        return bool(self.get("local_mode"))

    def as_dict(self):
        return {k: self.get(k) for k in _DEFAULTS}


config = KTConfig()
