"""Custom routing endpoint: user-supplied URL (skip Service creation) or a
sub-selector (e.g. Ray head only). Reference parity: compute/endpoint.py.

In-cluster the selector becomes the call Service's label selector
(provisioning/manifests.py:build_service_manifests); on the local driver
the selector narrows the pod list by index/role convention."""


class Endpoint:
    def __init__(self, url=None, selector=None, port=None):
        if url and selector:
            raise ValueError("Endpoint takes url OR selector, not both")
        if not url and not selector:
            raise ValueError("Endpoint needs url or selector")
        self.url = url
        self.selector = selector
        self.port = port

    def to_service_config(self):
        if self.url:
            return {"type": "url", "url": self.url}
        return {"type": "selector", "selector": dict(self.selector)}

    def select_hosts(self, hosts):
        """Local-driver routing: narrow the live pod list. Conventions:
        'pod-index': N -> that pod; 'role': 'head' -> pod 0 (RayCluster
        head). Unknown selectors route to the full set unchanged."""
        if not self.selector or not hosts:
            return hosts
        idx = self.selector.get("pod-index")
        if idx is not None:
            return [hosts[int(idx) % len(hosts)]]
        if self.selector.get("role") == "head":
            return [hosts[0]]
        return hosts

    def resolve(self, default_url=None, hosts=None):
        if self.url:
            return self.url
        if hosts:
            sel = self.select_hosts(list(hosts))
            if sel:
                return f"http://{sel[0]}"
        return default_url
