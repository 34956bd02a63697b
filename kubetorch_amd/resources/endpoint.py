"""Custom routing endpoint: user-supplied URL (skip Service creation) or a
sub-selector (e.g. Ray head only). Reference parity: compute/endpoint.py."""


class Endpoint:
    def __init__(self, url=None, selector=None, port=None):
        if not url and not selector:
            raise ValueError("Endpoint needs url or selector")
        self.url = url
        self.selector = selector
        self.port = port

    def resolve(self, default_url=None):
        return self.url or default_url
