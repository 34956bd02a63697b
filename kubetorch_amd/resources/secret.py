"""Secrets: values or provider presets materialized as K8s Secrets, exposed
as env vars or file mounts. (Reference parity: resources/secrets/.)"""
import os


class Secret:
    def __init__(self, name, values=None, provider=None, as_env=True,
                 mount_path=None):
        self.name = name
        self.provider = provider
        self.as_env = as_env
        self.mount_path = mount_path or f"/secrets/{name}"
        self.values = dict(values or {})
        if provider and not self.values:
            self.values = _provider_values(provider)

    @property
    def k8s_name(self):
        return f"kt-secret-{self.name}".lower().replace("_", "-")

    def to_manifest(self, namespace):
        import base64

        return {
            "apiVersion": "v1",
            "kind": "Secret",
            "metadata": {"name": self.k8s_name, "namespace": namespace},
            "type": "Opaque",
            "data": {k: base64.b64encode(str(v).encode()).decode()
                     for k, v in self.values.items()},
        }


# provider presets: env vars (and optional credential files) per provider
PROVIDERS = {
    "anthropic": ["ANTHROPIC_API_KEY"],
    "openai": ["OPENAI_API_KEY"],
    "huggingface": ["HF_TOKEN", "HUGGING_FACE_HUB_TOKEN"],
    "wandb": ["WANDB_API_KEY"],
    "aws": ["AWS_ACCESS_KEY_ID", "AWS_SECRET_ACCESS_KEY", "AWS_SESSION_TOKEN"],
    "gcp": ["GOOGLE_APPLICATION_CREDENTIALS"],
    "azure": ["AZURE_CLIENT_ID", "AZURE_CLIENT_SECRET", "AZURE_TENANT_ID"],
    "github": ["GITHUB_TOKEN"],
    "cohere": ["COHERE_API_KEY"],
    "pinecone": ["PINECONE_API_KEY"],
    "langchain": ["LANGCHAIN_API_KEY"],
    "lambda": ["LAMBDA_API_KEY"],
    "kubeconfig": ["KUBECONFIG"],
    "ssh": ["SSH_PRIVATE_KEY"],
}


def _provider_values(provider):
    keys = PROVIDERS.get(provider)
    if keys is None:
        raise ValueError(f"unknown secret provider {provider!r}; "
                         f"known: {sorted(PROVIDERS)}")
    return {k: os.environ[k] for k in keys if k in os.environ}


def secret_factory(provider, name=None, **kw):
    return Secret(name or provider, provider=provider, **kw)
