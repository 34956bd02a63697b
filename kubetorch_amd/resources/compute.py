"""Compute: the resource spec a Module deploys onto.

MI355X-first: `gpus=` maps to amd.com/gpu, `gpu_type=` to the AMD device
plugin's product label, worker image defaults to PyTorch-ROCm. (Reference
parity: resources/compute/compute.py — 2798 LoC there; the manifest
mechanics live in provisioning/manifests.py here.)"""
import copy

from kubetorch_amd import constants as C
from kubetorch_amd.resources.autoscaling import AutoscalingConfig
from kubetorch_amd.resources.images import DEFAULT_PYTORCH_ROCM
from kubetorch_amd.config import config
from kubetorch_amd.provisioning import manifests as M


class Compute:
    def __init__(self, cpus=None, memory=None, gpus=0, gpu_type=None,
                 gpu_memory=None, disk_size=None,
                 image=None, env=None, namespace=None, volumes=None,
                 secrets=None, shared_memory="8Gi", node_selector=None,
                 annotations=None, inactivity_ttl=None, queue=None,
                 launch_timeout=C.LAUNCH_TIMEOUT, endpoint=None,
                 local=None, allowed_serialization=None):
        self.cpus = cpus
        self.memory = memory
        self.gpus = gpus
        self.gpu_type = gpu_type
        # fractional-GPU memory cap: a whole amd.com/gpu is still requested,
        # the annotation caps usable HBM (reference: gpu_memory annotation)
        self.gpu_memory = gpu_memory
        if gpu_memory and not gpus:
            self.gpus = 1
        self.disk_size = disk_size
        self.image = image  # Image object or image id string
        self.env = dict(env or {})
        self.namespace = namespace or config.namespace
        self.volumes = list(volumes or [])
        self.secrets = list(secrets or [])
        self.shared_memory = shared_memory
        self.node_selector = node_selector
        self.annotations = dict(annotations or {})
        self.inactivity_ttl = inactivity_ttl
        self.queue = queue
        self.launch_timeout = launch_timeout
        self.endpoint = endpoint
        # serialization formats the pod will accept for /call bodies.
        # Default json (reference parity: KT_ALLOWED_SERIALIZATION); pass
        # ["json", "pickle"] to allow pickled args from trusted clients.
        self.allowed_serialization = list(allowed_serialization or ["json"])
        self.local = config.local_mode if local is None else local
        self.kind = "deployment"
        self.replicas = 1
        self.distributed_config = None
        self.autoscaling = None
        self._raw_manifest = None

    # -- fluent configuration --------------------------------------------------
    def distribute(self, framework="pytorch", workers=1, num_proc=None,
                   quorum_timeout=C.QUORUM_TIMEOUT):
        """Configure the SPMD launcher: `workers` pods x `num_proc` local
        ranks (auto = one per MI355X GPU via torch.cuda.device_count())."""
        if self.autoscaling is not None:
            raise ValueError(
                "distribute() and autoscale() are mutually exclusive: SPMD "
                "ranks need a stable worker set")
        if framework == "ray":
            self.kind = "raycluster"
            self.replicas = workers
            self.distributed_config = {"type": "ray", "workers": workers}
            return self
        self.replicas = workers
        self.distributed_config = {
            "type": framework,
            "workers": workers,
            "num_proc": num_proc,
            "quorum_timeout": quorum_timeout,
        }
        return self

    def autoscale(self, autoscaling=None, **kw):
        if self.distributed_config is not None:
            raise ValueError(
                "autoscale() and distribute() are mutually exclusive: SPMD "
                "ranks need a stable worker set (scale jobs by re-deploying "
                "with a different workers=)")
        self.kind = "knative"
        self.autoscaling = autoscaling or AutoscalingConfig(**kw)
        return self

    @classmethod
    def from_manifest(cls, manifest, namespace=None, **kw):
        """BYO compute: a prebuilt manifest (Deployment / PyTorchJob / ...)."""
        comp = cls(namespace=namespace or manifest.get("metadata", {})
                   .get("namespace", "default"), **kw)
        comp._raw_manifest = copy.deepcopy(manifest)
        kind = manifest.get("kind", "Deployment").lower()
        comp.kind = kind
        if kind == "pytorchjob":
            specs = manifest.get("spec", {}).get("pytorchReplicaSpecs", {})
            workers = sum(s.get("replicas", 0) for s in specs.values())
            if workers > 1:
                comp.distributed_config = {
                    "type": "pytorch", "workers": workers,
                    "num_proc": int(manifest["spec"].get("nprocPerNode", 1) or 1),
                }
        return comp

    # -- manifest --------------------------------------------------------------
    def image_id(self):
        if self.image is None:
            return config.get("image") or DEFAULT_PYTORCH_ROCM
        return getattr(self.image, "image_id", None) or str(self.image)

    def pod_env(self):
        env = dict(self.env)
        if self.inactivity_ttl:
            env["KT_INACTIVITY_TTL"] = str(self.inactivity_ttl)
        return env

    def to_manifest(self, service_name, username=None, module=None):
        if self._raw_manifest is not None:
            m = copy.deepcopy(self._raw_manifest)
            # the service owns the name: the driver registers pods under
            # metadata.name, and the client queries by service name — a
            # BYO manifest keeping its own name would orphan its pods
            m.setdefault("metadata", {})["name"] = service_name
            return m
        pod_kw = dict(
            env=self.pod_env(), cpus=self.cpus, memory=self.memory,
            gpus=self.gpus, gpu_type=self.gpu_type,
            disk_size=self.disk_size,
            shared_memory=self.shared_memory, volumes=self.volumes,
            secrets=self.secrets, node_selector=self.node_selector,
        )
        ann = dict(self.annotations)
        if self.inactivity_ttl:
            ann[C.INACTIVITY_TTL_ANNOTATION] = str(self.inactivity_ttl)
        if self.gpu_memory:
            ann["gpu-memory"] = str(self.gpu_memory)
        if self.kind == "knative":
            return M.build_knative_manifest(
                service_name, self.namespace, self.image_id(),
                autoscaling=self.autoscaling, username=username,
                annotations=ann, **pod_kw)
        if self.kind == "raycluster":
            return M.build_raycluster_manifest(
                service_name, self.namespace, self.image_id(),
                workers=max(0, self.replicas - 1), username=username, **pod_kw)
        if self.kind == "pytorchjob":
            dc = self.distributed_config or {}
            return M.build_pytorchjob_manifest(
                service_name, self.namespace, self.image_id(),
                workers=dc.get("workers", 1),
                num_proc=dc.get("num_proc") or 1,
                username=username, queue=self.queue, **pod_kw)
        return M.build_deployment_manifest(
            service_name, self.namespace, self.image_id(),
            replicas=self.replicas, username=username, module=module,
            annotations=ann, queue=self.queue, **pod_kw)

    def image_setup_contents(self):
        img = self.image
        if img is not None and hasattr(img, "contents"):
            return img.contents()
        return ""

    # -- getter/setter surface (reference: compute.py:934-956 — setters
    # -- also mutate a BYO manifest in place so from_manifest computes stay
    # -- editable without rebuilding) ---------------------------------------
    def _pod_spec_of_raw(self):
        m = self._raw_manifest
        if m is None:
            return None
        spec = m.get("spec", {})
        if "template" in spec:          # Deployment
            return spec["template"].get("spec")
        if "pytorchReplicaSpecs" in spec:  # PyTorchJob: first replica spec
            for rs in spec["pytorchReplicaSpecs"].values():
                return rs.get("template", {}).get("spec")
        return None

    def _patch_resource(self, key, value, request_only=False):
        pod = self._pod_spec_of_raw()
        if not pod:
            return
        for c in pod.get("containers", []):
            res = c.setdefault("resources", {})
            res.setdefault("requests", {})[key] = str(value)
            if not request_only:
                res.setdefault("limits", {})[key] = str(value)

    def set_gpus(self, gpus):
        self.gpus = gpus
        self._patch_resource(C.GPU_RESOURCE, gpus)
        return self

    def set_cpus(self, cpus):
        self.cpus = cpus
        self._patch_resource("cpu", cpus, request_only=True)
        return self

    def set_memory(self, memory):
        self.memory = memory
        self._patch_resource("memory", memory)
        return self

    def set_gpu_type(self, gpu_type):
        self.gpu_type = gpu_type
        pod = self._pod_spec_of_raw()
        if pod is not None:
            pod.setdefault("nodeSelector", {})[C.GPU_PRODUCT_LABEL] = gpu_type
        return self

    def set_image(self, image):
        self.image = image
        pod = self._pod_spec_of_raw()
        if pod is not None:
            for c in pod.get("containers", []):
                c["image"] = self.image_id()
        return self

    def set_replicas(self, replicas):
        self.replicas = replicas
        if self._raw_manifest is not None and \
                "replicas" in self._raw_manifest.get("spec", {}):
            self._raw_manifest["spec"]["replicas"] = replicas
        return self
