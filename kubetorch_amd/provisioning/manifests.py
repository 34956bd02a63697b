"""Kubernetes manifest builders (MI355X-first: amd.com/gpu resources, AMD
device-plugin labels, AMD accelerator anti-affinity). Reference parity:
provisioning/utils.py:418,476,542 and templates/pod_template.yaml — but
built as plain dicts, no Jinja."""
import copy

from kubetorch_amd import constants as C

AMD_ACCEL_LABELS = ("amd.com/gpu.product-name", "amd.com/gpu.device-id")

SUPPORTED_TRAINING_JOBS = ("pytorchjob", "tfjob", "mxjob", "xgboostjob")


def build_pod_spec(service_name, image, command=None, env=None, cpus=None,
                   memory=None, gpus=0, gpu_type=None, disk_size=None,
                   shared_memory="8Gi",
                   volumes=(), secrets=(), node_selector=None,
                   gpu_anti_affinity=True, port=C.SERVER_PORT,
                   inactivity_ttl=None):
    resources = {"requests": {}, "limits": {}}
    if cpus:
        resources["requests"]["cpu"] = str(cpus)
    if memory:
        resources["requests"]["memory"] = str(memory)
        resources["limits"]["memory"] = str(memory)
    if gpus:
        resources["requests"][C.GPU_RESOURCE] = str(gpus)
        resources["limits"][C.GPU_RESOURCE] = str(gpus)
    if disk_size:
        resources["requests"]["ephemeral-storage"] = str(disk_size)
        resources["limits"]["ephemeral-storage"] = str(disk_size)

    env_list = [{"name": k, "value": str(v)} for k, v in (env or {}).items()]
    env_list += [
        {"name": "POD_NAME", "valueFrom": {"fieldRef": {"fieldPath": "metadata.name"}}},
        {"name": "POD_NAMESPACE", "valueFrom": {"fieldRef": {"fieldPath": "metadata.namespace"}}},
        {"name": "POD_IP", "valueFrom": {"fieldRef": {"fieldPath": "status.podIP"}}},
    ]

    container = {
        "name": "kubetorch",
        "image": image,
        "ports": [{"name": "http", "containerPort": port}],
        "env": env_list,
        "resources": resources,
        "command": command or [
            "python", "-m", "kubetorch_amd.serving.http_server",
            "--port", str(port),
        ],
        "readinessProbe": {
            "httpGet": {"path": "/health", "port": port},
            "initialDelaySeconds": 2, "periodSeconds": 5,
        },
        "livenessProbe": {
            "httpGet": {"path": "/health", "port": port},
            "initialDelaySeconds": 20, "periodSeconds": 20,
        },
        "securityContext": {"capabilities": {"add": ["SYS_PTRACE"]}},
        "volumeMounts": [{"name": "dshm", "mountPath": "/dev/shm"}],
    }
    vols = [{"name": "dshm",
             "emptyDir": {"medium": "Memory", "sizeLimit": shared_memory}}]
    for v in volumes:
        container["volumeMounts"].append({"name": v.name, "mountPath": v.mount_path})
        vols.append({"name": v.name,
                     "persistentVolumeClaim": {"claimName": v.claim_name}})
    for s in secrets:
        if getattr(s, "as_env", True):
            container.setdefault("envFrom", []).append(
                {"secretRef": {"name": s.k8s_name}})
        else:
            container["volumeMounts"].append(
                {"name": s.k8s_name, "mountPath": s.mount_path})
            vols.append({"name": s.k8s_name, "secret": {"secretName": s.k8s_name}})

    spec = {"containers": [container], "volumes": vols}
    if node_selector or gpu_type:
        sel = dict(node_selector or {})
        if gpu_type:
            sel[C.GPU_PRODUCT_LABEL] = gpu_type
        spec["nodeSelector"] = sel
    if not gpus and gpu_anti_affinity:
        spec["affinity"] = {
            "nodeAffinity": {
                "requiredDuringSchedulingIgnoredDuringExecution": {
                    "nodeSelectorTerms": [{
                        "matchExpressions": [
                            {"key": lbl, "operator": "DoesNotExist"}
                            for lbl in AMD_ACCEL_LABELS
                        ]
                    }]
                }
            }
        }
    return spec


def _labels(service_name, username=None, module=None, version=None):
    labels = {C.SERVICE_LABEL: service_name}
    if username:
        labels[C.USERNAME_LABEL] = username
    if module:
        labels[C.MODULE_LABEL] = module
    if version:
        labels[C.VERSION_LABEL] = version
    return labels


def build_deployment_manifest(service_name, namespace, image, replicas=1,
                              username=None, module=None, annotations=None,
                              queue=None, **pod_kw):
    labels = _labels(service_name, username, module)
    if queue:
        labels[C.KUEUE_QUEUE_LABEL] = queue
    meta = {"name": service_name, "namespace": namespace, "labels": labels}
    if annotations:
        meta["annotations"] = dict(annotations)
    return {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": meta,
        "spec": {
            "replicas": replicas,
            "selector": {"matchLabels": {C.SERVICE_LABEL: service_name}},
            "template": {
                "metadata": {"labels": dict(labels)},
                "spec": build_pod_spec(service_name, image, **pod_kw),
            },
        },
    }


def build_service_manifests(service_name, namespace, port=C.SERVER_PORT,
                            selector=None):
    """ClusterIP service + headless service (peer discovery). A custom
    `selector` routes call traffic to a pod subset (e.g. the Ray head —
    reference: Endpoint(selector=...)); the headless service keeps the
    full pod set so rank discovery still sees every worker."""
    base = {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {"name": service_name, "namespace": namespace,
                     "labels": _labels(service_name)},
        "spec": {
            "selector": dict(selector or {C.SERVICE_LABEL: service_name}),
            "ports": [{"port": port, "targetPort": port}],
        },
    }
    headless = copy.deepcopy(base)
    headless["metadata"]["name"] = f"{service_name}-headless"
    headless["spec"]["selector"] = {C.SERVICE_LABEL: service_name}
    headless["spec"]["clusterIP"] = "None"
    return base, headless


def build_knative_manifest(service_name, namespace, image, autoscaling=None,
                           username=None, annotations=None, **pod_kw):
    ann = dict(annotations or {})
    if autoscaling is not None:
        ann.update(autoscaling.to_annotations())
    return {
        "apiVersion": "serving.knative.dev/v1",
        "kind": "Service",
        "metadata": {"name": service_name, "namespace": namespace,
                     "labels": _labels(service_name, username)},
        "spec": {
            "template": {
                "metadata": {"annotations": ann,
                             "labels": _labels(service_name, username)},
                "spec": build_pod_spec(service_name, image, **pod_kw),
            }
        },
    }


def build_raycluster_manifest(service_name, namespace, image, workers=0,
                              username=None, **pod_kw):
    pod = build_pod_spec(service_name, image, **pod_kw)
    return {
        "apiVersion": "ray.io/v1",
        "kind": "RayCluster",
        "metadata": {"name": service_name, "namespace": namespace,
                     "labels": _labels(service_name, username)},
        "spec": {
            "headGroupSpec": {
                "rayStartParams": {"dashboard-host": "0.0.0.0"},
                "template": {"spec": pod},
            },
            "workerGroupSpecs": [{
                "groupName": "workers",
                "replicas": workers,
                "rayStartParams": {},
                "template": {"spec": copy.deepcopy(pod)},
            }] if workers else [],
        },
    }


def build_training_job_manifest(kind, service_name, namespace, image,
                                workers=1, username=None, queue=None,
                                **pod_kw):
    """TFJob / MXJob / XGBoostJob (Kubeflow training operators); PyTorchJob
    has its own builder below. (Reference: SUPPORTED_TRAINING_JOBS.)"""
    kind_map = {"tfjob": ("TFJob", "tfReplicaSpecs"),
                "mxjob": ("MXJob", "mxReplicaSpecs"),
                "xgboostjob": ("XGBoostJob", "xgbReplicaSpecs")}
    if kind not in kind_map:
        raise ValueError(f"unsupported training job kind {kind!r}")
    k8s_kind, spec_key = kind_map[kind]
    pod = build_pod_spec(service_name, image, **pod_kw)
    labels = _labels(service_name, username)
    if queue:
        labels[C.KUEUE_QUEUE_LABEL] = queue
    return {
        "apiVersion": "kubeflow.org/v1",
        "kind": k8s_kind,
        "metadata": {"name": service_name, "namespace": namespace,
                     "labels": labels},
        "spec": {
            "runPolicy": {"suspend": bool(queue)},
            spec_key: {
                "Worker": {"replicas": workers,
                           "template": {"metadata": {"labels": dict(labels)},
                                        "spec": pod}},
            },
        },
    }


def build_pytorchjob_manifest(service_name, namespace, image, workers=1,
                              num_proc=8, username=None, queue=None, **pod_kw):
    pod = build_pod_spec(service_name, image, **pod_kw)
    labels = _labels(service_name, username)
    if queue:
        labels[C.KUEUE_QUEUE_LABEL] = queue
    return {
        "apiVersion": "kubeflow.org/v1",
        "kind": "PyTorchJob",
        "metadata": {"name": service_name, "namespace": namespace,
                     "labels": labels},
        "spec": {
            "nprocPerNode": str(num_proc),
            "runPolicy": {"suspend": bool(queue)},
            "pytorchReplicaSpecs": {
                "Master": {"replicas": 1,
                           "template": {"metadata": {"labels": dict(labels)},
                                        "spec": pod}},
                "Worker": {"replicas": max(0, workers - 1),
                           "template": {"metadata": {"labels": dict(labels)},
                                        "spec": copy.deepcopy(pod)}},
            },
        },
    }
