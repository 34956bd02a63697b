"""Production deploy surface in one example: image steps executed in the
running pod (no rebuild), provider secrets as env + file mounts, a PVC
cache volume, and per-deploy config. Everything runs on the local driver
with KT_LOCAL_MODE=true; on Kubernetes the same code materializes real
Secrets/PVCs and amd.com/gpu requests."""
import os

import kubetorch_amd as kt


def report():
    """Runs inside the pod: prove each production feature materialized."""
    cache_dir = os.environ.get("KT_VOLUME_MOUNT_MODEL_CACHE")
    marker = os.path.join(cache_dir, "warmed") if cache_dir else None
    first_boot = marker is not None and not os.path.exists(marker)
    if marker and first_boot:
        with open(marker, "w") as f:
            f.write("1")  # survives pod restarts on the PVC
    return {
        "image_env": os.environ.get("DEPLOY_STAGE"),
        "hf_token_env": os.environ.get("HF_TOKEN"),
        "cache_mounted": cache_dir is not None,
        "cache_cold": first_boot,
    }


if __name__ == "__main__":
    hf = kt.Secret("hf", values={"HF_TOKEN": "hf_example"})
    cache = kt.Volume("model-cache", size="10Gi")

    image = (kt.images.pytorch()            # rocm/pytorch base
             .set_env_vars({"DEPLOY_STAGE": "prod"})
             .run_bash("echo image-step-ran"))

    remote = kt.fn(report).to(kt.Compute(
        cpus=1, image=image, secrets=[hf], volumes=[cache],
        inactivity_ttl="30m"))
    try:
        out = remote()
        print(f"deployed: {out}")
        assert out["image_env"] == "prod"
        assert out["hf_token_env"] == "hf_example"
        assert out["cache_mounted"]
    finally:
        remote.teardown()
