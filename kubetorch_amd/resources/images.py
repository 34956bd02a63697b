"""Stock images — ROCm-first (reference ships nvcr.io pytorch images;
here the default worker is PyTorch-ROCm gfx950). Reference parity:
resources/images/images.py."""
from kubetorch_amd.resources.image import Image

DEFAULT_PYTORCH_ROCM = "rocm/pytorch:latest"


def pytorch(image_id=DEFAULT_PYTORCH_ROCM):
    """PyTorch-ROCm worker image for MI355X (gfx950)."""
    return Image(image_id=image_id, name="pytorch-rocm")


def python(version="3.10"):
    return Image(image_id=f"python:{version}-slim", name=f"python{version}")


def debian():
    return Image(image_id="debian:bookworm-slim", name="debian")


def ubuntu():
    return Image(image_id="ubuntu:24.04", name="ubuntu")


def ray(image_id="rayproject/ray:latest"):
    return Image(image_id=image_id, name="ray")
