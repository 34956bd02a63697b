"""PVC-backed volumes (reference parity: resources/volumes/volume.py)."""


class Volume:
    def __init__(self, name, size="10Gi", mount_path=None, access_mode="ReadWriteOnce",
                 storage_class=None, existing_claim=None):
        self.name = name
        self.size = size
        self.mount_path = mount_path or f"/mnt/{name}"
        self.access_mode = access_mode
        self.storage_class = storage_class
        self.claim_name = existing_claim or name

    def to_pvc_manifest(self, namespace):
        spec = {
            "accessModes": [self.access_mode],
            "resources": {"requests": {"storage": self.size}},
        }
        if self.storage_class:
            spec["storageClassName"] = self.storage_class
        return {
            "apiVersion": "v1",
            "kind": "PersistentVolumeClaim",
            "metadata": {"name": self.claim_name, "namespace": namespace},
            "spec": spec,
        }
