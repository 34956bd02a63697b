"""PVC-backed volumes (reference parity: resources/volumes/volume.py).

At deploy time the controller creates the PVC (unless it binds an
existing claim); `existing_pv=` pins the claim to a pre-provisioned
PersistentVolume by name (the reference's existing-PV bind). On the
local driver a host directory stands in for the mount, exposed to the
pod as KT_VOLUME_MOUNT_<NAME>."""


class Volume:
    def __init__(self, name, size="10Gi", mount_path=None, access_mode="ReadWriteOnce",
                 storage_class=None, existing_claim=None, existing_pv=None):
        self.name = name
        self.size = size
        self.mount_path = mount_path or f"/mnt/{name}"
        self.access_mode = access_mode
        self.storage_class = storage_class
        self.existing_claim = existing_claim
        self.existing_pv = existing_pv
        self.claim_name = existing_claim or name

    @property
    def needs_create(self):
        return self.existing_claim is None

    def to_pvc_manifest(self, namespace):
        spec = {
            "accessModes": [self.access_mode],
            "resources": {"requests": {"storage": self.size}},
        }
        if self.storage_class:
            spec["storageClassName"] = self.storage_class
        if self.existing_pv:
            # bind to a pre-provisioned PV: pin volumeName and disable
            # dynamic provisioning so the claim can only match that PV
            spec["volumeName"] = self.existing_pv
            spec.setdefault("storageClassName", "")
        return {
            "apiVersion": "v1",
            "kind": "PersistentVolumeClaim",
            "metadata": {"name": self.claim_name, "namespace": namespace},
            "spec": spec,
        }
