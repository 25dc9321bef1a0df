"""Model packaging: Model / ModelVersion registry.

Reference: the Model/ModelVersion CRDs + ModelVersion controller
(apis/model/v1alpha1/, controllers/model/modelversion_controller.go):
on job success a ModelVersion is created from the job's output storage,
a Kaniko pod bakes `FROM busybox; COPY build/ /torch-on-k8s-model` and
pushes an image, and Model.LatestVersion advances.

Node-native redesign: there is no registry daemon on the box, so an
"image" is an OCI-layout-shaped local artifact: the checkpoint directory
is packed into a content-addressed tar.gz layer with a manifest.json
recording the model path (TORCH_ON_K8S_MODEL_PATH parity,
model/v1alpha1/constants.go:24-27), and Model.latest_version advances.
Storage providers mirror the reference's LocalStorage/NFS split
(pkg/storage/): both are directory roots with provenance metadata.
"""
from __future__ import annotations

import hashlib
import json
import os
import tarfile
import time
from dataclasses import dataclass, field

MODEL_PATH_ENV = "TORCH_ON_K8S_MODEL_PATH"
MODEL_IMAGE_PATH = "/torch-on-k8s-model"  # path baked into the artifact


@dataclass
class ModelVersion:
    model: str
    version: str
    image_ref: str          # local artifact path (registry analog)
    digest: str = ""
    build_phase: str = "Created"   # Created|Building|Succeeded|Failed
    source_job: str | None = None
    ts: float = field(default_factory=time.time)


@dataclass
class Model:
    name: str
    description: str = ""
    latest_version: str | None = None
    versions: dict = field(default_factory=dict)


class StorageProvider:
    """Reference pkg/storage interface analog: yields the model output
    dir a job's tasks mount, and where artifacts land."""

    def __init__(self, root: str, kind: str = "local"):
        self.root = root
        self.kind = kind  # "local" (hostPath analog) | "nfs"
        os.makedirs(root, exist_ok=True)

    def job_output_dir(self, job_name: str) -> str:
        d = os.path.join(self.root, "outputs", job_name)
        os.makedirs(d, exist_ok=True)
        return d

    def artifact_dir(self) -> str:
        d = os.path.join(self.root, "artifacts")
        os.makedirs(d, exist_ok=True)
        return d


class ModelRegistry:
    """Model/ModelVersion store + the 'image builder'."""

    def __init__(self, storage: StorageProvider):
        self.storage = storage
        self.models: dict[str, Model] = {}

    def ensure_model(self, name: str) -> Model:
        # reference modelversion_controller.go:114-163
        if name not in self.models:
            self.models[name] = Model(name)
        return self.models[name]

    def create_version_for_job(self, job, src_dir: str | None = None) -> ModelVersion | None:
        """mv-<job>-<uid5> naming parity (job.go:462-508)."""
        if not job.model_name:
            return None
        version = f"mv-{job.name}-{str(job.uid).zfill(5)[-5:]}"
        src = src_dir or self.storage.job_output_dir(job.name)
        return self.build_version(job.model_name, version, src,
                                  source_job=job.name)

    def build_version(self, model_name: str, version: str, src_dir: str,
                      source_job: str | None = None) -> ModelVersion:
        """The Kaniko-pod analog: pack src_dir into a layer tar.gz +
        manifest; content digest = sha256 of the layer."""
        model = self.ensure_model(model_name)
        out = os.path.join(self.storage.artifact_dir(), model_name)
        os.makedirs(out, exist_ok=True)
        layer = os.path.join(out, f"{version}.tar.gz")
        mv = ModelVersion(model=model_name, version=version, image_ref=layer,
                          source_job=source_job, build_phase="Building")
        model.versions[version] = mv
        try:
            with tarfile.open(layer, "w:gz") as tf:
                tf.add(src_dir, arcname=MODEL_IMAGE_PATH.lstrip("/"))
            h = hashlib.sha256()
            with open(layer, "rb") as f:
                for chunk in iter(lambda: f.read(1 << 20), b""):
                    h.update(chunk)
            mv.digest = "sha256:" + h.hexdigest()
            manifest = {
                "schemaVersion": 2,
                "model": model_name,
                "version": version,
                "sourceJob": source_job,
                "layers": [{"path": os.path.basename(layer),
                            "digest": mv.digest}],
                "modelPath": MODEL_IMAGE_PATH,
                "created": mv.ts,
            }
            with open(os.path.join(out, f"{version}.manifest.json"), "w") as f:
                json.dump(manifest, f, indent=2)
            mv.build_phase = "Succeeded"
            model.latest_version = version   # Model.LatestVersion advance
        except OSError:
            mv.build_phase = "Failed"
        return mv

    def get_version(self, model_name: str, version: str) -> ModelVersion | None:
        m = self.models.get(model_name)
        return m.versions.get(version) if m else None

    def extract(self, model_name: str, version: str, dest: str) -> str:
        """'docker run' analog: unpack the artifact for serving/resume."""
        mv = self.get_version(model_name, version)
        if mv is None or mv.build_phase != "Succeeded":
            raise KeyError(f"no built version {model_name}:{version}")
        os.makedirs(dest, exist_ok=True)
        with tarfile.open(mv.image_ref, "r:gz") as tf:
            tf.extractall(dest)
        return os.path.join(dest, MODEL_IMAGE_PATH.lstrip("/"))
