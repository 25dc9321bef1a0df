"""Model packaging: Model / ModelVersion registry.

Reference: the Model/ModelVersion CRDs + ModelVersion controller
(apis/model/v1alpha1/, controllers/model/modelversion_controller.go):
on job success a ModelVersion is created from the job's output storage,
a Kaniko pod bakes `FROM busybox; COPY build/ /torch-on-k8s-model` and
pushes an image, and Model.LatestVersion advances.

Node-native redesign: there is no registry daemon on the box, so an
"image" is a REAL OCI image layout on disk (opencontainers image-spec:
oci-layout + index.json + blobs/sha256/{layer,config,manifest}) that
`podman load` / `skopeo copy oci:<dir>` accept as-is — pushing to a
registry is one skopeo invocation away, which is the Kaniko-push parity
point (modelversion_controller.go:286-406). The config blob carries
TORCH_ON_K8S_MODEL_PATH (model/v1alpha1/constants.go:24-27); build
lifecycle is Building -> Succeeded/Failed and Model.latest_version only
advances on success. Storage providers mirror the reference's
LocalStorage/NFS split (pkg/storage/).
"""
from __future__ import annotations

import gzip
import hashlib
import json
import os
import shutil
import tarfile
import time
from dataclasses import dataclass, field

MODEL_PATH_ENV = "TORCH_ON_K8S_MODEL_PATH"
MODEL_IMAGE_PATH = "/torch-on-k8s-model"  # path baked into the artifact


def _safe_extract(tf: tarfile.TarFile, dest: str) -> None:
    """extractall with member validation (py3.10 tarfile has no filter
    arg): artifacts under the models dir are data, not trusted input —
    reject absolute paths, '..' traversal, and symlinks/hardlinks that
    escape dest."""
    root = os.path.realpath(dest)
    for m in tf.getmembers():
        target = os.path.realpath(os.path.join(root, m.name))
        if not (target == root or target.startswith(root + os.sep)):
            raise RuntimeError(f"unsafe tar member path: {m.name!r}")
        if m.issym() or m.islnk():
            linkt = m.linkname
            base = os.path.dirname(target) if m.issym() else root
            link_target = os.path.realpath(os.path.join(base, linkt))
            if not link_target.startswith(root + os.sep):
                raise RuntimeError(
                    f"unsafe tar link: {m.name!r} -> {linkt!r}")
        if m.isdev():
            raise RuntimeError(f"device node in artifact: {m.name!r}")
    tf.extractall(dest)


@dataclass
class ModelVersion:
    model: str
    version: str
    image_ref: str          # OCI image layout dir (registry analog)
    digest: str = ""
    build_phase: str = "Created"   # Created|Building|Succeeded|Failed
    source_job: str | None = None
    storage: dict = field(default_factory=dict)  # provider provenance
    ts: float = field(default_factory=time.time)


@dataclass
class Model:
    name: str
    description: str = ""
    latest_version: str | None = None
    versions: dict = field(default_factory=dict)


class StorageProvider:
    """Reference pkg/storage interface analog (storage/interface.go:26-35):
    yields the model output dir a job's tasks mount, where artifacts
    land, and the provenance the reference records in PV specs."""

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

    def provenance(self) -> dict:
        """What the reference bakes into the PV spec; consumers (task
        env injection, ModelVersion records) read this to know where
        the bytes physically live."""
        return {"kind": self.kind, "root": self.root}

    def task_env(self) -> dict:
        """Env vars injected into tasks that mount this storage."""
        return {"TOK_STORAGE_KIND": self.kind, "TOK_STORAGE_ROOT": self.root}


class LocalStorageProvider(StorageProvider):
    """hostPath PV pinned by node affinity
    (localstorage/local_storage.go:36-109): on the single-node design
    the affinity collapses to recording WHICH node owns the bytes so a
    multi-node deployment can schedule readers onto it."""

    def __init__(self, root: str, node_name: str | None = None):
        super().__init__(root, kind="local")
        import socket
        self.node_name = node_name or socket.gethostname()

    def provenance(self) -> dict:
        return {"kind": self.kind, "root": self.root,
                "nodeAffinity": {"kubernetes.io/hostname": self.node_name}}

    def task_env(self) -> dict:
        env = super().task_env()
        env["TOK_STORAGE_NODE"] = self.node_name
        return env


class NFSStorageProvider(StorageProvider):
    """NFS-backed PV (nfs/nfs.go:36-89): carries server + export path
    provenance; any node may mount it (no affinity)."""

    def __init__(self, root: str, server: str, path: str):
        super().__init__(root, kind="nfs")
        self.server = server
        self.path = path

    def provenance(self) -> dict:
        return {"kind": self.kind, "root": self.root,
                "server": self.server, "path": self.path}

    def task_env(self) -> dict:
        env = super().task_env()
        env["TOK_STORAGE_NFS_SERVER"] = self.server
        env["TOK_STORAGE_NFS_PATH"] = self.path
        return env


def storage_from_spec(root: str, spec: dict | None) -> StorageProvider:
    """Registry parity (storage/registry/registry.go:36-44): pick the
    provider by which field set is present in the storage spec."""
    spec = spec or {}
    if "nfs" in spec:
        nfs = spec["nfs"]
        return NFSStorageProvider(root, server=nfs.get("server", ""),
                                  path=nfs.get("path", "/"))
    if "localStorage" in spec:
        ls = spec["localStorage"]
        return LocalStorageProvider(ls.get("path", root),
                                    node_name=ls.get("nodeName"))
    return LocalStorageProvider(root)


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

    @staticmethod
    def _sha256_file(path: str) -> str:
        h = hashlib.sha256()
        with open(path, "rb") as f:
            for chunk in iter(lambda: f.read(1 << 20), b""):
                h.update(chunk)
        return h.hexdigest()

    def build_version(self, model_name: str, version: str, src_dir: str,
                      source_job: str | None = None) -> ModelVersion:
        """The Kaniko-pod analog: bake src_dir into an OCI image layout
        (`FROM scratch; COPY src /torch-on-k8s-model; ENV
        TORCH_ON_K8S_MODEL_PATH=...`), Building -> Succeeded/Failed."""
        model = self.ensure_model(model_name)
        image_dir = os.path.join(self.storage.artifact_dir(), model_name,
                                 version)
        mv = ModelVersion(model=model_name, version=version,
                          image_ref=image_dir, source_job=source_job,
                          storage=self.storage.provenance(),
                          build_phase="Building")
        model.versions[version] = mv
        try:
            if not os.path.isdir(src_dir):
                raise OSError(f"source dir missing: {src_dir}")
            blobs = os.path.join(image_dir, "blobs", "sha256")
            os.makedirs(blobs, exist_ok=True)

            # layer: tar of src_dir at MODEL_IMAGE_PATH; OCI config needs
            # the UNCOMPRESSED digest (diff_id), the manifest the
            # compressed one
            tmp_tar = os.path.join(image_dir, ".layer.tar")
            with tarfile.open(tmp_tar, "w") as tf:
                tf.add(src_dir, arcname=MODEL_IMAGE_PATH.lstrip("/"))
            diff_id = self._sha256_file(tmp_tar)
            tmp_gz = tmp_tar + ".gz"
            with open(tmp_tar, "rb") as fin, \
                    gzip.GzipFile(filename="", mode="wb",
                                  fileobj=open(tmp_gz, "wb"),
                                  mtime=0) as fout:
                shutil.copyfileobj(fin, fout)
            os.unlink(tmp_tar)
            layer_digest = self._sha256_file(tmp_gz)
            layer_size = os.path.getsize(tmp_gz)
            os.replace(tmp_gz, os.path.join(blobs, layer_digest))

            def put_blob(obj) -> tuple[str, int]:
                data = json.dumps(obj).encode()
                d = hashlib.sha256(data).hexdigest()
                with open(os.path.join(blobs, d), "wb") as f:
                    f.write(data)
                return d, len(data)

            config_digest, config_size = put_blob({
                "created": time.strftime(
                    "%Y-%m-%dT%H:%M:%SZ", time.gmtime(mv.ts)),
                "architecture": "amd64",
                "os": "linux",
                "config": {
                    "Env": [f"{MODEL_PATH_ENV}={MODEL_IMAGE_PATH}"],
                    "Labels": {
                        "io.torch-on-k8s-amd.model": model_name,
                        "io.torch-on-k8s-amd.version": version,
                        "io.torch-on-k8s-amd.source-job": source_job or "",
                    },
                },
                "rootfs": {"type": "layers",
                           "diff_ids": [f"sha256:{diff_id}"]},
                "history": [{"created_by":
                             f"COPY {os.path.basename(src_dir)} "
                             f"{MODEL_IMAGE_PATH}"}],
            })
            manifest_digest, manifest_size = put_blob({
                "schemaVersion": 2,
                "mediaType": "application/vnd.oci.image.manifest.v1+json",
                "config": {
                    "mediaType": "application/vnd.oci.image.config.v1+json",
                    "digest": f"sha256:{config_digest}",
                    "size": config_size,
                },
                "layers": [{
                    "mediaType":
                        "application/vnd.oci.image.layer.v1.tar+gzip",
                    "digest": f"sha256:{layer_digest}",
                    "size": layer_size,
                }],
            })
            with open(os.path.join(image_dir, "oci-layout"), "w") as f:
                json.dump({"imageLayoutVersion": "1.0.0"}, f)
            with open(os.path.join(image_dir, "index.json"), "w") as f:
                json.dump({
                    "schemaVersion": 2,
                    "manifests": [{
                        "mediaType":
                            "application/vnd.oci.image.manifest.v1+json",
                        "digest": f"sha256:{manifest_digest}",
                        "size": manifest_size,
                        "annotations": {
                            "org.opencontainers.image.ref.name":
                                f"{model_name}:{version}",
                        },
                    }],
                }, f)
            mv.digest = "sha256:" + manifest_digest
            mv.build_phase = "Succeeded"
            model.latest_version = version   # Model.LatestVersion advance
        except (OSError, tarfile.TarError):
            mv.build_phase = "Failed"
            shutil.rmtree(image_dir, ignore_errors=True)
        return mv

    def reindex(self) -> int:
        """Rebuild the in-memory Model/ModelVersion view from the OCI
        layouts on disk (a fresh process — e.g. a serving daemon —
        pointed at the manager's artifact store). Returns the number of
        versions found."""
        found = 0
        root = self.storage.artifact_dir()
        for model_name in sorted(os.listdir(root)):
            mdir = os.path.join(root, model_name)
            if not os.path.isdir(mdir):
                continue
            for version in sorted(os.listdir(mdir)):
                vdir = os.path.join(mdir, version)
                idx = os.path.join(vdir, "index.json")
                if not os.path.isfile(idx):
                    continue
                try:
                    with open(idx) as f:
                        digest = json.load(f)["manifests"][0]["digest"]
                except (OSError, ValueError, KeyError, IndexError):
                    continue
                model = self.ensure_model(model_name)
                if version not in model.versions:
                    model.versions[version] = ModelVersion(
                        model=model_name, version=version, image_ref=vdir,
                        digest=digest, build_phase="Succeeded",
                        storage=self.storage.provenance())
                    model.latest_version = version  # sorted: last wins
                found += 1
        return found

    def get_version(self, model_name: str, version: str) -> ModelVersion | None:
        m = self.models.get(model_name)
        return m.versions.get(version) if m else None

    def extract(self, model_name: str, version: str, dest: str) -> str:
        """'podman run' analog: walk index -> manifest -> layers and
        unpack the rootfs for serving/resume."""
        mv = self.get_version(model_name, version)
        if mv is None or mv.build_phase != "Succeeded":
            raise KeyError(f"no built version {model_name}:{version}")
        blobs = os.path.join(mv.image_ref, "blobs", "sha256")
        with open(os.path.join(mv.image_ref, "index.json")) as f:
            index = json.load(f)
        mdigest = index["manifests"][0]["digest"].split(":", 1)[1]
        with open(os.path.join(blobs, mdigest)) as f:
            manifest = json.load(f)
        os.makedirs(dest, exist_ok=True)
        for layer in manifest["layers"]:
            ldigest = layer["digest"].split(":", 1)[1]
            with tarfile.open(os.path.join(blobs, ldigest), "r:gz") as tf:
                _safe_extract(tf, dest)
        return os.path.join(dest, MODEL_IMAGE_PATH.lstrip("/"))
