"""Kubernetes cluster management (parity: python/scannerpy/kube.py, which
provisioned GKE clusters with machine configs, deployed master/worker pods
and autoscaled them; here cloud-agnostic: MachineConfig/ClusterConfig
dataclasses generate the manifests programmatically — the static YAML in
deploy/ is the rendered default — and `Cluster` drives kubectl. The
kubectl runner is injectable so everything is testable without a cluster;
pricing tables are out of scope (cloud-specific)."""
import json
import shlex
import subprocess

from .common import ScannerException

DEFAULT_IMAGE = "YOUR_REGISTRY/scanner-amd:latest"


class MachineConfig:
    """Per-pod resources (parity: kube.py MachineType/MachineConfig)."""

    # USD/hour rate card for cost estimation (parity: the reference's
    # hardcoded GCP price tables, kube.py:124-177). Override per
    # deployment — cloud MI3xx list prices move frequently.
    RATES = {"cpu": 0.032, "mem_gb": 0.0043, "gpu": 2.50}
    PREEMPTIBLE_DISCOUNT = 0.30  # spot/preemptible multiplier

    def __init__(self, cpus=8, memory_gb=32, gpus_per_node=8,
                 gpu_resource="amd.com/gpu", preemptible=False):
        self.cpus = cpus
        self.memory_gb = memory_gb
        self.gpus_per_node = gpus_per_node
        self.gpu_resource = gpu_resource
        self.preemptible = preemptible

    def price(self):
        """Estimated USD/hour for one pod of this shape."""
        p = (self.RATES["cpu"] * self.cpus +
             self.RATES["mem_gb"] * self.memory_gb +
             self.RATES["gpu"] * self.gpus_per_node)
        return p * (self.PREEMPTIBLE_DISCOUNT if self.preemptible else 1.0)


class ClusterConfig:
    """Whole-cluster shape (parity: kube.py ClusterConfig)."""

    def __init__(self, id="scanner", num_workers=2,
                 master=None, worker=None, image=DEFAULT_IMAGE,
                 shared_claim="scanner-shared", db_path="/shared/db",
                 master_port=5001, namespace="default",
                 autoscale_max=0, storage_type="posix", bucket=""):
        self.id = id
        self.num_workers = num_workers
        self.master = master or MachineConfig(gpus_per_node=0)
        self.worker = worker or MachineConfig()
        self.image = image
        self.shared_claim = shared_claim
        self.db_path = db_path
        self.master_port = master_port
        self.namespace = namespace
        # >0: also emit an HPA capping worker replicas
        self.autoscale_max = autoscale_max
        # "s3": cluster runs on the object-store backend (bucket required;
        # db_path becomes a key prefix); "posix": the RWX shared volume
        self.storage_type = storage_type
        self.bucket = bucket

    def price(self, no_master=False):
        """Estimated whole-cluster USD/hour (parity: reference
        ClusterConfig.price kube.py:209)."""
        p = 0.0 if no_master else self.master.price()
        return p + self.worker.price() * self.num_workers


def _resources(mc):
    lim = {"cpu": str(mc.cpus), "memory": f"{mc.memory_gb}Gi"}
    if mc.gpus_per_node:
        lim[mc.gpu_resource] = str(mc.gpus_per_node)
    return {"limits": lim}


def master_manifests(cfg):
    name = f"{cfg.id}-master"
    dep = {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": {"name": name, "namespace": cfg.namespace},
        "spec": {
            "replicas": 1,
            "selector": {"matchLabels": {"app": name}},
            "template": {
                "metadata": {"labels": {"app": name}},
                "spec": {
                    "containers": [{
                        "name": "master",
                        "image": cfg.image,
                        "command": ["python", "-m", "scanner_amd.master",
                                    "--db-path", cfg.db_path, "--addr",
                                    f"0.0.0.0:{cfg.master_port}",
                                    "--storage-type", cfg.storage_type,
                                    "--bucket", cfg.bucket],
                        "ports": [{"containerPort": cfg.master_port}],
                        "resources": _resources(cfg.master),
                        "volumeMounts": [{"name": "shared",
                                          "mountPath": "/shared"}],
                    }],
                    "volumes": [{"name": "shared",
                                 "persistentVolumeClaim":
                                     {"claimName": cfg.shared_claim}}],
                },
            },
        },
    }
    svc = {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {"name": name, "namespace": cfg.namespace},
        "spec": {"selector": {"app": name},
                 "ports": [{"port": cfg.master_port,
                            "targetPort": cfg.master_port}]},
    }
    return [dep, svc]


def worker_manifests(cfg):
    name = f"{cfg.id}-worker"
    dep = {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": {"name": name, "namespace": cfg.namespace},
        "spec": {
            "replicas": cfg.num_workers,
            "selector": {"matchLabels": {"app": name}},
            "template": {
                "metadata": {"labels": {"app": name}},
                "spec": {
                    "containers": [{
                        "name": "worker",
                        "image": cfg.image,
                        "command": ["python", "-m", "scanner_amd.worker",
                                    "--master",
                                    f"{cfg.id}-master:{cfg.master_port}",
                                    "--db-path", cfg.db_path,
                                    "--instances",
                                    str(max(1,
                                            cfg.worker.gpus_per_node)),
                                    "--storage-type", cfg.storage_type,
                                    "--bucket", cfg.bucket],
                        "resources": _resources(cfg.worker),
                        "volumeMounts": [{"name": "shared",
                                          "mountPath": "/shared"}],
                    }],
                    "volumes": [{"name": "shared",
                                 "persistentVolumeClaim":
                                     {"claimName": cfg.shared_claim}}],
                },
            },
        },
    }
    out = [dep]
    if cfg.autoscale_max > cfg.num_workers:
        out.append({
            "apiVersion": "autoscaling/v2",
            "kind": "HorizontalPodAutoscaler",
            "metadata": {"name": name, "namespace": cfg.namespace},
            "spec": {
                "scaleTargetRef": {"apiVersion": "apps/v1",
                                   "kind": "Deployment", "name": name},
                "minReplicas": cfg.num_workers,
                "maxReplicas": cfg.autoscale_max,
                "metrics": [{
                    "type": "Resource",
                    "resource": {"name": "cpu",
                                 "target": {"type": "Utilization",
                                            "averageUtilization": 80}},
                }],
            },
        })
    return out


class Cluster:
    """Deploy/scale/tear down a scanner cluster on Kubernetes (parity:
    kube.py Cluster). `runner` is invoked with kubectl argv and stdin
    text; inject a fake for tests / dry runs."""

    def __init__(self, config=None, runner=None):
        self.config = config or ClusterConfig()
        self._run = runner or self._kubectl

    @staticmethod
    def _kubectl(argv, stdin_text=None):
        p = subprocess.run(["kubectl"] + argv, input=stdin_text,
                           capture_output=True, text=True, timeout=120)
        if p.returncode != 0:
            raise ScannerException(
                f"kubectl {' '.join(map(shlex.quote, argv))} failed: "
                f"{p.stderr.strip()[-500:]}")
        return p.stdout

    def manifests(self):
        return master_manifests(self.config) + worker_manifests(self.config)

    def deploy(self):
        docs = "\n---\n".join(json.dumps(m) for m in self.manifests())
        self._run(["apply", "-n", self.config.namespace, "-f", "-"], docs)
        return self

    def scale_workers(self, replicas):
        self.config.num_workers = replicas
        self._run(["scale", "-n", self.config.namespace,
                   f"deployment/{self.config.id}-worker",
                   f"--replicas={replicas}"])

    def master_address(self):
        return f"{self.config.id}-master:{self.config.master_port}"

    def delete(self):
        for kind, name in [("deployment", f"{self.config.id}-worker"),
                           ("hpa", f"{self.config.id}-worker"),
                           ("deployment", f"{self.config.id}-master"),
                           ("service", f"{self.config.id}-master")]:
            try:
                self._run(["delete", "-n", self.config.namespace,
                           kind, name, "--ignore-not-found"])
            except ScannerException:
                pass
