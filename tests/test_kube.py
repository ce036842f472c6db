"""Kubernetes cluster management (scanner_amd/kube.py — parity:
scannerpy kube.py cluster provisioning) with an injected kubectl runner:
manifests and command flow verified without a cluster."""
import json

from scanner_amd.kube import Cluster, ClusterConfig, MachineConfig


class FakeKubectl:
    def __init__(self):
        self.calls = []

    def __call__(self, argv, stdin_text=None):
        self.calls.append((argv, stdin_text))
        return ""


def test_manifest_shapes():
    cfg = ClusterConfig(id="t", num_workers=3,
                        worker=MachineConfig(cpus=16, memory_gb=64,
                                             gpus_per_node=8),
                        autoscale_max=6)
    c = Cluster(cfg, runner=FakeKubectl())
    ms = c.manifests()
    kinds = [m["kind"] for m in ms]
    assert kinds == ["Deployment", "Service", "Deployment",
                     "HorizontalPodAutoscaler"]
    worker = ms[2]
    assert worker["spec"]["replicas"] == 3
    ctr = worker["spec"]["template"]["spec"]["containers"][0]
    assert ctr["resources"]["limits"]["amd.com/gpu"] == "8"
    assert "--master" in ctr["command"]
    assert "t-master:5001" in ctr["command"]
    hpa = ms[3]
    assert hpa["spec"]["maxReplicas"] == 6
    # master pod requests no GPUs
    mctr = ms[0]["spec"]["template"]["spec"]["containers"][0]
    assert "amd.com/gpu" not in mctr["resources"]["limits"]


def test_deploy_scale_delete_flow():
    fake = FakeKubectl()
    c = Cluster(ClusterConfig(id="s", num_workers=2), runner=fake)
    c.deploy()
    c.scale_workers(5)
    c.delete()
    assert fake.calls[0][0][:2] == ["apply", "-n"]
    docs = fake.calls[0][1]
    assert docs and all(json.loads(d) for d in docs.split("\n---\n"))
    assert any("--replicas=5" in a for a in fake.calls[1][0])
    assert c.config.num_workers == 5
    deletes = [a for a, _ in fake.calls[2:]]
    assert len(deletes) == 4
    assert c.master_address() == "s-master:5001"


def test_cluster_price_estimate():
    """Cost estimation (parity: reference ClusterConfig.price)."""
    from scanner_amd.kube import ClusterConfig, MachineConfig
    cfg = ClusterConfig(
        num_workers=4,
        master=MachineConfig(cpus=4, memory_gb=16, gpus_per_node=0),
        worker=MachineConfig(cpus=16, memory_gb=128, gpus_per_node=8))
    m = 0.032 * 4 + 0.0043 * 16
    w = 0.032 * 16 + 0.0043 * 128 + 2.50 * 8
    assert abs(cfg.price() - (m + 4 * w)) < 1e-9
    assert abs(cfg.price(no_master=True) - 4 * w) < 1e-9
    spot = MachineConfig(cpus=16, memory_gb=128, gpus_per_node=8,
                         preemptible=True)
    assert abs(spot.price() - w * 0.30) < 1e-9
