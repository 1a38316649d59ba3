"""End-to-end (in-process cluster) tests of the LWS create path.

Mirrors the reference integration suite's create scenarios
(test/integration/controllers/leaderworkerset_test.go) against the lws_amd
substrate with FakeRuntime node agents standing in for kubelet.
"""
import pytest

from lws_amd.api import leaderworkerset as lwsapi
from tests.conftest import lws_condition, make_lws, wait_for


def test_lws_create_defaults(cluster):
    lws = make_lws(replicas=None, size=None)
    lws.spec.replicas = None
    lws.spec.leader_worker_template.size = None
    created = cluster.store.create(lws)
    assert created.spec.replicas == 1
    assert created.spec.leader_worker_template.size == 1
    assert created.spec.leader_worker_template.restart_policy == \
        "RecreateGroupOnPodRestart"
    assert created.spec.startup_policy == "LeaderCreated"
    assert created.spec.rollout_strategy.type == "RollingUpdate"
    ruc = created.spec.rollout_strategy.rolling_update_configuration
    assert (ruc.partition, ruc.max_unavailable, ruc.max_surge) == (0, 1, 0)
    assert created.spec.network_config.subdomain_policy == "Shared"


def test_lws_validation_rejects_bad_spec(cluster):
    from lws_amd.cluster.store import InvalidError

    bad = make_lws(name="Bad_Name")
    with pytest.raises(InvalidError):
        cluster.store.create(bad)

    bad2 = make_lws()
    bad2.spec.rollout_strategy.type = "RollingUpdate"
    from lws_amd.api.leaderworkerset import RollingUpdateConfiguration
    bad2.spec.rollout_strategy.rolling_update_configuration = \
        RollingUpdateConfiguration(partition=0, max_unavailable=0, max_surge=0)
    with pytest.raises(InvalidError):
        cluster.store.create(bad2)


def test_lws_group_becomes_ready(cluster):
    lws = make_lws(replicas=2, size=2)
    cluster.store.create(lws)

    # leader STS exists with the right shape
    sts = wait_for(lambda: cluster.store.try_get("StatefulSet", "default",
                                                 "my-lws"),
                   desc="leader statefulset")
    assert sts.spec.replicas == 2
    assert sts.spec.pod_management_policy == "Parallel"
    assert sts.spec.service_name == "my-lws"
    assert sts.metadata.annotations[lwsapi.REPLICAS_ANNOTATION_KEY] == "2"
    assert sts.spec.template.metadata.labels[lwsapi.WORKER_INDEX_LABEL_KEY] == "0"
    assert sts.spec.template.metadata.annotations[
        lwsapi.SIZE_ANNOTATION_KEY] == "2"

    # shared headless service
    svc = wait_for(lambda: cluster.store.try_get("Service", "default", "my-lws"),
                   desc="headless service")
    assert svc.spec.cluster_ip == "None"
    assert svc.spec.publish_not_ready_addresses

    # leader pods + worker statefulsets + worker pods
    def all_pods():
        pods = cluster.store.list("Pod", "default")
        return pods if len(pods) == 4 else None
    pods = wait_for(all_pods, desc="4 pods (2 groups x size 2)")
    names = sorted(p.metadata.name for p in pods)
    assert names == ["my-lws-0", "my-lws-0-1", "my-lws-1", "my-lws-1-1"]

    # webhook-injected identity + env
    by_name = {p.metadata.name: p for p in pods}
    leader0 = by_name["my-lws-0"]
    worker01 = by_name["my-lws-0-1"]
    assert leader0.metadata.labels[lwsapi.GROUP_INDEX_LABEL_KEY] == "0"
    assert leader0.metadata.labels[lwsapi.WORKER_INDEX_LABEL_KEY] == "0"
    assert lwsapi.GROUP_UNIQUE_HASH_LABEL_KEY in leader0.metadata.labels
    assert worker01.metadata.labels[lwsapi.WORKER_INDEX_LABEL_KEY] == "1"
    env = {e.name: e.value for e in worker01.spec.containers[0].env}
    assert env[lwsapi.LWS_LEADER_ADDRESS] == "my-lws-0.my-lws.default"
    assert env[lwsapi.LWS_GROUP_SIZE] == "2"
    assert env[lwsapi.LWS_WORKER_INDEX] == "1"
    env_l = {e.name: e.value for e in leader0.spec.containers[0].env}
    assert env_l[lwsapi.LWS_WORKER_INDEX] == "0"

    # worker pods annotated with leader name
    assert worker01.metadata.annotations[
        lwsapi.LEADER_POD_NAME_ANNOTATION_KEY] == "my-lws-0"

    # Available condition + status counts
    def available():
        cur = cluster.get_lws("default", "my-lws")
        cond = lws_condition(cur, "Available")
        return cur if cond is not None and cond.status == "True" else None
    cur = wait_for(available, desc="Available condition")
    assert cur.status.ready_replicas == 2
    assert cur.status.updated_replicas == 2
    assert cur.status.replicas == 2
    assert "worker-index=0" in cur.status.hpa_pod_selector


def test_size_one_no_worker_sts(cluster):
    lws = make_lws(name="solo", replicas=1, size=1)
    cluster.store.create(lws)

    def available():
        cur = cluster.get_lws("default", "solo")
        cond = lws_condition(cur, "Available")
        return cur if cond is not None and cond.status == "True" else None
    wait_for(available, desc="Available for size-1")
    # worker sts named after leader pod must NOT exist
    assert cluster.store.try_get("StatefulSet", "default", "solo-0") is None
    pods = cluster.store.list("Pod", "default")
    assert [p.metadata.name for p in pods] == ["solo-0"]


def test_leader_template_used_for_leader(cluster):
    from lws_amd.api.core import Container, PodSpec, PodTemplateSpec

    lws = make_lws(name="lt", replicas=1, size=2)
    lws.spec.leader_worker_template.leader_template = PodTemplateSpec(
        spec=PodSpec(containers=[Container(name="leader-main",
                                           image="leader:latest")]))
    cluster.store.create(lws)
    pods = wait_for(lambda: (lambda ps: ps if len(ps) == 2 else None)(
        cluster.store.list("Pod", "default")), desc="2 pods")
    by_name = {p.metadata.name: p for p in pods}
    assert by_name["lt-0"].spec.containers[0].image == "leader:latest"
    assert by_name["lt-0-1"].spec.containers[0].image == "engine:latest"


def test_scale_up_and_down(cluster):
    lws = make_lws(name="scale", replicas=1, size=2)
    cluster.store.create(lws)
    wait_for(lambda: len(cluster.store.list("Pod", "default")) == 2,
             desc="initial 2 pods")

    from tests.conftest import retry_update
    retry_update(cluster.store, "LeaderWorkerSet", "default", "scale",
                 lambda o: setattr(o.spec, "replicas", 3))
    wait_for(lambda: len(cluster.store.list("Pod", "default")) == 6,
             desc="scale up to 6 pods")

    retry_update(cluster.store, "LeaderWorkerSet", "default", "scale",
                 lambda o: setattr(o.spec, "replicas", 1))
    wait_for(lambda: len(cluster.store.list("Pod", "default")) == 2,
             desc="scale down to 2 pods", timeout=30)
    names = sorted(p.metadata.name for p in cluster.store.list("Pod", "default"))
    assert names == ["scale-0", "scale-0-1"]


def test_volume_claim_templates_create_pvcs(cluster):
    from lws_amd.api.core import (PersistentVolumeClaim,
                                  PersistentVolumeClaimSpec,
                                  ResourceRequirements,
                                  StatefulSetPersistentVolumeClaimRetentionPolicy)
    from lws_amd.api.meta import ObjectMeta

    lws = make_lws(name="pvc", replicas=1, size=2)
    vct = PersistentVolumeClaim(
        metadata=ObjectMeta(name="model-cache"),
        spec=PersistentVolumeClaimSpec(
            access_modes=["ReadWriteOnce"],
            resources=ResourceRequirements(requests={"storage": "10Gi"})))
    lws.spec.leader_worker_template.volume_claim_templates = [vct]
    lws.spec.leader_worker_template.persistent_volume_claim_retention_policy = \
        StatefulSetPersistentVolumeClaimRetentionPolicy(when_deleted="Delete")
    cluster.store.create(lws)

    def pvcs():
        got = cluster.store.list("PersistentVolumeClaim", "default")
        return got if len(got) == 2 else None
    got = wait_for(pvcs, desc="2 PVCs (leader + worker)", timeout=20)
    names = sorted(p.metadata.name for p in got)
    assert names == ["model-cache-pvc-0", "model-cache-pvc-0-1"]
    assert got[0].spec.resources.requests["storage"] == "10Gi"
