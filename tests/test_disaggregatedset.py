"""DisaggregatedSet end-to-end (in-process cluster) tests.

Mirrors the reference DS e2e scenarios
(test/e2e/disaggregatedset/e2e_test.go): simple create, lockstep rolling
update with coordinated drain, role scaling, revision-aware services,
external scalers via the /scale-shaped RoleScaler, slice scale-out.
"""
import pytest

from lws_amd.api import disaggregatedset as dsapi
from lws_amd.api import leaderworkerset as lwsapi
from tests.conftest import wait_for


def make_ds(name="my-ds", namespace="default", roles=None, slices=None,
            placement=None):
    from lws_amd.api.core import Container, PodSpec, PodTemplateSpec
    from lws_amd.api.disaggregatedset import (DisaggregatedRoleSpec,
                                              DisaggregatedSet,
                                              DisaggregatedSetSpec)
    from lws_amd.api.leaderworkerset import (LeaderWorkerSetSpec,
                                             LeaderWorkerTemplate)
    from lws_amd.api.meta import ObjectMeta

    roles = roles or [("prefill", 1, 2), ("decode", 1, 2)]
    role_specs = []
    for entry in roles:
        rname, replicas, size = entry[:3]
        role_specs.append(DisaggregatedRoleSpec(
            name=rname,
            spec=LeaderWorkerSetSpec(
                replicas=replicas,
                leader_worker_template=LeaderWorkerTemplate(
                    size=size,
                    worker_template=PodTemplateSpec(
                        spec=PodSpec(containers=[
                            Container(name="main", image="engine:v1")]))))))
    ds = DisaggregatedSet()
    ds.metadata = ObjectMeta(name=name, namespace=namespace)
    ds.spec = DisaggregatedSetSpec(roles=role_specs, slices=slices,
                                   placement_policy=placement)
    return ds


def ds_available(cluster, name="my-ds", ns="default"):
    cur = cluster.store.try_get(dsapi.KIND, ns, name)
    if cur is None:
        return None
    cond = next((c for c in cur.status.conditions if c.type == "Available"),
                None)
    if cond is None or cond.status != "True":
        return None
    return cur


def test_ds_validation(cluster):
    from lws_amd.cluster.store import InvalidError

    bad = make_ds(roles=[("only-one", 1, 1)])
    with pytest.raises(InvalidError):
        cluster.store.create(bad)

    mixed = make_ds(roles=[("a", 1, 1), ("b", 0, 1)])
    with pytest.raises(InvalidError):
        cluster.store.create(mixed)


def test_ds_create_becomes_available(cluster):
    ds = make_ds()
    cluster.store.create(ds)

    cur = wait_for(lambda: ds_available(cluster), desc="DS Available",
                   timeout=40)
    assert {rs.name for rs in cur.status.role_statuses} == \
        {"prefill", "decode"}
    for rs in cur.status.role_statuses:
        assert rs.replicas == 1 and rs.ready_replicas == 1 \
            and rs.updated_replicas == 1

    # children LWS named <ds>-<slice>-<rev8>-<role>, owned, labeled
    lws_list = cluster.store.list(lwsapi.KIND, "default")
    assert len(lws_list) == 2
    for lws in lws_list:
        parts = lws.metadata.name.split("-")
        assert parts[0] == "my" and parts[1] == "ds" and parts[2] == "0"
        assert len(parts[3]) == 8  # revision hash
        labels = lws.metadata.labels
        assert labels[dsapi.SET_NAME_LABEL_KEY] == "my-ds"
        assert labels[dsapi.SLICE_LABEL_KEY] == "0"
        ref = lws.metadata.owner_references[0]
        assert ref.kind == "DisaggregatedSet" and ref.controller

    # revision-aware services <lws>-prv
    def services_ready():
        svcs = [s for s in cluster.store.list("Service", "default")
                if s.metadata.name.endswith("-prv")]
        return svcs if len(svcs) == 2 else None
    svcs = wait_for(services_ready, desc="role services", timeout=20)
    for svc in svcs:
        assert svc.spec.selector[dsapi.ROLE_LABEL_KEY] in ("prefill", "decode")
        assert svc.spec.selector[dsapi.REVISION_LABEL_KEY]


def test_ds_lockstep_rolling_update(cluster):
    ds = make_ds(roles=[("prefill", 2, 1), ("decode", 3, 1)])
    cluster.store.create(ds)
    wait_for(lambda: ds_available(cluster), desc="initial Available",
             timeout=60)
    old_lws = {l.metadata.name for l in cluster.store.list(lwsapi.KIND,
                                                           "default")}

    from tests.conftest import retry_update

    def set_images(o):
        for role in o.spec.roles:
            role.spec.leader_worker_template.worker_template.spec \
                .containers[0].image = "engine:v2"
    retry_update(cluster.store, dsapi.KIND, "default", "my-ds", set_images)

    def rolled():
        c = ds_available(cluster)
        if c is None:
            return None
        lws_list = cluster.store.list(lwsapi.KIND, "default")
        if {l.metadata.name for l in lws_list} & old_lws:
            return None  # old revision LWS must be cleaned up
        if len(lws_list) != 2:
            return None
        for l in lws_list:
            if l.spec.leader_worker_template.worker_template.spec \
                    .containers[0].image != "engine:v2":
                return None
        return c
    wait_for(rolled, desc="lockstep rollout complete", timeout=120)
    cur = cluster.store.get(dsapi.KIND, "default", "my-ds")
    by_name = {rs.name: rs for rs in cur.status.role_statuses}
    assert by_name["prefill"].replicas == 2
    assert by_name["decode"].replicas == 3


def test_ds_role_scale(cluster):
    ds = make_ds(roles=[("prefill", 1, 1), ("decode", 2, 1)])
    cluster.store.create(ds)
    wait_for(lambda: ds_available(cluster), desc="Available", timeout=60)

    from tests.conftest import retry_update
    retry_update(cluster.store, dsapi.KIND, "default", "my-ds",
                 lambda o: setattr(o.spec.roles[1].spec, "replicas", 4))

    def scaled():
        c = ds_available(cluster)
        if c is None:
            return None
        by_name = {rs.name: rs for rs in c.status.role_statuses}
        return c if by_name["decode"].replicas == 4 else None
    wait_for(scaled, desc="decode scaled to 4", timeout=60)


def test_ds_external_scaler(cluster):
    from lws_amd.api.disaggregatedset import RoleScaling

    ds = make_ds(roles=[("prefill", 1, 1), ("decode", 0, 1)])
    ds.spec.roles[1].scaling = RoleScaling(mode="External")
    ds.spec.roles[1].spec.replicas = None
    cluster.store.create(ds)

    # scaler auto-created, seeded with 1 (fresh role)
    scaler = wait_for(lambda: cluster.store.try_get(
        dsapi.SCALER_KIND, "default", "my-ds-decode"), desc="scaler created",
        timeout=30)
    assert scaler.spec.replicas == 1
    ref = scaler.metadata.owner_references[0]
    assert ref.kind == "DisaggregatedSet" and ref.controller

    wait_for(lambda: ds_available(cluster), desc="Available", timeout=60)

    # external autoscaler writes spec.replicas (the /scale path)
    from tests.conftest import retry_update
    retry_update(cluster.store, dsapi.SCALER_KIND, "default", "my-ds-decode",
                 lambda o: setattr(o.spec, "replicas", 3))

    def scaled():
        c = ds_available(cluster)
        if c is None:
            return None
        by_name = {rs.name: rs for rs in c.status.role_statuses}
        return c if by_name["decode"].replicas == 3 else None
    wait_for(scaled, desc="external scale to 3", timeout=60)

    # status written back for HPA ratio math (leader-only selector)
    def status_written():
        s = cluster.store.get(dsapi.SCALER_KIND, "default", "my-ds-decode")
        return s if s.status.replicas == 3 and "worker-index=0" in \
            s.status.selector else None
    wait_for(status_written, desc="scaler status", timeout=30)


def test_ds_slices():
    from lws_amd.cluster.cluster import LwsCluster, make_nodes

    c = LwsCluster(nodes=make_nodes(1, gpus_per_node=8)).start()
    try:
        ds = make_ds(roles=[("prefill", 1, 1), ("decode", 1, 1)], slices=2)
        c.store.create(ds)
        wait_for(lambda: ds_available(c), desc="2-slice Available", timeout=90)
        lws_list = c.store.list(lwsapi.KIND, "default")
        assert len(lws_list) == 4  # 2 roles x 2 slices
        slices = {l.metadata.labels[dsapi.SLICE_LABEL_KEY] for l in lws_list}
        assert slices == {"0", "1"}
        # status aggregates across slices
        cur = c.store.get(dsapi.KIND, "default", "my-ds")
        for rs in cur.status.role_statuses:
            assert rs.replicas == 2

        # slice scale-UP adds slice-2 objects and re-reaches Available
        from tests.conftest import retry_update
        retry_update(c.store, dsapi.KIND, "default", "my-ds",
                     lambda o: setattr(o.spec, "slices", 3))

        def three_slices():
            lws_list = c.store.list(lwsapi.KIND, "default")
            slices = {l.metadata.labels[dsapi.SLICE_LABEL_KEY]
                      for l in lws_list}
            return (lws_list if len(lws_list) == 6 and
                    slices == {"0", "1", "2"} and ds_available(c) else None)
        wait_for(three_slices, desc="slice scale-up to 3", timeout=90)

        # slice scale-down removes slice-1/2 objects
        retry_update(c.store, dsapi.KIND, "default", "my-ds",
                     lambda o: setattr(o.spec, "slices", 1))
        def one_slice():
            lws_list = c.store.list(lwsapi.KIND, "default")
            if len(lws_list) != 2:
                return None
            return all(l.metadata.labels[dsapi.SLICE_LABEL_KEY] == "0"
                       for l in lws_list) or None
        wait_for(one_slice, desc="slice-1/2 cleanup", timeout=60)
    finally:
        c.stop()


def test_ds_role_add_and_remove(cluster):
    from tests.conftest import retry_update

    ds = make_ds(roles=[("prefill", 1, 1), ("decode", 1, 1)])
    cluster.store.create(ds)
    wait_for(lambda: ds_available(cluster), desc="Available", timeout=60)

    # add a third role (template change -> new revision, lockstep rollout)
    def add_role(o):
        import copy
        from lws_amd.api import serde
        new_role = serde.from_dict(type(o.spec.roles[0]),
                                   serde.to_dict(o.spec.roles[0]))
        new_role.name = "cache"
        o.spec.roles.append(new_role)
    retry_update(cluster.store, dsapi.KIND, "default", "my-ds", add_role)

    def three_roles():
        c = ds_available(cluster)
        if c is None:
            return None
        names = {rs.name for rs in c.status.role_statuses}
        return c if names == {"prefill", "decode", "cache"} else None
    wait_for(three_roles, desc="role added + Available", timeout=120)

    # remove it again; its LWS must drain away and status drop the entry
    retry_update(cluster.store, dsapi.KIND, "default", "my-ds",
                 lambda o: o.spec.roles.pop())

    def two_roles():
        c = ds_available(cluster)
        if c is None:
            return None
        names = {rs.name for rs in c.status.role_statuses}
        if names != {"prefill", "decode"}:
            return None
        lws_roles = {l.metadata.labels[dsapi.ROLE_LABEL_KEY]
                     for l in cluster.store.list(lwsapi.KIND, "default")}
        return c if "cache" not in lws_roles else None
    wait_for(two_roles, desc="role removed + drained", timeout=120)


def test_ds_exclusive_slice_placement():
    from lws_amd.api.disaggregatedset import PlacementPolicy
    from lws_amd.cluster.cluster import LwsCluster, make_nodes

    topo = "topology.lws.amd.com/island"
    # 2 islands x 4 GPUs; 2 slices of (prefill 1 + decode 1 pods, 1 GPU each)
    c = LwsCluster(nodes=make_nodes(2, gpus_per_node=4)).start()
    try:
        ds = make_ds(roles=[("prefill", 1, 1), ("decode", 1, 1)], slices=2,
                     placement=PlacementPolicy(type="ExclusiveSlice",
                                               topology=topo))
        for role in ds.spec.roles:
            role.spec.leader_worker_template.worker_template.spec \
                .containers[0].resources.requests = {"amd.com/gpu": 1}
        c.store.create(ds)
        wait_for(lambda: ds_available(c), desc="2-slice placed Available",
                 timeout=90)
        pods = c.store.list("Pod", "default")
        assert len(pods) == 4
        by_slice = {}
        for p in pods:
            sl = p.metadata.labels[dsapi.SLICE_LABEL_KEY]
            node = c.node(p.node_name)
            by_slice.setdefault(sl, set()).add(node.metadata.labels[topo])
        # each slice's roles co-located on ONE island; slices on DIFFERENT
        assert all(len(v) == 1 for v in by_slice.values()), by_slice
        assert len({next(iter(v)) for v in by_slice.values()}) == 2, by_slice
    finally:
        c.stop()
