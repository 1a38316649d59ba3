"""Per-leader-pod reconciler.

Behavioral port of reference pkg/controllers/pod_controller.go: restart
policy enforcement (all-or-nothing group recreation via foreground leader
deletion), worker StatefulSet creation from the revision-applied spec
(ordinals start at 1, replicas=size-1), per-replica headless Service for
UniquePerReplica, gang PodGroup creation, and exclusive-placement node
pinning of workers onto the leader's topology domain.
"""
from __future__ import annotations

from typing import Optional

from ..api import leaderworkerset as lwsapi
from ..api import serde
from ..api.core import (RollingUpdateStatefulSetStrategy, Service, ServiceSpec,
                        StatefulSet, StatefulSetOrdinals, StatefulSetSpec,
                        StatefulSetUpdateStrategy)
from ..api.meta import LabelSelector, OwnerReference
from ..cluster.controller import Controller, Manager
from ..cluster.statefulset_controller import parse_parent_and_ordinal
from ..cluster.store import AlreadyExistsError, NotFoundError, Store
from ..utils import revision as revisionutils
from ..utils.podutils import (container_restarted, leader_pod, pod_deleted,
                              pod_running_and_ready)
from .leaderworkerset_controller import _merge_metadata


class PodReconciler:
    def __init__(self, manager: Manager, scheduler_provider=None,
                 node_lookup=None, recorder=None) -> None:
        from ..cluster.events import NullRecorder
        self.store: Store = manager.store
        self.record = recorder or NullRecorder()
        self.scheduler_provider = scheduler_provider
        # node_lookup(node_name) -> Node (for topology label resolution)
        self.node_lookup = node_lookup or (lambda name: None)
        self.ctrl = Controller("pod", self.reconcile)
        manager.add_controller(self.ctrl)
        manager.watch("Pod", self.ctrl, self._map_pod)

    def _map_pod(self, event: str, pod) -> list[tuple[str, str]]:
        # event filter on SetNameLabelKey (pod_controller.go:463-477)
        if lwsapi.SET_NAME_LABEL_KEY in (pod.metadata.labels or {}):
            return [(pod.metadata.namespace, pod.metadata.name)]
        return []

    # ------------------------------------------------------------------
    def reconcile(self, namespace: str, name: str) -> Optional[float]:
        pod = self.store.try_get("Pod", namespace, name)
        if pod is None:
            return None
        labels = pod.metadata.labels or {}
        lws_name = labels.get(lwsapi.SET_NAME_LABEL_KEY)
        if not lws_name or lwsapi.WORKER_INDEX_LABEL_KEY not in labels:
            return None
        lws = self.store.try_get(lwsapi.KIND, namespace, lws_name)
        if lws is None:
            return None

        leader_deleted = self._handle_restart_policy(pod, lws)
        if leader_deleted:
            return None
        if not leader_pod(pod):
            return None
        if (pod.metadata.annotations or {}).get(
                lwsapi.LEADER_POD_NAME_ANNOTATION_KEY):
            # guard against misidentified leader (reference issue #391)
            return None
        if pod.metadata.deletion_timestamp is not None:
            return None

        if lws.spec.network_config is not None and \
                lws.spec.network_config.subdomain_policy == \
                lwsapi.SubdomainPolicy.UniquePerReplica:
            self._create_headless_service(
                lws, pod.metadata.name,
                {lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
                 lwsapi.GROUP_INDEX_LABEL_KEY:
                     labels.get(lwsapi.GROUP_INDEX_LABEL_KEY, "")},
                owner=pod)

        if self.scheduler_provider is not None:
            self.scheduler_provider.create_pod_group_if_not_exists(lws, pod)

        if lws.spec.leader_worker_template.size == 1:
            return None

        if lws.spec.startup_policy == lwsapi.StartupPolicyType.LeaderReady and \
                not pod_running_and_ready(pod):
            return None

        revision = revisionutils.get_revision(
            self.store, lws, revisionutils.get_revision_key(pod))
        if revision is None:
            return 1.0  # revision not created yet; requeue

        sts = self._construct_worker_statefulset(pod, lws, revision)

        topology_key = (lws.metadata.annotations or {}).get(
            lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY)
        if topology_key:
            if not pod.node_name:
                return None  # leader not scheduled yet
            topo_value = self._topology_value_from_pod(pod, topology_key)
            if topo_value is None:
                raise RuntimeError(
                    f"node for pod {pod.metadata.name} lacks topology label "
                    f"{topology_key}")
            sts.spec.template.spec.node_selector = dict(
                sts.spec.template.spec.node_selector or {})
            sts.spec.template.spec.node_selector[topology_key] = topo_value

        if self.store.try_get("StatefulSet", namespace, pod.metadata.name) is None:
            try:
                self.store.create(sts)
            except AlreadyExistsError:
                pass
        return None

    # ------------------------------------------------------------------
    def _handle_restart_policy(self, pod, lws) -> bool:
        """pod_controller.go:204-266."""
        policy = lws.spec.leader_worker_template.restart_policy
        if policy not in (lwsapi.RestartPolicyType.RecreateGroupOnPodRestart,
                          lwsapi.RestartPolicyType.RecreateGroupAfterStart):
            return False
        if not container_restarted(pod) and not pod_deleted(pod):
            return False

        pending = self._pending_pods_in_group(
            pod, lws.spec.leader_worker_template.size)
        has_annotation = lwsapi.RECREATE_GROUP_AFTER_START_ANNOTATION_KEY in \
            (lws.metadata.annotations or {})
        if pending and (policy == lwsapi.RestartPolicyType.RecreateGroupAfterStart
                        or has_annotation):
            return False

        if not leader_pod(pod):
            leader_name, ordinal = parse_parent_and_ordinal(pod.metadata.name)
            if ordinal == -1:
                raise ValueError(f"parsing pod name for pod {pod.metadata.name}")
            leader = self.store.try_get("Pod", pod.metadata.namespace, leader_name)
            if leader is None:
                return False
            if revisionutils.get_revision_key(leader) != \
                    revisionutils.get_revision_key(pod):
                return False
            if not self._worker_pod_belongs_to_leader(pod, leader):
                return False
        else:
            leader = pod
        if leader.metadata.deletion_timestamp is not None:
            return True
        try:
            self.store.delete("Pod", leader.metadata.namespace,
                              leader.metadata.name, propagation="Foreground")
        except NotFoundError:
            pass
        self.record.eventf(
            lws, "Normal", "RecreateGroup",
            f"Worker pod {pod.metadata.name} failed, deleted leader pod "
            f"{leader.metadata.name} to recreate group "
            f"{(leader.metadata.labels or {}).get(lwsapi.GROUP_INDEX_LABEL_KEY, '')}")
        return True

    def _worker_pod_belongs_to_leader(self, pod, leader) -> bool:
        """pod_controller.go:268-295 — stale worker-STS guard."""
        owner = next((r for r in pod.metadata.owner_references if r.controller),
                     None)
        if owner is None:
            return False
        if owner.kind == "Pod":
            return owner.name == leader.metadata.name and \
                owner.uid == leader.metadata.uid
        if owner.kind != "StatefulSet":
            return False
        worker_sts = self.store.try_get("StatefulSet", pod.metadata.namespace,
                                        owner.name)
        if worker_sts is None or worker_sts.metadata.uid != owner.uid:
            return False
        sts_owner = next((r for r in worker_sts.metadata.owner_references
                          if r.controller), None)
        if sts_owner is None:
            return False
        return sts_owner.kind == "Pod" and \
            sts_owner.name == leader.metadata.name and \
            sts_owner.uid == leader.metadata.uid

    def _pending_pods_in_group(self, pod, group_size: int) -> bool:
        """pod_controller.go:333-357."""
        labels = pod.metadata.labels or {}
        pods = self.store.list("Pod", pod.metadata.namespace, label_selector={
            lwsapi.SET_NAME_LABEL_KEY: labels.get(lwsapi.SET_NAME_LABEL_KEY, ""),
            lwsapi.GROUP_INDEX_LABEL_KEY:
                labels.get(lwsapi.GROUP_INDEX_LABEL_KEY, "")})
        if group_size != len(pods):
            return True
        return any(p.status.phase == "Pending" for p in pods)

    def _topology_value_from_pod(self, pod, topology_key: str) -> Optional[str]:
        node = self.node_lookup(pod.node_name)
        if node is None:
            return None
        return node.metadata.labels.get(topology_key)

    # ------------------------------------------------------------------
    def _construct_worker_statefulset(self, leader, lws, revision) -> StatefulSet:
        """pod_controller.go:381-461 — worker STS from the revision-applied
        spec; ordinals start at 1; replicas = size-1."""
        current_lws = revisionutils.apply_revision(lws, revision)
        template = serde.deep_copy(
            current_lws.spec.leader_worker_template.worker_template)
        leader_labels = leader.metadata.labels or {}
        selector_map = {
            lwsapi.GROUP_INDEX_LABEL_KEY:
                leader_labels.get(lwsapi.GROUP_INDEX_LABEL_KEY, ""),
            lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
            lwsapi.GROUP_UNIQUE_HASH_LABEL_KEY:
                leader_labels.get(lwsapi.GROUP_UNIQUE_HASH_LABEL_KEY, "")}
        label_map = dict(selector_map)
        label_map[lwsapi.REVISION_KEY] = revisionutils.get_revision_key(leader)
        template.metadata.labels = _merge_metadata(template.metadata.labels,
                                                   label_map)
        pod_annotations = {
            lwsapi.SIZE_ANNOTATION_KEY:
                str(current_lws.spec.leader_worker_template.size),
            lwsapi.LEADER_POD_NAME_ANNOTATION_KEY: leader.metadata.name}
        lws_ann = lws.metadata.annotations or {}
        if lws_ann.get(lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY):
            pod_annotations[lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY] = \
                lws_ann[lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY]
        sgp = current_lws.spec.leader_worker_template.sub_group_policy
        if sgp is not None:
            if sgp.type is not None:
                pod_annotations[lwsapi.SUBGROUP_POLICY_TYPE_ANNOTATION_KEY] = sgp.type
            pod_annotations[lwsapi.SUBGROUP_SIZE_ANNOTATION_KEY] = \
                str(sgp.sub_group_size)
            if lws_ann.get(lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY):
                pod_annotations[lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY] = \
                    lws_ann[lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY]
        template.metadata.annotations = _merge_metadata(
            template.metadata.annotations, pod_annotations)

        service_name = leader.metadata.name
        nc = current_lws.spec.network_config
        if nc is None or nc.subdomain_policy == lwsapi.SubdomainPolicy.Shared:
            service_name = lws.metadata.name

        sts = StatefulSet()
        sts.metadata.name = leader.metadata.name
        sts.metadata.namespace = leader.metadata.namespace
        sts.metadata.labels = _merge_metadata(lws.metadata.labels, label_map)
        sts.metadata.annotations = dict(lws.metadata.annotations or {})
        sts.metadata.owner_references = [OwnerReference(
            api_version="v1", kind="Pod", name=leader.metadata.name,
            uid=leader.metadata.uid, controller=True,
            block_owner_deletion=True)]
        sts.spec = StatefulSetSpec(
            replicas=current_lws.spec.leader_worker_template.size - 1,
            service_name=service_name,
            pod_management_policy="Parallel",
            template=template,
            ordinals=StatefulSetOrdinals(start=1),
            update_strategy=StatefulSetUpdateStrategy(
                type="RollingUpdate",
                rolling_update=RollingUpdateStatefulSetStrategy(partition=0)),
            selector=LabelSelector(match_labels=selector_map),
            volume_claim_templates=serde.deep_copy(
                current_lws.spec.leader_worker_template.volume_claim_templates),
            persistent_volume_claim_retention_policy=serde.deep_copy(
                current_lws.spec.leader_worker_template
                .persistent_volume_claim_retention_policy))
        return sts

    def _create_headless_service(self, lws, name: str,
                                 selector: dict[str, str], owner) -> None:
        if self.store.try_get("Service", lws.metadata.namespace, name):
            return
        svc = Service()
        svc.metadata.name = name
        svc.metadata.namespace = lws.metadata.namespace
        svc.spec = ServiceSpec(cluster_ip="None", selector=selector,
                               publish_not_ready_addresses=True)
        svc.metadata.owner_references = [OwnerReference(
            api_version="v1", kind="Pod", name=owner.metadata.name,
            uid=owner.metadata.uid, controller=True, block_owner_deletion=True)]
        try:
            self.store.create(svc)
        except AlreadyExistsError:
            pass
