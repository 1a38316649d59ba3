"""LeaderWorkerSet reconciler.

Behavioral port of reference pkg/controllers/leaderworkerset_controller.go:
revision management, the 5-case rolling-update partition/replica calculus
(incl. maxSurge burst + gradual reclaim), SSA of the leader StatefulSet,
the shared headless Service, and status/conditions.
"""
from __future__ import annotations

from typing import Optional

from ..api import leaderworkerset as lwsapi
from ..api import serde
from ..api.core import (RollingUpdateStatefulSetStrategy, Service,
                        ServiceSpec, StatefulSet, StatefulSetSpec,
                        StatefulSetUpdateStrategy)
from ..api.leaderworkerset import LeaderWorkerSet
from ..api.meta import (LabelSelector, OwnerReference, format_label_selector,
                        get_int_or_percent, new_condition)
from ..cluster.controller import Controller, Manager
from ..cluster.statefulset_controller import statefulset_ready
from ..cluster.store import AlreadyExistsError, Store
from ..utils import revision as revisionutils
from ..utils.podutils import pod_running_and_ready

FIELD_MANAGER = "lws"


def _merge_metadata(base: dict[str, str], overrides: dict[str, str]) -> dict[str, str]:
    """pkg/controllers/metadata.go:21-30."""
    out = dict(base or {})
    out.update(overrides)
    return out


def make_condition(cond_type: str, lws: LeaderWorkerSet):
    if cond_type == lwsapi.LeaderWorkerSetConditionType.Available:
        return new_condition("Available", "True", "AllGroupsReady",
                             "All replicas are ready",
                             lws.metadata.generation)
    if cond_type == lwsapi.LeaderWorkerSetConditionType.UpdateInProgress:
        return new_condition("UpdateInProgress", "True", "GroupsUpdating",
                             "Rolling Upgrade is in progress",
                             lws.metadata.generation)
    return new_condition("Progressing", "True", "GroupsProgressing",
                         "Replicas are progressing",
                         lws.metadata.generation)


def set_conditions(lws: LeaderWorkerSet, conditions) -> bool:
    """Exclusive condition handling: Available/Progressing are mutually
    exclusive; UpdateInProgress cleared when rollout done (mirrors
    setCondition/exclusiveConditionTypes in the reference)."""
    changed = False
    exclusive = {"Available", "Progressing"}
    for cond in conditions:
        existing = {c.type: c for c in lws.status.conditions}
        cur = existing.get(cond.type)
        if cur is not None and cur.status == cond.status and \
                cur.reason == cond.reason:
            continue
        new_list = [c for c in lws.status.conditions if c.type != cond.type]
        if cond.type in exclusive:
            other = exclusive - {cond.type}
            new_list = [c for c in new_list
                        if not (c.type in other and c.status == "True")]
        new_list.append(cond)
        lws.status.conditions = new_list
        changed = True
    cond_types = {c.type for c in conditions}
    if lwsapi.LeaderWorkerSetConditionType.Available in cond_types:
        before = len(lws.status.conditions)
        lws.status.conditions = [c for c in lws.status.conditions
                                 if c.type != "UpdateInProgress"]
        if len(lws.status.conditions) != before:
            changed = True
    return changed


def non_zero(v: int) -> int:
    return max(0, v)


def calculate_rolling_update_replicas(lws_replicas: int, max_surge: int,
                                      max_unavailable: int,
                                      unready_replicas: int) -> int:
    """leaderworkerset_controller.go:691-702."""
    burst = lws_replicas + max_surge
    if unready_replicas <= max_surge:
        required_surge = non_zero(unready_replicas - max_unavailable)
        return lws_replicas + required_surge
    return burst


def calculate_continuous_ready_replicas(states) -> int:
    """leaderworkerset_controller.go:704-713 — ready+updated tail length."""
    count = 0
    for ready, updated in reversed(states):
        if not ready or not updated:
            break
        count += 1
    return count


def rolling_update_partition(states, sts_replicas: int, rolling_step: int,
                             current_partition: int) -> int:
    """leaderworkerset_controller.go:649-679 — monotonic partition with
    maxUnavailable accounting and stuck-rollout unblocking."""
    continuous_ready = calculate_continuous_ready_replicas(states)
    rolling_step_partition = non_zero(sts_replicas - continuous_ready - rolling_step)
    unavailable = sum(1 for idx in range(rolling_step_partition)
                      if not states[idx][0])
    partition = rolling_step_partition + unavailable
    idx = min(partition, sts_replicas - 1)
    while idx >= rolling_step_partition:
        ready, updated = states[idx]
        if not ready or updated:
            partition = idx
        else:
            break
        idx -= 1
    return min(partition, current_partition)


def calculate_lws_unready_replicas(states, lws_replicas: int) -> int:
    """leaderworkerset_controller.go:681-689."""
    unready = 0
    for idx in range(lws_replicas):
        if idx >= len(states) or not states[idx][0] or not states[idx][1]:
            unready += 1
    return unready


def sort_by_index(items, index_fn, length: int):
    """pkg/utils/utils.go SortByIndex — fixed-length placement by index."""
    out = [None] * length
    for item in items:
        try:
            idx = index_fn(item)
        except (TypeError, ValueError):
            continue
        if 0 <= idx < length:
            out[idx] = item
    return out


class LeaderWorkerSetReconciler:
    def __init__(self, manager: Manager, recorder=None) -> None:
        from ..cluster.events import NullRecorder
        self.store: Store = manager.store
        self.record = recorder or NullRecorder()
        self.ctrl = Controller("leaderworkerset", self.reconcile)
        self._revision_cache: dict = {}
        manager.add_controller(self.ctrl)
        manager.watch(lwsapi.KIND, self.ctrl)
        manager.watch("StatefulSet", self.ctrl, self._map_owned)
        manager.watch("Pod", self.ctrl, self._map_pod)

    def _map_owned(self, event: str, obj) -> list[tuple[str, str]]:
        name = (obj.metadata.labels or {}).get(lwsapi.SET_NAME_LABEL_KEY)
        if name:
            return [(obj.metadata.namespace, name)]
        return []

    def _map_pod(self, event: str, pod) -> list[tuple[str, str]]:
        labels = pod.metadata.labels or {}
        name = labels.get(lwsapi.SET_NAME_LABEL_KEY)
        if name and labels.get(lwsapi.WORKER_INDEX_LABEL_KEY) == "0":
            return [(pod.metadata.namespace, name)]
        return []

    # ------------------------------------------------------------------
    def reconcile(self, namespace: str, name: str) -> Optional[float]:
        lws = self.store.try_get(lwsapi.KIND, namespace, name)
        if lws is None or lws.metadata.deletion_timestamp is not None:
            return None

        leader_sts = self.store.try_get("StatefulSet", namespace, name)
        if leader_sts is not None and leader_sts.metadata.deletion_timestamp is not None:
            return 5.0

        revision = self._get_or_create_revision(leader_sts, lws)
        updated_revision = self._get_updated_revision(leader_sts, lws, revision)
        lws_updated = updated_revision is not None
        if lws_updated:
            revision = revisionutils.create_revision(self.store, updated_revision)
            self.record.eventf(lws, "Normal", "CreatingRevision",
                               f"Creating revision with key "
                               f"{revisionutils.get_revision_key(revision)} "
                               "for updated LWS")
        revision_key = revisionutils.get_revision_key(revision)

        partition, replicas = self._rolling_update_parameters(
            lws, leader_sts, revision_key, lws_updated)

        old_partition = None
        if leader_sts is not None and \
                leader_sts.spec.update_strategy.rolling_update is not None:
            old_partition = leader_sts.spec.update_strategy.rolling_update.partition
        self._ssa_leader_statefulset(lws, partition, replicas, revision_key)
        if leader_sts is None:
            self.record.eventf(lws, "Normal", "GroupsProgressing",
                               f"Created leader statefulset {lws.metadata.name}")
        elif not lws_updated and old_partition is not None and \
                partition != old_partition:
            self.record.eventf(lws, "Normal", "GroupsUpdating",
                               f"Updating replicas {partition} to "
                               f"{old_partition - 1} (inclusive)")
        self._reconcile_headless_services(lws)

        update_done = self._update_status(lws, revision_key)
        if update_done:
            revisionutils.truncate_revisions(self.store, lws, revision_key)
        return None

    # ------------------------------------------------------------------
    def _get_or_create_revision(self, leader_sts, lws):
        """leaderworkerset_controller.go:728-751."""
        revision_key = ""
        if leader_sts is not None:
            revision_key = revisionutils.get_revision_key(leader_sts)
        existing = revisionutils.get_revision(self.store, lws, revision_key)
        if existing is not None:
            return existing
        revision = revisionutils.new_revision(self.store, lws, revision_key)
        return revisionutils.create_revision(self.store, revision)

    def _get_updated_revision(self, leader_sts, lws, revision):
        """leaderworkerset_controller.go:753-772."""
        if leader_sts is None:
            return None
        current = revisionutils.new_revision(self.store, lws, "")
        if not revisionutils.equal_revision(current, revision):
            if revisionutils.set_matches_revision(lws, current, revision,
                                                  self._revision_cache):
                return None
            return current
        return None

    # ------------------------------------------------------------------
    def _rolling_update_parameters(self, lws, sts, revision_key: str,
                                   lws_updated: bool) -> tuple[int, int]:
        """leaderworkerset_controller.go:286-379 — 5-case calculus."""
        lws_replicas = lws.spec.replicas
        ruc = lws.spec.rollout_strategy.rolling_update_configuration
        lws_partition = ruc.partition or 0

        def clamp(partition: int, replicas: int) -> tuple[int, int]:
            return max(partition, lws_partition), replicas

        # Case 1: leader sts not created yet
        if sts is None:
            return clamp(0, lws_replicas)

        sts_replicas = sts.spec.replicas
        max_surge = get_int_or_percent(ruc.max_surge, lws_replicas, True)
        max_unavailable = get_int_or_percent(ruc.max_unavailable, lws_replicas, False)
        max_surge = min(max_surge, lws_replicas)
        burst_replicas = lws_replicas + max_surge

        def want_replicas(unready: int) -> int:
            return calculate_rolling_update_replicas(
                lws_replicas, max_surge, max_unavailable, unready)

        # Case 2: new rolling update
        if lws_updated:
            partition = min(lws_replicas, sts_replicas)
            if sts_replicas < lws_replicas:
                return clamp(partition, lws_replicas)
            return clamp(partition, want_replicas(lws_replicas))

        partition = 0
        if sts.spec.update_strategy.rolling_update is not None:
            partition = sts.spec.update_strategy.rolling_update.partition or 0
        rolling_update_completed = partition == 0 and sts_replicas == lws_replicas
        # Case 3: steady state
        if rolling_update_completed:
            return clamp(0, lws_replicas)
        if sts_replicas < lws_replicas:
            return clamp(partition, lws_replicas)

        states = self._get_replica_states(lws, sts_replicas, revision_key)
        unready = calculate_lws_unready_replicas(states, lws_replicas)

        original = int((sts.metadata.annotations or {}).get(
            lwsapi.REPLICAS_ANNOTATION_KEY, lws_replicas))
        # Case 4: replicas changed during rolling update
        if original != lws_replicas:
            partition = min(partition, burst_replicas)
            return clamp(partition, want_replicas(unready))

        # Case 5: advance the partition
        rolling_step = max_unavailable
        rolling_step += max_surge - (burst_replicas - sts_replicas)
        partition = rolling_update_partition(states, sts_replicas,
                                             rolling_step, partition)
        return clamp(partition, want_replicas(unready))

    def _get_replica_states(self, lws, sts_replicas: int,
                            revision_key: str) -> list[tuple[bool, bool]]:
        """leaderworkerset_controller.go:582-647 — per-index (ready, updated)."""
        ns = lws.metadata.namespace
        leader_pods = self.store.list("Pod", ns, label_selector={
            lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
            lwsapi.WORKER_INDEX_LABEL_KEY: "0"}, copy=False)
        sorted_pods = sort_by_index(
            leader_pods,
            lambda p: int(p.metadata.labels[lwsapi.GROUP_INDEX_LABEL_KEY]),
            sts_replicas)
        sts_list = self.store.list("StatefulSet", ns, label_selector={
            lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name}, copy=False)
        sorted_sts = sort_by_index(
            sts_list,
            lambda s: int((s.metadata.labels or {}).get(
                lwsapi.GROUP_INDEX_LABEL_KEY, "x")),
            sts_replicas)
        no_worker_sts = lws.spec.leader_worker_template.size == 1

        states = []
        for idx in range(sts_replicas):
            nominated = f"{lws.metadata.name}-{idx}"
            pod = sorted_pods[idx]
            sts = sorted_sts[idx]
            if pod is None or pod.metadata.name != nominated or \
                    (not no_worker_sts and
                     (sts is None or sts.metadata.name != nominated)):
                states.append((False, False))
                continue
            leader_updated = revisionutils.get_revision_key(pod) == revision_key
            leader_ready = pod_running_and_ready(pod)
            if no_worker_sts:
                states.append((leader_ready, leader_updated))
                continue
            workers_updated = revisionutils.get_revision_key(sts) == revision_key
            workers_ready = statefulset_ready(sts)
            states.append((leader_ready and workers_ready,
                           leader_updated and workers_updated))
        return states

    # ------------------------------------------------------------------
    def _ssa_leader_statefulset(self, lws, partition: int, replicas: int,
                                revision_key: str) -> None:
        """constructLeaderStatefulSetApplyConfiguration + SSA
        (leaderworkerset_controller.go:381-417, 775-876)."""
        t = lws.spec.leader_worker_template
        template_src = t.leader_template if t.leader_template is not None \
            else t.worker_template
        template = serde.deep_copy(template_src)
        template.metadata.labels = _merge_metadata(
            template.metadata.labels,
            {lwsapi.WORKER_INDEX_LABEL_KEY: "0",
             lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
             lwsapi.REVISION_KEY: revision_key})
        pod_annotations = {lwsapi.SIZE_ANNOTATION_KEY: str(t.size)}
        lws_ann = lws.metadata.annotations or {}
        if lws_ann.get(lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY):
            pod_annotations[lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY] = \
                lws_ann[lwsapi.EXCLUSIVE_KEY_ANNOTATION_KEY]
        if t.sub_group_policy is not None:
            pod_annotations[lwsapi.SUBGROUP_POLICY_TYPE_ANNOTATION_KEY] = \
                t.sub_group_policy.type or lwsapi.SubGroupPolicyType.LeaderWorker
            pod_annotations[lwsapi.SUBGROUP_SIZE_ANNOTATION_KEY] = \
                str(t.sub_group_policy.sub_group_size)
            if lws_ann.get(lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY):
                pod_annotations[lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY] = \
                    lws_ann[lwsapi.SUBGROUP_EXCLUSIVE_KEY_ANNOTATION_KEY]
        if lws.spec.network_config is not None and \
                lws.spec.network_config.subdomain_policy == \
                lwsapi.SubdomainPolicy.UniquePerReplica:
            pod_annotations[lwsapi.SUBDOMAIN_POLICY_ANNOTATION_KEY] = \
                lwsapi.SubdomainPolicy.UniquePerReplica
        if lws_ann.get(lwsapi.RECREATE_GROUP_AFTER_START_ANNOTATION_KEY):
            pod_annotations[lwsapi.RECREATE_GROUP_AFTER_START_ANNOTATION_KEY] = \
                lws_ann[lwsapi.RECREATE_GROUP_AFTER_START_ANNOTATION_KEY]
        template.metadata.annotations = _merge_metadata(
            template.metadata.annotations, pod_annotations)

        lws_replicas = lws.spec.replicas
        ruc = lws.spec.rollout_strategy.rolling_update_configuration
        mu = get_int_or_percent(ruc.max_unavailable, lws_replicas, False)
        ms = min(get_int_or_percent(ruc.max_surge, lws_replicas, True),
                 lws_replicas)
        sts_max_unavailable = max(1, mu + ms)

        sts = StatefulSet()
        sts.metadata.name = lws.metadata.name
        sts.metadata.namespace = lws.metadata.namespace
        sts.metadata.labels = _merge_metadata(
            lws.metadata.labels,
            {lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
             lwsapi.REVISION_KEY: revision_key})
        sts.metadata.annotations = _merge_metadata(
            lws.metadata.annotations,
            {lwsapi.REPLICAS_ANNOTATION_KEY: str(lws.spec.replicas)})
        sts.metadata.owner_references = [OwnerReference(
            api_version=lws.api_version, kind=lws.kind, name=lws.metadata.name,
            uid=lws.metadata.uid, controller=True, block_owner_deletion=True)]
        sts.spec = StatefulSetSpec(
            replicas=replicas,
            service_name=lws.metadata.name,
            pod_management_policy="Parallel",
            template=template,
            update_strategy=StatefulSetUpdateStrategy(
                type=lws.spec.rollout_strategy.type,
                rolling_update=RollingUpdateStatefulSetStrategy(
                    partition=partition,
                    max_unavailable=sts_max_unavailable)),
            selector=LabelSelector(match_labels={
                lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
                lwsapi.WORKER_INDEX_LABEL_KEY: "0"}),
            volume_claim_templates=serde.deep_copy(t.volume_claim_templates),
            persistent_volume_claim_retention_policy=serde.deep_copy(
                t.persistent_volume_claim_retention_policy))
        self.store.apply(sts, field_manager=FIELD_MANAGER)

    def _reconcile_headless_services(self, lws) -> None:
        """CreateHeadlessServiceIfNotExists for Shared subdomain
        (controller_utils.go:33-65)."""
        if lws.spec.network_config is not None and \
                lws.spec.network_config.subdomain_policy == \
                lwsapi.SubdomainPolicy.UniquePerReplica:
            return
        if self.store.try_get("Service", lws.metadata.namespace,
                              lws.metadata.name) is not None:
            return
        svc = Service()
        svc.metadata.name = lws.metadata.name
        svc.metadata.namespace = lws.metadata.namespace
        svc.spec = ServiceSpec(
            cluster_ip="None",
            selector={lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name},
            publish_not_ready_addresses=True)
        svc.metadata.owner_references = [OwnerReference(
            api_version=lws.api_version, kind=lws.kind, name=lws.metadata.name,
            uid=lws.metadata.uid, controller=True, block_owner_deletion=True)]
        try:
            self.store.create(svc)
        except AlreadyExistsError:
            pass

    # ------------------------------------------------------------------
    def _update_status(self, lws, revision_key: str) -> bool:
        """updateStatus + updateConditions
        (leaderworkerset_controller.go:420-573)."""
        ns = lws.metadata.namespace
        sts = self.store.try_get("StatefulSet", ns, lws.metadata.name)
        if sts is None:
            return False
        lws = self.store.try_get(lwsapi.KIND, ns, lws.metadata.name)
        if lws is None:
            return False
        update_status = False
        if lws.status.replicas != sts.status.replicas:
            lws.status.replicas = sts.status.replicas
            update_status = True
        if lws.status.observed_generation != lws.metadata.generation:
            lws.status.observed_generation = lws.metadata.generation
            update_status = True
        if not lws.status.hpa_pod_selector:
            lws.status.hpa_pod_selector = format_label_selector(LabelSelector(
                match_labels={lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
                              lwsapi.WORKER_INDEX_LABEL_KEY: "0"}))
            update_status = True

        update_conditions, update_done = self._update_conditions(lws, revision_key)
        if update_status or update_conditions:
            self.store.update_status(lws)
        return update_done

    def _update_conditions(self, lws, revision_key: str) -> tuple[bool, bool]:
        """leaderworkerset_controller.go:420-513."""
        ns = lws.metadata.namespace
        leader_pods = self.store.list("Pod", ns, label_selector={
            lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name,
            lwsapi.WORKER_INDEX_LABEL_KEY: "0"}, copy=False)
        # one indexed bulk fetch of the worker STSes instead of a deep-
        # copying try_get per replica (read-only lister refs)
        sts_by_name = {
            o.metadata.name: o
            for o in self.store.list("StatefulSet", ns, label_selector={
                lwsapi.SET_NAME_LABEL_KEY: lws.metadata.name}, copy=False)}
        ready_count = updated_count = ready_non_burst = 0
        part_updated_non_burst = part_current_non_burst = part_updated_ready = 0
        no_worker_sts = lws.spec.leader_worker_template.size == 1
        lws_partition = (lws.spec.rollout_strategy
                         .rolling_update_configuration.partition or 0)
        lws_replicas = lws.spec.replicas

        for pod in leader_pods:
            try:
                index = int(pod.metadata.labels[lwsapi.GROUP_INDEX_LABEL_KEY])
            except (KeyError, ValueError):
                continue
            sts = None
            if not no_worker_sts:
                sts = sts_by_name.get(pod.metadata.name)
                if sts is None:
                    continue
            if index < lws_replicas and index >= lws_partition:
                part_current_non_burst += 1
            ready = updated = False
            if (no_worker_sts or statefulset_ready(sts)) and \
                    pod_running_and_ready(pod):
                ready = True
                ready_count += 1
            if (no_worker_sts or revisionutils.get_revision_key(sts) ==
                    revision_key) and \
                    revisionutils.get_revision_key(pod) == revision_key:
                updated = True
                updated_count += 1
                if index < lws_replicas and index >= lws_partition:
                    part_updated_non_burst += 1
            if index < lws_replicas:
                if ready:
                    ready_non_burst += 1
                if index >= lws_partition and ready and updated:
                    part_updated_ready += 1

        update_status = False
        if lws.status.ready_replicas != ready_count:
            lws.status.ready_replicas = ready_count
            update_status = True
        if lws.status.updated_replicas != updated_count:
            lws.status.updated_replicas = updated_count
            update_status = True

        conditions = []
        if part_updated_non_burst < part_current_non_burst:
            conditions.append(make_condition(
                lwsapi.LeaderWorkerSetConditionType.UpdateInProgress, lws))
            conditions.append(make_condition(
                lwsapi.LeaderWorkerSetConditionType.Progressing, lws))
        elif ready_non_burst == lws_replicas and \
                part_updated_ready == part_current_non_burst:
            conditions.append(make_condition(
                lwsapi.LeaderWorkerSetConditionType.Available, lws))
        else:
            conditions.append(make_condition(
                lwsapi.LeaderWorkerSetConditionType.Progressing, lws))

        update_done = (lws_partition == 0 and
                       part_updated_ready == lws_replicas)
        update_condition = set_conditions(lws, conditions)
        return update_status or update_condition, update_done
