"""Pod → VariantAutoscaling mapping.

Parity: reference internal/collector/source/pod_va_mapper.go:33-100 — walks
the owner chain Pod → ReplicaSet → Deployment and returns the VA whose
scaleTargetRef matches. This implementation also falls back to matching
pod labels against deployment selectors (our fake-cluster pods are owned
directly by Deployments; real clusters interpose a ReplicaSet).
"""
from __future__ import annotations

from typing import Dict, Optional

from ..kube.fake import FakeCluster
from ..kube.objects import Deployment, Pod


class PodVAMapper:
    def __init__(self, cluster: FakeCluster):
        self.cluster = cluster

    def find_va_for_pod(
        self,
        pod_name: str,
        namespace: str,
        deployments: Dict[str, Deployment],
    ) -> str:
        """Return the variant (deployment/VA) name owning this pod, or ""."""
        pod: Optional[Pod] = self.cluster.try_get("Pod", namespace, pod_name)
        if pod is None:
            # Pod not in cluster view (e.g. emulated metrics): fall back to
            # name-prefix matching against the deployment set.
            for _, deploy in deployments.items():
                if pod_name.startswith(deploy.name + "-"):
                    return deploy.name
            return ""

        # Owner chain: Pod → (ReplicaSet →) Deployment
        owner = self._resolve_deployment_owner(pod, namespace)
        if owner and any(d.name == owner for d in deployments.values()):
            return owner

        # Selector match fallback
        for _, deploy in deployments.items():
            if deploy.namespace != namespace:
                continue
            sel = deploy.selector or deploy.template.labels
            if sel and all(
                pod.metadata.labels.get(k) == v for k, v in sel.items()
            ):
                return deploy.name
        return ""

    def _resolve_deployment_owner(self, pod: Pod, namespace: str) -> str:
        for ref in pod.metadata.owner_references:
            kind = ref.get("kind", "")
            name = ref.get("name", "")
            if kind == "Deployment":
                return name
            if kind == "ReplicaSet":
                rs = self.cluster.try_get("ReplicaSet", namespace, name)
                if rs is not None:
                    for rref in rs.metadata.owner_references:
                        if rref.get("kind") == "Deployment":
                            return rref.get("name", "")
                # ReplicaSet name convention: <deployment>-<hash>
                if "-" in name:
                    return name.rsplit("-", 1)[0]
        return ""
