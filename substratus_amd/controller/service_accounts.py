"""ServiceAccount reconciler (reference internal/controller/
service_accounts_controller.go:16-66): five well-known SAs, created/updated
with the cloud principal annotation; BindIdentity via SCI when the principal
is not yet bound."""

from __future__ import annotations

from ..cloud.base import Cloud
from ..kube.memapi import MemKube

CONTAINER_BUILDER_SA = "container-builder"
MODELLER_SA = "modeller"
MODEL_SERVER_SA = "model-server"
NOTEBOOK_SA = "notebook"
DATA_LOADER_SA = "data-loader"

ALL_SAS = (CONTAINER_BUILDER_SA, MODELLER_SA, MODEL_SERVER_SA, NOTEBOOK_SA,
           DATA_LOADER_SA)


def reconcile_service_account(kube: MemKube, cloud: Cloud, sci, name: str,
                              namespace: str = "default") -> dict:
    sa = kube.try_get("ServiceAccount", name, namespace) or {
        "apiVersion": "v1", "kind": "ServiceAccount",
        "metadata": {"name": name, "namespace": namespace},
    }
    principal, bound = cloud.get_principal(sa)  # before annotating
    cloud.associate_principal(sa)
    if principal and not bound and sci is not None:
        sci.bind_identity(ksa=name, namespace=namespace, principal=principal)
    return kube.apply(sa)
