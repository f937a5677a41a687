"""Kubernetes integration layer.

The reference is a controller-runtime operator wired to a real cluster
(cmd/main.go:131-297, config/crd/bases/*.yaml, charts/). This package is
the MI355X stack's equivalent, built directly on the apiserver REST API
(no client-go port):

  serde      dataclass model ⇄ CRD JSON (tensor-fusion.ai/v1)
  crdgen     CRD manifests generated from the dataclass model → deploy/crds
  client     minimal typed REST client (CRUD, PATCH, watch streams,
             subresources, bindings) for kubeconfig or in-cluster auth
  informer   list+watch cache with relist-on-410 and resync
  bridge     K8sStore — the embedded Store write-through-backed by the
             apiserver, so every control-plane component runs unchanged
             against a real cluster
  deviceplugin  kubelet device-plugin (gRPC v1beta1) advertising
             tensor-fusion.ai/index-N
  kubelet_checkpoint  coexistence detector for foreign device plugins
  fake_apiserver  wire-faithful in-process apiserver for tests (this build
             environment has no network, so no kind cluster; the client
             and manifests are written for a real apiserver)
"""
from .client import ApiError, K8sClient  # noqa: F401
from .serde import from_k8s, to_k8s  # noqa: F401
