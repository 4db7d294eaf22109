"""kubeflow_amd — MI355X-native ML platform runtime.

A from-scratch, single-node rebuild of the Kubeflow platform (reference:
google/kubeflow meta-repo) for 8×AMD Instinct MI355X (CDNA4 / gfx950):

- The Kubernetes control plane (kube-apiserver + CRD controllers) collapses
  into an in-process object store (`kubeflow_amd.api`) with watch/pub-sub,
  reconcilers (`kubeflow_amd.controllers`), and a gang scheduler + process
  launcher (`kubeflow_amd.scheduler`) that starts one process per GPU with
  RCCL over xGMI wiring (torch.distributed, backend "nccl" == RCCL on ROCm).
- The workload runtime (`kubeflow_amd.runtime`, `kubeflow_amd.models`,
  `kubeflow_amd.ops`) is the half Kubeflow delegates to sibling repos
  (training-operator, KServe, Katib, Pipelines): PyTorchJob training with
  hand-written CDNA4 HIP kernels for the hot ops, an InferenceService engine
  with dynamic batching, Katib HPO trials and Pipeline DAG runs.

Public API surface mirrors the CRD shapes of the reference
(metadata/spec/status + conditions[], see
/root/reference/components/notebook-controller/api/v1beta1/notebook_types.go:27-71).
"""

__version__ = "0.1.0"
