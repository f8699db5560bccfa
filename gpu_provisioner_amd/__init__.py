"""gpu_provisioner_amd — an MI355X-first Karpenter cloud-provider controller for AKS.

A brand-new, from-scratch implementation of the capability surface of
Azure/gpu-provisioner: a Kubernetes controller that implements
the Karpenter ``CloudProvider`` contract over the ``karpenter.sh/v1 NodeClaim``
CRD and materializes each kaito-labeled NodeClaim as a single-VM AKS agent
pool — with the GPU surface AMD-native from the start (MI355X SKU catalog,
ROCm/amdgpu bootstrap, ``amd.com/gpu`` initialization gating, and a HIP
node-agent for on-node GPU health validation).

Unlike the reference (a patched vendored fork of karpenter-core in Go), the
controller runtime here — typed kube client, informers, rate-limited
workqueues, leader election, metrics, health probes — is implemented
first-class and sized to the problem.
"""

__version__ = "0.1.0"
