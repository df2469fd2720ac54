"""rbg-mi355x — MI355X-native disaggregated-serving orchestrator.

A from-scratch framework with the capabilities of sgl-project/rbg
(RoleBasedGroup): the v1alpha2 object model and `rbg.workloads.x-k8s.io`
contract, reconciled directly onto the GPUs of one MI355X node, with
PyTorch-ROCm + hand-written CDNA4 HIP-kernel serving engines underneath.
"""
__version__ = "0.1.0"
