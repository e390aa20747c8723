"""MI355X kubelet device plugin.

Role parity with the reference's pkg/gpu/nvidia (manager.go, beta_plugin.go,
gpusharing/, mig/, health_check/, metrics/, version_visibility/) re-designed
for AMD: /dev/kfd + /dev/dri/renderD* device nodes, amd.com/gpu resource,
SPX/DPX/QPX/CPX compute partitions instead of MIG, AMD-SMI RAS/ECC/xGMI events
instead of Xids, and CU-mask env fencing instead of CUDA MPS.
"""

RESOURCE_NAME = "amd.com/gpu"  # parity: resourceName, manager.go:67
