"""Cloud node provisioning.

Reference: internal/cloudprovider/ — GPUNodeProvider interface
(types/type.go:23-33: Create/Terminate/GetNodeStatus + pricing), AWS EC2,
Alibaba ECS and Karpenter NodeClaim implementations, and a mock for
tests. The MI355X build keeps the interface and the mock/karpenter-shaped
flows; real cloud SDK calls are deployment plumbing, represented by the
same provider contract.
"""
from .provider import (GPUNodeProvider, InstanceType, MockProvider,
                       PRICING_TABLE, cheapest_instance_for)

__all__ = ["GPUNodeProvider", "InstanceType", "MockProvider",
           "PRICING_TABLE", "cheapest_instance_for"]
