"""Cloud node provisioning.

Reference: internal/cloudprovider/ — GPUNodeProvider interface
(types/type.go:23-33: Create/Terminate/GetNodeStatus + pricing) with
AWS EC2 (SigV4 Query API, aws.py), Alibaba ECS (ACS RPC signature,
alibaba.py), Karpenter NodeClaim choreography (karpenter.py) and the
mock used by tests/the expander. No cloud SDKs exist in this image, so
the AWS/Alibaba providers speak the wire protocols directly with
hand-rolled signing — endpoints are injectable, which is also how the
wire-level fakes in tests/test_cloudproviders.py exercise them.
"""
from .alibaba import AlibabaConfig, AlibabaProvider
from .aws import AwsConfig, AwsProvider
from .karpenter import KarpenterProvider
from .provider import (GPUNodeProvider, InstanceType, MockProvider,
                       PRICING_TABLE, cheapest_instance_for)

__all__ = ["GPUNodeProvider", "InstanceType", "MockProvider",
           "PRICING_TABLE", "cheapest_instance_for",
           "AwsProvider", "AwsConfig", "AlibabaProvider", "AlibabaConfig",
           "KarpenterProvider"]
