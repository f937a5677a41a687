"""Alibaba Cloud ECS node provider — the RPC-style OpenAPI over HTTP.

Reference: internal/cloudprovider/alibaba/ecs.go (+ pricing). Requests
are signed with the classic ACS RPC signature (HMAC-SHA1 over the
sorted, percent-encoded query string). Endpoint injectable for tests.
"""
from __future__ import annotations

import base64
import datetime
import hashlib
import hmac
import json
import urllib.parse
import uuid
from dataclasses import dataclass, field
from typing import Dict, Optional

import requests

from .provider import GPUNodeProvider

API_VERSION = "2014-05-26"


def _pct(s: str) -> str:
    return urllib.parse.quote(s, safe="~")


def rpc_signature(method: str, params: Dict[str, str],
                  secret: str) -> str:
    """ACS RPC signature: HMAC-SHA1 of the canonicalized query."""

    canon = "&".join(f"{_pct(k)}={_pct(v)}"
                     for k, v in sorted(params.items()))
    to_sign = f"{method}&{_pct('/')}&{_pct(canon)}"
    digest = hmac.new((secret + "&").encode(), to_sign.encode(),
                      hashlib.sha1).digest()
    return base64.b64encode(digest).decode()


@dataclass
class AlibabaConfig:
    region: str = "cn-hangzhou"
    access_key_id: str = ""
    access_key_secret: str = ""
    endpoint: str = ""  # default https://ecs.<region>.aliyuncs.com
    image_id: str = "rocm-mi355x-image"
    vswitch_id: str = ""
    node_labels: Dict[str, str] = field(default_factory=dict)

    @property
    def url(self) -> str:
        return self.endpoint or f"https://ecs.{self.region}.aliyuncs.com"


INSTANCE_TYPE_MAP = {
    "mi355x.1g": "ecs.gn8a.8xlarge",
    "mi355x.2g": "ecs.gn8a.16xlarge",
    "mi355x.4g": "ecs.gn8a.32xlarge",
    "mi355x.8g": "ecs.ebmgn8a.64xlarge",
}

# $/h stand-ins mirroring the reference's static pricing data files
ECS_PRICING = {
    "ecs.gn8a.8xlarge": 4.4,
    "ecs.gn8a.16xlarge": 8.5,
    "ecs.gn8a.32xlarge": 16.4,
    "ecs.ebmgn8a.64xlarge": 31.9,
}


class AlibabaProvider(GPUNodeProvider):
    def __init__(self, cfg: AlibabaConfig):
        self.cfg = cfg
        self._s = requests.Session()

    def _call(self, action: str, params: Dict[str, str]) -> dict:
        base = {
            "Action": action,
            "Version": API_VERSION,
            "Format": "JSON",
            "RegionId": self.cfg.region,
            "AccessKeyId": self.cfg.access_key_id,
            "SignatureMethod": "HMAC-SHA1",
            "SignatureVersion": "1.0",
            "SignatureNonce": uuid.uuid4().hex,
            "Timestamp": datetime.datetime.utcnow().strftime(
                "%Y-%m-%dT%H:%M:%SZ"),
            **params,
        }
        base["Signature"] = rpc_signature(
            "GET", base, self.cfg.access_key_secret)
        r = self._s.get(self.cfg.url, params=base, timeout=30)
        if r.status_code >= 400:
            raise RuntimeError(f"ECS {action}: {r.status_code} "
                               f"{r.text[:300]}")
        return r.json()

    # ------------------------------------------------------- interface

    def create_node(self, claim) -> str:
        itype = INSTANCE_TYPE_MAP.get(
            getattr(claim, "instance_type", "mi355x.8g"),
            getattr(claim, "instance_type", ""))
        params = {
            "ImageId": self.cfg.image_id,
            "InstanceType": itype,
            "Tag.1.Key": "tensor-fusion.ai/managed-by",
            "Tag.1.Value": "tensor-fusion",
        }
        if self.cfg.vswitch_id:
            params["VSwitchId"] = self.cfg.vswitch_id
        out = self._call("CreateInstance", params)
        iid = out.get("InstanceId", "")
        if not iid:
            raise RuntimeError(f"CreateInstance: {json.dumps(out)[:200]}")
        self._call("StartInstance", {"InstanceId": iid})
        return iid

    def terminate_node(self, instance_id: str) -> None:
        self._call("DeleteInstance", {"InstanceId": instance_id,
                                      "Force": "true"})

    def node_status(self, instance_id: str) -> Optional[str]:
        out = self._call("DescribeInstanceStatus",
                         {"InstanceId.1": instance_id})
        statuses = (out.get("InstanceStatuses", {})
                    .get("InstanceStatus", []))
        for st in statuses:
            if st.get("InstanceId") == instance_id:
                if st.get("Status") == "Running":
                    return f"node-{instance_id}"
                return None
        return None

    def price_per_hour(self, instance_type: str) -> Optional[float]:
        return ECS_PRICING.get(instance_type)
