"""AWS EC2 node provider — the EC2 Query API over plain HTTP.

Reference: internal/cloudprovider/aws/ec2.go (RunInstances /
TerminateInstances / DescribeInstances via the AWS SDK). No boto3 in
this image, so the provider signs requests itself (SigV4, hmac/sha256
from the stdlib) and speaks the Query protocol directly — the endpoint
is injectable, which is also how the tests run a wire-level fake.
"""
from __future__ import annotations

import datetime
import hashlib
import hmac
import urllib.parse
import xml.etree.ElementTree as ET
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import requests

from .provider import GPUNodeProvider

API_VERSION = "2016-11-15"


# ------------------------------------------------------------- signing


def _sign(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode(), hashlib.sha256).digest()


def sigv4_headers(method: str, url: str, body: str, region: str,
                  access_key: str, secret_key: str,
                  service: str = "ec2",
                  now: Optional[datetime.datetime] = None) -> Dict[str, str]:
    """AWS Signature Version 4 for a form-encoded POST."""

    u = urllib.parse.urlparse(url)
    host = u.netloc
    now = now or datetime.datetime.utcnow()
    amz_date = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")
    payload_hash = hashlib.sha256(body.encode()).hexdigest()
    canonical_headers = (f"content-type:application/x-www-form-urlencoded\n"
                         f"host:{host}\nx-amz-date:{amz_date}\n")
    signed_headers = "content-type;host;x-amz-date"
    canonical_request = "\n".join([
        method, u.path or "/", "", canonical_headers, signed_headers,
        payload_hash])
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    string_to_sign = "\n".join([
        "AWS4-HMAC-SHA256", amz_date, scope,
        hashlib.sha256(canonical_request.encode()).hexdigest()])
    k = _sign(("AWS4" + secret_key).encode(), datestamp)
    k = _sign(k, region)
    k = _sign(k, service)
    k = _sign(k, "aws4_request")
    signature = hmac.new(k, string_to_sign.encode(),
                         hashlib.sha256).hexdigest()
    auth = (f"AWS4-HMAC-SHA256 Credential={access_key}/{scope}, "
            f"SignedHeaders={signed_headers}, Signature={signature}")
    return {"Content-Type": "application/x-www-form-urlencoded",
            "X-Amz-Date": amz_date, "Authorization": auth}


# ------------------------------------------------------------- provider


@dataclass
class AwsConfig:
    region: str = "us-east-1"
    access_key: str = ""
    secret_key: str = ""
    endpoint: str = ""  # default https://ec2.<region>.amazonaws.com
    ami: str = "ami-rocm-mi355x"
    subnet_id: str = ""
    security_group: str = ""
    node_labels: Dict[str, str] = field(default_factory=dict)

    @property
    def url(self) -> str:
        return self.endpoint or f"https://ec2.{self.region}.amazonaws.com"


# reference's instance-type mapping for GPU nodes (ec2.go instance table)
INSTANCE_TYPE_MAP = {
    "mi355x.1g": "g7a.4xlarge",
    "mi355x.2g": "g7a.8xlarge",
    "mi355x.4g": "g7a.24xlarge",
    "mi355x.8g": "p6a.48xlarge",
}


class AwsProvider(GPUNodeProvider):
    def __init__(self, cfg: AwsConfig):
        self.cfg = cfg
        self._s = requests.Session()

    # ---------------------------------------------------------- query

    def _call(self, action: str, params: Dict[str, str]) -> ET.Element:
        body_params = {"Action": action, "Version": API_VERSION, **params}
        body = urllib.parse.urlencode(sorted(body_params.items()))
        headers = sigv4_headers("POST", self.cfg.url, body,
                                self.cfg.region, self.cfg.access_key,
                                self.cfg.secret_key)
        r = self._s.post(self.cfg.url, data=body, headers=headers,
                         timeout=30)
        if r.status_code >= 400:
            raise RuntimeError(f"EC2 {action}: {r.status_code} "
                               f"{r.text[:300]}")
        # EC2 XML namespaces vary by version: strip them
        text = r.text.replace(
            f'xmlns="http://ec2.amazonaws.com/doc/{API_VERSION}/"', "")
        return ET.fromstring(text)

    # ------------------------------------------------------- interface

    def create_node(self, claim) -> str:
        itype = INSTANCE_TYPE_MAP.get(
            getattr(claim, "instance_type", "mi355x.8g"),
            getattr(claim, "instance_type", ""))
        params = {
            "ImageId": self.cfg.ami,
            "InstanceType": itype,
            "MinCount": "1",
            "MaxCount": "1",
            "TagSpecification.1.ResourceType": "instance",
            "TagSpecification.1.Tag.1.Key": "tensor-fusion.ai/managed-by",
            "TagSpecification.1.Tag.1.Value": "tensor-fusion",
            "TagSpecification.1.Tag.2.Key": "tensor-fusion.ai/claim",
            "TagSpecification.1.Tag.2.Value": getattr(claim, "name", "")
            or getattr(getattr(claim, "meta", None), "name", ""),
        }
        if self.cfg.subnet_id:
            params["SubnetId"] = self.cfg.subnet_id
        if self.cfg.security_group:
            params["SecurityGroupId.1"] = self.cfg.security_group
        root = self._call("RunInstances", params)
        iid = root.findtext(".//instanceId")
        if not iid:
            raise RuntimeError("RunInstances returned no instanceId")
        return iid

    def terminate_node(self, instance_id: str) -> None:
        self._call("TerminateInstances", {"InstanceId.1": instance_id})

    def node_status(self, instance_id: str) -> Optional[str]:
        root = self._call("DescribeInstances",
                          {"InstanceId.1": instance_id})
        state = root.findtext(".//instanceState/name") or ""
        if state != "running":
            return None
        # the node joins the cluster under its private DNS name
        return (root.findtext(".//privateDnsName")
                or f"node-{instance_id}")

    def list_managed_instances(self) -> List[str]:
        root = self._call("DescribeInstances", {
            "Filter.1.Name": "tag:tensor-fusion.ai/managed-by",
            "Filter.1.Value.1": "tensor-fusion",
        })
        return [e.text for e in root.findall(".//instanceId") if e.text]
