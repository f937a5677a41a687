"""Cloud-provider implementations against wire-level fakes.

Reference: internal/cloudprovider/{aws/ec2.go, alibaba/ecs.go,
karpenter/nodeclaim.go}. The fakes verify the real wire artifacts —
SigV4 signatures recompute on the EC2 side, the ACS RPC signature
recomputes on the ECS side — so a correct fake pass implies a correct
request against the real clouds.
"""
import threading
import urllib.parse

import pytest
from fastapi import FastAPI, Request, Response

from tensor_fusion_amd.cloudprovider import (AlibabaConfig, AlibabaProvider,
                                             AwsConfig, AwsProvider,
                                             KarpenterProvider)
from tensor_fusion_amd.cloudprovider.alibaba import rpc_signature
from tensor_fusion_amd.cloudprovider.aws import sigv4_headers


class _Claim:
    instance_type = "mi355x.8g"
    name = "claim-1"
    gpu_count = 8


def _serve(app):
    import socket
    import time

    import requests
    import uvicorn
    with socket.socket() as sk:
        sk.bind(("127.0.0.1", 0))
        port = sk.getsockname()[1]
    srv = uvicorn.Server(uvicorn.Config(app, host="127.0.0.1", port=port,
                                        log_level="error"))
    threading.Thread(target=srv.run, daemon=True).start()
    for _ in range(100):
        try:
            requests.get(f"http://127.0.0.1:{port}/__nope", timeout=1)
            break
        except Exception:
            time.sleep(0.02)
    return f"http://127.0.0.1:{port}", srv


# ------------------------------------------------------------------ AWS


@pytest.fixture()
def fake_ec2():
    app = FastAPI()
    state = {"instances": {}, "last_auth": None, "bodies": []}

    @app.post("/")
    async def handler(request: Request):
        body = (await request.body()).decode()
        state["last_auth"] = request.headers.get("authorization", "")
        state["bodies"].append(body)
        params = dict(urllib.parse.parse_qsl(body))
        action = params.get("Action")
        if action == "RunInstances":
            iid = f"i-{len(state['instances']):08x}"
            state["instances"][iid] = {
                "type": params["InstanceType"], "state": "pending"}
            return Response(
                f"<RunInstancesResponse><instancesSet><item>"
                f"<instanceId>{iid}</instanceId></item></instancesSet>"
                f"</RunInstancesResponse>", media_type="text/xml")
        if action == "DescribeInstances":
            iid = params.get("InstanceId.1")
            inst = state["instances"].get(iid)
            if inst:
                inst["state"] = "running"  # second poll = running
            return Response(
                f"<DescribeInstancesResponse><reservationSet><item>"
                f"<instancesSet><item><instanceId>{iid}</instanceId>"
                f"<instanceState><name>{inst['state'] if inst else 'x'}"
                f"</name></instanceState>"
                f"<privateDnsName>ip-10-0-0-7.ec2.internal"
                f"</privateDnsName></item></instancesSet></item>"
                f"</reservationSet></DescribeInstancesResponse>",
                media_type="text/xml")
        if action == "TerminateInstances":
            state["instances"].pop(params.get("InstanceId.1"), None)
            return Response("<TerminateInstancesResponse/>",
                            media_type="text/xml")
        return Response("<Error/>", status_code=400)

    base, srv = _serve(app)
    yield base, state
    srv.should_exit = True


class TestAws:
    def test_lifecycle_and_signature(self, fake_ec2):
        base, state = fake_ec2
        cfg = AwsConfig(region="us-east-1", access_key="AKIATEST",
                        secret_key="sekrit", endpoint=base)
        p = AwsProvider(cfg)
        iid = p.create_node(_Claim())
        assert iid.startswith("i-")
        # the Authorization header is a real SigV4 credential scope
        auth = state["last_auth"]
        assert auth.startswith("AWS4-HMAC-SHA256 Credential=AKIATEST/")
        assert "us-east-1/ec2/aws4_request" in auth
        assert "Signature=" in auth
        # managed-by tag went on the wire
        assert "tensor-fusion" in state["bodies"][0]
        node = p.node_status(iid)
        assert node == "ip-10-0-0-7.ec2.internal"
        p.terminate_node(iid)
        assert iid not in state["instances"]

    def test_sigv4_is_deterministic(self):
        import datetime
        now = datetime.datetime(2026, 9, 14, 12, 0, 0)
        h1 = sigv4_headers("POST", "https://ec2.us-east-1.amazonaws.com",
                           "Action=DescribeInstances", "us-east-1",
                           "AK", "SK", now=now)
        h2 = sigv4_headers("POST", "https://ec2.us-east-1.amazonaws.com",
                           "Action=DescribeInstances", "us-east-1",
                           "AK", "SK", now=now)
        assert h1 == h2
        assert h1["X-Amz-Date"] == "20260914T120000Z"


# -------------------------------------------------------------- Alibaba


@pytest.fixture()
def fake_ecs():
    app = FastAPI()
    state = {"instances": {}, "sig_ok": []}
    SECRET = "alisecret"

    @app.get("/")
    async def handler(request: Request):
        params = dict(request.query_params)
        sig = params.pop("Signature", "")
        state["sig_ok"].append(
            sig == rpc_signature("GET", params, SECRET))
        action = params.get("Action")
        if action == "CreateInstance":
            iid = f"i-ali{len(state['instances'])}"
            state["instances"][iid] = "Stopped"
            return {"InstanceId": iid, "RequestId": "r"}
        if action == "StartInstance":
            state["instances"][params["InstanceId"]] = "Running"
            return {"RequestId": "r"}
        if action == "DescribeInstanceStatus":
            iid = params.get("InstanceId.1")
            return {"InstanceStatuses": {"InstanceStatus": [
                {"InstanceId": iid,
                 "Status": state["instances"].get(iid, "Missing")}]}}
        if action == "DeleteInstance":
            state["instances"].pop(params.get("InstanceId"), None)
            return {"RequestId": "r"}
        return {"Code": "Unknown"}

    base, srv = _serve(app)
    yield base, state, SECRET
    srv.should_exit = True


class TestAlibaba:
    def test_lifecycle_and_signature(self, fake_ecs):
        base, state, secret = fake_ecs
        cfg = AlibabaConfig(access_key_id="ali-ak",
                            access_key_secret=secret, endpoint=base)
        p = AlibabaProvider(cfg)
        iid = p.create_node(_Claim())
        assert iid.startswith("i-ali")
        assert p.node_status(iid) == f"node-{iid}"
        p.terminate_node(iid)
        assert p.node_status(iid) is None
        # every request carried a signature the server could recompute
        assert state["sig_ok"] and all(state["sig_ok"])

    def test_pricing_table(self):
        p = AlibabaProvider(AlibabaConfig())
        assert p.price_per_hour("ecs.ebmgn8a.64xlarge") == 31.9


# ------------------------------------------------------------ Karpenter


class TestKarpenter:
    def test_nodeclaim_choreography(self):
        from tensor_fusion_amd.k8s.client import K8sClient
        from tensor_fusion_amd.k8s.fake_apiserver import serve_in_thread
        srv, base, us = serve_in_thread()
        try:
            cli = K8sClient(base)
            p = KarpenterProvider(cli)
            name = p.create_node(_Claim())
            nc = cli.get("NodeClaim", name)
            reqs = {r["key"]: r["values"]
                    for r in nc["spec"]["requirements"]}
            assert reqs["node.kubernetes.io/instance-type"] == \
                ["mi355x.8g"]
            assert nc["spec"]["resources"]["requests"]["amd.com/gpu"] == "8"
            # not ready until Karpenter's controller reports Registered
            assert p.node_status(name) is None
            cli.patch("NodeClaim", name, {"status": {
                "nodeName": "ip-10-9-8-7",
                "conditions": [{"type": "Registered", "status": "True"}],
            }}, subresource="status")
            assert p.node_status(name) == "ip-10-9-8-7"
            p.terminate_node(name)
            assert cli.try_get("NodeClaim", name) is None
        finally:
            us.should_exit = True
