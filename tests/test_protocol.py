"""Wire-protocol tests: v1beta1 messages + device-list codec.

Oracle: the programmatically-built protobuf classes (which the reference's
generated Go code and the kubelet both round-trip through)."""

import pytest

from gpushare_amd import consts
from gpushare_amd.device import fakedev
from gpushare_amd.deviceplugin import v1beta1 as api


def test_device_roundtrip():
    d = api.Device(ID="amd-x-_-0", health=consts.HEALTHY)
    d2 = api.Device.FromString(d.SerializeToString())
    assert d2.ID == "amd-x-_-0" and d2.health == "Healthy"


def test_register_request_fields():
    r = api.RegisterRequest(
        version="v1beta1",
        endpoint="amdgpushare.sock",
        resource_name=consts.RESOURCE_NAME,
    )
    r2 = api.RegisterRequest.FromString(r.SerializeToString())
    assert r2.resource_name == "aliyun.com/gpu-mem"
    assert r2.version == "v1beta1"


def test_allocate_request_grouping():
    req = api.AllocateRequest()
    cr = req.container_requests.add()
    cr.devicesIDs.extend(["a-_-0", "a-_-1"])
    cr2 = req.container_requests.add()
    cr2.devicesIDs.append("a-_-2")
    parsed = api.AllocateRequest.FromString(req.SerializeToString())
    assert [list(c.devicesIDs) for c in parsed.container_requests] == [
        ["a-_-0", "a-_-1"],
        ["a-_-2"],
    ]


def test_container_allocate_response_maps_and_devices():
    c = api.ContainerAllocateResponse(
        envs={"HIP_VISIBLE_DEVICES": "0"},
        devices=[
            api.DeviceSpec(
                container_path="/dev/kfd", host_path="/dev/kfd", permissions="rw"
            )
        ],
        annotations={"k": "v"},
    )
    c2 = api.ContainerAllocateResponse.FromString(c.SerializeToString())
    assert c2.envs["HIP_VISIBLE_DEVICES"] == "0"
    assert c2.devices[0].host_path == "/dev/kfd"
    assert c2.annotations["k"] == "v"


# --------------------------------------------------------------------------- #
# codec: native vs python vs protobuf byte-identical
# --------------------------------------------------------------------------- #

IDS = [f"amd-{i:016x}-_-{j}" for i in range(4) for j in range(17)]


def _protobuf_encode(ids, unhealthy):
    return api.ListAndWatchResponse(
        devices=[
            api.Device(
                ID=d,
                health=consts.UNHEALTHY if k in unhealthy else consts.HEALTHY,
            )
            for k, d in enumerate(ids)
        ]
    ).SerializeToString()


@pytest.mark.parametrize("unhealthy", [set(), {0}, {3, 17, 67}, set(range(68))])
def test_python_encoder_matches_protobuf(unhealthy):
    assert fakedev.encode_list_python(IDS, unhealthy) == _protobuf_encode(
        IDS, unhealthy
    )


@pytest.mark.parametrize("unhealthy", [[], [5], [1, 50]])
def test_native_codec_matches_protobuf(unhealthy):
    _devlist = pytest.importorskip("gpushare_amd._devlist")
    codec = _devlist.DeviceListCodec(IDS)
    assert codec.encode(unhealthy) == _protobuf_encode(IDS, set(unhealthy))


def test_native_codec_rejects_long_ids():
    _devlist = pytest.importorskip("gpushare_amd._devlist")
    with pytest.raises(ValueError):
        _devlist.DeviceListCodec(["x" * 64])


def test_codec_scale_2304_devices():
    """8×MI355X node: 2,304 fake devices must encode fast and correctly."""
    ids = [f"amd-{i:016x}-_-{j}" for i in range(8) for j in range(288)]
    codec = fakedev.make_codec(ids)
    payload = codec.encode([])
    assert payload == _protobuf_encode(ids, set())
    parsed = api.ListAndWatchResponse.FromString(payload)
    assert len(parsed.devices) == 2304


class TestFastHTTPRobustness:
    def test_garbage_request_drops_connection_not_server(self):
        """A malformed request line closes that connection; the server
        keeps serving other clients."""
        import socket

        from gpushare_amd.cluster.fasthttp import FastHTTPServer

        srv = FastHTTPServer(lambda m, p, b: (200, b'{"ok":true}')).start()
        try:
            bad = socket.create_connection(("127.0.0.1", srv.port))
            bad.sendall(b"NOT-HTTP\r\n\r\n")
            bad.settimeout(2)
            assert bad.recv(128) == b""  # dropped
            bad.close()

            from gpushare_amd.cluster.httpconn import HttpSession

            s = HttpSession(f"http://127.0.0.1:{srv.port}")
            status, body = s.request("GET", "/x")
            assert status == 200 and body == b'{"ok":true}'
            s.close()
        finally:
            srv.stop()

    def test_handler_exception_returns_500_keeps_connection(self):
        from gpushare_amd.cluster.fasthttp import FastHTTPServer
        from gpushare_amd.cluster.httpconn import HttpSession

        calls = []

        def handler(m, p, b):
            calls.append(p)
            if p == "/boom":
                raise RuntimeError("kaboom")
            return 200, b"{}"

        srv = FastHTTPServer(handler).start()
        try:
            s = HttpSession(f"http://127.0.0.1:{srv.port}")
            status, body = s.request("GET", "/boom")
            assert status == 500
            # same keep-alive connection still serves
            status, _ = s.request("GET", "/ok")
            assert status == 200
            s.close()
        finally:
            srv.stop()

    def test_extender_bad_json_returns_500(self):
        from gpushare_amd.cluster.httpconn import HttpSession
        from gpushare_amd.cluster.kubeclient import FakeKubeClient
        from gpushare_amd.extender.core import GPUShareExtender
        from gpushare_amd.extender.server import ExtenderServer

        ext = GPUShareExtender(FakeKubeClient("n"), resync_interval=3600)
        ext.register_node("n", [16])
        srv = ExtenderServer(ext).start()
        try:
            s = HttpSession(srv.url)
            status, _ = s.request(
                "POST", "/gpushare-scheduler/filter", body=b"{not json",
                headers={"Content-Type": "application/json"},
            )
            assert status == 500
            # extender still healthy
            status, _ = s.request("GET", "/gpushare-scheduler/packing")
            assert status == 200
            s.close()
        finally:
            srv.stop()


class TestTLS:
    def test_https_session_and_kubelet_client(self, tmp_path):
        """Production kubelet mode: HTTPS with a self-signed serving cert
        and verification off (reference forces insecure, client.go:75-99).
        Covers make_ssl_context + the HTTPS branch of HttpSession and
        KubeletClient's default scheme."""
        import json
        import ssl
        import subprocess

        from gpushare_amd.cluster.fasthttp import FastHTTPServer
        from gpushare_amd.cluster.httpconn import HttpSession
        from gpushare_amd.cluster.kubeclient import KubeletClient

        cert = tmp_path / "tls.crt"
        key = tmp_path / "tls.key"
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
             "-days", "1", "-subj", "/CN=127.0.0.1",
             "-addext", "subjectAltName=IP:127.0.0.1",
             "-keyout", str(key), "-out", str(cert)],
            check=True, capture_output=True,
        )
        ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
        ctx.load_cert_chain(str(cert), str(key))

        podlist = {"kind": "PodList", "items": [
            {"metadata": {"name": "p1", "namespace": "default"},
             "spec": {"nodeName": "n"}, "status": {"phase": "Running"}}
        ]}

        def handler(method, path, body):
            assert path.startswith("/pods")
            return 200, json.dumps(podlist).encode()

        srv = FastHTTPServer(handler, ssl_context=ctx).start()
        try:
            s = HttpSession(f"https://127.0.0.1:{srv.port}", verify=False)
            status, body = s.request("GET", "/pods/")
            assert status == 200 and json.loads(body) == podlist
            s.close()

            kc = KubeletClient(address="127.0.0.1", port=srv.port, token="t")
            assert kc.get_node_running_pods() == podlist
            kc.close()

            # verified TLS is the DEFAULT: the same self-signed
            # server must be REJECTED without an explicit verify=False,
            # and accepted when its cert is pinned as the CA
            s_default = HttpSession(f"https://127.0.0.1:{srv.port}")
            with pytest.raises(ssl.SSLError):
                s_default.request("GET", "/pods/")
            s_default.close()

            s_pinned = HttpSession(
                f"https://127.0.0.1:{srv.port}", verify=str(cert)
            )
            status, body = s_pinned.request("GET", "/pods/")
            assert status == 200
            s_pinned.close()
        finally:
            srv.stop()

    def test_kubeconfig_certificate_authority_data(self, tmp_path):
        """_auto_config must honour inline certificate-authority-data
        (base64 PEM) — previously unparsed, which silently downgraded
        such kubeconfigs to unverified TLS."""
        import base64
        import os
        import ssl
        import subprocess

        import yaml

        from gpushare_amd.cluster.kubeclient import RestKubeClient

        cert = tmp_path / "ca.crt"
        key = tmp_path / "ca.key"
        subprocess.run(
            ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
             "-days", "1", "-subj", "/CN=127.0.0.1",
             "-addext", "subjectAltName=IP:127.0.0.1",
             "-keyout", str(key), "-out", str(cert)],
            check=True, capture_output=True,
        )
        kubeconfig = tmp_path / "kubeconfig"
        kubeconfig.write_text(yaml.safe_dump({
            "current-context": "c",
            "contexts": [{"name": "c",
                          "context": {"cluster": "cl", "user": "u"}}],
            "clusters": [{"name": "cl", "cluster": {
                "server": "https://127.0.0.1:6443",
                "certificate-authority-data": base64.b64encode(
                    cert.read_bytes()
                ).decode(),
            }}],
            "users": [{"name": "u", "user": {"token": "tok"}}],
        }))
        old = os.environ.get("KUBECONFIG")
        os.environ["KUBECONFIG"] = str(kubeconfig)
        try:
            server, token, verify = RestKubeClient._auto_config()
            assert server == "https://127.0.0.1:6443"
            assert token == "tok"
            assert isinstance(verify, ssl.SSLContext)
            assert verify.verify_mode == ssl.CERT_REQUIRED
        finally:
            if old is None:
                os.environ.pop("KUBECONFIG", None)
            else:
                os.environ["KUBECONFIG"] = old


def test_deploy_manifests_parse_and_reference_contract():
    """Every deploy manifest parses as YAML and carries the wire-contract
    names (resource, socket dir, NODE_NAME downward API)."""
    import glob
    import os

    import yaml

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    files = sorted(glob.glob(os.path.join(repo, "deploy", "**", "*.yaml"),
                             recursive=True))
    assert len(files) >= 6
    docs = {}
    for f in files:
        docs[f] = [d for d in yaml.safe_load_all(open(f)) if d]
        assert docs[f], f

    blob = "\n".join(open(f).read() for f in files)
    assert consts.RESOURCE_NAME in blob
    assert "/var/lib/kubelet/device-plugins" in blob
    assert "NODE_NAME" in blob
    assert "prioritizeVerb" in blob       # scheduler policy wiring
    assert "/var/lib/gpushare" in blob    # memguard hostPath
