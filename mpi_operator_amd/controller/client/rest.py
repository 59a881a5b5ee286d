"""REST KubeClient — talks to a real apiserver via `requests`.

Supports in-cluster config (serviceaccount token) and kubeconfig files
(the reference uses client-go's BuildConfigFromFlags,
cmd/mpi-operator/app/server.go:102-114). Includes a watch helper for the
informer-style event loop in server.py."""
from __future__ import annotations

import json
import os
import ssl
import time
from typing import Iterator, Optional

from .base import (GVR, ApiError, Conflict, KubeClient, NotFound,
                   ResourceClient)

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class RestConfig:
    def __init__(self, host: str, token: str | None = None,
                 ca_file: str | None = None, cert: tuple | None = None,
                 verify=True):
        self.host = host.rstrip("/")
        self.token = token
        self.ca_file = ca_file
        self.cert = cert
        self.verify = ca_file if ca_file else verify

    @classmethod
    def in_cluster(cls) -> "RestConfig":
        host = os.environ["KUBERNETES_SERVICE_HOST"]
        port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
        with open(os.path.join(SA_DIR, "token")) as f:
            token = f.read().strip()
        return cls(f"https://{host}:{port}", token=token,
                   ca_file=os.path.join(SA_DIR, "ca.crt"))

    @classmethod
    def from_kubeconfig(cls, path: str | None = None, context: str | None = None) -> "RestConfig":
        import tempfile
        import base64

        import yaml

        path = path or os.environ.get("KUBECONFIG", os.path.expanduser("~/.kube/config"))
        with open(path) as f:
            cfg = yaml.safe_load(f)
        ctx_name = context or cfg.get("current-context")
        ctx = next(c["context"] for c in cfg["contexts"] if c["name"] == ctx_name)
        cluster = next(c["cluster"] for c in cfg["clusters"] if c["name"] == ctx["cluster"])
        user = next(u["user"] for u in cfg["users"] if u["name"] == ctx["user"])

        def materialize(data_key, file_key, obj):
            if file_key in obj:
                return obj[file_key]
            if data_key in obj:
                f = tempfile.NamedTemporaryFile(delete=False, suffix=".pem")
                f.write(base64.b64decode(obj[data_key]))
                f.close()
                return f.name
            return None

        ca = materialize("certificate-authority-data", "certificate-authority", cluster)
        cert = materialize("client-certificate-data", "client-certificate", user)
        key = materialize("client-key-data", "client-key", user)
        token = user.get("token")
        return cls(cluster["server"], token=token, ca_file=ca,
                   cert=(cert, key) if cert and key else None,
                   verify=ca if ca else not cluster.get("insecure-skip-tls-verify", False))

    @classmethod
    def auto(cls) -> "RestConfig":
        if os.path.exists(os.path.join(SA_DIR, "token")):
            return cls.in_cluster()
        return cls.from_kubeconfig()


def _path(gvr: GVR, namespace: Optional[str], name: Optional[str] = None,
          subresource: Optional[str] = None) -> str:
    base = f"/api/{gvr.version}" if not gvr.group else f"/apis/{gvr.group}/{gvr.version}"
    p = base
    if gvr.namespaced and namespace:
        p += f"/namespaces/{namespace}"
    p += f"/{gvr.resource}"
    if name:
        p += f"/{name}"
    if subresource:
        p += f"/{subresource}"
    return p


class RestResourceClient(ResourceClient):
    def __init__(self, parent: "RestKubeClient", gvr: GVR):
        self.p = parent
        self.gvr = gvr

    def _req(self, method, path, **kw):
        return self.p.request(method, path, **kw)

    def get(self, namespace, name):
        return self._req("GET", _path(self.gvr, namespace, name))

    def list(self, namespace, label_selector=None):
        params = {}
        if label_selector:
            params["labelSelector"] = ",".join(f"{k}={v}" for k, v in label_selector.items())
        out = self._req("GET", _path(self.gvr, namespace), params=params)
        return out.get("items", [])

    def list_with_rv(self, namespace, label_selector=None):
        """List returning (items, resourceVersion) — the resume point for a
        watch (client-go ListWatch semantics)."""
        params = {}
        if label_selector:
            params["labelSelector"] = ",".join(f"{k}={v}" for k, v in label_selector.items())
        out = self._req("GET", _path(self.gvr, namespace), params=params)
        return out.get("items", []), out.get("metadata", {}).get("resourceVersion")

    def create(self, namespace, obj):
        return self._req("POST", _path(self.gvr, namespace), json=obj)

    def update(self, namespace, obj):
        return self._req("PUT", _path(self.gvr, namespace, obj["metadata"]["name"]), json=obj)

    def update_status(self, namespace, obj):
        return self._req("PUT", _path(self.gvr, namespace, obj["metadata"]["name"], "status"),
                         json=obj)

    def delete(self, namespace, name):
        self._req("DELETE", _path(self.gvr, namespace, name))

    def watch(self, namespace, resource_version: str | None = None,
              timeout_s: int = 300) -> Iterator[dict]:
        """Yields watch events: {"type": "ADDED|MODIFIED|DELETED", "object": {...}}."""
        params = {"watch": "true", "timeoutSeconds": str(timeout_s)}
        if resource_version:
            params["resourceVersion"] = resource_version
        resp = self.p.raw_request("GET", _path(self.gvr, namespace), params=params,
                                  stream=True)
        for line in resp.iter_lines():
            if line:
                yield json.loads(line)


class RestKubeClient(KubeClient):
    def __init__(self, config: RestConfig | None = None, user_agent: str = "mpi-operator",
                 qps: float = 5.0, burst: int = 10):
        """qps/burst: client-side token-bucket throttle, the
        --kube-api-qps/--kube-api-burst contract (reference options.go:87-88,
        client-go flowcontrol defaults 5/10)."""
        import requests

        from ..ratelimit import TokenBucket

        self.config = config or RestConfig.auto()
        self.session = requests.Session()
        if self.config.token:
            self.session.headers["Authorization"] = f"Bearer {self.config.token}"
        self.session.headers["User-Agent"] = user_agent
        if self.config.cert:
            self.session.cert = self.config.cert
        self.session.verify = self.config.verify
        self._clients: dict[GVR, RestResourceClient] = {}
        self._throttle = TokenBucket(qps, burst)

    def raw_request(self, method, path, **kw):
        # watches hold a streaming connection open — don't charge the bucket
        if not kw.get("stream"):
            self._throttle.wait()
        resp = self.session.request(method, self.config.host + path, **kw)
        if resp.status_code == 404:
            raise NotFound(path)
        if resp.status_code == 409:
            raise Conflict(path)
        if resp.status_code >= 400:
            raise ApiError(resp.status_code, resp.text[:500])
        return resp

    def request(self, method, path, **kw):
        return self.raw_request(method, path, **kw).json()

    def resource(self, gvr: GVR) -> RestResourceClient:
        if gvr not in self._clients:
            self._clients[gvr] = RestResourceClient(self, gvr)
        return self._clients[gvr]

    def crd_exists(self, name: str) -> bool:
        try:
            self.request("GET", f"/apis/apiextensions.k8s.io/v1/customresourcedefinitions/{name}")
            return True
        except NotFound:
            return False
