"""Cloud-provider detection via instance metadata services.

Reference: pkg/providers (Detector interface: Provider/PublicIPv4/
PrivateIPv4/VMEnvironment/InstanceID + RegionDetector — providers.go:11-37)
with per-cloud IMDS packages for AWS (IMDSv2 token flow), GCP
(Metadata-Flavor header), Azure (JSON instance endpoint), OCI (Bearer
Oracle), Nebius (metadata.nebius.internal) and NScale (OpenStack
meta_data.json). Air-gapped nodes simply detect nothing — every probe has
a short timeout and failures are silent.

All HTTP goes through injectable ``http_get``/``http_put`` callables so the
detectors are fully testable offline (the reference mocks its imds
packages the same way).
"""

from __future__ import annotations

import json
from dataclasses import asdict, dataclass
from typing import Callable, List, Optional

DEFAULT_TIMEOUT = 2.0


@dataclass
class Info:
    """reference: providers.Info (providers.go:40-48)."""

    provider: str = ""
    public_ip: str = ""
    private_ip: str = ""
    region: str = ""
    vm_environment: str = ""
    instance_id: str = ""

    def to_dict(self) -> dict:
        return asdict(self)


def _httpx_get(url: str, headers: Optional[dict] = None,
               timeout: float = DEFAULT_TIMEOUT) -> Optional[str]:
    import httpx

    try:
        r = httpx.get(url, headers=headers or {}, timeout=timeout)
        if r.status_code == 200:
            return r.text
    except Exception:
        pass
    return None


def _httpx_put(url: str, headers: Optional[dict] = None,
               timeout: float = DEFAULT_TIMEOUT) -> Optional[str]:
    import httpx

    try:
        r = httpx.put(url, headers=headers or {}, timeout=timeout)
        if r.status_code == 200:
            return r.text
    except Exception:
        pass
    return None


class AWSDetector:
    """IMDSv2: PUT a session token, then GET metadata paths with it
    (reference: pkg/providers/aws/imds/imds.go)."""

    name = "aws"
    token_url = "http://169.254.169.254/latest/api/token"
    meta_url = "http://169.254.169.254/latest/meta-data"

    def __init__(self, http_get: Callable = _httpx_get,
                 http_put: Callable = _httpx_put):
        self._get = http_get
        self._put = http_put

    def detect(self) -> Optional[Info]:
        token = self._put(
            self.token_url,
            headers={"X-aws-ec2-metadata-token-ttl-seconds": "21600"},
        )
        if not token:
            return None
        h = {"X-aws-ec2-metadata-token": token.strip()}

        def meta(path: str) -> str:
            return (self._get(f"{self.meta_url}/{path}", headers=h) or "").strip()

        instance_id = meta("instance-id")
        if not instance_id:
            return None
        return Info(
            provider="aws",
            instance_id=instance_id,
            region=meta("placement/region"),
            public_ip=meta("public-ipv4"),
            private_ip=meta("local-ipv4"),
            vm_environment="EC2",
        )


class GCPDetector:
    """Metadata-Flavor: Google header (reference: pkg/providers/gcp/imds)."""

    name = "gcp"
    meta_url = "http://metadata.google.internal/computeMetadata/v1"

    def __init__(self, http_get: Callable = _httpx_get):
        self._get = http_get

    def detect(self) -> Optional[Info]:
        h = {"Metadata-Flavor": "Google"}

        def meta(path: str) -> str:
            return (self._get(f"{self.meta_url}/{path}", headers=h) or "").strip()

        instance_id = meta("instance/id")
        if not instance_id:
            return None
        zone = meta("instance/zone")  # projects/<n>/zones/<zone>
        zone = zone.rsplit("/", 1)[-1]
        region = zone.rsplit("-", 1)[0] if "-" in zone else zone
        return Info(
            provider="gcp",
            instance_id=instance_id,
            region=region,
            private_ip=meta("instance/network-interfaces/0/ip"),
            public_ip=meta(
                "instance/network-interfaces/0/access-configs/0/external-ip"
            ),
            vm_environment="GCE",
        )


class AzureDetector:
    """JSON instance endpoint with Metadata: true header
    (reference: pkg/providers/azure/imds)."""

    name = "azure"
    meta_url = (
        "http://169.254.169.254/metadata/instance?api-version=2021-02-01"
    )

    def __init__(self, http_get: Callable = _httpx_get):
        self._get = http_get

    def detect(self) -> Optional[Info]:
        body = self._get(self.meta_url, headers={"Metadata": "true"})
        if not body:
            return None
        try:
            doc = json.loads(body)
        except json.JSONDecodeError:
            return None
        compute = doc.get("compute") or {}
        if not compute.get("vmId"):
            return None
        public_ip = private_ip = ""
        for iface in (doc.get("network") or {}).get("interface", []):
            for addr in (iface.get("ipv4") or {}).get("ipAddress", []):
                private_ip = private_ip or addr.get("privateIpAddress", "")
                public_ip = public_ip or addr.get("publicIpAddress", "")
        return Info(
            provider="azure",
            instance_id=compute.get("vmId", ""),
            region=compute.get("location", ""),
            vm_environment=compute.get("azEnvironment", ""),
            public_ip=public_ip,
            private_ip=private_ip,
        )


class OCIDetector:
    """Authorization: Bearer Oracle (reference: pkg/providers/oci/imds)."""

    name = "oci"
    meta_url = "http://169.254.169.254/opc/v2"

    def __init__(self, http_get: Callable = _httpx_get):
        self._get = http_get

    def detect(self) -> Optional[Info]:
        h = {"Authorization": "Bearer Oracle"}
        body = self._get(f"{self.meta_url}/instance/", headers=h)
        if not body:
            return None
        try:
            doc = json.loads(body)
        except json.JSONDecodeError:
            return None
        if not doc.get("id"):
            return None
        private_ip = (self._get(
            f"{self.meta_url}/vnics/0/privateIp", headers=h) or "").strip('" \n')
        return Info(
            provider="oci",
            instance_id=doc.get("id", ""),
            region=doc.get("canonicalRegionName", doc.get("region", "")),
            vm_environment=doc.get("shape", ""),
            private_ip=private_ip,
        )


class NebiusDetector:
    """metadata.nebius.internal/v1 (reference: pkg/providers/nebius/imds)."""

    name = "nebius"
    meta_url = "http://metadata.nebius.internal/v1"

    def __init__(self, http_get: Callable = _httpx_get):
        self._get = http_get

    def detect(self) -> Optional[Info]:
        instance_id = (self._get(f"{self.meta_url}/instance/id") or "").strip()
        if not instance_id:
            return None
        return Info(
            provider="nebius",
            instance_id=instance_id,
            private_ip=(self._get(
                f"{self.meta_url}/instance/network-interfaces/0/ip") or "").strip(),
            vm_environment="nebius",
        )


class NScaleDetector:
    """OpenStack meta_data.json on the EC2-compatible endpoint
    (reference: pkg/providers/nscale/doc.go)."""

    name = "nscale"
    meta_url = "http://169.254.169.254/openstack/latest/meta_data.json"

    def __init__(self, http_get: Callable = _httpx_get):
        self._get = http_get

    def detect(self) -> Optional[Info]:
        body = self._get(self.meta_url)
        if not body:
            return None
        try:
            doc = json.loads(body)
        except json.JSONDecodeError:
            return None
        if not doc.get("uuid"):
            return None
        return Info(
            provider="nscale",
            instance_id=doc.get("uuid", ""),
            region=doc.get("availability_zone", ""),
            vm_environment="openstack",
        )


def default_detectors(http_get: Callable = _httpx_get,
                      http_put: Callable = _httpx_put) -> List:
    return [
        AWSDetector(http_get, http_put),
        GCPDetector(http_get),
        AzureDetector(http_get),
        OCIDetector(http_get),
        NebiusDetector(http_get),
        NScaleDetector(http_get),
    ]


def detect(detectors: Optional[List] = None) -> Optional[Info]:
    """First detector that answers wins (reference: pkg/providers/detect.go
    probes each provider and returns the first success)."""
    for d in detectors if detectors is not None else default_detectors():
        try:
            info = d.detect()
        except Exception:
            info = None
        if info is not None:
            return info
    return None
