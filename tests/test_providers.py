"""Cloud-provider IMDS detectors (reference: pkg/providers/*/imds —
injected-HTTP fakes, the reference's mock-imds test approach)."""

import json

from gpud_amd.pkg.providers import (
    AWSDetector,
    AzureDetector,
    GCPDetector,
    NebiusDetector,
    NScaleDetector,
    OCIDetector,
    detect,
)


def test_aws_imdsv2_token_flow():
    calls = []

    def put(url, headers=None, timeout=None):
        calls.append(("PUT", url, headers))
        assert headers["X-aws-ec2-metadata-token-ttl-seconds"] == "21600"
        return "TOKEN123\n"

    def get(url, headers=None, timeout=None):
        assert headers["X-aws-ec2-metadata-token"] == "TOKEN123"
        path = url.rsplit("meta-data/", 1)[1]
        return {
            "instance-id": "i-0abc",
            "placement/region": "us-east-1",
            "public-ipv4": "3.4.5.6",
            "local-ipv4": "10.0.0.7",
        }.get(path)

    info = AWSDetector(http_get=get, http_put=put).detect()
    assert info.provider == "aws"
    assert info.instance_id == "i-0abc"
    assert info.region == "us-east-1"
    assert info.public_ip == "3.4.5.6"
    assert info.private_ip == "10.0.0.7"


def test_aws_no_token_means_not_aws():
    info = AWSDetector(
        http_get=lambda *a, **k: None, http_put=lambda *a, **k: None
    ).detect()
    assert info is None


def test_gcp_zone_to_region():
    def get(url, headers=None, timeout=None):
        assert headers["Metadata-Flavor"] == "Google"
        path = url.split("computeMetadata/v1/", 1)[1]
        return {
            "instance/id": "123456",
            "instance/zone": "projects/99/zones/us-central1-a",
            "instance/network-interfaces/0/ip": "10.1.2.3",
            "instance/network-interfaces/0/access-configs/0/external-ip": "34.1.2.3",
        }.get(path)

    info = GCPDetector(http_get=get).detect()
    assert info.provider == "gcp"
    assert info.region == "us-central1"
    assert info.private_ip == "10.1.2.3"


def test_azure_json_document():
    doc = {
        "compute": {
            "vmId": "vm-42",
            "location": "westus2",
            "azEnvironment": "AZUREPUBLICCLOUD",
        },
        "network": {
            "interface": [
                {
                    "ipv4": {
                        "ipAddress": [
                            {
                                "privateIpAddress": "10.9.8.7",
                                "publicIpAddress": "20.1.2.3",
                            }
                        ]
                    }
                }
            ]
        },
    }

    def get(url, headers=None, timeout=None):
        assert headers["Metadata"] == "true"
        return json.dumps(doc)

    info = AzureDetector(http_get=get).detect()
    assert info.provider == "azure"
    assert info.instance_id == "vm-42"
    assert info.region == "westus2"
    assert info.vm_environment == "AZUREPUBLICCLOUD"
    assert info.private_ip == "10.9.8.7"
    assert info.public_ip == "20.1.2.3"


def test_oci_bearer_oracle():
    def get(url, headers=None, timeout=None):
        assert headers["Authorization"] == "Bearer Oracle"
        if url.endswith("/instance/"):
            return json.dumps(
                {"id": "ocid1.instance.x", "canonicalRegionName": "us-ashburn-1",
                 "shape": "BM.GPU.MI300X.8"}
            )
        if url.endswith("privateIp"):
            return '"10.5.5.5"'
        return None

    info = OCIDetector(http_get=get).detect()
    assert info.provider == "oci"
    assert info.instance_id == "ocid1.instance.x"
    assert info.region == "us-ashburn-1"
    assert info.private_ip == "10.5.5.5"


def test_nebius_and_nscale():
    def get_nebius(url, headers=None, timeout=None):
        if url.endswith("/instance/id"):
            return "computeinstance-abc"
        if url.endswith("/0/ip"):
            return "10.2.3.4"
        return None

    info = NebiusDetector(http_get=get_nebius).detect()
    assert info.provider == "nebius"
    assert info.instance_id == "computeinstance-abc"

    def get_nscale(url, headers=None, timeout=None):
        return json.dumps({"uuid": "os-uuid-1", "availability_zone": "az1"})

    info = NScaleDetector(http_get=get_nscale).detect()
    assert info.provider == "nscale"
    assert info.region == "az1"


def test_detect_first_success_wins_and_offline_none():
    class Never:
        def detect(self):
            return None

    class Boom:
        def detect(self):
            raise RuntimeError("imds exploded")

    class Found:
        def detect(self):
            from gpud_amd.pkg.providers import Info

            return Info(provider="aws", instance_id="i-1")

    assert detect([Never(), Boom(), Found()]).provider == "aws"
    assert detect([Never(), Boom()]) is None
    assert detect([]) is None


def test_machine_info_inventory_fields():
    """The deepened inventory: CPU vendor, NIC MAC + virtual-interface
    filter, mounted-disk mapping fields."""
    from gpud_amd.pkg import machine_info as mi

    assert mi._cpu_vendor()  # some vendor string on any linux host
    nic = mi._nic_info()
    for iface in nic.private_ip_interfaces:
        assert "mac" in iface and "ip" in iface
        assert not iface["interface"].startswith(("docker", "veth", "cali"))
    disk = mi._disk_info()
    for d in disk.block_devices:
        assert {"name", "type", "size", "used", "mount_point", "fstype",
                "serial", "model", "parent_device"} <= set(d)
