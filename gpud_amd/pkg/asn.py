"""Public-IP ASN lookup for network-provider labeling.

Reference: pkg/asn/asn.go — primary HTTP lookup (hackertarget-style
``api/aslookup`` JSON) with retries and a fallback, feeding the
machine-info ``provider`` normalization (machine_info.go:339-389). Both
sources need egress; on an air-gapped node every path degrades to None
and machine-info simply omits the provider, exactly like the reference's
error path.

Injection points mirror the reference (``lookup_primary`` /
``lookup_fallback`` module attributes) so tests swap the network calls.
"""

from __future__ import annotations

import json
import time
from dataclasses import dataclass
from typing import Callable, Optional

import httpx

from .log import logger

MAX_RETRIES = 3
RETRY_DELAY_SECONDS = 3.0
PRIMARY_URL = "https://api.hackertarget.com/aslookup/?q={ip}&output=json"


@dataclass
class ASLookup:
    asn: str = ""
    asn_name: str = ""
    asn_range: str = ""
    country: str = ""
    ip: str = ""


def _fetch_primary(ip: str, timeout: float = 10.0) -> ASLookup:
    r = httpx.get(PRIMARY_URL.format(ip=ip), timeout=timeout)
    r.raise_for_status()
    d = json.loads(r.text)
    return ASLookup(
        asn=str(d.get("asn", "")),
        asn_name=d.get("asn_name", ""),
        asn_range=d.get("asn_range", ""),
        country=d.get("country", ""),
        ip=d.get("ip", ip),
    )


# injection points (reference: lookupPrimary/lookupFallback swap in tests)
lookup_primary: Callable[[str], ASLookup] = _fetch_primary
lookup_fallback: Optional[Callable[[str], ASLookup]] = None


def get_as_lookup(
    ip: str, sleep: Callable[[float], None] = time.sleep
) -> Optional[ASLookup]:
    """Retrying lookup with fallback; None when every path fails (the
    air-gapped default)."""
    last_err: Optional[Exception] = None
    for attempt in range(1, MAX_RETRIES + 1):
        try:
            resp = lookup_primary(ip)
            if resp.asn_name:
                return resp
            # empty primary: try the fallback before retrying
            if lookup_fallback is not None:
                try:
                    fb = lookup_fallback(ip)
                    if fb.asn_name:
                        return fb
                except Exception as e:  # noqa: BLE001
                    last_err = e
            if attempt == MAX_RETRIES:
                return resp  # reference returns the empty response at the end
        except Exception as e:  # noqa: BLE001 — no egress is the common case
            last_err = e
            if lookup_fallback is not None:
                try:
                    return lookup_fallback(ip)
                except Exception as fe:  # noqa: BLE001
                    last_err = fe
        if attempt < MAX_RETRIES:
            sleep(RETRY_DELAY_SECONDS)
    logger.debug("ASN lookup failed for %s: %s", ip, last_err)
    return None


# provider-name normalization (reference: asn.NormalizeASNName — lowercase
# the AS org and strip legal suffixes so "AMAZON-02, Inc." and "amazon"
# label the same provider)
_STRIP_TOKENS = (", inc.", ", inc", " inc.", " inc", ", llc", " llc",
                 ", ltd", " ltd", " corp.", " corp", " co.")


def normalize_asn_name(name: str) -> str:
    n = name.strip().lower()
    for tok in _STRIP_TOKENS:
        if n.endswith(tok):
            n = n[: -len(tok)]
    # common "PROVIDER-NN" org formats collapse to the provider word
    for sep in ("-", " "):
        head = n.split(sep, 1)[0]
        if head in ("amazon", "google", "microsoft", "oracle", "tencent",
                    "alibaba", "ovh", "hetzner", "lambda", "coreweave",
                    "crusoe", "vultr", "equinix"):
            return head
    return n
