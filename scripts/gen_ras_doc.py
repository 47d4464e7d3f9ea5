#!/usr/bin/env python3
"""Generate docs/RAS_CATALOG.md from the catalog (kept in sync by
tests/test_kmsg_ras.py::test_ras_doc_in_sync)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from gpud_amd.pkg import ras_catalog  # noqa: E402


def render() -> str:
    lines = [
        "# amdgpu RAS kernel-message catalog",
        "",
        "The AMD-native equivalent of gpud's NVRM Xid table"
        " (auto-generated from `gpud_amd/pkg/ras_catalog.py` by"
        " `scripts/gen_ras_doc.py` — do not edit by hand).",
        "",
        "`critical` entries drive the error-ras health state machine;"
        " entries marked injectable can be replayed through"
        " `inject-fault --ras-event <name>` on healthy hardware.",
        "",
        "| Event | Severity | Critical | Injectable | Suggested actions | Description |",
        "|---|---|---|---|---|---|",
    ]
    for d in ras_catalog.CATALOG:
        lines.append(
            "| `{}` | {} | {} | {} | {} | {} |".format(
                d.name,
                d.event_type,
                "yes" if d.critical else "",
                "yes" if d.name in ras_catalog.INJECTABLE else "",
                ", ".join(d.repair_actions) or "—",
                d.description.replace("\n", " ").replace("|", "\\|"),
            )
        )
    lines.append("")
    return "\n".join(lines)


if __name__ == "__main__":
    out = os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "docs",
        "RAS_CATALOG.md",
    )
    with open(out, "w") as f:
        f.write(render())
    print(f"wrote {out}")
