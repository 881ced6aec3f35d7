#!/usr/bin/env python3
"""Generate THIRD_PARTY_NOTICES.md from what actually ships.

Role parity with the reference's notices tooling (it rebuilds notices
from the *built image* and fails closed on unresolved licenses); this
implementation scans Python ``*.dist-info`` directories either from a
local site-packages tree (``--site-packages``, used in CI and tests) or
extracted out of a container image (``--image``, docker CLI required).

Fails (exit 1) when a shipped distribution has no resolvable license
text or when a License-File entry escapes its dist-info directory.
"""

from __future__ import annotations

import argparse
import email.parser
import subprocess
import sys
import tempfile
from pathlib import Path
from typing import Dict, List, Optional

HEADER = """# Third-Party Notices

This file lists the third-party Python distributions shipped in the
k8s-cc-manager-amd container image and their licenses. It is generated
by `hack/gen_third_party_notices.py`; regenerate with
`make -f deployments/container/Makefile third-party-notices`.
"""


class NoticesError(Exception):
    pass


def parse_metadata(dist_info: Path) -> Optional[Dict]:
    meta_path = dist_info / "METADATA"
    if not meta_path.exists():
        return None
    msg = email.parser.Parser().parsestr(
        meta_path.read_text(errors="replace").split("\n\n", 1)[0]
    )
    name = msg.get("Name") or dist_info.name.split("-")[0]
    version = msg.get("Version") or "unknown"
    license_expr = msg.get("License-Expression") or ""
    license_field = msg.get("License") or ""
    classifiers = msg.get_all("Classifier") or []
    cls_licenses = [
        c.split("::")[-1].strip()
        for c in classifiers
        if c.startswith("License ::") and "OSI Approved" in c
    ]
    license_name = (
        license_expr
        or (license_field if license_field and license_field != "UNKNOWN" and len(license_field) < 120 else "")
        or (cls_licenses[0] if cls_licenses else "")
    )
    # collect license texts
    texts: List[str] = []
    file_names = msg.get_all("License-File") or []
    candidates = set(file_names)
    for pattern in ("LICENSE*", "COPYING*", "licenses/*"):
        for p in dist_info.glob(pattern):
            candidates.add(str(p.relative_to(dist_info)))
    for rel in sorted(candidates):
        p = (dist_info / rel).resolve()
        if not str(p).startswith(str(dist_info.resolve())):
            raise NoticesError(
                f"{name}: License-File entry escapes dist-info: {rel}"
            )
        if p.is_file():
            texts.append(p.read_text(errors="replace"))
    if not license_name and license_field and len(license_field) > 120:
        # whole license text inline in the License field
        texts.append(license_field)
        license_name = license_field.splitlines()[0][:60]
    return {
        "name": name,
        "version": version,
        "license": license_name or "UNRESOLVED",
        "texts": texts,
    }


def collect(site_packages: Path) -> List[Dict]:
    dists = []
    for dist_info in sorted(site_packages.glob("*.dist-info")):
        meta = parse_metadata(dist_info)
        if meta:
            dists.append(meta)
    return dists


def render(dists: List[Dict]) -> str:
    out = [HEADER]
    out.append("\n## Summary\n")
    out.append("| Distribution | Version | License |")
    out.append("|---|---|---|")
    for d in dists:
        out.append(f"| {d['name']} | {d['version']} | {d['license']} |")
    out.append("\n## License texts\n")
    for d in dists:
        out.append(f"### {d['name']} {d['version']}\n")
        if d["texts"]:
            for text in d["texts"]:
                out.append("```text")
                out.append(text.rstrip())
                out.append("```")
        else:
            out.append(f"License: {d['license']} (no bundled text)")
        out.append("")
    return "\n".join(out) + "\n"


def site_packages_from_image(image: str) -> Path:
    """Export an image's /app/site-packages via docker."""
    tmp = Path(tempfile.mkdtemp(prefix="notices-"))
    cid = subprocess.run(
        ["docker", "create", image], capture_output=True, text=True, check=True
    ).stdout.strip()
    try:
        subprocess.run(
            ["docker", "cp", f"{cid}:/app/site-packages", str(tmp / "sp")],
            check=True,
            capture_output=True,
        )
    finally:
        subprocess.run(["docker", "rm", cid], capture_output=True)
    return tmp / "sp"


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    group = ap.add_mutually_exclusive_group(required=True)
    group.add_argument("--site-packages", help="local site-packages tree to scan")
    group.add_argument("--image", help="container image to extract and scan")
    ap.add_argument("--output", default="THIRD_PARTY_NOTICES.md")
    ap.add_argument(
        "--allow-unresolved",
        action="store_true",
        help="do not fail on distributions without license text/name",
    )
    args = ap.parse_args(argv)

    sp = (
        Path(args.site_packages)
        if args.site_packages
        else site_packages_from_image(args.image)
    )
    try:
        dists = collect(sp)
    except NoticesError as e:
        print(f"error: {e}", file=sys.stderr)
        return 1

    unresolved = [d["name"] for d in dists if d["license"] == "UNRESOLVED" and not d["texts"]]
    if unresolved and not args.allow_unresolved:
        print(f"error: unresolved licenses: {unresolved}", file=sys.stderr)
        return 1

    Path(args.output).write_text(render(dists))
    print(f"wrote {args.output}: {len(dists)} distributions")
    return 0


if __name__ == "__main__":
    sys.exit(main())
