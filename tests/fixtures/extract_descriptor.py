"""Regenerate k8s_deviceplugin_v1beta1.fdp.bin from a generated Go file.

Usage: python extract_descriptor.py /path/to/api.pb.go [out.bin]

Upstream protoc embeds the gzipped FileDescriptorProto in the generated Go
as `var fileDescriptorApi = []byte{...}`; this pulls it out and gunzips it.
"""

import gzip
import hashlib
import re
import sys


def extract(go_source: str) -> bytes:
    m = re.search(r"var fileDescriptorApi = \[\]byte\{(.*?)\n\}", go_source, re.S)
    if not m:
        raise SystemExit("no fileDescriptorApi byte literal found")
    data = bytes(int(t, 16) for t in re.findall(r"0x([0-9a-fA-F]{2})", m.group(1)))
    return gzip.decompress(data)


if __name__ == "__main__":
    src = open(sys.argv[1]).read()
    out = sys.argv[2] if len(sys.argv) > 2 else "k8s_deviceplugin_v1beta1.fdp.bin"
    raw = extract(src)
    with open(out, "wb") as fh:
        fh.write(raw)
    print(f"{len(raw)} bytes sha256={hashlib.sha256(raw).hexdigest()}")
