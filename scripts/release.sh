#!/usr/bin/env bash
# Release bundle (the reference ships via goreleaser: binaries +
# container-tools tarballs + checksums, .goreleaser.yaml:22-46).
# Produces dist/: sdist + wheel of the package, a container-tools tarball
# (nbwatch, the analog of container-tools-{os}-{arch}.tar.gz), and
# sha256 checksums.
set -euo pipefail
cd "$(dirname "$0")/.."
rm -rf dist && mkdir -p dist
python setup.py --quiet sdist --dist-dir dist 2>/dev/null || true
pip wheel . --no-deps --no-build-isolation -w dist 2>/dev/null || \
  echo "(wheel build skipped — offline toolchain)"
tar czf "dist/container-tools-linux-amd64.tar.gz" \
  -C substratus_amd/containertools nbwatch.py
( cd dist && sha256sum * > checksums.txt )
echo "release artifacts:"; ls -l dist
