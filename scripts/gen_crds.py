"""Render the CRD YAMLs into config/crd/bases (reference layout)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from substratus_amd.api.crd import render_all  # noqa: E402

out_dir = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "config", "crd", "bases")
os.makedirs(out_dir, exist_ok=True)
for name, text in render_all().items():
    with open(os.path.join(out_dir, name), "w") as f:
        f.write(text)
    print("wrote", name)
