"""Generate the CRD manifests for every volcano API group into
deploy/crds/ (reference installer/helm/chart/volcano/crd/)."""
import os
import sys

import yaml

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from volcano_amd.store.k8s import GVK, crd_manifest  # noqa: E402

out_dir = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "deploy", "crds")
os.makedirs(out_dir, exist_ok=True)
for kind, (g, v, plural) in sorted(GVK.items()):
    man = crd_manifest(kind)
    if man is None:
        continue
    path = os.path.join(out_dir, f"{plural}.{g}.yaml")
    with open(path, "w") as f:
        yaml.safe_dump(man, f, sort_keys=False)
    print("wrote", path)
