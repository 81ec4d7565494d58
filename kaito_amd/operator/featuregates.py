"""Feature gates — parity with the reference's pkg/featuregates
(featuregates.go:28-63; defaults from pkg/utils/consts/consts.go:59-69)."""
from __future__ import annotations

from typing import Dict

DEFAULT_GATES: Dict[str, bool] = {
    "vLLM": True,                              # engine on (ours: kaito-amd)
    "disableNodeAutoProvisioning": False,
    "gatewayAPIInferenceExtension": False,
    "enableInferenceSetController": True,
    "enableMIG": False,                        # MI355X: SR-IOV partitioning
    "enableAccelerator": False,
    "enableMultiRoleInferenceController": False,
    "ModelMirror": False,
    "ModelStreaming": False,
    "enableBaseImageAutoUpgrade": False,
}


class FeatureGateError(ValueError):
    pass


def parse_feature_gates(spec: str) -> Dict[str, bool]:
    """Parse "a=true,b=false" with validation against known gates."""
    gates = dict(DEFAULT_GATES)
    if not spec:
        return gates
    for part in spec.split(","):
        part = part.strip()
        if not part:
            continue
        if "=" not in part:
            raise FeatureGateError(f"malformed feature gate {part!r}")
        k, v = part.split("=", 1)
        k = k.strip()
        if k not in gates:
            raise FeatureGateError(
                f"unknown feature gate {k!r}; known: {sorted(gates)}")
        vl = v.strip().lower()
        if vl not in ("true", "false"):
            raise FeatureGateError(f"feature gate {k!r} must be true/false")
        gates[k] = vl == "true"
    return gates
