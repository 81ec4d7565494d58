"""RAGEngine v1alpha1 ↔ v1beta1 conversion (reference:
api/v1alpha1/ragengine_conversion.go — v1beta1 is the storage/hub
version; the only structural change is Storage: v1alpha1 keeps
persistentVolumeClaim/mountPath FLAT, v1beta1 nests them under
`persistentVolume`)."""
from __future__ import annotations

import copy
from typing import Dict


def ragengine_to_v1beta1(obj: Dict) -> Dict:
    """v1alpha1 RAGEngine object dict → v1beta1 (hub)."""
    out = copy.deepcopy(obj)
    out["apiVersion"] = "kaito.sh/v1beta1"
    spec = out.get("spec") or {}
    st = spec.get("storage")
    if st and ("persistentVolumeClaim" in st or "mountPath" in st):
        pv = {}
        for k in ("persistentVolumeClaim", "mountPath"):
            if k in st:
                pv[k] = st.pop(k)
        if pv:
            st["persistentVolume"] = pv
    return out


def ragengine_to_v1alpha1(obj: Dict) -> Dict:
    """v1beta1 RAGEngine object dict → v1alpha1 (spoke)."""
    out = copy.deepcopy(obj)
    out["apiVersion"] = "kaito.sh/v1alpha1"
    spec = out.get("spec") or {}
    st = spec.get("storage")
    if st and "persistentVolume" in st:
        pv = st.pop("persistentVolume") or {}
        for k in ("persistentVolumeClaim", "mountPath"):
            if k in pv:
                st[k] = pv[k]
    return out


def convert_ragengine(obj: Dict, target_version: str) -> Dict:
    if target_version in ("v1beta1", "kaito.sh/v1beta1"):
        return ragengine_to_v1beta1(obj)
    if target_version in ("v1alpha1", "kaito.sh/v1alpha1"):
        return ragengine_to_v1alpha1(obj)
    raise ValueError(f"unknown RAGEngine version {target_version!r}")
