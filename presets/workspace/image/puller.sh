#!/bin/sh
# OCI model-weight puller (reference parity: pkg/workspace/image/puller.sh
# pull/expand/relocate) — skopeo copy of a weights artifact into the
# mounted volume.
set -eu
IMG="$1"          # oci artifact ref, e.g. registry/models/llama-3-8b:v1
DEST="${2:-/workspace/weights}"
TMP="$DEST/.pull.tmp"
mkdir -p "$TMP"
echo "0% pulling $IMG"
skopeo copy "docker://$IMG" "oci:$TMP:latest"
echo "70% expanding layers"
for blob in "$TMP"/blobs/sha256/*; do
  if tar -tf "$blob" >/dev/null 2>&1; then
    tar -xf "$blob" -C "$DEST"
  fi
done
echo "95% relocating"
rm -rf "$TMP"
echo "100% done"
