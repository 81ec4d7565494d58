#!/bin/sh
# OCI adapter pusher (reference parity: pkg/workspace/image/pusher.go +
# the docker-push sidecar in preset_tuning.go:283-368) — waits for the
# tuning job's completion marker, packs the output adapter directory into
# a single-layer OCI artifact and pushes it with skopeo.
set -eu
SRC="${1:-/mnt/results}"        # tuning output dir (adapter_*.safetensors)
IMG="$2"                        # destination ref, e.g. registry/adapters/x:v1
MARKER="$SRC/fine_tuning_completed.txt"

echo "waiting for tuning completion marker $MARKER"
while [ ! -f "$MARKER" ]; do sleep 10; done

TMP=$(mktemp -d)
trap 'rm -rf "$TMP"' EXIT
tar -cf "$TMP/layer.tar" -C "$SRC" .
mkdir -p "$TMP/oci/blobs/sha256"
DIGEST=$(sha256sum "$TMP/layer.tar" | cut -d' ' -f1)
SIZE=$(wc -c < "$TMP/layer.tar")
mv "$TMP/layer.tar" "$TMP/oci/blobs/sha256/$DIGEST"
CONFIG='{"architecture":"amd64","os":"linux","rootfs":{"type":"layers","diff_ids":["sha256:'$DIGEST'"]}}'
CDIGEST=$(printf '%s' "$CONFIG" | sha256sum | cut -d' ' -f1)
CSIZE=$(printf '%s' "$CONFIG" | wc -c)
printf '%s' "$CONFIG" > "$TMP/oci/blobs/sha256/$CDIGEST"
MANIFEST='{"schemaVersion":2,"mediaType":"application/vnd.oci.image.manifest.v1+json","config":{"mediaType":"application/vnd.oci.image.config.v1+json","digest":"sha256:'$CDIGEST'","size":'$CSIZE'},"layers":[{"mediaType":"application/vnd.oci.image.layer.v1.tar","digest":"sha256:'$DIGEST'","size":'$SIZE'}]}'
MDIGEST=$(printf '%s' "$MANIFEST" | sha256sum | cut -d' ' -f1)
MSIZE=$(printf '%s' "$MANIFEST" | wc -c)
printf '%s' "$MANIFEST" > "$TMP/oci/blobs/sha256/$MDIGEST"
printf '{"schemaVersion":2,"manifests":[{"mediaType":"application/vnd.oci.image.manifest.v1+json","digest":"sha256:%s","size":%s}]}' "$MDIGEST" "$MSIZE" > "$TMP/oci/index.json"
printf '{"imageLayoutVersion":"1.0.0"}' > "$TMP/oci/oci-layout"
echo "pushing adapter artifact to $IMG"
skopeo copy "oci:$TMP/oci:latest" "docker://$IMG"
echo "push complete"
